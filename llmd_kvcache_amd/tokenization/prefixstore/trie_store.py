"""Character-trie alternative prefix store.

Parity with reference pkg/tokenization/prefixstore/trie_store.go: each trie
node stores the id/index of the last token fully contained at that character
position (:29-35); insert scans tokens while walking characters (:85-136);
lookup walks characters collecting newly-contained tokens (:142-174).  Not
the default backend (slower than the LRU block store).
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional, Sequence, Tuple


class _TrieNode:
    __slots__ = ("children", "token_id", "token_index")

    def __init__(self) -> None:
        self.children: Dict[str, "_TrieNode"] = {}
        self.token_id: Optional[int] = None
        self.token_index: int = -1


class TrieTokenStore:
    def __init__(self) -> None:
        self.root = _TrieNode()
        self._mu = threading.Lock()

    def add_tokenization(
        self,
        prompt: str,
        tokens: Sequence[int],
        offsets: Sequence[Tuple[int, int]],
    ) -> None:
        if not prompt or not tokens:
            return
        with self._mu:
            node = self.root
            token_idx = 0
            prompt_bytes = prompt.encode("utf-8")
            # Walk byte positions; record the last token whose [_, high)
            # offset ends at or before position+1.
            for pos in range(len(prompt_bytes)):
                ch = prompt_bytes[pos : pos + 1]
                nxt = node.children.get(ch)
                if nxt is None:
                    nxt = _TrieNode()
                    node.children[ch] = nxt
                node = nxt
                while (
                    token_idx < len(tokens) and offsets[token_idx][1] <= pos + 1
                ):
                    node.token_id = tokens[token_idx]
                    node.token_index = token_idx
                    token_idx += 1

    def find_longest_contained_tokens(
        self, prompt: str
    ) -> Tuple[List[int], float]:
        contained: List[int] = []
        prompt_bytes = prompt.encode("utf-8")
        if not prompt_bytes:
            return contained, 0.0
        node = self.root
        last_token_index = -1
        covered = 0
        for pos in range(len(prompt_bytes)):
            ch = prompt_bytes[pos : pos + 1]
            nxt = node.children.get(ch)
            if nxt is None:
                break
            node = nxt
            if nxt.token_index > last_token_index:
                last_token_index = nxt.token_index
                covered = pos + 1
        if last_token_index < 0:
            return contained, 0.0
        # Re-walk is avoided by storing only the running ids; collect by index
        # is not possible without token list, so nodes carry ids incrementally.
        # Collect along the path again:
        node = self.root
        for pos in range(covered):
            node = node.children[prompt_bytes[pos : pos + 1]]
            if node.token_index >= 0 and node.token_id is not None:
                if len(contained) <= node.token_index:
                    contained.extend(
                        [0] * (node.token_index + 1 - len(contained))
                    )
                contained[node.token_index] = node.token_id
        return contained, covered / len(prompt_bytes)
