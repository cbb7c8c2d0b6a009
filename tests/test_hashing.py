"""Golden tests for the vLLM-compatible hashing primitives.

FNV-64a vectors are the published reference vectors (draft-eastlake-fnv);
the CBOR bytes are hand-derived from RFC 8949 canonical encoding rules and
match Go fxamacker/cbor CanonicalEncOptions for the fixed payload shape
``[uint64, []uint32, nil]`` (reference token_processor.go:94-112).
"""

from llmd_kvcache_amd.utils import hashing


class TestFnv64a:
    def test_empty(self):
        assert hashing.fnv1a_64(b"") == 0xCBF29CE484222325

    def test_vectors(self):
        # Published FNV-1a 64 test vectors.
        assert hashing.fnv1a_64(b"a") == 0xAF63DC4C8601EC8C
        assert hashing.fnv1a_64(b"foobar") == 0x85944171F73967E8

    def test_fnv32a(self):
        assert hashing.fnv1a_32(b"") == 0x811C9DC5
        assert hashing.fnv1a_32(b"a") == 0xE40C292C


class TestCanonicalCbor:
    def test_small_ints_inline(self):
        # [0, [1, 2], null] -> 83 00 82 01 02 f6
        assert hashing.cbor_chunk_payload(0, [1, 2]) == bytes.fromhex(
            "830082010 2f6".replace(" ", "")
        )

    def test_shortest_form_boundaries(self):
        # parent 23 -> 0x17 inline; 24 -> 0x18 0x18; 255 -> 0x18 0xff;
        # 256 -> 0x19 0x0100; 2^16 -> 0x1a; 2^32 -> 0x1b
        assert hashing.cbor_encode_uint(23) == b"\x17"
        assert hashing.cbor_encode_uint(24) == b"\x18\x18"
        assert hashing.cbor_encode_uint(255) == b"\x18\xff"
        assert hashing.cbor_encode_uint(256) == b"\x19\x01\x00"
        assert hashing.cbor_encode_uint(65535) == b"\x19\xff\xff"
        assert hashing.cbor_encode_uint(65536) == b"\x1a\x00\x01\x00\x00"
        assert hashing.cbor_encode_uint(2**32 - 1) == b"\x1a\xff\xff\xff\xff"
        assert (
            hashing.cbor_encode_uint(2**32)
            == b"\x1b\x00\x00\x00\x01\x00\x00\x00\x00"
        )

    def test_array_header_sizes(self):
        # 16-token chunk -> array header 0x90 (0x80 | 16 < 24)
        payload = hashing.cbor_chunk_payload(0, list(range(16)))
        assert payload[0] == 0x83
        assert payload[2] == 0x80 | 16
        # 24-token chunk -> 0x98 0x18
        payload = hashing.cbor_chunk_payload(0, [0] * 24)
        assert payload[2:4] == b"\x98\x18"
        # 256-token chunk -> 0x99 0x0100
        payload = hashing.cbor_chunk_payload(0, [0] * 256)
        assert payload[2:5] == b"\x99\x01\x00"

    def test_typical_chunk_bytes(self):
        # parent = large uint64, tokens in various ranges, null tail
        parent = 0xCBF29CE484222325
        tokens = [5, 200, 70000, 4_000_000_000]
        payload = hashing.cbor_chunk_payload(parent, tokens)
        expected = (
            b"\x83"
            + b"\x1b" + parent.to_bytes(8, "big")
            + b"\x84"
            + b"\x05"
            + b"\x18\xc8"
            + b"\x1a\x00\x01\x11\x70"
            + b"\x1a\xee\x6b\x28\x00"
            + b"\xf6"
        )
        assert payload == expected

    def test_chunk_hash_is_fnv_of_payload(self):
        parent = 12345
        tokens = [1, 2, 3]
        payload = hashing.cbor_chunk_payload(parent, tokens)
        assert hashing.chunk_hash(parent, tokens) == hashing.fnv1a_64(payload)


class TestInitHash:
    def test_empty_seed(self):
        assert hashing.init_hash("") == 0xCBF29CE484222325

    def test_seed_string(self):
        assert hashing.init_hash("a") == 0xAF63DC4C8601EC8C
