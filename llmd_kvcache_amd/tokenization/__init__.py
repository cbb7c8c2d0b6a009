from .pool import TokenizationConfig, TokenizationPool  # noqa: F401
from .tokenizer import (  # noqa: F401
    CachedTokenizer,
    CompositeTokenizer,
    HFTokenizerConfig,
    LocalTokenizerConfig,
    TokenizationError,
    Tokenizer,
)
