from . import chat_completions  # noqa: F401
