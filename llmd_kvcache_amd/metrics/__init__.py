from . import collector  # noqa: F401
