from . import coord  # noqa: F401
