from . import hocon  # noqa: F401
