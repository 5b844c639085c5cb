from . import golden  # noqa: F401
