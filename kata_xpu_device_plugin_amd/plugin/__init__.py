from . import api  # noqa: F401
