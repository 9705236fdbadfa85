from . import losses, vtrace  # noqa: F401
