from . import kge  # noqa: F401
