from .stopwatch import Stopwatch  # noqa: F401
