from .scheduler import HivedScheduler  # noqa: F401
