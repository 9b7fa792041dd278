from .server import create_app, run_server  # noqa: F401
