from .server import ViewerApp, main  # noqa: F401
