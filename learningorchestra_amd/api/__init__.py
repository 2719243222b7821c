from .app import create_app  # noqa: F401
