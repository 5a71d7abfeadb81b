from .rest import build_app  # noqa: F401
