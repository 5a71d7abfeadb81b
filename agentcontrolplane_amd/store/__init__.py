from .store import (  # noqa: F401
    ConflictError,
    AlreadyExistsError,
    NotFoundError,
    ResourceStore,
    WatchEvent,
)
