from .dist import (  # noqa: F401
    get_rank,
    get_world_size,
    init_distributed,
    is_distributed,
)
