from .misc import Timer, human_bytes, set_seed  # noqa: F401
