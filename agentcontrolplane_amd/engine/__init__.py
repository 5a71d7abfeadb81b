from .config import EngineConfig, ModelConfig  # noqa: F401
from .request import ChatResult, SamplingParams  # noqa: F401


def create_engine(*args, **kwargs):
    from .engine import InferenceEngine

    return InferenceEngine(*args, **kwargs)
