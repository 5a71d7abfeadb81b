from .llama import LlamaForCausalLM  # noqa: F401
from .mixtral import MixtralForCausalLM  # noqa: F401


def create_model(model_config, engine_config, device, tp_rank: int = 0, tp_world: int = 1):
    if model_config.is_moe:
        return MixtralForCausalLM(model_config, engine_config, device, tp_rank, tp_world)
    return LlamaForCausalLM(model_config, engine_config, device, tp_rank, tp_world)
