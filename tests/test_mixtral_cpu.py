"""Mixtral (sparse MoE) model tier on CPU: routing correctness + engine
integration with the tiny-moe preset (BASELINE.json config 5 at test scale)."""
import pytest
import torch

from agentcontrolplane_amd.engine.batch import FlatBatch, SeqMeta
from agentcontrolplane_amd.engine.config import PRESETS, EngineConfig
from agentcontrolplane_amd.engine.engine import InferenceEngine
from agentcontrolplane_amd.engine.request import SamplingParams
from agentcontrolplane_amd.models.mixtral import MixtralForCausalLM


def _batch(T, device="cpu"):
    meta = SeqMeta(
        seq_id=1, query_len=T, seq_len=T, ctx_len=0,
        block_table=list(range((T + 15) // 16)), needs_logits=True,
    )
    return FlatBatch(
        token_ids=torch.randint(0, 512, (T,)),
        positions=torch.arange(T),
        slot_mapping=torch.arange(T),
        prefills=[meta],
        num_prefill_tokens=T,
        decode_seq_ids=[],
        decode_block_tables=None,
        decode_seq_lens=None,
        logit_rows=torch.tensor([T - 1]),
        sample_seq_ids=[1],
    )


def test_moe_routing_weights_sum():
    """Top-k routing weights renormalize to 1 and mix exactly k experts."""
    torch.manual_seed(0)
    cfg = PRESETS["tiny-moe"]
    ecfg = EngineConfig(model="tiny-moe", device="cpu", num_kv_blocks=64)
    model = MixtralForCausalLM(cfg, ecfg, "cpu")
    model.random_init(0)
    x = torch.randn(5, cfg.hidden_size)
    out = model._moe_mlp(0, x)
    assert out.shape == x.shape
    assert torch.isfinite(out).all()


def test_moe_matches_dense_when_one_expert():
    """A 1-expert 'MoE' must equal the dense SwiGLU MLP with those weights."""
    import dataclasses

    import torch.nn.functional as F

    from agentcontrolplane_amd import ops

    torch.manual_seed(1)
    cfg = dataclasses.replace(PRESETS["tiny-moe"], num_experts=1, num_experts_per_tok=1)
    ecfg = EngineConfig(model="tiny-moe", device="cpu", num_kv_blocks=64)
    model = MixtralForCausalLM(cfg, ecfg, "cpu")
    model.random_init(0)
    x = torch.randn(7, cfg.hidden_size)
    got = model._moe_mlp(0, x)
    want = F.linear(
        ops.swiglu(F.linear(x, model.expert_gate_up[0][0])), model.expert_down[0][0]
    )
    assert (got - want).abs().max().item() < 1e-5


def test_moe_model_forward():
    torch.manual_seed(2)
    cfg = PRESETS["tiny-moe"]
    ecfg = EngineConfig(model="tiny-moe", device="cpu", num_kv_blocks=64)
    model = MixtralForCausalLM(cfg, ecfg, "cpu")
    model.random_init(0)
    model.allocate_kv_cache(64, 16)
    logits = model.forward(_batch(12))
    assert logits.shape == (1, cfg.vocab_size)
    assert torch.isfinite(logits).all()


def test_moe_engine_end_to_end():
    eng = InferenceEngine(
        EngineConfig(
            model="tiny-moe", device="cpu", num_kv_blocks=1024, kv_block_size=16,
            max_prefill_tokens=256, request_timeout_s=120,
        )
    )
    try:
        res = eng.chat(
            [{"role": "user", "content": "route me"}],
            tools=[{"type": "function", "function": {"name": "t__x", "parameters": {}}}],
            sampling=SamplingParams(max_tokens=48, temperature=0.8, tool_choice="required"),
        )
        assert res.finish_reason == "tool_calls"
        assert res.tool_calls[0]["function"]["name"] == "t__x"
    finally:
        eng.stop()


def test_dense_moe_path_matches_sparse():
    """The all-experts dense (graph-capturable) decode path must equal the
    sparse routing loop."""
    torch.manual_seed(5)
    cfg = PRESETS["tiny-moe"]
    ecfg = EngineConfig(model="tiny-moe", device="cpu", num_kv_blocks=64)
    model = MixtralForCausalLM(cfg, ecfg, "cpu")
    model.random_init(0)
    x = torch.randn(9, cfg.hidden_size)
    model.dense_moe_threshold = 0
    sparse = model._moe_mlp(0, x.clone())
    model.dense_moe_threshold = 160
    dense = model._moe_mlp(0, x.clone())
    assert (sparse - dense).abs().max().item() < 1e-4
