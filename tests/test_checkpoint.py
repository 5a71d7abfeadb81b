"""Checkpoint round-trip: save a random model (HF safetensors naming),
reload it (including a TP slice), forward must be identical."""
import dataclasses

import torch

from agentcontrolplane_amd.engine.batch import FlatBatch, SeqMeta
from agentcontrolplane_amd.engine.config import PRESETS, EngineConfig
from agentcontrolplane_amd.models.llama import LlamaForCausalLM
from agentcontrolplane_amd.models.mixtral import MixtralForCausalLM
from agentcontrolplane_amd.models.weights import load_checkpoint, save_checkpoint


def _batch(T, vocab):
    meta = SeqMeta(seq_id=1, query_len=T, seq_len=T, ctx_len=0,
                   block_table=[0], needs_logits=True)
    g = torch.Generator().manual_seed(3)
    return FlatBatch(
        token_ids=torch.randint(0, vocab, (T,), generator=g),
        positions=torch.arange(T), slot_mapping=torch.arange(T),
        prefills=[meta], num_prefill_tokens=T, decode_seq_ids=[],
        decode_block_tables=None, decode_seq_lens=None,
        logit_rows=torch.tensor([T - 1]), sample_seq_ids=[1],
    )


def test_llama_checkpoint_roundtrip(tmp_path):
    cfg = dataclasses.replace(PRESETS["tiny"], dtype="float32")
    ecfg = EngineConfig(model="tiny", device="cpu", num_kv_blocks=64)
    m1 = LlamaForCausalLM(cfg, ecfg, "cpu")
    m1.random_init(0)
    m1.allocate_kv_cache(64, 16)
    save_checkpoint(m1, str(tmp_path / "ckpt"))
    m2 = LlamaForCausalLM(cfg, ecfg, "cpu")
    load_checkpoint(m2, str(tmp_path / "ckpt"))
    m2.allocate_kv_cache(64, 16)
    l1 = m1.forward(_batch(8, cfg.vocab_size))
    l2 = m2.forward(_batch(8, cfg.vocab_size))
    assert torch.equal(l1, l2)


def test_llama_checkpoint_tp_slice(tmp_path):
    cfg = dataclasses.replace(PRESETS["tiny"], dtype="float32")
    ecfg = EngineConfig(model="tiny", device="cpu", num_kv_blocks=64)
    m1 = LlamaForCausalLM(cfg, ecfg, "cpu")
    m1.random_init(0)
    save_checkpoint(m1, str(tmp_path / "ckpt"))
    shard = LlamaForCausalLM(cfg, ecfg, "cpu", tp_rank=1, tp_world=2)
    load_checkpoint(shard, str(tmp_path / "ckpt"))
    # rank-1 q slice == second half of the full q projection
    qd = cfg.num_heads * cfg.head_dim
    half = qd // 2
    assert torch.equal(shard.layers[0].qkv[:half], m1.layers[0].qkv[half:qd])


def test_mixtral_checkpoint_roundtrip(tmp_path):
    cfg = PRESETS["tiny-moe"]
    ecfg = EngineConfig(model="tiny-moe", device="cpu", num_kv_blocks=64)
    m1 = MixtralForCausalLM(cfg, ecfg, "cpu")
    m1.random_init(0)
    m1.allocate_kv_cache(64, 16)
    save_checkpoint(m1, str(tmp_path / "ckpt"))
    m2 = MixtralForCausalLM(cfg, ecfg, "cpu")
    load_checkpoint(m2, str(tmp_path / "ckpt"))
    m2.allocate_kv_cache(64, 16)
    l1 = m1.forward(_batch(8, cfg.vocab_size))
    l2 = m2.forward(_batch(8, cfg.vocab_size))
    assert torch.equal(l1, l2)


def test_random_init_tp_replicated_consistency():
    """Replicated tensors (embed, lm_head, routers) must be identical
    across TP ranks; sharded tensors are each rank's own slice."""
    from agentcontrolplane_amd.engine.config import PRESETS, EngineConfig
    from agentcontrolplane_amd.models.llama import LlamaForCausalLM
    from agentcontrolplane_amd.models.mixtral import MixtralForCausalLM

    ecfg = EngineConfig(model="tiny", device="cpu")
    r0 = LlamaForCausalLM(PRESETS["tiny"], ecfg, "cpu", tp_rank=0, tp_world=2)
    r1 = LlamaForCausalLM(PRESETS["tiny"], ecfg, "cpu", tp_rank=1, tp_world=2)
    r0.random_init(7)
    r1.random_init(7)
    assert torch.equal(r0.embed, r1.embed)
    assert torch.equal(r0.lm_head, r1.lm_head)
    assert not torch.equal(r0.layers[0].qkv, r1.layers[0].qkv)

    m0 = MixtralForCausalLM(PRESETS["tiny-moe"], ecfg, "cpu", tp_rank=0, tp_world=2)
    m1 = MixtralForCausalLM(PRESETS["tiny-moe"], ecfg, "cpu", tp_rank=1, tp_world=2)
    m0.random_init(7)
    m1.random_init(7)
    assert torch.equal(m0.routers[0], m1.routers[0])
    assert not torch.equal(m0.expert_gate_up[0], m1.expert_gate_up[0])
