"""Checkpoint save/load (safetensors, HF Llama/Mixtral naming).

There is no network in this deployment, so benchmarks run random-init
weights — but the loading path is real and round-trip tested: ``save``
writes a model in the Hugging Face naming scheme (sharded safetensors +
index) and ``load`` restores it, including TP re-sharding on the way in,
so a real Llama-3/Mixtral checkpoint placed on disk drops straight in.
"""
from __future__ import annotations

import json
import os
from typing import Dict

import torch


def _llama_name_map(model) -> Dict[str, torch.Tensor]:
    """Our weights ↔ HF names.  QKV and gate_up are stored fused in-engine
    and split on save / fused on load."""
    cfg = model.cfg
    out: Dict[str, torch.Tensor] = {
        "model.embed_tokens.weight": model.embed,
        "model.norm.weight": model.final_norm,
    }
    if not cfg.tie_embeddings:
        out["lm_head.weight"] = model.lm_head
    qd = model.n_heads * model.head_dim * model.tp_world
    kvd = model.n_kv_heads * model.head_dim * model.tp_world
    for i, lw in enumerate(model.layers):
        p = f"model.layers.{i}."
        out[p + "input_layernorm.weight"] = lw.input_norm
        out[p + "post_attention_layernorm.weight"] = lw.post_norm
        out[p + "self_attn.q_proj.weight"] = lw.qkv[: qd // model.tp_world]
        out[p + "self_attn.k_proj.weight"] = lw.qkv[
            qd // model.tp_world : (qd + kvd) // model.tp_world
        ]
        out[p + "self_attn.v_proj.weight"] = lw.qkv[(qd + kvd) // model.tp_world :]
        if lw.qkv_bias is not None:
            out[p + "self_attn.q_proj.bias"] = lw.qkv_bias[: qd // model.tp_world]
            out[p + "self_attn.k_proj.bias"] = lw.qkv_bias[
                qd // model.tp_world : (qd + kvd) // model.tp_world
            ]
            out[p + "self_attn.v_proj.bias"] = lw.qkv_bias[(qd + kvd) // model.tp_world :]
        out[p + "self_attn.o_proj.weight"] = lw.o
        if lw.gate_up is not None:
            inter = lw.gate_up.shape[0] // 2
            out[p + "mlp.gate_proj.weight"] = lw.gate_up[:inter]
            out[p + "mlp.up_proj.weight"] = lw.gate_up[inter:]
            out[p + "mlp.down_proj.weight"] = lw.down
    # MoE (Mixtral naming)
    if getattr(model, "routers", None):
        for i in range(cfg.num_layers):
            p = f"model.layers.{i}.block_sparse_moe."
            out[p + "gate.weight"] = model.routers[i]
            for e in range(cfg.num_experts):
                inter = model.expert_gate_up[i].shape[1] // 2
                out[p + f"experts.{e}.w1.weight"] = model.expert_gate_up[i][e][:inter]
                out[p + f"experts.{e}.w3.weight"] = model.expert_gate_up[i][e][inter:]
                out[p + f"experts.{e}.w2.weight"] = model.expert_down[i][e]
    return out


def save_checkpoint(model, path: str, max_shard_bytes: int = 4 << 30) -> None:
    from safetensors.torch import save_file

    if getattr(model, "tp_world", 1) > 1:
        # each rank holds only its shard under the full HF tensor name;
        # writing those would silently produce a corrupt checkpoint
        raise ValueError(
            "save_checkpoint requires tp_world == 1 (gather shards first)"
        )
    os.makedirs(path, exist_ok=True)
    tensors = _llama_name_map(model)
    shards, cur, cur_bytes = [], {}, 0
    for name, t in tensors.items():
        tb = t.numel() * t.element_size()
        if cur and cur_bytes + tb > max_shard_bytes:
            shards.append(cur)
            cur, cur_bytes = {}, 0
        cur[name] = t.contiguous().cpu()
        cur_bytes += tb
    if cur:
        shards.append(cur)
    index = {"metadata": {"total_size": sum(t.numel() * t.element_size() for t in tensors.values())}, "weight_map": {}}
    for si, shard in enumerate(shards):
        fname = f"model-{si + 1:05d}-of-{len(shards):05d}.safetensors"
        save_file(shard, os.path.join(path, fname))
        for name in shard:
            index["weight_map"][name] = fname
    with open(os.path.join(path, "model.safetensors.index.json"), "w") as f:
        json.dump(index, f)


def load_checkpoint(model, path: str) -> None:
    """Load HF-named safetensors into the model, fusing QKV/gate_up and
    taking this rank's TP slice."""
    from safetensors.torch import load_file

    idx_path = os.path.join(path, "model.safetensors.index.json")
    if os.path.exists(idx_path):
        with open(idx_path) as f:
            weight_map = json.load(f)["weight_map"]
        files = sorted(set(weight_map.values()))
    else:
        files = sorted(
            f for f in os.listdir(path) if f.endswith(".safetensors")
        )
    tensors: Dict[str, torch.Tensor] = {}
    for fname in files:
        tensors.update(load_file(os.path.join(path, fname)))

    cfg = model.cfg
    r, w = model.tp_rank, model.tp_world
    hd = cfg.head_dim
    q_per, kv_per = cfg.num_heads // w, cfg.num_kv_heads // w
    inter_per = cfg.intermediate_size // w
    dev, dt = model.device, model.dtype

    def get(name):
        if name not in tensors:
            raise KeyError(f"checkpoint missing {name}")
        return tensors[name]

    model.embed = get("model.embed_tokens.weight").to(dev, dt)
    model.final_norm = get("model.norm.weight").to(dev, dt)
    model.lm_head = (
        model.embed if cfg.tie_embeddings else get("lm_head.weight").to(dev, dt)
    )
    from .llama import LlamaLayerWeights

    model.layers = []
    is_moe = cfg.num_experts > 0
    if is_moe:
        model.routers, model.expert_gate_up, model.expert_down = [], [], []
    for i in range(cfg.num_layers):
        p = f"model.layers.{i}."
        lw = LlamaLayerWeights()
        lw.input_norm = get(p + "input_layernorm.weight").to(dev, dt)
        lw.post_norm = get(p + "post_attention_layernorm.weight").to(dev, dt)
        qw = get(p + "self_attn.q_proj.weight")
        kw = get(p + "self_attn.k_proj.weight")
        vw = get(p + "self_attn.v_proj.weight")
        lw.qkv = torch.cat(
            [
                qw[r * q_per * hd : (r + 1) * q_per * hd],
                kw[r * kv_per * hd : (r + 1) * kv_per * hd],
                vw[r * kv_per * hd : (r + 1) * kv_per * hd],
            ]
        ).to(dev, dt)
        if cfg.attention_bias:
            qb = get(p + "self_attn.q_proj.bias")
            kb = get(p + "self_attn.k_proj.bias")
            vb = get(p + "self_attn.v_proj.bias")
            lw.qkv_bias = torch.cat(
                [
                    qb[r * q_per * hd : (r + 1) * q_per * hd],
                    kb[r * kv_per * hd : (r + 1) * kv_per * hd],
                    vb[r * kv_per * hd : (r + 1) * kv_per * hd],
                ]
            ).to(dev, dt)
        ow = get(p + "self_attn.o_proj.weight")
        lw.o = ow[:, r * q_per * hd : (r + 1) * q_per * hd].to(dev, dt)
        if not is_moe:
            gw = get(p + "mlp.gate_proj.weight")
            uw = get(p + "mlp.up_proj.weight")
            lw.gate_up = torch.cat(
                [gw[r * inter_per : (r + 1) * inter_per], uw[r * inter_per : (r + 1) * inter_per]]
            ).to(dev, dt)
            lw.down = get(p + "mlp.down_proj.weight")[:, r * inter_per : (r + 1) * inter_per].to(dev, dt)
        else:
            lw.gate_up = None
            lw.down = None
            mp = p + "block_sparse_moe."
            model.routers.append(get(mp + "gate.weight").to(dev, dt))
            gus, dns = [], []
            for e in range(cfg.num_experts):
                w1 = get(mp + f"experts.{e}.w1.weight")[r * inter_per : (r + 1) * inter_per]
                w3 = get(mp + f"experts.{e}.w3.weight")[r * inter_per : (r + 1) * inter_per]
                w2 = get(mp + f"experts.{e}.w2.weight")[:, r * inter_per : (r + 1) * inter_per]
                gus.append(torch.cat([w1, w3]))
                dns.append(w2)
            model.expert_gate_up.append(torch.stack(gus).to(dev, dt))
            model.expert_down.append(torch.stack(dns).to(dev, dt))
        model.layers.append(lw)
