"""Llama-family model over the MI355X op set.

Architecture (Llama-3): RMSNorm → fused-QKV proj → RoPE (rotate-half,
θ=500000) → GQA attention on the paged KV cache → O proj → RMSNorm →
SwiGLU MLP, pre-norm residuals, untied LM head.

Execution model: a flat FlatBatch per step (mixed prefill chunks + decode
tokens).  GEMMs go through torch.nn.functional.linear (hipBLASLt on ROCm);
everything between the GEMMs — rmsnorm, rope+KV-scatter, attention over the
paged cache, SwiGLU — is the op set in ``agentcontrolplane_amd.ops`` (HIP
kernels on a GPU, fp32 torch references on CPU).

Tensor parallelism: head-parallel attention + column/row-parallel MLP;
rank r holds heads [r·Hq/w, (r+1)·Hq/w) and MLP columns likewise; the
all-reduce after o-proj and down-proj is issued by the caller (engine
worker) so communication can overlap with the next layer's norm.
"""
from __future__ import annotations

import math
from typing import List, Optional

import torch
import torch.nn.functional as F

from .. import ops
from ..engine.batch import FlatBatch
from ..engine.config import EngineConfig, ModelConfig
from ..ops.reference import build_cos_sin


def _dtype_of(cfg: ModelConfig):
    return {"bfloat16": torch.bfloat16, "float16": torch.float16, "float32": torch.float32}[
        cfg.dtype
    ]


class LlamaLayerWeights:
    __slots__ = ("input_norm", "qkv", "qkv_bias", "o", "post_norm", "gate_up", "down")

    def __init__(self):
        # Qwen2-style attention bias; None for Llama/Mixtral
        self.qkv_bias = None


class LlamaForCausalLM:
    def __init__(
        self,
        cfg: ModelConfig,
        ecfg: EngineConfig,
        device: str,
        tp_rank: int = 0,
        tp_world: int = 1,
    ):
        self.cfg = cfg
        self.ecfg = ecfg
        self.device = torch.device(device)
        self.dtype = _dtype_of(cfg)
        self.tp_rank = tp_rank
        self.tp_world = tp_world
        assert cfg.num_heads % tp_world == 0 and cfg.num_kv_heads % tp_world == 0, (
            "TP degree must divide head counts"
        )
        self.n_heads = cfg.num_heads // tp_world
        self.n_kv_heads = cfg.num_kv_heads // tp_world
        self.inter = cfg.intermediate_size // tp_world
        self.head_dim = cfg.head_dim
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        self.layers: List[LlamaLayerWeights] = []
        self.embed: Optional[torch.Tensor] = None
        self.lm_head: Optional[torch.Tensor] = None
        self.final_norm: Optional[torch.Tensor] = None
        self.cos_sin = build_cos_sin(
            cfg.max_position, cfg.head_dim, cfg.rope_theta, self.device
        )
        # per-layer paged KV caches, allocated by allocate_kv_cache
        self.k_caches: List[torch.Tensor] = []
        self.v_caches: List[torch.Tensor] = []
        # optional TP all-reduce hook installed by the parallel engine
        self.all_reduce = None

    # ------------------------------------------------------------- weights

    def random_init(self, seed: int = 0) -> None:
        """Random-init weights of the real shapes (synthetic bench: no
        network for checkpoints — BASELINE.md).  Init in manageable chunks
        directly on the device.

        TP consistency: REPLICATED tensors (embed, lm_head, norms) must be
        bit-identical on every rank — they feed the all-reduced residual
        stream — so they come from a rank-independent generator.  SHARDED
        tensors are each rank's own slice of the logical full model, so a
        rank-offset seed just picks which random slice this rank holds."""
        g_rep = torch.Generator(device=self.device).manual_seed(seed)
        g = torch.Generator(device=self.device).manual_seed(seed + 1 + self.tp_rank)
        cfg = self.cfg
        h = cfg.hidden_size
        std = 0.02

        def randw(*shape, gen=None):
            w = torch.empty(shape, dtype=self.dtype, device=self.device)
            w.normal_(0.0, std, generator=gen or g)
            return w

        self.embed = randw(cfg.vocab_size, h, gen=g_rep)
        self.lm_head = self.embed if cfg.tie_embeddings else randw(cfg.vocab_size, h, gen=g_rep)
        self.final_norm = torch.ones(h, dtype=self.dtype, device=self.device)
        qd = self.n_heads * self.head_dim
        kvd = self.n_kv_heads * self.head_dim
        for _ in range(cfg.num_layers):
            lw = LlamaLayerWeights()
            lw.input_norm = torch.ones(h, dtype=self.dtype, device=self.device)
            lw.qkv = randw(qd + 2 * kvd, h)
            if self.cfg.attention_bias:
                b = torch.empty(qd + 2 * kvd, dtype=self.dtype, device=self.device)
                b.normal_(0.0, std, generator=g)
                lw.qkv_bias = b
            lw.o = randw(h, qd)
            lw.post_norm = torch.ones(h, dtype=self.dtype, device=self.device)
            lw.gate_up = randw(2 * self.inter, h)
            lw.down = randw(h, self.inter)
            self.layers.append(lw)

    def load_safetensors(self, path: str) -> None:
        """Load an HF-named safetensors checkpoint dir (round-trip tested in
        tests/test_checkpoint.py; handles TP slicing)."""
        from .weights import load_checkpoint

        load_checkpoint(self, path)

    # ------------------------------------------------------------ KV cache

    def allocate_kv_cache(self, num_blocks: int, block_size: int) -> None:
        # uninitialized on purpose: every slot is written (rope+scatter)
        # before any attention reads it; zero-filling 200+ GB costs ~35 ms/GB
        shape = (num_blocks, block_size, self.n_kv_heads, self.head_dim)
        self.k_caches = [
            torch.empty(shape, dtype=self.dtype, device=self.device)
            for _ in range(self.cfg.num_layers)
        ]
        self.v_caches = [
            torch.empty(shape, dtype=self.dtype, device=self.device)
            for _ in range(self.cfg.num_layers)
        ]

    def kv_block_bytes(self, block_size: int) -> int:
        return (
            2 * self.cfg.num_layers * block_size * self.n_kv_heads * self.head_dim
            * self.dtype.itemsize
        )

    # -------------------------------------------------------------- forward

    def _attention(self, layer_idx: int, q, batch: FlatBatch):
        """q: [N, Hq, D] post-rope; returns [N, Hq, D] (one preallocated
        output; prefill and decode kernels write disjoint row ranges)."""
        kc, vc = self.k_caches[layer_idx], self.v_caches[layer_idx]
        N = q.shape[0]
        out = torch.empty(
            (N, self.n_heads, self.head_dim), dtype=self.dtype, device=q.device
        )
        row = batch.num_prefill_tokens
        if batch.prefills:
            ops.attention_prefill_batch(q[:row], kc, vc, batch, self.scale, out=out[:row])
        if batch.num_decode:
            ops.attention_decode_batch(q[row:], kc, vc, batch, self.scale, out=out[row:])
        return out

    def forward(self, batch: FlatBatch) -> torch.Tensor:
        """Returns logits [len(batch.sample_seq_ids), vocab]."""
        cfg = self.cfg
        N = batch.num_tokens
        x = F.embedding(batch.token_ids, self.embed)
        residual = None
        qd = self.n_heads * self.head_dim
        kvd = self.n_kv_heads * self.head_dim
        for li, lw in enumerate(self.layers):
            if residual is None:
                residual = x
                normed = ops.rmsnorm(x, lw.input_norm, cfg.rms_eps)
            else:
                normed, residual = ops.fused_add_rmsnorm(x, residual, lw.input_norm, cfg.rms_eps)
            qkv = ops.linear(normed, lw.qkv)
            if lw.qkv_bias is not None:
                qkv = qkv + lw.qkv_bias
            # strided views into the fused GEMM output: the HIP kernels take
            # a token stride, so no .contiguous() copies on the hot path
            row = qkv.stride(0)
            q = qkv.as_strided((N, self.n_heads, self.head_dim), (row, self.head_dim, 1))
            k = qkv.as_strided(
                (N, self.n_kv_heads, self.head_dim), (row, self.head_dim, 1), qd
            )
            v = qkv.as_strided(
                (N, self.n_kv_heads, self.head_dim), (row, self.head_dim, 1), qd + kvd
            )
            q, k = ops.rope_and_cache(
                q, k, v, batch.positions, batch.slot_mapping,
                self.k_caches[li], self.v_caches[li], self.cos_sin,
            )
            attn = self._attention(li, q, batch)
            o = ops.linear(attn.reshape(N, qd), lw.o)
            if self.all_reduce is not None:
                o = self.all_reduce(o)
            normed2, residual = ops.fused_add_rmsnorm(o, residual, lw.post_norm, cfg.rms_eps)
            gate_up = ops.linear(normed2, lw.gate_up)
            mlp = ops.linear(ops.swiglu(gate_up), lw.down)
            if self.all_reduce is not None:
                mlp = self.all_reduce(mlp)
            x = mlp
        normed, _ = ops.fused_add_rmsnorm(x, residual, self.final_norm, cfg.rms_eps)
        sel = normed[batch.logit_rows]
        return ops.linear(sel, self.lm_head).float()
