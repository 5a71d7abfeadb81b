"""Mixtral-family (sparse MoE) model.

Same attention stack as Llama; the dense SwiGLU MLP becomes a top-2-of-8
expert mixture: router linear → softmax over experts → top-k weights
renormalized → weighted sum of the chosen experts' SwiGLU outputs.

Single-GPU execution groups tokens by expert and runs one gate_up/down GEMM
pair per expert with >0 tokens.  Under expert parallelism
(parallel/moe.py), experts are sharded across ranks and token routing goes
through an RCCL all-to-all over xGMI.
"""
from __future__ import annotations

import os

from typing import List

import torch
import torch.nn.functional as F

from .. import ops
from ..engine.config import EngineConfig, ModelConfig
from .llama import LlamaForCausalLM


class MixtralForCausalLM(LlamaForCausalLM):
    def __init__(self, cfg: ModelConfig, ecfg: EngineConfig, device: str,
                 tp_rank: int = 0, tp_world: int = 1):
        super().__init__(cfg, ecfg, device, tp_rank, tp_world)
        self.num_experts = cfg.num_experts
        self.top_k = cfg.num_experts_per_tok
        self.routers: List[torch.Tensor] = []
        self.expert_gate_up: List[torch.Tensor] = []   # per layer: [E, 2I, H]
        self.expert_down: List[torch.Tensor] = []      # per layer: [E, H, I]
        # expert-parallel dispatch hook installed by parallel/moe.py
        self.moe_dispatch = None

    def random_init(self, seed: int = 0) -> None:
        super().random_init(seed)
        # routers are REPLICATED (every rank must route identically in EP);
        # expert weights are sharded/owned per rank
        g_rep = torch.Generator(device=self.device).manual_seed(seed + 500)
        g = torch.Generator(device=self.device).manual_seed(seed + 1000 + self.tp_rank)
        h = self.cfg.hidden_size
        std = 0.02

        def randw(*shape, gen=None):
            w = torch.empty(shape, dtype=self.dtype, device=self.device)
            w.normal_(0.0, std, generator=gen or g)
            return w

        self.routers = []
        self.expert_gate_up = []
        self.expert_down = []
        for lw in self.layers:
            # the dense MLP weights are replaced per-expert; free them
            lw.gate_up = None
            lw.down = None
            self.routers.append(randw(self.num_experts, h, gen=g_rep))
            self.expert_gate_up.append(randw(self.num_experts, 2 * self.inter, h))
            self.expert_down.append(randw(self.num_experts, h, self.inter))

    #: below this many tokens the dense path (ALL experts via batched
    #: GEMMs, static control flow — hipGraph-capturable) beats the sparse
    #: loop (~5·E launches of small ragged GEMMs).  The r02 Mixtral
    #: profile prices dense as the win through ~1k tokens (sparse leaves
    #: the GPU 28% busy at 64-task serving), but admitting N≈1k FAULTS on
    #: hardware under BOTH dense formulations tried (the old stride-0
    #: expand+transpose AND the plain strided [E,N,*] rewrite) — the
    #: fault is suspected inside the batched-GEMM library at
    #: [8, ~1k, 28672]-class shapes and needs a standalone repro before
    #: this threshold moves (ACP_MOE_DENSE_THRESHOLD overrides for
    #: experiments; docs/roadmap.md tracks the grouped-GEMM fix).
    dense_moe_threshold = int(os.environ.get("ACP_MOE_DENSE_THRESHOLD", "160"))

    def _moe_mlp(self, li: int, x: torch.Tensor) -> torch.Tensor:
        """x: [N, H] → [N, H] via top-k expert mixture."""
        router_logits = F.linear(x, self.routers[li]).float()       # [N, E]
        probs = torch.softmax(router_logits, dim=-1)
        topw, topi = torch.topk(probs, self.top_k, dim=-1)          # [N, k]
        topw = topw / topw.sum(dim=-1, keepdim=True)
        if self.moe_dispatch is not None:
            return self.moe_dispatch(li, x, topi, topw.to(x.dtype))
        if x.shape[0] <= self.dense_moe_threshold:
            return self._moe_dense(li, x, topi, topw)
        out = torch.zeros_like(x)
        flat_expert = topi.reshape(-1)                              # [N*k]
        flat_rows = (
            torch.arange(x.shape[0], device=x.device).repeat_interleave(self.top_k)
        )
        flat_w = topw.reshape(-1).to(x.dtype)
        for e in range(self.num_experts):
            sel = (flat_expert == e).nonzero(as_tuple=True)[0]
            if sel.numel() == 0:
                continue
            rows = flat_rows[sel]
            xe = x[rows]
            gate_up = ops.linear(xe, self.expert_gate_up[li][e])
            ye = ops.linear(ops.swiglu(gate_up), self.expert_down[li][e])
            out.index_add_(0, rows, ye * flat_w[sel].unsqueeze(-1))
        return out

    def _moe_dense(self, li: int, x: torch.Tensor, topi, topw) -> torch.Tensor:
        """All-experts batched GEMMs + routed combine (same numerics as the
        sparse loop: non-selected experts get weight 0).

        Formulated [E, N, *] end-to-end: the weights provide the batch dim
        (op_B transposes handled inside rocBLAS) and the activations stay
        in row-major [E, N, C] throughout — no stride-0 expanded operand
        and no transpose-contiguous round trips (the earlier
        bmm-[E,2I,N]-then-transpose formulation copied two [E, N, 2I]
        intermediates per layer, and its expanded operand faulted on
        hardware when admitted at N≈1k)."""
        E = self.num_experts
        N, H = x.shape
        # [E, N, H] @ [E, H, 2I] -> [E, N, 2I]; the small explicit replica
        # of x (E·N·H, ~64 MB at N=1k) buys plain strided batched GEMMs —
        # no stride-0 operands anywhere
        xe = x.unsqueeze(0).expand(E, N, H).contiguous()
        gate_up = torch.matmul(xe, self.expert_gate_up[li].transpose(1, 2))
        act = ops.swiglu(gate_up.contiguous())                       # [E, N, I]
        # [E, N, I] @ [E, I, H] -> [E, N, H]
        y = torch.matmul(act, self.expert_down[li].transpose(1, 2))
        # routing weights as a dense [N, E] matrix
        w = torch.zeros(N, E, dtype=x.dtype, device=x.device)
        w.scatter_(1, topi, topw.to(x.dtype))
        # out[n, h] = sum_e w[n, e] * y[e, n, h]
        return torch.einsum("ne,enh->nh", w, y)

    def forward(self, batch) -> torch.Tensor:
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
            x = self._moe_mlp(li, normed2)
            if self.all_reduce is not None and self.moe_dispatch is None:
                # TP-sharded experts (intermediate split): partial sums reduce
                # here; EP dispatch returns fully-combined rows instead
                x = self.all_reduce(x)
        normed, _ = ops.fused_add_rmsnorm(x, residual, self.final_norm, cfg.rms_eps)
        sel = normed[batch.logit_rows]
        return ops.linear(sel, self.lm_head).float()
