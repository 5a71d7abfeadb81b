"""Op dispatch.

Two implementations of every engine op:

- ``reference`` — plain PyTorch fp32, runs on CPU.  The numerics oracle for
  every HIP kernel test and the compute path for CPU-only engine tests.
- ``hip`` — hand-written CDNA4 HIP kernels (csrc/*.hip) through the in-tree
  ``_C`` extension.  The ONLY path on a GPU: if a tensor is on cuda and the
  extension is missing, ops raise instead of silently falling back, so a GPU
  run that passes is guaranteed to have run the native kernels.
"""
from __future__ import annotations

import torch

_HIP = None
_HIP_ERR: Exception | None = None


def _hip():
    global _HIP, _HIP_ERR
    if _HIP is None and _HIP_ERR is None:
        try:
            from . import hip as m

            _HIP = m
        except Exception as e:  # pragma: no cover
            _HIP_ERR = e
    if _HIP is None:
        raise RuntimeError(
            f"HIP ops extension not available on a GPU tensor path: {_HIP_ERR}\n"
            "Build it with: python build_ext.py (PYTORCH_ROCM_ARCH=gfx950)"
        )
    return _HIP


def _impl(t: torch.Tensor):
    if t.is_cuda:
        return _hip()
    from . import reference as m

    return m


import os

# Measured A/B (tools/gemm_bench.py, profiles/r01_gemm_ab.md): the TunableOp-
# tuned hipBLASLt stream-K kernels hold 1.3-1.6 PF on every engine projection
# shape; the hand-written 8-phase kernel (csrc/gemm_bf16.hip) reaches 1.1 PF
# (zero LDS conflicts, barrier/issue-stall bound).  The library stays the
# default; set ACP_GEMM_MIN_M to route M >= threshold to the custom kernel.
_GEMM_MIN_M = int(os.environ.get("ACP_GEMM_MIN_M", str(1 << 30)))


def linear(x, w):
    """x[*, K] @ w[N, K]^T → [*, N].  Large-M bf16 shapes go to the
    hand-written 8-phase MFMA GEMM (csrc/gemm_bf16.hip); everything else
    (decode-sized M, ragged shards, CPU) to torch.nn.functional.linear."""
    if (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and x.dim() == 2
        and x.shape[0] >= _GEMM_MIN_M
        and w.shape[0] % 256 == 0
        and w.shape[1] % 128 == 0
    ):
        return _hip().gemm_bf16(x, w)
    return torch.nn.functional.linear(x, w)


def rmsnorm(x, weight, eps: float = 1e-5):
    return _impl(x).rmsnorm(x, weight, eps)


def fused_add_rmsnorm(x, residual, weight, eps: float = 1e-5):
    """returns (normed, new_residual) where new_residual = x + residual."""
    return _impl(x).fused_add_rmsnorm(x, residual, weight, eps)


def rope_and_cache(q, k, v, positions, slot_mapping, k_cache, v_cache, cos_sin):
    """Apply rotary embedding to q,k in place and scatter k,v into the paged
    cache at slot_mapping.  q:[N, Hq, D] k/v:[N, Hkv, D]."""
    return _impl(q).rope_and_cache(q, k, v, positions, slot_mapping, k_cache, v_cache, cos_sin)


def attention_prefill(q, k_cache, v_cache, block_table, seq_len, ctx_len, scale):
    """Causal attention for ONE sequence's prefill chunk (test/oracle entry).
    q: [Lq, Hq, D]; KV read from the paged cache; ctx_len = tokens already
    cached before this chunk (the chunk's causal offset)."""
    return _impl(q).attention_prefill(q, k_cache, v_cache, block_table, seq_len, ctx_len, scale)


def attention_prefill_batch(q, k_cache, v_cache, batch, scale, out=None):
    """Causal attention for ALL prefill chunks of a FlatBatch in one launch.
    q: [num_prefill_tokens, Hq, D]."""
    return _impl(q).attention_prefill_batch(q, k_cache, v_cache, batch, scale, out)


def attention_decode_batch(q, k_cache, v_cache, batch, scale, out=None):
    """One-token-per-sequence paged attention.  q: [B, Hq, D]."""
    return _impl(q).attention_decode_batch(q, k_cache, v_cache, batch, scale, out)


def swiglu(gate_up):
    """gate_up: [N, 2*I] (gate | up interleaved as two halves) → [N, I]."""
    return _impl(gate_up).swiglu(gate_up)


def softmax_sample(logits, temperatures, top_ks, top_ps, gen, mask=None,
                   uniforms=None, mask_map=None):
    """Batch sampling: temperature/top-k/top-p with optional boolean vocab
    masks (constrained decoding; compact [M, V] rows + mask_map [B]
    indirection for BPE-scale vocabularies) and optional precomputed
    per-row uniforms (per-request seeded generators).  Returns [B] long."""
    return _impl(logits).softmax_sample(
        logits, temperatures, top_ks, top_ps, gen, mask, uniforms, mask_map
    )
