"""GPU op dispatch into the hand-written CDNA4 kernels (csrc/*.hip).

No fallbacks: a cuda-tensor call that cannot reach a kernel raises, so a
passing GPU run proves the native path ran (the round-end loader check
looks for agentcontrolplane_amd/_C.so in the process maps).
"""
from __future__ import annotations

import torch

from .. import _C  # in-tree extension; ImportError surfaces via ops.__init__


def rmsnorm(x, weight, eps: float = 1e-5):
    out = torch.empty_like(x)
    _C.rmsnorm(out, x, weight, eps)
    return out


def fused_add_rmsnorm(x, residual, weight, eps: float = 1e-5):
    out = torch.empty_like(x)
    _C.fused_add_rmsnorm(out, residual, x, weight, eps)  # residual updated in place
    return out, residual


def rope_and_cache(q, k, v, positions, slot_mapping, k_cache, v_cache, cos_sin):
    _C.rope_cache(q, k, v, positions, slot_mapping, k_cache, v_cache, cos_sin)
    return q, k


def attention_prefill(q, k_cache, v_cache, block_table, seq_len, ctx_len, scale):
    """Single-sequence entry (tests): wrap into a one-seq batched call."""
    from ..engine.batch import FlatBatch, SeqMeta

    meta = SeqMeta(
        seq_id=0, query_len=q.shape[0], seq_len=seq_len, ctx_len=ctx_len,
        block_table=list(block_table), needs_logits=True,
    )
    dummy = torch.empty(0)
    batch = FlatBatch(
        token_ids=torch.empty(q.shape[0], device=q.device),
        positions=dummy, slot_mapping=dummy, prefills=[meta],
        num_prefill_tokens=q.shape[0], decode_seq_ids=[],
        decode_block_tables=None, decode_seq_lens=None,
        logit_rows=dummy, sample_seq_ids=[],
    )
    return attention_prefill_batch(q, k_cache, v_cache, batch, scale)


import os

_PREFILL_IMPL = os.environ.get("ACP_PREFILL_IMPL", "mfma")  # mfma | v0


def _attn_out(q):
    return torch.empty(q.shape, dtype=q.dtype, device=q.device)


def attention_prefill_batch(q, k_cache, v_cache, batch, scale, out=None):
    if out is None:
        out = _attn_out(q)
    if _PREFILL_IMPL == "mfma":
        meta = batch.prefill_meta(tile_q=128)
        _C.prefill_attn_mfma(
            out, q, k_cache, v_cache, meta.block_tables, meta.seq_lens, meta.ctx_lens,
            meta.row_starts, meta.tile_seq, meta.tile_q0, scale,
        )
    else:
        meta = batch.prefill_meta(tile_q=64)
        _C.prefill_attn(
            out, q, k_cache, v_cache, meta.block_tables, meta.seq_lens, meta.ctx_lens,
            meta.row_starts, meta.tile_seq, meta.tile_q0, scale,
        )
    return out


def attention_decode_batch(q, k_cache, v_cache, batch, scale, out=None):
    if out is None:
        out = _attn_out(q)
    _C.decode_attn(
        out, q, k_cache, v_cache, batch.decode_tables_i32(), batch.decode_lens_i32(),
        scale,
    )
    return out


def attention_decode_raw(q, k_cache, v_cache, block_tables, seq_lens, scale):
    out = _attn_out(q)
    _C.decode_attn(
        out, q, k_cache, v_cache, block_tables.int().contiguous(),
        seq_lens.int().contiguous(), scale,
    )
    return out


def gemm_bf16(x, w):
    out = torch.empty((x.shape[0], w.shape[0]), dtype=x.dtype, device=x.device)
    _C.gemm_bf16(out, x.contiguous(), w)
    return out


def swiglu(gate_up):
    inter = gate_up.shape[-1] // 2
    out = torch.empty(
        (*gate_up.shape[:-1], inter), dtype=gate_up.dtype, device=gate_up.device
    )
    _C.swiglu(out, gate_up)
    return out


def softmax_sample(logits, temperatures, top_ks, top_ps, gen, mask=None,
                   uniforms=None, mask_map=None):
    B, V = logits.shape
    out = torch.empty(B, dtype=torch.long, device=logits.device)
    if uniforms is None:
        uniforms = torch.rand(B, device=logits.device, generator=gen)
    if V > 512:
        # BPE-scale vocab: LDS radix-select kernel, bf16 logits unconverted
        # (csrc/sampling_fullvocab.hip); masks are compact [M, V] rows with
        # mask_map [B] indirection (-1 = unconstrained row)
        if logits.dtype not in (torch.bfloat16, torch.float32):
            logits = logits.float()
        if mask is not None and mask_map is None:
            mask_map = torch.arange(B, dtype=torch.int32, device=logits.device)
        _C.sample_fullvocab(
            out, logits.contiguous(), temperatures.float().contiguous(),
            top_ks.long().contiguous(), top_ps.float().contiguous(),
            uniforms.contiguous(),
            mask.contiguous() if mask is not None else None,
            mask_map.contiguous() if mask is not None else None,
        )
        return out
    _C.sample(
        out, logits.float().contiguous(), temperatures.float().contiguous(),
        top_ks.long().contiguous(), top_ps.float().contiguous(),
        uniforms.contiguous(), mask.contiguous() if mask is not None else None,
    )
    return out
