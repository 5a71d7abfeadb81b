"""Plain-PyTorch fp32 reference implementations.

These are the numerics oracles the HIP kernels are tested against
(tests/test_ops_gpu.py compares each kernel to its reference at fp32), and
the compute path for CPU-only engine tests.  Written for clarity, not speed.
"""
from __future__ import annotations

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * weight.float()
    return out.to(x.dtype)


def fused_add_rmsnorm(x, residual, weight, eps: float = 1e-5):
    new_residual = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm(new_residual, weight, eps), new_residual


def build_cos_sin(max_position: int, head_dim: int, theta: float, device, dtype=torch.float32):
    """[max_position, head_dim/2] cos and sin tables (rotate-half RoPE)."""
    inv_freq = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, device=device, dtype=torch.float32) / head_dim)
    )
    t = torch.arange(max_position, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)
    return torch.stack([freqs.cos(), freqs.sin()], dim=0).to(dtype)  # [2, P, D/2]


def _rope_rotate_half(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    # x: [N, H, D]; cos/sin: [N, D/2]
    d2 = x.shape[-1] // 2
    x1, x2 = x[..., :d2].float(), x[..., d2:].float()
    c = cos.unsqueeze(1)
    s = sin.unsqueeze(1)
    out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
    return out.to(x.dtype)


def rope_and_cache(q, k, v, positions, slot_mapping, k_cache, v_cache, cos_sin):
    """q:[N,Hq,D] k,v:[N,Hkv,D]; cos_sin:[2,P,D/2];
    k_cache/v_cache: [num_blocks*block_size, Hkv, D] flat-slot view."""
    cos = cos_sin[0][positions]
    sin = cos_sin[1][positions]
    q.copy_(_rope_rotate_half(q, cos, sin))
    k.copy_(_rope_rotate_half(k, cos, sin))
    _flat(k_cache)[slot_mapping] = k.to(k_cache.dtype)
    _flat(v_cache)[slot_mapping] = v.to(v_cache.dtype)
    return q, k


def _gather_kv(k_cache, v_cache, block_table, seq_len, block_size):
    """Contiguous [S, Hkv, D] K and V for one sequence."""
    slots = []
    for pos in range(seq_len):
        b = block_table[pos // block_size]
        slots.append(int(b) * block_size + pos % block_size)
    idx = torch.tensor(slots, device=k_cache.device, dtype=torch.long)
    return k_cache[idx], v_cache[idx]


def attention_prefill(q, k_cache, v_cache, block_table, seq_len, ctx_len, scale):
    """q: [Lq, Hq, D]; returns [Lq, Hq, D].  Causal: query i (global position
    ctx_len+i) attends to kv positions 0..ctx_len+i."""
    Lq, Hq, D = q.shape
    block_size = _block_size_of(k_cache)
    k, v = _gather_kv(_flat(k_cache), _flat(v_cache), block_table, seq_len, block_size)
    Hkv = k.shape[1]
    rep = Hq // Hkv
    kf = k.float().repeat_interleave(rep, dim=1)   # [S, Hq, D]
    vf = v.float().repeat_interleave(rep, dim=1)
    qf = q.float()
    # scores [Hq, Lq, S]
    scores = torch.einsum("lhd,shd->hls", qf, kf) * scale
    pos_q = torch.arange(ctx_len, ctx_len + Lq, device=q.device).view(1, Lq, 1)
    pos_k = torch.arange(seq_len, device=q.device).view(1, 1, seq_len)
    scores = scores.masked_fill(pos_k > pos_q, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    out = torch.einsum("hls,shd->lhd", probs, vf)
    return out.to(q.dtype)


def attention_prefill_batch(q, k_cache, v_cache, batch, scale, out=None):
    """Loop the per-sequence oracle over a FlatBatch's prefill chunks."""
    outs = []
    row = 0
    for m in batch.prefills:
        outs.append(
            attention_prefill(
                q[row : row + m.query_len], k_cache, v_cache, m.block_table,
                m.seq_len, m.ctx_len, scale,
            )
        )
        row += m.query_len
    res = torch.cat(outs, dim=0) if len(outs) > 1 else outs[0]
    if out is not None:
        out.copy_(res)
        return out
    return res


def attention_decode_batch(q, k_cache, v_cache, batch, scale, out=None):
    """q: [B, Hq, D]; decode tables/lens come from the FlatBatch."""
    res = attention_decode_raw(
        q, k_cache, v_cache, batch.decode_block_tables, batch.decode_seq_lens, scale
    )
    if out is not None:
        out.copy_(res)
        return out
    return res


def attention_decode_raw(q, k_cache, v_cache, block_tables, seq_lens, scale):
    B, Hq, D = q.shape
    outs = []
    for i in range(B):
        out = attention_prefill(
            q[i : i + 1],
            k_cache,
            v_cache,
            block_tables[i].tolist(),
            int(seq_lens[i]),
            int(seq_lens[i]) - 1,
            scale,
        )
        outs.append(out)
    return torch.cat(outs, dim=0)


def _flat(cache):
    """Cache stored [num_blocks, block_size, Hkv, D] → flat [slots, Hkv, D]."""
    if cache.dim() == 4:
        nb, bs, h, d = cache.shape
        return cache.view(nb * bs, h, d)
    return cache


def _block_size_of(cache):
    if cache.dim() == 4:
        return cache.shape[1]
    raise ValueError("pass the 4-D [num_blocks, block_size, H, D] cache")


def swiglu(gate_up: torch.Tensor) -> torch.Tensor:
    gate, up = gate_up.float().chunk(2, dim=-1)
    return (torch.nn.functional.silu(gate) * up).to(gate_up.dtype)


def softmax_sample(logits, temperatures, top_ks, top_ps, gen, mask=None,
                   uniforms=None, mask_map=None):
    """logits: [B, V] float; temperatures/top_ps: [B] float; top_ks: [B] long
    (0 = off); mask: [B, V] bool (True = allowed) or None; uniforms: [B]
    precomputed draws (per-request seeds) — inverse-CDF like the HIP kernel.
    Greedy when temperature == 0."""
    logits = logits.float().clone()
    B, V = logits.shape
    if mask is not None and mask_map is not None:
        # expand the compact [M, V] mask rows (hip kernel indirection)
        for i in range(B):
            mi = int(mask_map[i])
            if mi >= 0:
                logits[i].masked_fill_(~mask[mi], float("-inf"))
    elif mask is not None:
        logits.masked_fill_(~mask, float("-inf"))
    out = torch.empty(B, dtype=torch.long, device=logits.device)
    greedy = temperatures <= 0
    if greedy.any():
        out[greedy] = logits[greedy].argmax(-1)
    rows = (~greedy).nonzero(as_tuple=True)[0]
    for i in rows.tolist():
        row = logits[i] / temperatures[i]
        k = int(top_ks[i])
        if k > 0 and k < V:
            kth = torch.topk(row, k).values[-1]
            row[row < kth] = float("-inf")
        p = float(top_ps[i])
        if p < 1.0:
            sorted_logits, sorted_idx = torch.sort(row, descending=True)
            probs = torch.softmax(sorted_logits, dim=-1)
            cum = torch.cumsum(probs, dim=-1)
            cut = cum - probs > p  # keep tokens until cumulative prob exceeds p
            sorted_logits[cut] = float("-inf")
            row = torch.full_like(row, float("-inf"))
            row[sorted_idx] = sorted_logits
        probs = torch.softmax(row, dim=-1)
        if uniforms is not None:
            cdf = torch.cumsum(probs, dim=-1)
            out[i] = int(torch.searchsorted(cdf, uniforms[i].to(cdf) * cdf[-1]).clamp(max=probs.numel() - 1))
        else:
            out[i] = torch.multinomial(probs, 1, generator=gen).squeeze(-1)
    return out
