"""Independent fp64 brute-force cross-checks of the fp32 torch oracles in
ops/reference.py — the oracles are what every HIP kernel is validated
against, so they get their own ground truth (explicit python loops, no
shared einsum/softmax code paths)."""
import math

import numpy as np
import torch

from agentcontrolplane_amd.ops import reference as ref


def paged_cache(S, Hkv, D, block_size, seed):
    rng = np.random.default_rng(seed)
    nblocks = (S + block_size - 1) // block_size + 2
    k = torch.tensor(rng.standard_normal((nblocks, block_size, Hkv, D)), dtype=torch.float32)
    v = torch.tensor(rng.standard_normal((nblocks, block_size, Hkv, D)), dtype=torch.float32)
    table = list(rng.permutation(nblocks)[: (S + block_size - 1) // block_size])
    return k, v, [int(b) for b in table]


def gather(cache, table, S, bs):
    out = []
    for i in range(S):
        out.append(cache[table[i // bs], i % bs].numpy().astype(np.float64))
    return np.stack(out)  # [S, Hkv, D]


def brute_attention(qn, kn, vn, ctx, scale):
    """Explicit loops: query i (global pos ctx+i) attends kv 0..ctx+i."""
    Lq, Hq, D = qn.shape
    S = kn.shape[0]
    Hkv = kn.shape[1]
    rep = Hq // Hkv
    out = np.zeros((Lq, Hq, D))
    for i in range(Lq):
        limit = ctx + i + 1
        for h in range(Hq):
            kv_h = h // rep
            scores = np.array(
                [np.dot(qn[i, h], kn[s, kv_h]) * scale for s in range(limit)]
            )
            scores -= scores.max()
            w = np.exp(scores)
            w /= w.sum()
            for s in range(limit):
                out[i, h] += w[s] * vn[s, kv_h]
    return out


def test_prefill_oracle_vs_brute_force():
    S, ctx, Hq, Hkv, D, bs = 23, 9, 4, 2, 16, 4
    Lq = S - ctx
    torch.manual_seed(0)
    q = torch.randn(Lq, Hq, D)
    k_cache, v_cache, table = paged_cache(S, Hkv, D, bs, 1)
    scale = 1.0 / math.sqrt(D)
    got = ref.attention_prefill(q, k_cache, v_cache, table, S, ctx, scale).numpy()
    kn = gather(k_cache, table, S, bs)
    vn = gather(v_cache, table, S, bs)
    want = brute_attention(q.numpy().astype(np.float64), kn, vn, ctx, scale)
    assert np.abs(got - want).max() < 1e-4


def test_decode_oracle_vs_brute_force():
    Hq, Hkv, D, bs = 4, 2, 16, 4
    lens = [7, 13, 22]
    B = len(lens)
    torch.manual_seed(2)
    q = torch.randn(B, Hq, D)
    caches = [paged_cache(s, Hkv, D, bs, 10 + i) for i, (s) in enumerate(lens)]
    # merge into one cache pool with disjoint tables
    nb = sum(c[0].shape[0] for c in caches)
    k_cache = torch.zeros(nb, bs, Hkv, D)
    v_cache = torch.zeros(nb, bs, Hkv, D)
    tables, off = [], 0
    for (kc, vc, t) in caches:
        k_cache[off : off + kc.shape[0]] = kc
        v_cache[off : off + vc.shape[0]] = vc
        tables.append([b + off for b in t])
        off += kc.shape[0]
    maxb = max(len(t) for t in tables)
    bt = torch.tensor([t + [0] * (maxb - len(t)) for t in tables], dtype=torch.long)
    sl = torch.tensor(lens, dtype=torch.long)
    scale = 1.0 / math.sqrt(D)
    got = ref.attention_decode_raw(q, k_cache, v_cache, bt, sl, scale).numpy()
    for i, s in enumerate(lens):
        kn = gather(k_cache, tables[i], s, bs)
        vn = gather(v_cache, tables[i], s, bs)
        want = brute_attention(
            q[i : i + 1].numpy().astype(np.float64), kn, vn, s - 1, scale
        )[0]
        assert np.abs(got[i] - want).max() < 1e-4, i


def test_rmsnorm_and_swiglu_oracles_vs_brute_force():
    torch.manual_seed(3)
    x = torch.randn(5, 32)
    w = torch.randn(32)
    got = ref.rmsnorm(x, w, eps=1e-5).numpy()
    xn = x.numpy().astype(np.float64)
    want = xn / np.sqrt((xn ** 2).mean(-1, keepdims=True) + 1e-5) * w.numpy()
    assert np.abs(got - want).max() < 1e-4

    gu = torch.randn(5, 64)
    got = ref.swiglu(gu).numpy()
    g = gu[:, :32].numpy().astype(np.float64)
    u = gu[:, 32:].numpy().astype(np.float64)
    want = (g / (1 + np.exp(-g))) * u
    assert np.abs(got - want).max() < 1e-4


def test_rope_oracle_vs_brute_force():
    """Rotate-half RoPE: pair (x[i], x[i+D/2]) rotates by pos·θ^(-2i/D)."""
    N, H, D, P = 6, 2, 16, 40
    theta = 10000.0
    torch.manual_seed(4)
    q = torch.randn(N, H, D)
    k = torch.randn(N, 1, D)
    v = torch.randn(N, 1, D)
    q0, k0 = q.numpy().astype(np.float64).copy(), k.numpy().astype(np.float64).copy()
    pos = torch.tensor([0, 1, 5, 17, 2, 39], dtype=torch.long)
    nb = 8
    kc = torch.zeros(nb, 4, 1, D)
    vc = torch.zeros(nb, 4, 1, D)
    slots = torch.tensor([3, 9, 12, 20, 25, 31], dtype=torch.long)
    cs = ref.build_cos_sin(P, D, theta, "cpu")
    ref.rope_and_cache(q, k, v, pos, slots, kc, vc, cs)
    for n in range(N):
        p = int(pos[n])
        for h in range(H):
            for i in range(D // 2):
                ang = p / theta ** (2 * i / D)
                c, s = math.cos(ang), math.sin(ang)
                want_lo = q0[n, h, i] * c - q0[n, h, i + D // 2] * s
                want_hi = q0[n, h, i + D // 2] * c + q0[n, h, i] * s
                assert abs(float(q[n, h, i]) - want_lo) < 1e-4, (n, h, i)
                assert abs(float(q[n, h, i + D // 2]) - want_hi) < 1e-4
    # K landed rope'd at the right flat slots; V raw
    flat_k = kc.view(-1, 1, D)
    flat_v = vc.view(-1, 1, D)
    for n in range(N):
        p = int(pos[n])
        ang = p / theta ** 0.0  # i = 0 pair
        c, s = math.cos(ang), math.sin(ang)
        want = k0[n, 0, 0] * c - k0[n, 0, D // 2] * s
        assert abs(float(flat_k[slots[n], 0, 0]) - want) < 1e-4
        assert torch.allclose(flat_v[slots[n]], v[n])
