"""GPU numerics: every CDNA4 kernel against its plain-PyTorch fp32 reference
(SURVEY.md §4 implication: kernel-level golden tests are a new test tier).

All tests are @gpu — run via gpurun on an MI355X.
"""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from agentcontrolplane_amd.ops import reference as ref

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")


def hip():
    from agentcontrolplane_amd.ops import hip as m

    return m


def maxerr(a, b):
    return (a.float() - b.float()).abs().max().item()


@requires_gpu
@pytest.mark.parametrize("n,h", [(1, 4096), (17, 4096), (256, 8192), (33, 256)])
def test_rmsnorm(n, h):
    x = torch.randn(n, h, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(h, dtype=torch.bfloat16, device="cuda").abs() + 0.5
    out = hip().rmsnorm(x, w, 1e-5)
    want = ref.rmsnorm(x.cpu(), w.cpu(), 1e-5)
    assert maxerr(out.cpu(), want) < 0.05


@requires_gpu
def test_fused_add_rmsnorm():
    n, h = 64, 4096
    x = torch.randn(n, h, dtype=torch.bfloat16, device="cuda")
    r = torch.randn(n, h, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(h, dtype=torch.bfloat16, device="cuda").abs() + 0.5
    xc, rc = x.cpu().clone(), r.cpu().clone()
    out, new_r = hip().fused_add_rmsnorm(x, r, w, 1e-5)
    want_out, want_r = ref.fused_add_rmsnorm(xc, rc, w.cpu(), 1e-5)
    assert maxerr(new_r.cpu(), want_r) < 0.05
    assert maxerr(out.cpu(), want_out) < 0.05


@requires_gpu
@pytest.mark.parametrize("n,inter", [(4, 14336), (129, 512)])
def test_swiglu(n, inter):
    gu = torch.randn(n, 2 * inter, dtype=torch.bfloat16, device="cuda")
    out = hip().swiglu(gu)
    want = ref.swiglu(gu.cpu())
    assert maxerr(out.cpu(), want) < 0.05


def _make_cache(num_blocks, bs, hkv, d, device, dtype=torch.bfloat16, seed=0):
    g = torch.Generator().manual_seed(seed)
    k = torch.randn(num_blocks, bs, hkv, d, generator=g, dtype=torch.float32).to(dtype)
    v = torch.randn(num_blocks, bs, hkv, d, generator=g, dtype=torch.float32).to(dtype)
    return k.to(device), v.to(device)


@requires_gpu
def test_rope_cache():
    N, Hq, Hkv, D, P = 37, 4, 2, 128, 512
    torch.manual_seed(0)
    q = torch.randn(N, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(N, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(N, Hkv, D, dtype=torch.bfloat16, device="cuda")
    kc, vc = _make_cache(16, 16, Hkv, D, "cuda")
    kc.zero_(); vc.zero_()
    pos = torch.randint(0, P, (N,), device="cuda", dtype=torch.long)
    slots = torch.randperm(16 * 16, device="cuda")[:N].long()
    cos_sin = ref.build_cos_sin(P, D, 500000.0, "cuda")
    qc, kcpu, vcpu = q.cpu().clone(), k.cpu().clone(), v.cpu().clone()
    kc_cpu = torch.zeros(16, 16, Hkv, D, dtype=torch.bfloat16)
    vc_cpu = torch.zeros_like(kc_cpu)
    q2, k2 = hip().rope_and_cache(q, k, v, pos, slots, kc, vc, cos_sin)
    ref.rope_and_cache(
        qc, kcpu, vcpu, pos.cpu(), slots.cpu(), kc_cpu, vc_cpu, cos_sin.cpu()
    )
    assert maxerr(q2.cpu(), qc) < 0.03
    assert maxerr(k2.cpu(), kcpu) < 0.03
    assert maxerr(kc.cpu(), kc_cpu) < 0.03
    assert maxerr(vc.cpu(), vc_cpu) < 0.03


@requires_gpu
@pytest.mark.parametrize("G", [1, 2, 4, 8])
def test_decode_attn(G):
    torch.manual_seed(1)
    B, Hkv, D, bs = 5, 2, 128, 16
    Hq = Hkv * G
    num_blocks = 64
    kc, vc = _make_cache(num_blocks, bs, Hkv, D, "cuda")
    seq_lens = torch.tensor([1, 17, 63, 200, 333], dtype=torch.long)
    max_blocks = int((seq_lens.max() + bs - 1) // bs)
    # random non-overlapping block tables
    perm = torch.randperm(num_blocks)
    tables = torch.zeros(B, max_blocks, dtype=torch.long)
    off = 0
    for i in range(B):
        nb = int((seq_lens[i] + bs - 1) // bs)
        tables[i, :nb] = perm[off : off + nb]
        off += nb
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    scale = 1.0 / math.sqrt(D)
    out = hip().attention_decode_raw(
        q, kc, vc, tables.to("cuda"), seq_lens.to("cuda"), scale
    )
    want = ref.attention_decode_raw(
        q.cpu().float(), kc.cpu().float(), vc.cpu().float(), tables, seq_lens, scale
    )
    assert maxerr(out.cpu(), want) < 0.03


@requires_gpu
def test_mfma_fragment_layout_probe():
    """The assumed A/B/D lane maps for v_mfma_f32_32x32x16_bf16 must
    reproduce a plain matmul (asymmetric operands per guide G9/rule 16)."""
    import torch

    from agentcontrolplane_amd import _C

    torch.manual_seed(7)
    a = (torch.randn(32, 16) * 0.5).bfloat16().cuda()
    b = (torch.randn(16, 32) * 0.5).bfloat16().cuda()
    d = _C.mfma_probe(a, b)
    want = a.cpu().float() @ b.cpu().float()
    err = (d.cpu() - want).abs().max().item()
    assert err < 0.05, f"fragment layout mismatch: max err {err}"


@requires_gpu
@pytest.mark.parametrize("impl", ["v0", "mfma"])
def test_prefill_attn_multi_seq(impl, monkeypatch):
    """Three chunks with different ctx offsets in one batched launch."""
    from agentcontrolplane_amd.engine.batch import FlatBatch, SeqMeta
    from agentcontrolplane_amd.ops import hip as hip_mod

    monkeypatch.setattr(hip_mod, "_PREFILL_IMPL", impl)
    torch.manual_seed(2)
    Hq, Hkv, D, bs = 4, 2, 128, 16
    num_blocks = 128
    kc, vc = _make_cache(num_blocks, bs, Hkv, D, "cuda")
    # (query_len, ctx_len)
    shapes = [(70, 0), (1, 129), (65, 63)]
    perm = torch.randperm(num_blocks)
    metas = []
    off = 0
    rows = 0
    for qlen, ctx in shapes:
        S = qlen + ctx
        nb = (S + bs - 1) // bs
        metas.append(
            SeqMeta(
                seq_id=0, query_len=qlen, seq_len=S, ctx_len=ctx,
                block_table=perm[off : off + nb].tolist(), needs_logits=True,
            )
        )
        off += nb
        rows += qlen
    dummy = torch.empty(0)
    batch = FlatBatch(
        token_ids=torch.empty(rows, device="cuda"), positions=dummy,
        slot_mapping=dummy, prefills=metas, num_prefill_tokens=rows,
        decode_seq_ids=[], decode_block_tables=None, decode_seq_lens=None,
        logit_rows=dummy, sample_seq_ids=[],
    )
    q = torch.randn(rows, Hq, D, dtype=torch.bfloat16, device="cuda")
    scale = 1.0 / math.sqrt(D)
    out = hip().attention_prefill_batch(q, kc, vc, batch, scale)
    # oracle per sequence on fp32 CPU
    row = 0
    for m in metas:
        want = ref.attention_prefill(
            q[row : row + m.query_len].cpu().float(), kc.cpu().float(),
            vc.cpu().float(), m.block_table, m.seq_len, m.ctx_len, scale,
        )
        err = maxerr(out[row : row + m.query_len].cpu(), want)
        assert err < 0.03, f"chunk {m}: err {err}"
        row += m.query_len


@requires_gpu
def test_sample_greedy_and_filters():
    torch.manual_seed(3)
    B, V = 64, 261
    logits = torch.randn(B, V, device="cuda")
    gen = torch.Generator(device="cuda").manual_seed(0)
    zeros = torch.zeros(B, device="cuda")
    ones = torch.ones(B, device="cuda")
    offk = torch.zeros(B, dtype=torch.long, device="cuda")
    # greedy == argmax
    out = hip().softmax_sample(logits, zeros, offk, ones, gen)
    assert torch.equal(out, logits.argmax(-1))
    # top-k: sampled ids within the row's top-k set
    k = 5
    out = hip().softmax_sample(logits, ones, torch.full_like(offk, k), ones, gen)
    topk = logits.topk(k, dim=-1).indices
    for i in range(B):
        assert out[i] in topk[i]
    # top-p → 0 degenerates to argmax
    out = hip().softmax_sample(logits, ones, offk, torch.full_like(ones, 1e-6), gen)
    assert torch.equal(out, logits.argmax(-1))
    # grammar mask respected
    mask = torch.zeros(B, V, dtype=torch.bool, device="cuda")
    mask[:, 7] = True
    mask[:, 100] = True
    out = hip().softmax_sample(logits, ones, offk, ones, gen, mask)
    assert all(int(t) in (7, 100) for t in out.tolist())


@requires_gpu
def test_model_forward_gpu_vs_cpu_reference():
    """Whole tiny model: GPU kernels vs CPU fp32 reference on identical
    weights — the kernels compose correctly."""
    from agentcontrolplane_amd.engine.batch import FlatBatch, SeqMeta
    from agentcontrolplane_amd.engine.config import PRESETS, EngineConfig
    from agentcontrolplane_amd.models.llama import LlamaForCausalLM

    ecfg = EngineConfig(model="tiny-gpu", num_kv_blocks=64, kv_block_size=16)
    mcfg = PRESETS["tiny-gpu"]
    gpu_model = LlamaForCausalLM(mcfg, ecfg, "cuda")
    gpu_model.random_init(0)
    gpu_model.allocate_kv_cache(64, 16)

    import dataclasses

    cpu_cfg = dataclasses.replace(mcfg, dtype="float32")
    cpu_model = LlamaForCausalLM(cpu_cfg, ecfg, "cpu")
    cpu_model.random_init(0)
    cpu_model.allocate_kv_cache(64, 16)
    # copy weights (fp32 promotion)
    cpu_model.embed = gpu_model.embed.cpu().float()
    cpu_model.lm_head = gpu_model.lm_head.cpu().float()
    cpu_model.final_norm = gpu_model.final_norm.cpu().float()
    for cl, gl in zip(cpu_model.layers, gpu_model.layers):
        for f in ("input_norm", "qkv", "o", "post_norm", "gate_up", "down"):
            setattr(cl, f, getattr(gl, f).cpu().float())

    T = 33
    tokens = torch.randint(0, mcfg.vocab_size, (T,))
    meta = SeqMeta(
        seq_id=1, query_len=T, seq_len=T, ctx_len=0,
        block_table=list(range(4)), needs_logits=True,
    )

    def make_batch(device):
        return FlatBatch(
            token_ids=tokens.to(device), positions=torch.arange(T, device=device),
            slot_mapping=torch.arange(T, device=device), prefills=[meta],
            num_prefill_tokens=T, decode_seq_ids=[], decode_block_tables=None,
            decode_seq_lens=None,
            logit_rows=torch.tensor([T - 1], device=device), sample_seq_ids=[1],
        )

    glog = gpu_model.forward(make_batch("cuda")).cpu()
    clog = cpu_model.forward(make_batch("cpu"))
    scale = clog.abs().max().item()
    err = (glog - clog).abs().max().item()
    assert err / max(scale, 1e-6) < 0.08, f"rel err {err / scale} (scale {scale})"


@requires_gpu
@pytest.mark.parametrize(
    "m,n,k",
    [
        (256, 256, 128),      # single tile, one 8-phase iteration
        (300, 512, 256),      # ragged M (row-clamp + store guard)
        (1024, 1536, 4096),   # multi-tile, deep K
        (2048, 768, 1664),    # K % 128 == 0 but K % 256 != 0; odd tile grid
    ],
)
def test_gemm_bf16(m, n, k):
    """Hand-written 256²-tile 8-phase MFMA GEMM vs fp32 matmul oracle."""
    torch.manual_seed(11)
    x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    got = hip().gemm_bf16(x, w)
    want = x.float() @ w.float().t()
    # bf16 inputs, fp32 accumulation: error scales with sqrt(K)
    tol = 0.03 * math.sqrt(k)  # bf16 output rounding of ~4.5σ results
    assert maxerr(got, want) < tol, maxerr(got, want)


@pytest.mark.gpu
def test_sample_fullvocab_vs_oracle():
    """csrc/sampling_fullvocab.hip vs the fp32 reference over a Llama-3
    sized vocabulary: greedy exact, draws match the oracle's inverse CDF
    for given uniforms, top-k/top-p respect the oracle's kept sets, and
    compact masks (mask_map indirection) are honored exactly."""
    from agentcontrolplane_amd.ops import reference

    torch.manual_seed(0)
    device = "cuda"
    B, V = 64, 128256
    gen = torch.Generator(device=device)
    logits = torch.randn(B, V, device=device, dtype=torch.float32) * 3
    uniforms = torch.rand(B, device=device)
    zeros = torch.zeros(B, device=device)
    ones = torch.ones(B, device=device)
    offk = torch.zeros(B, dtype=torch.long, device=device)

    # greedy: exact argmax match
    out = hip().softmax_sample(logits, zeros, offk, ones, gen, uniforms=uniforms)
    assert torch.equal(out, logits.argmax(-1))

    # temperature-only fp32: draws match the oracle inverse CDF
    ref = reference.softmax_sample(
        logits.cpu(), ones.cpu() * 0.8, offk.cpu(), ones.cpu(), None,
        uniforms=uniforms.cpu())
    out = hip().softmax_sample(logits, ones * 0.8, offk, ones, gen,
                               uniforms=uniforms)
    frac = (out.cpu() == ref).float().mean().item()
    assert frac >= 0.95, f"only {frac:.2%} of draws matched the oracle"

    # top-k: every draw inside the oracle's kept set {x >= kth}
    k = 50
    out = hip().softmax_sample(logits, ones, torch.full_like(offk, k), ones,
                               gen, uniforms=uniforms)
    kth = torch.topk(logits, k, dim=-1).values[:, -1]
    picked = logits[torch.arange(B, device=device), out]
    assert (picked >= kth - 1e-6).all()

    # top-p: draws inside the oracle nucleus
    p = 0.7
    out = hip().softmax_sample(logits, ones, offk, ones * p, gen,
                               uniforms=uniforms)
    probs = torch.softmax(logits, -1)
    sp, si = torch.sort(probs, descending=True, dim=-1)
    cum = sp.cumsum(-1)
    for i in range(B):
        keep_n = int((cum[i] - sp[i] <= p).sum())
        kept = set(si[i, :max(1, keep_n)].tolist())
        assert int(out[i]) in kept

    # bf16 logits path + compact mask rows with indirection
    lb = logits.bfloat16()
    n_masked = 8
    mask = torch.zeros(n_masked, V, dtype=torch.bool, device=device)
    allowed_ids = torch.randint(0, V, (n_masked, 37), device=device)
    for j in range(n_masked):
        mask[j, allowed_ids[j]] = True
    mask_map = torch.full((B,), -1, dtype=torch.int32, device=device)
    mask_map[: n_masked] = torch.arange(n_masked, dtype=torch.int32, device=device)
    out = hip().softmax_sample(lb, ones, offk, ones, gen, mask,
                               uniforms=uniforms, mask_map=mask_map)
    for j in range(n_masked):
        assert bool(mask[j, out[j]]), "masked row sampled a disallowed token"
    # unmasked rows unaffected by the compact mask
    out_ref = hip().softmax_sample(lb, ones, offk, ones, gen,
                                   uniforms=uniforms)
    assert torch.equal(out[n_masked:], out_ref[n_masked:])


@pytest.mark.gpu
def test_tp_shard_path_math_on_gpu():
    """TP assembly path on one GPU (driver-checkable without multi-GPU):
    a world=1 'shard' built through shard_from_full + the all_reduce hook
    must equal the plain model exactly, and world=2 shards must run the
    CDNA4 kernels at halved head counts without shape faults (the gloo
    CPU test proves the world=2 math)."""
    import dataclasses

    from agentcontrolplane_amd.engine.batch import FlatBatch, SeqMeta
    from agentcontrolplane_amd.engine.config import PRESETS, EngineConfig
    from agentcontrolplane_amd.models.llama import LlamaForCausalLM
    from agentcontrolplane_amd.parallel.tp import shard_from_full

    cfg = dataclasses.replace(PRESETS["tiny-gpu"])
    ecfg = EngineConfig(model="tiny-gpu", device="cuda", num_kv_blocks=64)
    torch.manual_seed(0)
    full = LlamaForCausalLM(cfg, ecfg, "cuda")
    full.random_init(0)
    full.allocate_kv_cache(64, 16)

    T = 24
    g = torch.Generator().manual_seed(42)
    tokens = torch.randint(0, cfg.vocab_size, (T,), generator=g).cuda()
    meta = SeqMeta(seq_id=1, query_len=T, seq_len=T, ctx_len=0,
                   block_table=[0, 1], needs_logits=True)

    def mk():
        return FlatBatch(
            token_ids=tokens.clone(), positions=torch.arange(T, device="cuda"),
            slot_mapping=torch.arange(T, device="cuda"), prefills=[meta],
            num_prefill_tokens=T, decode_seq_ids=[],
            decode_block_tables=None, decode_seq_lens=None,
            logit_rows=torch.tensor([T - 1], device="cuda"), sample_seq_ids=[1],
        )

    ref_logits = full.forward(mk()).float()

    shard1 = LlamaForCausalLM(cfg, ecfg, "cuda", tp_rank=0, tp_world=1)
    shard_from_full(full, shard1, 0, 1)
    shard1.allocate_kv_cache(64, 16)
    shard1.all_reduce = lambda t: t  # world=1 collective = identity
    out1 = shard1.forward(mk()).float()
    assert torch.equal(out1, ref_logits), "world=1 shard path diverged"

    # world=2 shards: halved Hq/Hkv shapes through the HIP kernels
    for rank in (0, 1):
        sh = LlamaForCausalLM(cfg, ecfg, "cuda", tp_rank=rank, tp_world=2)
        shard_from_full(full, sh, rank, 2)
        sh.allocate_kv_cache(64, 16)
        sh.all_reduce = lambda t: t  # partial sums: shape/sanity only
        out = sh.forward(mk()).float()
        assert out.shape == ref_logits.shape
        assert torch.isfinite(out).all()


@pytest.mark.gpu
def test_attention_bias_serving_on_gpu():
    """Qwen2-style QKV bias through the CDNA4 kernel path: the bias add
    precedes the strided q/k/v views, so the fused-QKV stride handling
    must hold on the result tensor."""
    from agentcontrolplane_amd.engine.config import EngineConfig
    from agentcontrolplane_amd.engine.engine import InferenceEngine
    from agentcontrolplane_amd.engine.request import SamplingParams

    eng = InferenceEngine(
        EngineConfig(model="tiny-gpu-bias", device="cuda", num_kv_blocks=128),
        start=True,
    )
    try:
        assert eng.model.layers[0].qkv_bias is not None
        res = eng.chat(
            [{"role": "user", "content": "bias check"}],
            sampling=SamplingParams(max_tokens=8, temperature=0.8),
        )
        assert res.completion_tokens <= 8
    finally:
        eng.stop()
