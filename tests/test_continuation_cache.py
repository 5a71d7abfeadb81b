"""Continuation cache: a request extending a finished conversation adopts
its KV prefix; outputs must be IDENTICAL to a cold run (KV is a derived
cache — reuse may never change numerics)."""
import pytest
import torch

from agentcontrolplane_amd.engine.config import EngineConfig
from agentcontrolplane_amd.engine.engine import InferenceEngine
from agentcontrolplane_amd.engine.request import SamplingParams


def make_engine(**kw):
    cfg = dict(
        model="tiny", device="cpu", num_kv_blocks=2048, kv_block_size=16,
        max_prefill_tokens=512, request_timeout_s=120,
    )
    cfg.update(kw)
    return InferenceEngine(EngineConfig(**cfg))


def test_continuation_adopts_and_matches_cold():
    eng = make_engine()
    try:
        prompt1 = list(range(100, 180))  # 80 tokens
        r1 = eng.generate(prompt1, SamplingParams(max_tokens=8, temperature=0))
        # turn 2 extends turn 1's stream (prompt + output + new suffix)
        prompt2 = prompt1 + r1.output_ids + [7, 8, 9, 10, 11]
        r2 = eng.generate(prompt2, SamplingParams(max_tokens=8, temperature=0))
        m = eng.metrics()
        assert m["continuation_hits"] == 1
        assert m["continuation_tokens_saved"] >= 80
    finally:
        eng.stop()
    # cold engine, same turn-2 prompt → identical greedy output
    eng2 = make_engine()
    try:
        r2_cold = eng2.generate(prompt2, SamplingParams(max_tokens=8, temperature=0))
        assert r2.output_ids == r2_cold.output_ids
    finally:
        eng2.stop()


def test_no_adoption_on_mismatched_prefix():
    eng = make_engine()
    try:
        eng.generate(list(range(100, 180)), SamplingParams(max_tokens=4, temperature=0))
        # same first block... but then diverges inside block 2
        p2 = list(range(100, 120)) + [9] * 60
        eng.generate(p2, SamplingParams(max_tokens=4, temperature=0))
        m = eng.metrics()
        # only the first block (16 tokens) may be adopted
        assert m["continuation_tokens_saved"] in (0, 16)
        # whole different first block: no hit
        eng.generate([3] * 80, SamplingParams(max_tokens=4, temperature=0))
    finally:
        eng.stop()


def test_retired_evicted_under_pressure():
    # pool of 64 blocks; conversations of 5 blocks each; the retired pool
    # must be reclaimed rather than blocking new admissions
    eng = make_engine(num_kv_blocks=64)
    try:
        for i in range(20):
            prompt = [i * 7 % 250] * 70 + list(range(i, i + 10))
            r = eng.generate(prompt, SamplingParams(max_tokens=4, temperature=0))
            assert len(r.output_ids) > 0
        m = eng.metrics()
        assert m["retired_seqs"] <= 64 // 5
    finally:
        eng.stop()


def test_block_manager_adopt_prefix_bookkeeping():
    from agentcontrolplane_amd.engine.kv import PyBlockManager

    bm = PyBlockManager(num_blocks=16, block_size=4)
    bm.add_seq(1)
    bm.append_tokens(1, 14)  # 4 blocks
    t1 = bm.block_table(1)
    bm.adopt_prefix(2, 1, 2, 8)
    assert not bm.has_seq(1)
    assert bm.block_table(2) == t1[:2]
    assert bm.seq_len(2) == 8
    assert bm.free_blocks == 16 - 2
    # appends continue from the adopted prefix
    slots = bm.append_tokens(2, 1)
    assert slots[0] // 4 not in t1[:2] or slots[0] // 4 == t1[2]


def test_shared_prefix_multiple_adopters():
    """Sibling requests with the same prefix share the retiree's blocks
    (refcounted) — all outputs identical to cold runs."""
    eng = make_engine()
    try:
        base = list(range(50, 114))  # 64 tokens = 4 full blocks
        r0 = eng.generate(base + [1, 2, 3], SamplingParams(max_tokens=4, temperature=0))
        outs = []
        for suffix in ([4, 5, 6], [7, 8, 9], [10, 11, 12]):
            outs.append(
                eng.generate(base + suffix, SamplingParams(max_tokens=4, temperature=0))
            )
        m = eng.metrics()
        assert m["continuation_hits"] >= 3
    finally:
        eng.stop()
    # each against a cold engine
    for suffix, r in zip(([4, 5, 6], [7, 8, 9], [10, 11, 12]), outs):
        eng2 = make_engine()
        try:
            cold = eng2.generate(base + suffix, SamplingParams(max_tokens=4, temperature=0))
            assert r.output_ids == cold.output_ids, f"suffix {suffix}"
        finally:
            eng2.stop()


def test_share_prefix_refcounts():
    from agentcontrolplane_amd.engine.kv import PyBlockManager

    bm = PyBlockManager(num_blocks=16, block_size=4)
    bm.add_seq(1)
    bm.append_tokens(1, 8)  # 2 blocks
    t1 = bm.block_table(1)
    bm.share_prefix(2, 1, 2, 8)
    bm.share_prefix(3, 1, 2, 8)
    assert bm.block_table(2) == t1 and bm.block_table(3) == t1
    assert bm.free_blocks == 14
    bm.free_seq(1)
    assert bm.free_blocks == 14  # still referenced
    bm.free_seq(2)
    assert bm.free_blocks == 14
    bm.free_seq(3)
    assert bm.free_blocks == 16
    # appends after sharing go to fresh blocks
    bm.add_seq(9)
    bm.append_tokens(9, 8)
    bm.share_prefix(10, 9, 2, 8)
    slots = bm.append_tokens(10, 1)
    assert slots[0] // 4 not in bm.block_table(9)
