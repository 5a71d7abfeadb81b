"""Speculative step overlap (async scheduling): greedy outputs must be
identical with and without the pipeline; finish/rollback edges covered."""
import threading

import pytest

from agentcontrolplane_amd.engine.config import EngineConfig
from agentcontrolplane_amd.engine.engine import InferenceEngine
from agentcontrolplane_amd.engine.request import SamplingParams


def make_engine(async_sched: bool, **kw):
    cfg = dict(
        model="tiny", device="cpu", num_kv_blocks=2048, kv_block_size=16,
        max_prefill_tokens=256, request_timeout_s=180,
        async_scheduling=async_sched,
    )
    cfg.update(kw)
    return InferenceEngine(EngineConfig(**cfg))


def _run_batch(eng, n=12, max_tokens=17, prompt_len=40):
    results = [None] * n
    def go(i):
        prompt = [(i * 31 + j) % 250 for j in range(prompt_len + i)]
        results[i] = eng.generate(prompt, SamplingParams(max_tokens=max_tokens, temperature=0))
    ts = [threading.Thread(target=go, args=(i,)) for i in range(n)]
    for t in ts: t.start()
    for t in ts: t.join()
    return [r.output_ids for r in results]


def test_greedy_equivalence_with_speculation():
    e1 = make_engine(False)
    try:
        want = _run_batch(e1)
    finally:
        e1.stop()
    e2 = make_engine(True)
    try:
        got = _run_batch(e2)
        m = e2.metrics()
        assert m.get("spec_steps", 0) > 0, "speculation never engaged"
    finally:
        e2.stop()
    assert want == got


def test_speculation_with_eot_finishers():
    """Sequences that stop by EOT mid-stream (max_tokens staggered so
    rollback rows occur) — equivalence still holds."""
    e1 = make_engine(False)
    try:
        want = _run_batch(e1, n=10, max_tokens=9)
    finally:
        e1.stop()
    e2 = make_engine(True)
    try:
        got = _run_batch(e2, n=10, max_tokens=9)
    finally:
        e2.stop()
    assert want == got


def test_speculation_with_constrained_decodes():
    """Grammar-constrained decodes stay in the pipeline (the sampler launch
    of a grammar-carrying speculative step is deferred until the
    predecessor's commit advances the PDA states); results stay valid and
    speculative steps actually happen."""
    import json

    eng = make_engine(True)
    tools = [{"type": "function", "function": {
        "name": "t__x",
        "parameters": {"type": "object", "properties": {"m": {"type": "string"}},
                       "required": ["m"]}}}]
    try:
        res = eng.chat(
            [{"role": "user", "content": "call the tool"}], tools=tools,
            sampling=SamplingParams(max_tokens=48, temperature=0.8, tool_choice="required"),
        )
        assert res.finish_reason == "tool_calls"
        json.loads(res.tool_calls[0]["function"]["arguments"])
        assert eng.metrics().get("spec_steps", 0) > 0
    finally:
        eng.stop()


def test_speculation_under_kv_pressure_bails_cleanly():
    e = make_engine(True, num_kv_blocks=56)
    try:
        outs = _run_batch(e, n=8, max_tokens=12, prompt_len=60)
        assert all(len(o) > 0 for o in outs)
    finally:
        e.stop()


def test_pop_last_token_units():
    from agentcontrolplane_amd.engine.kv import PyBlockManager

    bm = PyBlockManager(num_blocks=8, block_size=4)
    bm.add_seq(1)
    bm.append_tokens(1, 4)
    bm.append_tokens(1, 1)  # starts block 2
    assert len(bm.block_table(1)) == 2
    bm.pop_last_token(1)
    assert bm.seq_len(1) == 4 and len(bm.block_table(1)) == 1
    bm.pop_last_token(1)
    assert bm.seq_len(1) == 3 and len(bm.block_table(1)) == 1


def test_stress_mixed_grammar_pressure_equivalence():
    """Randomized end-to-end stress: mixed free-run + grammar-constrained
    greedy requests with ragged prompts under a tiny KV pool (preemption +
    speculation bail-outs interleave).  Outputs must be identical with the
    pipeline on and off."""
    import json
    import random

    rng = random.Random(1234)
    tools = [{"type": "function",
              "function": {"name": "srv__go", "description": "", "parameters": {}}}]
    jobs = []
    for i in range(24):
        jobs.append({
            "prompt_len": rng.randrange(8, 120),
            "max_tokens": rng.randrange(4, 40),
            "tool": rng.random() < 0.3,
            "seed": i,
        })

    def run(async_sched):
        eng = make_engine(async_sched, num_kv_blocks=96, kv_block_size=8,
                          max_prefill_tokens=128)
        try:
            results = [None] * len(jobs)

            def go(i, j):
                if j["tool"]:
                    # call ids come from a global counter → compare content only
                    results[i] = [
                        (tc["function"]["name"], tc["function"]["arguments"])
                        for tc in eng.chat(
                            [{"role": "user", "content": "x" * j["prompt_len"]}],
                            tools=tools,
                            sampling=SamplingParams(
                                max_tokens=64, temperature=0, tool_choice="required"),
                        ).tool_calls
                    ]
                else:
                    prompt = [(i * 17 + k) % 250 for k in range(j["prompt_len"])]
                    results[i] = eng.generate(
                        prompt, SamplingParams(max_tokens=j["max_tokens"], temperature=0)
                    ).output_ids

            ts = [threading.Thread(target=go, args=(i, j)) for i, j in enumerate(jobs)]
            for t in ts:
                t.start()
            for t in ts:
                t.join()
            return results
        finally:
            eng.stop()

    a, b = run(False), run(True)
    for i, (x, y) in enumerate(zip(a, b)):
        assert x == y, (i, jobs[i], x, y)
    # tool-call results must be valid constrained JSON
    for i, j in enumerate(jobs):
        if j["tool"]:
            assert a[i] and a[i][0][0] == "srv__go"
            json.loads(a[i][0][1])


def test_seeded_and_penalized_equivalence():
    """Seeded and penalty-carrying requests are in the deferred-sampler
    class: outputs identical with the pipeline on and off."""
    def run(async_sched):
        eng = make_engine(async_sched)
        try:
            outs = []
            for i in range(6):
                prompt = [(i * 13 + k) % 250 for k in range(30 + i)]
                outs.append(eng.generate(
                    prompt,
                    SamplingParams(max_tokens=14, temperature=1.0, seed=100 + i,
                                   frequency_penalty=0.5, presence_penalty=0.2),
                ).output_ids)
            return outs
        finally:
            eng.stop()

    assert run(False) == run(True)


def test_long_soak_staggered_mixed_arrivals():
    """Staggered (non-wave) arrivals of free/tool/seeded/abandoned-stream
    requests under a tiny pool: no errors, no KV leak, and the pipeline
    keeps engaging (spec_steps > 0) despite the stateful-sampling mix."""
    import json
    import random
    import time as _t

    eng = make_engine(True, num_kv_blocks=96, kv_block_size=8,
                      max_prefill_tokens=96)
    params = {"type": "object", "properties": {"m": {"type": "string"}},
              "required": ["m"]}
    tools = [{"type": "function", "function": {"name": "s__t", "parameters": params}}]
    errors = []
    lock = threading.Lock()
    rng = random.Random(424242)

    def job(i, kind):
        try:
            if kind == "tool":
                r = eng.chat(
                    [{"role": "user", "content": "x" * rng.randrange(5, 60)}],
                    tools=tools,
                    sampling=SamplingParams(max_tokens=48, temperature=0.8,
                                            tool_choice="required"),
                )
                json.loads(r.tool_calls[0]["function"]["arguments"])
            elif kind == "stream":
                for n, _ in enumerate(eng.chat_stream(
                    [{"role": "user", "content": "s" * 20}],
                    sampling=SamplingParams(max_tokens=24, temperature=1.0),
                )):
                    if n > 4:
                        break  # abandon → cancellation path
            else:
                eng.generate(
                    [rng.randrange(250) for _ in range(rng.randrange(5, 80))],
                    SamplingParams(max_tokens=rng.randrange(2, 20),
                                   temperature=1.0, seed=i,
                                   frequency_penalty=0.3),
                )
        except Exception as e:  # noqa: BLE001
            with lock:
                errors.append((i, kind, repr(e)))

    threads = []
    for i in range(60):
        kind = rng.choice(["tool", "stream", "free", "free"])
        t = threading.Thread(target=job, args=(i, kind))
        t.start()
        threads.append(t)
        _t.sleep(rng.random() * 0.01)
    for t in threads:
        t.join()
    assert not errors, errors[:4]
    _t.sleep(0.5)  # let stream cancellations drain
    while eng.scheduler.retired:
        assert eng.scheduler._evict_one_retired()
    assert not eng.scheduler.running and not eng.scheduler.waiting
    assert eng.bm.used_blocks == 0
    assert eng.metrics().get("spec_steps", 0) > 0
    eng.stop()
