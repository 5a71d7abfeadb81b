"""Request router: multi-Task + sub-agent fan-out across GPUs.

SURVEY.md §2.3's "request router" component: for data-parallel serving a
single control-plane process drives one InferenceEngine per GPU (no
collectives — DP needs none) and routes each chat turn to the least-
loaded engine.  Sub-agent delegation (executor.go:176-242 creates child
Tasks) then shards across the node automatically: every child's turns
enter the pool independently.

The pool quacks like one engine (`chat / chat_async / chat_stream /
metrics / stop / cfg / tokenizer`), so the `local` LLM provider
(llmclient/local.py), the REST server's OpenAI endpoint, and the LLM
controller's live probe all work unchanged — `ControlPlane(engine=
EnginePool.build(...))`.

Multi-PROCESS scaling (one rank per GPU over RCCL, `bench.py --gpus N`
under torchrun) remains the benchmarked weak-scaling path; the pool is
the in-process routing alternative for `serve --gpus N` deployments and
the component the reference lacks a counterpart for.
"""
from __future__ import annotations

import threading
from typing import List


class EnginePool:
    """Least-loaded router over per-GPU engines."""

    def __init__(self, engines: List):
        if not engines:
            raise ValueError("EnginePool needs at least one engine")
        self.engines = engines
        self._lock = threading.Lock()
        self._inflight = [0] * len(engines)
        self._routed = [0] * len(engines)

    # ------------------------------------------------------------ building

    @classmethod
    def build(cls, config, n_gpus: int) -> "EnginePool":
        """One engine per visible GPU (device cuda:0..n-1), sharing one
        EngineConfig shape."""
        import dataclasses

        from ..engine.engine import InferenceEngine

        engines = []
        for i in range(n_gpus):
            cfg_i = dataclasses.replace(config, device=f"cuda:{i}")
            engines.append(InferenceEngine(cfg_i))
        return cls(engines)

    # ------------------------------------------------------------- routing

    def _pick(self) -> int:
        with self._lock:
            i = min(range(len(self.engines)), key=lambda j: self._inflight[j])
            self._inflight[i] += 1
            self._routed[i] += 1
            return i

    def _done(self, i: int) -> None:
        with self._lock:
            self._inflight[i] -= 1

    # ------------------------------------------------------- engine facade

    @property
    def cfg(self):
        return self.engines[0].cfg

    @property
    def tokenizer(self):
        return self.engines[0].tokenizer

    def chat(self, messages, tools=None, sampling=None):
        i = self._pick()
        try:
            return self.engines[i].chat(messages, tools, sampling)
        finally:
            self._done(i)

    def chat_async(self, messages, tools, sampling, callback):
        i = self._pick()

        def _cb(result, error):
            self._done(i)
            callback(result, error)

        return self.engines[i].chat_async(messages, tools, sampling, _cb)

    def chat_stream(self, messages, tools=None, sampling=None):
        i = self._pick()
        try:
            yield from self.engines[i].chat_stream(messages, tools, sampling)
        finally:
            self._done(i)

    def generate(self, *a, **kw):
        i = self._pick()
        try:
            return self.engines[i].generate(*a, **kw)
        finally:
            self._done(i)

    def metrics(self) -> dict:
        """Aggregate counters over the pool + per-engine routing stats."""
        agg: dict = {}
        for e in self.engines:
            for k, v in e.metrics().items():
                if isinstance(v, (int, float)):
                    agg[k] = agg.get(k, 0) + v
        n = len(self.engines)
        for k in ("kv_occupancy", "tokens_per_s", "uptime_s"):
            if k in agg:
                agg[k] /= n
        agg["pool_size"] = n
        with self._lock:
            agg["pool_inflight"] = sum(self._inflight)
            for i, r in enumerate(self._routed):
                agg[f"pool_routed_{i}"] = r
        return agg

    def stop(self) -> None:
        for e in self.engines:
            e.stop()

    def start(self) -> None:
        for e in self.engines:
            e.start()
