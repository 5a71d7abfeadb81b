"""hipGraph-captured decode steps.

Decode-only engine steps are host-bound: ~10 ms of Python/launch overhead
per step against ~5 ms of GPU work (profiles/r01).  Each decode step runs
the same 32-layer kernel sequence with only tensor *contents* changing, so
the whole forward is captured once per batch-size bucket as a hipGraph
(torch.cuda.CUDAGraph is hipGraph on ROCm) and replayed with refreshed
inputs — one launch instead of ~400.

Buckets pad the decode batch up to {4, 8, …, max}; padding rows point at a
dedicated scratch KV block (beyond the allocator's range) so their cache
writes land nowhere meaningful, and their logits are never sampled.
"""
from __future__ import annotations

from typing import Dict, Optional

import torch

from .batch import FlatBatch

BUCKETS = [4, 8, 16, 32, 64, 96, 128, 192, 256, 384, 512, 768, 1024]


class DecodeGraphRunner:
    def __init__(self, model, max_batch: int, max_blocks_per_seq: int,
                 scratch_block: int, kv_block_size: int):
        self.model = model
        self.device = model.device
        self.max_blocks = max_blocks_per_seq
        self.scratch_slot = scratch_block * kv_block_size
        self.buckets = [b for b in BUCKETS if b <= max_batch]
        if not self.buckets or self.buckets[-1] < max_batch:
            self.buckets.append(max_batch)
        self._graphs: Dict[int, dict] = {}
        self._pool = None

    def _bucket_for(self, n: int) -> Optional[int]:
        for b in self.buckets:
            if n <= b:
                return b
        return None

    def _build(self, b: int) -> dict:
        dev = self.device
        static = {
            "token_ids": torch.zeros(b, dtype=torch.long, device=dev),
            "positions": torch.zeros(b, dtype=torch.long, device=dev),
            "slot_mapping": torch.full((b,), self.scratch_slot, dtype=torch.long, device=dev),
            "block_tables": torch.zeros(b, self.max_blocks, dtype=torch.int32, device=dev),
            "seq_lens": torch.ones(b, dtype=torch.int32, device=dev),
            "logit_rows": torch.arange(b, dtype=torch.long, device=dev),
        }
        batch = FlatBatch(
            token_ids=static["token_ids"],
            positions=static["positions"],
            slot_mapping=static["slot_mapping"],
            prefills=[],
            num_prefill_tokens=0,
            decode_seq_ids=list(range(b)),
            decode_block_tables=None,
            decode_seq_lens=None,
            logit_rows=static["logit_rows"],
            sample_seq_ids=list(range(b)),
        )
        # the attention op reads the int32 views; install the static ones
        batch._decode_tables_i32 = static["block_tables"]
        batch._decode_lens_i32 = static["seq_lens"]

        # warmup on a side stream (the documented capture recipe)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            self.model.forward(batch)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        graph = torch.cuda.CUDAGraph()
        if self._pool is None:
            with torch.cuda.graph(graph):
                out = self.model.forward(batch)
            self._pool = graph.pool()
        else:
            with torch.cuda.graph(graph, pool=self._pool):
                out = self.model.forward(batch)
        return {"graph": graph, "static": static, "out": out, "b": b}

    def run(self, batch: FlatBatch) -> Optional[torch.Tensor]:
        """Replay the bucketed graph for a decode-only batch; returns the
        logits rows for the real sequences, or None if no bucket fits."""
        n = batch.num_decode
        b = self._bucket_for(n)
        if b is None:
            return None
        if b not in self._graphs:
            self._graphs[b] = self._build(b)
        g = self._graphs[b]
        st = g["static"]
        st["token_ids"][:n].copy_(batch.token_ids, non_blocking=True)
        st["positions"][:n].copy_(batch.positions, non_blocking=True)
        st["slot_mapping"][:n].copy_(batch.slot_mapping, non_blocking=True)
        if n < b:
            st["slot_mapping"][n:].fill_(self.scratch_slot)
            st["seq_lens"][n:].fill_(1)
        tables = batch.decode_tables_i32()
        w = tables.shape[1]
        st["block_tables"][:n, :w].copy_(tables, non_blocking=True)
        st["seq_lens"][:n].copy_(batch.decode_lens_i32(), non_blocking=True)
        g["graph"].replay()
        return g["out"][:n]
