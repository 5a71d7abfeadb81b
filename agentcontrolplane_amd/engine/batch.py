"""Flat-batch representation for mixed prefill+decode engine steps.

One engine step runs ONE model forward over a flat token batch that mixes
prefill chunks (variable-length, one per admitted sequence) and decode
tokens (one per running sequence) — continuous batching.  Rows are ordered
[prefill chunk 0 | prefill chunk 1 | … | decode tokens].
"""
from __future__ import annotations

import dataclasses
from typing import List, Optional

import torch


@dataclasses.dataclass
class SeqMeta:
    seq_id: int
    query_len: int        # tokens of this sequence in this step
    seq_len: int          # total cached length after this step
    ctx_len: int          # cached length before this step (causal offset)
    block_table: List[int]
    needs_logits: bool    # prompt fully processed after this chunk


@dataclasses.dataclass
class FlatBatch:
    token_ids: torch.Tensor      # [N] long
    positions: torch.Tensor      # [N] long
    slot_mapping: torch.Tensor   # [N] long — flat KV slot per token
    prefills: List[SeqMeta]
    num_prefill_tokens: int
    # decode part (query_len == 1 each), rows [num_prefill_tokens:]
    decode_seq_ids: List[int]
    decode_block_tables: Optional[torch.Tensor]  # [Bd, max_blocks] long
    decode_seq_lens: Optional[torch.Tensor]      # [Bd] long
    # rows of the hidden state that need logits (prefill-final rows + all
    # decode rows), aligned with sample_seq_ids
    logit_rows: torch.Tensor
    sample_seq_ids: List[int]

    # speculative step: decode-row tokens come from the PREVIOUS step's
    # GPU-resident sampled tokens (gathered by these row indices) instead of
    # committed CPU values — the engine fills token_ids[num_prefill:] before
    # the forward launch
    spec_src_rows: Optional[torch.Tensor] = None
    #: set by the engine's sample launch; successors gather their decode
    #: tokens from it
    _tokens_gpu: Optional[torch.Tensor] = None

    # lazily-built device metadata, shared by every layer's kernels
    _prefill_meta: Optional["PrefillMeta"] = None
    #: pinned-staging helper (engine/stage.py); set by the scheduler so the
    #: per-step metadata uploads are async instead of blocking hipMemcpys
    _stager: Optional[object] = None
    #: host-side lists mirroring the device tensors (set by the scheduler)
    #: so batch_to_wire never syncs
    _host: Optional[dict] = None
    _decode_tables_i32: Optional[torch.Tensor] = None
    _decode_lens_i32: Optional[torch.Tensor] = None

    @property
    def num_tokens(self) -> int:
        return int(self.token_ids.shape[0])

    @property
    def num_decode(self) -> int:
        return len(self.decode_seq_ids)

    def prefill_meta(self, tile_q: int = 64) -> "PrefillMeta":
        if self._prefill_meta is None:
            self._prefill_meta = {}
        if tile_q not in self._prefill_meta:
            self._prefill_meta[tile_q] = PrefillMeta.build(
                self.prefills, self.token_ids.device, tile_q, self._stager
            )
        return self._prefill_meta[tile_q]

    def decode_tables_i32(self) -> torch.Tensor:
        if self._decode_tables_i32 is None:
            self._decode_tables_i32 = self.decode_block_tables.int().contiguous()
        return self._decode_tables_i32

    def decode_lens_i32(self) -> torch.Tensor:
        if self._decode_lens_i32 is None:
            self._decode_lens_i32 = self.decode_seq_lens.int().contiguous()
        return self._decode_lens_i32


def batch_to_wire(batch: FlatBatch) -> dict:
    """Compact picklable form for the TP metadata broadcast (rank 0 runs the
    scheduler; other ranks rebuild the batch and run the sharded forward).
    Prefers the scheduler's host-side lists (``_host``) so the broadcast
    never D2H-syncs the just-uploaded device tensors."""
    h = getattr(batch, "_host", None) or {}

    def of(name, tensor):
        v = h.get(name)
        return v if v is not None else (tensor.tolist() if tensor is not None else None)

    return {
        "token_ids": of("token_ids", batch.token_ids),
        "positions": of("positions", batch.positions),
        "slot_mapping": of("slot_mapping", batch.slot_mapping),
        "prefills": [
            (m.seq_id, m.query_len, m.seq_len, m.ctx_len, m.block_table, m.needs_logits)
            for m in batch.prefills
        ],
        "num_prefill_tokens": batch.num_prefill_tokens,
        "decode_seq_ids": batch.decode_seq_ids,
        "decode_block_tables": of("decode_block_tables", batch.decode_block_tables),
        "decode_seq_lens": of("decode_seq_lens", batch.decode_seq_lens),
        "logit_rows": of("logit_rows", batch.logit_rows),
        "sample_seq_ids": batch.sample_seq_ids,
    }


def batch_from_wire(d: dict, device) -> FlatBatch:
    t = lambda x: torch.tensor(x, device=device, dtype=torch.long)  # noqa: E731
    return FlatBatch(
        token_ids=t(d["token_ids"]),
        positions=t(d["positions"]),
        slot_mapping=t(d["slot_mapping"]),
        prefills=[
            SeqMeta(seq_id=s, query_len=q, seq_len=sl, ctx_len=c, block_table=bt,
                    needs_logits=nl)
            for (s, q, sl, c, bt, nl) in d["prefills"]
        ],
        num_prefill_tokens=d["num_prefill_tokens"],
        decode_seq_ids=d["decode_seq_ids"],
        decode_block_tables=(
            t(d["decode_block_tables"]) if d["decode_block_tables"] is not None else None
        ),
        decode_seq_lens=(
            t(d["decode_seq_lens"]) if d["decode_seq_lens"] is not None else None
        ),
        logit_rows=t(d["logit_rows"]),
        sample_seq_ids=d["sample_seq_ids"],
    )


@dataclasses.dataclass
class PrefillMeta:
    """Device tensors for the batched prefill-attention kernel, built once
    per step and reused by all layers."""

    block_tables: torch.Tensor  # [num_seqs, max_blocks] int32
    seq_lens: torch.Tensor      # [num_seqs] int32 (after this chunk)
    ctx_lens: torch.Tensor      # [num_seqs] int32 (before this chunk)
    row_starts: torch.Tensor    # [num_seqs] int32 — first q row per seq
    tile_seq: torch.Tensor      # [num_tiles] int32
    tile_q0: torch.Tensor       # [num_tiles] int32
    tile_q: int

    @staticmethod
    def build(prefills: List[SeqMeta], device, tile_q: int = 64,
              stager=None) -> "PrefillMeta":
        max_blocks = max((len(m.block_table) for m in prefills), default=1)
        tables, seq_lens, ctx_lens, row_starts = [], [], [], []
        tile_seq, tile_q0 = [], []
        row = 0
        for i, m in enumerate(prefills):
            tables.append(m.block_table + [0] * (max_blocks - len(m.block_table)))
            seq_lens.append(m.seq_len)
            ctx_lens.append(m.ctx_len)
            row_starts.append(row)
            for q0 in range(0, m.query_len, tile_q):
                tile_seq.append(i)
                tile_q0.append(q0)
            row += m.query_len
        if stager is not None:
            t = lambda n, x: stager.tensor(f"pm{tile_q}_{n}", x, "int32")  # noqa: E731
        else:
            t = lambda n, x: torch.tensor(x, device=device, dtype=torch.int32)  # noqa: E731
        return PrefillMeta(
            block_tables=t("bt", tables),
            seq_lens=t("sl", seq_lens),
            ctx_lens=t("cl", ctx_lens),
            row_starts=t("rs", row_starts),
            tile_seq=t("ts", tile_seq),
            tile_q0=t("tq", tile_q0),
            tile_q=tile_q,
        )
