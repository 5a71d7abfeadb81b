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

    @property
    def num_tokens(self) -> int:
        return int(self.token_ids.shape[0])

    @property
    def num_decode(self) -> int:
        return len(self.decode_seq_ids)
