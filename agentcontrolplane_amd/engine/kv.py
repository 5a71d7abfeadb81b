"""Paged KV-cache block management.

The engine's KV cache is paged vLLM-style: per layer a pool of fixed-size
blocks (``block_size`` tokens each); every sequence holds an ordered block
table.  The pool is sized for the GPU's free HBM (288 GB on MI355X) at
engine start.

The allocator itself is native C++ (csrc/block_manager.cpp, exposed through
the ``_C`` extension) — free-list allocation, block-table bookkeeping and
slot-mapping computation run outside the GIL for thousands of concurrent
sequences.  ``PyBlockManager`` is the pure-python equivalent used on CPU and
as the numerics oracle for the C++ one (tests assert identical behavior).
"""
from __future__ import annotations

from typing import Dict, List


class OutOfBlocksError(RuntimeError):
    pass


class PyBlockManager:
    """Free-list block allocator + per-sequence block tables."""

    def __init__(self, num_blocks: int, block_size: int):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))
        self._tables: Dict[int, List[int]] = {}
        self._lens: Dict[int, int] = {}
        # copy-on-nothing sharing: a block referenced by >1 table is
        # immutable (full) and freed when its count drops to 0
        self._refs: Dict[int, int] = {}
        #: bumps whenever any block table changes shape — callers cache
        #: derived tensors against it
        self.table_epoch = 0

    def _release_block(self, b: int) -> None:
        n = self._refs.get(b, 1) - 1
        if n <= 0:
            self._refs.pop(b, None)
            self._free.append(b)
        else:
            self._refs[b] = n

    # ------------------------------------------------------------ lifecycle

    def add_seq(self, seq_id: int) -> None:
        if seq_id in self._tables:
            raise KeyError(f"seq {seq_id} already exists")
        self._tables[seq_id] = []
        self._lens[seq_id] = 0

    def free_seq(self, seq_id: int) -> None:
        blocks = self._tables.pop(seq_id, None)
        self._lens.pop(seq_id, None)
        if blocks:
            for b in reversed(blocks):
                self._release_block(b)
            self.table_epoch += 1

    def has_seq(self, seq_id: int) -> bool:
        return seq_id in self._tables

    # ----------------------------------------------------------- allocation

    def can_append(self, seq_id: int, n_tokens: int) -> bool:
        need = self._blocks_needed(seq_id, n_tokens)
        return need <= len(self._free)

    def _blocks_needed(self, seq_id: int, n_tokens: int) -> int:
        cur_len = self._lens[seq_id]
        have = len(self._tables[seq_id])
        need_total = (cur_len + n_tokens + self.block_size - 1) // self.block_size
        return max(0, need_total - have)

    def append_tokens(self, seq_id: int, n_tokens: int) -> List[int]:
        """Reserve space for n new tokens; returns the flat slot index for
        each token (block_id * block_size + offset)."""
        need = self._blocks_needed(seq_id, n_tokens)
        if need > len(self._free):
            raise OutOfBlocksError(
                f"seq {seq_id}: need {need} blocks, {len(self._free)} free"
            )
        table = self._tables[seq_id]
        if need:
            self.table_epoch += 1
        for _ in range(need):
            table.append(self._free.pop())
        slots = []
        start = self._lens[seq_id]
        for i in range(n_tokens):
            pos = start + i
            slots.append(table[pos // self.block_size] * self.block_size + pos % self.block_size)
        self._lens[seq_id] = start + n_tokens
        return slots

    def pop_last_token(self, seq_id: int) -> None:
        """Undo the last append_tokens(seq, 1) — the speculative-step
        rollback when the previous step finished the sequence."""
        cur = self._lens[seq_id]
        assert cur > 0
        self._lens[seq_id] = cur - 1
        table = self._tables[seq_id]
        if (cur - 1) % self.block_size == 0 and len(table) * self.block_size >= cur:
            # the popped token was alone in the trailing block
            b = table.pop()
            self._release_block(b)
            self.table_epoch += 1

    def adopt_prefix(self, new_seq_id: int, old_seq_id: int, n_blocks: int,
                     n_tokens: int) -> None:
        """Transfer the first n_blocks of old's table to a NEW sequence (the
        continuation cache: a finished conversation's KV prefix is reused by
        the turn that extends it); the remainder of old is freed."""
        if new_seq_id in self._tables:
            raise KeyError(f"seq {new_seq_id} already exists")
        old = self._tables.pop(old_seq_id, None)
        self._lens.pop(old_seq_id, None)
        if old is None:
            raise KeyError(f"seq {old_seq_id} not found")
        assert n_blocks <= len(old) and n_tokens <= n_blocks * self.block_size
        self._tables[new_seq_id] = old[:n_blocks]
        self._lens[new_seq_id] = n_tokens
        for b in reversed(old[n_blocks:]):
            self._release_block(b)
        self.table_epoch += 1

    def share_prefix(self, new_seq_id: int, donor_seq_id: int, n_blocks: int,
                     n_tokens: int) -> None:
        """Register a NEW sequence whose first n_blocks alias the donor's
        (refcounted; the shared blocks are full and immutable — appends only
        ever touch blocks past n_tokens).  The donor keeps its table."""
        if new_seq_id in self._tables:
            raise KeyError(f"seq {new_seq_id} already exists")
        donor = self._tables.get(donor_seq_id)
        if donor is None:
            raise KeyError(f"seq {donor_seq_id} not found")
        assert n_blocks <= len(donor) and n_tokens == n_blocks * self.block_size
        shared = donor[:n_blocks]
        for b in shared:
            self._refs[b] = self._refs.get(b, 1) + 1
        self._tables[new_seq_id] = list(shared)
        self._lens[new_seq_id] = n_tokens
        self.table_epoch += 1

    def ref_count(self, block: int) -> int:
        return self._refs.get(block, 1)

    # -------------------------------------------------------------- queries

    def block_table(self, seq_id: int) -> List[int]:
        return list(self._tables[seq_id])

    def seq_len(self, seq_id: int) -> int:
        return self._lens[seq_id]

    @property
    def free_blocks(self) -> int:
        return len(self._free)

    @property
    def used_blocks(self) -> int:
        return self.num_blocks - len(self._free)

    def occupancy(self) -> float:
        return self.used_blocks / max(1, self.num_blocks)


def make_block_manager(num_blocks: int, block_size: int, prefer_native: bool = True):
    """Return the C++ block manager when the extension is built, else the
    python one (CPU tests)."""
    if prefer_native:
        try:
            from .. import _C  # built by build_ext.py

            return _C.BlockManager(num_blocks, block_size)
        except ImportError:
            pass
    return PyBlockManager(num_blocks, block_size)
