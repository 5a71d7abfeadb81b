"""Pinned-memory host→device staging for the per-step batch metadata.

``torch.tensor(list, device="cuda")`` performs a BLOCKING hipMemcpy: with
the speculative pipeline 1-2 steps deep it drains the stream on every call
(measured: 5.2 s of an 11 s engine-busy bench — profiles/r01).  The stager
keeps one pinned staging buffer per (name, rotation slot) and one device
buffer per name, fills the pinned view on the host, and issues a single
``copy_(non_blocking=True)`` — a true async H2D on the current stream.

Rotation depth 3 > pipeline depth 2 guarantees a slot's previous copy has
been consumed (the commit of step t syncs the stream past step t's copies
before the slot is reused at t+3; stream order protects the device side).
"""
from __future__ import annotations

import os
from typing import Dict, Tuple

import numpy as np
import torch

# staging scope, A/B'd on MI355X (default bench @128 tasks / @1k tasks):
#   "numpy" (default) — np.asarray → .to(device) everywhere.  The original
#       torch.tensor(python_list, device="cuda") cost 5.2 s/bench in list
#       parsing + blocking copies; the numpy route alone recovers it
#       (84.7 steps/s @128, 66.7 @1k).
#   "params" — additionally stages cross-step cached tensors (sampling
#       params, decode tables) through pinned memory: ties @128 (85.7) but
#       collapses @1k (40.8) — MB-scale pinned re-allocs for the 1000-row
#       decode tables are device-synchronizing.
#   "full" — pinned rotating buffers for every per-step tensor: per-op
#       staging cost outweighs the drains it avoids (60.4 @128).
_MODE = os.environ.get("ACP_STAGE_MODE", "numpy")

_TORCH_DTYPE = {
    np.dtype(np.int64): torch.int64,
    np.dtype(np.int32): torch.int32,
    np.dtype(np.float32): torch.float32,
    np.dtype(np.bool_): torch.bool,
}


class HostStager:
    def __init__(self, device, depth: int = 3):
        self.device = torch.device(device)
        self.cuda = self.device.type == "cuda"
        self.depth = depth
        self._slot = 0
        self._fresh_n = 0  # distinct pinned slots for same-step fresh() calls
        # (name, slot) -> pinned buffer; name -> device buffer
        self._pinned: Dict[Tuple, torch.Tensor] = {}
        self._dev: Dict[str, torch.Tensor] = {}

    def step(self) -> None:
        """Rotate staging slots; call once per engine step."""
        self._slot = (self._slot + 1) % self.depth
        self._fresh_n = 0

    def _stage_pinned(self, key, arr: np.ndarray, tdt) -> torch.Tensor:
        n = arr.size
        buf = self._pinned.get(key)
        if buf is None or buf.numel() < n or buf.dtype != tdt:
            buf = torch.empty(max(1024, 2 * n), dtype=tdt, pin_memory=True)
            self._pinned[key] = buf
        buf[:n].copy_(torch.from_numpy(arr.reshape(-1)))
        return buf

    def tensor(self, name: str, data, dtype) -> torch.Tensor:
        """Device tensor valid for THIS step only (rotating buffers)."""
        arr = np.asarray(data, dtype=dtype)
        tdt = _TORCH_DTYPE[arr.dtype]
        if not self.cuda:
            return torch.from_numpy(arr.copy())
        if _MODE != "full":
            return torch.from_numpy(arr).to(self.device, non_blocking=True)
        n = arr.size
        buf = self._stage_pinned((name, self._slot), arr, tdt)
        dev = self._dev.get(name)
        if dev is None or dev.numel() < n or dev.dtype != tdt:
            dev = torch.empty(max(1024, 2 * n), dtype=tdt, device=self.device)
            self._dev[name] = dev
        dev[:n].copy_(buf[:n], non_blocking=True)
        return dev[:n].view(*arr.shape)

    def persistent(self, capacity: int, tail=(), np_dtype=np.float32, depth: int = 3):
        return PersistentStage(self.device, capacity, tail, np_dtype, depth)

    def fresh(self, data, dtype) -> torch.Tensor:
        """Freshly-allocated device tensor (safe to cache across steps),
        still staged through pinned memory with an async copy."""
        arr = np.asarray(data, dtype=dtype)
        tdt = _TORCH_DTYPE[arr.dtype]
        if not self.cuda:
            return torch.from_numpy(arr.copy())
        if _MODE != "full" and _MODE != "params":
            return torch.from_numpy(arr).to(self.device)
        self._fresh_n += 1
        buf = self._stage_pinned(("__fresh__", self._fresh_n, self._slot), arr, tdt)
        dev = torch.empty(arr.shape, dtype=tdt, device=self.device)
        dev.view(-1).copy_(buf[: arr.size], non_blocking=True)
        return dev


class PersistentStage:
    """Fixed-capacity pinned→device staging pair for per-step sampling
    metadata (temperatures/top-k/top-p, grammar masks).

    Round 1 rebuilt these as fresh unpinned host tensors whenever the
    sample-batch membership changed — at 1k concurrency that is almost
    every step, and an unpinned H2D copy is synchronous, draining the
    speculative pipeline (~10 ms/step measured in BENCH_r01's
    ``ls_params`` bucket).  Here both sides are allocated once: fill the
    pinned numpy view, then ``commit(n)`` issues one async copy.  ``depth``
    pinned slots rotate per commit (> pipeline depth 2, so a slot's
    previous copy has been consumed before reuse; stream order protects
    the single device buffer).
    """

    def __init__(self, device, capacity: int, tail=(), np_dtype=np.float32, depth: int = 3):
        self.device = torch.device(device)
        self.cuda = self.device.type == "cuda"
        self.depth = depth
        tdt = _TORCH_DTYPE[np.dtype(np_dtype)]
        shape = (depth, capacity) + tuple(tail)
        self._pin = torch.empty(shape, dtype=tdt, pin_memory=self.cuda)
        self._np = self._pin.numpy()
        self._dev = (
            torch.empty(shape[1:], dtype=tdt, device=self.device) if self.cuda else None
        )
        self._slot = 0

    def host(self) -> np.ndarray:
        """The current slot's pinned numpy view — fill rows [0, n)."""
        return self._np[self._slot]

    def commit(self, n: int) -> torch.Tensor:
        """Async-copy the first n rows to the device; returns the device view
        (valid until overwritten by a later commit, which stream order
        sequences after every launched consumer)."""
        slot = self._slot
        self._slot = (self._slot + 1) % self.depth
        if not self.cuda:
            return self._pin[slot, :n].clone()
        self._dev[:n].copy_(self._pin[slot, :n], non_blocking=True)
        return self._dev[:n]
