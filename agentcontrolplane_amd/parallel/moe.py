"""Expert parallelism: MoE all-to-all over RCCL/xGMI.

Experts are sharded across ranks (rank r owns experts [r·E/w, (r+1)·E/w));
each MoE layer routes its tokens to the owning rank with an all-to-all,
computes locally, and returns the expert outputs with the reverse
all-to-all (BASELINE.json config 5: Mixtral 8×7B, one expert group per
GPU of the 8-GPU node — the all-to-all rides the 7×153 GB/s xGMI links
point-to-point, which is exactly the traffic shape EP produces: every rank
exchanges a different row set with every other rank).

Routing weights stay on the source rank and are applied at combine time, so
the wire carries only hidden-state rows.  On the "nccl" (RCCL) backend the
exchange is ``all_to_all_single``; gloo (CPU tests) has no all-to-all, so a
send/recv emulation keeps the tests backend-portable.
"""
from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist
import torch.nn.functional as F


def _exchange(parts: List[torch.Tensor], group=None) -> List[torch.Tensor]:
    """parts[r] goes to rank r; returns what every rank sent to us."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    backend = dist.get_backend(group)
    if backend == "nccl":
        send_counts = torch.tensor([p.shape[0] for p in parts], device=parts[0].device)
        recv_counts = torch.empty_like(send_counts)
        dist.all_to_all_single(recv_counts, send_counts, group=group)
        send = torch.cat(parts, dim=0)
        cols = send.shape[1] if send.dim() > 1 else 1
        recv = torch.empty(
            int(recv_counts.sum()), *send.shape[1:], dtype=send.dtype, device=send.device
        )
        dist.all_to_all_single(
            recv, send,
            output_split_sizes=recv_counts.tolist(),
            input_split_sizes=send_counts.tolist(),
            group=group,
        )
        out, off = [], 0
        for r in range(world):
            n = int(recv_counts[r])
            out.append(recv[off : off + n])
            off += n
        return out
    # gloo (CPU tests): no all_to_all — object all-gather emulation
    gathered: List[List[torch.Tensor]] = [None] * world  # type: ignore[list-item]
    dist.all_gather_object(gathered, list(parts), group=group)
    return [gathered[src][rank] for src in range(world)]


class ExpertParallelDispatcher:
    """Installs as model.moe_dispatch; owns this rank's expert slice."""

    def __init__(self, model, group=None):
        self.model = model
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        E = model.num_experts
        assert E % self.world == 0, "experts must divide the EP world"
        self.e_per = E // self.world
        self.e_lo = self.rank * self.e_per
        model.moe_dispatch = self.dispatch

    def _owner(self, e: torch.Tensor) -> torch.Tensor:
        return e // self.e_per

    def dispatch(self, li: int, x: torch.Tensor, topi: torch.Tensor,
                 topw: torch.Tensor) -> torch.Tensor:
        """x: [N, H]; topi/topw: [N, k].  Returns the combined MoE output."""
        from .. import ops

        N, k = topi.shape
        flat_e = topi.reshape(-1)                       # [N*k]
        flat_rows = torch.arange(N, device=x.device).repeat_interleave(k)
        owners = self._owner(flat_e)

        # sort by owner so per-rank slices are contiguous
        order = torch.argsort(owners, stable=True)
        flat_e_s = flat_e[order]
        rows_s = flat_rows[order]
        owners_s = owners[order]
        counts = torch.bincount(owners_s, minlength=self.world)

        send_x, send_e = [], []
        off = 0
        for r in range(self.world):
            n = int(counts[r])
            sel = order[off : off + n]
            send_x.append(x[flat_rows[sel]])
            send_e.append(flat_e[sel].to(torch.float32).unsqueeze(1))  # ship as f32 col
            off += n

        recv_x = _exchange(send_x, self.group)
        recv_e = _exchange(send_e, self.group)

        # compute local experts on every received row
        results = []
        for r in range(self.world):
            xr = recv_x[r]
            er = recv_e[r].squeeze(1).to(torch.long) - self.e_lo
            yr = torch.empty_like(xr)
            for le in range(self.e_per):
                sel = (er == le).nonzero(as_tuple=True)[0]
                if sel.numel() == 0:
                    continue
                ge = self.model.expert_gate_up[li][self.e_lo + le]
                dn = self.model.expert_down[li][self.e_lo + le]
                yr[sel] = F.linear(ops.swiglu(F.linear(xr[sel], ge)), dn)
            results.append(yr)

        back = _exchange(results, self.group)

        out = torch.zeros_like(x)
        flat_w = topw.reshape(-1)
        off = 0
        for r in range(self.world):
            n = int(counts[r])
            sel = order[off : off + n]
            out.index_add_(0, flat_rows[sel], back[r] * flat_w[sel].unsqueeze(1))
            off += n
        return out


def shard_experts_from_full(full, shard, rank: int, world: int) -> None:
    """Copy the full Mixtral's routers + this rank's expert slice (tests /
    checkpoint loading).  Attention weights shard like TP=1 (replicated)."""
    shard.routers = [r.clone() for r in full.routers]
    shard.expert_gate_up = [t.clone() for t in full.expert_gate_up]
    shard.expert_down = [t.clone() for t in full.expert_down]
