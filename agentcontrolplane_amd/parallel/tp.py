"""Tensor parallelism over RCCL/xGMI.

One process per GPU (`torch.distributed`, backend "nccl" = RCCL on ROCm):
the model is head-parallel in attention and column/row-parallel in the MLP
(models/llama.py holds its rank's shard), with one all-reduce after the
o-projection and one after the down-projection per layer — 2 all-reduces
per layer on the 7×153 GB/s xGMI mesh, issued through the ``all_reduce``
hook so kernels and collectives share the stream order torch.distributed
manages.

Serving topology (BASELINE.json config 3 — Llama-3-70B TP=8): rank 0 runs
the control plane + scheduler + sampler; every step's batch metadata is
broadcast (``batch_to_wire``); all ranks execute the sharded forward; the
final hidden states are identical after the last all-reduce, so the
replicated LM head gives identical logits and only rank 0 samples.

``shard_from_full`` slices a full model's weights into a rank's shard —
used by the correctness tests (TP forward must equal the single-model
forward bit-for-bit up to collective reduction order) and by checkpoint
loading when a weights source exists.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from ..engine.batch import batch_from_wire

STOP = {"op": "stop"}


def make_all_reduce(group=None):
    def _all_reduce(t: torch.Tensor) -> torch.Tensor:
        dist.all_reduce(t, group=group)
        return t

    return _all_reduce


def shard_from_full(full, shard, rank: int, world: int) -> None:
    """Copy rank `rank`'s slice of a full (tp_world=1) Llama model's weights
    into `shard` (a model built with tp_world=world)."""
    cfg = full.cfg
    hd = cfg.head_dim
    nq, nkv = cfg.num_heads, cfg.num_kv_heads
    q_per, kv_per = nq // world, nkv // world
    inter = cfg.intermediate_size // world

    def to_dev(t):
        return t.to(device=shard.device, dtype=shard.dtype)

    shard.embed = to_dev(full.embed)
    shard.lm_head = shard.embed if cfg.tie_embeddings else to_dev(full.lm_head)
    shard.final_norm = to_dev(full.final_norm)
    shard.layers = []
    from ..models.llama import LlamaLayerWeights

    for fl in full.layers:
        sl = LlamaLayerWeights()
        sl.input_norm = to_dev(fl.input_norm)
        sl.post_norm = to_dev(fl.post_norm)
        q_w = fl.qkv[: nq * hd]
        k_w = fl.qkv[nq * hd : (nq + nkv) * hd]
        v_w = fl.qkv[(nq + nkv) * hd :]
        sl.qkv = to_dev(
            torch.cat(
                [
                    q_w[rank * q_per * hd : (rank + 1) * q_per * hd],
                    k_w[rank * kv_per * hd : (rank + 1) * kv_per * hd],
                    v_w[rank * kv_per * hd : (rank + 1) * kv_per * hd],
                ]
            )
        )
        sl.o = to_dev(fl.o[:, rank * q_per * hd : (rank + 1) * q_per * hd])
        gate = fl.gate_up[: cfg.intermediate_size]
        up = fl.gate_up[cfg.intermediate_size :]
        sl.gate_up = to_dev(
            torch.cat(
                [
                    gate[rank * inter : (rank + 1) * inter],
                    up[rank * inter : (rank + 1) * inter],
                ]
            )
        )
        sl.down = to_dev(fl.down[:, rank * inter : (rank + 1) * inter])
        shard.layers.append(sl)


def broadcast_step(wire: Optional[dict], src: int = 0, group=None) -> Optional[dict]:
    """Rank 0 sends the step metadata (or STOP); workers receive it."""
    box = [wire]
    dist.broadcast_object_list(box, src=src, group=group)
    return box[0]


def run_tp_worker(engine) -> None:
    """Worker loop for ranks > 0: rebuild each broadcast batch and run the
    sharded forward (graph-replayed for decode-only steps, mirroring rank
    0's dispatch so collectives stay aligned)."""
    device = engine.device
    while True:
        wire = broadcast_step(None)
        if wire is None or wire.get("op") == "stop":
            return
        batch = batch_from_wire(wire, device)
        logits = None
        if engine.graph_runner is not None and not batch.prefills:
            logits = engine.graph_runner.run(batch)
        if logits is None:
            engine.model.forward(batch)
