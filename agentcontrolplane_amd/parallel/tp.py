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
        if getattr(fl, "qkv_bias", None) is not None:
            qb = fl.qkv_bias[: nq * hd]
            kb = fl.qkv_bias[nq * hd : (nq + nkv) * hd]
            vb = fl.qkv_bias[(nq + nkv) * hd :]
            sl.qkv_bias = to_dev(
                torch.cat(
                    [
                        qb[rank * q_per * hd : (rank + 1) * q_per * hd],
                        kb[rank * kv_per * hd : (rank + 1) * kv_per * hd],
                        vb[rank * kv_per * hd : (rank + 1) * kv_per * hd],
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


_META_GROUP = None
_META_INIT = False


def _meta_group():
    """A gloo side-group for the metadata broadcast: the step wire is
    consumed on the host (workers rebuild host lists + device tensors), so
    a CPU-tensor broadcast avoids the GPU round-trip a NCCL broadcast
    would force — and tensor broadcast replaces round 1's per-step
    ``broadcast_object_list`` pickling."""
    global _META_GROUP, _META_INIT
    if not _META_INIT:
        _META_INIT = True
        if dist.get_backend() != "gloo":
            _META_GROUP = dist.new_group(backend="gloo")
    return _META_GROUP


def _pack_wire(wire: Optional[dict]) -> torch.Tensor:
    """Flatten the step metadata into one int64 tensor (layout mirrored by
    _unpack_wire).  op codes: 0 = step, 1 = stop."""
    if wire is None or wire.get("op") == "stop":
        return torch.tensor([1], dtype=torch.int64)
    out = [0]
    tok = wire["token_ids"]
    out.append(len(tok))
    out.extend(tok)
    out.extend(wire["positions"])
    out.extend(wire["slot_mapping"])
    out.append(wire["num_prefill_tokens"])
    prefills = wire["prefills"]
    out.append(len(prefills))
    for (sid, q, sl, c, bt, nl) in prefills:
        out.extend((sid, q, sl, c, 1 if nl else 0, len(bt)))
        out.extend(bt)
    dec = wire["decode_seq_ids"]
    out.append(len(dec))
    out.extend(dec)
    tables = wire["decode_block_tables"]
    if tables is None:
        out.append(-1)
    else:
        width = len(tables[0]) if tables else 0
        out.append(width)
        for row in tables:
            out.extend(row)
    lens = wire["decode_seq_lens"]
    out.append(-1 if lens is None else len(lens))
    if lens is not None:
        out.extend(lens)
    lr = wire["logit_rows"]
    out.append(len(lr))
    out.extend(lr)
    ss = wire["sample_seq_ids"]
    out.append(len(ss))
    out.extend(ss)
    return torch.tensor(out, dtype=torch.int64)


def _unpack_wire(t: torch.Tensor) -> Optional[dict]:
    data = t.tolist()
    pos = 0

    def take(n):
        nonlocal pos
        v = data[pos:pos + n]
        pos += n
        return v

    def one():
        return take(1)[0]

    if one() == 1:
        return STOP
    n = one()
    token_ids = take(n)
    positions = take(n)
    slot_mapping = take(n)
    num_prefill_tokens = one()
    prefills = []
    for _ in range(one()):
        sid, q, sl, c, nl, btn = take(6)
        prefills.append((sid, q, sl, c, take(btn), bool(nl)))
    n_dec = one()
    decode_seq_ids = take(n_dec)
    width = one()
    tables = None
    if width >= 0:
        tables = [take(width) for _ in range(n_dec)]
    n_lens = one()
    lens = take(n_lens) if n_lens >= 0 else None
    logit_rows = take(one())
    sample_seq_ids = take(one())
    return {
        "token_ids": token_ids,
        "positions": positions,
        "slot_mapping": slot_mapping,
        "prefills": prefills,
        "num_prefill_tokens": num_prefill_tokens,
        "decode_seq_ids": decode_seq_ids,
        "decode_block_tables": tables,
        "decode_seq_lens": lens,
        "logit_rows": logit_rows,
        "sample_seq_ids": sample_seq_ids,
    }


def broadcast_step(wire: Optional[dict], src: int = 0, group=None) -> Optional[dict]:
    """Rank 0 sends the step metadata (or STOP); workers receive it.  One
    size broadcast + one packed int64 payload broadcast over the gloo
    side-group — no pickling, no GPU round-trip."""
    g = group if group is not None else _meta_group()
    if dist.get_rank() == src:
        payload = _pack_wire(wire)
        size = torch.tensor([payload.numel()], dtype=torch.int64)
        dist.broadcast(size, src=src, group=g)
        dist.broadcast(payload, src=src, group=g)
        return wire
    size = torch.tensor([0], dtype=torch.int64)
    dist.broadcast(size, src=src, group=g)
    payload = torch.empty(int(size[0]), dtype=torch.int64)
    dist.broadcast(payload, src=src, group=g)
    return _unpack_wire(payload)


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
