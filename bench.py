#!/usr/bin/env python3
"""Flagship benchmark: agent-loop steps/sec on the MI355X-native control
plane + inference engine (BASELINE.json metric: "agent-loop steps/sec +
p50 Task latency, Llama-3-8B, 1k concurrent Tasks"; the reference publishes
no numbers — BASELINE.md establishes them here).

One bench step = every one of C concurrent Tasks per GPU completes one full
agent-loop wave: LLM turn with a grammar-constrained tool call → MCP tool
execution → ToolCall join → LLM answer turn → FinalAnswer.  Each wave is
2 LLM turns per task, so one step completes 2·C agent-loop steps per GPU.
value = whole-job agent-loop steps/sec across all N GPUs.

Scaling is weak data parallelism (request sharding): each rank runs its own
control plane + engine on its GPU with C tasks (config 4's sharding mode);
--tp N instead runs one tensor-parallel engine across the ranks.  Synthetic
data: random-init weights of the named architecture, deterministic
pseudo-random prompts (no network for checkpoints or datasets).

Contract: W untimed warmup steps, then EXACTLY K timed steps bracketed by
dist barrier + torch.cuda.synchronize on both sides; elapsed = MAX over
ranks; rank 0 prints one JSON line.
"""
from __future__ import annotations

import argparse
import json
import os
import random
import statistics
import string
import sys
import time


def make_prompt(rng: random.Random, n_chars: int = 600) -> str:
    words = []
    total = 0
    while total < n_chars:
        w = "".join(rng.choice(string.ascii_lowercase) for _ in range(rng.randint(3, 9)))
        words.append(w)
        total += len(w) + 1
    return " ".join(words)


def run_wave(cp, agent_name: str, concurrency: int, rng: random.Random,
             timeout_s: float = 1200.0):
    """Create C tasks, wait until all reach FinalAnswer; returns latencies.

    Completion is watch-driven: round 1 polled every task's full object
    every 20 ms, which cost ~50 s of GIL + store-lock time per wave and
    starved the engine thread — the harness must not be the bottleneck of
    the thing it measures."""
    import queue as _queue

    from agentcontrolplane_amd.api.types import TASK, TaskPhase, make_resource

    t_wave0 = time.monotonic()
    watch_q = cp.store.watch(kinds={TASK})
    names = []
    t_create = {}
    try:
        for i in range(concurrency):
            name = f"bench-{rng.randrange(1 << 30):08x}-{i}"
            cp.store.create(
                make_resource(
                    TASK,
                    name,
                    spec={"agentRef": {"name": agent_name}, "userMessage": make_prompt(rng)},
                )
            )
            t_create[name] = time.monotonic()
            names.append(name)
        t_created = time.monotonic()
        deadline = time.monotonic() + timeout_s
        latencies = {}
        pending = set(names)
        while pending:
            try:
                ev = watch_q.get(timeout=max(0.0, min(5.0, deadline - time.monotonic())))
            except _queue.Empty:
                if time.monotonic() > deadline:
                    raise TimeoutError(
                        f"{len(pending)} tasks unfinished (of {concurrency})"
                    ) from None
                continue
            name = ev.obj.get("metadata", {}).get("name")
            if name not in pending:
                continue
            phase = ev.obj.get("status", {}).get("phase")
            if phase == TaskPhase.FINAL_ANSWER:
                latencies[name] = time.monotonic() - t_create[name]
                pending.discard(name)
            elif phase == TaskPhase.FAILED:
                raise RuntimeError(
                    f"task {name} failed: {ev.obj['status'].get('error')}"
                )
    finally:
        cp.store.stop_watch(watch_q)
    # forensics: the wave is gated by its slowest task — dump its event
    # timeline (stderr; the stdout JSON line stays clean)
    t_polled = time.monotonic()
    slowest = max(latencies, key=latencies.get)
    ls = sorted(latencies.values())
    print(
        f"[wave] n={len(ls)} create={t_created - t_wave0:.2f}s "
        f"poll={t_polled - t_created:.2f}s p50={ls[len(ls)//2]:.2f}s "
        f"p95={ls[max(0,int(0.95*len(ls))-1)]:.2f}s max={ls[-1]:.2f}s ({slowest})",
        file=sys.stderr,
        flush=True,
    )
    if ls[-1] > 3 * ls[len(ls) // 2]:
        evs = cp.store.events_for(slowest)
        t_prev = None
        for e in evs:
            ts = e.get("lastTimestamp", "")
            print(f"[straggler] {ts} {e['reason']}: {e['message'][:80]}", file=sys.stderr)
    # cleanup so the next wave starts from an empty store
    profile = os.environ.get("ACP_BENCH_PROFILE") == "1"
    if profile:
        import cProfile
        import io
        import pstats

        pr = cProfile.Profile()
        pr.enable()
    for name in names:
        cp.store.delete(TASK, name)
    if profile:
        pr.disable()
        st = pstats.Stats(pr)
        st.sort_stats("cumulative")
        buf = io.StringIO()
        st.stream = buf
        st.print_stats(15)
        print(buf.getvalue()[:4000], file=sys.stderr, flush=True)
    print(f"[wave] cleanup={time.monotonic() - t_polled:.2f}s", file=sys.stderr, flush=True)
    return list(latencies.values())


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--model", default="llama3-8b")
    p.add_argument(
        "--concurrency", type=int, default=1000,
        help="tasks per GPU (default = the 1k-concurrent-Tasks config BASELINE.json names)",
    )
    p.add_argument("--decode-tokens", type=int, default=32)
    p.add_argument("--tp", type=int, default=1, help="tensor-parallel degree (else DP sharding)")
    p.add_argument("--device", default=None)
    p.add_argument("--kv-blocks", type=int, default=None)
    p.add_argument("--prefill-budget", type=int, default=16384)
    p.add_argument(
        "--sub-agents", type=int, default=0,
        help="N child agents under the bench agent (BASELINE.json config 4: "
        "the engine's constrained decoding may delegate; children run their "
        "own full loops)",
    )
    p.add_argument(
        "--approval-gate", action="store_true",
        help="route the MCP tool through a human-approval ContactChannel "
        "(BASELINE.json config 5's gate; approvals auto-resolve)",
    )
    p.add_argument(
        "--no-wal", action="store_true",
        help="disable the durable WAL (default on: the reference checkpoints "
        "every transition to etcd, so the measured loop includes durability)",
    )
    args = p.parse_args()

    import torch

    from agentcontrolplane_amd.api.types import AGENT, LLM, MCP_SERVER, make_resource
    from agentcontrolplane_amd.engine.config import EngineConfig
    from agentcontrolplane_amd.engine.engine import InferenceEngine
    from agentcontrolplane_amd.parallel.dist import init_distributed
    from agentcontrolplane_amd.runtime import ControlPlane

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = init_distributed()
    if device == "cuda":
        torch.cuda.set_device(local_rank)
    import torch.distributed as dist

    is_dist = world > 1

    model = args.model
    if device == "cpu":
        model = "tiny"  # CPU smoke of the bench harness itself

    ecfg = EngineConfig(
        model=model,
        device=device,
        num_kv_blocks=args.kv_blocks if args.kv_blocks else (1024 if device == "cpu" else None),
        max_prefill_tokens=args.prefill_budget,
        tensor_parallel=args.tp,
        seed=1234 + rank,
        request_timeout_s=1200,
    )
    if args.tp > 1:
        # TP serving (config 3): rank 0 runs the control plane + scheduler;
        # ranks 1..tp-1 follow the metadata broadcast (parallel/tp.py)
        assert world == args.tp, "launch with torchrun --nproc-per-node == --tp"
        engine = InferenceEngine(ecfg, start=(rank == 0))
        if rank != 0:
            from agentcontrolplane_amd.parallel.tp import run_tp_worker

            run_tp_worker(engine)
            return
    else:
        engine = InferenceEngine(ecfg)

    wal_path = None
    if not args.no_wal:
        import tempfile

        wal_path = os.path.join(
            tempfile.mkdtemp(prefix="acp-bench-"), f"wal-rank{rank}.jsonl"
        )
    cp = ControlPlane(
        engine=engine, auto_approve="approve", llm_probe=False, wal_path=wal_path
    )
    cp.start()
    try:
        cp.store.create(
            make_resource(
                LLM,
                "bench-llm",
                spec={
                    "provider": "local",
                    "parameters": {
                        "model": model,
                        "maxTokens": args.decode_tokens,
                        "temperature": "0.8",
                    },
                },
            )
        )
        llm = cp.store.get(LLM, "bench-llm")
        llm["status"].update({"ready": True, "status": "Ready"})
        cp.store.update_status(llm)
        cp.mcp.register_inproc("tools", {"noop": lambda **_: "ok"})
        mcp_spec = {"transport": "inproc"}
        if args.approval_gate:
            cp.store.create(
                make_resource(
                    "Secret", "hl-key", spec={"data": {"k": "hl-bench"}}, api_version="v1"
                )
            )
            cp.store.create(
                make_resource(
                    "ContactChannel",
                    "bench-approvals",
                    spec={
                        "type": "slack",
                        "apiKeyFrom": {"secretKeyRef": {"name": "hl-key", "key": "k"}},
                        "slack": {"channelOrUserID": "CBENCH"},
                    },
                )
            )
            mcp_spec["approvalContactChannel"] = {"name": "bench-approvals"}
        cp.store.create(make_resource(MCP_SERVER, "tools", spec=mcp_spec))
        agent_spec = {
            "llmRef": {"name": "bench-llm"},
            "system": "You are a benchmark agent. Use tools when offered.",
            "mcpServers": [{"name": "tools"}],
        }
        if args.sub_agents:
            subs = []
            for i in range(args.sub_agents):
                cp.store.create(
                    make_resource(
                        AGENT,
                        f"bench-child-{i}",
                        spec={
                            "llmRef": {"name": "bench-llm"},
                            "system": f"You are specialist {i}.",
                            "mcpServers": [{"name": "tools"}],
                        },
                    )
                )
                subs.append({"name": f"bench-child-{i}"})
            agent_spec["subAgents"] = subs
        cp.store.create(make_resource(AGENT, "bench-agent", spec=agent_spec))
        # wait for the agent to validate
        t0 = time.monotonic()
        while time.monotonic() - t0 < 60:
            a = cp.store.get(AGENT, "bench-agent")
            if a.get("status", {}).get("ready"):
                break
            time.sleep(0.05)
        else:
            raise TimeoutError("agent never became ready")

        rng = random.Random(42 + rank)
        conc = args.concurrency if device != "cpu" else 8

        for _ in range(args.warmup):
            run_wave(cp, "bench-agent", conc, rng)

        dp_sync = is_dist and args.tp <= 1  # TP workers sit in their own loop
        if dp_sync:
            dist.barrier()
        if device == "cuda":
            torch.cuda.synchronize()
        t_start = time.monotonic()
        all_lat = []
        for _ in range(args.steps):
            all_lat.extend(run_wave(cp, "bench-agent", conc, rng))
        if device == "cuda":
            torch.cuda.synchronize()
        if dp_sync:
            dist.barrier()
        elapsed = time.monotonic() - t_start
        if dp_sync:
            # nccl collectives need device tensors
            t = torch.tensor([elapsed], device="cuda" if device == "cuda" else "cpu")
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            elapsed = float(t[0])

        agent_steps = 2 * conc * args.steps  # 2 LLM turns per task per wave
        total_steps = agent_steps * (1 if args.tp > 1 else world)
        value = total_steps / elapsed
        p50 = statistics.median(all_lat)
        em = engine.metrics()
        if rank == 0:
            out = {
                "metric": "agent-loop steps/sec",
                "value": round(value, 3),
                "unit": "steps/s",
                "n_gpus": world if device == "cuda" else args.gpus,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": round(1000.0 * elapsed / args.steps, 2),
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": "bf16" if device == "cuda" else "fp32",
                "data": "synthetic (random-init weights, pseudo-random prompts; no network)",
                "config": {
                    "model": model,
                    "global_batch": conc * (1 if args.tp > 1 else world),
                    "seq_len": 8192,
                    "parallelism": (f"tp{args.tp}" if args.tp > 1 else f"dp{world}"),
                    "concurrency_per_gpu": conc,
                    "decode_tokens": args.decode_tokens,
                    "llm_turns_per_task": 2,
                    "p50_task_latency_s": round(p50, 3),
                    "p95_task_latency_s": round(
                        sorted(all_lat)[max(0, int(0.95 * len(all_lat)) - 1)], 3
                    ),
                    "engine_tokens_per_s": round(
                        em["generated_tokens"] / max(em["busy_time_s"], 1e-9), 1
                    ),
                    "prompt_tokens_total": em["prompt_tokens"],
                    "generated_tokens_total": em["generated_tokens"],
                    "engine_steps": em["steps"],
                    "engine_busy_s": round(em["busy_time_s"], 2),
                    "engine_sched_s": round(em["sched_time_s"], 2),
                    "engine_compute_s": round(em["compute_time_s"], 2),
                    "engine_sample_s": round(em["sample_time_s"], 2),
                    "engine_sample_launch_s": round(em.get("sample_launch_time_s", 0.0), 2),
                    "engine_mask_s": round(em.get("mask_time_s", 0.0), 2),
                    "engine_ls": [round(em.get(k, 0.0), 2) for k in ("ls_gather_s", "ls_params_s", "ls_kernel_s")],
                    "engine_spec_steps": em.get("spec_steps", 0),
                    "engine_graph_steps": em.get("graph_steps", 0),
                    "kv_occupancy": round(em["kv_occupancy"], 4),
                    "preemptions": em["preemptions"],
                },
            }
            print(json.dumps(out))
    finally:
        cp.stop()
        engine.stop()


if __name__ == "__main__":
    main()
