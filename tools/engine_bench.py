#!/usr/bin/env python3
"""Raw engine throughput microbench (no control plane): submit R requests,
measure prefill and decode rates.  The rocprofv3 target for kernel-level
profiling (profiles/ keeps the committed summaries)."""
import argparse
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--requests", type=int, default=128)
    p.add_argument("--prompt-tokens", type=int, default=512)
    p.add_argument("--decode-tokens", type=int, default=64)
    p.add_argument("--constrained", action="store_true")
    p.add_argument("--kv-blocks", type=int, default=None)
    p.add_argument("--prefill-budget", type=int, default=8192)
    p.add_argument("--device", default=None)
    args = p.parse_args()

    import torch

    from agentcontrolplane_amd.engine.config import EngineConfig
    from agentcontrolplane_amd.engine.engine import InferenceEngine
    from agentcontrolplane_amd.engine.request import InferenceRequest, SamplingParams

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    model = args.model if device == "cuda" else "tiny"
    eng = InferenceEngine(
        EngineConfig(
            model=model,
            device=device,
            num_kv_blocks=args.kv_blocks,
            max_prefill_tokens=args.prefill_budget,
            request_timeout_s=1200,
        )
    )
    tools = [
        {"type": "function", "function": {"name": "t__x", "description": "", "parameters": {}}}
    ]
    try:
        sp = SamplingParams(
            max_tokens=args.decode_tokens,
            temperature=0.8,
            tool_choice="required" if args.constrained else "none",
        )
        # warmup
        eng.generate(list(range(64)), SamplingParams(max_tokens=4, temperature=0.8))
        t0 = time.monotonic()
        reqs = []
        for i in range(args.requests):
            prompt = [(i * 37 + j * 13) % 256 for j in range(args.prompt_tokens)]
            r = InferenceRequest(prompt, sp, constrained=args.constrained,
                                 tools=tools if args.constrained else [])
            eng.submit(r)
            reqs.append(r)
        for r in reqs:
            r.wait(1200)
        if device == "cuda":
            torch.cuda.synchronize()
        dt = time.monotonic() - t0
        m = eng.metrics()
        total_prompt = args.requests * args.prompt_tokens
        total_gen = sum(len(r.output_ids) for r in reqs)
        print(
            f"requests={args.requests} elapsed={dt:.2f}s "
            f"prefill={total_prompt} tok ({total_prompt / dt:.0f} tok/s overall) "
            f"decode={total_gen} tok ({total_gen / dt:.0f} tok/s overall) "
            f"steps={m['steps']} sched={m['sched_time_s']:.2f}s "
            f"compute={m['compute_time_s']:.2f}s sample={m['sample_time_s']:.2f}s"
        )
    finally:
        eng.stop()


if __name__ == "__main__":
    main()
