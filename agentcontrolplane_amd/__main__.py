"""CLI — the cmd/main.go equivalent.

    python -m agentcontrolplane_amd serve [--wal PATH] [--model llama3-8b]
        [--device cuda|cpu|none] [--port 8082] [--auto-approve]
    python -m agentcontrolplane_amd apply -f manifest.yaml [-f more.yaml]
        [--server http://127.0.0.1:8082]
    python -m agentcontrolplane_amd get <kind> [name]

``serve`` wires the store + six reconcilers + MCP manager + optional local
engine + the REST API on :8082 (the reference's api_service port).
``apply`` accepts the same YAML manifests the reference's CRDs use
(acp.humanlayer.dev/v1alpha1 Agent/LLM/Task/…), posted into a running
server's store through a small admin endpoint, so existing manifests run
unchanged without a kubernetes cluster.
"""
from __future__ import annotations

import argparse
import json
import sys


def cmd_serve(args) -> None:
    import uvicorn

    from .engine.config import EngineConfig
    from .runtime import ControlPlane

    engine = None
    if args.device != "none":
        import torch

        device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
        model = args.model if device == "cuda" else "tiny"
        ecfg = EngineConfig(
            model=model,
            device=device,
            checkpoint_path=args.checkpoint,
            num_kv_blocks=args.kv_blocks,
        )
        n_gpus = getattr(args, "gpus", 1) or 1
        if device.startswith("cuda") and n_gpus > 1:
            # DP request routing across this node's GPUs (parallel/router)
            from .parallel.router import EnginePool

            engine = EnginePool.build(ecfg, n_gpus)
        else:
            from .engine.engine import InferenceEngine

            engine = InferenceEngine(ecfg)
    store = None
    if getattr(args, "kube", False):
        # apiserver-backed store: in-cluster credentials or --kube-url
        from .store.kube import KubeStore

        store = KubeStore(base_url=args.kube_url or None)
    import os as _os

    cp = ControlPlane(
        wal_path=args.wal,
        engine=engine,
        auto_approve="approve" if args.auto_approve else None,
        store=store,
        pod_name=_os.environ.get("POD_NAME", "acp-controller-0"),
    )
    cp.start(leader_elect=getattr(args, "leader_elect", False))
    app = cp.rest_app
    from .server.admin import add_admin_routes

    add_admin_routes(app, cp.store)
    print(f"acp-amd serving on :{args.port} (engine={'on' if engine else 'off'})")
    try:
        uvicorn.run(app, host=args.host, port=args.port, log_level="warning")
    finally:
        cp.stop()
        if engine is not None:
            engine.stop()


def cmd_apply(args) -> None:
    import yaml

    try:
        import httpx

        client = httpx.Client(base_url=args.server, timeout=30)
    except ImportError:  # pragma: no cover
        print("httpx required for apply", file=sys.stderr)
        sys.exit(1)
    for path in args.files:
        with open(path) as f:
            for doc in yaml.safe_load_all(f):
                if not doc:
                    continue
                r = client.post("/admin/resources", json=doc)
                name = doc.get("metadata", {}).get("name", "?")
                kind = doc.get("kind", "?")
                if r.status_code in (200, 201):
                    print(f"{kind.lower()}/{name} {'configured' if r.status_code == 200 else 'created'}")
                else:
                    print(f"{kind.lower()}/{name} error: {r.text}", file=sys.stderr)


# printer columns per kind (the reference's kubebuilder printcolumn UX,
# e.g. task_types.go:195-207)
_COLUMNS = {
    "Task": [
        ("NAME", lambda o: o["metadata"]["name"]),
        ("AGENT", lambda o: o.get("spec", {}).get("agentRef", {}).get("name", "")),
        ("PHASE", lambda o: o.get("status", {}).get("phase", "")),
        ("READY", lambda o: str(o.get("status", {}).get("ready", False))),
        ("PREVIEW", lambda o: o.get("status", {}).get("userMsgPreview", "")[:40]),
    ],
    "Agent": [
        ("NAME", lambda o: o["metadata"]["name"]),
        ("READY", lambda o: str(o.get("status", {}).get("ready", False))),
        ("STATUS", lambda o: o.get("status", {}).get("status", "")),
        ("DETAIL", lambda o: o.get("status", {}).get("statusDetail", "")[:48]),
    ],
    "LLM": [
        ("NAME", lambda o: o["metadata"]["name"]),
        ("PROVIDER", lambda o: o.get("spec", {}).get("provider", "")),
        ("READY", lambda o: str(o.get("status", {}).get("ready", False))),
        ("DETAIL", lambda o: o.get("status", {}).get("statusDetail", "")[:48]),
    ],
    "ToolCall": [
        ("NAME", lambda o: o["metadata"]["name"]),
        ("PHASE", lambda o: o.get("status", {}).get("phase", "")),
        ("TASK", lambda o: o.get("spec", {}).get("taskRef", {}).get("name", "")),
        ("TOOL", lambda o: o.get("spec", {}).get("toolRef", {}).get("name", "")),
        ("ERROR", lambda o: o.get("status", {}).get("error", "")[:40]),
    ],
    "MCPServer": [
        ("NAME", lambda o: o["metadata"]["name"]),
        ("TRANSPORT", lambda o: o.get("spec", {}).get("transport", "")),
        ("CONNECTED", lambda o: str(o.get("status", {}).get("connected", False))),
        ("TOOLS", lambda o: str(len(o.get("status", {}).get("tools", []) or []))),
    ],
    "ContactChannel": [
        ("NAME", lambda o: o["metadata"]["name"]),
        ("TYPE", lambda o: o.get("spec", {}).get("type", "")),
        ("READY", lambda o: str(o.get("status", {}).get("ready", False))),
        ("DETAIL", lambda o: o.get("status", {}).get("statusDetail", "")[:48]),
    ],
}


def _print_table(kind: str, objs) -> None:
    cols = _COLUMNS.get(kind)
    if not cols:
        print(json.dumps(objs, indent=2))
        return
    rows = [[str(fn(o)) for _, fn in cols] for o in objs]
    widths = [
        max(len(h), *(len(r[i]) for r in rows)) if rows else len(h)
        for i, (h, _) in enumerate(cols)
    ]
    print("  ".join(h.ljust(w) for (h, _), w in zip(cols, widths)))
    for r in rows:
        print("  ".join(c.ljust(w) for c, w in zip(r, widths)))


def cmd_get(args) -> None:
    import httpx

    client = httpx.Client(base_url=args.server, timeout=30)
    url = f"/admin/resources/{args.kind}"
    if args.name:
        url += f"/{args.name}"
    r = client.get(url, params={"namespace": args.namespace})
    data = r.json()
    if args.output == "json" or args.name:
        print(json.dumps(data, indent=2))
        return
    from .api.types import KINDS

    canonical = {k.lower(): k for k in KINDS}
    canonical.update({k.lower() + "s": k for k in KINDS})
    _print_table(canonical.get(args.kind.lower(), args.kind), data)


def cmd_delete(args) -> None:
    import httpx

    client = httpx.Client(base_url=args.server, timeout=30)
    r = client.delete(
        f"/admin/resources/{args.kind}/{args.name}", params={"namespace": args.namespace}
    )
    if r.status_code >= 400:
        print(r.text, file=sys.stderr)
        sys.exit(1)
    print(r.json().get("deleted", ""))


def main() -> None:
    p = argparse.ArgumentParser(prog="agentcontrolplane_amd")
    sub = p.add_subparsers(dest="cmd", required=True)

    s = sub.add_parser("serve", help="run the control plane + engine + REST API")
    s.add_argument("--host", default="127.0.0.1")
    s.add_argument("--port", type=int, default=8082)
    s.add_argument("--wal", default=None, help="WAL path for durable state")
    s.add_argument("--model", default="llama3-8b")
    s.add_argument("--checkpoint", default=None, help="safetensors checkpoint dir")
    s.add_argument("--device", default=None, help="cuda | cpu | none (no engine)")
    s.add_argument("--kv-blocks", type=int, default=None)
    s.add_argument("--gpus", type=int, default=1,
                   help="DP-route requests across N GPUs of this node "
                   "(one engine per GPU, least-loaded routing)")
    s.add_argument("--auto-approve", action="store_true")
    s.add_argument("--kube", action="store_true",
                   help="back the store with the Kubernetes apiserver "
                   "(in-cluster credentials or --kube-url)")
    s.add_argument("--kube-url", default=None)
    s.add_argument("--leader-elect", action="store_true",
                   help="multi-replica active/passive via a coordination Lease")
    s.set_defaults(fn=cmd_serve)

    a = sub.add_parser("apply", help="apply CRD-style YAML manifests")
    a.add_argument("-f", "--files", action="append", required=True)
    a.add_argument("--server", default="http://127.0.0.1:8082")
    a.set_defaults(fn=cmd_apply)

    g = sub.add_parser("get", help="get resources")
    g.add_argument("kind")
    g.add_argument("name", nargs="?")
    g.add_argument("--namespace", default="default")
    g.add_argument("--server", default="http://127.0.0.1:8082")
    g.add_argument("-o", "--output", default="table", choices=["table", "json"])
    g.set_defaults(fn=cmd_get)

    d = sub.add_parser("delete", help="delete a resource (cascades to owned children)")
    d.add_argument("kind")
    d.add_argument("name")
    d.add_argument("--namespace", default="default")
    d.add_argument("--server", default="http://127.0.0.1:8082")
    d.set_defaults(fn=cmd_delete)

    args = p.parse_args()
    args.fn(args)


if __name__ == "__main__":
    main()
