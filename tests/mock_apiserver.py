"""In-process mock Kubernetes apiserver for KubeStore tests.

A thin HTTP shim translating the Kubernetes REST wire (CRUD + status
subresource + labelSelector + chunked watch streams + ownerReference
cascade) onto a local ResourceStore — the same role envtest's apiserver
plays for the reference's controller tests (task/suite_test.go:56-60).

Create/update requests for ACP kinds are validated against the generated
CRD schemas (config/crd/bases/), so applying the reference's sample
manifests here also proves those manifests validate against this repo's
CRDs."""
from __future__ import annotations

import glob
import json
import os
import re
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import yaml

from agentcontrolplane_amd.store import (
    AlreadyExistsError,
    ConflictError,
    NotFoundError,
    ResourceStore,
)

_PLURALS = {
    "llms": "LLM", "agents": "Agent", "tasks": "Task", "toolcalls": "ToolCall",
    "mcpservers": "MCPServer", "contactchannels": "ContactChannel",
    "secrets": "Secret", "events": "Event", "leases": "Lease",
}

_PATH_RE = re.compile(
    r"^/(?:api/v1|apis/[^/]+/[^/]+)/namespaces/(?P<ns>[^/]+)/(?P<plural>[^/]+)"
    r"(?:/(?P<name>[^/]+))?(?:/(?P<sub>status))?$"
)


# ------------------------------------------------------- schema validation


def _load_crd_schemas():
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = {}
    for path in glob.glob(os.path.join(root, "config", "crd", "bases", "*.yaml")):
        with open(path) as f:
            for doc in yaml.safe_load_all(f):
                if not doc or doc.get("kind") != "CustomResourceDefinition":
                    continue  # e.g. the kustomization.yaml in the same dir
                kind = doc["spec"]["names"]["kind"]
                out[kind] = doc["spec"]["versions"][0]["schema"]["openAPIV3Schema"]
    return out


def validate(obj, schema, path="$"):
    """Minimal openAPIV3Schema validator (types, required, enum, items).

    Unknown fields are PRUNED in place, matching the apiserver's structural
    schema behavior — stale sample-manifest fields apply cleanly, as on a
    real cluster."""
    errors = []
    t = schema.get("type")
    if t == "object":
        if not isinstance(obj, dict):
            return [f"{path}: expected object, got {type(obj).__name__}"]
        props = schema.get("properties", {})
        for req in schema.get("required", []):
            if req not in obj:
                errors.append(f"{path}.{req}: required")
        if not schema.get("x-kubernetes-preserve-unknown-fields"):
            for k in [k for k in obj if props and k not in props]:
                obj.pop(k)  # structural-schema pruning
            for k, v in obj.items():
                if k in props:
                    errors.extend(validate(v, props[k], f"{path}.{k}"))
    elif t == "array":
        if not isinstance(obj, list):
            return [f"{path}: expected array"]
        for i, it in enumerate(obj):
            errors.extend(validate(it, schema.get("items", {}), f"{path}[{i}]"))
    elif t == "string":
        if not isinstance(obj, str):
            return [f"{path}: expected string"]
        if "enum" in schema and obj not in schema["enum"]:
            errors.append(f"{path}: {obj!r} not in {schema['enum']}")
    elif t == "integer":
        if not isinstance(obj, int) or isinstance(obj, bool):
            return [f"{path}: expected integer"]
    elif t == "boolean":
        if not isinstance(obj, bool):
            return [f"{path}: expected boolean"]
    return errors


class MockAPIServer:
    def __init__(self, port: int = 0):
        self.store = ResourceStore()
        self._port = port
        self.schemas = _load_crd_schemas()
        self._gen = [0]
        outer = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, *a):
                pass

            def _send(self, code, obj):
                data = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(data)))
                self.end_headers()
                self.wfile.write(data)

            def _status(self, code, reason, msg):
                self._send(code, {"kind": "Status", "status": "Failure",
                                  "reason": reason, "message": msg, "code": code})

            def _body(self):
                n = int(self.headers.get("Content-Length", 0))
                return json.loads(self.rfile.read(n) or b"{}")

            # ---------------------------------------------------------- GET
            def do_GET(self):  # noqa: N802
                from urllib.parse import parse_qs, urlparse

                u = urlparse(self.path)
                m = _PATH_RE.match(u.path)
                if not m:
                    if u.path == "/api/v1/namespaces":
                        self._send(200, {"items": []})
                        return
                    self._status(404, "NotFound", u.path)
                    return
                q = parse_qs(u.query)
                kind = _PLURALS.get(m["plural"])
                if kind is None:
                    self._status(404, "NotFound", m["plural"])
                    return
                ns, name = m["ns"], m["name"]
                if name:
                    obj = outer.store.get(kind, name, ns)
                    if obj is None:
                        self._status(404, "NotFound", f"{kind} {name}")
                    else:
                        self._send(200, obj)
                    return
                if q.get("watch", ["false"])[0] == "true":
                    self._watch(kind)
                    return
                sel = None
                if "labelSelector" in q:
                    sel = dict(kv.split("=", 1) for kv in q["labelSelector"][0].split(","))
                items = outer.store.list(kind, ns, label_selector=sel)
                self._send(200, {"kind": f"{kind}List",
                                 "metadata": {"resourceVersion": "0"},
                                 "items": items})

            def _watch(self, kind):
                wq = outer.store.watch(kinds={kind})
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Transfer-Encoding", "chunked")
                self.end_headers()
                try:
                    import queue as _q

                    while True:
                        try:
                            ev = wq.get(timeout=0.5)
                        except _q.Empty:
                            # heartbeat chunk keeps the connection alive
                            self.wfile.write(b"1\r\n\n\r\n")
                            self.wfile.flush()
                            continue
                        line = json.dumps({"type": ev.type, "object": ev.obj}) + "\n"
                        data = line.encode()
                        self.wfile.write(f"{len(data):x}\r\n".encode() + data + b"\r\n")
                        self.wfile.flush()
                except (BrokenPipeError, ConnectionResetError, OSError):
                    pass
                finally:
                    outer.store.stop_watch(wq)

            # --------------------------------------------------------- POST
            def do_POST(self):  # noqa: N802
                m = _PATH_RE.match(self.path.split("?")[0])
                if self.path.split("?")[0] == "/api/v1/namespaces":
                    self._send(201, self._body())
                    return
                if not m:
                    self._status(404, "NotFound", self.path)
                    return
                kind = _PLURALS.get(m["plural"])
                obj = self._body()
                obj["kind"] = kind
                obj.setdefault("metadata", {})["namespace"] = m["ns"]
                gn = obj["metadata"].pop("generateName", None)
                if gn and not obj["metadata"].get("name"):
                    outer._gen[0] += 1
                    obj["metadata"]["name"] = f"{gn}{outer._gen[0]:06d}"
                errs = outer._validate(kind, obj)
                if errs:
                    self._status(422, "Invalid", "; ".join(errs[:5]))
                    return
                try:
                    out = outer.store.create(obj)
                except AlreadyExistsError as e:
                    self._status(409, "AlreadyExists", str(e))
                    return
                self._send(201, out)

            # ---------------------------------------------------------- PUT
            def do_PUT(self):  # noqa: N802
                m = _PATH_RE.match(self.path.split("?")[0])
                if not m or not m["name"]:
                    self._status(404, "NotFound", self.path)
                    return
                kind = _PLURALS.get(m["plural"])
                obj = self._body()
                obj["kind"] = kind
                obj.setdefault("metadata", {})["namespace"] = m["ns"]
                obj["metadata"]["name"] = m["name"]
                errs = outer._validate(kind, obj)
                if errs:
                    self._status(422, "Invalid", "; ".join(errs[:5]))
                    return
                try:
                    if m["sub"] == "status":
                        out = outer.store.update_status(obj)
                    elif kind == "Lease":
                        # leases PUT whole-object (no spec/status split)
                        cur = outer.store.get(kind, m["name"], m["ns"])
                        if cur is None:
                            self._status(404, "NotFound", m["name"])
                            return
                        out = outer.store.update(obj)
                    else:
                        out = outer.store.update(obj)
                except NotFoundError as e:
                    self._status(404, "NotFound", str(e))
                    return
                except ConflictError as e:
                    self._status(409, "Conflict", str(e))
                    return
                self._send(200, out)

            # ------------------------------------------------------- DELETE
            def do_DELETE(self):  # noqa: N802
                m = _PATH_RE.match(self.path.split("?")[0])
                if not m or not m["name"]:
                    self._status(404, "NotFound", self.path)
                    return
                kind = _PLURALS.get(m["plural"])
                if outer.store.delete(kind, m["name"], m["ns"]):
                    self._send(200, {"kind": "Status", "status": "Success"})
                else:
                    self._status(404, "NotFound", m["name"])

        self._srv = ThreadingHTTPServer(("127.0.0.1", self._port), Handler)
        self._thread = threading.Thread(target=self._srv.serve_forever, daemon=True)
        self._thread.start()
        self.base = f"http://127.0.0.1:{self._srv.server_address[1]}"

    def _validate(self, kind, obj):
        schema = self.schemas.get(kind)
        if schema is None:
            return []
        spec_schema = schema.get("properties", {}).get("spec", {})
        return validate(obj.get("spec", {}), spec_schema, "$.spec")

    def shutdown(self):
        self._srv.shutdown()
        self._srv.server_close()  # release the listening socket (restarts
        # re-bind the same port in the reconnect test)
