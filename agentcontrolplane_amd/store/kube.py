"""Kubernetes-apiserver-backed ResourceStore.

The reference runs against a real apiserver/etcd through controller-runtime
(SURVEY.md L0); this module gives the MI355X control plane the same
deployment option: the identical store interface (`create/get/list/update/
update_status/delete/watch/record_event/acquire_lease/...`) implemented
over the Kubernetes REST API, so `ControlPlane(store=KubeStore(...))`
reconciles CRs that `kubectl apply` created — the north star's "kubectl
apply of existing Agent/Task manifests runs unchanged".

Wire mapping:
- ACP kinds  -> /apis/acp.humanlayer.dev/v1alpha1/namespaces/{ns}/{plural}
               (CRDs from config/crd/bases/, status as a subresource)
- Secret     -> /api/v1/namespaces/{ns}/secrets
- Event      -> /api/v1/namespaces/{ns}/events
- Lease      -> /apis/coordination.k8s.io/v1/namespaces/{ns}/leases
  (task locking, task/state_machine.go:1069-1145)
- watches    -> chunked `?watch=true` streams, one reader thread per
               resource type, re-listing on 410 Gone

HTTP status mapping keeps the local-store error contract: 404 ->
NotFoundError, 409 -> ConflictError (the controllers' conflict-retry and
IgnoreNotFound paths work unchanged against either backend).

Auth: in-cluster service-account token when present
(/var/run/secrets/kubernetes.io/serviceaccount), else an explicit
base_url/token/ca (tests run against an in-process mock apiserver —
tests/test_kube_store.py)."""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Any, Dict, Iterable, List, Optional

from ..api.types import EVENT, LEASE, SECRET, now_iso
from .store import (
    AlreadyExistsError,
    ConflictError,
    NotFoundError,
    WatchEvent,
    _Watch,
    _copy,
)

GROUP = "acp.humanlayer.dev"
VERSION = "v1alpha1"

# kind -> (api prefix, plural)
ACP_KINDS = {
    "LLM": "llms",
    "Agent": "agents",
    "Task": "tasks",
    "ToolCall": "toolcalls",
    "MCPServer": "mcpservers",
    "ContactChannel": "contactchannels",
}
CORE_KINDS = {SECRET: "secrets", EVENT: "events"}


class KubeStore:
    """ResourceStore-compatible facade over a Kubernetes apiserver."""

    def __init__(self, base_url: Optional[str] = None, token: Optional[str] = None,
                 verify: Any = True, transport=None, watch_kinds: Optional[List[str]] = None):
        import httpx

        sa = "/var/run/secrets/kubernetes.io/serviceaccount"
        if base_url is None and os.path.exists(f"{sa}/token"):
            host = os.environ.get("KUBERNETES_SERVICE_HOST", "kubernetes.default.svc")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            base_url = f"https://{host}:{port}"
            token = open(f"{sa}/token").read().strip()
            if os.path.exists(f"{sa}/ca.crt"):
                verify = f"{sa}/ca.crt"
        if base_url is None:
            raise ValueError("KubeStore needs base_url (or in-cluster credentials)")
        self.base = base_url.rstrip("/")
        headers = {"Content-Type": "application/json"}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        self._client = httpx.Client(
            base_url=self.base, headers=headers, verify=verify,
            timeout=httpx.Timeout(10.0, read=None), transport=transport,
            trust_env=False,
        )
        self._lock = threading.Lock()
        self._watches: List[_Watch] = []
        self._watch_threads: List[threading.Thread] = []
        self._stop = False
        self._watch_kinds = watch_kinds or (list(ACP_KINDS) + [EVENT])
        # single-namespace deployment scope (the manager's POD_NAMESPACE,
        # like the reference's namespaced kustomize install); cluster-wide
        # informers would need the un-namespaced list/watch paths
        self.namespace = os.environ.get("POD_NAMESPACE", "default")

    # ------------------------------------------------------------------ paths

    def _path(self, kind: str, namespace: str, name: str = "", sub: str = "") -> str:
        if kind in ACP_KINDS:
            p = f"/apis/{GROUP}/{VERSION}/namespaces/{namespace}/{ACP_KINDS[kind]}"
        elif kind in CORE_KINDS:
            p = f"/api/v1/namespaces/{namespace}/{CORE_KINDS[kind]}"
        elif kind == LEASE:
            p = f"/apis/coordination.k8s.io/v1/namespaces/{namespace}/leases"
        else:
            raise NotFoundError(f"no API mapping for kind {kind!r}")
        if name:
            p += f"/{name}"
        if sub:
            p += f"/{sub}"
        return p

    @staticmethod
    def _check(r, kind: str, name: str):
        if r.status_code == 404:
            raise NotFoundError(f"{kind} {name} not found")
        if r.status_code == 409:
            body = r.text
            if "AlreadyExists" in body or '"reason":"AlreadyExists"' in body:
                raise AlreadyExistsError(f"{kind} {name} already exists")
            raise ConflictError(f"{kind} {name}: {body[:200]}")
        if r.status_code >= 400:
            raise RuntimeError(f"apiserver {r.status_code} for {kind} {name}: {r.text[:300]}")
        return r.json() if r.content else None

    # ------------------------------------------------------------ primitives

    def create(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        obj = _copy(obj)
        kind = obj["kind"]
        ns = obj.setdefault("metadata", {}).setdefault("namespace", "default")
        obj.setdefault("apiVersion", self._api_version(kind))
        # the apiserver assigns uid/resourceVersion/creationTimestamp
        for k in ("uid", "resourceVersion", "creationTimestamp"):
            obj["metadata"].pop(k, None)
        status = obj.pop("status", None)
        r = self._client.post(self._path(kind, ns), json=obj)
        out = self._check(r, kind, obj["metadata"].get("name", ""))
        if status:
            # SetupWithStatus-style force write (test fixtures start
            # mid-state-machine; the apiserver drops status on create)
            out["status"] = status
            out = self.update_status(out)
        out.setdefault("status", {})
        return out

    @staticmethod
    def _api_version(kind: str) -> str:
        if kind in ACP_KINDS:
            return f"{GROUP}/{VERSION}"
        if kind == LEASE:
            return "coordination.k8s.io/v1"
        return "v1"

    def get(self, kind: str, name: str, namespace: str = "default") -> Optional[Dict[str, Any]]:
        r = self._client.get(self._path(kind, namespace, name))
        if r.status_code == 404:
            return None
        out = self._check(r, kind, name)
        out.setdefault("status", {})
        return out

    def list(self, kind: str, namespace: Optional[str] = "default",
             label_selector: Optional[Dict[str, str]] = None) -> List[Dict[str, Any]]:
        params = {}
        if label_selector:
            params["labelSelector"] = ",".join(f"{k}={v}" for k, v in label_selector.items())
        if namespace is None:
            namespace = self.namespace  # single-namespace deployment scope
        r = self._client.get(self._path(kind, namespace), params=params)
        if r.status_code == 404:
            return []
        out = self._check(r, kind, "")
        items = out.get("items", []) or []
        for it in items:
            it.setdefault("kind", kind)
            it.setdefault("status", {})
        return items

    def update(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        obj = _copy(obj)
        kind, m = obj["kind"], obj["metadata"]
        ns, name = m.get("namespace", "default"), m["name"]
        obj.pop("status", None)  # status is a subresource
        obj.setdefault("apiVersion", self._api_version(kind))
        r = self._client.put(self._path(kind, ns, name), json=obj)
        out = self._check(r, kind, name)
        out.setdefault("status", {})
        return out

    def update_status(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        obj = _copy(obj)
        kind, m = obj["kind"], obj["metadata"]
        ns, name = m.get("namespace", "default"), m["name"]
        obj.setdefault("apiVersion", self._api_version(kind))
        r = self._client.put(self._path(kind, ns, name, sub="status"), json=obj)
        out = self._check(r, kind, name)
        out.setdefault("status", {})
        return out

    def delete(self, kind: str, name: str, namespace: str = "default") -> bool:
        r = self._client.delete(self._path(kind, namespace, name))
        if r.status_code == 404:
            return False
        self._check(r, kind, name)
        return True

    # --------------------------------------------------------------- watches

    def watch(self, kinds: Optional[Iterable[str]] = None):
        w = _Watch(kinds)
        with self._lock:
            self._watches.append(w)
            if not self._watch_threads:
                self._start_watch_threads()
        return w.q

    def stop_watch(self, q) -> None:
        with self._lock:
            for w in self._watches:
                if w.q is q:
                    w.alive = False
            self._watches = [w for w in self._watches if w.alive]

    def _start_watch_threads(self) -> None:
        for kind in self._watch_kinds:
            t = threading.Thread(
                target=self._watch_loop, args=(kind,),
                name=f"kube-watch-{kind.lower()}", daemon=True,
            )
            t.start()
            self._watch_threads.append(t)

    def _notify(self, ev: WatchEvent) -> None:
        with self._lock:
            watchers = list(self._watches)
        for w in watchers:
            if w.alive and (w.kinds is None or ev.kind in w.kinds):
                w.q.put(ev)

    def _watch_loop(self, kind: str) -> None:
        import httpx

        ns = self.namespace
        rv = None
        while not self._stop:
            try:
                if rv is None:
                    r = self._client.get(self._path(kind, ns))
                    body = r.json()
                    rv = body.get("metadata", {}).get("resourceVersion", "0")
                    for it in body.get("items", []) or []:
                        it.setdefault("kind", kind)
                        it.setdefault("status", {})
                        self._notify(WatchEvent("ADDED", kind, it))
                with self._client.stream(
                    "GET", self._path(kind, ns),
                    params={"watch": "true", "resourceVersion": rv},
                ) as r:
                    if r.status_code == 410:
                        rv = None
                        continue
                    for line in r.iter_lines():
                        if self._stop:
                            return
                        if not line.strip():
                            continue
                        ev = json.loads(line)
                        typ, obj = ev.get("type"), ev.get("object", {})
                        if typ == "ERROR":
                            rv = None
                            break
                        obj.setdefault("kind", kind)
                        obj.setdefault("status", {})
                        new_rv = obj.get("metadata", {}).get("resourceVersion")
                        if new_rv:
                            rv = new_rv
                        if typ in ("ADDED", "MODIFIED", "DELETED"):
                            self._notify(WatchEvent(typ, kind, obj))
            except (httpx.HTTPError, json.JSONDecodeError, OSError):
                time.sleep(0.5)
                rv = None

    # ---------------------------------------------------------------- events

    def record_event(self, involved: Dict[str, Any], event_type: str,
                     reason: str, message: str) -> None:
        m = involved.get("metadata", {})
        ns = m.get("namespace", "default")
        ev = {
            "apiVersion": "v1",
            "kind": EVENT,
            "metadata": {"generateName": f'{m.get("name", "obj")}.', "namespace": ns},
            "involvedObject": {
                "kind": involved.get("kind"),
                "name": m.get("name"),
                "namespace": ns,
                "uid": m.get("uid", ""),
            },
            "type": event_type,
            "reason": reason,
            "message": message,
            "count": 1,
            "lastTimestamp": now_iso(),
            "source": {"component": "acp-controller"},
        }
        try:
            self._client.post(self._path(EVENT, ns), json=ev)
        except Exception:
            pass  # events are best-effort, like recorder.Event

    def events_for(self, involved_name: str, namespace: str = "default") -> List[Dict[str, Any]]:
        out = [
            e for e in self.list(EVENT, namespace)
            if e.get("involvedObject", {}).get("name") == involved_name
        ]
        out.sort(key=lambda e: str(e["metadata"].get("resourceVersion", "0")))
        return out

    # ---------------------------------------------------------------- leases

    def acquire_lease(self, name: str, holder: str, duration_s: float,
                      namespace: str = "default") -> bool:
        now = time.time()
        cur = self.get(LEASE, name, namespace)
        body = {
            "apiVersion": "coordination.k8s.io/v1",
            "kind": LEASE,
            "metadata": {"name": name, "namespace": namespace},
            "spec": {
                "holderIdentity": holder,
                "leaseDurationSeconds": int(duration_s),
                "renewTime": now,
            },
        }
        if cur is None:
            try:
                self._client.post(self._path(LEASE, namespace), json=body)
                return True
            except Exception:
                return False
        spec = cur.get("spec", {})
        expired = now - float(spec.get("renewTime", 0) or 0) > float(
            spec.get("leaseDurationSeconds", duration_s) or duration_s
        )
        if spec.get("holderIdentity") != holder and not expired:
            return False
        body["metadata"]["resourceVersion"] = cur["metadata"].get("resourceVersion")
        try:
            r = self._client.put(self._path(LEASE, namespace, name), json=body)
            return 200 <= r.status_code < 300
        except Exception:
            return False

    def release_lease(self, name: str, holder: str, namespace: str = "default") -> None:
        cur = self.get(LEASE, name, namespace)
        if cur is not None and cur.get("spec", {}).get("holderIdentity") == holder:
            self.delete(LEASE, name, namespace)

    # ------------------------------------------------------------- utilities

    def ensure_namespace(self, namespace: str) -> None:
        try:
            self._client.post("/api/v1/namespaces",
                              json={"apiVersion": "v1", "kind": "Namespace",
                                    "metadata": {"name": namespace}})
        except Exception:
            pass

    def stats(self) -> Dict[str, int]:
        return {k: len(self.list(k)) for k in ACP_KINDS}

    def close(self) -> None:
        self._stop = True
        self._client.close()
