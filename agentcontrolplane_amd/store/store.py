"""Durable resource store with watches — the etcd/apiserver role.

The reference persists every resource and status transition to etcd through
the Kubernetes apiserver, and controllers react through informer watches
(acp/cmd/main.go:208-320 wires the manager's caches).  This
module is the MI355X-native equivalent: a process-local, thread-safe,
WAL-durable object store with

- optimistic concurrency via ``metadata.resourceVersion`` (update conflicts
  surface as ``ConflictError``, mirroring apiserver 409s that the reference's
  conflict-retry loops handle, e.g. agent/state_machine.go:162-204),
- label-filtered list (the reference lists ToolCalls by
  ``{task, toolcallrequest}`` labels, task/state_machine.go:291-341),
- watch streams (ADDED/MODIFIED/DELETED) delivered to subscriber queues —
  these drive the controllers' workqueues, replacing the reference's
  poll-with-RequeueAfter joins with watch-triggered reconciles,
- an append-only JSONL WAL + snapshot for crash recovery (etcd's durability
  role): every write is appended; ``load()`` replays.

Status is a subresource: ``update_status`` only replaces ``status`` (like
``client.Status().Update``), ``update`` only replaces spec/metadata.
"""
from __future__ import annotations

import dataclasses
import json
import os
import queue
import threading
import time
from typing import Any, Dict, Iterable, List, Optional, Tuple

from ..api.types import API_VERSION, EVENT, LEASE, new_object_meta, now_iso


def _copy(o):
    """Fast deep copy for JSON-shaped data (dict/list/scalars only).

    ``copy.deepcopy`` pays memo bookkeeping on every node; store objects are
    pure JSON trees, so a recursive rebuild is ~4x faster — this runs on
    every get/list/update under the store lock, so it is hot at 1k
    concurrent Tasks."""
    if isinstance(o, dict):
        return {k: _copy(v) for k, v in o.items()}
    if isinstance(o, list):
        return [_copy(v) for v in o]
    return o


class NotFoundError(KeyError):
    pass


class AlreadyExistsError(ValueError):
    pass


class ConflictError(RuntimeError):
    """resourceVersion mismatch — caller should re-get and retry."""


@dataclasses.dataclass
class WatchEvent:
    type: str  # ADDED | MODIFIED | DELETED
    kind: str
    obj: Dict[str, Any]


class _Watch:
    __slots__ = ("kinds", "q", "alive")

    def __init__(self, kinds: Optional[Iterable[str]]):
        self.kinds = set(kinds) if kinds else None
        self.q: "queue.Queue[WatchEvent]" = queue.Queue()
        self.alive = True


class ResourceStore:
    """Thread-safe namespaced object store with watches and a WAL.

    ``wal_path=None`` keeps everything in memory (unit tests, benchmarks that
    measure the non-durable floor); with a path, every mutation is appended as
    one JSON line and fsync'd per ``fsync`` policy ("always" | "interval" |
    "never").
    """

    def __init__(
        self,
        wal_path: Optional[str] = None,
        fsync: str = "interval",
        fsync_interval_s: float = 0.05,
        compact_bytes: int = 256 << 20,
    ):
        self._lock = threading.RLock()
        # kind -> namespace -> name -> obj
        self._data: Dict[str, Dict[str, Dict[str, Dict[str, Any]]]] = {}
        # owner uid -> [(kind, ns, name)] — O(children) cascade deletes
        self._owned_by: Dict[str, List[Tuple[str, str, str]]] = {}
        # per-namespace FIFO of event names for TTL-style capping
        self._event_fifo: Dict[str, List[str]] = {}
        self.max_events_per_namespace = 20000
        # O(1) event aggregation: (ns, involved, type, reason, message) ->
        # event name.  A linear bucket scan here cost ~tens of seconds per
        # 1k-task bench wave once the bucket filled (round-1 wave drift).
        self._event_index: Dict[Tuple, str] = {}
        # (ns, involved name) -> [event names] — events_for and delete-GC
        self._events_by_obj: Dict[Tuple[str, str], List[str]] = {}
        # serialized "put" lines for O(live) compaction without re-dumping
        self._snap_lines: Dict[Tuple[str, str, str], str] = {}
        self._rv = 0
        self._watches: List[_Watch] = []
        self._wal_path = wal_path
        self._wal_file = None
        self._fsync = fsync
        self._fsync_interval = fsync_interval_s
        self._last_fsync = 0.0
        # auto-compaction: rewrite the log as a snapshot once it exceeds
        # this size (etcd's compaction role); 0 disables
        self._compact_bytes = compact_bytes
        self._wal_written = 0
        if wal_path:
            os.makedirs(os.path.dirname(os.path.abspath(wal_path)), exist_ok=True)
            if os.path.exists(wal_path):
                self._replay(wal_path)
            self._wal_file = open(wal_path, "a", encoding="utf-8")

    # ------------------------------------------------------------------ WAL

    def _replay(self, path: str) -> None:
        good_end = 0  # byte offset after the last valid line
        with open(path, "rb") as f:
            raw = f.read()
        offset = 0
        # parse from the raw bytes: a non-UTF-8 byte in a torn/corrupt line
        # must be skipped like bad JSON, not abort recovery (text-mode
        # iteration would raise UnicodeDecodeError and lose everything)
        for raw_line in raw.split(b"\n"):
            raw_len = len(raw_line) + 1  # the split consumed the newline
            if offset + len(raw_line) >= len(raw):
                raw_len = len(raw_line)  # last fragment had no newline
            line_b = raw_line.strip()
            if not line_b:
                offset += raw_len
                good_end = offset
                continue
            try:
                rec = json.loads(line_b.decode("utf-8"))
            except (json.JSONDecodeError, UnicodeDecodeError):
                offset += raw_len
                continue  # torn tail write after a crash
            offset += raw_len
            good_end = offset
            op, obj = rec.get("op"), rec.get("obj")
            if not obj:
                continue
            kind = obj.get("kind")
            m = obj.get("metadata", {})
            ns, name = m.get("namespace", "default"), m.get("name")
            if not kind or not name:
                continue
            bucket = self._data.setdefault(kind, {}).setdefault(ns, {})
            if op == "delete":
                bucket.pop(name, None)
            else:
                bucket[name] = obj
            self._rv = max(self._rv, int(m.get("resourceVersion", 0)))
        # truncate away any torn tail so the next append starts a fresh
        # line (appending onto a torn fragment would corrupt BOTH records)
        if good_end < len(raw):
            with open(path, "r+b") as f:
                f.truncate(good_end)
        # rebuild the owner index from the replayed state
        self._owned_by.clear()
        for kind, nss in self._data.items():
            for ns, objs in nss.items():
                for name, obj in objs.items():
                    for ref in obj.get("metadata", {}).get("ownerReferences", []) or []:
                        if ref.get("uid"):
                            self._owned_by.setdefault(ref["uid"], []).append(
                                (kind, ns, name)
                            )
        # rebuild the event aggregation/involved-object indexes
        for ns, objs in self._data.get(EVENT, {}).items():
            ordered = sorted(
                objs.items(), key=lambda kv: int(kv[1]["metadata"].get("resourceVersion", 0))
            )
            fifo = self._event_fifo.setdefault(ns, [])
            for name, ev in ordered:
                io = ev.get("involvedObject", {})
                self._event_index[
                    (ns, io.get("name"), ev.get("type"), ev.get("reason"), ev.get("message"))
                ] = name
                self._events_by_obj.setdefault((ns, io.get("name")), []).append(name)
                fifo.append(name)

    def _append_wal(self, op: str, obj: Dict[str, Any]) -> None:
        if self._wal_file is None:
            return
        line = json.dumps({"op": op, "obj": obj}, separators=(",", ":")) + "\n"
        key = (
            obj.get("kind", ""),
            obj.get("metadata", {}).get("namespace", "default"),
            obj.get("metadata", {}).get("name", ""),
        )
        if op == "put":
            self._snap_lines[key] = line
        else:
            self._snap_lines.pop(key, None)
        self._wal_file.write(line)
        self._wal_written += len(line)
        if self._compact_bytes and self._wal_written > self._compact_bytes:
            self._wal_written = 0
            self.compact()
            return
        self._wal_file.flush()
        if self._fsync == "always":
            os.fsync(self._wal_file.fileno())
        elif self._fsync == "interval":
            now = time.monotonic()
            if now - self._last_fsync >= self._fsync_interval:
                os.fsync(self._wal_file.fileno())
                self._last_fsync = now

    def compact(self) -> None:
        """Rewrite the WAL as one snapshot line per live object.

        Uses the serialized-line cache where possible: round 1's compaction
        re-dumped every live object (plus fsync) while holding the store
        lock — a multi-second stall that showed up as episodic 2x wave-time
        spikes in the driver bench."""
        if not self._wal_path:
            return
        with self._lock:
            tmp = self._wal_path + ".tmp"
            with open(tmp, "w", encoding="utf-8") as f:
                for kind, nss in self._data.items():
                    for ns, objs in nss.items():
                        for name, obj in objs.items():
                            line = self._snap_lines.get((kind, ns, name))
                            if line is None:
                                line = (
                                    json.dumps(
                                        {"op": "put", "obj": obj}, separators=(",", ":")
                                    )
                                    + "\n"
                                )
                            f.write(line)
                f.flush()
                os.fsync(f.fileno())
            if self._wal_file:
                self._wal_file.close()
            os.replace(tmp, self._wal_path)
            self._wal_file = open(self._wal_path, "a", encoding="utf-8")

    def close(self) -> None:
        with self._lock:
            if self._wal_file:
                self._wal_file.close()
                self._wal_file = None

    # ------------------------------------------------------------ primitives

    def _notify(self, ev: WatchEvent) -> None:
        for w in self._watches:
            if w.alive and (w.kinds is None or ev.kind in w.kinds):
                w.q.put(ev)

    def create(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        obj = _copy(obj)
        kind = obj["kind"]
        m = obj.setdefault("metadata", {})
        ns = m.setdefault("namespace", "default")
        name = m["name"]
        if "uid" not in m or "creationTimestamp" not in m:
            base = new_object_meta(name, ns)
            m.setdefault("uid", base["uid"])
            m.setdefault("creationTimestamp", base["creationTimestamp"])
        obj.setdefault("apiVersion", API_VERSION)
        obj.setdefault("spec", {})
        obj.setdefault("status", {})
        with self._lock:
            bucket = self._data.setdefault(kind, {}).setdefault(ns, {})
            if name in bucket:
                raise AlreadyExistsError(f"{kind} {ns}/{name} already exists")
            self._rv += 1
            m["resourceVersion"] = self._rv
            m["generation"] = 1
            bucket[name] = obj
            for ref in m.get("ownerReferences", []) or []:
                if ref.get("uid"):
                    self._owned_by.setdefault(ref["uid"], []).append((kind, ns, name))
            self._append_wal("put", obj)
            out = _copy(obj)
            self._notify(WatchEvent("ADDED", kind, out))
        return out

    def get(self, kind: str, name: str, namespace: str = "default") -> Optional[Dict[str, Any]]:
        with self._lock:
            obj = self._data.get(kind, {}).get(namespace, {}).get(name)
            return _copy(obj) if obj is not None else None

    def list(
        self,
        kind: str,
        namespace: Optional[str] = "default",
        label_selector: Optional[Dict[str, str]] = None,
    ) -> List[Dict[str, Any]]:
        with self._lock:
            nss = self._data.get(kind, {})
            spaces = [namespace] if namespace is not None else list(nss.keys())
            out = []
            for ns in spaces:
                for obj in nss.get(ns, {}).values():
                    if label_selector:
                        labels = obj.get("metadata", {}).get("labels", {}) or {}
                        if any(labels.get(k) != v for k, v in label_selector.items()):
                            continue
                    out.append(_copy(obj))
            return out

    def _check_rv(self, current: Dict[str, Any], incoming: Dict[str, Any]) -> None:
        want = incoming.get("metadata", {}).get("resourceVersion")
        have = current.get("metadata", {}).get("resourceVersion")
        if want is not None and int(want) != int(have):
            raise ConflictError(
                f'{incoming.get("kind")} {incoming["metadata"].get("name")}: '
                f"resourceVersion conflict (have {have}, got {want})"
            )

    def update(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        """Replace spec + metadata (labels etc.); status is untouched."""
        return self._update(obj, which="main")

    def update_status(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        """Replace only status — the Status().Update subresource semantics."""
        return self._update(obj, which="status")

    def _update(self, obj: Dict[str, Any], which: str) -> Dict[str, Any]:
        kind = obj["kind"]
        m = obj.get("metadata", {})
        ns, name = m.get("namespace", "default"), m["name"]
        with self._lock:
            bucket = self._data.get(kind, {}).get(ns, {})
            cur = bucket.get(name)
            if cur is None:
                raise NotFoundError(f"{kind} {ns}/{name} not found")
            self._check_rv(cur, obj)
            if which == "status" and cur.get("status") == obj.get("status", {}):
                # no-op status write: like the apiserver, identical content
                # does not bump resourceVersion or emit a watch event —
                # without this, reconcilers that rewrite unchanged status
                # self-trigger forever through their own MODIFIED events
                return _copy(cur)
            self._rv += 1
            if which == "status":
                cur["status"] = _copy(obj.get("status", {}))
            else:
                cur["spec"] = _copy(obj.get("spec", {}))
                newm = _copy(obj.get("metadata", {}))
                keep = cur["metadata"]
                for k in ("labels", "annotations", "ownerReferences", "deletionTimestamp"):
                    if k in newm:
                        keep[k] = newm[k]
                    elif k in keep and k not in newm:
                        keep.pop(k, None)
                keep["generation"] = int(keep.get("generation", 1)) + 1
            cur["metadata"]["resourceVersion"] = self._rv
            self._append_wal("put", cur)
            out = _copy(cur)
            self._notify(WatchEvent("MODIFIED", kind, out))
        return out

    def delete(self, kind: str, name: str, namespace: str = "default") -> bool:
        with self._lock:
            bucket = self._data.get(kind, {}).get(namespace, {})
            obj = bucket.pop(name, None)
            if obj is None:
                return False
            self._append_wal("delete", obj)
            self._notify(WatchEvent("DELETED", kind, _copy(obj)))
            # cascade via the owner index (k8s GC role), O(children)
            uid = obj.get("metadata", {}).get("uid")
            if uid:
                for (k2, ns2, n2) in self._owned_by.pop(uid, []):
                    self.delete(k2, n2, ns2)
            # GC the object's events (k8s expires events by TTL; here the
            # involved object's deletion is the expiry signal — without
            # this, bench-style create/delete churn pins the event bucket
            # at its cap and every aggregation lookup degrades)
            if kind != EVENT:
                for ev_name in list(self._events_by_obj.get((namespace, name), [])):
                    self._drop_event(namespace, ev_name)
        return True

    # --------------------------------------------------------------- watches

    def watch(self, kinds: Optional[Iterable[str]] = None) -> "queue.Queue[WatchEvent]":
        w = _Watch(kinds)
        with self._lock:
            self._watches.append(w)
        return w.q

    def stop_watch(self, q: "queue.Queue[WatchEvent]") -> None:
        with self._lock:
            for w in self._watches:
                if w.q is q:
                    w.alive = False
            self._watches = [w for w in self._watches if w.alive]

    # ---------------------------------------------------------------- events

    def record_event(
        self,
        involved: Dict[str, Any],
        event_type: str,
        reason: str,
        message: str,
    ) -> None:
        """Kubernetes-Events equivalent — the user-facing execution history
        (the reference has ~40 recorder.Event call sites, e.g.
        task/state_machine.go:224,628,731)."""
        m = involved.get("metadata", {})
        ns = m.get("namespace", "default")
        ikey = (ns, m.get("name"), event_type, reason, message)
        with self._lock:
            # k8s-style event aggregation: an identical (object, reason,
            # message) event bumps count/lastTimestamp — O(1) via the index
            # (a linear bucket scan here cost tens of seconds per 1k-task
            # bench wave once the bucket filled)
            ev_name = self._event_index.get(ikey)
            if ev_name is not None:
                ev = self._data.get(EVENT, {}).get(ns, {}).get(ev_name)
                if ev is not None:
                    ev["count"] = int(ev.get("count", 1)) + 1
                    ev["lastTimestamp"] = now_iso()
                    # durable + visible: rv bump so watchers order it, WAL
                    # append so a crash does not lose the count
                    self._rv += 1
                    ev["metadata"]["resourceVersion"] = self._rv
                    self._append_wal("put", ev)
                    self._notify(WatchEvent("MODIFIED", EVENT, _copy(ev)))
                    return
                self._event_index.pop(ikey, None)
            name = f'{m.get("name", "obj")}.{self._rv + 1}'
            ev = {
                "apiVersion": "v1",
                "kind": EVENT,
                "metadata": new_object_meta(name, ns),
                "spec": {},
                "status": {},
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
            }
            bucket = self._data.setdefault(EVENT, {}).setdefault(ns, {})
            self._rv += 1
            ev["metadata"]["resourceVersion"] = self._rv
            bucket[name] = ev
            self._event_index[ikey] = name
            self._events_by_obj.setdefault((ns, m.get("name")), []).append(name)
            fifo = self._event_fifo.setdefault(ns, [])
            fifo.append(name)
            # TTL-style cap (k8s events expire after 1h; here: count-bound)
            while len(fifo) > self.max_events_per_namespace:
                self._drop_event(ns, fifo.pop(0))
            self._append_wal("put", ev)
            self._notify(WatchEvent("ADDED", EVENT, _copy(ev)))

    def _drop_event(self, ns: str, name: str) -> None:
        """Remove one event + its index entries (lock held)."""
        ev = self._data.get(EVENT, {}).get(ns, {}).pop(name, None)
        if ev is None:
            return
        io = ev.get("involvedObject", {})
        self._event_index.pop(
            (ns, io.get("name"), ev.get("type"), ev.get("reason"), ev.get("message")),
            None,
        )
        bucket = self._events_by_obj.get((ns, io.get("name")))
        if bucket is not None:
            try:
                bucket.remove(name)
            except ValueError:
                pass
            if not bucket:
                self._events_by_obj.pop((ns, io.get("name")), None)
        self._snap_lines.pop((EVENT, ns, name), None)

    def events_for(self, involved_name: str, namespace: str = "default") -> List[Dict[str, Any]]:
        with self._lock:
            bucket = self._data.get(EVENT, {}).get(namespace, {})
            out = [
                _copy(bucket[n])
                for n in self._events_by_obj.get((namespace, involved_name), [])
                if n in bucket
            ]
            out.sort(key=lambda e: int(e["metadata"].get("resourceVersion", 0)))
            return out

    # ---------------------------------------------------------------- leases

    def acquire_lease(
        self,
        name: str,
        holder: str,
        duration_s: float,
        namespace: str = "default",
    ) -> bool:
        """coordination.k8s.io Lease semantics for task locking
        (task/state_machine.go:1069-1145): acquire if absent, expired, or
        held by the same holder (renew).  Returns False when held by another
        live holder."""
        now = time.time()
        with self._lock:
            bucket = self._data.setdefault(LEASE, {}).setdefault(namespace, {})
            cur = bucket.get(name)
            if cur is not None:
                spec = cur.get("spec", {})
                expired = now - float(spec.get("renewTime", 0)) > float(
                    spec.get("leaseDurationSeconds", duration_s)
                )
                if spec.get("holderIdentity") != holder and not expired:
                    return False
            self._rv += 1
            lease = {
                "apiVersion": "coordination.k8s.io/v1",
                "kind": LEASE,
                "metadata": {
                    "name": name,
                    "namespace": namespace,
                    "resourceVersion": self._rv,
                },
                "spec": {
                    "holderIdentity": holder,
                    "leaseDurationSeconds": duration_s,
                    "renewTime": now,
                },
                "status": {},
            }
            bucket[name] = lease
            self._append_wal("put", lease)
            return True

    def release_lease(self, name: str, holder: str, namespace: str = "default") -> None:
        with self._lock:
            bucket = self._data.get(LEASE, {}).get(namespace, {})
            cur = bucket.get(name)
            if cur is not None and cur.get("spec", {}).get("holderIdentity") == holder:
                bucket.pop(name, None)
                self._append_wal("delete", cur)

    # ------------------------------------------------------------- utilities

    def ensure_namespace(self, namespace: str) -> None:
        # namespaces are implicit buckets; kept for API parity with createTask's
        # ensureNamespaceExists (server.go)
        return

    def stats(self) -> Dict[str, int]:
        with self._lock:
            return {
                kind: sum(len(objs) for objs in nss.values())
                for kind, nss in self._data.items()
            }
