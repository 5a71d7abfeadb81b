"""Controller manager: workqueues + watch wiring.

The reference is a controller-runtime Manager (acp/cmd/main.go:208-320): each
reconciler gets a rate-limited workqueue fed by informer watches, plus
``Owns()`` mappings that requeue the owner when an owned object changes.
This module is the native equivalent:

- one ``_WorkQueue`` per reconciler: deduplicating, with delayed requeue
  (``Result(requeue_after=…)``) via a heap timer,
- a single store watch thread that routes ADDED/MODIFIED/DELETED events to
  the owning controller's queue *and* to controllers that declared an
  ``owns`` mapping (e.g. the Task controller owns ToolCalls: a ToolCall
  status flip immediately requeues the parent Task — replacing the
  reference's 5-second polling joins, the dominant latency term in
  BASELINE.md's control-plane floor),
- N worker threads per reconciler, panic-isolated; a reconcile error
  requeues with backoff like controller-runtime's default rate limiter.
"""
from __future__ import annotations

import dataclasses
import heapq
import logging
import threading
import time
import traceback
from typing import Dict, List, Optional, Tuple

from ..store import ConflictError, NotFoundError, ResourceStore, WatchEvent

logger = logging.getLogger("acp.manager")


@dataclasses.dataclass
class Result:
    requeue: bool = False
    requeue_after: float = 0.0  # seconds


class Reconciler:
    """Base reconciler: subclasses set ``kind`` and implement ``reconcile``."""

    kind: str = ""
    #: kinds whose events map to this controller's queue through ``map_owned``
    owns: Tuple[str, ...] = ()
    workers: int = 2

    def __init__(self, store: ResourceStore, recorder=None, **kwargs):
        self.store = store

    def reconcile(self, name: str, namespace: str) -> Result:
        raise NotImplementedError

    def map_owned(self, ev: WatchEvent) -> Optional[Tuple[str, str]]:
        """Map an owned object's event to (name, namespace) of *this* kind's
        object to requeue.  Default: follow the controller ownerReference."""
        for ref in ev.obj.get("metadata", {}).get("ownerReferences", []) or []:
            if ref.get("kind") == self.kind:
                return ref["name"], ev.obj["metadata"].get("namespace", "default")
        return None


class _WorkQueue:
    """Deduplicating delay-capable workqueue (controller-runtime's role)."""

    def __init__(self) -> None:
        self._cv = threading.Condition()
        self._ready: List[Tuple[str, str]] = []
        self._ready_set: set = set()
        self._delayed: List[Tuple[float, Tuple[str, str]]] = []
        self._in_flight: set = set()
        self._dirty_while_in_flight: set = set()
        self._shutdown = False

    def add(self, key: Tuple[str, str]) -> None:
        with self._cv:
            if key in self._in_flight:
                # controller-runtime semantics: never reconcile one key on two
                # workers at once — remember the event, requeue at done()
                self._dirty_while_in_flight.add(key)
            elif key not in self._ready_set:
                self._ready.append(key)
                self._ready_set.add(key)
            self._cv.notify()

    def add_after(self, key: Tuple[str, str], delay: float) -> None:
        if delay <= 0:
            self.add(key)
            return
        with self._cv:
            heapq.heappush(self._delayed, (time.monotonic() + delay, key))
            self._cv.notify()

    def get(self, timeout: float = 0.5) -> Optional[Tuple[str, str]]:
        with self._cv:
            deadline = time.monotonic() + timeout
            while True:
                now = time.monotonic()
                while self._delayed and self._delayed[0][0] <= now:
                    _, key = heapq.heappop(self._delayed)
                    if key in self._in_flight:
                        self._dirty_while_in_flight.add(key)
                    elif key not in self._ready_set:
                        self._ready.append(key)
                        self._ready_set.add(key)
                if self._ready:
                    key = self._ready.pop(0)
                    self._ready_set.discard(key)
                    self._in_flight.add(key)
                    return key
                if self._shutdown:
                    return None
                wait = deadline - now
                if self._delayed:
                    wait = min(wait, self._delayed[0][0] - now)
                if wait <= 0:
                    return None
                self._cv.wait(wait)

    def done(self, key: Tuple[str, str]) -> None:
        """Mark a key's reconcile finished; requeue if events arrived meanwhile."""
        with self._cv:
            self._in_flight.discard(key)
            if key in self._dirty_while_in_flight:
                self._dirty_while_in_flight.discard(key)
                if key not in self._ready_set:
                    self._ready.append(key)
                    self._ready_set.add(key)
                self._cv.notify()

    def shutdown(self) -> None:
        with self._cv:
            self._shutdown = True
            self._cv.notify_all()

    def __len__(self) -> int:
        with self._cv:
            return len(self._ready) + len(self._delayed)


class ControllerManager:
    """Runs reconcilers against a store — the controller-runtime Manager role."""

    def __init__(self, store: ResourceStore):
        self.store = store
        self._reconcilers: List[Reconciler] = []
        self._queues: Dict[str, _WorkQueue] = {}
        self._threads: List[threading.Thread] = []
        self._watch_q = None
        self._running = False
        self._error_backoff: Dict[Tuple[str, str, str], float] = {}

    def register(self, rec: Reconciler) -> None:
        rec._manager = self  # back-ref for async completion requeues
        self._reconcilers.append(rec)
        self._queues[rec.kind] = _WorkQueue()

    def start(self) -> None:
        self._running = True
        # seed: reconcile everything that already exists (informer initial list)
        for rec in self._reconcilers:
            for obj in self.store.list(rec.kind, namespace=None):
                self._queues[rec.kind].add(
                    (obj["metadata"]["name"], obj["metadata"].get("namespace", "default"))
                )
        self._watch_q = self.store.watch()  # all kinds
        t = threading.Thread(target=self._watch_loop, name="acp-watch", daemon=True)
        t.start()
        self._threads.append(t)
        for rec in self._reconcilers:
            for i in range(rec.workers):
                t = threading.Thread(
                    target=self._worker_loop,
                    args=(rec,),
                    name=f"acp-{rec.kind.lower()}-{i}",
                    daemon=True,
                )
                t.start()
                self._threads.append(t)

    def stop(self) -> None:
        self._running = False
        for q in self._queues.values():
            q.shutdown()
        if self._watch_q is not None:
            self.store.stop_watch(self._watch_q)
        for t in self._threads:
            t.join(timeout=3.0)
        self._threads.clear()

    # ------------------------------------------------------------------ loops

    def _watch_loop(self) -> None:
        while self._running:
            try:
                ev = self._watch_q.get(timeout=0.25)
            except Exception:
                continue
            if ev is None:
                continue
            try:
                self._route(ev)
            except Exception:
                logger.error("watch routing failed:\n%s", traceback.format_exc())

    def _route(self, ev: WatchEvent) -> None:
        name = ev.obj.get("metadata", {}).get("name")
        ns = ev.obj.get("metadata", {}).get("namespace", "default")
        if not name:
            return
        for rec in self._reconcilers:
            if rec.kind == ev.kind:
                if ev.type != "DELETED":
                    self._queues[rec.kind].add((name, ns))
                else:
                    hook = getattr(rec, "on_deleted", None)
                    if hook is not None:
                        hook(name, ns)
            if ev.kind in rec.owns:
                mapped = rec.map_owned(ev)
                if mapped is not None:
                    self._queues[rec.kind].add(mapped)

    def _worker_loop(self, rec: Reconciler) -> None:
        q = self._queues[rec.kind]
        while self._running:
            key = q.get(timeout=0.5)
            if key is None:
                continue
            name, ns = key
            try:
                res = rec.reconcile(name, ns)
                self._error_backoff.pop((rec.kind, name, ns), None)
                q.done(key)
                if res is not None and (res.requeue or res.requeue_after > 0):
                    q.add_after(key, res.requeue_after)
            except NotFoundError:
                # expected race: the object was deleted while a reconcile
                # was in flight — the reference's client.IgnoreNotFound
                # (e.g. task_controller.go Reconcile's Get error path);
                # drop the key, no traceback, no requeue
                self._error_backoff.pop((rec.kind, name, ns), None)
                q.done(key)
            except ConflictError:
                # optimistic-concurrency conflict: re-read and retry, the
                # apiserver-409 path the reference handles with
                # RetryOnConflict loops — requeue quickly, quietly
                q.done(key)
                q.add_after(key, 0.01)
            except Exception:
                logger.error(
                    "reconcile %s %s/%s failed:\n%s", rec.kind, ns, name, traceback.format_exc()
                )
                bk = self._error_backoff.get((rec.kind, name, ns), 0.005)
                bk = min(bk * 2, 8.0)
                self._error_backoff[(rec.kind, name, ns)] = bk
                q.done(key)
                q.add_after(key, bk)

    # ------------------------------------------------------------------ util

    def enqueue(self, kind: str, name: str, namespace: str = "default") -> None:
        self._queues[kind].add((name, namespace))

    def queue_depths(self) -> Dict[str, int]:
        return {k: len(q) for k, q in self._queues.items()}
