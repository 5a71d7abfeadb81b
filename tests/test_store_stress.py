"""Concurrency stress over the store: mixed mutators + compaction + a
watch consumer running simultaneously, then invariant checks — the
sanitizer-tier coverage for the hand-rolled locking (SURVEY §5 race
detection; the reference leans on -race + envtest, neither exists
here)."""
from __future__ import annotations

import os
import queue
import random
import threading

from agentcontrolplane_amd.api.types import make_resource
from agentcontrolplane_amd.store import (
    AlreadyExistsError,
    ConflictError,
    NotFoundError,
    ResourceStore,
)


def test_store_stress_with_compaction(tmp_path):
    wal = str(tmp_path / "stress-wal.jsonl")
    # tiny compaction threshold: dozens of compactions during the run
    store = ResourceStore(wal_path=wal, compact_bytes=64 << 10)
    stop = threading.Event()
    errors = []
    created_total = [0]
    deleted_total = [0]
    lock = threading.Lock()

    watch_q = store.watch(kinds={"Task"})
    seen_events = [0]

    def watcher():
        while not stop.is_set():
            try:
                watch_q.get(timeout=0.05)
                seen_events[0] += 1
            except queue.Empty:
                pass

    def worker(wid: int):
        rng = random.Random(wid)
        mine = []
        try:
            for i in range(150):
                op = rng.random()
                if op < 0.45 or not mine:
                    name = f"t-{wid}-{i}"
                    store.create(make_resource(
                        "Task", name,
                        spec={"agentRef": {"name": "a"}, "userMessage": "x" * 200},
                    ))
                    store.record_event(
                        {"kind": "Task", "metadata": {"name": name,
                                                      "namespace": "default"}},
                        "Normal", "Created", f"by {wid}",
                    )
                    mine.append(name)
                    with lock:
                        created_total[0] += 1
                elif op < 0.75:
                    name = rng.choice(mine)
                    obj = store.get("Task", name)
                    if obj is None:
                        continue
                    obj["status"] = {"phase": "Pending",
                                     "statusDetail": f"u{i}"}
                    try:
                        store.update_status(obj)
                    except (ConflictError, NotFoundError):
                        pass  # expected races
                elif op < 0.9:
                    store.list("Task", label_selector=None)
                    store.events_for(rng.choice(mine))
                else:
                    name = mine.pop(rng.randrange(len(mine)))
                    if store.delete("Task", name):
                        with lock:
                            deleted_total[0] += 1
        except Exception as e:  # noqa: BLE001
            errors.append((wid, repr(e)))

    wt = threading.Thread(target=watcher, daemon=True)
    wt.start()
    threads = [threading.Thread(target=worker, args=(w,)) for w in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    stop.set()
    wt.join(timeout=5)

    assert not errors, errors[:3]
    live = store.list("Task")
    assert len(live) == created_total[0] - deleted_total[0]
    assert seen_events[0] > 0
    # events for deleted tasks were GC'd; live tasks kept theirs
    live_names = {o["metadata"]["name"] for o in live}
    for o in live[:20]:
        evs = store.events_for(o["metadata"]["name"])
        assert evs and evs[0]["reason"] == "Created"
    for e in store.list("Event"):
        assert e["involvedObject"]["name"] in live_names

    # crash-replay equivalence after all that churn + compactions
    store.close()
    assert os.path.getsize(wal) > 0
    replayed = ResourceStore(wal_path=wal)
    live2 = replayed.list("Task")
    assert {o["metadata"]["name"] for o in live2} == live_names
    # statuses survive too
    by_name = {o["metadata"]["name"]: o for o in live2}
    for o in live:
        assert by_name[o["metadata"]["name"]]["status"] == o["status"]
    replayed.close()
