"""CLI/admin surface: YAML manifests (the reference's CRD shapes) applied
through the admin endpoint drive the reconcilers; example script runs."""
import subprocess
import sys
import time

import pytest
import yaml
from fastapi.testclient import TestClient

from agentcontrolplane_amd.api.types import AGENT, LLM, TASK, TaskPhase
from agentcontrolplane_amd.runtime import ControlPlane
from agentcontrolplane_amd.server.admin import add_admin_routes

from conftest import wait_for


@pytest.fixture
def cp():
    plane = ControlPlane(auto_approve="approve")
    plane.start()
    app = plane.rest_app
    add_admin_routes(app, plane.store)
    yield plane
    plane.stop()


MANIFESTS = """
apiVersion: acp.humanlayer.dev/v1alpha1
kind: LLM
metadata:
  name: yaml-llm
spec:
  provider: mock
---
apiVersion: acp.humanlayer.dev/v1alpha1
kind: Agent
metadata:
  name: yaml-agent
spec:
  llmRef:
    name: yaml-llm
  system: from yaml
---
apiVersion: acp.humanlayer.dev/v1alpha1
kind: Task
metadata:
  name: yaml-task
spec:
  agentRef:
    name: yaml-agent
  userMessage: "hello from a manifest"
"""


def test_apply_yaml_manifests_runs_loop(cp):
    client = TestClient(cp.rest_app)
    for doc in yaml.safe_load_all(MANIFESTS):
        r = client.post("/admin/resources", json=doc)
        assert r.status_code == 201, r.text
    task = wait_for(
        lambda: (cp.store.get(TASK, "yaml-task") or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER
        and cp.store.get(TASK, "yaml-task"),
        timeout=30,
    )
    assert task["status"]["output"] == "mock final answer"
    # upsert: re-apply with a changed system prompt → 200 configured
    doc = list(yaml.safe_load_all(MANIFESTS))[1]
    doc["spec"]["system"] = "updated"
    r = client.post("/admin/resources", json=doc)
    assert r.status_code == 200
    assert cp.store.get(AGENT, "yaml-agent")["spec"]["system"] == "updated"
    # admin reads (plural + lowercase forms)
    assert client.get("/admin/resources/tasks").json()
    assert client.get("/admin/resources/Task/yaml-task").status_code == 200
    assert client.get("/admin/events/yaml-task").json()


def test_sample_manifests_parse():
    import glob

    files = glob.glob("config/samples/*.yaml")
    assert files, "sample manifests missing"
    for f in files:
        for doc in yaml.safe_load_all(open(f)):
            assert doc.get("apiVersion") and doc.get("kind") and doc["metadata"]["name"]


def test_example_script_runs():
    out = subprocess.run(
        [sys.executable, "examples/simple_agent.py"],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stderr
    assert "FinalAnswer" in out.stdout


def test_admin_delete_cascades(cp):
    """DELETE /admin/resources/<kind>/<name> removes the object and its
    owned children (the store's owner index)."""
    client = TestClient(cp.rest_app)
    for doc in yaml.safe_load_all(MANIFESTS):
        client.post("/admin/resources", json=doc)
    wait_for(
        lambda: (cp.store.get(TASK, "yaml-task") or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER or None,
        timeout=20,
    )
    r = client.delete("/admin/resources/Task/yaml-task")
    assert r.status_code == 200 and r.json()["deleted"] == "Task/yaml-task"
    assert cp.store.get(TASK, "yaml-task") is None
    r = client.delete("/admin/resources/Task/yaml-task")
    assert r.status_code == 404


def test_admin_rejects_malformed(cp):
    client = TestClient(cp.rest_app)
    r = client.post("/admin/resources", json={"kind": "Task"})  # no metadata.name
    assert r.status_code == 400
    r = client.get("/admin/resources/NoSuchKind")
    assert r.status_code in (200, 404)  # empty list or not-found, never 500
    r = client.delete("/admin/resources/Task/never-existed")
    assert r.status_code == 404


def test_serve_subprocess_boots_and_runs_loop(tmp_path):
    """The actual `python -m agentcontrolplane_amd serve` CLI path: boot on
    a free port with no engine, drive the mock-LLM loop over HTTP."""
    import socket

    import httpx

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    proc = subprocess.Popen(
        [sys.executable, "-m", "agentcontrolplane_amd", "serve",
         "--port", str(port), "--device", "none", "--auto-approve",
         "--wal", str(tmp_path / "wal.jsonl")],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
    )
    try:
        client = httpx.Client(base_url=f"http://127.0.0.1:{port}", timeout=20)
        deadline = time.time() + 30
        while time.time() < deadline:
            try:
                if client.get("/status").status_code == 200:
                    break
            except Exception:
                time.sleep(0.2)
        else:
            raise AssertionError("serve never came up")
        for doc in yaml.safe_load_all(MANIFESTS):
            r = client.post("/admin/resources", json=doc)
            assert r.status_code == 201, r.text
        deadline = time.time() + 30
        while time.time() < deadline:
            t = client.get("/admin/resources/Task/yaml-task").json()
            if t.get("status", {}).get("phase") == "FinalAnswer":
                break
            time.sleep(0.25)
        else:
            raise AssertionError(f"task never finished: {t.get('status', {})}")
        assert t["status"]["output"] == "mock final answer"
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=5)
        except subprocess.TimeoutExpired:
            proc.kill()
