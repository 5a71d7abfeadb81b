"""Admin resource endpoints — the apiserver surface the CLI's ``apply`` and
``get`` use.  Accepts the reference's CRD YAML/JSON shapes verbatim
(apiVersion acp.humanlayer.dev/v1alpha1, kind/metadata/spec), upserting
into the store; reconcilers react through watches exactly as if the object
came from kubectl."""
from __future__ import annotations

import json

from fastapi import Request
from fastapi.responses import JSONResponse

from ..api.types import KINDS


def add_admin_routes(app, store) -> None:
    @app.post("/admin/resources")
    async def upsert_resource(request: Request):
        doc = json.loads(await request.body())
        kind = doc.get("kind")
        name = doc.get("metadata", {}).get("name")
        if not kind or not name:
            return JSONResponse({"error": "kind and metadata.name required"}, status_code=400)
        ns = doc.get("metadata", {}).get("namespace", "default")
        doc.setdefault("spec", {})
        doc.setdefault("status", {})
        existing = store.get(kind, name, ns)
        if existing is None:
            store.create(doc)
            return JSONResponse({"created": f"{kind}/{name}"}, status_code=201)
        existing["spec"] = doc["spec"]
        if doc.get("metadata", {}).get("labels") is not None:
            existing["metadata"]["labels"] = doc["metadata"]["labels"]
        store.update(existing)
        return {"configured": f"{kind}/{name}"}

    @app.get("/admin/resources/{kind}")
    def list_resources(kind: str, namespace: str = "default"):
        kind = _canonical(kind)
        return store.list(kind, namespace)

    @app.get("/admin/resources/{kind}/{name}")
    def get_resource(kind: str, name: str, namespace: str = "default"):
        kind = _canonical(kind)
        obj = store.get(kind, name, namespace)
        if obj is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        return obj

    @app.delete("/admin/resources/{kind}/{name}")
    def delete_resource(kind: str, name: str, namespace: str = "default"):
        k = _canonical(kind)
        if store.get(k, name, namespace) is None:
            return JSONResponse({"error": f"{k} {name} not found"}, status_code=404)
        store.delete(k, name, namespace)
        return {"deleted": f"{k}/{name}"}

    @app.get("/admin/events/{name}")
    def get_events(name: str, namespace: str = "default"):
        return store.events_for(name, namespace)


def _canonical(kind: str) -> str:
    lower = {k.lower(): k for k in KINDS}
    lower.update({k.lower() + "s": k for k in KINDS})
    return lower.get(kind.lower(), kind)
