"""API server — HTTP exposure of the object store with the admission
chain on the write path (the kube-apiserver role for out-of-process
clients: vcctl, remote controllers, dashboards).

Endpoints:
  GET    /apis/{kind}                     list (optionally ?namespace=)
  GET    /apis/{kind}/{ns}/{name}         get
  POST   /apis/{kind}                     create (admission-checked)
  PUT    /apis/{kind}/{ns}/{name}         update (admission-checked)
  DELETE /apis/{kind}/{ns}/{name}         delete
  GET    /watch?since=RV&kinds=a,b        event journal replay (long-poll)
  GET    /healthz, /metrics
"""

from __future__ import annotations

import threading
from typing import Optional

from ..api.objects import KINDS, from_dict, to_dict
from ..utils.metrics import METRICS
from ..webhooks import AdmissionError, default_chain
from .store import ObjectStore


def create_app(store: ObjectStore, with_admission: bool = True):
    from fastapi import Body, FastAPI, HTTPException
    from fastapi.responses import PlainTextResponse

    chain = default_chain(store) if with_admission else None
    app = FastAPI(title="volcano-amd apiserver")

    def _kind(kind: str):
        cls = KINDS.get(kind)
        if cls is None:
            raise HTTPException(404, f"unknown kind {kind}")
        return cls

    @app.get("/healthz")
    def healthz():
        return {"ok": True, "resourceVersion": store.resource_version}

    @app.get("/metrics", response_class=PlainTextResponse)
    def metrics():
        return METRICS.export_prometheus()

    @app.get("/apis/{kind}")
    def list_objs(kind: str, namespace: Optional[str] = None):
        _kind(kind)
        return {"items": [to_dict(o) for o in store.list(kind, namespace)]}

    def _legacy_get(kind: str, ns: str, name: str):
        _kind(kind)
        obj = store.get(kind, ns, name)
        if obj is None:
            raise HTTPException(404, f"{kind} {ns}/{name} not found")
        return to_dict(obj)

    @app.post("/apis/{kind}")
    def create_obj(kind: str, data: dict = Body(...)):
        obj = from_dict(_kind(kind), data)
        try:
            if chain is not None:
                chain.admit(kind, obj, "CREATE")
            store.create(kind, obj)
        except AdmissionError as e:
            raise HTTPException(400, f"admission denied: {e}")
        except KeyError as e:
            raise HTTPException(409, str(e))
        return to_dict(obj)

    @app.put("/apis/{kind}/{ns}/{name}")
    def update_obj(kind: str, ns: str, name: str, data: dict = Body(...)):
        obj = from_dict(_kind(kind), data)
        try:
            if chain is not None:
                chain.admit(kind, obj, "UPDATE")
            store.update(kind, obj)
        except AdmissionError as e:
            raise HTTPException(400, f"admission denied: {e}")
        except KeyError as e:
            raise HTTPException(404, str(e))
        return to_dict(obj)

    @app.delete("/apis/{kind}/{ns}/{name}")
    def delete_obj(kind: str, ns: str, name: str):
        _kind(kind)
        obj = store.delete(kind, ns, name)
        if obj is None:
            raise HTTPException(404, f"{kind} {ns}/{name} not found")
        return {"deleted": True}

    @app.get("/watch")
    def watch(since: int = 0, kinds: str = ""):
        ks = tuple(k for k in kinds.split(",") if k) or None
        evs = store.journal_since(since, ks)
        return {"resourceVersion": store.resource_version,
                "events": [{"rv": v, "type": t, "kind": k, "object": o}
                           for (v, t, k, o) in evs]}

    # ---- Kubernetes-style group/version paths --------
    # /apis/{group}/{version}/[namespaces/{ns}/]{plural}[/{name}]
    # plus /api/v1/... for core kinds; ?watch=1 streams newline-delimited
    # JSON watch events (the k8s chunked-watch shape).
    import json as _json

    from fastapi.responses import StreamingResponse

    from .k8s import kind_for, to_manifest, from_manifest, api_version

    def _k8s_kind(group: str, version: str, plural: str) -> str:
        kind = kind_for(group, version, plural)
        if kind is None:
            raise HTTPException(
                404, f"unknown resource {group}/{version}/{plural}")
        return kind

    def _k8s_list(kind: str, ns, watch, since):
        if watch:
            return _watch_stream(kind, since)
        items = [to_manifest(o) for o in store.list(kind, ns)]
        return {"apiVersion": api_version(kind), "kind": f"{kind}List",
                "metadata": {"resourceVersion":
                             str(store.resource_version)},
                "items": items}

    def _watch_stream(kind: str, since: int):
        def gen():
            rv = since          # 0 = full journal replay (k8s rv="0")
            import time as _t
            deadline = _t.monotonic() + 30.0       # bounded stream
            while _t.monotonic() < deadline:
                evs = store.journal_since(rv, (kind,))
                for (v, t, k, o) in evs:
                    rv = max(rv, v)
                    typ = {"ADDED": "ADDED", "MODIFIED": "MODIFIED",
                           "DELETED": "DELETED"}.get(t, t)
                    try:
                        man = to_manifest(from_manifest(o, k))
                    except Exception:
                        man = o
                    yield _json.dumps({"type": typ, "object": man}) + "\n"
                if evs:
                    continue
                _t.sleep(0.05)
        return StreamingResponse(gen(), media_type="application/json")

    def _k8s_get(kind: str, ns: str, name: str):
        obj = store.get(kind, ns, name)
        if obj is None:
            raise HTTPException(404, f"{kind} {ns}/{name} not found")
        return to_manifest(obj)

    def _k8s_create(kind: str, ns: Optional[str], data: dict):
        try:
            obj = from_manifest(data, kind)
        except (KeyError, TypeError, ValueError) as e:
            raise HTTPException(400, f"bad manifest: {e}")
        if ns and not obj.meta.namespace:
            obj.meta.namespace = ns
        try:
            if chain is not None:
                chain.admit(kind, obj, "CREATE")
            store.create(kind, obj)
        except AdmissionError as e:
            raise HTTPException(400, f"admission denied: {e}")
        except KeyError as e:
            raise HTTPException(409, str(e))
        return to_manifest(obj)

    def _k8s_update(kind: str, ns: str, name: str, data: dict):
        try:
            obj = from_manifest(data, kind)
        except (KeyError, TypeError, ValueError) as e:
            raise HTTPException(400, f"bad manifest: {e}")
        try:
            if chain is not None:
                chain.admit(kind, obj, "UPDATE")
            store.update(kind, obj)
        except AdmissionError as e:
            raise HTTPException(400, f"admission denied: {e}")
        except KeyError as e:
            raise HTTPException(404, str(e))
        return to_manifest(obj)

    def _k8s_delete(kind: str, ns: str, name: str):
        obj = store.delete(kind, ns, name)
        if obj is None:
            raise HTTPException(404, f"{kind} {ns}/{name} not found")
        return {"status": "Success", "kind": kind,
                "details": {"name": name}}

    @app.get("/apis")
    def api_groups():
        from .k8s import GVK
        groups = sorted({g for (g, _, _) in GVK.values() if g})
        return {"kind": "APIGroupList",
                "groups": [{"name": g} for g in groups]}

    @app.get("/apis/{group}/{version}/{plural}")
    def k8s_list_cluster(group: str, version: str, plural: str,
                         watch: int = 0, since: int = 0):
        # legacy shape shares this arity: /apis/{kind}/{ns}/{name}
        if group in KINDS:
            return _legacy_get(group, version, plural)
        return _k8s_list(_k8s_kind(group, version, plural), None,
                         watch, since)

    @app.post("/apis/{group}/{version}/{plural}")
    def k8s_create_cluster(group: str, version: str, plural: str,
                           data: dict = Body(...)):
        return _k8s_create(_k8s_kind(group, version, plural), None, data)

    @app.get("/apis/{group}/{version}/namespaces/{ns}/{plural}")
    def k8s_list_ns(group: str, version: str, plural: str, ns: str,
                    watch: int = 0, since: int = 0):
        return _k8s_list(_k8s_kind(group, version, plural), ns, watch, since)

    @app.post("/apis/{group}/{version}/namespaces/{ns}/{plural}")
    def k8s_create_ns(group: str, version: str, plural: str, ns: str,
                      data: dict = Body(...)):
        return _k8s_create(_k8s_kind(group, version, plural), ns, data)

    @app.get("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}")
    def k8s_get(group: str, version: str, plural: str, ns: str, name: str):
        return _k8s_get(_k8s_kind(group, version, plural), ns, name)

    @app.put("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}")
    def k8s_put(group: str, version: str, plural: str, ns: str, name: str,
                data: dict = Body(...)):
        return _k8s_update(_k8s_kind(group, version, plural), ns, name, data)

    @app.delete("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}")
    def k8s_del(group: str, version: str, plural: str, ns: str, name: str):
        return _k8s_delete(_k8s_kind(group, version, plural), ns, name)

    # core group (/api/v1): Pod/Node/PV/PVC/ResourceQuota
    @app.get("/api/v1/{plural}")
    def core_list(plural: str, watch: int = 0, since: int = 0):
        return _k8s_list(_k8s_kind("", "v1", plural), None, watch, since)

    @app.post("/api/v1/{plural}")
    def core_create(plural: str, data: dict = Body(...)):
        return _k8s_create(_k8s_kind("", "v1", plural), None, data)

    @app.get("/api/v1/namespaces/{ns}/{plural}")
    def core_list_ns(plural: str, ns: str, watch: int = 0, since: int = 0):
        return _k8s_list(_k8s_kind("", "v1", plural), ns, watch, since)

    @app.post("/api/v1/namespaces/{ns}/{plural}")
    def core_create_ns(plural: str, ns: str, data: dict = Body(...)):
        return _k8s_create(_k8s_kind("", "v1", plural), ns, data)

    @app.get("/api/v1/namespaces/{ns}/{plural}/{name}")
    def core_get(plural: str, ns: str, name: str):
        return _k8s_get(_k8s_kind("", "v1", plural), ns, name)

    @app.put("/api/v1/namespaces/{ns}/{plural}/{name}")
    def core_put(plural: str, ns: str, name: str, data: dict = Body(...)):
        return _k8s_update(_k8s_kind("", "v1", plural), ns, name, data)

    @app.delete("/api/v1/namespaces/{ns}/{plural}/{name}")
    def core_del(plural: str, ns: str, name: str):
        return _k8s_delete(_k8s_kind("", "v1", plural), ns, name)

    return app


def serve(store: ObjectStore, host: str = "127.0.0.1", port: int = 8343,
          background: bool = True):
    import uvicorn
    app = create_app(store)
    config = uvicorn.Config(app, host=host, port=port, log_level="warning")
    server = uvicorn.Server(config)
    if background:
        t = threading.Thread(target=server.run, daemon=True)
        t.start()
        return server
    server.run()
