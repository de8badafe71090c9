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

    @app.get("/apis/{kind}/{ns}/{name}")
    def get_obj(kind: str, ns: str, name: str):
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
