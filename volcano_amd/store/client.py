"""HTTP client for the apiserver — ObjectStore-shaped surface for
out-of-process components (vcctl, remote controllers)."""

from __future__ import annotations

from typing import Callable, List, Optional

from ..api.objects import KINDS, from_dict, to_dict


class ApiError(Exception):
    def __init__(self, status: int, message: str):
        super().__init__(f"{status}: {message}")
        self.status = status


class StoreClient:
    def __init__(self, base_url: str = "http://127.0.0.1:8343",
                 timeout: float = 10.0):
        import httpx
        self.base = base_url.rstrip("/")
        self._http = httpx.Client(timeout=timeout)

    def _raise(self, r):
        if r.status_code >= 400:
            try:
                msg = r.json().get("detail", r.text)
            except Exception:
                msg = r.text
            raise ApiError(r.status_code, msg)
        return r

    def create(self, kind: str, obj):
        r = self._raise(self._http.post(f"{self.base}/apis/{kind}",
                                        json=to_dict(obj)))
        return from_dict(KINDS[kind], r.json())

    def update(self, kind: str, obj):
        r = self._raise(self._http.put(
            f"{self.base}/apis/{kind}/{obj.meta.namespace}/{obj.meta.name}",
            json=to_dict(obj)))
        return from_dict(KINDS[kind], r.json())

    def delete(self, kind: str, namespace: str, name: str) -> bool:
        r = self._http.delete(f"{self.base}/apis/{kind}/{namespace}/{name}")
        return r.status_code < 400

    def get(self, kind: str, namespace: str, name: str):
        r = self._http.get(f"{self.base}/apis/{kind}/{namespace}/{name}")
        if r.status_code == 404:
            return None
        self._raise(r)
        return from_dict(KINDS[kind], r.json())

    def list(self, kind: str, namespace: Optional[str] = None,
             selector: Optional[Callable] = None) -> List[object]:
        params = {"namespace": namespace} if namespace else {}
        r = self._raise(self._http.get(f"{self.base}/apis/{kind}",
                                       params=params))
        out = [from_dict(KINDS[kind], d) for d in r.json()["items"]]
        if selector is not None:
            out = [o for o in out if selector(o)]
        return out

    def watch_since(self, rv: int, kinds: tuple = ()):
        r = self._raise(self._http.get(
            f"{self.base}/watch",
            params={"since": rv, "kinds": ",".join(kinds)}))
        data = r.json()
        events = [(e["rv"], e["type"], e["kind"],
                   from_dict(KINDS[e["kind"]], e["object"]))
                  for e in data["events"]]
        return data["resourceVersion"], events

    def healthz(self) -> bool:
        try:
            return self._http.get(f"{self.base}/healthz").status_code == 200
        except Exception:
            return False
