"""HTTP client for the cluster store, duck-typed to MemStore.

Controllers take either a MemStore (in-process; tests) or a StoreClient
(separate store server process; production single-node deployment) — the
method surface is identical.
"""

from __future__ import annotations

import json
import threading
from typing import Any, Dict, Iterator, List, Optional

import httpx

from fma_amd.store import objects as ob
from fma_amd.store.memstore import (ApiError, Conflict, Invalid,
                                    NotFound, RevisionTooOld, WatchEvent)


def _raise_for(code: int, message: str) -> None:
    if code == 404:
        raise NotFound(message)
    if code == 409:
        raise Conflict(message)
    if code == 410:
        raise RevisionTooOld(message)
    if code == 422:
        raise Invalid(message)
    raise ApiError(code, message)


class StoreClient:
    def __init__(self, base_url: str, actor: str = "anonymous",
                 timeout: float = 30.0):
        self.base = base_url.rstrip("/")
        self.actor = actor
        self._client = httpx.Client(
            timeout=timeout, headers={"X-FMA-Actor": actor})

    # -- CRUD ---------------------------------------------------------------

    def create(self, obj: Dict[str, Any], actor: Optional[str] = None
               ) -> Dict[str, Any]:
        r = self._client.post(f"{self.base}/apis/{obj['kind']}", json=obj,
                              headers=self._hdr(actor))
        if r.status_code != 201:
            _raise_for(r.status_code, r.json().get("error", r.text))
        return r.json()

    def get(self, kind: str, name: str, namespace: str = "default"
            ) -> Dict[str, Any]:
        r = self._client.get(f"{self.base}/apis/{kind}/{namespace}/{name}")
        if r.status_code != 200:
            _raise_for(r.status_code, r.text)
        return r.json()

    def try_get(self, kind: str, name: str, namespace: str = "default"
                ) -> Optional[Dict[str, Any]]:
        try:
            return self.get(kind, name, namespace)
        except NotFound:
            return None

    def list(self, kind: str, namespace: Optional[str] = "default",
             label_selector: Optional[Dict[str, str]] = None
             ) -> List[Dict[str, Any]]:
        params: Dict[str, Any] = {}
        if namespace is None:
            params["all_namespaces"] = "true"
        else:
            params["namespace"] = namespace
        if label_selector:
            params["labels"] = ",".join(f"{k}={v}"
                                        for k, v in label_selector.items())
        r = self._client.get(f"{self.base}/apis/{kind}", params=params)
        if r.status_code != 200:
            _raise_for(r.status_code, r.text)
        return r.json()["items"]

    def list_revision(self) -> int:
        r = self._client.get(f"{self.base}/healthz")
        return int(r.json().get("revision", 0))

    def update(self, obj: Dict[str, Any], actor: Optional[str] = None,
               expect_uid: Optional[str] = None,
               expect_rv: Optional[str] = None,
               subresource: Optional[str] = None) -> Dict[str, Any]:
        params: Dict[str, Any] = {}
        if subresource:
            params["subresource"] = subresource
        if expect_uid:
            params["uid"] = expect_uid
        if expect_rv:
            params["rv"] = expect_rv
        r = self._client.put(
            f"{self.base}/apis/{obj['kind']}/{ob.namespace_of(obj)}/"
            f"{ob.name_of(obj)}", json=obj, params=params,
            headers=self._hdr(actor))
        if r.status_code != 200:
            _raise_for(r.status_code, r.json().get("error", r.text))
        return r.json()

    def patch(self, kind: str, name: str, patch: Dict[str, Any],
              namespace: str = "default", actor: Optional[str] = None,
              strategic: bool = False) -> Dict[str, Any]:
        r = self._client.patch(
            f"{self.base}/apis/{kind}/{namespace}/{name}", json=patch,
            params={"strategic": int(strategic)}, headers=self._hdr(actor))
        if r.status_code != 200:
            _raise_for(r.status_code, r.json().get("error", r.text))
        return r.json()

    def delete(self, kind: str, name: str, namespace: str = "default",
               actor: Optional[str] = None,
               expect_uid: Optional[str] = None,
               expect_rv: Optional[str] = None) -> None:
        params: Dict[str, Any] = {}
        if expect_uid:
            params["uid"] = expect_uid
        if expect_rv:
            params["rv"] = expect_rv
        r = self._client.delete(
            f"{self.base}/apis/{kind}/{namespace}/{name}", params=params,
            headers=self._hdr(actor))
        if r.status_code != 200:
            _raise_for(r.status_code, r.json().get("error", r.text))

    def _hdr(self, actor: Optional[str]) -> Optional[Dict[str, str]]:
        return {"X-FMA-Actor": actor} if actor else None

    # -- indexes ------------------------------------------------------------

    def add_index(self, kind: str, name: str, fn) -> None:
        """No-op: the store server pre-installs the standard indexes
        (store/server.py create_app); registration is server-side."""

    def index_get(self, kind: str, index_name: str, key: str,
                  namespace: Optional[str] = "default"
                  ) -> List[Dict[str, Any]]:
        params: Dict[str, Any] = {"key": key}
        if namespace is not None:
            params["namespace"] = namespace
        r = self._client.get(f"{self.base}/index/{kind}/{index_name}",
                             params=params)
        if r.status_code != 200:
            raise KeyError(r.json().get("error", r.text))
        return r.json()["items"]

    # -- watch --------------------------------------------------------------

    def watch(self, since: int = 0, kinds: Optional[List[str]] = None,
              stop: Optional[threading.Event] = None,
              timeout: Optional[float] = None) -> Iterator[WatchEvent]:
        params: Dict[str, Any] = {"since": since}
        if kinds:
            params["kinds"] = ",".join(kinds)
        while True:
            if stop is not None and stop.is_set():
                return
            try:
                with self._client.stream("GET", f"{self.base}/watch",
                                         params=params, timeout=None) as r:
                    for line in r.iter_lines():
                        if stop is not None and stop.is_set():
                            return
                        if not line:
                            continue
                        d = json.loads(line)
                        if d.get("code") == 410:
                            raise RevisionTooOld(d.get("error", "revision too old"))
                        params["since"] = max(params["since"], d["revision"])
                        yield WatchEvent(d["revision"], d["type"], d["kind"],
                                         d["object"])
            except httpx.HTTPError:
                if stop is not None and stop.is_set():
                    return
                import time
                time.sleep(0.5)  # store restarting; reconnect
