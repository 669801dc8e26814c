"""KubeStore: the FMA stack's Kubernetes backend.

Speaks the Kubernetes REST wire protocol (core v1 + fma.llm-d.ai/v1alpha1)
with the exact method surface of MemStore/StoreClient, so both controllers
and the node agent run against a real apiserver — or the in-tree
kubeapiserver double — without a line of change:

    create / get / try_get / list / update / delete / watch /
    list_revision / add_index / index_get

Design notes for a REAL cluster:

- ``list_revision()`` captures a per-resource resourceVersion snapshot and
  returns an opaque integer token; ``watch(since=token)`` opens one
  k8s watch stream per resource from that snapshot and multiplexes them
  into a single event iterator (the informer pattern). A token this store
  did not hand out — or a 410 Gone from the server — raises
  RevisionTooOld, and the caller re-LISTs, exactly like a reflector.
- update(expect_rv=...) maps to the k8s optimistic-concurrency contract
  (resourceVersion in the object; 409 on mismatch). expect_uid on delete
  maps to DeleteOptions preconditions; on update it is emulated with a
  read-and-compare (the controllers combine it with expect_rv, which
  still closes the race).
- ``index_get`` evaluates the standard index functions client-side over a
  list (an informer cache would hold these; against the in-process
  MemStore path the controllers get true O(1) store-side indexes).
- The admission ``actor`` maps to authentication: FMA controller actors
  are sent as ServiceAccount usernames matching the
  ValidatingAdmissionPolicy exemption pattern; anything else is sent
  verbatim and subject to the deny rules.
"""

from __future__ import annotations

import json
import queue
import threading
from typing import Any, Dict, Iterator, List, Optional

import httpx

from fma_amd.store import objects as ob
from fma_amd.store.indexes import POD_INDEXES
from fma_amd.store.memstore import (AlreadyExists, ApiError, Conflict,
                                    Invalid, NotFound, RevisionTooOld,
                                    WatchEvent)

#: kind -> (path prefix builder info: group, version, resource, namespaced)
RESOURCE_MAP = {
    "Pod": ("", "v1", "pods", True),
    "ConfigMap": ("", "v1", "configmaps", True),
    "Event": ("", "v1", "events", True),
    "Node": ("", "v1", "nodes", False),
    "InferenceServerConfig":
        ("fma.llm-d.ai", "v1alpha1", "inferenceserverconfigs", True),
    "LauncherConfig": ("fma.llm-d.ai", "v1alpha1", "launcherconfigs", True),
    "LauncherPopulationPolicy":
        ("fma.llm-d.ai", "v1alpha1", "launcherpopulationpolicies", True),
}

#: in-process actors that authenticate as FMA controller service accounts
#: (matching the VAP exemption regex "…:[^:]*-fma-controllers$")
FMA_SERVICE_ACCOUNTS = {
    "dual-pods-controller":
        "system:serviceaccount:fma-system:dual-pods-fma-controllers",
    "launcher-populator":
        "system:serviceaccount:fma-system:populator-fma-controllers",
    "node-agent":
        "system:serviceaccount:fma-system:node-agent-fma-controllers",
    "system": "system:serviceaccount:fma-system:system-fma-controllers",
}


def _raise_status(r: httpx.Response) -> None:
    try:
        msg = r.json().get("message", r.text)
    except Exception:  # noqa: BLE001
        msg = r.text
    code = r.status_code
    if code == 404:
        raise NotFound(msg)
    if code == 409:
        if "already exists" in msg:
            raise AlreadyExists(msg)
        raise Conflict(msg)
    if code == 410:
        raise RevisionTooOld(msg)
    if code == 422:
        raise Invalid(msg)
    raise ApiError(code, msg)


#: standard in-cluster service-account mount (k8s convention)
SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class KubeStore:
    def __init__(self, base_url: str, actor: str = "system",
                 timeout: float = 30.0, bearer_token: Optional[str] = None,
                 ca_cert: Optional[str] = None):
        self.base = base_url.rstrip("/")
        self.default_actor = actor
        self._bearer = bearer_token
        # reads and watch streams need the token too, so it rides on the
        # client as a default header (writes still go through _hdr)
        base_headers = ({"Authorization": f"Bearer {bearer_token}"}
                        if bearer_token else {})
        self._client = httpx.Client(timeout=timeout, headers=base_headers,
                                    verify=ca_cert if ca_cert else True)
        self._mu = threading.Lock()
        self._tokens: Dict[int, Dict[str, str]] = {}
        self._next_token = 1

    @classmethod
    def in_cluster(cls, actor: str = "system", timeout: float = 30.0,
                   sa_dir: str = SA_DIR) -> "KubeStore":
        """Build from the in-Pod service-account mount, the way client-go's
        rest.InClusterConfig does (the reference controllers run this way
        via their Helm chart; ours deploy/charts/fma-amd does the same).
        Returns the store; `in_cluster_namespace` gives the namespace."""
        import os
        host = os.environ["KUBERNETES_SERVICE_HOST"]
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        if ":" in host and not host.startswith("["):
            host = f"[{host}]"  # IPv6 service host
        with open(os.path.join(sa_dir, "token")) as f:
            token = f.read().strip()
        ca = os.path.join(sa_dir, "ca.crt")
        return cls(f"https://{host}:{port}", actor=actor, timeout=timeout,
                   bearer_token=token,
                   ca_cert=ca if os.path.exists(ca) else None)

    @staticmethod
    def in_cluster_namespace(sa_dir: str = SA_DIR) -> str:
        import os
        try:
            with open(os.path.join(sa_dir, "namespace")) as f:
                return f.read().strip()
        except OSError:
            return "default"

    # -- plumbing -----------------------------------------------------------

    def _path(self, kind: str, namespace: Optional[str],
              name: Optional[str] = None,
              subresource: Optional[str] = None) -> str:
        group, version, resource, namespaced = RESOURCE_MAP[kind]
        root = "/api/v1" if not group else f"/apis/{group}/{version}"
        if namespaced:
            p = f"{root}/namespaces/{namespace or 'default'}/{resource}"
        else:
            p = f"{root}/{resource}"
        if name:
            p += f"/{name}"
        if subresource:
            p += f"/{subresource}"
        return p

    def _hdr(self, actor: Optional[str]) -> Dict[str, str]:
        a = actor or self.default_actor
        if self._bearer:
            # real apiserver: the SA token IS the identity; the username
            # the VAP bindings see comes from token review, not a header
            return {"Authorization": f"Bearer {self._bearer}"}
        return {"X-Remote-User": FMA_SERVICE_ACCOUNTS.get(a, a)}

    # -- CRUD ---------------------------------------------------------------

    def create(self, obj: Dict[str, Any], actor: str = None
               ) -> Dict[str, Any]:
        kind = obj.get("kind", "")
        if kind not in RESOURCE_MAP:
            raise Invalid(f"unmapped kind {kind!r}")
        r = self._client.post(
            self.base + self._path(kind, ob.namespace_of(obj)),
            json=obj, headers=self._hdr(actor))
        if r.status_code != 201:
            _raise_status(r)
        return r.json()

    def get(self, kind: str, name: str, namespace: str = "default"
            ) -> Dict[str, Any]:
        r = self._client.get(self.base + self._path(kind, namespace, name))
        if r.status_code != 200:
            _raise_status(r)
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
        return self._list_raw(kind, namespace, label_selector)["items"]

    def _list_raw(self, kind: str, namespace: Optional[str] = "default",
                  label_selector: Optional[Dict[str, str]] = None
                  ) -> Dict[str, Any]:
        params = {}
        if label_selector:
            params["labelSelector"] = ",".join(
                f"{k}={v}" for k, v in label_selector.items())
        group, version, resource, namespaced = RESOURCE_MAP[kind]
        ns = namespace if namespaced else None
        r = self._client.get(self.base + self._path(kind, ns),
                             params=params)
        if r.status_code != 200:
            _raise_status(r)
        return r.json()

    def update(self, obj: Dict[str, Any], actor: str = None,
               expect_uid: Optional[str] = None,
               expect_rv: Optional[str] = None,
               subresource: Optional[str] = None) -> Dict[str, Any]:
        kind = obj.get("kind", "")
        name = ob.name_of(obj)
        ns = ob.namespace_of(obj)
        if expect_uid is not None:
            cur = self.get(kind, name, ns)
            if ob.uid_of(cur) != expect_uid:
                raise Conflict(f"uid mismatch on {kind}/{name}")
        body = ob.deepcopy(obj)
        if expect_rv is not None:
            ob.meta(body)["resourceVersion"] = expect_rv
        r = self._client.put(
            self.base + self._path(kind, ns, name, subresource),
            json=body, headers=self._hdr(actor))
        if r.status_code != 200:
            _raise_status(r)
        return r.json()

    def patch(self, kind: str, name: str, patch: Dict[str, Any],
              namespace: str = "default", actor: str = None,
              strategic: bool = False) -> Dict[str, Any]:
        """Server-side merge patch with the real k8s content types
        (application/merge-patch+json or …strategic-merge-patch+json) —
        no resourceVersion, so concurrent patches of disjoint fields
        cannot lose each other."""
        ctype = ("application/strategic-merge-patch+json" if strategic
                 else "application/merge-patch+json")
        headers = dict(self._hdr(actor))
        headers["Content-Type"] = ctype
        r = self._client.request(
            "PATCH", self.base + self._path(kind, namespace, name),
            content=json.dumps(patch).encode(), headers=headers)
        if r.status_code != 200:
            _raise_status(r)
        return r.json()

    def delete(self, kind: str, name: str, namespace: str = "default",
               actor: str = None, expect_uid: Optional[str] = None,
               expect_rv: Optional[str] = None) -> None:
        body = None
        if expect_uid or expect_rv:
            pre = {}
            if expect_uid:
                pre["uid"] = expect_uid
            if expect_rv:
                pre["resourceVersion"] = expect_rv
            body = {"preconditions": pre}
        r = self._client.request(
            "DELETE", self.base + self._path(kind, namespace, name),
            json=body, headers=self._hdr(actor))
        if r.status_code != 200:
            _raise_status(r)

    # -- indexes ------------------------------------------------------------

    def add_index(self, kind: str, name: str, fn) -> None:
        """No-op: index functions are evaluated client-side (a real
        deployment would hold them in an informer cache)."""

    def index_get(self, kind: str, index_name: str, key: str,
                  namespace: Optional[str] = "default"
                  ) -> List[Dict[str, Any]]:
        fn = POD_INDEXES.get(index_name)
        if kind != "Pod" or fn is None:
            raise KeyError(f"no index {index_name!r} on {kind}")
        return [o for o in self.list(kind, namespace)
                if key in (fn(o) or [])]

    # -- watch --------------------------------------------------------------

    def list_revision(self) -> int:
        """Capture a per-resource resourceVersion snapshot; the returned
        token is only valid for a subsequent watch(since=token)."""
        cursors: Dict[str, str] = {}
        for kind in RESOURCE_MAP:
            try:
                raw = self._list_raw(kind, None
                                     if not RESOURCE_MAP[kind][3]
                                     else "default")
                cursors[kind] = raw.get("metadata", {}).get(
                    "resourceVersion", "0")
            except ApiError:
                cursors[kind] = "0"
        with self._mu:
            token = self._next_token
            self._next_token += 1
            self._tokens[token] = cursors
            # bound memory: keep the last 64 snapshots
            for old in sorted(self._tokens)[:-64]:
                del self._tokens[old]
        return token

    def watch(self, since: int = 0, kinds: Optional[List[str]] = None,
              stop: Optional[threading.Event] = None,
              timeout: Optional[float] = None) -> Iterator[WatchEvent]:
        with self._mu:
            cursors = self._tokens.get(since)
        if cursors is None:
            raise RevisionTooOld(
                f"watch token {since} unknown (stale snapshot); re-LIST")
        kind_list = kinds or list(RESOURCE_MAP)
        q: "queue.Queue" = queue.Queue()
        local_stop = threading.Event()
        threads = []
        for kind in kind_list:
            th = threading.Thread(
                target=self._watch_one, daemon=True,
                args=(kind, cursors.get(kind, "0"), q, local_stop))
            th.start()
            threads.append(th)
        seq = since * 1_000_000  # locally monotonic event revisions
        try:
            while True:
                if stop is not None and stop.is_set():
                    return
                try:
                    item = q.get(timeout=timeout if timeout is not None
                                 else 0.5)
                except queue.Empty:
                    if timeout is not None:
                        return
                    continue
                if isinstance(item, Exception):
                    raise item
                seq += 1
                yield WatchEvent(seq, item["type"],
                                 item["object"].get("kind", ""),
                                 item["object"])
        finally:
            local_stop.set()

    def _watch_one(self, kind: str, rv: str, out: "queue.Queue",
                   stop: threading.Event) -> None:
        group, version, resource, namespaced = RESOURCE_MAP[kind]
        path = self._path(kind, "default" if namespaced else None)
        cursor = rv
        while not stop.is_set():
            try:
                with self._client.stream(
                        "GET", self.base + path,
                        params={"watch": "1", "resourceVersion": cursor,
                                "allowWatchBookmarks": "true"},
                        # bounded read: bookmarks keep a live stream fed,
                        # so a timeout means the server idled/died — we
                        # reconnect from the cursor, and an abandoned
                        # watcher can observe its stop event instead of
                        # blocking forever (thread-leak fix)
                        timeout=httpx.Timeout(5.0, read=8.0)) as r:
                    if r.status_code == 410:
                        out.put(RevisionTooOld(f"{kind} watch expired"))
                        return
                    if r.status_code != 200:
                        out.put(ApiError(r.status_code, f"watch {kind}"))
                        return
                    for line in r.iter_lines():
                        if stop.is_set():
                            return
                        if not line:
                            continue
                        ev = json.loads(line)
                        if ev.get("type") == "ERROR":
                            code = (ev.get("object") or {}).get("code")
                            if code == 410:
                                out.put(RevisionTooOld(
                                    f"{kind} watch expired"))
                                return
                            # transient server-side error: reconnect from
                            # the cursor like any dropped stream
                            break
                        obj = ev["object"]
                        new_rv = ob.rv_of(obj)
                        if new_rv:
                            cursor = new_rv
                        if ev.get("type") == "BOOKMARK":
                            continue  # cursor keep-alive only, no event
                        out.put(ev)
            except httpx.HTTPError:
                if stop.is_set():
                    return
                import time
                time.sleep(0.3)  # apiserver hiccup; reconnect from cursor
