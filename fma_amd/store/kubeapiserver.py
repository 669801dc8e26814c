"""Kubernetes API server test double: the real k8s wire protocol over
MemStore.

The reference's e2e runs on kind — a real apiserver in containers
(reference test/e2e/run-launcher-based.sh:1-80). This environment has no
container runtime, so the equivalent seam is an in-tree apiserver double
that speaks the Kubernetes REST protocol faithfully enough for the FMA
stack to run *unchanged through a Kubernetes client* (KubeStore):

  GET/POST          /api/v1/namespaces/{ns}/{pods,configmaps,events}
  GET/PUT/DELETE    /api/v1/namespaces/{ns}/{resource}/{name}[/status]
  GET/POST          /api/v1/nodes[/{name}]               (cluster-scoped)
  GET/POST/...      /apis/fma.llm-d.ai/v1alpha1/namespaces/{ns}/{crds}
  ?watch=1&resourceVersion=N    NDJSON watch streams (410 via ERROR event)
  ?labelSelector=k%3Dv,k2%3Dv2  equality selectors
  DELETE preconditions ({"preconditions": {"uid":…, "resourceVersion":…}})

Authn is front-proxy style: the username comes from the X-Remote-User
header. On Pod UPDATE the ValidatingAdmissionPolicy YAML artifacts from
``manifests/validating-admission-policies/`` are evaluated with the CEL
evaluator (fma_amd/store/cel.py) — the deny rules tests assert are the
shipped cluster artifacts, not a parallel Python implementation.

Error bodies are k8s Status objects; resourceVersions are MemStore's
global revision (valid per k8s semantics: RVs are opaque and may be
shared across resources, as etcd's global revision is).
"""

from __future__ import annotations

import asyncio
import json
import os
from typing import Any, Dict, List, Optional, Tuple

import yaml
from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse

from concurrent.futures import ThreadPoolExecutor

from fma_amd.store import cel
from fma_amd.store import objects as ob
from fma_amd.store.memstore import (ApiError, Conflict, Invalid, MemStore,
                                    RevisionTooOld)

#: resource name <-> kind for everything the FMA stack touches
CORE_RESOURCES = {
    "pods": ("Pod", True),
    "configmaps": ("ConfigMap", True),
    "events": ("Event", True),
    "nodes": ("Node", False),
}
FMA_GROUP = "fma.llm-d.ai"
FMA_VERSION = "v1alpha1"
FMA_RESOURCES = {
    "inferenceserverconfigs": ("InferenceServerConfig", True),
    "launcherconfigs": ("LauncherConfig", True),
    "launcherpopulationpolicies": ("LauncherPopulationPolicy", True),
}

KIND_TO_RESOURCE = {kind: (res, namespaced, "")
                    for res, (kind, namespaced) in CORE_RESOURCES.items()}
KIND_TO_RESOURCE.update(
    {kind: (res, namespaced, f"{FMA_GROUP}/{FMA_VERSION}")
     for res, (kind, namespaced) in FMA_RESOURCES.items()})

VAP_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__)))),
    "manifests", "validating-admission-policies")


def status_body(code: int, message: str, reason: str = "") -> Dict[str, Any]:
    return {"kind": "Status", "apiVersion": "v1", "status": "Failure",
            "code": code, "message": message, "reason": reason}


def _err(e: ApiError) -> JSONResponse:
    reason = {404: "NotFound", 409: "Conflict", 410: "Gone",
              422: "Invalid"}.get(e.code, "")
    return JSONResponse(status_body(e.code, e.message, reason),
                        status_code=e.code)


class AdmissionPolicies:
    """Loads and evaluates the shipped VAP YAML + bindings."""

    def __init__(self, manifest_dir: str = VAP_DIR):
        self.policies: Dict[str, Dict[str, Any]] = {}
        self.bound: List[Dict[str, Any]] = []
        self._compiled: Dict[str, List[Tuple[Any, str]]] = {}
        if not os.path.isdir(manifest_dir):
            return
        docs: List[Dict[str, Any]] = []
        for fn in sorted(os.listdir(manifest_dir)):
            if not fn.endswith((".yaml", ".yml")):
                continue
            with open(os.path.join(manifest_dir, fn)) as f:
                docs.extend(d for d in yaml.safe_load_all(f) if d)
        for d in docs:
            if d.get("kind") == "ValidatingAdmissionPolicy":
                name = d["metadata"]["name"]
                self.policies[name] = d
                self._compiled[name] = [
                    (cel.compile_expr(v["expression"]),
                     v.get("message", "denied by policy"))
                    for v in d["spec"].get("validations", [])]
            elif d.get("kind") == "ValidatingAdmissionPolicyBinding":
                self.bound.append(d)

    def check_update(self, resource: str, group: str, old: Dict[str, Any],
                     new: Dict[str, Any], username: str) -> Optional[str]:
        """Returns a deny message or None."""
        env = {"object": new, "oldObject": old,
               "request": {"userInfo": {"username": username}}}
        for binding in self.bound:
            pol_name = binding["spec"]["policyName"]
            if pol_name not in self.policies:
                continue
            rules = (binding["spec"].get("matchResources") or {}).get(
                "resourceRules", [])
            if rules and not any(
                    resource in r.get("resources", [])
                    and group in (r.get("apiGroups") or [""])
                    and "UPDATE" in r.get("operations", [])
                    for r in rules):
                continue
            for expr, message in self._compiled.get(pol_name, []):
                try:
                    ok = expr(env)
                except cel.CelError as e:
                    # failurePolicy: Fail
                    return f"policy {pol_name} evaluation error: {e}"
                if ok is not True:
                    return f"{pol_name}: {message}"
        return None


def _parse_label_selector(sel: Optional[str]) -> Optional[Dict[str, str]]:
    if not sel:
        return None
    out = {}
    for part in sel.split(","):
        if "=" in part:
            k, v = part.split("=", 1)
            out[k.strip().lstrip("=")] = v.strip()
    return out or None


def create_app(store: Optional[MemStore] = None,
               vap_dir: str = VAP_DIR) -> FastAPI:
    app = FastAPI(title="fma-amd kube apiserver double")
    st = store or MemStore()
    app.state.store = st
    # dedicated executor for watch polling: every live watch stream
    # holds a worker for up to its poll interval, and the loop's default
    # executor (~cpu+4 threads) starves under one-stream-per-kind
    # informers from several controllers
    watch_pool = ThreadPoolExecutor(max_workers=64,
                                    thread_name_prefix="kube-watch")
    app.state.watch_pool = watch_pool

    @app.on_event("shutdown")
    async def _shutdown_pool():
        watch_pool.shutdown(wait=False, cancel_futures=True)

    # explicit close for test servers whose lifespan shutdown may not
    # fire (thread-leak fix: 64 idle pool workers per double otherwise
    # outlive their server for the whole pytest session)
    app.state.close = lambda: watch_pool.shutdown(wait=False,
                                                  cancel_futures=True)
    # structural CRD validation, as a real apiserver's schema would
    from fma_amd.store.admission import crd_schema_policy
    if crd_schema_policy not in st._admission:
        st.add_admission_hook(crd_schema_policy)
    vap = AdmissionPolicies(vap_dir)
    app.state.vap = vap

    def username_of(request: Request) -> str:
        return request.headers.get("X-Remote-User", "system:anonymous")

    def resolve(resource: str, group: str) -> Tuple[str, bool]:
        table = CORE_RESOURCES if not group else FMA_RESOURCES
        if resource not in table:
            raise ApiError(404, f"the server could not find the requested "
                                f"resource {group or 'core'}/{resource}")
        return table[resource]

    # ---- shared handlers --------------------------------------------------

    async def handle_list_or_watch(request: Request, resource: str,
                                   group: str, ns: Optional[str]):
        kind, namespaced = resolve(resource, group)
        q = request.query_params
        sel = _parse_label_selector(q.get("labelSelector"))
        if q.get("watch") in ("1", "true"):
            since = int(q.get("resourceVersion") or 0)
            bookmarks = q.get("allowWatchBookmarks") in ("1", "true")
            loop = asyncio.get_running_loop()

            async def stream():
                cursor = since
                while True:
                    if await request.is_disconnected():
                        return
                    # global revision read BEFORE polling: any event of
                    # this kind at or below it lands in this batch, so an
                    # empty batch makes it a safe bookmark cursor
                    rev0 = st.list_revision()
                    try:
                        batch = await loop.run_in_executor(
                            watch_pool, lambda: list(st.watch(
                                since=cursor, kinds=[kind], timeout=1.0)))
                    except RuntimeError:
                        return  # server shutting down (executor closed)
                    except RevisionTooOld as e:
                        yield json.dumps({
                            "type": "ERROR",
                            "object": status_body(410, e.message, "Gone"),
                        }) + "\n"
                        return
                    if bookmarks and not batch:
                        # keep-alive + cursor advance: moves idle
                        # watchers past history churned by other kinds
                        # (no 410 on reconnect) and keeps the stream
                        # fed so clients can use bounded read timeouts
                        cursor = max(cursor, rev0)
                        yield json.dumps({
                            "type": "BOOKMARK",
                            "object": {"kind": kind, "metadata": {
                                "resourceVersion": str(cursor)}}}) + "\n"
                    for ev in batch:
                        cursor = max(cursor, ev.revision)
                        obj = ev.obj
                        if namespaced and ns is not None \
                                and ob.namespace_of(obj) != ns:
                            continue
                        if sel and any(
                                ob.labels_of(obj).get(k) != v
                                for k, v in sel.items()):
                            continue
                        yield json.dumps({
                            "type": {"ADDED": "ADDED", "MODIFIED": "MODIFIED",
                                     "DELETED": "DELETED"}[ev.type],
                            "object": obj}) + "\n"

            return StreamingResponse(stream(),
                                     media_type="application/json")
        items = st.list(kind, namespace=ns if namespaced else None,
                        label_selector=sel)
        return {"kind": f"{kind}List",
                "apiVersion": "v1" if not group
                else f"{group}/{FMA_VERSION}",
                "metadata": {"resourceVersion": str(st.list_revision())},
                "items": items}

    async def handle_create(request: Request, resource: str, group: str,
                            ns: Optional[str]):
        kind, namespaced = resolve(resource, group)
        body = await request.json()
        body["kind"] = kind
        if namespaced and ns:
            body.setdefault("metadata", {})["namespace"] = ns
        try:
            created = st.create(body, actor=username_of(request))
        except ApiError as e:
            return _err(e)
        return JSONResponse(created, status_code=201)

    def handle_get(resource: str, group: str, ns: Optional[str], name: str):
        kind, namespaced = resolve(resource, group)
        try:
            return st.get(kind, name, ns if namespaced else "default")
        except ApiError as e:
            return _err(e)

    async def handle_put(request: Request, resource: str, group: str,
                         ns: Optional[str], name: str,
                         subresource: Optional[str] = None):
        kind, namespaced = resolve(resource, group)
        body = await request.json()
        body["kind"] = kind
        namespace = ns if namespaced else "default"
        body.setdefault("metadata", {})["namespace"] = namespace
        user = username_of(request)
        try:
            old = st.get(kind, name, namespace)
            if subresource != "status":
                deny = vap.check_update(resource, group, old, body, user)
                if deny:
                    return JSONResponse(
                        status_body(422, deny, "Invalid"), status_code=422)
            updated = st.update(body, actor=user, subresource=subresource)
        except ApiError as e:
            return _err(e)
        return updated

    async def handle_patch(request: Request, resource: str, group: str,
                           ns: Optional[str], name: str):
        """Server-side apply of merge / strategic-merge patches, content
        type selected the k8s way. VAP admission sees the MERGED result
        (as a real apiserver's admission chain does)."""
        kind, namespaced = resolve(resource, group)
        namespace = ns if namespaced else "default"
        ctype = request.headers.get("content-type", "")
        strategic = "strategic-merge-patch" in ctype
        if not strategic and "merge-patch" not in ctype:
            return JSONResponse(
                status_body(415, f"unsupported patch type {ctype!r}",
                            "UnsupportedMediaType"), status_code=415)
        body = await request.json()
        user = username_of(request)
        from fma_amd.store.merge import merge_patch, strategic_merge
        fn = strategic_merge if strategic else merge_patch
        try:
            for _ in range(16):
                old = st.get(kind, name, namespace)
                new = fn(old, body)
                if not isinstance(new, dict):
                    raise Invalid("patch must produce an object")
                deny = vap.check_update(resource, group, old, new, user)
                if deny:
                    return JSONResponse(
                        status_body(422, deny, "Invalid"), status_code=422)
                try:
                    return st.update(new, actor=user,
                                     expect_rv=ob.rv_of(old))
                except Conflict:
                    continue
            raise Conflict(f"patch on {kind} {namespace}/{name} kept "
                           "conflicting after 16 attempts")
        except ApiError as e:
            return _err(e)

    async def handle_delete(request: Request, resource: str, group: str,
                            ns: Optional[str], name: str):
        kind, namespaced = resolve(resource, group)
        try:
            raw = await request.body()
        except Exception:  # pragma: no cover
            raw = b""
        uid = rv = None
        if raw:
            try:
                pre = (json.loads(raw) or {}).get("preconditions") or {}
                uid, rv = pre.get("uid"), pre.get("resourceVersion")
            except json.JSONDecodeError:
                pass
        try:
            st.delete(kind, name, ns if namespaced else "default",
                      actor=username_of(request), expect_uid=uid,
                      expect_rv=rv)
        except ApiError as e:
            return _err(e)
        return status_body(200, "deleted", "")

    # ---- core v1 (namespaced) --------------------------------------------

    @app.get("/api/v1/namespaces/{ns}/{resource}")
    async def core_list(request: Request, ns: str, resource: str):
        try:
            return await handle_list_or_watch(request, resource, "", ns)
        except ApiError as e:
            return _err(e)

    @app.post("/api/v1/namespaces/{ns}/{resource}")
    async def core_create(request: Request, ns: str, resource: str):
        try:
            return await handle_create(request, resource, "", ns)
        except ApiError as e:
            return _err(e)

    @app.get("/api/v1/namespaces/{ns}/{resource}/{name}")
    def core_get(ns: str, resource: str, name: str):
        try:
            return handle_get(resource, "", ns, name)
        except ApiError as e:
            return _err(e)

    @app.put("/api/v1/namespaces/{ns}/{resource}/{name}")
    async def core_put(request: Request, ns: str, resource: str, name: str):
        try:
            return await handle_put(request, resource, "", ns, name)
        except ApiError as e:
            return _err(e)

    @app.put("/api/v1/namespaces/{ns}/{resource}/{name}/status")
    async def core_put_status(request: Request, ns: str, resource: str,
                              name: str):
        try:
            return await handle_put(request, resource, "", ns, name,
                                    subresource="status")
        except ApiError as e:
            return _err(e)

    @app.patch("/api/v1/namespaces/{ns}/{resource}/{name}")
    async def core_patch(request: Request, ns: str, resource: str,
                         name: str):
        return await handle_patch(request, resource, "", ns, name)

    @app.delete("/api/v1/namespaces/{ns}/{resource}/{name}")
    async def core_delete(request: Request, ns: str, resource: str,
                          name: str):
        try:
            return await handle_delete(request, resource, "", ns, name)
        except ApiError as e:
            return _err(e)

    # ---- core v1 cluster-scoped (nodes) ----------------------------------

    @app.get("/api/v1/nodes")
    async def nodes_list(request: Request):
        return await handle_list_or_watch(request, "nodes", "", None)

    @app.post("/api/v1/nodes")
    async def nodes_create(request: Request):
        return await handle_create(request, "nodes", "", None)

    @app.get("/api/v1/nodes/{name}")
    def nodes_get(name: str):
        return handle_get("nodes", "", None, name)

    @app.put("/api/v1/nodes/{name}")
    async def nodes_put(request: Request, name: str):
        return await handle_put(request, "nodes", "", None, name)

    @app.put("/api/v1/nodes/{name}/status")
    async def nodes_put_status(request: Request, name: str):
        return await handle_put(request, "nodes", "", None, name,
                                subresource="status")

    @app.delete("/api/v1/nodes/{name}")
    async def nodes_delete(request: Request, name: str):
        return await handle_delete(request, "nodes", "", None, name)

    # ---- fma.llm-d.ai/v1alpha1 CRDs ---------------------------------------

    PREFIX = f"/apis/{FMA_GROUP}/{FMA_VERSION}"

    @app.get(PREFIX + "/namespaces/{ns}/{resource}")
    async def fma_list(request: Request, ns: str, resource: str):
        try:
            return await handle_list_or_watch(request, resource, FMA_GROUP,
                                              ns)
        except ApiError as e:
            return _err(e)

    @app.post(PREFIX + "/namespaces/{ns}/{resource}")
    async def fma_create(request: Request, ns: str, resource: str):
        try:
            return await handle_create(request, resource, FMA_GROUP, ns)
        except ApiError as e:
            return _err(e)

    @app.get(PREFIX + "/namespaces/{ns}/{resource}/{name}")
    def fma_get(ns: str, resource: str, name: str):
        try:
            return handle_get(resource, FMA_GROUP, ns, name)
        except ApiError as e:
            return _err(e)

    @app.put(PREFIX + "/namespaces/{ns}/{resource}/{name}")
    async def fma_put(request: Request, ns: str, resource: str, name: str):
        try:
            return await handle_put(request, resource, FMA_GROUP, ns, name)
        except ApiError as e:
            return _err(e)

    @app.patch(PREFIX + "/namespaces/{ns}/{resource}/{name}")
    async def fma_patch(request: Request, ns: str, resource: str,
                        name: str):
        return await handle_patch(request, resource, FMA_GROUP, ns, name)

    @app.put(PREFIX + "/namespaces/{ns}/{resource}/{name}/status")
    async def fma_put_status(request: Request, ns: str, resource: str,
                             name: str):
        try:
            return await handle_put(request, resource, FMA_GROUP, ns, name,
                                    subresource="status")
        except ApiError as e:
            return _err(e)

    @app.delete(PREFIX + "/namespaces/{ns}/{resource}/{name}")
    async def fma_delete(request: Request, ns: str, resource: str,
                         name: str):
        try:
            return await handle_delete(request, resource, FMA_GROUP, ns,
                                       name)
        except ApiError as e:
            return _err(e)

    # ---- discovery-ish ----------------------------------------------------

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "revision": st.list_revision()}

    @app.get("/version")
    def version():
        return {"major": "1", "minor": "33",
                "gitVersion": "v1.33.0-fma-double"}

    return app


def main() -> None:  # pragma: no cover - exercised by live e2e
    import argparse

    import uvicorn

    ap = argparse.ArgumentParser("fma-kube-apiserver-double")
    ap.add_argument("--port", type=int, default=6443)
    ap.add_argument("--host", default="127.0.0.1")
    args = ap.parse_args()
    uvicorn.run(create_app(), host=args.host, port=args.port,
                log_level="warning")


if __name__ == "__main__":  # pragma: no cover
    main()
