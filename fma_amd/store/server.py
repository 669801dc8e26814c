"""HTTP front-end for the cluster store (the stack's mini-apiserver).

Exposes MemStore's semantics over REST so the controllers, node agent,
launcher notifier and CLI tools can run as separate processes against one
state store — the single-node analog of kube-apiserver in the reference's
deployment. The wire surface is deliberately small:

  POST   /apis/{kind}                          create (body: object)
  GET    /apis/{kind}?namespace=&labels=k=v,..  list (+ ?revision response)
  GET    /apis/{kind}/{ns}/{name}              get
  PUT    /apis/{kind}/{ns}/{name}              update (?subresource=status)
  DELETE /apis/{kind}/{ns}/{name}              delete (?uid=&rv=)
  GET    /watch?since=N&kinds=a,b              NDJSON watch stream
  GET    /healthz

The ``actor`` for admission comes from the X-FMA-Actor header.
"""

from __future__ import annotations

import asyncio
import json
from typing import Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse

from fma_amd.store.memstore import ApiError, MemStore, RevisionTooOld


def create_app(store: Optional[MemStore] = None) -> FastAPI:
    app = FastAPI(title="fma-amd cluster store")
    st = store or MemStore()
    app.state.store = st
    # standard Pod indexes are served here so StoreClient.index_get is a
    # store-side dict hit (reference informer indexers, controller.go:129-159)
    from fma_amd.store.indexes import install_pod_indexes
    install_pod_indexes(st)

    def actor_of(request: Request) -> str:
        return request.headers.get("X-FMA-Actor", "anonymous")

    def err(e: ApiError) -> JSONResponse:
        return JSONResponse({"error": e.message}, status_code=e.code)

    @app.get("/healthz")
    def healthz():
        return {"status": "OK", "revision": st.list_revision()}

    @app.post("/apis/{kind}")
    async def create(kind: str, request: Request):
        body = await request.json()
        body["kind"] = kind
        try:
            return JSONResponse(st.create(body, actor=actor_of(request)),
                                status_code=201)
        except ApiError as e:
            return err(e)

    @app.get("/apis/{kind}")
    def list_objs(kind: str, namespace: str = "default",
                  labels: str = "", all_namespaces: bool = False):
        sel = None
        if labels:
            sel = dict(kv.split("=", 1) for kv in labels.split(",") if kv)
        items = st.list(kind, None if all_namespaces else namespace, sel)
        return {"items": items, "revision": st.list_revision()}

    @app.get("/apis/{kind}/{ns}/{name}")
    def get_obj(kind: str, ns: str, name: str):
        try:
            return st.get(kind, name, ns)
        except ApiError as e:
            return err(e)

    @app.put("/apis/{kind}/{ns}/{name}")
    async def update(kind: str, ns: str, name: str, request: Request,
                     subresource: Optional[str] = None,
                     uid: Optional[str] = None, rv: Optional[str] = None):
        body = await request.json()
        body["kind"] = kind
        try:
            return st.update(body, actor=actor_of(request),
                             expect_uid=uid, expect_rv=rv,
                             subresource=subresource)
        except ApiError as e:
            return err(e)

    @app.patch("/apis/{kind}/{ns}/{name}")
    async def patch(kind: str, ns: str, name: str, request: Request,
                    strategic: int = 0):
        body = await request.json()
        try:
            return st.patch(kind, name, body, ns,
                            actor=actor_of(request),
                            strategic=bool(strategic))
        except ApiError as e:
            return err(e)

    @app.delete("/apis/{kind}/{ns}/{name}")
    def delete(kind: str, ns: str, name: str, request: Request,
               uid: Optional[str] = None, rv: Optional[str] = None):
        try:
            st.delete(kind, name, ns, actor=actor_of(request),
                      expect_uid=uid, expect_rv=rv)
            return {"status": "ok"}
        except ApiError as e:
            return err(e)

    @app.get("/index/{kind}/{iname}")
    def index_get(kind: str, iname: str, key: str,
                  namespace: Optional[str] = "default"):
        try:
            return {"items": st.index_get(kind, iname, key, namespace)}
        except KeyError as e:
            return JSONResponse({"error": str(e)}, status_code=404)

    @app.get("/watch")
    async def watch(request: Request, since: int = 0, kinds: str = ""):
        kind_list = [k for k in kinds.split(",") if k] or None
        loop = asyncio.get_running_loop()

        async def stream():
            cursor = since
            while True:
                if await request.is_disconnected():
                    return
                try:
                    batch = await loop.run_in_executor(
                        None, lambda: list(st.watch(since=cursor,
                                                    kinds=kind_list,
                                                    timeout=1.0)))
                except RevisionTooOld as e:
                    # Terminal line: status already streamed, so signal 410
                    # in-band; the client raises and its caller re-LISTs.
                    yield json.dumps({"error": e.message, "code": 410}) + "\n"
                    return
                for ev in batch:
                    cursor = max(cursor, ev.revision)
                    yield json.dumps({
                        "revision": ev.revision, "type": ev.type,
                        "kind": ev.kind, "object": ev.obj}) + "\n"

        return StreamingResponse(stream(), media_type="application/x-ndjson")

    return app


def main() -> None:
    import argparse

    import uvicorn

    from fma_amd.store.admission import install_policies

    ap = argparse.ArgumentParser("fma-store")
    ap.add_argument("--port", type=int, default=8081)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--no-admission", action="store_true")
    args = ap.parse_args()
    st = MemStore()
    if not args.no_admission:
        install_policies(st)
    uvicorn.run(create_app(st), host=args.host, port=args.port,
                log_level="warning")


if __name__ == "__main__":
    main()
