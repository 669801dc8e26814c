"""Outbound HTTP for controllers: one shared helper + a fake for tests.

Mirrors the reference's doHTTP discipline (reference pkg/controller/
dual-pods/inference-server.go:2195-2254): a single 30 s-timeout client for
every stub/launcher/server call, JSON in/out, latency observation, and the
DR-20 log convention — every outbound call logs its start timestamp
(``httpCallStartTime``) so log-based tracing can reconstruct timelines
(reference DESIGN_RULES.md:139-163).

Tests inject :class:`FakeHttp`, which routes URLs to in-process ASGI test
clients or plain callables — the same seam the reference gets from httptest
servers.
"""

from __future__ import annotations

import json as jsonlib
import logging
import time
from typing import Any, Callable, Dict, Optional

logger = logging.getLogger("fma.http")

TIMEOUT_SECONDS = 30.0


class HttpResult:
    def __init__(self, status: int, body: Any):
        self.status = status
        self.body = body

    @property
    def ok(self) -> bool:
        return 200 <= self.status < 300


class HttpAdapter:
    """Production adapter over httpx."""

    def __init__(self, observe: Optional[Callable[[str, str, int, float],
                                                  None]] = None):
        import httpx
        self._client = httpx.Client(timeout=TIMEOUT_SECONDS)
        self._observe = observe

    def request(self, method: str, url: str, *, purpose: str = "",
                json: Any = None, params: Optional[Dict[str, Any]] = None,
                headers: Optional[Dict[str, str]] = None,
                content: Optional[bytes] = None) -> HttpResult:
        start = time.time()
        logger.debug("http call", extra={"httpCallStartTime": start,
                                         "method": method, "url": url,
                                         "purpose": purpose})
        try:
            r = self._client.request(method, url, json=json, params=params,
                                     headers=headers, content=content)
            status = r.status_code
            try:
                body = r.json()
            except Exception:
                body = r.text
        except Exception as e:
            status, body = 0, str(e)
        if self._observe:
            self._observe(purpose, method, status, time.time() - start)
        return HttpResult(status, body)


class FakeHttp:
    """Routes http://host:port/... to registered handlers.

    register("1.2.3.4:8001", test_client) — anything with .request(method,
    path, json=...) (fastapi TestClient) or a callable
    (method, path, json, params) -> (status, body).
    """

    def __init__(self) -> None:
        self.routes: Dict[str, Any] = {}
        self.calls: list = []

    def register(self, hostport: str, handler: Any) -> None:
        self.routes[hostport] = handler

    def request(self, method: str, url: str, *, purpose: str = "",
                json: Any = None, params: Optional[Dict[str, Any]] = None,
                headers: Optional[Dict[str, str]] = None,
                content: Optional[bytes] = None) -> HttpResult:
        self.calls.append((method, url, purpose))
        assert url.startswith("http://"), url
        rest = url[len("http://"):]
        hostport, _, path = rest.partition("/")
        path = "/" + path
        handler = self.routes.get(hostport)
        if handler is None:
            return HttpResult(0, f"connection refused: {hostport}")
        if callable(handler) and not hasattr(handler, "request"):
            # 4-arg fake handlers predate headers/content; pass content as
            # json payload for simplicity
            status, body = handler(method, path,
                                   json if content is None else content,
                                   params)
            return HttpResult(status, body)
        r = handler.request(method, path, json=json, params=params,
                            headers=headers, content=content)
        try:
            body = r.json()
        except Exception:
            body = r.text
        return HttpResult(r.status_code, body)


class LauncherClient:
    """Typed client for the launcher REST API (reference
    pkg/controller/dual-pods/launcherclient.go:29-281)."""

    def __init__(self, http, base_url: str):
        self.http = http
        self.base = base_url.rstrip("/")

    def list_instances(self) -> HttpResult:
        return self.http.request(
            "GET", self.base + "/v2/vllm/instances",
            purpose="launcher-list")

    def get_instance(self, iid: str) -> HttpResult:
        return self.http.request(
            "GET", f"{self.base}/v2/vllm/instances/{iid}",
            purpose="launcher-get")

    def create_named_instance(self, iid: str, config: Dict[str, Any]
                              ) -> HttpResult:
        return self.http.request(
            "PUT", f"{self.base}/v2/vllm/instances/{iid}",
            purpose="launcher-create", json=config)

    def delete_instance(self, iid: str) -> HttpResult:
        return self.http.request(
            "DELETE", f"{self.base}/v2/vllm/instances/{iid}",
            purpose="launcher-delete")


def dump_json(obj: Any) -> str:
    return jsonlib.dumps(obj, sort_keys=True, separators=(",", ":"))
