"""Prometheus metrics, name-compatible with the reference's dashboards.

Metric names, labels and buckets follow the reference exactly so existing
dashboards/alerts port over (reference pkg/controller/dual-pods/
controller.go:204-293, docs/metrics.md:14-212; populator metrics
pkg/controller/launcher-populator/metrics.go:36-310).
"""

from __future__ import annotations


try:
    from prometheus_client import (Counter, Gauge, Histogram,
                                   start_http_server)
    HAVE_PROM = True
except Exception:  # pragma: no cover
    HAVE_PROM = False

_METRICS = {}


def _get_or_make(cls, name, doc, labels=(), **kw):
    if not HAVE_PROM:
        return _Noop()
    key = name
    if key not in _METRICS:
        _METRICS[key] = cls(name, doc, labelnames=list(labels), **kw)
    return _METRICS[key]


class _Noop:
    def labels(self, *a, **kw):
        return self

    def observe(self, *a):
        pass

    def inc(self, *a):
        pass

    def dec(self, *a):
        pass

    def set(self, *a):
        pass


# reference controller.go:268
ACTUATION_BUCKETS = (0, 1, 3, 5, 7.5, 10, 15, 30, 60, 120, 240, 480, 960, 1920)
# reference controller.go:275
LAUNCHER_CREATE_BUCKETS = (0.001, 0.005, 0.01, 0.05, 0.1, 0.5, 1, 2.5, 5)
# reference controller.go:284
HTTP_LATENCY_BUCKETS = (0.001, 0.01, 0.1, 0.3, 1, 3, 10, 30, 90, 270, 810)


def actuation_seconds():
    return _get_or_make(
        Histogram, "fma_actuation_seconds",
        "requester container start -> readiness relay, by actuation path",
        ("path", "instancesDeleted", "isc_name"),
        buckets=ACTUATION_BUCKETS)


def launcher_create_seconds():
    return _get_or_make(
        Histogram, "fma_launcher_create_seconds",
        "kube API create-Pod call latency for launchers",
        buckets=LAUNCHER_CREATE_BUCKETS)


def http_latency_seconds():
    return _get_or_make(
        Histogram, "fma_http_latency_seconds",
        "controller -> stub/launcher/server HTTP latency",
        ("purpose", "method", "status_code"),
        buckets=HTTP_LATENCY_BUCKETS)


def requester_count():
    return _get_or_make(Gauge, "fma_requester_count",
                        "server-requesting Pods known to the controller")


def isc_count():
    return _get_or_make(Gauge, "fma_isc_count",
                        "InferenceServerConfig objects known")


def duality():
    return _get_or_make(
        Gauge, "fma_duality",
        "1 per (requester, provider) binding; labels join to GPU metrics",
        ("requester_name", "provider_name", "node"))


def launcher_pod_count():
    return _get_or_make(
        Gauge, "fma_launcher_pod_count",
        "launcher Pods by LauncherConfig and phase",
        ("lcfg_name", "phase"))


def serve_metrics(port: int = 8002) -> None:
    """Expose /metrics (reference pkg/observability/prom-and-debug.go:34-79
    serves :8002; pprof has no Python analog — py-spy attaches externally)."""
    if HAVE_PROM:
        start_http_server(port)


def queue_adds_total():
    return _get_or_make(
        Counter, "fma_dpc_innerqueue_adds_total",
        "items added to the dual-pods inner queue (reference: k8s "
        "workqueue metrics, docs/metrics.md)", ("name",))


def queue_depth():
    return _get_or_make(
        Gauge, "fma_dpc_innerqueue_depth",
        "current depth of the dual-pods inner queue", ("name",))


def queue_retries_total():
    return _get_or_make(
        Counter, "fma_dpc_innerqueue_retries_total",
        "rate-limited retries on the dual-pods inner queue", ("name",))


def queue_queue_duration_seconds():
    return _get_or_make(
        Histogram, "fma_dpc_innerqueue_queue_duration_seconds",
        "time items wait in the inner queue before processing", ("name",),
        buckets=(.001, .01, .1, 1, 10, 60))


def queue_work_duration_seconds():
    return _get_or_make(
        Histogram, "fma_dpc_innerqueue_work_duration_seconds",
        "time spent processing inner-queue items", ("name",),
        buckets=(.001, .01, .1, 1, 10, 60))


def serve_debug(port: int = 8003):
    """The reference's debug listener analog (prom-and-debug.go:68-79
    serves Go /debug/pprof on :8003). Go's pprof has no direct Python
    equivalent; this serves the operational 90% — "where is it stuck" —
    as /debug/threads (a live stack dump of every thread), /debug/vars
    (gc + thread counts), and /debug/profile?seconds=S&hz=H — a sampling
    CPU profiler emitting flamegraph "folded" lines (the same role as
    /debug/pprof/profile). Returns the server (daemon thread).
    """
    import gc
    import http.server
    import json
    import sys
    import threading
    import time
    import traceback
    from collections import Counter as _Counter
    from urllib.parse import parse_qs, urlparse

    def sample_profile(seconds: float, hz: float) -> str:
        """Folded-stack samples across every thread (skipping ours)."""
        me = threading.get_ident()
        agg: _Counter = _Counter()
        names = {t.ident: t.name for t in threading.enumerate()}
        deadline = time.monotonic() + seconds
        period = 1.0 / max(hz, 1.0)
        while time.monotonic() < deadline:
            for tid, frame in sys._current_frames().items():
                if tid == me:
                    continue
                stack = []
                f = frame
                while f is not None:
                    stack.append(f"{f.f_code.co_name} "
                                 f"({f.f_code.co_filename.rsplit('/', 1)[-1]}"
                                 f":{f.f_lineno})")
                    f = f.f_back
                key = names.get(tid, "?") + ";" + ";".join(reversed(stack))
                agg[key] += 1
            time.sleep(period)
        return "".join(f"{k} {v}\n" for k, v in agg.most_common())

    class Handler(http.server.BaseHTTPRequestHandler):
        def do_GET(self):  # noqa: N802
            if self.path.startswith("/debug/profile"):
                q = parse_qs(urlparse(self.path).query)
                seconds = min(float(q.get("seconds", ["5"])[0]), 60.0)
                hz = min(float(q.get("hz", ["100"])[0]), 1000.0)
                body = sample_profile(seconds, hz).encode()
                ctype = "text/plain"
            elif self.path.startswith("/debug/threads"):
                names = {t.ident: t.name for t in threading.enumerate()}
                out = []
                for tid, frame in sys._current_frames().items():
                    out.append(f"--- thread {tid} ({names.get(tid, '?')})")
                    out.extend(x.rstrip()
                               for x in traceback.format_stack(frame))
                body = ("\n".join(out) + "\n").encode()
                ctype = "text/plain"
            elif self.path.startswith("/debug/vars"):
                body = json.dumps({
                    "threads": threading.active_count(),
                    "gc": gc.get_count(),
                }).encode()
                ctype = "application/json"
            else:
                self.send_error(404)
                return
            self.send_response(200)
            self.send_header("Content-Type", ctype)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):  # quiet
            pass

    srv = http.server.ThreadingHTTPServer(("0.0.0.0", port), Handler)
    threading.Thread(target=srv.serve_forever, daemon=True,
                     name="fma-debug-http").start()
    return srv


def observe_http(purpose: str, method: str, status: int, seconds: float
                 ) -> None:
    http_latency_seconds().labels(purpose, method, str(status)).observe(seconds)
