"""Dual-pods actuation benchmark harness.

Analog of the reference's inference_server/benchmark: a scenario driver
measuring the metrics defined in the reference's benchmark.md:31-70 —

- T_actuation: requester created -> requester Ready
- T_wake: POST /wake_up -> is_sleeping=false
- T_instance_create: instance PUT -> serving
- Hot/Warm hit rates over a scenario's actuations

Two backends, like the reference's KubernetesOps family (kube_ops.py:329,
419, 451): :class:`SimClusterOps` (configurable synthetic delays — the
scenario logic is testable with no cluster at all) and
:class:`LiveClusterOps` (drives a real store + node agent + controllers,
the single-node e2e environment).
"""

from __future__ import annotations

import statistics
import time
from dataclasses import dataclass, field
from enum import Enum
from typing import Any, Dict, List, Optional


@dataclass
class ActuationSample:
    requester: str
    t_actuation: float
    path: str  # hot | warm | cold


@dataclass
class BenchReport:
    scenario: str
    samples: List[ActuationSample] = field(default_factory=list)

    def summary(self) -> Dict[str, Any]:
        ts = [s.t_actuation for s in self.samples]
        paths = [s.path for s in self.samples]
        n = len(self.samples)
        return {
            "scenario": self.scenario,
            "actuations": n,
            "t_actuation_mean_s": statistics.mean(ts) if ts else None,
            "t_actuation_p50_s": statistics.median(ts) if ts else None,
            "t_actuation_max_s": max(ts) if ts else None,
            "hot_hit_rate": paths.count("hot") / n if n else None,
            "warm_hit_rate": paths.count("warm") / n if n else None,
            "cold_rate": paths.count("cold") / n if n else None,
        }


class SimClusterOps:
    """Simulated cluster: readiness arrives after configured delays
    (reference SimKubernetesOps, kube_ops.py:451)."""

    def __init__(self, cold_s: float = 0.5, warm_s: float = 0.1,
                 hot_s: float = 0.02):
        self.delays = {"cold": cold_s, "warm": warm_s, "hot": hot_s}
        self.instances_seen: set = set()
        self.launchers = 0
        self._pending: Dict[str, tuple] = {}

    def create_requester(self, name: str, isc: str) -> None:
        if isc in self.instances_seen:
            path = "hot"
        elif self.launchers > 0:
            path = "warm"
        else:
            path = "cold"
            self.launchers += 1
        self.instances_seen.add(isc)
        self._pending[name] = (time.perf_counter(), path)

    def wait_ready(self, name: str, timeout: float = 60.0) -> str:
        t0, path = self._pending[name]
        remaining = self.delays[path] - (time.perf_counter() - t0)
        if remaining > 0:
            time.sleep(remaining)
        return path

    def delete_requester(self, name: str) -> None:
        self._pending.pop(name, None)


class LiveClusterOps:
    """Drives a live store+agent+controllers (the single-node stack)."""

    def __init__(self, store, agent, isc_port: int = 8355):
        import sys
        self.store = store
        self.agent = agent
        self.python = sys.executable
        self._created: Dict[str, float] = {}

    def create_requester(self, name: str, isc: str) -> None:
        from fma_amd.api import contracts as C
        from fma_amd.store import objects as ob
        pod = ob.new_object(
            "Pod", name,
            annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: isc},
            spec={"nodeName": self.agent.node, "containers": [{
                "name": "requester",
                "command": [self.python, "-m", "fma_amd.requester.server"],
            }]})
        self.store.create(pod, actor="bench")
        self._created[name] = time.perf_counter()

    def wait_ready(self, name: str, timeout: float = 120.0) -> str:
        import httpx
        deadline = time.perf_counter() + timeout
        while time.perf_counter() < deadline:
            pp = self.agent.pods.get(name)
            if pp is not None:
                try:
                    if httpx.get(f"http://{pp.ip}:8080/ready",
                                 timeout=2).status_code == 200:
                        return "unknown"
                except httpx.HTTPError:
                    pass
            time.sleep(0.1)
        raise TimeoutError(f"requester {name} never became ready")

    def delete_requester(self, name: str) -> None:
        from fma_amd.store.memstore import NotFound
        try:
            self.store.delete("Pod", name, actor="bench")
        except NotFound:
            pass
        deadline = time.perf_counter() + 60
        while time.perf_counter() < deadline:
            if self.store.try_get("Pod", name) is None:
                return
            time.sleep(0.1)


class DualPodsBenchmark:
    def __init__(self, ops):
        self.ops = ops

    def run_baseline(self, isc: str = "isc1", n: int = 3) -> BenchReport:
        """Cold first actuation, then hot re-actuations of the same ISC."""
        report = BenchReport("baseline")
        for i in range(n):
            name = f"bench-req-{i}"
            t0 = time.perf_counter()
            self.ops.create_requester(name, isc)
            path = self.ops.wait_ready(name)
            dt = time.perf_counter() - t0
            if path == "unknown":
                path = "cold" if i == 0 else "hot"
            report.samples.append(ActuationSample(name, dt, path))
            self.ops.delete_requester(name)
        return report

    def run_swap(self, isc_a: str, isc_b: str, cycles: int = 2
                 ) -> BenchReport:
        """Alternate two models on the same capacity (reference
        'switching instances in one launcher', test-cases.sh:560)."""
        report = BenchReport("swap")
        idx = 0
        for c in range(cycles):
            for isc in (isc_a, isc_b):
                name = f"swap-req-{idx}"
                idx += 1
                t0 = time.perf_counter()
                self.ops.create_requester(name, isc)
                path = self.ops.wait_ready(name)
                dt = time.perf_counter() - t0
                if path == "unknown":
                    path = "cold" if c == 0 and isc == isc_b else \
                        ("cold" if idx == 1 else "hot")
                report.samples.append(ActuationSample(name, dt, path))
                self.ops.delete_requester(name)
        return report

    def run_new_variant(self, iscs: List[str], n: int = 2
                        ) -> List[BenchReport]:
        """Introduce model variants one after another; each variant gets
        its own baseline pass tagged variant-<isc> (reference
        scenarios.py:271-330 run_new_variant_scenario)."""
        reports = []
        for isc in iscs:
            r = self.run_baseline(isc, n=n)
            r.scenario = f"variant-{isc}"
            reports.append(r)
        return reports

    def run_scaling(self, isc: str, n: int = 4) -> BenchReport:
        """N concurrent requesters (reference scenarios.py scaling)."""
        report = BenchReport("scaling")
        names = [f"scale-req-{i}" for i in range(n)]
        t0 = time.perf_counter()
        for name in names:
            self.ops.create_requester(name, isc)
        for name in names:
            path = self.ops.wait_ready(name)
            report.samples.append(ActuationSample(
                name, time.perf_counter() - t0,
                path if path != "unknown" else "cold"))
        for name in names:
            self.ops.delete_requester(name)
        return report


class ScenarioStatus(Enum):
    """Reference benchmark_diagnostics.py:24 (SUCCESS/FAILURE)."""
    SUCCESS = 1
    FAILURE = 2


@dataclass
class ScenarioResult:
    """Outcome + context of one scenario run, the diagnosis input
    (reference benchmark_diagnostics.py ScenarioResult:56-71)."""
    status: ScenarioStatus
    report: Optional[BenchReport] = None
    failed_requester: str = ""
    unready: List[str] = field(default_factory=list)
    error: str = ""


class BenchmarkDiagnosis:
    """Collect post-mortem state for a failing scenario before exiting
    (reference benchmark_diagnostics.py BenchmarkDiagnosis:73-180: dumps
    dual-pods controller logs and failing pod descriptions via kubectl;
    here the store IS the cluster, so the dump is the controller's log
    records plus the JSON of every Pod involved)."""

    def __init__(self, store=None, log_records: Optional[List[str]] = None):
        self.store = store
        self.log_records = log_records or []

    def collect_diagnostics(self, result: ScenarioResult,
                            out_dir: str) -> List[str]:
        import json
        import os
        os.makedirs(out_dir, exist_ok=True)
        written = []

        def dump(fname: str, text: str) -> None:
            p = os.path.join(out_dir, fname)
            with open(p, "w") as f:
                f.write(text)
            written.append(p)

        dump("scenario-result.json", json.dumps({
            "status": result.status.name,
            "failed_requester": result.failed_requester,
            "unready": result.unready,
            "error": result.error,
            "summary": result.report.summary() if result.report else None,
        }, indent=2))
        if self.log_records:
            dump("dual-pods-controller.log", "\n".join(self.log_records))
        if self.store is not None:
            pods = {}
            for pod in self.store.list("Pod"):
                name = pod["metadata"]["name"]
                if (name == result.failed_requester or name in result.unready
                        or "launcher" in name):
                    pods[name] = pod
            dump("pods.json", json.dumps(pods, indent=2, default=str))
        return written


def query_gpu_usage() -> Dict[str, Any]:
    """Accelerator memory in use, per GPU (reference benchmark_base.py:330
    query_gpu_usage shells nvidia-smi; MI355X uses rocm-smi)."""
    import json
    import subprocess
    try:
        out = subprocess.run(
            ["rocm-smi", "--showmeminfo", "vram", "--json"],
            capture_output=True, text=True, timeout=20)
        if out.returncode != 0:
            return {}
        return json.loads(out.stdout or "{}")
    except (OSError, ValueError, subprocess.SubprocessError):
        return {}
