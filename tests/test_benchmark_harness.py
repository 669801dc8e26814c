"""Benchmark harness scenario logic (sim backend, no cluster)."""

from fma_amd.benchmark.harness import DualPodsBenchmark, SimClusterOps


def test_baseline_cold_then_hot():
    bench = DualPodsBenchmark(SimClusterOps(cold_s=0.05, warm_s=0.02,
                                            hot_s=0.005))
    rep = bench.run_baseline(n=3)
    s = rep.summary()
    assert s["actuations"] == 3
    assert rep.samples[0].path == "cold"
    assert all(x.path == "hot" for x in rep.samples[1:])
    assert s["hot_hit_rate"] == 2 / 3
    assert rep.samples[0].t_actuation > rep.samples[1].t_actuation


def test_swap_scenario():
    bench = DualPodsBenchmark(SimClusterOps(cold_s=0.03, warm_s=0.01,
                                            hot_s=0.002))
    rep = bench.run_swap("a", "b", cycles=2)
    s = rep.summary()
    assert s["actuations"] == 4
    # first actuation of each model is cold/warm, repeats are hot
    assert rep.samples[2].path == "hot" and rep.samples[3].path == "hot"


def test_scaling_scenario():
    bench = DualPodsBenchmark(SimClusterOps())
    rep = bench.run_scaling("m", n=3)
    assert rep.summary()["actuations"] == 3


def test_new_variant_scenario():
    """Each variant is a tagged baseline pass; the first actuation of a
    new variant on warmed capacity is warm, its repeat hot (reference
    scenarios.py:271 run_new_variant_scenario)."""
    bench = DualPodsBenchmark(SimClusterOps(cold_s=0.03, warm_s=0.01,
                                            hot_s=0.002))
    reports = bench.run_new_variant(["m1", "m2"], n=2)
    assert [r.scenario for r in reports] == ["variant-m1", "variant-m2"]
    assert reports[0].samples[0].path == "cold"
    assert reports[0].samples[1].path == "hot"
    assert reports[1].samples[0].path == "warm"


def test_diagnosis_collects_failure_state(tmp_path):
    from fma_amd.benchmark.harness import (BenchmarkDiagnosis, BenchReport,
                                           ScenarioResult, ScenarioStatus)
    from fma_amd.store.memstore import MemStore
    from fma_amd.store import objects as ob
    store = MemStore()
    store.create(ob.new_object("Pod", "req-x", spec={}))
    store.create(ob.new_object("Pod", "launcher-1", spec={}))
    store.create(ob.new_object("Pod", "bystander", spec={}))
    diag = BenchmarkDiagnosis(store, log_records=["r1 bound", "r1 woke"])
    res = ScenarioResult(status=ScenarioStatus.FAILURE,
                         report=BenchReport("baseline"),
                         failed_requester="req-x", unready=["req-x"],
                         error="never became ready")
    files = diag.collect_diagnostics(res, str(tmp_path / "diag"))
    names = {f.rsplit("/", 1)[-1] for f in files}
    assert names == {"scenario-result.json", "dual-pods-controller.log",
                     "pods.json"}
    import json
    pods = json.loads(open(str(tmp_path / "diag" / "pods.json")).read())
    assert set(pods) == {"req-x", "launcher-1"}  # bystander excluded
    sr = json.loads(
        open(str(tmp_path / "diag" / "scenario-result.json")).read())
    assert sr["status"] == "FAILURE" and sr["failed_requester"] == "req-x"


def test_query_gpu_usage_no_gpu_is_empty_or_dict():
    from fma_amd.benchmark.harness import query_gpu_usage
    assert isinstance(query_gpu_usage(), dict)


def test_autoscale_demo_runs(tmp_path):
    """tools/demo_autoscale.py (the reference's demo-fma-hpa analog)
    completes a 1->2->1 scale cycle through the real single-node stack
    and reports per-replica actuation latencies."""
    import json
    import os
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    res = subprocess.run(
        [sys.executable, "tools/demo_autoscale.py", "--curve", "1,2,1",
         "--log-dir", str(tmp_path)],
        capture_output=True, text=True, timeout=240, cwd=root,
        env=dict(os.environ, FMA_FAKE_GPU="1"))
    assert res.returncode == 0, res.stderr[-2000:]
    out = json.loads(res.stdout)
    assert out["scale_ups"] == 2
    assert out["first_s"] > 0
    downs = [e for e in out["events"] if e["event"] == "scale-down"]
    assert len(downs) == 1  # tick 2 scales 2 -> 1 (teardown is unrecorded)
