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
