"""Launcher selection/reclaim policy tests incl. properties (hypothesis)."""

from hypothesis import given, settings
from hypothesis import strategies as st

from fma_amd.controller.dualpods.selection import (InstanceView, LauncherView,
                                                   select_or_reclaim)

TARGET = "Itargeti"


def lv(name, insts=(), ready=True, bound=False, max_instances=2, **kw):
    return LauncherView(name=name, ready=ready, bound=bound,
                        max_instances=max_instances,
                        instances=list(insts), **kw)


def iv(iid, status="running", port=8000, last_used=0.0):
    return InstanceView(iid, status, port, last_used)


def test_hot_beats_everything():
    launchers = [
        lv("warm", []),
        lv("hot", [iv(TARGET)]),
    ]
    r = select_or_reclaim(launchers, TARGET, 8000)
    assert r.launcher.name == "hot"
    assert r.has_sleeping_instance
    assert not r.deletions


def test_stopped_target_is_not_hot():
    r = select_or_reclaim([lv("l1", [iv(TARGET, status="stopped")])],
                          TARGET, 8000)
    # stopped target doesn't count as sleeping; launcher has capacity
    assert r.launcher.name == "l1"
    assert not r.has_sleeping_instance


def test_warm_when_capacity():
    r = select_or_reclaim([lv("l1", [iv("Iotheri", port=9000)])],
                          TARGET, 8000)
    assert r.launcher.name == "l1"
    assert not r.has_sleeping_instance
    assert not r.deletions


def test_port_conflict_forces_reclaim():
    r = select_or_reclaim([lv("l1", [iv("Iotheri", port=8000)])],
                          TARGET, 8000)
    assert r.launcher.name == "l1"
    assert ("l1", "Iotheri") in r.deletions


def test_full_launcher_reclaims_lru():
    insts = [iv("Ia", port=9001, last_used=100),
             iv("Ib", port=9002, last_used=50)]  # Ib older
    r = select_or_reclaim([lv("l1", insts, max_instances=2)], TARGET, 8000)
    assert r.deletions == [("l1", "Ib")]


def test_best_plan_has_most_victims():
    a = lv("few", [iv("Ix", port=8000)], max_instances=2)
    b = lv("many", [iv("Iy", port=8000), iv("Iz", port=8000)],
           max_instances=2)
    r = select_or_reclaim([a, b], TARGET, 8000)
    assert r.launcher.name == "many"
    assert len(r.deletions) == 2


def test_malformed_port_repaired_first():
    r = select_or_reclaim([lv("l1", [iv("Ibad", port=None)])], TARGET, 8000)
    assert r.retry
    assert r.deletions == [("l1", "Ibad")]
    assert r.launcher is None


def test_not_ready_launchers_mean_retry():
    r = select_or_reclaim([lv("l1", ready=False)], TARGET, 8000)
    assert r.retry and r.launcher is None and not r.deletions


def test_nothing_available():
    r = select_or_reclaim([lv("l1", bound=True)], TARGET, 8000)
    assert not r.retry and r.launcher is None


def test_skips_bound_failed_deleting():
    launchers = [lv("b", bound=True), lv("f", failed=True),
                 lv("d", deleting=True)]
    r = select_or_reclaim(launchers, TARGET, 8000)
    assert r.launcher is None


@st.composite
def launcher_strategy(draw):
    n = draw(st.integers(0, 4))
    insts = []
    for i in range(n):
        iid = draw(st.sampled_from(
            [TARGET, f"Ii{i}a", f"Ii{i}b"]))
        if any(x.instance_id == iid for x in insts):
            continue
        insts.append(iv(
            iid,
            status=draw(st.sampled_from(["running", "stopped"])),
            port=draw(st.sampled_from([8000, 9000, 9001])),
            last_used=draw(st.floats(0, 1000, allow_nan=False))))
    return lv(draw(st.text("abc", min_size=1, max_size=4)),
              insts,
              ready=draw(st.booleans()),
              bound=draw(st.booleans()),
              max_instances=draw(st.integers(1, 3)))


@settings(max_examples=200, deadline=None)
@given(st.lists(launcher_strategy(), max_size=5))
def test_selection_invariants(launchers):
    # unique names
    seen = set()
    uniq = []
    for l in launchers:
        if l.name not in seen:
            seen.add(l.name)
            uniq.append(l)
    r = select_or_reclaim(uniq, TARGET, 8000)
    by_name = {l.name: l for l in uniq}
    if r.launcher is not None:
        sel = r.launcher
        assert not sel.bound and not sel.failed and not sel.deleting
        assert sel.ready
    # never delete a LIVE target instance (a stopped one is dead weight
    # and may be cleaned up)
    for (lname, iid) in r.deletions:
        assert lname in by_name
        if iid == TARGET:
            assert any(i.instance_id == TARGET and i.status == "stopped"
                       for i in by_name[lname].instances)
    if r.has_sleeping_instance:
        assert any(i.instance_id == TARGET and i.status != "stopped"
                   for i in r.launcher.instances)
        assert not r.deletions
    # after performing the plan's deletions, the selected launcher can
    # host the target: either it already holds it, or its remaining
    # instance count is < max and no remaining instance uses the port
    if r.launcher is not None and not r.has_sleeping_instance:
        deleted = {iid for (_, iid) in r.deletions}
        remaining = [i for i in r.launcher.instances
                     if i.instance_id not in deleted
                     and i.instance_id != TARGET]
        assert len(remaining) <= r.launcher.max_instances - 1
        assert all(i.port != 8000 for i in remaining)
