"""Launcher selection & reclaim (pure logic).

Re-implements selectOrReclaimLauncherPod (reference pkg/controller/
dual-pods/inference-server.go:783-1011) as a pure function over launcher
snapshots, returning both the selection and the instance deletions the
caller must perform (the reference interleaves HTTP deletes; hoisting them
makes the policy property-testable):

- Priority 1 (hot): a launcher already holding the (non-stopped) target
  instance -> bind + wake only.
- Priority 2 (warm): a launcher with spare capacity (< MaxInstances) and
  no port conflict.
- Priority 3 (reclaim): delete victims — port-conflicting instances
  always, then least-recently-used others; the best plan is the one with
  the MOST victims (empty the fullest launcher), ties broken by oldest
  LRU victim then instance ID (reference compareReclaimPlans:995-1000).
- A launcher reporting an instance with no usable port is repaired by
  deleting that instance and retrying with a fresh snapshot (:849-871).
- Not-ready launchers make the result retryable instead of triggering a
  cold start (:831-836).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple


@dataclass
class InstanceView:
    instance_id: str
    status: str          # "running" | "stopped"
    port: Optional[int]  # None == malformed / no usable port
    last_used: float = 0.0


@dataclass
class LauncherView:
    name: str
    pod: Any = None
    ready: bool = True
    bound: bool = False
    failed: bool = False
    deleting: bool = False
    max_instances: int = 1
    instances: List[InstanceView] = field(default_factory=list)


@dataclass
class SelectionResult:
    launcher: Optional[LauncherView] = None
    has_sleeping_instance: bool = False
    retry: bool = False
    #: (launcher_name, instance_id) deletions the caller must perform
    #: BEFORE using the selection
    deletions: List[Tuple[str, str]] = field(default_factory=list)


def _cmp_last_used(a: str, at: float, b: str, bt: float) -> int:
    if at < bt:
        return -1
    if bt < at:
        return 1
    return -1 if a < b else (1 if a > b else 0)


def pick_instance_victims(candidates: List[str],
                          last_used: Dict[str, float],
                          limit: int) -> List[str]:
    """Up to `limit` least-recently-used candidates
    (reference pickInstanceVictims:963-976)."""
    if limit <= 0:
        return []
    import functools
    ordered = sorted(candidates, key=functools.cmp_to_key(
        lambda a, b: _cmp_last_used(a, last_used.get(a, 0.0),
                                    b, last_used.get(b, 0.0))))
    return ordered[:limit]


@dataclass
class _Plan:
    launcher: LauncherView
    victims: List[str]
    lru_id: str
    lru_time: float


def _plan_lru(victims: List[str], last_used: Dict[str, float]
              ) -> Tuple[str, float]:
    lru_id = victims[0]
    lru_t = last_used.get(lru_id, 0.0)
    for v in victims[1:]:
        vt = last_used.get(v, 0.0)
        if _cmp_last_used(v, vt, lru_id, lru_t) < 0:
            lru_id, lru_t = v, vt
    return lru_id, lru_t


def _cmp_plans(a: _Plan, b: _Plan) -> int:
    if len(a.victims) != len(b.victims):
        return len(b.victims) - len(a.victims)  # MORE victims sorts first
    return _cmp_last_used(a.lru_id, a.lru_time, b.lru_id, b.lru_time)


def select_or_reclaim(launchers: List[LauncherView],
                      target_instance_id: str,
                      desired_port: int,
                      last_used: Optional[Dict[str, float]] = None,
                      ) -> SelectionResult:
    # LRU times come from the instance views; an explicit map (the
    # controller's cache, which may know about instances no longer listed)
    # overrides them.
    lu: Dict[str, float] = {}
    for l in launchers:
        for inst in l.instances:
            lu[inst.instance_id] = inst.last_used
    lu.update(last_used or {})

    candidate_with_capacity: Optional[LauncherView] = None
    capacity_deletions: List[Tuple[str, str]] = []
    some_not_ready = False
    best_plan: Optional[_Plan] = None

    for lv in launchers:
        if lv.failed or lv.deleting or lv.bound:
            continue
        if not lv.ready:
            some_not_ready = True
            continue

        max_others = lv.max_instances - 1
        has_sleeping = False
        port_conflicts: List[str] = []
        others: List[str] = []
        stopped: List[str] = []  # dead instances: free deletions
        for inst in lv.instances:
            if inst.port is None:
                # repair: delete the malformed instance, retry fresh
                return SelectionResult(
                    retry=True, deletions=[(lv.name, inst.instance_id)])
            if inst.status == "stopped":
                # a stopped instance only occupies a slot (and, for the
                # target id, would 409 a named re-create): always delete
                # (the reference deletes these in syncLauncherInstances
                # before selection, inference-server.go:2129-2151)
                stopped.append(inst.instance_id)
                continue
            if inst.instance_id == target_instance_id:
                has_sleeping = True
                continue
            if inst.port == desired_port:
                port_conflicts.append(inst.instance_id)
            else:
                others.append(inst.instance_id)
        if has_sleeping:
            return SelectionResult(launcher=lv, has_sleeping_instance=True)

        total_live = len(port_conflicts) + len(others)
        if not port_conflicts and total_live <= max_others:
            if candidate_with_capacity is None:
                candidate_with_capacity = lv
                capacity_deletions = [(lv.name, s) for s in stopped]
            continue

        to_delete = max(total_live - max_others, 1)
        victims = port_conflicts + pick_instance_victims(
            others, lu, to_delete - len(port_conflicts))
        if not victims:
            continue
        lru_id, lru_t = _plan_lru(victims, lu)
        plan = _Plan(lv, victims + stopped, lru_id, lru_t)
        if best_plan is None or _cmp_plans(plan, best_plan) < 0:
            best_plan = plan

    if candidate_with_capacity is not None:
        return SelectionResult(launcher=candidate_with_capacity,
                               deletions=capacity_deletions)
    if best_plan is not None:
        return SelectionResult(
            launcher=best_plan.launcher,
            deletions=[(best_plan.launcher.name, v)
                       for v in best_plan.victims])
    if some_not_ready:
        return SelectionResult(retry=True)
    return SelectionResult()
