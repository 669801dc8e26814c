"""Launcher-populator controller: proactive launcher Pod population.

Re-implements the reference's two-queue architecture (reference
pkg/controller/launcher-populator/populator.go:56-132):

- a single-worker digest queue is the SOLE writer of the digested policy
  (node -> LauncherConfig -> desired count), fed by LPP/LC/Node events;
  per-(node,LC) desired = max over matching LPPs
  (digest-updater.go:107-196);
- key workers reconcile one (node, lcName) each: categorize existing
  launcher Pods (bound / live-unbound-current / stale-by-template-hash /
  deleting), delete stale and excess with UID+RV preconditions, create
  the missing difference (populator.go:402-572);
- key workers start only after the initial digest batch drains
  (onDigestSyncProcessed, populator.go:356-363);
- pending expectations bridge informer lag after creates/deletes
  (pending_expectations.go:31-157);
- stuck-launcher phases with metric + label (metrics.go:244-310).
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Set, Tuple

from fma_amd.api import contracts
from fma_amd.api.types import LauncherPopulationPolicySpec
from fma_amd.controller import metrics
from fma_amd.controller.populator.podtemplate import (
    build_node_independent_template, build_launcher_pod, specialize_to_node)
from fma_amd.controller.workqueue import InitialSyncTracker, QueueAndWorkers
from fma_amd.store import objects as ob
from fma_amd.store.memstore import (Conflict, MemStore, NotFound,
                                    RevisionTooOld)

HANDS_OFF = -1  # LC missing/malformed: do not create or delete

STUCK_SCHEDULING_SECONDS = 120.0    # reference metrics.go thresholds
STUCK_STARTING_SECONDS = 450.0


@dataclass
class LcDigest:
    template: Optional[Dict[str, Any]] = None
    template_hash: str = ""
    error: Optional[str] = None


@dataclass
class DigestedPolicy:
    """RWMutex-protected maps (reference digested-policy.go:32-205)."""
    lock: threading.Lock = field(default_factory=threading.Lock)
    #: node -> lcName -> desired count (max over LPPs)
    desired: Dict[str, Dict[str, int]] = field(default_factory=dict)
    #: lcName -> digest
    lcs: Dict[str, LcDigest] = field(default_factory=dict)
    #: lppName -> set of (node, lcName) it contributes to
    lpp_keys: Dict[str, Set[Tuple[str, str]]] = field(default_factory=dict)
    #: lppName -> {(node, lcName): count}
    lpp_counts: Dict[str, Dict[Tuple[str, str], int]] = field(
        default_factory=dict)

    def snapshot_for_key(self, node: str, lc_name: str
                         ) -> Tuple[int, Optional[LcDigest]]:
        with self.lock:
            lc = self.lcs.get(lc_name)
            if lc is None or lc.error:
                return HANDS_OFF, lc
            count = max((c.get((node, lc_name), 0)
                         for c in self.lpp_counts.values()), default=0)
            return count, lc


class LauncherPopulator:
    def __init__(self, store: MemStore, namespace: str = "default",
                 key_workers: int = 4, clock=time):
        self.store = store
        self.ns = namespace
        self.clock = clock
        self.policy = DigestedPolicy()
        self.key_queue: QueueAndWorkers = QueueAndWorkers(
            "populator-keys", key_workers, self._process_key)
        self.digest_queue: QueueAndWorkers = QueueAndWorkers(
            "populator-digest", 1, self._process_digest)
        self._sync = InitialSyncTracker(self._on_digest_synced)
        self._keys_started = threading.Event()
        self._stop = threading.Event()
        self._watch_thread: Optional[threading.Thread] = None
        # pending expectations: uids we created/deleted not yet observed
        self._expected_creates: Dict[Tuple[str, str], Set[str]] = {}
        self._expected_deletes: Dict[Tuple[str, str], Set[str]] = {}
        self._expect_stamp: Dict[Tuple[str, str], float] = {}

    # -- wiring --------------------------------------------------------

    def start(self) -> None:
        self.digest_queue.start()
        for lpp in self.store.list("LauncherPopulationPolicy", self.ns):
            item = ("lpp", ob.name_of(lpp))
            self._sync.register(item)
            self.digest_queue.queue.add(item)
        for lc in self.store.list("LauncherConfig", self.ns):
            item = ("lc", ob.name_of(lc))
            self._sync.register(item)
            self.digest_queue.queue.add(item)
        for node in self.store.list("Node", self.ns):
            item = ("node", ob.name_of(node))
            self._sync.register(item)
            self.digest_queue.queue.add(item)
        self._sync.start()
        self._watch_thread = threading.Thread(target=self._watch_loop,
                                              daemon=True)
        self._watch_thread.start()

    def _on_digest_synced(self) -> None:
        # key workers start only after the initial digest drains
        self.key_queue.start()
        self._keys_started.set()
        with self.policy.lock:
            keys = {(n, lc) for n, per in self.policy.desired.items()
                    for lc in per}
            for counts in self.policy.lpp_counts.values():
                keys.update(counts.keys())
        for key in keys:
            self.key_queue.queue.add(key)

    def stop(self) -> None:
        self._stop.set()
        self.digest_queue.stop()
        self.key_queue.stop()

    def _watch_loop(self) -> None:
        # Informer loop with 410 recovery: on history eviction, re-enqueue
        # a full digest pass and resume from the current list revision.
        while not self._stop.is_set():
            since = self.store.list_revision()
            try:
                self._watch_once(since)
                return  # stop was set
            except RevisionTooOld:
                self._requeue_all()

    def _requeue_all(self) -> None:
        for kind, tag in (("LauncherConfig", "lc"),
                          ("LauncherPopulationPolicy", "lpp"),
                          ("Node", "node")):
            for o in self.store.list(kind, self.ns):
                self.digest_queue.queue.add((tag, ob.name_of(o)))
        if self._keys_started.is_set():
            with self.policy.lock:
                keys = {(n, lc) for n, per in self.policy.desired.items()
                        for lc in per}
            for key in keys:
                self.key_queue.queue.add(key)

    def _watch_once(self, since: int) -> None:
        for ev in self.store.watch(
                since=since, stop=self._stop,
                kinds=["Pod", "LauncherConfig",
                       "LauncherPopulationPolicy", "Node"]):
            if ev.kind == "LauncherPopulationPolicy":
                self.digest_queue.queue.add(("lpp", ob.name_of(ev.obj)))
            elif ev.kind == "LauncherConfig":
                self.digest_queue.queue.add(("lc", ob.name_of(ev.obj)))
            elif ev.kind == "Node":
                self.digest_queue.queue.add(("node", ob.name_of(ev.obj)))
            elif ev.kind == "Pod" and self._keys_started.is_set():
                lbl = ob.labels_of(ev.obj)
                if lbl.get(contracts.COMPONENT_LABEL) == \
                        contracts.LAUNCHER_COMPONENT:
                    node = ob.pod_node_name(ev.obj) or \
                        lbl.get(contracts.NODE_NAME_LABEL, "")
                    lc_name = lbl.get(contracts.LAUNCHER_CONFIG_NAME_LABEL,
                                      "")
                    if node and lc_name:
                        self._observe_pod_event(ev, (node, lc_name))
                        self.key_queue.queue.add((node, lc_name))

    def _observe_pod_event(self, ev, key: Tuple[str, str]) -> None:
        uid = ob.uid_of(ev.obj)
        if ev.type == "ADDED":
            self._expected_creates.get(key, set()).discard(uid)
        elif ev.type == "DELETED":
            self._expected_deletes.get(key, set()).discard(uid)

    # -- digest side (single worker == sole writer) ----------------------

    def _process_digest(self, item: Tuple[str, str]) -> bool:
        kind, name = item
        try:
            if kind == "lc":
                self._update_digest_for_lc(name)
            elif kind == "lpp":
                self._update_digest_for_lpp(name)
            elif kind == "node":
                self._update_digest_for_node(name)
        finally:
            self._sync.mark_processed(item)
        return False

    def _update_digest_for_lc(self, name: str) -> None:
        """SOLE validator of LC templates; writes LC.status (reference
        digest-updater.go:42-97)."""
        lc = self.store.try_get("LauncherConfig", name, self.ns)
        with self.policy.lock:
            if lc is None:
                self.policy.lcs.pop(name, None)
            else:
                digest = LcDigest()
                try:
                    tmpl = build_node_independent_template(lc)
                    digest.template = tmpl
                    digest.template_hash = tmpl["metadata"]["annotations"][
                        contracts.LAUNCHER_TEMPLATE_HASH_ANNOTATION]
                except Exception as e:  # malformed template
                    digest.error = f"invalid pod template: {e}"
                self.policy.lcs[name] = digest
            affected = {key for counts in self.policy.lpp_counts.values()
                        for key in counts if key[1] == name}
        if lc is not None:
            self._write_status("LauncherConfig", name,
                               [self.policy.lcs[name].error]
                               if self.policy.lcs[name].error else [])
        if self._keys_started.is_set():
            for key in affected:
                self.key_queue.queue.add(key)

    def _update_digest_for_lpp(self, name: str) -> None:
        """SOLE runner of node matching; writes LPP.status (reference
        digest-updater.go:107-196)."""
        lpp = self.store.try_get("LauncherPopulationPolicy", name, self.ns)
        errors: List[str] = []
        new_counts: Dict[Tuple[str, str], int] = {}
        if lpp is not None:
            try:
                spec = LauncherPopulationPolicySpec(**lpp["spec"])
                nodes = self._matching_nodes(spec)
                for cfl in spec.countForLauncher:
                    for node in nodes:
                        new_counts[(node, cfl.launcherConfigName)] = \
                            cfl.launcherCount
            except Exception as e:
                errors.append(str(e))
        with self.policy.lock:
            old = set(self.policy.lpp_counts.get(name, {}))
            if lpp is None:
                self.policy.lpp_counts.pop(name, None)
            else:
                self.policy.lpp_counts[name] = new_counts
            changed = old | set(new_counts)
        if lpp is not None:
            self._write_status("LauncherPopulationPolicy", name, errors)
        if self._keys_started.is_set():
            for key in changed:
                self.key_queue.queue.add(key)

    def _update_digest_for_node(self, name: str) -> None:
        """Replays all LPPs for one node (reference
        digest-updater.go:202-233)."""
        with self.policy.lock:
            lpp_names = list(self.policy.lpp_counts)
        for lpp_name in lpp_names:
            self._update_digest_for_lpp(lpp_name)
        # a brand-new node may match LPPs not yet digested for it
        for lpp in self.store.list("LauncherPopulationPolicy", self.ns):
            if ob.name_of(lpp) not in lpp_names:
                self._update_digest_for_lpp(ob.name_of(lpp))

    def _matching_nodes(self, spec: LauncherPopulationPolicySpec
                        ) -> List[str]:
        out = []
        for node in self.store.list("Node", self.ns):
            labels = ob.labels_of(node)
            allocatable = node.get("status", {}).get("allocatable", {})
            if spec.enhancedNodeSelector.matches_node(labels, allocatable):
                out.append(ob.name_of(node))
        return out

    def _write_status(self, kind: str, name: str, errors: List[str]) -> None:
        obj = self.store.try_get(kind, name, self.ns)
        if obj is None:
            return
        want = {"observedGeneration": ob.meta(obj).get("generation", 0),
                "errors": errors}
        if obj.get("status") == want:
            return
        obj["status"] = want
        try:
            self.store.update(obj, actor="launcher-populator",
                              subresource="status")
        except (Conflict, NotFound):
            pass

    # -- key side --------------------------------------------------------

    def _launchers_for_key(self, node: str, lc_name: str
                           ) -> List[Dict[str, Any]]:
        out = []
        for pod in self.store.list(
                "Pod", self.ns,
                label_selector={contracts.COMPONENT_LABEL:
                                contracts.LAUNCHER_COMPONENT,
                                contracts.LAUNCHER_CONFIG_NAME_LABEL:
                                lc_name}):
            if (ob.pod_node_name(pod) or ob.labels_of(pod).get(
                    contracts.NODE_NAME_LABEL)) == node:
                out.append(pod)
        return out

    def _process_key(self, key: Tuple[str, str]) -> bool:
        node, lc_name = key
        desired, lc_digest = self.policy.snapshot_for_key(node, lc_name)
        pods = self._launchers_for_key(node, lc_name)
        self._record_phases(lc_name, pods)
        self._schedule_phase_flip(key, pods)
        if desired == HANDS_OFF or lc_digest is None:
            return False

        # expectations: if we still await prior creates/deletes, wait
        # (with timeout fallback, reference pending_expectations.go:31-157)
        stamp = self._expect_stamp.get(key, 0)
        pending = self._expected_creates.get(key, set()) | \
            self._expected_deletes.get(key, set())
        if pending:
            if self.clock.time() - stamp < 5.0:
                return True
            self._expected_creates.pop(key, None)
            self._expected_deletes.pop(key, None)

        bound, current, stale, deleting = [], [], [], []
        for pod in pods:
            if ob.is_deleting(pod):
                deleting.append(pod)
            elif ob.annotations_of(pod).get(contracts.REQUESTER_ANNOTATION):
                bound.append(pod)
            elif ob.pod_phase(pod) == "Failed":
                stale.append(pod)  # crashed unbound launcher: replace it
            elif ob.annotations_of(pod).get(
                    contracts.LAUNCHER_TEMPLATE_HASH_ANNOTATION) != \
                    lc_digest.template_hash:
                stale.append(pod)
            else:
                current.append(pod)

        # delete stale unbound launchers (UID+RV preconditions)
        if stale:
            for pod in stale:
                self._delete_pod(key, pod)
            return True  # requeue after deletions (reference :438-572)

        total_live = len(bound) + len(current)
        if total_live > desired and current:
            excess = total_live - desired
            for pod in current[:excess]:
                self._delete_pod(key, pod)
            return True

        diff = desired - total_live
        if diff > 0:
            lc = self.store.try_get("LauncherConfig", lc_name, self.ns)
            if lc is None:
                return False
            created = self._expected_creates.setdefault(key, set())
            for i in range(diff):
                pod = build_launcher_pod(
                    lc, node,
                    name_suffix=f"{int(self.clock.time()*1e3) % 10**9}-{i}")
                t0 = self.clock.time()
                try:
                    out = self.store.create(pod, actor="launcher-populator")
                    created.add(ob.uid_of(out))
                except Conflict:
                    continue
                metrics.launcher_create_seconds().observe(
                    self.clock.time() - t0)
            self._expect_stamp[key] = self.clock.time()
        return False

    def _delete_pod(self, key: Tuple[str, str], pod: Dict[str, Any]) -> None:
        try:
            self.store.delete("Pod", ob.name_of(pod), self.ns,
                              actor="launcher-populator",
                              expect_uid=ob.uid_of(pod),
                              expect_rv=ob.rv_of(pod))
            self._expected_deletes.setdefault(key, set()).add(ob.uid_of(pod))
            self._expect_stamp[key] = self.clock.time()
        except (NotFound, Conflict):
            pass

    def _schedule_phase_flip(self, key: Tuple[str, str],
                             pods: List[Dict[str, Any]]) -> None:
        """Timed re-reconcile exactly at the next stuck-phase transition —
        no polling sweep (reference metrics.go:303-310,
        reportStuckLaunchers populator.go:579-626)."""
        soonest: Optional[float] = None
        now = self.clock.time()
        for pod in pods:
            if ob.annotations_of(pod).get(contracts.REQUESTER_ANNOTATION):
                continue
            created = ob.meta(pod).get("creationTimestamp") or now
            scheduled = bool(ob.pod_node_name(pod))
            if not scheduled:
                flip = created + STUCK_SCHEDULING_SECONDS
            elif not ob.pod_is_ready(pod):
                flip = created + STUCK_STARTING_SECONDS
            else:
                continue
            if flip > now and (soonest is None or flip < soonest):
                soonest = flip
        if soonest is not None:
            self.key_queue.queue.add_after(key, soonest - now + 0.05)

    # -- stuck detection (reference metrics.go:244-310) -------------------

    def launcher_phase(self, pod: Dict[str, Any],
                       template_hash: Optional[str] = None) -> str:
        if ob.annotations_of(pod).get(contracts.REQUESTER_ANNOTATION):
            return "bound"
        if template_hash is not None and ob.annotations_of(pod).get(
                contracts.LAUNCHER_TEMPLATE_HASH_ANNOTATION) != template_hash:
            return "stale"
        created = ob.meta(pod).get("creationTimestamp") or self.clock.time()
        age = self.clock.time() - created
        scheduled = bool(ob.pod_node_name(pod))
        if not scheduled and age > STUCK_SCHEDULING_SECONDS:
            return "stuck_scheduling"
        if scheduled and not ob.pod_is_ready(pod) and \
                age > STUCK_STARTING_SECONDS:
            return "stuck_starting"
        return "unbound"

    def _record_phases(self, lc_name: str, pods: List[Dict[str, Any]]
                       ) -> None:
        with self.policy.lock:
            digest = self.policy.lcs.get(lc_name)
        th = digest.template_hash if digest else None
        counts: Dict[str, int] = {}
        for pod in pods:
            phase = self.launcher_phase(pod, th)
            counts[phase] = counts.get(phase, 0) + 1
            if phase.startswith("stuck"):
                cur = self.store.try_get("Pod", ob.name_of(pod), self.ns)
                if cur is not None and ob.labels_of(cur).get(
                        contracts.LAUNCHER_STUCK_LABEL) != phase:
                    ob.labels_of(cur)[contracts.LAUNCHER_STUCK_LABEL] = phase
                    try:
                        self.store.update(cur, actor="launcher-populator")
                    except (Conflict, NotFound):
                        pass
                    from fma_amd.controller.events import (
                        REASON_LAUNCHER_STUCK, record_event)
                    record_event(
                        self.store, pod, REASON_LAUNCHER_STUCK,
                        f"launcher Pod {ob.name_of(pod)} is {phase}",
                        actor="launcher-populator", namespace=self.ns)
        for phase in ("bound", "unbound", "stale", "stuck_scheduling",
                      "stuck_starting"):
            metrics.launcher_pod_count().labels(lc_name, phase).set(
                counts.get(phase, 0))


__all__ = ["LauncherPopulator", "DigestedPolicy", "HANDS_OFF",
           "build_node_independent_template", "specialize_to_node"]
