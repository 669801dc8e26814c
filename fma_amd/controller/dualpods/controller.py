"""Dual-pods controller: binds server-requesting Pods to server-providing
Pods and orchestrates sleep/wake.

Re-implements the reference's control loop (reference pkg/controller/
dual-pods/controller.go + inference-server.go) against the pluggable
cluster store:

- event classification (careAbout, controller.go:580-593): bound provider /
  unbound launcher / requester / ignore;
- per-node serialization: one outer queue of node names, each node's items
  drained under that node's lock, oldest first (controller.go:404-424,
  inference-server.go:92-143);
- the inference-server reconcile (inference-server.go:171-763): finalizer
  upkeep, deletion mirroring, GPU discovery via the requester stub,
  launcher-based and direct actuation, bind/wake/sleep, readiness relay,
  actuation-path metrics;
- restart recovery purely from Pod annotations (controller.go:64-99).

State the controller is allowed to keep in memory is a cache; every
binding fact lives in Pod metadata (the store is the ACID substrate,
reference docs/dual-pods.md:729-737).
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Set, Tuple

import logging

from fma_amd.api import contracts
from fma_amd.controller import metrics
from fma_amd.controller.dualpods import nominal as nominal_mod
from fma_amd.controller.dualpods.identity import instance_id as compute_iid
from fma_amd.controller.dualpods.selection import (InstanceView,
                                                   LauncherView,
                                                   select_or_reclaim)
from fma_amd.controller.httpadapter import LauncherClient
from fma_amd.controller.workqueue import NodeQueueAndWorkers
from fma_amd.store import objects as ob
from fma_amd.store.indexes import install_pod_indexes
from fma_amd.store.memstore import (Conflict, MemStore, NotFound,
                                    RevisionTooOld)

REQUESTER_FINALIZER = "dual-pods.llm-d.ai/requester-protection"
PROVIDER_FINALIZER = "dual-pods.llm-d.ai/provider-protection"

logger = logging.getLogger("fma.dualpods")

RETRY = True
DONE = False
#: wait states (server booting, pod starting) re-queue through the same
#: exponential backoff, but the dual-pods queue caps it at 2 s (the
#: reference caps at 20 s, which shows up directly in cold T_actuation)
WAIT = True


@dataclass
class ServerData:
    """Per-inference-server cache (reference controller.go:455-518)."""
    uid: str
    requester_name: str
    gpus: Optional[List[str]] = None
    instance_id: Optional[str] = None
    port: Optional[int] = None
    needed_new_launcher: bool = False
    needed_new_instance: bool = False
    readiness_relayed: bool = False
    deleted_instances: Set[str] = field(default_factory=set)
    instance_last_used: Dict[str, float] = field(default_factory=dict)
    log_position: int = 0


@dataclass
class ControllerConfig:
    namespace: str = "default"
    sleeper_limit: int = 1
    num_workers: int = 2
    accelerator_sleeping_memory_limit_mib: Optional[int] = None
    launcher_port: int = contracts.LAUNCHER_SERVICE_PORT


class DualPodsController:
    def __init__(self, store: MemStore, http, config: ControllerConfig = None,
                 clock=time):
        self.store = store
        install_pod_indexes(store)  # O(1) lookups (ref controller.go:129-159)
        self.http = http
        self.cfg = config or ControllerConfig()
        self.clock = clock
        self.ns = self.cfg.namespace
        self.server_data: Dict[str, ServerData] = {}
        self.node_locks: Dict[str, threading.Lock] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        # two-level per-node queue (reference controller.go:404-424):
        # the node is element [1] of every item tuple; one worker drains
        # one node at a time, ready items oldest-first
        self.workers = NodeQueueAndWorkers(
            "dualpods", self.cfg.num_workers, self._process,
            node_of=lambda item: item[1], max_backoff=2.0,
            metrics_name="dualpods")
        self._watch_thread: Optional[threading.Thread] = None

    # ------------------------------------------------------------------
    # wiring
    # ------------------------------------------------------------------

    def start(self) -> None:
        self.workers.start()
        self.resync()
        self._watch_thread = threading.Thread(target=self._watch_loop,
                                              daemon=True)
        self._watch_thread.start()

    def stop(self) -> None:
        self._stop.set()
        self.workers.stop()

    def resync(self) -> None:
        for pod in self.store.list("Pod", self.ns):
            self._enqueue_for(pod)
        for isc in self.store.list("InferenceServerConfig", self.ns):
            self._enqueue_isc_gc(ob.name_of(isc))

    def _watch_loop(self) -> None:
        # Informer loop with 410 recovery: when the bounded event history
        # evicts past our cursor, re-LIST (resync) and resume from the
        # list revision, as a Kubernetes reflector would.
        while not self._stop.is_set():
            since = self.store.list_revision()
            try:
                self._watch_once(since)
                return  # stop was set
            except RevisionTooOld:
                self.resync()

    def _watch_once(self, since: int) -> None:
        # replay everything from before start too (resync covered it);
        # only the kinds this controller reacts to (a KubeStore opens one
        # watch stream per kind — unneeded kinds waste apiserver slots)
        for ev in self.store.watch(
                since=since, stop=self._stop,
                kinds=["Pod", "InferenceServerConfig"]):
            if ev.kind == "Pod":
                self._enqueue_for(ev.obj)
            elif ev.kind == "InferenceServerConfig":
                metrics.isc_count().set(
                    len(self.store.list("InferenceServerConfig", self.ns)))
                if ev.type in ("MODIFIED", "DELETED"):
                    self._enqueue_isc_gc(ob.name_of(ev.obj))
                # requesters referencing it may now be actionable
                for pod in self.store.index_get(
                        "Pod", "inferenceserverconfig",
                        ob.name_of(ev.obj), self.ns):
                    self._enqueue_for(pod)

    # classification (reference careAbout, controller.go:580-593)
    def classify(self, pod: Dict[str, Any]) -> str:
        ann = ob.annotations_of(pod)
        lbl = ob.labels_of(pod)
        if ann.get(contracts.REQUESTER_ANNOTATION):
            return "provider"
        if lbl.get(contracts.COMPONENT_LABEL) == contracts.LAUNCHER_COMPONENT:
            return "launcher"
        if ann.get(contracts.INFERENCE_SERVER_CONFIG_ANNOTATION) or \
                ann.get(contracts.SERVER_PATCH_ANNOTATION):
            return "requester"
        return "ignore"

    def _enqueue_for(self, pod: Dict[str, Any]) -> None:
        kind = self.classify(pod)
        node = ob.pod_node_name(pod) or ob.labels_of(pod).get(
            contracts.NODE_NAME_LABEL, "")
        if kind == "requester":
            self.workers.queue.add(
                ("infsvr", node, ob.uid_of(pod), ob.name_of(pod)))
        elif kind == "provider":
            req = ob.annotations_of(pod).get(contracts.REQUESTER_ANNOTATION,
                                             "")
            parts = req.split(" ", 1)
            if len(parts) == 2:
                self.workers.queue.add(("infsvr", node, parts[0], parts[1]))
        elif kind == "launcher":
            self.workers.queue.add(("launcher", node, ob.name_of(pod)))

    def _enqueue_isc_gc(self, isc_name: str) -> None:
        nodes = {ob.pod_node_name(p)
                 for p in self.store.list("Pod", self.ns)
                 if self.classify(p) == "launcher"}
        for node in nodes:
            if node:
                self.workers.queue.add(("iscgc", node, isc_name))

    def _node_lock(self, node: str) -> threading.Lock:
        with self._lock:
            return self.node_locks.setdefault(node, threading.Lock())

    def _process(self, item: Tuple) -> bool:
        if self._stop.is_set():
            # a stopped controller must not keep writing: a replacement
            # (restart) may already be reconciling the same objects
            return DONE
        kind, node = item[0], item[1]
        with self._node_lock(node):
            if kind == "infsvr":
                return self.reconcile_inference_server(node, item[2], item[3])
            if kind == "launcher":
                return self.reconcile_unbound_launcher(node, item[2])
            if kind == "iscgc":
                return self.reconcile_isc_gc(node, item[2])
        return DONE

    # ------------------------------------------------------------------
    # lookups
    # ------------------------------------------------------------------

    def _find_provider_for(self, uid: str, name: str
                           ) -> Optional[Dict[str, Any]]:
        want = f"{uid} {name}"
        hits = self.store.index_get("Pod", "requester", want, self.ns)
        return hits[0] if hits else None

    def _launchers_on_node(self, node: str, lc_name: Optional[str] = None
                           ) -> List[Dict[str, Any]]:
        out = []
        for pod in self.store.index_get("Pod", "launcherNode", node, self.ns):
            if lc_name and ob.labels_of(pod).get(
                    contracts.LAUNCHER_CONFIG_NAME_LABEL) != lc_name:
                continue
            out.append(pod)
        return out

    def _sdata(self, uid: str, name: str) -> ServerData:
        with self._lock:
            if uid not in self.server_data:
                self.server_data[uid] = ServerData(uid, name)
            return self.server_data[uid]

    def _launcher_client(self, pod: Dict[str, Any]) -> LauncherClient:
        ip = ob.pod_ip(pod)
        return LauncherClient(self.http,
                              f"http://{ip}:{self.cfg.launcher_port}")

    def _stub_url(self, requester: Dict[str, Any]) -> str:
        port = ob.annotations_of(requester).get(
            contracts.ADMIN_PORT_ANNOTATION, contracts.ADMIN_PORT_DEFAULT)
        return f"http://{ob.pod_ip(requester)}:{port}"

    # ------------------------------------------------------------------
    # the inference-server reconcile (reference inference-server.go:171-763)
    # ------------------------------------------------------------------

    def reconcile_inference_server(self, node: str, uid: str, name: str
                                   ) -> bool:
        requester = self.store.try_get("Pod", name, self.ns)
        if requester is not None and ob.uid_of(requester) != uid:
            requester = None  # a different incarnation; treat old as gone

        if requester is None:
            # requester gone entirely: unbind any provider still pointing
            # at it (provider-finalizer case, reference :257-290)
            provider = self._find_provider_for(uid, name)
            if provider is not None:
                return self._ensure_unbound(provider, None)
            with self._lock:
                self.server_data.pop(uid, None)
            return DONE

        if ob.is_deleting(requester):
            provider = self._find_provider_for(uid, name)
            if provider is not None:
                retry = self._ensure_unbound(provider, requester)
                if retry:
                    return RETRY
            return self._remove_requester_finalizer(requester)

        # mirror exogenous provider deletion: provider was bound but
        # vanished while we thought the server was up (reference :257-290)
        # -- detected below when bound lookup fails while sdata says bound.

        if not self._ensure_requester_finalizer(requester):
            return RETRY

        node = ob.pod_node_name(requester)
        if not node:
            return DONE  # not scheduled yet; Pod update will re-enqueue

        sdata = self._sdata(uid, name)
        if sdata.gpus is None:
            gpus = self._query_gpus(requester)
            if gpus is None:
                return WAIT  # stub not answering yet
            sdata.gpus = gpus
        metrics.requester_count().set(sum(
            1 for p in self.store.list("Pod", self.ns)
            if self.classify(p) == "requester"))

        ann = ob.annotations_of(requester)
        if ann.get(contracts.INFERENCE_SERVER_CONFIG_ANNOTATION):
            return self._reconcile_launcher_based(node, requester, sdata)
        return self._reconcile_direct(node, requester, sdata)

    def _query_gpus(self, requester: Dict[str, Any]) -> Optional[List[str]]:
        if not ob.pod_ip(requester):
            return None
        r = self.http.request(
            "GET", self._stub_url(requester) + contracts.ACCELERATOR_QUERY_PATH,
            purpose="gpu-query")
        if not r.ok or not isinstance(r.body, list):
            return None
        return [str(u) for u in r.body]

    def _set_requester_status(self, requester: Dict[str, Any],
                              errors: List[str]) -> None:
        import json
        cur = self.store.try_get("Pod", ob.name_of(requester), self.ns)
        if cur is None:
            return
        want = json.dumps({"Errors": errors}) if errors else None
        anns = ob.annotations_of(cur)
        if errors:
            if anns.get(contracts.STATUS_ANNOTATION) == want:
                return
            anns[contracts.STATUS_ANNOTATION] = want
        else:
            if contracts.STATUS_ANNOTATION not in anns:
                return
            anns.pop(contracts.STATUS_ANNOTATION)
        try:
            self.store.update(cur, actor="dual-pods-controller")
        except (Conflict, NotFound):
            pass

    # -- launcher-based path (reference :672-762, 425-601) ---------------

    def _reconcile_launcher_based(self, node: str, requester: Dict[str, Any],
                                  sdata: ServerData) -> bool:
        isc_name = ob.annotations_of(requester)[
            contracts.INFERENCE_SERVER_CONFIG_ANNOTATION]
        isc = self.store.try_get("InferenceServerConfig", isc_name, self.ns)
        if isc is None:
            self._set_requester_status(
                requester, [f"InferenceServerConfig {isc_name!r} not found"])
            return DONE  # ISC event will re-enqueue
        msc = isc["spec"]["modelServerConfig"]
        lc_name = isc["spec"].get("launcherConfigName", "")
        iid = compute_iid(msc, sdata.gpus or [])
        sdata.instance_id = iid
        sdata.port = int(msc["port"])

        uid, name = ob.uid_of(requester), ob.name_of(requester)
        provider = self._find_provider_for(uid, name)
        if provider is None:
            if self._was_bound(requester):
                # the provider vanished while bound: mirror the deletion so
                # the requester's owner re-creates it (reference
                # inference-server.go:257-290, docs/dual-pods.md:750-769)
                return self._mirror_provider_deletion(requester)
            return self._bind_to_launcher(node, requester, sdata, isc, lc_name)
        return self._run_bound_launcher(node, requester, provider, sdata,
                                        isc)

    def _was_bound(self, requester: Dict[str, Any]) -> bool:
        """The dual label on the requester is the durable record that a
        binding existed (survives controller restarts)."""
        return contracts.DUAL_LABEL in ob.labels_of(requester)

    def _mirror_provider_deletion(self, requester: Dict[str, Any]) -> bool:
        self.http.request(
            "POST", self._stub_url(requester) + contracts.BECOME_UNREADY_PATH,
            purpose="become-unready")
        try:
            self.store.delete("Pod", ob.name_of(requester), self.ns,
                              actor="dual-pods-controller",
                              expect_uid=ob.uid_of(requester))
        except (NotFound, Conflict):
            pass
        return RETRY  # continue into the deletion flow (finalizer release)

    def _launcher_views(self, pods: List[Dict[str, Any]], sdata: ServerData
                        ) -> Tuple[List[LauncherView], bool]:
        views = []
        any_unready = False
        for pod in pods:
            lv = LauncherView(
                name=ob.name_of(pod), pod=pod,
                ready=bool(ob.pod_ip(pod)) and ob.pod_is_ready(pod),
                bound=bool(ob.annotations_of(pod).get(
                    contracts.REQUESTER_ANNOTATION)),
                failed=ob.pod_phase(pod) == "Failed",
                deleting=ob.is_deleting(pod),
                max_instances=int(ob.annotations_of(pod).get(
                    "dual-pods.llm-d.ai/max-instances", "1")))
            if lv.ready and not lv.bound and not lv.failed and not lv.deleting:
                r = self._launcher_client(pod).list_instances()
                if not r.ok:
                    lv.ready = False
                    any_unready = True
                else:
                    for inst in r.body.get("instances", []):
                        if inst["instance_id"] in sdata.deleted_instances:
                            continue
                        port = _instance_port(inst)
                        lv.instances.append(InstanceView(
                            instance_id=inst["instance_id"],
                            status=inst.get("status", "stopped"),
                            port=port,
                            last_used=sdata.instance_last_used.get(
                                inst["instance_id"], 0.0)))
            views.append(lv)
        return views, any_unready

    def _bind_to_launcher(self, node: str, requester: Dict[str, Any],
                          sdata: ServerData, isc: Dict[str, Any],
                          lc_name: str) -> bool:
        pods = self._launchers_on_node(node, lc_name or None)
        views, _ = self._launcher_views(pods, sdata)
        sel = select_or_reclaim(views, sdata.instance_id, sdata.port,
                                sdata.instance_last_used)
        for lname, iid in sel.deletions:
            pod = next(p for p in pods if ob.name_of(p) == lname)
            r = self._launcher_client(pod).delete_instance(iid)
            if r.ok or r.status == 404:
                sdata.deleted_instances.add(iid)
        if sel.retry:
            return WAIT  # launchers not ready / repairs pending
        if sel.launcher is None:
            created = self._create_launcher(node, isc)
            sdata.needed_new_launcher = True
            return WAIT if created else DONE
        if not sel.has_sleeping_instance:
            sdata.needed_new_instance = True
        return self._bind(requester, sel.launcher.pod, sdata, isc)

    def _create_launcher(self, node: str, isc: Dict[str, Any]) -> bool:
        """Cold start: create a launcher Pod from the LauncherConfig
        template (the populator normally pre-creates them)."""
        lc_name = isc["spec"].get("launcherConfigName", "")
        lc = self.store.try_get("LauncherConfig", lc_name, self.ns)
        if lc is None:
            return False
        from fma_amd.controller.populator.podtemplate import \
            build_launcher_pod
        pod = build_launcher_pod(lc, node,
                                 name_suffix=str(int(self.clock.time() * 1e3)))
        t0 = self.clock.time()
        try:
            self.store.create(pod, actor="dual-pods-controller")
        except Conflict:
            pass
        metrics.launcher_create_seconds().observe(self.clock.time() - t0)
        return True

    def _bind(self, requester: Dict[str, Any], launcher_pod: Dict[str, Any],
              sdata: ServerData, isc: Dict[str, Any]) -> bool:
        """One atomic update makes the launcher Pod the record of the
        binding (reference bind, inference-server.go:1431-1484)."""
        import json
        cur = self.store.try_get("Pod", ob.name_of(launcher_pod), self.ns)
        if cur is None:
            return RETRY
        anns = ob.annotations_of(cur)
        if anns.get(contracts.REQUESTER_ANNOTATION):
            return RETRY  # someone else won; re-evaluate
        uid, name = ob.uid_of(requester), ob.name_of(requester)
        anns[contracts.REQUESTER_ANNOTATION] = f"{uid} {name}"
        anns[contracts.INSTANCE_ID_ANNOTATION] = sdata.instance_id
        anns[contracts.SERVER_PORT_ANNOTATION] = str(sdata.port)
        anns[contracts.SERVER_CONFIG_ANNOTATION] = json.dumps(
            self._server_config(isc, sdata), sort_keys=True)
        anns[contracts.ISC_ROUTING_METADATA_ANNOTATION] = json.dumps({
            "labels": isc["spec"]["modelServerConfig"].get("labels", {}),
            "annotations": isc["spec"]["modelServerConfig"].get(
                "annotations", {}),
        }, sort_keys=True)
        anns[contracts.LAUNCHER_BASED_ANNOTATION] = "true"
        ob.labels_of(cur)[contracts.DUAL_LABEL] = name
        fin = ob.finalizers_of(cur)
        if PROVIDER_FINALIZER not in fin:
            fin.append(PROVIDER_FINALIZER)
        try:
            self.store.update(cur, actor="dual-pods-controller",
                              expect_rv=ob.rv_of(cur))
        except Conflict:
            return RETRY
        logger.info("bound requester %s to launcher %s (instance %s)",
                    name, ob.name_of(cur), sdata.instance_id)
        # FYI labels on the requester (reference :1431-1484)
        self._apply_requester_fyi(requester, ob.name_of(cur),
                                  sdata.instance_id)
        return RETRY  # continue on the bound path next pass

    def _apply_requester_fyi(self, requester: Dict[str, Any],
                             provider_name: str, iid: Optional[str]) -> None:
        cur = self.store.try_get("Pod", ob.name_of(requester), self.ns)
        if cur is None or ob.uid_of(cur) != ob.uid_of(requester):
            return
        lbl = ob.labels_of(cur)
        lbl[contracts.DUAL_LABEL] = provider_name
        if iid:
            lbl[contracts.INSTANCE_LABEL] = iid[:63]
        anns = ob.annotations_of(cur)
        sd = self.server_data.get(ob.uid_of(requester))
        if sd and sd.gpus:
            anns[contracts.ACCELERATORS_ANNOTATION] = ",".join(sd.gpus)
        try:
            self.store.update(cur, actor="dual-pods-controller")
        except (Conflict, NotFound):
            pass

    def _server_config(self, isc: Dict[str, Any], sdata: ServerData
                       ) -> Dict[str, Any]:
        msc = isc["spec"]["modelServerConfig"]
        return {
            "options": f"{msc.get('options', '')} --port {msc['port']}".strip(),
            "gpu_uuids": sdata.gpus or [],
            "env_vars": msc.get("env_vars", {}),
            "annotations": {
                contracts.CONFIG_ISC_NAME_KEY: ob.name_of(isc),
                contracts.CONFIG_INFERENCE_PORT_KEY: str(msc["port"]),
            },
        }

    def _run_bound_launcher(self, node: str, requester: Dict[str, Any],
                            provider: Dict[str, Any], sdata: ServerData,
                            isc: Dict[str, Any]) -> bool:
        self._warn_outdated_routing(provider, isc)
        client = self._launcher_client(provider)
        r = client.get_instance(sdata.instance_id)
        if r.status == 404:
            cfg = self._server_config(isc, sdata)
            cr = client.create_named_instance(sdata.instance_id, cfg)
            if cr.status not in (201, 409):
                return RETRY
            sdata.needed_new_instance = True
            return WAIT  # instance process is booting
        if not r.ok:
            return WAIT
        if r.body.get("status") == contracts.INSTANCE_STATUS_STOPPED:
            # bound instance died: delete the requester so its owner
            # re-creates it (reference :454-507)
            try:
                self.store.delete("Pod", ob.name_of(requester), self.ns,
                                  actor="dual-pods-controller")
            except NotFound:
                pass
            return DONE

        # instance is running: is it asleep?
        ip = ob.pod_ip(provider)
        base = f"http://{ip}:{sdata.port}"
        sr = self.http.request("GET", base + contracts.IS_SLEEPING_PATH,
                               purpose="query-sleeping")
        if not sr.ok:
            return WAIT  # server still starting
        if sr.body.get("is_sleeping"):
            if not self._accel_memory_low_enough(requester, sdata):
                return RETRY
            wr = self.http.request("POST", base + contracts.WAKE_UP_PATH,
                                   purpose="wake")
            if not wr.ok:
                return RETRY
            logger.info("woke instance %s on %s (%.3fs)", sdata.instance_id,
                        ob.name_of(provider),
                        float(wr.body.get("seconds", 0) or 0)
                        if isinstance(wr.body, dict) else 0.0)
        self._apply_bound_labels(provider, isc)
        sdata.instance_last_used[sdata.instance_id] = self.clock.time()
        self._relay_instance_log(requester, provider, sdata)
        return self._relay_readiness(requester, sdata)

    def _relay_instance_log(self, requester: Dict[str, Any],
                            provider: Dict[str, Any],
                            sdata: ServerData) -> None:
        """Pipe new server-log bytes to the requester's /v1/set-log sink so
        the requesting Pod surfaces its server's output (the consumer of
        the SPI's SetLogPath contract, reference pkg/spi/interface.go:52-66;
        position-deduplicated on the stub side, so re-relays are
        harmless)."""
        iid = sdata.instance_id
        if not iid:
            return
        client = self._launcher_client(provider)
        r = self.http.request(
            "GET", f"{client.base}/v2/vllm/instances/{iid}/log",
            purpose="log-fetch",
            headers={"Range": f"bytes={sdata.log_position}-"})
        if r.status not in (200, 206) or not isinstance(r.body, str) \
                or not r.body:
            return
        chunk = r.body
        sr = self.http.request(
            "POST", self._stub_url(requester) + contracts.SET_LOG_PATH,
            purpose="log-relay",
            params={contracts.LOG_START_POS_PARAM: sdata.log_position},
            content=chunk.encode("utf-8", "ignore"))
        if sr.ok:
            sdata.log_position += len(chunk.encode("utf-8", "ignore"))

    def _warn_outdated_routing(self, provider: Dict[str, Any],
                               isc: Dict[str, Any]) -> None:
        """Routing metadata is frozen at bind time; if the ISC changed
        since, surface it as an Event (reference docs/dual-pods.md:687-717:
        OutdatedRoutingMetadata)."""
        import json
        stored = ob.annotations_of(provider).get(
            contracts.ISC_ROUTING_METADATA_ANNOTATION)
        msc_now = isc["spec"]["modelServerConfig"]
        fresh = json.dumps({"labels": msc_now.get("labels", {}),
                            "annotations": msc_now.get("annotations", {})},
                           sort_keys=True)
        if stored is not None and stored != fresh:
            from fma_amd.controller.events import (REASON_OUTDATED_ROUTING,
                                                   record_event)
            record_event(
                self.store, provider, REASON_OUTDATED_ROUTING,
                f"bound provider {ob.name_of(provider)} carries routing "
                f"metadata older than InferenceServerConfig "
                f"{ob.name_of(isc)}; it will refresh at next bind",
                actor="dual-pods-controller", namespace=self.ns)

    def _accel_memory_low_enough(self, requester: Dict[str, Any],
                                 sdata: ServerData) -> bool:
        limit = self.cfg.accelerator_sleeping_memory_limit_mib
        if limit is None:
            return True
        r = self.http.request(
            "GET",
            self._stub_url(requester) + contracts.ACCELERATOR_MEMORY_QUERY_PATH,
            purpose="accel-memory")
        if not r.ok or not isinstance(r.body, dict):
            return True
        return all(int(v) <= limit * (1 << 20) for v in r.body.values())

    def _apply_bound_labels(self, provider: Dict[str, Any],
                            isc: Dict[str, Any]) -> None:
        cur = self.store.try_get("Pod", ob.name_of(provider), self.ns)
        if cur is None:
            return
        lbl = ob.labels_of(cur)
        changed = lbl.get(contracts.SLEEPING_LABEL) != "false"
        lbl[contracts.SLEEPING_LABEL] = "false"
        msc = isc["spec"]["modelServerConfig"]
        for k, v in msc.get("labels", {}).items():
            if lbl.get(k) != v:
                lbl[k] = v
                changed = True
        anns = ob.annotations_of(cur)
        for k, v in msc.get("annotations", {}).items():
            if anns.get(k) != v:
                anns[k] = v
                changed = True
        if changed:
            try:
                self.store.update(cur, actor="dual-pods-controller")
            except (Conflict, NotFound):
                pass

    def _configure_proxy(self, requester: Dict[str, Any],
                         sdata: ServerData) -> None:
        """Point the requester's TCP reverse proxy at the serving endpoint
        so traffic addressed to the requester Pod reaches the provider
        (the reference's release-0.7 feature; stub side
        pkg/server/requester/proxy/server.go:39-217). 409 == already
        configured, which is fine (configure-once semantics)."""
        provider = self._find_provider_for(ob.uid_of(requester),
                                           ob.name_of(requester))
        if provider is None or not sdata.port:
            return
        ip = ob.pod_ip(provider)
        if not ip:
            return
        self.http.request(
            "PUT", self._stub_url(requester) + contracts.PROXY_CONFIG_PATH,
            purpose="proxy-config",
            json={"address": ip, "port": int(sdata.port)})

    def _relay_readiness(self, requester: Dict[str, Any], sdata: ServerData
                         ) -> bool:
        self._configure_proxy(requester, sdata)
        r = self.http.request(
            "POST", self._stub_url(requester) + contracts.BECOME_READY_PATH,
            purpose="become-ready")
        if not r.ok:
            return RETRY
        if not sdata.readiness_relayed:
            sdata.readiness_relayed = True
            path = "cold" if sdata.needed_new_launcher else \
                ("warm" if sdata.needed_new_instance else "hot")
            start = ob.pod_container_start_time(requester) or \
                self.clock.time()
            isc_name = ob.annotations_of(requester).get(
                contracts.INFERENCE_SERVER_CONFIG_ANNOTATION, "")
            metrics.actuation_seconds().labels(
                path, str(len(sdata.deleted_instances)), isc_name
            ).observe(max(self.clock.time() - start, 0.0))
            provider = self._find_provider_for(ob.uid_of(requester),
                                               ob.name_of(requester))
            if provider is not None:
                metrics.duality().labels(
                    ob.name_of(requester), ob.name_of(provider),
                    ob.pod_node_name(requester)).set(1)
        return DONE

    # -- direct (launcher-less) path (reference :618-669) ----------------

    def _gpu_indices(self, node: str, gpus: List[str]) -> Optional[List[int]]:
        import json
        cm = self.store.try_get("ConfigMap", contracts.GPU_MAP_CONFIGMAP,
                                self.ns)
        if cm is None:
            return None
        blob = cm.get("data", {}).get(node)
        if not blob:
            return None
        mapping = json.loads(blob)
        try:
            return [int(mapping[u]) for u in gpus]
        except KeyError:
            return None

    def _reconcile_direct(self, node: str, requester: Dict[str, Any],
                          sdata: ServerData) -> bool:
        uid, name = ob.uid_of(requester), ob.name_of(requester)
        provider = self._find_provider_for(uid, name)
        if provider is None and self._was_bound(requester):
            return self._mirror_provider_deletion(requester)
        patch = ob.annotations_of(requester)[contracts.SERVER_PATCH_ANNOTATION]
        indices = self._gpu_indices(node, sdata.gpus or [])
        if indices is None:
            self._set_requester_status(
                requester, [f"gpu-map has no entry for node {node!r}"])
            return RETRY
        try:
            desired, nom_hash = nominal_mod.build_nominal_provider(
                requester, patch, node, sdata.gpus or [], indices,
                provider_name=f"{name}-server")
        except nominal_mod.NominalError as e:
            self._set_requester_status(requester, [str(e)])
            return DONE
        self._set_requester_status(requester, [])

        if provider is not None:
            return self._run_bound_direct(requester, provider, sdata)

        # find a matching sleeper by nominal hash (reference :624-642)
        sleeper = None
        for pod in self.store.index_get("Pod", "nominal", nom_hash, self.ns):
            if not ob.annotations_of(pod).get(contracts.REQUESTER_ANNOTATION) \
                    and not ob.is_deleting(pod):
                sleeper = pod
                break
        if sleeper is not None:
            cur = sleeper
            anns = ob.annotations_of(cur)
            anns[contracts.REQUESTER_ANNOTATION] = f"{uid} {name}"
            ob.labels_of(cur)[contracts.DUAL_LABEL] = name
            fin = ob.finalizers_of(cur)
            if PROVIDER_FINALIZER not in fin:
                fin.append(PROVIDER_FINALIZER)
            try:
                self.store.update(cur, actor="dual-pods-controller",
                                  expect_rv=ob.rv_of(cur))
            except Conflict:
                return RETRY
            self._apply_requester_fyi(requester, ob.name_of(cur), None)
            return RETRY

        self._enforce_sleeper_budget(node, sdata)
        sdata.needed_new_launcher = True  # "cold" for metrics purposes
        try:
            self.store.create(desired, actor="dual-pods-controller")
        except Conflict:
            pass
        return RETRY

    def _run_bound_direct(self, requester: Dict[str, Any],
                          provider: Dict[str, Any], sdata: ServerData
                          ) -> bool:
        ip = ob.pod_ip(provider)
        if not ip or not ob.pod_is_ready(provider):
            return WAIT  # provider pod still starting
        port = _direct_server_port(provider)
        sdata.port = port  # for the proxy configuration at relay time
        base = f"http://{ip}:{port}"
        sr = self.http.request("GET", base + contracts.IS_SLEEPING_PATH,
                               purpose="query-sleeping")
        if not sr.ok:
            return RETRY
        if sr.body.get("is_sleeping"):
            wr = self.http.request("POST", base + contracts.WAKE_UP_PATH,
                                   purpose="wake")
            if not wr.ok:
                return RETRY
        cur = self.store.try_get("Pod", ob.name_of(provider), self.ns)
        if cur is not None and ob.labels_of(cur).get(
                contracts.SLEEPING_LABEL) != "false":
            ob.labels_of(cur)[contracts.SLEEPING_LABEL] = "false"
            try:
                self.store.update(cur, actor="dual-pods-controller")
            except (Conflict, NotFound):
                pass
        return self._relay_readiness(requester, sdata)

    def _enforce_sleeper_budget(self, node: str, sdata: ServerData) -> None:
        """Per-GPU cap on sleeping direct providers: for each GPU the new
        provider will occupy, delete the oldest sleepers using that GPU
        beyond sleeper_limit (reference enforceSleeperBudget,
        inference-server.go:1354-1428, which loops over GPUIndices; a
        node-wide count would evict 8x too aggressively on an 8-GPU node)."""
        def is_sleeper(p):
            return (ob.pod_node_name(p) == node
                    and not ob.annotations_of(p).get(
                        contracts.REQUESTER_ANNOTATION)
                    and not ob.is_deleting(p)
                    and ob.labels_of(p).get(contracts.SLEEPING_LABEL) == "true"
                    and ob.annotations_of(p).get(contracts.NOMINAL_ANNOTATION))

        victims: Dict[str, Dict[str, Any]] = {}
        for gpu in sdata.gpus or []:
            on_gpu = [p for p in self.store.index_get("Pod", "gpu", gpu,
                                                      self.ns)
                      if is_sleeper(p)]
            on_gpu.sort(key=lambda p: ob.meta(p).get("creationTimestamp") or 0)
            excess = len(on_gpu) - self.cfg.sleeper_limit
            for pod in on_gpu[:max(excess, 0)]:
                victims[ob.uid_of(pod)] = pod
        for pod in victims.values():
            try:
                self.store.delete("Pod", ob.name_of(pod), self.ns,
                                  actor="dual-pods-controller",
                                  expect_uid=ob.uid_of(pod))
            except (NotFound, Conflict):
                pass

    # ------------------------------------------------------------------
    # unbinding (reference ensureUnbound, :1667-1770)
    # ------------------------------------------------------------------

    def _ensure_unbound(self, provider: Dict[str, Any],
                        requester: Optional[Dict[str, Any]]) -> bool:
        import json
        cur = self.store.try_get("Pod", ob.name_of(provider), self.ns)
        if cur is None:
            return DONE
        anns = ob.annotations_of(cur)
        launcher_based = anns.get(contracts.LAUNCHER_BASED_ANNOTATION) == \
            "true" or ob.labels_of(cur).get(contracts.COMPONENT_LABEL) == \
            contracts.LAUNCHER_COMPONENT

        # 1. de-route BEFORE sleeping (reference :1680)
        routing = anns.get(contracts.ISC_ROUTING_METADATA_ANNOTATION)
        if routing:
            meta = json.loads(routing)
            lbl = ob.labels_of(cur)
            changed = False
            for k in meta.get("labels", {}):
                if k in lbl:
                    del lbl[k]
                    changed = True
            for k in meta.get("annotations", {}):
                if k in anns:
                    del anns[k]
                    changed = True
            if changed:
                try:
                    cur = self.store.update(cur, actor="dual-pods-controller")
                    anns = ob.annotations_of(cur)
                except (Conflict, NotFound):
                    return RETRY

        ip = ob.pod_ip(cur)
        port = anns.get(contracts.SERVER_PORT_ANNOTATION) or \
            _direct_server_port(cur)
        iid = anns.get(contracts.INSTANCE_ID_ANNOTATION)

        if launcher_based and iid:
            # obsolete instance (ISC changed since bind)? delete, else sleep
            if self._instance_is_obsolete(cur, iid):
                self._launcher_client(cur).delete_instance(iid)
            elif ip and port:
                r = self.http.request(
                    "POST", f"http://{ip}:{port}" + contracts.SLEEP_PATH,
                    purpose="sleep", params={"level": 1})
                if not r.ok and r.status != 0:
                    return RETRY
        elif not launcher_based and ip and port:
            self.http.request(
                "POST", f"http://{ip}:{port}" + contracts.SLEEP_PATH,
                purpose="sleep", params={"level": 1})

        # 2. one update clearing the binding (reference :1721-1765)
        anns.pop(contracts.REQUESTER_ANNOTATION, None)
        anns.pop(contracts.INSTANCE_ID_ANNOTATION, None)
        anns.pop(contracts.SERVER_PORT_ANNOTATION, None)
        anns.pop(contracts.SERVER_CONFIG_ANNOTATION, None)
        anns.pop(contracts.ISC_ROUTING_METADATA_ANNOTATION, None)
        lbl = ob.labels_of(cur)
        lbl.pop(contracts.DUAL_LABEL, None)
        lbl[contracts.SLEEPING_LABEL] = "true"
        fin = ob.finalizers_of(cur)
        if PROVIDER_FINALIZER in fin:
            fin.remove(PROVIDER_FINALIZER)
        try:
            self.store.update(cur, actor="dual-pods-controller",
                              expect_rv=ob.rv_of(cur))
        except Conflict:
            return RETRY
        except NotFound:
            pass
        logger.info("unbound provider %s (instance slept or deleted)",
                    ob.name_of(cur))
        if requester is not None:
            metrics.duality().labels(
                ob.name_of(requester), ob.name_of(cur),
                ob.pod_node_name(cur)).set(0)
        return DONE

    def _instance_is_obsolete(self, launcher_pod: Dict[str, Any],
                              iid: str) -> bool:
        """ISC changed after bind: the stored instance id no longer matches
        a fresh hash of the referenced ISC + stored GPUs
        (reference maybeDeleteObsoleteInstance, :1777-1836)."""
        import json
        anns = ob.annotations_of(launcher_pod)
        blob = anns.get(contracts.SERVER_CONFIG_ANNOTATION)
        if not blob:
            return False
        cfg = json.loads(blob)
        isc_name = cfg.get("annotations", {}).get(contracts.CONFIG_ISC_NAME_KEY)
        if not isc_name:
            return False
        isc = self.store.try_get("InferenceServerConfig", isc_name, self.ns)
        if isc is None:
            return True
        fresh = compute_iid(isc["spec"]["modelServerConfig"],
                            cfg.get("gpu_uuids", []))
        return fresh != iid

    # ------------------------------------------------------------------
    # finalizers
    # ------------------------------------------------------------------

    def _ensure_requester_finalizer(self, requester: Dict[str, Any]) -> bool:
        cur = self.store.try_get("Pod", ob.name_of(requester), self.ns)
        if cur is None:
            return False
        fin = ob.finalizers_of(cur)
        if REQUESTER_FINALIZER in fin:
            return True
        fin.append(REQUESTER_FINALIZER)
        try:
            self.store.update(cur, actor="dual-pods-controller")
            return True
        except (Conflict, NotFound):
            return False

    def _remove_requester_finalizer(self, requester: Dict[str, Any]) -> bool:
        cur = self.store.try_get("Pod", ob.name_of(requester), self.ns)
        if cur is None:
            return DONE
        fin = ob.finalizers_of(cur)
        if REQUESTER_FINALIZER not in fin:
            return DONE
        fin.remove(REQUESTER_FINALIZER)
        try:
            self.store.update(cur, actor="dual-pods-controller")
        except Conflict:
            return RETRY
        except NotFound:
            pass
        with self._lock:
            self.server_data.pop(ob.uid_of(requester), None)
        return DONE

    # ------------------------------------------------------------------
    # unbound-launcher sync + ISC GC
    # ------------------------------------------------------------------

    def reconcile_unbound_launcher(self, node: str, pod_name: str) -> bool:
        """Delete stopped instances on unbound launchers; keep the sleeping
        label truthful (reference syncLauncherInstances, :2091-2183)."""
        pod = self.store.try_get("Pod", pod_name, self.ns)
        if pod is None or ob.is_deleting(pod):
            return DONE
        if ob.annotations_of(pod).get(contracts.REQUESTER_ANNOTATION):
            return DONE  # bound: handled by its infsvr item
        if not ob.pod_ip(pod) or not ob.pod_is_ready(pod):
            return DONE
        client = self._launcher_client(pod)
        r = client.list_instances()
        if not r.ok:
            return DONE
        any_awake = False
        for inst in r.body.get("instances", []):
            if inst.get("status") == contracts.INSTANCE_STATUS_STOPPED:
                client.delete_instance(inst["instance_id"])
                continue
            # an unbound launcher's instances should all be sleeping; if a
            # server is awake (e.g. controller restarted mid-unbind), put
            # it back to sleep
            port = _instance_port(inst)
            if port:
                ip = ob.pod_ip(pod)
                sr = self.http.request(
                    "GET", f"http://{ip}:{port}" + contracts.IS_SLEEPING_PATH,
                    purpose="query-sleeping")
                if sr.ok and not sr.body.get("is_sleeping"):
                    self.http.request(
                        "POST", f"http://{ip}:{port}" + contracts.SLEEP_PATH,
                        purpose="sleep", params={"level": 1})
                    any_awake = True
        # every instance on an unbound launcher is (now) sleeping
        if ob.labels_of(pod).get(contracts.SLEEPING_LABEL) != "true":
            cur = self.store.try_get("Pod", pod_name, self.ns)
            if cur is not None:
                ob.labels_of(cur)[contracts.SLEEPING_LABEL] = "true"
                try:
                    self.store.update(cur, actor="dual-pods-controller")
                except (Conflict, NotFound):
                    pass
        return DONE

    def reconcile_isc_gc(self, node: str, isc_name: str) -> bool:
        """Delete sleeping, unbound instances made obsolete by an ISC spec
        change (reference instanceGCItem.process, :1587-1664)."""
        isc = self.store.try_get("InferenceServerConfig", isc_name, self.ns)
        for pod in self._launchers_on_node(node):
            if ob.annotations_of(pod).get(contracts.REQUESTER_ANNOTATION):
                continue
            if not ob.pod_ip(pod) or not ob.pod_is_ready(pod):
                continue
            client = self._launcher_client(pod)
            r = client.list_instances()
            if not r.ok:
                continue
            for inst in r.body.get("instances", []):
                inst_isc = (inst.get("annotations") or {}).get(
                    contracts.CONFIG_ISC_NAME_KEY)
                if inst_isc != isc_name:
                    continue
                if isc is None:
                    client.delete_instance(inst["instance_id"])
                    continue
                fresh = compute_iid(isc["spec"]["modelServerConfig"],
                                    inst.get("gpu_uuids") or [])
                if fresh != inst["instance_id"]:
                    client.delete_instance(inst["instance_id"])
        return DONE


def _instance_port(inst: Dict[str, Any]) -> Optional[int]:
    """Usable inference port of an instance (reference
    getVLLMInstancePort): the inference-port annotation, else --port in
    the options string; None marks the instance malformed."""
    ann = inst.get("annotations") or {}
    if contracts.CONFIG_INFERENCE_PORT_KEY in ann:
        try:
            return int(ann[contracts.CONFIG_INFERENCE_PORT_KEY])
        except (TypeError, ValueError):
            return None
    opts = (inst.get("options") or "").split()
    for i, tok in enumerate(opts):
        if tok == "--port" and i + 1 < len(opts):
            try:
                return int(opts[i + 1])
            except ValueError:
                return None
        if tok.startswith("--port="):
            try:
                return int(tok.split("=", 1)[1])
            except ValueError:
                return None
    return None


def _direct_server_port(pod: Dict[str, Any]) -> int:
    for c in ob.pod_containers(pod):
        if c.get("name") == contracts.INFERENCE_SERVER_CONTAINER:
            for p in c.get("ports", []):
                if p.get("containerPort"):
                    return int(p["containerPort"])
    return 8000
