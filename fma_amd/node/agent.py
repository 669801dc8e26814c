"""Node agent: the kubelet analog of the single-node deployment.

Watches the cluster store for Pods scheduled to this node and materializes
them as local processes. Each Pod gets its own loopback IP (127.0.<node>.N)
so every Pod has a private network identity on one host — launchers all
bind :8001 on their own IP, instances bind the ISC port on the launcher's
IP, exactly like per-Pod network namespaces in the reference's Kubernetes
deployment.

Responsibilities:
- spawn the Pod's main container command (launcher / requester stub /
  direct serving provider) with FMA_BIND_HOST + env from the container
  spec; the process group is killed when the Pod object goes away;
- publish status: podIP, phase Running, Ready from the readinessProbe
  (or process liveness when no probe is declared);
- run the notifier duty for launcher Pods (the reference's sidecar
  container) so instance-state changes become Pod annotation updates the
  controller's informer sees.
"""

from __future__ import annotations

import os
import signal
import subprocess
import sys
import threading
import time
from typing import Any, Dict, Optional

import httpx

from fma_amd.api import contracts
from fma_amd.launcher.notifier import PodNotifier
from fma_amd.store import objects as ob
from fma_amd.store.memstore import Conflict, NotFound, RevisionTooOld


class PodProcess:
    def __init__(self, pod: Dict[str, Any], ip: str,
                 proc: subprocess.Popen, probe: Optional[Dict[str, Any]]):
        self.pod_name = ob.name_of(pod)
        self.uid = ob.uid_of(pod)
        self.ip = ip
        self.proc = proc
        self.probe = probe
        self.restarts = 0
        self.started_at = time.time()
        self.next_restart = 0.0
        # k8s default is Always; "Never" lets a crash surface as Failed
        self.restart_policy = (pod.get("spec") or {}).get(
            "restartPolicy", "Always")
        self.notifier: Optional[PodNotifier] = None
        self.notifier_thread: Optional[threading.Thread] = None


class NodeAgent:
    def __init__(self, store, node_name: str, node_index: int = 1,
                 namespace: str = "default", log_dir: str = "/tmp",
                 extra_env: Optional[Dict[str, str]] = None):
        self.store = store
        self.node = node_name
        self.node_index = node_index
        self.ns = namespace
        self.log_dir = log_dir
        self.extra_env = dict(extra_env or {})
        self.pods: Dict[str, PodProcess] = {}
        self._next_ip = 2
        self._stop = threading.Event()
        self._threads = []

    # -- lifecycle -----------------------------------------------------------

    def start(self) -> None:
        for fn in (self._watch_loop, self._status_loop):
            t = threading.Thread(target=fn, daemon=True)
            t.start()
            self._threads.append(t)

    def stop(self) -> None:
        self._stop.set()
        for pp in list(self.pods.values()):
            self._kill(pp)
        self.pods.clear()

    def _alloc_ip(self) -> str:
        ip = f"127.{self.node_index}.{self._next_ip // 250}.{self._next_ip % 250 + 2}"
        self._next_ip += 1
        return ip

    # -- reconcile -----------------------------------------------------------

    def _watch_loop(self) -> None:
        while not self._stop.is_set():
            self._sync_all()
            try:
                for ev in self.store.watch(since=self.store.list_revision(),
                                           kinds=["Pod"], stop=self._stop):
                    self._handle(ev.type, ev.obj)
                return  # stop was set
            except RevisionTooOld:
                continue  # re-LIST and resume from the list revision

    def _sync_all(self) -> None:
        for pod in self.store.list("Pod", self.ns):
            self._handle("ADDED", pod)

    def _handle(self, ev_type: str, pod: Dict[str, Any]) -> None:
        name = ob.name_of(pod)
        mine = ob.pod_node_name(pod) == self.node
        if ev_type == "DELETED" or not mine or ob.is_deleting(pod):
            # deleting pods keep running until finalizers clear and the
            # object is gone (kubelet kills on deletion); we kill when the
            # object disappears OR is marked deleting with no containers
            if ev_type == "DELETED" and name in self.pods:
                self._kill(self.pods.pop(name))
            return
        if name in self.pods:
            existing = self.pods[name]
            if existing.uid != ob.uid_of(pod):
                self._kill(self.pods.pop(name))
            else:
                return
        cmd, env, probe = self._container_plan(pod)
        if cmd is None:
            return
        self._spawn(pod, cmd, env, probe)

    def _container_plan(self, pod: Dict[str, Any]):
        """Derive (command, env, readinessProbe) for the Pod's main
        container. Containers without an explicit command are skipped
        (images do not run here — commands name python modules)."""
        containers = ob.pod_containers(pod)
        # role priority: a provider's inference-server outranks the stub
        # container it inherited from the requester spec
        priority = (contracts.INFERENCE_SERVER_CONTAINER, "launcher",
                    "requester", "main", "stub")
        main = None
        for name in priority:
            for c in containers:
                if c.get("name") == name and c.get("command"):
                    main = c
                    break
            if main:
                break
        if main is None:
            main = next((c for c in containers if c.get("command")), None)
        if not main:
            return None, None, None
        env = {e["name"]: str(e.get("value", ""))
               for e in main.get("env", []) if "name" in e}
        return list(main["command"]) + list(main.get("args", [])), env, \
            main.get("readinessProbe")

    def _spawn(self, pod: Dict[str, Any], cmd, env: Dict[str, str],
               probe) -> None:
        ip = self._alloc_ip()
        full_env = dict(os.environ)
        full_env.update(self.extra_env)
        full_env.update(env)
        full_env["FMA_BIND_HOST"] = ip
        full_env["POD_IP"] = ip
        full_env["NODE_NAME"] = self.node
        full_env["POD_NAME"] = ob.name_of(pod)
        # unique per-pod marker inherited by EVERY descendant: the kill
        # path reaps by scanning /proc for it, the way a kubelet's cgroup
        # kill takes the whole tree. Without this, a crashed launcher
        # orphans its instance processes (they run in their own process
        # groups), which keep the model ports — and on real GPUs the
        # HBM — alive forever.
        full_env["FMA_POD_TREE"] = \
            f"{self.node}/{ob.name_of(pod)}/{ob.uid_of(pod)}"
        log_path = os.path.join(self.log_dir,
                                f"pod-{ob.name_of(pod)}.log")
        logf = open(log_path, "ab")
        preexec = None
        if os.environ.get("FMA_POD_PDEATHSIG") == "1":
            # test harness: pods die with the agent process, so a
            # hard-killed pytest run (timeout, SIGKILL) cannot leave
            # orphaned servers squatting on fixed ports for later runs
            def preexec():
                import ctypes
                ctypes.CDLL(None).prctl(1, signal.SIGKILL)  # PR_SET_PDEATHSIG
                os.setsid()
        proc = subprocess.Popen(cmd, env=full_env, stdout=logf,
                                stderr=subprocess.STDOUT,
                                start_new_session=preexec is None,
                                preexec_fn=preexec)
        pp = PodProcess(pod, ip, proc, probe)
        # kept for restartPolicy:Always respawns (kubelet semantics)
        pp.cmd = list(cmd)
        pp.full_env = full_env
        pp.log_path = log_path
        self.pods[ob.name_of(pod)] = pp
        self._patch_status(pp, phase="Running", ready=False, ip=ip)
        if ob.labels_of(pod).get(contracts.COMPONENT_LABEL) == \
                contracts.LAUNCHER_COMPONENT:
            self._start_notifier(pp)

    def _start_notifier(self, pp: PodProcess) -> None:
        """The state-change-reflector duty (reference sidecar,
        launcher_pod_notifier.py:135-194) run by the agent."""
        url = f"http://{pp.ip}:{contracts.LAUNCHER_SERVICE_PORT}"

        def patch(key: str, value: str) -> None:
            cur = self.store.try_get("Pod", pp.pod_name, self.ns)
            if cur is None or ob.uid_of(cur) != pp.uid:
                return
            try:
                # server-side merge patch: cannot lose a concurrent
                # controller update the way read-modify-write can
                # (the reference sidecar PATCHes for the same reason,
                # launcher_pod_notifier.py:135-194)
                self.store.patch(
                    "Pod", pp.pod_name,
                    {"metadata": {"annotations": {key: value}}}, self.ns,
                    actor="node-agent")
            except (Conflict, NotFound):
                pass

        pp.notifier = PodNotifier(url, patch, interval=0.5)
        t = threading.Thread(target=pp.notifier.run, daemon=True)
        t.start()
        pp.notifier_thread = t

    def _kill(self, pp: PodProcess) -> None:
        if pp.notifier:
            pp.notifier.stop()
        if pp.proc.poll() is None:
            try:
                os.killpg(pp.proc.pid, signal.SIGTERM)
            except (ProcessLookupError, PermissionError):
                pass
            try:
                pp.proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                try:
                    os.killpg(pp.proc.pid, signal.SIGKILL)
                except (ProcessLookupError, PermissionError):
                    pass
        self._reap_pod_tree(pp.pod_name, pp.uid)

    def _reap_pod_tree(self, pod_name: str, uid: str) -> int:
        """Kill every process carrying this pod's FMA_POD_TREE marker —
        the cgroup-kill analog. Catches instance processes that outlived
        a crashed launcher (they setpgrp into their own groups, so the
        launcher's killpg never reaches them). Exact-PID kills only, by
        environment marker, never by name pattern."""
        marker = f"{self.node}/{pod_name}/{uid}"
        needle = ("FMA_POD_TREE=" + marker).encode()
        victims = []
        me = os.getpid()
        for ent in os.listdir("/proc"):
            if not ent.isdigit() or int(ent) == me:
                continue
            try:
                with open(f"/proc/{ent}/environ", "rb") as f:
                    if needle in f.read().split(b"\0"):
                        victims.append(int(ent))
            except OSError:
                continue
        for pid in victims:
            try:
                os.kill(pid, signal.SIGTERM)
            except (ProcessLookupError, PermissionError):
                pass
        if victims:
            deadline = time.time() + 5
            while time.time() < deadline:
                if not any(os.path.exists(f"/proc/{p}") for p in victims):
                    break
                time.sleep(0.1)
            for pid in victims:
                try:
                    os.kill(pid, signal.SIGKILL)
                except (ProcessLookupError, PermissionError):
                    pass
        return len(victims)

    # -- status --------------------------------------------------------------

    def _probe_ready(self, pp: PodProcess) -> bool:
        if pp.proc.poll() is not None:
            return False
        probe = pp.probe
        if not probe or "httpGet" not in probe:
            return True  # no probe: process liveness is readiness
        hg = probe["httpGet"]
        url = f"http://{pp.ip}:{hg.get('port', 80)}{hg.get('path', '/')}"
        try:
            r = httpx.get(url, timeout=2)
            return r.status_code < 400
        except httpx.HTTPError:
            return False

    def _status_loop(self) -> None:
        while not self._stop.wait(0.15):
            for name, pp in list(self.pods.items()):
                exited = pp.proc.poll() is not None
                restartable = pp.restart_policy != "Never"
                if exited and pp.proc.returncode and restartable:
                    # restartPolicy Always (the k8s default the reference
                    # relies on: a crashed launcher container restarts in
                    # place and the dual-pods controller re-creates its
                    # instance). Reap the pod's whole process tree first
                    # so orphaned instances release their ports/HBM, then
                    # respawn with CrashLoop-style backoff. The Pod stays
                    # Running/NotReady through the gap, like kubelet.
                    now = time.time()
                    if pp.next_restart == 0.0:
                        pp.next_restart = now + min(
                            5.0, 0.5 * (2 ** pp.restarts))
                        self._patch_status(pp, phase="Running", ready=False,
                                           ip=pp.ip)
                        continue
                    if now < pp.next_restart:
                        continue
                    self._reap_pod_tree(pp.pod_name, pp.uid)
                    self._respawn(pp)
                    continue
                ready = False if exited else self._probe_ready(pp)
                phase = "Failed" if exited and pp.proc.returncode else \
                    ("Succeeded" if exited else "Running")
                self._patch_status(pp, phase=phase, ready=ready, ip=pp.ip)

    def _respawn(self, pp: PodProcess) -> None:
        logf = open(pp.log_path, "ab")
        logf.write(f"[agent] restart #{pp.restarts + 1} of "
                   f"{pp.pod_name}\n".encode())
        try:
            pp.proc = subprocess.Popen(pp.cmd, env=pp.full_env, stdout=logf,
                                       stderr=subprocess.STDOUT,
                                       start_new_session=True)
        except OSError:
            return
        pp.restarts += 1
        pp.started_at = time.time()
        pp.next_restart = 0.0
        self._patch_status(pp, phase="Running", ready=False, ip=pp.ip)

    def _patch_status(self, pp: PodProcess, phase: str, ready: bool,
                      ip: str) -> None:
        cur = self.store.try_get("Pod", pp.pod_name, self.ns)
        if cur is None or ob.uid_of(cur) != pp.uid:
            return
        status = cur.setdefault("status", {})
        old_restarts = (status.get("containerStatuses") or
                        [{}])[0].get("restartCount", 0)
        changed = (status.get("phase") != phase or
                   status.get("podIP") != ip or
                   old_restarts != pp.restarts or
                   ob.pod_is_ready(cur) != ready)
        if not changed:
            return
        status["phase"] = phase
        status["podIP"] = ip
        # kubelet-style restart accounting (PodIsInTrouble in the
        # reference, utils/pod-helper.go:44, keys off this)
        status["containerStatuses"] = [
            {"restartCount": pp.restarts, "ready": ready,
             # actuation latency is measured from container start, not Pod
             # creation (reference inference-server.go:574-591)
             "state": {"running": {"startedAt": pp.started_at}}}]
        ob.set_pod_ready(cur, ready)
        status.setdefault("startTime", ob.meta(cur).get("creationTimestamp"))
        try:
            self.store.update(cur, actor="node-agent", subresource="status")
        except (Conflict, NotFound):
            pass


def main() -> None:
    import argparse

    from fma_amd.store.client import StoreClient

    ap = argparse.ArgumentParser("fma-node-agent")
    ap.add_argument("--store-url", default="http://127.0.0.1:8081")
    ap.add_argument("--node-name", default=os.environ.get("NODE_NAME",
                                                          "node-1"))
    ap.add_argument("--node-index", type=int, default=1)
    ap.add_argument("--log-dir", default="/tmp")
    args = ap.parse_args()
    agent = NodeAgent(StoreClient(args.store_url, actor="node-agent"),
                      args.node_name, args.node_index,
                      log_dir=args.log_dir)
    agent.start()
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        agent.stop()
        sys.exit(0)


if __name__ == "__main__":
    main()
