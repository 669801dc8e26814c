"""Launcher Pod template construction (reference pkg/controller/utils/
pod-helper.go:143-400).

BuildNodeIndependentLauncherTemplate equivalent: canonicalize the
LauncherConfig's pod template (sorted named lists for a stable hash),
force the launcher identity labels, inject the launcher environment,
fixed liveness/readiness probes on :8001, zero GPU resources (launchers
see all GPUs via HIP_VISIBLE_DEVICES=all, bypassing the device plugin),
append the state-change-reflector notifier sidecar, and stamp the
node-independent template hash annotation. Node specialization pins the
hostname selector.
"""

from __future__ import annotations

import copy
from typing import Any, Dict, Optional

from fma_amd.api import contracts
from fma_amd.controller.dualpods.identity import template_hash
from fma_amd.store import objects as ob

LAUNCHER_CONTAINER = "launcher"
NOTIFIER_CONTAINER = "state-change-reflector"


def canonicalize_template(template: Dict[str, Any]) -> Dict[str, Any]:
    """Sort named lists (volumes, tolerations, ports, mounts, env) so
    semantically equal templates hash equal (reference
    canonicalizeTemplateForHash, pod-helper.go:143-197)."""
    t = copy.deepcopy(template)
    spec = t.setdefault("spec", {})

    def sort_named(lst, key="name"):
        if isinstance(lst, list) and all(isinstance(e, dict) for e in lst):
            return sorted(lst, key=lambda e: str(e.get(key, "")))
        return lst

    if "volumes" in spec:
        spec["volumes"] = sort_named(spec["volumes"])
    if "tolerations" in spec:
        spec["tolerations"] = sort_named(spec["tolerations"], key="key")
    for c in spec.get("containers", []):
        if "ports" in c:
            c["ports"] = sorted(
                c["ports"], key=lambda p: int(p.get("containerPort", 0)))
        if "volumeMounts" in c:
            c["volumeMounts"] = sort_named(c["volumeMounts"])
        if "env" in c:
            c["env"] = sort_named(c["env"])
    return t


def build_node_independent_template(lc: Dict[str, Any]) -> Dict[str, Any]:
    """reference BuildNodeIndependentLauncherTemplate, pod-helper.go:205-300."""
    spec_tmpl = lc["spec"].get("podTemplate", {})
    template: Dict[str, Any] = {
        "metadata": {
            "labels": dict(spec_tmpl.get("metadata", {}).get("labels", {})),
            "annotations": dict(
                spec_tmpl.get("metadata", {}).get("annotations", {})),
        },
        "spec": copy.deepcopy(spec_tmpl.get("spec", {})),
    }
    lbl = template["metadata"]["labels"]
    lbl[contracts.COMPONENT_LABEL] = contracts.LAUNCHER_COMPONENT
    lbl[contracts.LAUNCHER_CONFIG_NAME_LABEL] = ob.name_of(lc)
    lbl[contracts.SLEEPING_LABEL] = "true"

    spec = template["spec"]
    containers = spec.setdefault("containers", [])
    launcher = None
    for c in containers:
        if c.get("name") == LAUNCHER_CONTAINER:
            launcher = c
            break
    if launcher is None:
        launcher = {"name": LAUNCHER_CONTAINER,
                    "image": "fma-amd/launcher:latest",
                    "command": ["python", "-m", "fma_amd.launcher.service"]}
        containers.insert(0, launcher)
    # launcher environment (reference pod-helper.go:326-344, adapted to
    # ROCm: HIP_VISIBLE_DEVICES instead of NVIDIA_VISIBLE_DEVICES)
    ob.container_env_set(launcher, "PYTHONPATH", "/app")
    ob.container_env_set(launcher, contracts.VISIBLE_DEVICES_ENV,
                         contracts.ALL_DEVICES_ENV_VALUE)
    ob.container_env_set(launcher, "FMA_SERVER_DEV_MODE", "1")
    launcher["livenessProbe"] = {
        "httpGet": {"path": contracts.HEALTH_PATH,
                    "port": contracts.LAUNCHER_SERVICE_PORT},
        "periodSeconds": 10,
    }
    launcher["readinessProbe"] = {
        "httpGet": {"path": contracts.LAUNCHER_API_ROOT,
                    "port": contracts.LAUNCHER_SERVICE_PORT},
        "periodSeconds": 5,
    }
    # zero GPU extended resources (reference pod-helper.go:345-352)
    for section in ("limits", "requests"):
        res = launcher.setdefault("resources", {}).setdefault(section, {})
        res[contracts.GPU_RESOURCE_NAME] = "0"

    if not any(c.get("name") == NOTIFIER_CONTAINER for c in containers):
        containers.append({
            "name": NOTIFIER_CONTAINER,
            "image": launcher.get("image", "fma-amd/launcher:latest"),
            "command": ["python", "-m", "fma_amd.launcher.notifier"],
            "env": [
                {"name": "LAUNCHER_BASE_URL",
                 "value": f"http://127.0.0.1:{contracts.LAUNCHER_SERVICE_PORT}"},
            ],
        })

    canonical = canonicalize_template(template)
    h = template_hash(canonical)
    template["metadata"]["annotations"][
        contracts.LAUNCHER_TEMPLATE_HASH_ANNOTATION] = h
    template["metadata"]["annotations"][
        "dual-pods.llm-d.ai/max-instances"] = str(
            lc["spec"].get("maxInstances", 1))
    return template


def specialize_to_node(template: Dict[str, Any], node: str) -> Dict[str, Any]:
    """reference SpecializeLauncherTemplateToNode, pod-helper.go:303-322."""
    t = copy.deepcopy(template)
    spec = t.setdefault("spec", {})
    spec.setdefault("nodeSelector", {})["kubernetes.io/hostname"] = node
    spec["nodeName"] = node
    t["metadata"].setdefault("labels", {})[contracts.NODE_NAME_LABEL] = node
    return t


def build_launcher_pod(lc: Dict[str, Any], node: str,
                       name_suffix: Optional[str] = None) -> Dict[str, Any]:
    template = specialize_to_node(build_node_independent_template(lc), node)
    suffix = name_suffix or str(abs(hash((node, ob.uid_of(lc)))) % 100000)
    pod = ob.new_object(
        "Pod", f"{ob.name_of(lc)}-{node}-{suffix}"[:63],
        namespace=ob.namespace_of(lc),
        labels=template["metadata"]["labels"],
        annotations=template["metadata"]["annotations"],
        spec=template["spec"])
    ob.meta(pod)["ownerReferences"] = [{
        "apiVersion": f"{contracts.GROUP}/{contracts.VERSION}",
        "kind": "LauncherConfig",
        "name": ob.name_of(lc),
        "uid": ob.uid_of(lc),
    }]
    return pod
