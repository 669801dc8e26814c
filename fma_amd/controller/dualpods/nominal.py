"""Nominal server-providing Pod construction (direct, launcher-less path).

Reference: getNominalServerProvidingPod (pkg/controller/dual-pods/
inference-server.go:1843-1947). Steps preserved:

1. render the ``dual-pods.llm-d.ai/server-patch`` annotation as a template
   over ProviderData{NodeName, LocalVolume} (the reference uses Go
   text/template; we substitute the same ``{{.Field}}`` references);
2. strategic-merge-patch the result onto the de-individualized requester
   spec (containers and other named lists merge by their ``name`` key,
   exactly the part of Kubernetes strategic merge the contract uses);
3. pin nodeSelector to the requester's node, inject
   HIP_VISIBLE_DEVICES=<indices> into the inference-server container
   (CUDA_VISIBLE_DEVICES in the reference, :1917-1924), zero out the
   ``amd.com/gpu`` resource so the device plugin is bypassed (:1926-1934);
4. compute the nominal hash for sleeper lookup.
"""

from __future__ import annotations

import copy
import re
from typing import Any, Dict, List, Optional, Tuple

import yaml

from fma_amd.api import contracts
from fma_amd.controller.dualpods.identity import nominal_hash
from fma_amd.store import objects as ob
from fma_amd.store.merge import strategic_merge  # noqa: F401 (re-export)

_TEMPLATE_RE = re.compile(r"\{\{\s*\.(\w+)\s*\}\}")


class NominalError(Exception):
    pass


def render_template(text: str, provider_data: Dict[str, str]) -> str:
    def sub(m: re.Match) -> str:
        key = m.group(1)
        if key not in provider_data:
            raise NominalError(f"server-patch references unknown field .{key}")
        return str(provider_data[key])

    return _TEMPLATE_RE.sub(sub, text)


def deindividualize(requester: Dict[str, Any]) -> Dict[str, Any]:
    """Strip the requester's identity so the patch defines the provider:
    all annotations removed (per the contract comment, reference
    pkg/api/interface.go:40-44), names/uids/status cleared."""
    pod = ob.deepcopy(requester)
    meta = pod.get("metadata", {})
    meta.pop("annotations", None)
    meta.pop("uid", None)
    meta.pop("resourceVersion", None)
    meta.pop("creationTimestamp", None)
    meta.pop("deletionTimestamp", None)
    meta.pop("finalizers", None)
    meta.pop("ownerReferences", None)
    pod.pop("status", None)
    return pod


def build_nominal_provider(
        requester: Dict[str, Any],
        server_patch_yaml: str,
        node_name: str,
        gpu_uuids: List[str],
        gpu_indices: List[int],
        local_volume: str = "",
        provider_name: Optional[str] = None,
) -> Tuple[Dict[str, Any], str]:
    """Returns (provider_pod, nominal_hash)."""
    rendered = render_template(server_patch_yaml,
                               {"NodeName": node_name,
                                "LocalVolume": local_volume})
    try:
        patch = yaml.safe_load(rendered) or {}
    except yaml.YAMLError as e:
        raise NominalError(f"server-patch is not valid YAML: {e}") from e
    if not isinstance(patch, dict):
        raise NominalError("server-patch must be a mapping")

    base = deindividualize(requester)
    pod = strategic_merge(base, patch)
    meta = pod.setdefault("metadata", {})
    meta["name"] = provider_name or f"{ob.name_of(requester)}-server"
    meta["namespace"] = ob.namespace_of(requester)
    spec = pod.setdefault("spec", {})
    spec.setdefault("nodeSelector", {})["kubernetes.io/hostname"] = node_name
    spec["nodeName"] = node_name

    container = None
    for c in spec.get("containers", []):
        if c.get("name") == contracts.INFERENCE_SERVER_CONTAINER:
            container = c
            break
    if container is None:
        raise NominalError(
            f"patched spec has no {contracts.INFERENCE_SERVER_CONTAINER!r} "
            "container")
    ob.container_env_set(container, contracts.VISIBLE_DEVICES_ENV,
                         ",".join(str(i) for i in gpu_indices))
    # bypass the device plugin: the GPUs are already attributed to the
    # requester Pod (reference inference-server.go:1926-1934)
    for section in ("limits", "requests"):
        res = container.setdefault("resources", {}).setdefault(section, {})
        res[contracts.GPU_RESOURCE_NAME] = "0"

    h = nominal_hash(spec, gpu_uuids, node_name)
    anns = ob.annotations_of(pod)
    anns[contracts.NOMINAL_ANNOTATION] = h
    # record the GPUs this provider occupies so the per-GPU sleeper budget
    # can count sleepers by accelerator (reference indexes providers by GPU,
    # controller.go:129-159; annotation key pkg/api/interface.go)
    anns[contracts.ACCELERATORS_ANNOTATION] = ",".join(gpu_uuids)
    return pod, h
