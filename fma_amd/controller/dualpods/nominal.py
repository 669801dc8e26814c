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


def strategic_merge(base: Any, patch: Any) -> Any:
    """Strategic merge patch for the Pod-shaped subset: dicts merge
    recursively; lists of objects with a ``name`` key merge by name;
    other lists replace; ``None`` deletes a key."""
    if patch is None:
        return None
    if isinstance(base, dict) and isinstance(patch, dict):
        # pure merge: untouched branches are COPIED, never aliased, so
        # mutating the result can't corrupt the caller's base
        out = {k: copy.deepcopy(v) for k, v in base.items()
               if k not in patch}
        out.update({k: copy.deepcopy(base[k]) for k in patch
                    if k in base})
        for k, v in patch.items():
            if v is None:
                out.pop(k, None)
            elif k in out:
                merged = strategic_merge(out[k], v)
                if merged is None:
                    out.pop(k, None)
                else:
                    out[k] = merged
            else:
                out[k] = copy.deepcopy(v)
        return out
    if isinstance(base, list) and isinstance(patch, list):
        if all(isinstance(e, dict) and "name" in e for e in base + patch):
            by_name = {e["name"]: copy.deepcopy(e) for e in base}
            order = [e["name"] for e in base]
            for e in patch:
                if e["name"] in by_name:
                    by_name[e["name"]] = strategic_merge(by_name[e["name"]], e)
                else:
                    by_name[e["name"]] = copy.deepcopy(e)
                    order.append(e["name"])
            return [by_name[n] for n in order]
        return copy.deepcopy(patch)
    return copy.deepcopy(patch)


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
