"""Validating admission policies, enforced in-process.

Python equivalents of the reference's CEL ValidatingAdmissionPolicies
(reference config/validating-admission-policies/fma-immutable-fields.yaml:
1-33 and fma-bound-serverreqpod.yaml:1-30; rationale docs/dual-pods.md:
189-214): FMA-managed Pod metadata is frozen against everyone except the
FMA controllers, and a bound requester cannot switch its ISC.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

from fma_amd.api import contracts
from fma_amd.store.memstore import Invalid, MemStore

#: actors allowed to mutate FMA-managed metadata (the reference exempts
#: the FMA service accounts)
FMA_ACTORS = {"dual-pods-controller", "launcher-populator", "system",
              "node-agent"}

PROTECTED_ANNOTATIONS = (
    contracts.REQUESTER_ANNOTATION,
    contracts.STATUS_ANNOTATION,
    contracts.INSTANCE_ID_ANNOTATION,
    contracts.SERVER_PORT_ANNOTATION,
    contracts.SERVER_CONFIG_ANNOTATION,
    contracts.ISC_ROUTING_METADATA_ANNOTATION,
)
PROTECTED_LABELS = (
    contracts.DUAL_LABEL,
    contracts.LAUNCHER_CONFIG_NAME_LABEL,
)


def _meta_maps(obj: Optional[Dict[str, Any]]):
    meta = (obj or {}).get("metadata", {})
    return meta.get("annotations", {}) or {}, meta.get("labels", {}) or {}


def immutable_fields_policy(op: str, old: Optional[Dict[str, Any]],
                            new: Optional[Dict[str, Any]],
                            actor: str) -> None:
    """Deny non-FMA mutation of FMA-managed annotations/labels."""
    if actor in FMA_ACTORS:
        return
    if op not in ("UPDATE",):
        return
    if (old or {}).get("kind") != "Pod":
        return
    old_ann, old_lbl = _meta_maps(old)
    new_ann, new_lbl = _meta_maps(new)
    for key in PROTECTED_ANNOTATIONS:
        if old_ann.get(key) != new_ann.get(key):
            raise Invalid(
                f"annotation {key!r} is managed by FMA controllers and may "
                f"not be changed by {actor!r}")
    for key in PROTECTED_LABELS:
        if old_lbl.get(key) != new_lbl.get(key):
            raise Invalid(
                f"label {key!r} is managed by FMA controllers and may not "
                f"be changed by {actor!r}")


def bound_requester_policy(op: str, old: Optional[Dict[str, Any]],
                           new: Optional[Dict[str, Any]],
                           actor: str) -> None:
    """Deny changing a BOUND requester's ISC annotation (reference
    fma-bound-serverreqpod.yaml: the binding's identity derives from it)."""
    if op != "UPDATE" or (old or {}).get("kind") != "Pod":
        return
    old_ann, old_lbl = _meta_maps(old)
    new_ann, _ = _meta_maps(new)
    key = contracts.INFERENCE_SERVER_CONFIG_ANNOTATION
    if key not in old_ann:
        return
    if contracts.DUAL_LABEL not in old_lbl:
        return  # not bound: free to change
    if old_ann.get(key) != new_ann.get(key):
        raise Invalid(
            "cannot change the inference-server-config annotation of a "
            "bound server-requesting Pod")


#: kind -> pydantic spec model: structural CRD validation, the analog of
#: the apiserver's OpenAPI schema + listMapKey checks (the reference's
#: e2e expects e.g. an LPP with duplicate countForLauncher keys to be
#: REJECTED at create — test-cases.sh:266-296)
def crd_schema_policy(op: str, old, new, actor: str) -> None:
    if op not in ("CREATE", "UPDATE") or new is None:
        return
    kind = new.get("kind")
    from fma_amd.api import types as t
    models = {
        "InferenceServerConfig": t.InferenceServerConfigSpec,
        "LauncherConfig": t.LauncherConfigSpec,
        "LauncherPopulationPolicy": t.LauncherPopulationPolicySpec,
    }
    model = models.get(kind)
    if model is None:
        return
    try:
        model.model_validate(new.get("spec") or {})
    except Exception as e:  # pydantic.ValidationError
        raise Invalid(f"{kind} spec invalid: {e}") from e


def install_policies(store: MemStore) -> None:
    store.add_admission_hook(crd_schema_policy)
    store.add_admission_hook(immutable_fields_policy)
    store.add_admission_hook(bound_requester_policy)
