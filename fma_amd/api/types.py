"""CRD types of the FMA API group ``fma.llm-d.ai/v1alpha1``.

Mirrors the reference CRDs field-for-field (pydantic instead of Go structs):

- InferenceServerConfig: reference api/fma/v1alpha1/inferenceserverconfig_types.go:24-97
- LauncherConfig:        reference api/fma/v1alpha1/launcherconfig_types.go:24-91
- LauncherPopulationPolicy: reference api/fma/v1alpha1/launcherpopulationpolicy_types.go:25-126

The MI355X stack keeps these JSON-identical so a user's YAML manifests work
unchanged; only the cluster-resource identifiers differ (``amd.com/gpu``).
"""

from __future__ import annotations

import re
from typing import Any, Dict, List, Optional, Union

from pydantic import BaseModel, Field, field_validator

# -- Kubernetes quantity parsing (subset: plain ints, binary/decimal suffixes) --

_QUANTITY_RE = re.compile(r"^([+-]?[0-9]+(?:\.[0-9]+)?)(Ki|Mi|Gi|Ti|Pi|Ei|m|k|M|G|T|P|E)?$")
_SUFFIX = {
    None: 1,
    "m": 1e-3,
    "k": 1e3, "M": 1e6, "G": 1e9, "T": 1e12, "P": 1e15, "E": 1e18,
    "Ki": 2**10, "Mi": 2**20, "Gi": 2**30, "Ti": 2**40, "Pi": 2**50, "Ei": 2**60,
}


def parse_quantity(q: "str | int | float") -> float:
    """Parse a Kubernetes resource.Quantity into a float.

    Same surface semantics as apimachinery resource.Quantity for the formats
    the reference's ResourceRange comparisons need
    (launcherpopulationpolicy_types.go:104-126).
    """
    if isinstance(q, (int, float)):
        return float(q)
    m = _QUANTITY_RE.match(q.strip())
    if not m:
        raise ValueError(f"invalid quantity {q!r}")
    return float(m.group(1)) * _SUFFIX[m.group(2)]


class ModelServerConfig(BaseModel):
    """Parameters of one model-server instance
    (reference inferenceserverconfig_types.go:35-62)."""

    port: int = Field(ge=1, le=65535)
    options: str = ""
    env_vars: Dict[str, str] = Field(default_factory=dict)
    labels: Dict[str, str] = Field(default_factory=dict)
    annotations: Dict[str, str] = Field(default_factory=dict)

    @field_validator("labels", "annotations")
    @classmethod
    def _no_reserved_prefixes(cls, v: Dict[str, str]) -> Dict[str, str]:
        for key in v:
            for prefix in ("dual-pods.llm-d.ai/", "kubernetes.io/", "k8s.io/"):
                if key.startswith(prefix) or f"/{prefix}" in key:
                    raise ValueError(f"key {key!r} uses reserved prefix {prefix!r}")
        return v


class ObjectStatus(BaseModel):
    """Shared status shape: observedGeneration + errors
    (reference inferenceserverconfig_types.go:64-74)."""

    observedGeneration: int = 0
    errors: List[str] = Field(default_factory=list)


class InferenceServerConfigSpec(BaseModel):
    modelServerConfig: ModelServerConfig
    launcherConfigName: str


class EmbeddedObjectMeta(BaseModel):
    labels: Dict[str, str] = Field(default_factory=dict)
    annotations: Dict[str, str] = Field(default_factory=dict)


class EmbeddedPodTemplateSpec(BaseModel):
    """Pod template with explicitly declared metadata
    (reference launcherconfig_types.go:38-45). ``spec`` is an open PodSpec
    dict — the store validates only the parts the controllers consume."""

    metadata: EmbeddedObjectMeta = Field(default_factory=EmbeddedObjectMeta)
    spec: Dict[str, Any] = Field(default_factory=dict)


class LauncherConfigSpec(BaseModel):
    podTemplate: EmbeddedPodTemplateSpec = Field(default_factory=EmbeddedPodTemplateSpec)
    maxInstances: int = Field(ge=1)


class ResourceRange(BaseModel):
    """Inclusive min/max quantity bounds
    (reference launcherpopulationpolicy_types.go:104-121). Quantities may
    be strings with k8s suffixes ("512Gi") or plain numbers."""

    min: Optional[Union[str, int, float]] = None
    max: Optional[Union[str, int, float]] = None

    def contains(self, value: "str | int | float") -> bool:
        v = parse_quantity(value)
        if self.min is not None and v < parse_quantity(self.min):
            return False
        if self.max is not None and v > parse_quantity(self.max):
            return False
        return True


class LabelSelector(BaseModel):
    """metav1.LabelSelector subset: matchLabels + matchExpressions."""

    matchLabels: Dict[str, str] = Field(default_factory=dict)
    matchExpressions: List[Dict[str, Any]] = Field(default_factory=list)

    def matches(self, labels: Dict[str, str]) -> bool:
        for k, v in self.matchLabels.items():
            if labels.get(k) != v:
                return False
        for expr in self.matchExpressions:
            key = expr.get("key", "")
            op = expr.get("operator", "In")
            values = expr.get("values", []) or []
            if op == "In":
                if labels.get(key) not in values:
                    return False
            elif op == "NotIn":
                if labels.get(key) in values:
                    return False
            elif op == "Exists":
                if key not in labels:
                    return False
            elif op == "DoesNotExist":
                if key in labels:
                    return False
            else:
                raise ValueError(f"unknown selector operator {op!r}")
        return True


class EnhancedNodeSelector(BaseModel):
    """Label selector + allocatable-resource ranges
    (reference launcherpopulationpolicy_types.go:92-126, node matching
    pkg/controller/launcher-populator/node-matcher.go:26-44)."""

    labelSelector: LabelSelector = Field(default_factory=LabelSelector)
    allocatableResources: Dict[str, ResourceRange] = Field(default_factory=dict)

    def matches_node(self, labels: Dict[str, str], allocatable: Dict[str, Any]) -> bool:
        if not self.labelSelector.matches(labels):
            return False
        for resource_name, rng in self.allocatableResources.items():
            if resource_name not in allocatable:
                return False
            if not rng.contains(allocatable[resource_name]):
                return False
        return True


class CountForLauncher(BaseModel):
    launcherConfigName: str
    launcherCount: int


class LauncherPopulationPolicySpec(BaseModel):
    enhancedNodeSelector: EnhancedNodeSelector
    countForLauncher: List[CountForLauncher]

    @field_validator("countForLauncher")
    @classmethod
    def _unique_lc_names(cls, v: List[CountForLauncher]) -> List[CountForLauncher]:
        # listMapKey=launcherConfigName: one entry per LauncherConfig
        # (reference launcherpopulationpolicy_types.go listType=map)
        names = [c.launcherConfigName for c in v]
        if len(names) != len(set(names)):
            raise ValueError("countForLauncher entries must have unique launcherConfigName")
        return v


KIND_SHORT_NAMES = {
    "InferenceServerConfig": "isc",
    "LauncherConfig": "lcfg",
    "LauncherPopulationPolicy": "lpp",
}

SPEC_TYPES = {
    "InferenceServerConfig": InferenceServerConfigSpec,
    "LauncherConfig": LauncherConfigSpec,
    "LauncherPopulationPolicy": LauncherPopulationPolicySpec,
}
