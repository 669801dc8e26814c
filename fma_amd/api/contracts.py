"""Wire contracts of the FMA stack: annotation/label names, SPI paths, ports.

These constants are the cross-process protocol carried in Pod metadata and
HTTP paths. They intentionally match the reference contract so that clients
of llm-d-fast-model-actuation find the same surface here:

- annotation/label names: reference pkg/api/interface.go:47-135
- requester SPI paths:    reference pkg/spi/interface.go:34-89
- shared controller keys: reference pkg/controller/common/interface.go:19-51

Only the accelerator-facing identifiers differ, because this stack is
MI355X-native: GPU visibility is ``HIP_VISIBLE_DEVICES`` (not CUDA_*), the
extended resource is ``amd.com/gpu`` (not nvidia.com/gpu) and UUIDs come from
amd-smi/rocm-smi instead of pynvml/nvidia-smi.
"""

from __future__ import annotations

GROUP = "fma.llm-d.ai"
VERSION = "v1alpha1"

# ---------------------------------------------------------------------------
# Dual-pods annotation / label contract (reference pkg/api/interface.go:47-129)
# ---------------------------------------------------------------------------

#: Go-template -> strategic-merge-patch that derives the server-providing Pod
#: from the server-requesting Pod (mutually exclusive with ISC annotation).
SERVER_PATCH_ANNOTATION = "dual-pods.llm-d.ai/server-patch"

#: Name of the InferenceServerConfig the server-providing Pod uses
#: (mutually exclusive with the server-patch annotation).
INFERENCE_SERVER_CONFIG_ANNOTATION = "dual-pods.llm-d.ai/inference-server-config"

#: JSON ServerRequestingPodStatus maintained by the dual-pods controller.
STATUS_ANNOTATION = "dual-pods.llm-d.ai/status"

#: Port name/number on the requester to query for accelerator UUIDs.
ADMIN_PORT_ANNOTATION = "dual-pods.llm-d.ai/admin-port"
ADMIN_PORT_DEFAULT = "8081"

#: FYI list of accelerator UUIDs bound to the server (comma separated).
ACCELERATORS_ANNOTATION = "dual-pods.llm-d.ai/accelerators"

#: Marks a server-providing Pod as launcher-based.
LAUNCHER_BASED_ANNOTATION = "dual-pods.llm-d.ai/launcher-based"

#: "<requester UID> <requester name>" on a bound server-providing Pod; the
#: Pod object carrying this annotation is the ACID record of the binding.
REQUESTER_ANNOTATION = "dual-pods.llm-d.ai/requester"

#: FYI label present while bound; value is the name of the dual Pod.
DUAL_LABEL = "dual-pods.llm-d.ai/dual"

#: FYI label on a bound requester: the model-server instance ID.
INSTANCE_LABEL = "dual-pods.llm-d.ai/instance"

#: "true"/"false" on server-providing Pods: instance sleeping state.
SLEEPING_LABEL = "dual-pods.llm-d.ai/sleeping"

#: Container name the server patch describes (runs the inference server).
INFERENCE_SERVER_CONTAINER = "inference-server"

# Instance state externalized on launcher Pods so a restarted controller can
# recover bindings (reference pkg/controller/dual-pods/inference-server.go
# recoverInstanceStateFromLauncherPod:1236-1278).
INSTANCE_ID_ANNOTATION = "dual-pods.llm-d.ai/instance-id"
SERVER_PORT_ANNOTATION = "dual-pods.llm-d.ai/server-port"
SERVER_CONFIG_ANNOTATION = "dual-pods.llm-d.ai/vllm-config"
ISC_ROUTING_METADATA_ANNOTATION = "dual-pods.llm-d.ai/isc-routing-metadata"

#: Signature over a launcher's instance list, patched onto the launcher Pod
#: by the notifier sidecar (reference launcher_pod_notifier.py:31).
INSTANCE_SIGNATURE_ANNOTATION = "dual-pods.llm-d.ai/vllm-instance-signature"

# ---------------------------------------------------------------------------
# Shared controller constants (reference pkg/controller/common/interface.go)
# ---------------------------------------------------------------------------

COMPONENT_LABEL = "app.kubernetes.io/component"
LAUNCHER_COMPONENT = "launcher"
LAUNCHER_CONFIG_NAME_LABEL = "dual-pods.llm-d.ai/launcher-config-name"
NODE_NAME_LABEL = "dual-pods.llm-d.ai/node-name"
LAUNCHER_CONFIG_HASH_ANNOTATION = "dual-pods.llm-d.ai/launcher-config-hash"
#: nominal-provider hash of a direct (launcher-less) provider Pod, used for
#: sleeper lookup (reference pkg/controller/dual-pods/controller.go:102);
#: distinct from LAUNCHER_CONFIG_HASH_ANNOTATION, which is reserved for
#: launcher-based providers (reference inference-server.go:687)
NOMINAL_ANNOTATION = "dual-pods.llm-d.ai/nominal"
LAUNCHER_TEMPLATE_HASH_ANNOTATION = (
    "dual-pods.llm-d.ai/launcher-populator-template-hash"
)
LAUNCHER_STUCK_LABEL = "dual-pods.llm-d.ai/launcher-stuck"
LAUNCHER_SERVICE_PORT = 8001

# ---------------------------------------------------------------------------
# Requester SPI (reference pkg/spi/interface.go:34-89)
# ---------------------------------------------------------------------------

ACCELERATOR_QUERY_PATH = "/v1/dual-pods/accelerators"
ACCELERATOR_MEMORY_QUERY_PATH = "/v1/dual-pods/accelerator-memory-usage"
BECOME_READY_PATH = "/v1/become-ready"
BECOME_UNREADY_PATH = "/v1/become-unready"
READY_PATH = "/ready"
SET_LOG_PATH = "/v1/set-log"
LOG_START_POS_PARAM = "startPos"
PROXY_CONFIG_PATH = "/v1/proxy/config"

PROBES_PORT_DEFAULT = 8080
SPI_PORT_DEFAULT = 8081
#: fixed TCP reverse-proxy listen port (reference cmd/requester/main.go
#: --proxy-port default 8082): the controller does not consume the
#: listen_port returned by PUT /v1/proxy/config, so in-cluster traffic
#: must find the proxy at a known port
PROXY_PORT_DEFAULT = 8082

# ---------------------------------------------------------------------------
# Inference-server sleep contract (reference pkg/api/interface.go:131-135 and
# pkg/controller/dual-pods/inference-server.go:1497,1712,1985)
# ---------------------------------------------------------------------------

IS_SLEEPING_PATH = "/is_sleeping"
SLEEP_PATH = "/sleep"
WAKE_UP_PATH = "/wake_up"
HEALTH_PATH = "/health"

# ---------------------------------------------------------------------------
# MI355X-native accelerator identifiers (replacing the reference's NVIDIA ones,
# reference inference-server.go:1917-1934, utils/pod-helper.go:326-352)
# ---------------------------------------------------------------------------

GPU_RESOURCE_NAME = "amd.com/gpu"
VISIBLE_DEVICES_ENV = "HIP_VISIBLE_DEVICES"
ALL_DEVICES_ENV_VALUE = "all"

#: ConfigMap name mapping node -> JSON {uuid: index} for the direct path
#: (reference pkg/controller/dual-pods/controller.go:124).
GPU_MAP_CONFIGMAP = "gpu-map"

# Launcher REST API root (reference docs/launcher.md:191-532). "v2/vllm" kept
# verbatim so reference clients work unchanged against this launcher.
LAUNCHER_API_ROOT = "/v2/vllm/instances"

#: Instance status value reported once the server process has terminated
#: (reference pkg/controller/dual-pods/launcherclient.go:56).
INSTANCE_STATUS_STOPPED = "stopped"
INSTANCE_STATUS_RUNNING = "running"

#: Keys inside VllmConfig.annotations (reference launcherclient.go:50-51).
CONFIG_ISC_NAME_KEY = "isc-name"
CONFIG_INFERENCE_PORT_KEY = "inference-port"
