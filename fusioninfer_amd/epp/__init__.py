"""First-party Endpoint Picker (EPP) runtime.

The reference runs the upstream Gateway-API-Inference-Extension EPP image
(registry.k8s.io/.../epp:v1.2.1, reference pkg/router/epp.go:46) as an
Envoy ext-proc that scores worker endpoints per request. This package is
the first-party equivalent: it consumes the SAME EndpointPickerConfig
YAML the control plane renders (controlplane/router.py ==
reference strategy.go:51-165) and implements the same plugin graph —
prefix-cache LRU scorer, kv-cache-utilization scorer, queue scorer,
lora-affinity scorer, by-label filters, max-score-picker, and the PD
profile handler (prefill subrequest + decode routing with header).

Scorers read per-endpoint KV occupancy / queue depth from the engine's
/metrics surface (vllm:gpu_cache_usage_perc, vllm:num_requests_waiting —
server/api_server.py exports them).
"""

from fusioninfer_amd.epp.picker import (  # noqa: F401
    Endpoint,
    EndpointPicker,
    PickResult,
)
