"""Router stack rendering: EPP (Endpoint Picker) deployment + service +
RBAC + config, InferencePool, HTTPRoute.

Mirrors the reference builders (pkg/router/epp.go:34-361, strategy.go:27-165,
inferencepool.go:31-129, httproute.go:37-92): the EndpointPickerConfig YAML
is the Gateway-API-Inference-Extension plugin schema the EPP container
consumes — the five routing strategies map to the same plugin graphs, with
the user's `endpointPickerConfig` passthrough winning.
"""

from __future__ import annotations

import copy
import os
from typing import Any, Dict

import yaml

from fusioninfer_amd.controlplane.api import (
    PD_DISAGGREGATION,
    KV_CACHE_UTILIZATION,
    LORA_AFFINITY,
    QUEUE_SIZE,
    DECODER,
    PREFILLER,
    InferenceService,
    Role,
)
from fusioninfer_amd.controlplane.hashutil import compute_spec_hash
from fusioninfer_amd.controlplane.scheduling import is_pd_disaggregated
from fusioninfer_amd.controlplane.workload import (
    LABEL_COMPONENT_TYPE,
    LABEL_LWS_WORKER_INDEX,
    LABEL_SERVICE,
    LABEL_SPEC_HASH,
)

# ports (reference epp.go:34-47)
EPP_GRPC_PORT = 9002
EPP_HEALTH_PORT = 9003
EPP_METRICS_PORT = 9090
ENGINE_PORT = 8000

DEFAULT_EPP_IMAGE = "registry.k8s.io/gateway-api-inference-extension/epp:v1.2.1"


def get_epp_image() -> str:
    """EPP_IMAGE env override (reference epp.go:43-55)."""
    return os.environ.get("EPP_IMAGE", DEFAULT_EPP_IMAGE)


# ------------------------------------------------------------------ names
def pool_name(svc: InferenceService) -> str:
    return f"{svc.name}-pool"


def epp_name(svc: InferenceService) -> str:
    return f"{svc.name}-epp"


def epp_config_name(svc: InferenceService) -> str:
    return f"{svc.name}-epp-config"


def httproute_name(svc: InferenceService) -> str:
    return f"{svc.name}-route"


# ------------------------------------------------------------------ strategy
def _single_scorer_config(scorer: str) -> Dict[str, Any]:
    return {
        "apiVersion": "inference.networking.x-k8s.io/v1alpha1",
        "kind": "EndpointPickerConfig",
        "plugins": [{"type": scorer}, {"type": "max-score-picker"}],
        "schedulingProfiles": [
            {
                "name": "default",
                "plugins": [
                    {"pluginRef": "max-score-picker"},
                    {"pluginRef": scorer, "weight": 100},
                ],
            }
        ],
    }


def _prefix_cache_config() -> Dict[str, Any]:
    cfg = _single_scorer_config("prefix-cache-scorer")
    cfg["plugins"][0] = {
        "type": "prefix-cache-scorer",
        "parameters": {
            "blockSize": 5,
            "maxPrefixBlocksToMatch": 256,
            "lruCapacityPerServer": 31250,
        },
    }
    return cfg


def _pd_config() -> Dict[str, Any]:
    return {
        "apiVersion": "inference.networking.x-k8s.io/v1alpha1",
        "kind": "EndpointPickerConfig",
        "plugins": [
            {
                "type": "pd-profile-handler",
                "parameters": {
                    "threshold": 0,
                    "hashBlockSize": 5,
                    "primaryPort": ENGINE_PORT,
                },
            },
            {"type": "prefill-header-handler"},
            {
                "type": "by-label",
                "name": "prefill-pods",
                "parameters": {
                    "label": LABEL_COMPONENT_TYPE,
                    "validValues": [PREFILLER],
                },
            },
            {
                "type": "by-label",
                "name": "decode-pods",
                "parameters": {
                    "label": LABEL_COMPONENT_TYPE,
                    "validValues": [DECODER],
                },
            },
            {
                "type": "prefix-cache-scorer",
                "parameters": {
                    "hashBlockSize": 5,
                    "maxPrefixBlocksToMatch": 256,
                    "lruCapacityPerServer": 31250,
                },
            },
            {"type": "max-score-picker"},
        ],
        "schedulingProfiles": [
            {
                "name": "prefill",
                "plugins": [
                    {"pluginRef": "prefill-pods"},
                    {"pluginRef": "max-score-picker"},
                    {"pluginRef": "prefix-cache-scorer", "weight": 50},
                ],
            },
            {
                "name": "decode",
                "plugins": [
                    {"pluginRef": "decode-pods"},
                    {"pluginRef": "max-score-picker"},
                    {"pluginRef": "prefix-cache-scorer", "weight": 50},
                ],
            },
        ],
    }


_STRATEGY_SCORERS = {
    KV_CACHE_UTILIZATION: "kv-cache-utilization-scorer",
    QUEUE_SIZE: "queue-scorer",
    LORA_AFFINITY: "lora-affinity-scorer",
}


def generate_epp_config(svc: InferenceService, role: Role) -> str:
    """Strategy -> EndpointPickerConfig YAML. User passthrough wins
    (reference strategy.go:28-31); PD falls back to prefix-cache when the
    service is not actually PD (:120-124); default = prefix-cache."""
    if role.endpoint_picker_config:
        return role.endpoint_picker_config
    strategy = role.routing_strategy
    if strategy == PD_DISAGGREGATION:
        cfg = _pd_config() if is_pd_disaggregated(svc) else _prefix_cache_config()
    elif strategy in _STRATEGY_SCORERS:
        cfg = _single_scorer_config(_STRATEGY_SCORERS[strategy])
    else:
        cfg = _prefix_cache_config()
    return yaml.safe_dump(cfg, sort_keys=False)


# ------------------------------------------------------------------ EPP
def build_epp_configmap(svc: InferenceService, role: Role) -> Dict[str, Any]:
    cm = {
        "apiVersion": "v1",
        "kind": "ConfigMap",
        "metadata": {
            "name": epp_config_name(svc),
            "namespace": svc.namespace,
            "labels": {LABEL_SERVICE: svc.name},
        },
        "data": {"config.yaml": generate_epp_config(svc, role)},
    }
    cm["metadata"]["labels"][LABEL_SPEC_HASH] = compute_spec_hash(cm["data"])
    return cm


def build_epp_deployment(svc: InferenceService) -> Dict[str, Any]:
    name = epp_name(svc)
    labels = {"app": name, LABEL_SERVICE: svc.name}
    spec = {
        "replicas": 1,
        "strategy": {"type": "Recreate"},
        "selector": {"matchLabels": {"app": name}},
        "template": {
            "metadata": {"labels": dict(labels)},
            "spec": {
                "serviceAccountName": name,
                "containers": [
                    {
                        "name": "epp",
                        "image": get_epp_image(),
                        "args": [
                            f"--pool-name={pool_name(svc)}",
                            f"--pool-namespace={svc.namespace}",
                            "--config-file=/config/config.yaml",
                            "--v=4",
                        ],
                        "ports": [
                            {"containerPort": EPP_GRPC_PORT, "name": "grpc"},
                            {"containerPort": EPP_HEALTH_PORT, "name": "grpc-health"},
                            {"containerPort": EPP_METRICS_PORT, "name": "metrics"},
                        ],
                        "livenessProbe": {
                            "grpc": {"port": EPP_HEALTH_PORT, "service": "inference-extension"},
                            "initialDelaySeconds": 5,
                            "periodSeconds": 10,
                        },
                        "readinessProbe": {
                            "grpc": {"port": EPP_HEALTH_PORT, "service": "inference-extension"},
                            "initialDelaySeconds": 5,
                            "periodSeconds": 10,
                        },
                        "volumeMounts": [
                            {"name": "config", "mountPath": "/config", "readOnly": True}
                        ],
                    }
                ],
                "volumes": [
                    {
                        "name": "config",
                        "configMap": {"name": epp_config_name(svc)},
                    }
                ],
            },
        },
    }
    dep = {
        "apiVersion": "apps/v1",
        "kind": "Deployment",
        "metadata": {
            "name": name,
            "namespace": svc.namespace,
            "labels": dict(labels),
        },
        "spec": spec,
    }
    dep["metadata"]["labels"][LABEL_SPEC_HASH] = compute_spec_hash(spec)
    return dep


def build_epp_service(svc: InferenceService) -> Dict[str, Any]:
    name = epp_name(svc)
    spec = {
        "selector": {"app": name},
        "ports": [
            {"name": "grpc", "port": EPP_GRPC_PORT, "targetPort": EPP_GRPC_PORT},
            {"name": "grpc-health", "port": EPP_HEALTH_PORT, "targetPort": EPP_HEALTH_PORT},
            {"name": "metrics", "port": EPP_METRICS_PORT, "targetPort": EPP_METRICS_PORT},
        ],
        "type": "ClusterIP",
    }
    s = {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": {
            "name": name,
            "namespace": svc.namespace,
            "labels": {LABEL_SERVICE: svc.name},
        },
        "spec": spec,
    }
    s["metadata"]["labels"][LABEL_SPEC_HASH] = compute_spec_hash(spec)
    return s


def build_epp_service_account(svc: InferenceService) -> Dict[str, Any]:
    return {
        "apiVersion": "v1",
        "kind": "ServiceAccount",
        "metadata": {
            "name": epp_name(svc),
            "namespace": svc.namespace,
            "labels": {LABEL_SERVICE: svc.name},
        },
    }


def build_epp_role(svc: InferenceService) -> Dict[str, Any]:
    """RBAC rules the EPP needs (reference epp.go:279-327)."""
    return {
        "apiVersion": "rbac.authorization.k8s.io/v1",
        "kind": "Role",
        "metadata": {
            "name": epp_name(svc),
            "namespace": svc.namespace,
            "labels": {LABEL_SERVICE: svc.name},
        },
        "rules": [
            {"apiGroups": [""], "resources": ["pods"], "verbs": ["get", "list", "watch"]},
            {
                "apiGroups": ["inference.networking.k8s.io"],
                "resources": ["inferencepools"],
                "verbs": ["get", "list", "watch"],
            },
            {
                "apiGroups": ["inference.networking.x-k8s.io"],
                "resources": ["inferenceobjectives", "inferencemodels"],
                "verbs": ["get", "list", "watch"],
            },
            {
                "apiGroups": ["coordination.k8s.io"],
                "resources": ["leases"],
                "verbs": ["get", "list", "watch", "create", "update", "patch", "delete"],
            },
            {"apiGroups": [""], "resources": ["events"], "verbs": ["create", "patch"]},
        ],
    }


def build_epp_role_binding(svc: InferenceService) -> Dict[str, Any]:
    return {
        "apiVersion": "rbac.authorization.k8s.io/v1",
        "kind": "RoleBinding",
        "metadata": {
            "name": epp_name(svc),
            "namespace": svc.namespace,
            "labels": {LABEL_SERVICE: svc.name},
        },
        "roleRef": {
            "apiGroup": "rbac.authorization.k8s.io",
            "kind": "Role",
            "name": epp_name(svc),
        },
        "subjects": [
            {
                "kind": "ServiceAccount",
                "name": epp_name(svc),
                "namespace": svc.namespace,
            }
        ],
    }


# ------------------------------------------------------------------ pool
def build_pool_selector(svc: InferenceService) -> Dict[str, str]:
    """Select routable worker pods: the service label, the component type
    when exactly one worker role exists, and ONLY leader pods
    (worker-index 0) — reference inferencepool.go:81-99."""
    selector = {LABEL_SERVICE: svc.name}
    worker_roles = svc.worker_roles()
    if len(worker_roles) == 1:
        selector[LABEL_COMPONENT_TYPE] = worker_roles[0].component_type
    selector[LABEL_LWS_WORKER_INDEX] = "0"
    return selector


def build_inference_pool(svc: InferenceService) -> Dict[str, Any]:
    spec = {
        "targetPorts": [{"number": ENGINE_PORT}],
        "selector": {"matchLabels": build_pool_selector(svc)},
        "endpointPickerRef": {
            "name": epp_name(svc),
            "port": {"number": EPP_GRPC_PORT},
        },
    }
    pool = {
        "apiVersion": "inference.networking.k8s.io/v1",
        "kind": "InferencePool",
        "metadata": {
            "name": pool_name(svc),
            "namespace": svc.namespace,
            "labels": {LABEL_SERVICE: svc.name},
        },
        "spec": spec,
    }
    pool["metadata"]["labels"][LABEL_SPEC_HASH] = compute_spec_hash(spec)
    return pool


# ------------------------------------------------------------------ route
def build_httproute(svc: InferenceService, role: Role) -> Dict[str, Any]:
    """User HTTPRouteSpec preserved (parentRefs/hostnames), rules OVERRIDDEN
    with the single backendRef to the InferencePool (reference
    httproute.go:55-92)."""
    user_spec = copy.deepcopy(role.httproute) or {}
    spec: Dict[str, Any] = {
        k: v for k, v in user_spec.items() if k in ("parentRefs", "hostnames")
    }
    spec["rules"] = [
        {
            "matches": [{"path": {"type": "PathPrefix", "value": "/"}}],
            "backendRefs": [
                {
                    "group": "inference.networking.k8s.io",
                    "kind": "InferencePool",
                    "name": pool_name(svc),
                }
            ],
        }
    ]
    route = {
        "apiVersion": "gateway.networking.k8s.io/v1",
        "kind": "HTTPRoute",
        "metadata": {
            "name": httproute_name(svc),
            "namespace": svc.namespace,
            "labels": {LABEL_SERVICE: svc.name},
        },
        "spec": spec,
    }
    route["metadata"]["labels"][LABEL_SPEC_HASH] = compute_spec_hash(spec)
    return route
