"""Volcano gang-scheduling policy + PodGroup rendering.

Mirrors the reference (pkg/scheduling/podgroup.go): gang scheduling is
needed iff the service is PD-disaggregated (prefiller AND decoder roles
both present, :33-47) or any role is multi-node (:53-70). ONE shared
PodGroup per service: minTaskMember["{role}-{replicaIdx}"] = nodeCount,
minMember = sum (:101-135), minResources = per-pod container limits
summed across all pods (:159-190). GPU resource: amd.com/gpu.
"""

from __future__ import annotations

from typing import Any, Dict

from fusioninfer_amd.controlplane.api import (
    DECODER,
    PREFILLER,
    InferenceService,
    Role,
)
from fusioninfer_amd.controlplane.hashutil import compute_spec_hash
from fusioninfer_amd.controlplane.workload import LABEL_SPEC_HASH


def is_pd_disaggregated(svc: InferenceService) -> bool:
    """prefiller + decoder both present (reference podgroup.go:33-47)."""
    types = {r.component_type for r in svc.roles}
    return PREFILLER in types and DECODER in types


def needs_gang_scheduling(svc: InferenceService) -> bool:
    if is_pd_disaggregated(svc):
        return True
    return any(r.node_count() >= 2 for r in svc.worker_roles())


def needs_gang_scheduling_for_role(svc: InferenceService, role: Role) -> bool:
    """Per-role variant (reference :73-85): PD services gang every worker
    role; otherwise only multi-node roles."""
    if is_pd_disaggregated(svc):
        return True
    return role.node_count() >= 2


def podgroup_name(svc: InferenceService) -> str:
    return svc.name  # reference :193-195


def task_name(role: Role, replica_index: int) -> str:
    return f"{role.component_type}-{replica_index}"  # reference :200-202


def _add_quantities(total: Dict[str, float], limits: Dict[str, Any], factor: int):
    for k, val in limits.items():
        try:
            q = float(val)
        except (TypeError, ValueError):
            continue  # non-scalar quantities (e.g. "2Gi") summed separately
        total[k] = total.get(k, 0.0) + q * factor


def build_podgroup(svc: InferenceService) -> Dict[str, Any]:
    min_task_member: Dict[str, int] = {}
    min_member = 0
    min_resources: Dict[str, float] = {}
    for role in svc.worker_roles():
        nodes = role.node_count()
        for i in range(role.replicas):
            min_task_member[task_name(role, i)] = nodes
            min_member += nodes
        total_pods = role.replicas * nodes
        for c in (role.template or {}).get("spec", {}).get("containers", []):
            limits = c.get("resources", {}).get("limits", {})
            _add_quantities(min_resources, limits, total_pods)

    pg = {
        "apiVersion": "scheduling.volcano.sh/v1beta1",
        "kind": "PodGroup",
        "metadata": {
            "name": podgroup_name(svc),
            "namespace": svc.namespace,
            "labels": {},
        },
        "spec": {
            "minMember": min_member,
            "minTaskMember": min_task_member,
            "minResources": {
                k: (int(v) if float(v).is_integer() else v)
                for k, v in sorted(min_resources.items())
            },
        },
    }
    pg["metadata"]["labels"][LABEL_SPEC_HASH] = compute_spec_hash(pg["spec"])
    return pg


def get_replica_count(svc: InferenceService) -> int:
    return sum(r.replicas for r in svc.worker_roles())


def get_node_count(svc: InferenceService) -> int:
    return sum(r.replicas * r.node_count() for r in svc.worker_roles())
