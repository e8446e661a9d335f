"""LeaderWorkerSet rendering: one LWS per worker-role replica.

Semantics mirror the reference builder (pkg/workload/lws.go:73-270):
per-replica mode forces replicas=1 with size=nodeCount, LeaderCreated
startup policy + RollingUpdate, the five fusioninfer.io/* labels, Volcano
annotations, volcano schedulerName when gang-scheduled, and spec-hash
label computed post-build.

MI355X-native difference: multi-node command wrapping
emits torchrun/TCPStore rendezvous consuming the SAME LWS env contract
(LWS_LEADER_ADDRESS / LWS_WORKER_INDEX) instead of Ray cluster bootstrap
(reference lws.go:189-242) — the first-party engine forms its process
group over RCCL directly. Leader readiness = TCP probe on the rendezvous
port 29500 (reference probes Ray's 6379).
"""

from __future__ import annotations

import copy
from typing import Any, Dict, List, Optional

from fusioninfer_amd.controlplane.api import InferenceService, Role
from fusioninfer_amd.controlplane.hashutil import compute_spec_hash

# labels (reference lws.go:40-56)
LABEL_SERVICE = "fusioninfer.io/service"
LABEL_ROLE = "fusioninfer.io/role"
LABEL_REPLICA_INDEX = "fusioninfer.io/replica-index"
LABEL_COMPONENT_TYPE = "fusioninfer.io/component-type"
LABEL_MANAGED_BY = "fusioninfer.io/managed-by"
MANAGED_BY = "fusioninfer-amd-controller"
LABEL_SPEC_HASH = "fusioninfer.io/spec-hash"
LABEL_LWS_WORKER_INDEX = "leaderworkerset.sigs.k8s.io/worker-index"

# Volcano annotations (reference lws.go:50-53)
ANNOTATION_PODGROUP = "scheduling.k8s.io/group-name"
ANNOTATION_TASK_SPEC = "volcano.sh/task-spec"

RENDEZVOUS_PORT = 29500
GPU_RESOURCE = "amd.com/gpu"


def generate_lws_name(service_name: str, role: Role, index: int) -> str:
    """{svc}-{componentType}-{replicaIndex} (reference lws.go:260-265)."""
    return f"{service_name}-{role.component_type}-{index}"


def is_multi_node(role: Role) -> bool:
    return role.node_count() >= 2


def _component_labels(svc: InferenceService, role: Role, index: int) -> Dict[str, str]:
    return {
        LABEL_SERVICE: svc.name,
        LABEL_ROLE: role.component_type,
        LABEL_REPLICA_INDEX: str(index),
        LABEL_COMPONENT_TYPE: role.component_type,
        LABEL_MANAGED_BY: MANAGED_BY,
    }


def _gpus_per_pod(pod_spec: Dict[str, Any]) -> int:
    total = 0
    for c in pod_spec.get("containers", []):
        limits = c.get("resources", {}).get("limits", {})
        total += int(limits.get(GPU_RESOURCE, 0))
    return total


def wrap_leader_container(container: Dict[str, Any], node_count: int,
                          gpus_per_pod: int) -> None:
    """Leader: engine rank-0 node with torchrun-style rendezvous flags.

    (Replaces the reference's `ray start --head && vllm serve ...
    --distributed-executor-backend ray` wrap, lws.go:189-231.)
    """
    orig = " ".join(container.get("command", []) + container.get("args", []))
    container["command"] = ["sh", "-c"]
    container["args"] = [
        f"{orig} --nnodes {node_count} --node-rank 0 "
        f"--nproc-per-node {max(gpus_per_pod, 1)} "
        f"--master-addr $LWS_LEADER_ADDRESS --master-port {RENDEZVOUS_PORT}"
    ]
    container.setdefault("readinessProbe", {
        "tcpSocket": {"port": RENDEZVOUS_PORT},
        "initialDelaySeconds": 15,
        "periodSeconds": 10,
        "failureThreshold": 60,
    })
    ports = container.setdefault("ports", [])
    if not any(p.get("containerPort") == RENDEZVOUS_PORT for p in ports):
        ports.append({"containerPort": RENDEZVOUS_PORT, "name": "rendezvous"})


def wrap_worker_container(container: Dict[str, Any], node_count: int,
                          gpus_per_pod: int) -> None:
    """Worker node: same engine command joining the leader's TCPStore.

    (Replaces `ray start --address=$LWS_LEADER_ADDRESS:6379 --block`,
    reference lws.go:235-242.)
    """
    orig = " ".join(container.get("command", []) + container.get("args", []))
    container["command"] = ["sh", "-c"]
    container["args"] = [
        f"{orig} --nnodes {node_count} --node-rank $LWS_WORKER_INDEX "
        f"--nproc-per-node {max(gpus_per_pod, 1)} "
        f"--master-addr $LWS_LEADER_ADDRESS --master-port {RENDEZVOUS_PORT}"
    ]


def _build_pod_spec(role: Role, gang: bool, podgroup_name: Optional[str]) -> Dict[str, Any]:
    """Decode the user pod template and apply scheduling policy
    (reference lws.go:168-185)."""
    template = copy.deepcopy(role.template) or {"spec": {"containers": []}}
    pod_spec = template.get("spec", {})
    if gang:
        pod_spec["schedulerName"] = "volcano"
    return template


def build_lws(
    svc: InferenceService,
    role: Role,
    replica_index: int,
    gang_scheduled: bool = False,
    podgroup_name: Optional[str] = None,
    task_name: Optional[str] = None,
) -> Dict[str, Any]:
    """Render one LeaderWorkerSet for replica `replica_index` of `role`."""
    name = generate_lws_name(svc.name, role, replica_index)
    labels = _component_labels(svc, role, replica_index)
    node_count = role.node_count()

    leader_template = _build_pod_spec(role, gang_scheduled, podgroup_name)
    annotations: Dict[str, str] = {}
    if gang_scheduled and podgroup_name:
        annotations[ANNOTATION_PODGROUP] = podgroup_name
        annotations[ANNOTATION_TASK_SPEC] = task_name or f"{role.component_type}-{replica_index}"
    meta = leader_template.setdefault("metadata", {})
    meta.setdefault("labels", {}).update(labels)
    if annotations:
        meta.setdefault("annotations", {}).update(annotations)

    worker_template = None
    if is_multi_node(role):
        worker_template = copy.deepcopy(leader_template)
        gpus = _gpus_per_pod(leader_template.get("spec", {}))
        lc = leader_template["spec"]["containers"]
        wc = worker_template["spec"]["containers"]
        if lc:
            wrap_leader_container(lc[0], node_count, gpus)
        if wc:
            wrap_worker_container(wc[0], node_count, gpus)

    lws_spec: Dict[str, Any] = {
        # per-replica mode: each LWS manages exactly one replica group
        # (reference lws.go:87-93)
        "replicas": 1,
        "startupPolicy": "LeaderCreated",
        "rolloutStrategy": {
            "type": "RollingUpdate",
            "rollingUpdateConfiguration": {"maxSurge": 0, "maxUnavailable": 1},
        },
        "leaderWorkerTemplate": {
            "size": node_count,
            "leaderTemplate": leader_template,
            **(
                {"workerTemplate": worker_template}
                if worker_template is not None
                else {}
            ),
        },
    }
    lws = {
        "apiVersion": "leaderworkerset.x-k8s.io/v1",
        "kind": "LeaderWorkerSet",
        "metadata": {
            "name": name,
            "namespace": svc.namespace,
            "labels": dict(labels),
        },
        "spec": lws_spec,
    }
    # spec-hash computed post-build over the rendered spec (reference :160-163)
    lws["metadata"]["labels"][LABEL_SPEC_HASH] = compute_spec_hash(lws["spec"])
    return lws


def list_expected_lws_names(svc: InferenceService, role: Role) -> List[str]:
    return [generate_lws_name(svc.name, role, i) for i in range(role.replicas)]
