"""InferenceService API types (group fusioninfer.io/v1alpha1).

Mirrors the reference CRD schema: component types and routing strategies
(reference api/core/v1alpha1/inferenceservice_types.go:24-45), Role
(:58-105), Multinode (:114-119), ComponentStatus/phases (:121-165), and
status with conditions + components map (:167-183). Objects render to the
same YAML/JSON shapes; pod templates / HTTPRoute specs are carried as raw
dicts (the reference's RawExtension escape hatches, :74-104).
"""

from __future__ import annotations

import copy
import dataclasses
from typing import Any, Dict, List, Optional

GROUP = "fusioninfer.io"
VERSION = "v1alpha1"
API_VERSION = f"{GROUP}/{VERSION}"

# component types (reference :24-33)
ROUTER = "router"
PREFILLER = "prefiller"
DECODER = "decoder"
WORKER = "worker"
COMPONENT_TYPES = (ROUTER, PREFILLER, DECODER, WORKER)
WORKER_TYPES = (PREFILLER, DECODER, WORKER)

# routing strategies (reference :35-45)
PREFIX_CACHE = "prefix-cache"
KV_CACHE_UTILIZATION = "kv-cache-utilization"
QUEUE_SIZE = "queue-size"
LORA_AFFINITY = "lora-affinity"
PD_DISAGGREGATION = "pd-disaggregation"
ROUTING_STRATEGIES = (
    PREFIX_CACHE,
    KV_CACHE_UTILIZATION,
    QUEUE_SIZE,
    LORA_AFFINITY,
    PD_DISAGGREGATION,
)

# component phases (reference :131-145)
PHASE_PENDING = "Pending"
PHASE_DEPLOYING = "Deploying"
PHASE_RUNNING = "Running"
PHASE_FAILED = "Failed"


@dataclasses.dataclass
class Multinode:
    """nodeCount > 1 makes a role multi-node (reference :114-119)."""

    node_count: int = 1

    def to_dict(self) -> Dict[str, Any]:
        return {"nodeCount": self.node_count}

    @staticmethod
    def from_dict(d: Optional[Dict[str, Any]]) -> Optional["Multinode"]:
        if not d:
            return None
        return Multinode(node_count=int(d.get("nodeCount", 1)))


@dataclasses.dataclass
class Role:
    """One role of an InferenceService (reference :58-105)."""

    component_type: str = WORKER
    replicas: int = 1
    template: Optional[Dict[str, Any]] = None        # PodTemplateSpec (raw)
    multinode: Optional[Multinode] = None
    # router-only fields
    routing_strategy: Optional[str] = None
    endpoint_picker_config: Optional[str] = None      # raw YAML passthrough
    httproute: Optional[Dict[str, Any]] = None        # HTTPRouteSpec (raw)
    gateway: Optional[Dict[str, Any]] = None          # GatewaySpec (raw)

    def node_count(self) -> int:
        return self.multinode.node_count if self.multinode else 1

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {
            "componentType": self.component_type,
            "replicas": self.replicas,
        }
        if self.template is not None:
            d["template"] = copy.deepcopy(self.template)
        if self.multinode is not None:
            d["multinode"] = self.multinode.to_dict()
        if self.routing_strategy is not None:
            d["routingStrategy"] = self.routing_strategy
        if self.endpoint_picker_config is not None:
            d["endpointPickerConfig"] = self.endpoint_picker_config
        if self.httproute is not None:
            d["httpRoute"] = copy.deepcopy(self.httproute)
        if self.gateway is not None:
            d["gateway"] = copy.deepcopy(self.gateway)
        return d

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "Role":
        return Role(
            component_type=d.get("componentType", WORKER),
            replicas=int(d.get("replicas", 1)),
            template=d.get("template"),
            multinode=Multinode.from_dict(d.get("multinode")),
            routing_strategy=d.get("routingStrategy"),
            endpoint_picker_config=d.get("endpointPickerConfig"),
            httproute=d.get("httpRoute"),
            gateway=d.get("gateway"),
        )


@dataclasses.dataclass
class InferenceService:
    name: str
    namespace: str = "default"
    roles: List[Role] = dataclasses.field(default_factory=list)
    labels: Dict[str, str] = dataclasses.field(default_factory=dict)
    annotations: Dict[str, str] = dataclasses.field(default_factory=dict)
    uid: str = ""
    generation: int = 1
    #: resourceVersion at read time (optimistic concurrency through the
    #: typed client; empty on fresh objects — the store assigns it)
    resource_version: str = ""
    status: Dict[str, Any] = dataclasses.field(default_factory=dict)

    def worker_roles(self) -> List[Role]:
        return [r for r in self.roles if r.component_type in WORKER_TYPES]

    def router_roles(self) -> List[Role]:
        return [r for r in self.roles if r.component_type == ROUTER]

    def role(self, component_type: str) -> Optional[Role]:
        for r in self.roles:
            if r.component_type == component_type:
                return r
        return None

    def to_dict(self) -> Dict[str, Any]:
        return {
            "apiVersion": API_VERSION,
            "kind": "InferenceService",
            "metadata": {
                "name": self.name,
                "namespace": self.namespace,
                "labels": dict(self.labels),
                "annotations": dict(self.annotations),
                "uid": self.uid,
                "generation": self.generation,
                **({"resourceVersion": self.resource_version}
                   if self.resource_version else {}),
            },
            "spec": {"roles": [r.to_dict() for r in self.roles]},
            "status": copy.deepcopy(self.status),
        }

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "InferenceService":
        md = d.get("metadata", {})
        return InferenceService(
            name=md["name"],
            namespace=md.get("namespace", "default"),
            roles=[Role.from_dict(r) for r in d.get("spec", {}).get("roles", [])],
            labels=dict(md.get("labels", {})),
            annotations=dict(md.get("annotations", {})),
            uid=md.get("uid", ""),
            generation=int(md.get("generation", 1)),
            resource_version=md.get("resourceVersion", ""),
            status=copy.deepcopy(d.get("status", {})),
        )

    def spec_dict(self) -> Dict[str, Any]:
        return {"roles": [r.to_dict() for r in self.roles]}


@dataclasses.dataclass
class ModelLoader:
    """Scaffolded CRD mirroring the reference's stub ModelLoader
    (reference api/core/v1alpha1/modelloader_types.go:26-59 — a `Foo`
    placeholder spec shipped unimplemented; its controller is a no-op,
    modelloader_controller.go:49-55). Real weight materialization in this
    build lives engine-side (fusioninfer_amd/models/weight_loader.py)."""

    name: str
    namespace: str = "default"
    foo: Optional[str] = None

    def to_dict(self) -> Dict[str, Any]:
        return {
            "apiVersion": API_VERSION,
            "kind": "ModelLoader",
            "metadata": {"name": self.name, "namespace": self.namespace},
            "spec": {"foo": self.foo} if self.foo is not None else {},
        }

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "ModelLoader":
        return ModelLoader(
            name=d["metadata"]["name"],
            namespace=d["metadata"].get("namespace", "default"),
            foo=d.get("spec", {}).get("foo"),
        )
