"""CRD schema for InferenceService (fusioninfer.io/v1alpha1, amd.com/gpu).

Renders the OpenAPI v3 schema equivalent to the reference CRD
(config/crd/bases/fusioninfer.io_inferenceservices.yaml:1-267) with the
same groups/kinds/fields; RawExtension escape hatches map to
x-kubernetes-preserve-unknown-fields.
"""

from __future__ import annotations

import yaml

from fusioninfer_amd.controlplane.api import (
    COMPONENT_TYPES,
    GROUP,
    ROUTING_STRATEGIES,
    VERSION,
)


def inference_service_crd() -> dict:
    raw = {"type": "object", "x-kubernetes-preserve-unknown-fields": True}
    role_schema = {
        "type": "object",
        "required": ["componentType"],
        "properties": {
            "componentType": {"type": "string", "enum": list(COMPONENT_TYPES)},
            "replicas": {"type": "integer", "minimum": 0, "default": 1},
            "template": raw,
            "multinode": {
                "type": "object",
                "properties": {
                    "nodeCount": {"type": "integer", "minimum": 1}
                },
            },
            "routingStrategy": {
                "type": "string",
                "enum": list(ROUTING_STRATEGIES),
            },
            "endpointPickerConfig": {"type": "string"},
            "httpRoute": raw,
            "gateway": raw,
        },
    }
    component_status = {
        "type": "object",
        "properties": {
            "componentType": {"type": "string"},
            "replicas": {"type": "integer"},
            "readyReplicas": {"type": "integer"},
            "readyPods": {"type": "integer"},
            "phase": {
                "type": "string",
                "enum": ["Pending", "Deploying", "Running", "Failed"],
            },
        },
    }
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"inferenceservices.{GROUP}"},
        "spec": {
            "group": GROUP,
            "names": {
                "kind": "InferenceService",
                "listKind": "InferenceServiceList",
                "plural": "inferenceservices",
                "singular": "inferenceservice",
                "shortNames": ["isvc"],
            },
            "scope": "Namespaced",
            "versions": [
                {
                    "name": VERSION,
                    "served": True,
                    "storage": True,
                    "subresources": {"status": {}},
                    "schema": {
                        "openAPIV3Schema": {
                            "type": "object",
                            "properties": {
                                "spec": {
                                    "type": "object",
                                    "required": ["roles"],
                                    "properties": {
                                        "roles": {
                                            "type": "array",
                                            "items": role_schema,
                                        }
                                    },
                                },
                                "status": {
                                    "type": "object",
                                    "properties": {
                                        "conditions": {
                                            "type": "array",
                                            "items": {
                                                "type": "object",
                                                "x-kubernetes-preserve-unknown-fields": True,
                                            },
                                        },
                                        "components": {
                                            "type": "object",
                                            "additionalProperties": component_status,
                                        },
                                    },
                                },
                            },
                        }
                    },
                }
            ],
        },
    }


def render_crd_yaml() -> str:
    return yaml.safe_dump(inference_service_crd(), sort_keys=False)
