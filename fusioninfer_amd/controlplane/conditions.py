"""Status conditions (reference pkg/controller/condition.go:26-85):
Initialized / Active / Failed, with reasons and observedGeneration."""

from __future__ import annotations

import datetime
from typing import Any, Dict, List

COND_INITIALIZED = "Initialized"
COND_ACTIVE = "Active"
COND_FAILED = "Failed"

REASON_INITIALIZED = "Initialized"
REASON_PROCESSING = "Processing"
REASON_ALL_READY = "AllComponentsReady"
REASON_RECONCILE_ERROR = "ReconcileError"


def _now() -> str:
    return datetime.datetime.now(datetime.timezone.utc).strftime(
        "%Y-%m-%dT%H:%M:%SZ"
    )


def set_condition(
    status: Dict[str, Any],
    cond_type: str,
    cond_status: str,
    reason: str,
    message: str,
    generation: int,
) -> None:
    conds: List[Dict[str, Any]] = status.setdefault("conditions", [])
    for c in conds:
        if c["type"] == cond_type:
            if c["status"] != cond_status:
                c["lastTransitionTime"] = _now()
            c.update(
                status=cond_status,
                reason=reason,
                message=message,
                observedGeneration=generation,
            )
            return
    conds.append(
        {
            "type": cond_type,
            "status": cond_status,
            "reason": reason,
            "message": message,
            "observedGeneration": generation,
            "lastTransitionTime": _now(),
        }
    )


def has_condition(status: Dict[str, Any], cond_type: str) -> bool:
    return any(c["type"] == cond_type for c in status.get("conditions", []))


def set_init_condition(status, generation):
    set_condition(status, COND_INITIALIZED, "True", REASON_INITIALIZED,
                  "InferenceService initialized", generation)


def set_processing_condition(status, generation):
    set_condition(status, COND_ACTIVE, "False", REASON_PROCESSING,
                  "components are being deployed", generation)
    set_condition(status, COND_FAILED, "False", REASON_PROCESSING, "",
                  generation)


def set_active_condition(status, generation):
    set_condition(status, COND_ACTIVE, "True", REASON_ALL_READY,
                  "all components ready", generation)
    set_condition(status, COND_FAILED, "False", REASON_ALL_READY, "",
                  generation)


def set_failed_condition(status, generation, message):
    set_condition(status, COND_FAILED, "True", REASON_RECONCILE_ERROR,
                  message, generation)
    set_condition(status, COND_ACTIVE, "False", REASON_RECONCILE_ERROR,
                  message, generation)
