"""Deterministic spec hashing (the reference's change-detection mechanism,
pkg/util/hash.go:31-44: FNV-32a over a deep dump). We hash a canonical
JSON dump of the spec with FNV-32a and render hex — the hash VALUE differs
from Go's (different dump format) but the semantics are identical:
deterministic, spec-sensitive, metadata-insensitive."""

from __future__ import annotations

import json
from typing import Any

_FNV32_OFFSET = 0x811C9DC5
_FNV32_PRIME = 0x01000193


def fnv32a(data: bytes) -> int:
    h = _FNV32_OFFSET
    for b in data:
        h ^= b
        h = (h * _FNV32_PRIME) & 0xFFFFFFFF
    return h


def compute_spec_hash(obj: Any) -> str:
    """Hash any JSON-serializable object deterministically."""
    dump = json.dumps(obj, sort_keys=True, separators=(",", ":"), default=str)
    return format(fnv32a(dump.encode()), "08x")
