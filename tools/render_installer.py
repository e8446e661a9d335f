#!/usr/bin/env python3
"""Offline fallback for `kubectl kustomize config/default`: walk the
kustomization resource tree and emit the concatenated multi-doc YAML
(reference `make build-installer` -> dist/install.yaml)."""

import os
import sys

import yaml


def render(dirpath, seen=None):
    seen = seen if seen is not None else set()
    kfile = os.path.join(dirpath, "kustomization.yaml")
    docs = []
    with open(kfile) as f:
        kust = yaml.safe_load(f) or {}
    for res in kust.get("resources", []):
        path = os.path.normpath(os.path.join(dirpath, res))
        if path in seen:
            continue
        seen.add(path)
        if os.path.isdir(path):
            docs.extend(render(path, seen))
        else:
            with open(path) as f:
                docs.extend(d for d in yaml.safe_load_all(f) if d)
    return docs


if __name__ == "__main__":
    root = sys.argv[1] if len(sys.argv) > 1 else "config/default"
    print(yaml.safe_dump_all(render(root), sort_keys=False))
