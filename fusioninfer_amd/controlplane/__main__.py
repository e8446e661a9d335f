"""Control-plane CLI: `python -m fusioninfer_amd.controlplane <cmd>`.

Commands:
  crd                      — print the InferenceService CRD YAML
  render <svc.yaml>        — reconcile an InferenceService spec against an
                             in-memory apiserver and print every child
                             resource (LWS / PodGroup / EPP stack /
                             InferencePool / HTTPRoute) as YAML — the exact
                             objects a live controller would apply.

The live-cluster reconcile loop uses the same reconciler against a real
apiserver client; in this offline environment (no kubernetes client
package, no cluster) `render` is the verification surface, mirroring the
reference's envtest strategy.
"""

from __future__ import annotations

import sys

import yaml


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    if not argv or argv[0] in ("-h", "--help"):
        print(__doc__)
        return 0
    cmd = argv[0]
    if cmd == "crd":
        from fusioninfer_amd.controlplane.crd import render_crd_yaml

        print(render_crd_yaml())
        return 0
    if cmd == "render":
        from fusioninfer_amd.controlplane.api import InferenceService
        from fusioninfer_amd.controlplane.fake import FakeClient
        from fusioninfer_amd.controlplane.reconciler import (
            InferenceServiceReconciler,
        )

        with open(argv[1]) as f:
            docs = list(yaml.safe_load_all(f))
        client = FakeClient()
        rec = InferenceServiceReconciler(client)
        for doc in docs:
            if not doc:
                continue
            svc = InferenceService.from_dict(doc)
            client.create(svc.to_dict())
            rec.reconcile(svc.name, svc.namespace)
        out = []
        for kind in (
            "PodGroup",
            "LeaderWorkerSet",
            "ServiceAccount",
            "Role",
            "RoleBinding",
            "ConfigMap",
            "Deployment",
            "Service",
            "InferencePool",
            "HTTPRoute",
            "InferenceService",
        ):
            for obj in client.list(kind, namespace=None):
                out.append(obj)
        print(yaml.safe_dump_all(out, sort_keys=False))
        return 0
    print(f"unknown command {cmd!r}", file=sys.stderr)
    return 1


if __name__ == "__main__":
    sys.exit(main())
