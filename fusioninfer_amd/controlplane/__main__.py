"""Control-plane CLI: `python -m fusioninfer_amd.controlplane <cmd>`.

Commands:
  crd                      — print the InferenceService CRD YAML
  render <svc.yaml>        — reconcile an InferenceService spec against an
                             in-memory apiserver and print every child
                             resource (LWS / PodGroup / EPP stack /
                             InferencePool / HTTPRoute) as YAML — the exact
                             objects a live controller would apply.
  run [flags] [svc.yaml…]  — the manager entrypoint (reference
                             cmd/main.go:68-222): starts the watch-driven
                             reconcile loop with health probes, metrics and
                             optional leader election. Flags:
                               --health-probe-bind-address :8081
                               --metrics-bind-address :8080
                               --leader-elect
                               --apiserver URL   talk to a remote store
                                                 (httpapi) instead of an
                                                 in-process one
                               --serve-apiserver :PORT  also expose this
                                                 manager's in-memory store
                                                 over HTTP for kubectl-like
                                                 clients / other managers
                             Any svc.yaml files are applied at startup.
"""

from __future__ import annotations

import sys

import yaml


def _port_of(addr: str) -> int:
    return int(addr.rsplit(":", 1)[-1])


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    if not argv or argv[0] in ("-h", "--help"):
        print(__doc__)
        return 0
    cmd = argv[0]
    if cmd == "run":
        import argparse

        from fusioninfer_amd.controlplane.api import InferenceService
        from fusioninfer_amd.controlplane.fake import FakeClient
        from fusioninfer_amd.controlplane.manager import Manager

        p = argparse.ArgumentParser(prog="controlplane run")
        p.add_argument("--health-probe-bind-address", default=":8081")
        p.add_argument("--metrics-bind-address", default=":8080")
        p.add_argument("--leader-elect", action="store_true")
        p.add_argument("--apiserver", default=None,
                       help="HTTP store URL (httpapi); default in-process")
        p.add_argument("--serve-apiserver", default=None,
                       help="expose the in-process store on this :PORT")
        p.add_argument("--ports-file", default=None,
                       help="write the bound probe/metrics/apiserver "
                            "addresses here as JSON (e2e discovery)")
        p.add_argument("files", nargs="*")
        args = p.parse_args(argv[1:])

        if args.apiserver:
            from fusioninfer_amd.controlplane.httpapi import HTTPClient

            client = HTTPClient(args.apiserver)
        else:
            client = FakeClient()
        api_srv = None
        if args.serve_apiserver and not args.apiserver:
            from fusioninfer_amd.controlplane.httpapi import serve_store

            api_srv = serve_store(client, _port_of(args.serve_apiserver))
            print(f"apiserver listening on {api_srv.server_address}",
                  file=sys.stderr)
        mgr = Manager(
            client,
            probe_port=_port_of(args.health_probe_bind_address),
            metrics_port=_port_of(args.metrics_bind_address),
            leader_elect=args.leader_elect,
        ).start()
        for path in args.files:
            with open(path) as f:
                for doc in yaml.safe_load_all(f):
                    if doc:
                        svc = InferenceService.from_dict(doc)
                        client.create(svc.to_dict())
        print(
            f"manager running (probes {mgr.probe_addr}, metrics "
            f"{mgr.metrics_addr}); Ctrl-C to stop",
            file=sys.stderr,
        )
        if args.ports_file:
            import json

            with open(args.ports_file, "w") as f:
                json.dump({
                    "probe": list(mgr.probe_addr or ()),
                    "metrics": list(mgr.metrics_addr or ()),
                    "apiserver": list(api_srv.server_address)
                    if api_srv is not None else None,
                }, f)
        try:
            import signal
            import threading

            stop = threading.Event()
            signal.signal(signal.SIGTERM, lambda *a: stop.set())
            signal.signal(signal.SIGINT, lambda *a: stop.set())
            stop.wait()
        finally:
            mgr.stop()
            if api_srv is not None:
                api_srv.shutdown()
        return 0
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
