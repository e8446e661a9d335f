"""HTTP apiserver + client for the control plane.

Serves the in-memory store (fake.FakeClient) over REST so the manager can
run in a DIFFERENT process from the state — the closest offline analog of
controller-runtime talking to kube-apiserver (no kubernetes wheel / kind
cluster exists in this environment; reference cmd/main.go talks to a real
apiserver). Verbs map 1:1 to the store:

    POST   /apis/{kind}                      create        (409 exists)
    GET    /apis/{kind}/{ns}/{name}          get           (404)
    PUT    /apis/{kind}/{ns}/{name}          update        (409 conflict)
    PUT    /apis/{kind}/{ns}/{name}/status   update_status (409 conflict)
    DELETE /apis/{kind}/{ns}/{name}          delete (+ ownerRef GC)
    GET    /apis/{kind}?namespace=&selector= list
    GET    /watch?kinds=A,B&initial=1        JSON-lines event stream
    POST   /lease                            leader-election lease

HTTPClient implements the same verb surface as FakeClient (duck-typed),
so Manager/Reconciler run unchanged against either."""

from __future__ import annotations

import http.client
import http.server
import json
import threading
import urllib.parse
import urllib.request
from typing import Any, Dict, List, Optional

from fusioninfer_amd.controlplane.fake import (
    AlreadyExistsError,
    ConflictError,
    FakeClient,
    NotFoundError,
    Watch,
)


# ------------------------------------------------------------------ server

def serve_store(store: FakeClient, port: int = 0,
                host: str = "127.0.0.1") -> http.server.ThreadingHTTPServer:
    """Start serving `store` over HTTP; returns the server (its
    .server_address carries the bound port; .shutdown() stops it)."""

    class Handler(http.server.BaseHTTPRequestHandler):
        protocol_version = "HTTP/1.1"

        def log_message(self, *a):
            pass

        def _json(self, code: int, payload) -> None:
            body = json.dumps(payload).encode()
            self.send_response(code)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def _body(self) -> Dict[str, Any]:
            n = int(self.headers.get("Content-Length", 0))
            return json.loads(self.rfile.read(n) or b"{}")

        def _err(self, e: Exception) -> None:
            if isinstance(e, NotFoundError):
                self._json(404, {"error": "not found", "detail": str(e)})
            elif isinstance(e, ConflictError):
                self._json(409, {"error": "conflict", "detail": str(e)})
            elif isinstance(e, AlreadyExistsError):
                self._json(409, {"error": "already exists",
                                 "detail": str(e)})
            else:
                self._json(500, {"error": repr(e)})

        # -------------------------------------------------------- routes
        def do_POST(self):
            path = urllib.parse.urlparse(self.path).path
            try:
                if path == "/lease":
                    b = self._body()
                    got = store.acquire_lease(
                        b["name"], b["identity"], b["duration_s"], b["now"],
                        b.get("namespace", "default"),
                    )
                    self._json(200, {"acquired": got})
                    return
                parts = path.strip("/").split("/")
                if len(parts) == 2 and parts[0] == "apis":
                    self._json(201, store.create(self._body()))
                    return
                self._json(404, {"error": "bad path"})
            except Exception as e:
                self._err(e)

        def do_GET(self):
            url = urllib.parse.urlparse(self.path)
            q = urllib.parse.parse_qs(url.query)
            parts = url.path.strip("/").split("/")
            try:
                if parts[0] == "watch":
                    self._stream_watch(q)
                    return
                if parts[0] != "apis":
                    self._json(404, {"error": "bad path"})
                    return
                if len(parts) == 2:
                    ns = q.get("namespace", ["default"])[0]
                    ns = None if ns == "*" else ns
                    sel = None
                    if "selector" in q and q["selector"][0]:
                        sel = dict(
                            kv.split("=", 1)
                            for kv in q["selector"][0].split(",")
                        )
                    self._json(200, store.list(parts[1], ns, sel))
                    return
                if len(parts) == 4:
                    self._json(200, store.get(parts[1], parts[3], parts[2]))
                    return
                self._json(404, {"error": "bad path"})
            except Exception as e:
                self._err(e)

        def do_PUT(self):
            parts = urllib.parse.urlparse(self.path).path.strip("/").split("/")
            try:
                if len(parts) == 4 and parts[0] == "apis":
                    self._json(200, store.update(self._body()))
                elif len(parts) == 5 and parts[4] == "status":
                    self._json(200, store.update_status(self._body()))
                else:
                    self._json(404, {"error": "bad path"})
            except Exception as e:
                self._err(e)

        def do_DELETE(self):
            parts = urllib.parse.urlparse(self.path).path.strip("/").split("/")
            try:
                if len(parts) == 4 and parts[0] == "apis":
                    store.delete(parts[1], parts[3], parts[2])
                    self._json(200, {})
                else:
                    self._json(404, {"error": "bad path"})
            except Exception as e:
                self._err(e)

        def _stream_watch(self, q) -> None:
            kinds = None
            if "kinds" in q and q["kinds"][0]:
                kinds = q["kinds"][0].split(",")
            initial = q.get("initial", ["0"])[0] == "1"
            w = store.watch(kinds=kinds, send_initial=initial)
            try:
                self.send_response(200)
                self.send_header("Content-Type", "application/jsonlines")
                self.send_header("Transfer-Encoding", "chunked")
                self.end_headers()

                def write_chunk(data: bytes) -> None:
                    self.wfile.write(f"{len(data):x}\r\n".encode())
                    self.wfile.write(data + b"\r\n")
                    self.wfile.flush()

                # subscription established — unblocks the client's watch()
                write_chunk(b'{"type":"SYNC"}\n')
                while True:
                    ev = w.poll(timeout=5.0)
                    if ev is None:
                        write_chunk(b'{"type":"PING"}\n')  # keepalive
                        continue
                    etype, obj = ev
                    write_chunk(
                        json.dumps({"type": etype, "object": obj}).encode()
                        + b"\n"
                    )
            except (BrokenPipeError, ConnectionResetError, OSError):
                pass
            finally:
                store.stop_watch(w)

    srv = http.server.ThreadingHTTPServer((host, port), Handler)
    srv.daemon_threads = True
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    return srv


# ------------------------------------------------------------------ client

class HTTPClient:
    """FakeClient-compatible verb surface over the HTTP apiserver."""

    def __init__(self, base_url: str):
        self.base = base_url.rstrip("/")
        u = urllib.parse.urlparse(self.base)
        self._host = u.hostname
        self._port = u.port
        self._watches: List[Watch] = []

    # -------------------------------------------------------------- unary
    def _req(self, method: str, path: str, body=None):
        data = json.dumps(body).encode() if body is not None else None
        req = urllib.request.Request(
            self.base + path, data=data, method=method,
            headers={"Content-Type": "application/json"},
        )
        try:
            with urllib.request.urlopen(req, timeout=10) as r:
                return json.loads(r.read() or b"{}")
        except urllib.error.HTTPError as e:
            payload = {}
            try:
                payload = json.loads(e.read() or b"{}")
            except Exception:
                pass
            detail = payload.get("detail", payload.get("error", str(e)))
            if e.code == 404:
                raise NotFoundError(detail)
            if e.code == 409:
                if payload.get("error") == "already exists":
                    raise AlreadyExistsError(detail)
                raise ConflictError(detail)
            raise RuntimeError(f"{method} {path}: {e.code} {detail}")

    def create(self, obj):
        return self._req("POST", f"/apis/{obj['kind']}", obj)

    def get(self, kind, name, namespace="default"):
        return self._req("GET", f"/apis/{kind}/{namespace}/{name}")

    def try_get(self, kind, name, namespace="default"):
        try:
            return self.get(kind, name, namespace)
        except NotFoundError:
            return None

    def update(self, obj):
        md = obj["metadata"]
        return self._req(
            "PUT",
            f"/apis/{obj['kind']}/{md.get('namespace', 'default')}/{md['name']}",
            obj,
        )

    def update_status(self, obj):
        md = obj["metadata"]
        return self._req(
            "PUT",
            f"/apis/{obj['kind']}/{md.get('namespace', 'default')}"
            f"/{md['name']}/status",
            obj,
        )

    def delete(self, kind, name, namespace="default"):
        self._req("DELETE", f"/apis/{kind}/{namespace}/{name}")

    def list(self, kind, namespace="default", label_selector=None):
        ns = "*" if namespace is None else namespace
        q = f"?namespace={urllib.parse.quote(ns)}"
        if label_selector:
            sel = ",".join(f"{k}={v}" for k, v in label_selector.items())
            q += f"&selector={urllib.parse.quote(sel)}"
        return self._req("GET", f"/apis/{kind}{q}")

    def acquire_lease(self, name, identity, duration_s, now,
                      namespace="default"):
        return self._req("POST", "/lease", {
            "name": name, "identity": identity, "duration_s": duration_s,
            "now": now, "namespace": namespace,
        })["acquired"]

    # -------------------------------------------------------------- watch
    def watch(self, kinds: Optional[List[str]] = None,
              send_initial: bool = False) -> Watch:
        w = Watch(set(kinds) if kinds else None)
        q = []
        if kinds:
            q.append("kinds=" + ",".join(kinds))
        if send_initial:
            q.append("initial=1")
        path = "/watch" + ("?" + "&".join(q) if q else "")

        conn = http.client.HTTPConnection(self._host, self._port, timeout=30)
        synced = threading.Event()

        def pump():
            try:
                conn.request("GET", path)
                resp = conn.getresponse()
                buf = b""
                while not w.closed:
                    chunk = resp.read1(65536)
                    if not chunk:
                        break
                    buf += chunk
                    while b"\n" in buf:
                        line, buf = buf.split(b"\n", 1)
                        if not line.strip():
                            continue
                        ev = json.loads(line)
                        if ev.get("type") == "SYNC":
                            synced.set()
                            continue
                        if ev.get("type") == "PING":
                            continue
                        w._emit(ev["type"], ev["object"])
            except Exception:
                pass
            finally:
                synced.set()
                conn.close()

        t = threading.Thread(target=pump, daemon=True)
        t.start()
        # block until the server has registered the subscription: events
        # issued after watch() returns are then guaranteed delivered
        synced.wait(timeout=10.0)
        w._pump_thread = t
        w._conn = conn
        self._watches.append(w)
        return w

    def stop_watch(self, w: Watch) -> None:
        w.close()
        try:
            w._conn.close()
        except Exception:
            pass
        if w in self._watches:
            self._watches.remove(w)
