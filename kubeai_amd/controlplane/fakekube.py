"""In-process fake Kubernetes API server — the envtest analog.

The reference's integration tier boots envtest (kube-apiserver + etcd,
no kubelet) and runs the whole manager against it
(test/integration/main_test.go:83-157). This fake provides the same
contract for the K8s-backed control plane: namespaced REST storage with
resourceVersions, label-selector LIST, line-delimited WATCH streams,
merge-patch, finalizer-aware DELETE (sets deletionTimestamp, removes only
once finalizers are cleared), and the Model `scale` subresource. Pods
never run — tests flip readiness by PATCHing pod status, exactly like
envtest tests do (utils_test.go:118-132).

Also usable as a standalone dev server:
    python -m kubeai_amd.controlplane.fakekube --port 8443
"""
from __future__ import annotations

import copy
import itertools
import json
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional
from urllib.parse import parse_qs, urlparse


class _Storage:
    """Resource store keyed by (group/version/plural, namespace, name)."""

    def __init__(self):
        self.lock = threading.Lock()
        self.objs: dict[tuple, dict] = {}
        self.rv = itertools.count(1)
        self.uid = itertools.count(1)
        # watch subscribers: list of (prefix, queue-ish list + condition)
        self.watchers: list[tuple[tuple, "_WatchQueue"]] = []

    def bump(self, obj: dict) -> None:
        obj.setdefault("metadata", {})["resourceVersion"] = str(next(self.rv))

    def notify(self, key: tuple, etype: str, obj: dict) -> None:
        for prefix, q in list(self.watchers):
            if key[:len(prefix)] == prefix:
                q.put({"type": etype, "object": copy.deepcopy(obj)})


class _WatchQueue:
    def __init__(self):
        self.items: list[dict] = []
        self.cond = threading.Condition()
        self.closed = False

    def put(self, item: dict) -> None:
        with self.cond:
            self.items.append(item)
            self.cond.notify_all()

    def get(self, timeout: float) -> Optional[dict]:
        with self.cond:
            if not self.items:
                self.cond.wait(timeout)
            if self.items:
                return self.items.pop(0)
            return None


def _parse_resource_path(path: str):
    """-> (gvp, namespace, name, subresource) or None.

    /api/v1/namespaces/{ns}/{plural}[/{name}[/{sub}]]
    /apis/{group}/{version}/namespaces/{ns}/{plural}[/{name}[/{sub}]]
    """
    parts = [p for p in path.split("/") if p]
    if not parts:
        return None
    if parts[0] == "api" and len(parts) >= 4 and parts[2] == "namespaces":
        group, version, rest = "", parts[1], parts[3:]
    elif parts[0] == "apis" and len(parts) >= 5 and parts[3] == "namespaces":
        group, version, rest = parts[1], parts[2], parts[4:]
    else:
        return None
    ns = rest[0]
    if len(rest) < 2:
        return None
    plural = rest[1]
    name = rest[2] if len(rest) >= 3 else None
    sub = rest[3] if len(rest) >= 4 else None
    gvp = f"{group}/{version}/{plural}" if group else f"{version}/{plural}"
    return gvp, ns, name, sub


def _match_selector(obj: dict, selector: str) -> bool:
    labels = (obj.get("metadata") or {}).get("labels") or {}
    for term in selector.split(","):
        if not term:
            continue
        if "=" in term:
            k, _, v = term.partition("=")
            if labels.get(k.rstrip("=")) != v:
                return False
        else:  # existence
            if term not in labels:
                return False
    return True


def _merge_patch(target: dict, patch: dict) -> dict:
    for k, v in patch.items():
        if v is None:
            target.pop(k, None)
        elif isinstance(v, dict) and isinstance(target.get(k), dict):
            _merge_patch(target[k], v)
        else:
            target[k] = copy.deepcopy(v)
    return target


class FakeKubeApiServer:
    def __init__(self, port: int = 0, host: str = "127.0.0.1"):
        self.storage = _Storage()
        storage = self.storage

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, *a):  # quiet
                pass

            def _send(self, code: int, obj=None):
                body = json.dumps(obj).encode() if obj is not None else b""
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def _body(self) -> dict:
                n = int(self.headers.get("Content-Length") or 0)
                raw = self.rfile.read(n) if n else b"{}"
                return json.loads(raw or b"{}")

            # ---------------------------------------------------- GET
            def do_GET(self):
                u = urlparse(self.path)
                parsed = _parse_resource_path(u.path)
                if parsed is None:
                    return self._send(404, {"message": "not found"})
                gvp, ns, name, sub = parsed
                qs = parse_qs(u.query)
                if name is None and qs.get("watch", ["false"])[0] == "true":
                    return self._watch(gvp, ns)
                with storage.lock:
                    if name is None:
                        sel = qs.get("labelSelector", [""])[0]
                        items = [
                            copy.deepcopy(o)
                            for (g, n, _), o in sorted(storage.objs.items())
                            if g == gvp and n == ns
                            and (not sel or _match_selector(o, sel))
                        ]
                        return self._send(200, {
                            "kind": "List", "items": items,
                            "metadata": {"resourceVersion": str(next(storage.rv))},
                        })
                    obj = storage.objs.get((gvp, ns, name))
                    if obj is None:
                        return self._send(404, {"message": f"{name} not found"})
                    if sub == "scale":
                        return self._send(200, _scale_of(obj))
                    return self._send(200, copy.deepcopy(obj))

            def _watch(self, gvp, ns):
                q = _WatchQueue()
                key = (gvp, ns)
                with storage.lock:
                    storage.watchers.append((key, q))
                    # replay current state as ADDED (rv handling kept
                    # simple: informers here always re-list first)
                    snapshot = [
                        copy.deepcopy(o)
                        for (g, n, _), o in sorted(storage.objs.items())
                        if g == gvp and n == ns
                    ]
                try:
                    self.send_response(200)
                    self.send_header("Content-Type", "application/json")
                    self.send_header("Transfer-Encoding", "chunked")
                    self.end_headers()

                    def emit(ev):
                        data = json.dumps(ev).encode() + b"\n"
                        self.wfile.write(f"{len(data):x}\r\n".encode())
                        self.wfile.write(data + b"\r\n")
                        self.wfile.flush()

                    for o in snapshot:
                        emit({"type": "ADDED", "object": o})
                    while True:
                        ev = q.get(timeout=1.0)
                        if ev is not None:
                            emit(ev)
                except (BrokenPipeError, ConnectionResetError, OSError):
                    pass
                finally:
                    with storage.lock:
                        try:
                            storage.watchers.remove((key, q))
                        except ValueError:
                            pass

            # ---------------------------------------------------- POST
            def do_POST(self):
                parsed = _parse_resource_path(urlparse(self.path).path)
                if parsed is None:
                    return self._send(404, {"message": "not found"})
                gvp, ns, name, sub = parsed
                obj = self._body()
                name = (obj.get("metadata") or {}).get("name")
                if not name:
                    return self._send(422, {"message": "metadata.name required"})
                with storage.lock:
                    key = (gvp, ns, name)
                    if key in storage.objs:
                        return self._send(409, {"message": f"{name} exists"})
                    md = obj.setdefault("metadata", {})
                    md["namespace"] = ns
                    md.setdefault("uid", f"uid-{next(storage.uid)}")
                    md["generation"] = 1
                    md["creationTimestamp"] = _now()
                    storage.bump(obj)
                    storage.objs[key] = obj
                    storage.notify((gvp, ns), "ADDED", obj)
                    return self._send(201, copy.deepcopy(obj))

            # ---------------------------------------------------- PUT
            def do_PUT(self):
                parsed = _parse_resource_path(urlparse(self.path).path)
                if parsed is None:
                    return self._send(404, {"message": "not found"})
                gvp, ns, name, sub = parsed
                body = self._body()
                with storage.lock:
                    key = (gvp, ns, name)
                    old = storage.objs.get(key)
                    if old is None:
                        return self._send(404, {"message": f"{name} not found"})
                    if sub == "scale":
                        reps = ((body.get("spec") or {}).get("replicas"))
                        old.setdefault("spec", {})["replicas"] = reps
                        old["metadata"]["generation"] = (
                            old["metadata"].get("generation", 1) + 1
                        )
                        storage.bump(old)
                        storage.notify((gvp, ns), "MODIFIED", old)
                        return self._send(200, _scale_of(old))
                    md = body.setdefault("metadata", {})
                    md["namespace"] = ns
                    md["name"] = name
                    md.setdefault("uid", old["metadata"].get("uid"))
                    md["creationTimestamp"] = old["metadata"].get(
                        "creationTimestamp"
                    )
                    if old.get("spec") != body.get("spec"):
                        md["generation"] = old["metadata"].get("generation", 1) + 1
                    else:
                        md["generation"] = old["metadata"].get("generation", 1)
                    if old["metadata"].get("deletionTimestamp"):
                        md["deletionTimestamp"] = old["metadata"]["deletionTimestamp"]
                    storage.bump(body)
                    storage.objs[key] = body
                    self._finalize_if_due(key)
                    if key in storage.objs:
                        storage.notify((gvp, ns), "MODIFIED", body)
                        return self._send(200, copy.deepcopy(body))
                    return self._send(200, copy.deepcopy(body))

            # ---------------------------------------------------- PATCH
            def do_PATCH(self):
                parsed = _parse_resource_path(urlparse(self.path).path)
                if parsed is None:
                    return self._send(404, {"message": "not found"})
                gvp, ns, name, sub = parsed
                patch = self._body()
                with storage.lock:
                    key = (gvp, ns, name)
                    obj = storage.objs.get(key)
                    if obj is None:
                        return self._send(404, {"message": f"{name} not found"})
                    before_spec = copy.deepcopy(obj.get("spec"))
                    if sub == "status":
                        _merge_patch(obj.setdefault("status", {}),
                                     patch.get("status", patch))
                    else:
                        _merge_patch(obj, patch)
                    obj["metadata"]["name"] = name  # immutable
                    obj["metadata"]["namespace"] = ns
                    if obj.get("spec") != before_spec:
                        obj["metadata"]["generation"] = (
                            obj["metadata"].get("generation", 1) + 1
                        )
                    storage.bump(obj)
                    self._finalize_if_due(key)
                    if key in storage.objs:
                        storage.notify((gvp, ns), "MODIFIED", obj)
                        return self._send(200, copy.deepcopy(obj))
                    return self._send(200, copy.deepcopy(obj))

            # ---------------------------------------------------- DELETE
            def do_DELETE(self):
                parsed = _parse_resource_path(urlparse(self.path).path)
                if parsed is None:
                    return self._send(404, {"message": "not found"})
                gvp, ns, name, _ = parsed
                with storage.lock:
                    key = (gvp, ns, name)
                    obj = storage.objs.get(key)
                    if obj is None:
                        return self._send(404, {"message": f"{name} not found"})
                    if obj["metadata"].get("finalizers"):
                        # graceful deletion: mark + wait for finalizers
                        if not obj["metadata"].get("deletionTimestamp"):
                            obj["metadata"]["deletionTimestamp"] = _now()
                            storage.bump(obj)
                            storage.notify((gvp, ns), "MODIFIED", obj)
                        return self._send(200, copy.deepcopy(obj))
                    storage.objs.pop(key)
                    storage.notify((gvp, ns), "DELETED", obj)
                    return self._send(200, copy.deepcopy(obj))

            def _finalize_if_due(self, key):
                """deletionTimestamp set + finalizers emptied -> remove."""
                obj = storage.objs.get(key)
                if (
                    obj is not None
                    and obj["metadata"].get("deletionTimestamp")
                    and not obj["metadata"].get("finalizers")
                ):
                    storage.objs.pop(key)
                    storage.notify(key[:2], "DELETED", obj)

        self._httpd = ThreadingHTTPServer((host, port), Handler)
        self._httpd.daemon_threads = True
        self.port = self._httpd.server_address[1]
        self.url = f"http://{host}:{self.port}"
        self._thread = threading.Thread(
            target=self._httpd.serve_forever, daemon=True
        )

    def start(self) -> "FakeKubeApiServer":
        self._thread.start()
        return self

    def stop(self) -> None:
        self._httpd.shutdown()
        self._httpd.server_close()


def _scale_of(obj: dict) -> dict:
    # subresources.scale: specReplicasPath .spec.replicas,
    # statusReplicasPath .status.replicas.all (deploy/crds manifest)
    return {
        "apiVersion": "autoscaling/v1",
        "kind": "Scale",
        "metadata": {
            "name": obj["metadata"]["name"],
            "namespace": obj["metadata"].get("namespace"),
            "resourceVersion": obj["metadata"].get("resourceVersion"),
        },
        "spec": {"replicas": (obj.get("spec") or {}).get("replicas") or 0},
        "status": {
            "replicas": ((obj.get("status") or {}).get("replicas") or {}).get(
                "all"
            )
            or 0
        },
    }


def _now() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


def main():
    import argparse

    p = argparse.ArgumentParser()
    p.add_argument("--port", type=int, default=8443)
    p.add_argument("--host", default="127.0.0.1")
    args = p.parse_args()
    srv = FakeKubeApiServer(port=args.port, host=args.host).start()
    print(f"fake kube api server on {srv.url}")
    try:
        while True:
            time.sleep(60)
    except KeyboardInterrupt:
        srv.stop()


if __name__ == "__main__":
    main()
