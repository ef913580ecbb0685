"""HTTP facade over the in-memory API server: real k8s REST conventions.

Exposes a ``FakeApiServer`` over HTTP with the wire behavior ``HttpClient``
(and kubectl-style tooling) expects: group/version path prefixes,
namespaced + cluster-scoped routes, labelSelector filtering, JSON
merge-PATCH, optimistic-concurrency PUT (409 on conflict), and chunked
``?watch=true`` streams of watch events. This closes the loop on the
real-cluster client path in CI and lets every component run as a separate
process against one mini API server (`python -m
k8s_dra_driver_gpu_amd.k8s.httpserver --port 8001`).
"""

from __future__ import annotations

import json
import logging
import re
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Any, Dict, Optional, Tuple
from urllib.parse import parse_qs, urlparse

from .client import RESOURCE_INFO
from .fakeserver import ApiError, FakeApiServer, Gone, NotFound

logger = logging.getLogger("amddra.httpserver")

# path prefixes -> resource names (built from RESOURCE_INFO)
_ROUTE_RE = re.compile(
    r"^/(?:api|apis)(?:/(?P<group>[^/]+))?/(?P<version>v[0-9a-z]+)"
    r"(?:/namespaces/(?P<namespace>[^/]+))?/(?P<resource>[^/]+)(?:/(?P<name>[^/]+))?$"
)


def _parse_path(path: str) -> Optional[Tuple[str, str, str]]:
    """-> (resource, namespace, name) or None."""
    m = _ROUTE_RE.match(path)
    if not m:
        return None
    resource = m.group("resource")
    if resource not in RESOURCE_INFO:
        return None
    # /api/v1/... has no group segment; /apis/<group>/<version>/...
    return resource, m.group("namespace") or "", m.group("name") or ""


def _parse_selector(qs: Dict[str, list]) -> Optional[Dict[str, str]]:
    sel = qs.get("labelSelector", [None])[0]
    if not sel:
        return None
    out = {}
    for part in sel.split(","):
        if "=" in part:
            k, _, v = part.partition("=")
            out[k.strip()] = v.strip()
    return out


class _Handler(BaseHTTPRequestHandler):
    protocol_version = "HTTP/1.1"
    server_version = "amd-dra-mini-apiserver"

    @property
    def api(self) -> FakeApiServer:
        return self.server.api  # type: ignore[attr-defined]

    def log_message(self, fmt, *args):
        logger.debug(fmt, *args)

    def _send_json(self, code: int, obj: Any) -> None:
        body = json.dumps(obj).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def _error(self, e: Exception) -> None:
        code = e.code if isinstance(e, ApiError) else 500
        self._send_json(code, {"kind": "Status", "code": code, "message": str(e)})

    def _body(self) -> Dict[str, Any]:
        length = int(self.headers.get("Content-Length", "0"))
        return json.loads(self.rfile.read(length)) if length else {}

    # -- verbs ---------------------------------------------------------------

    def do_GET(self):
        url = urlparse(self.path)
        if url.path == "/version":
            self._send_json(200, {"major": "1", "minor": "33"})
            return
        if url.path in ("/readyz", "/healthz", "/livez"):
            self._send_json(200, {"status": "ok"})
            return
        parsed = _parse_path(url.path)
        if parsed is None:
            self._send_json(404, {"message": f"unknown path {url.path}"})
            return
        resource, namespace, name = parsed
        qs = parse_qs(url.query)
        try:
            if name:
                self._send_json(200, self.api.get(resource, name, namespace))
                return
            if qs.get("watch", ["false"])[0] == "true":
                tmo = float(qs.get("timeoutSeconds", ["0"])[0] or 0)
                rv = qs.get("resourceVersion", [None])[0]
                bookmarks = qs.get("allowWatchBookmarks", ["false"])[0] == "true"
                self._stream_watch(resource, namespace or None,
                                   _parse_selector(qs), timeout_s=tmo,
                                   resource_version=rv,
                                   allow_bookmarks=bookmarks)
                return
            items, rv = self.api.list_with_rv(
                resource, namespace or None, _parse_selector(qs))
            self._send_json(200, {"kind": "List",
                                  "metadata": {"resourceVersion": rv},
                                  "items": items})
        except Exception as e:  # noqa: BLE001
            self._error(e)

    def _stream_watch(self, resource, namespace, selector, timeout_s=0.0,
                      resource_version=None, allow_bookmarks=False):
        import time as _time

        try:
            watch = self.api.watch(
                resource, namespace, selector,
                resource_version=resource_version,
                bookmark_interval=5.0 if allow_bookmarks else None,
            )
        except Gone as e:
            # 410: the client must relist (client-go reflector semantics)
            self._send_json(410, {"kind": "Status", "code": 410, "message": str(e)})
            return
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.send_header("Transfer-Encoding", "chunked")
        self.end_headers()
        deadline = _time.monotonic() + timeout_s if timeout_s > 0 else None
        try:
            while True:
                if deadline is not None and _time.monotonic() > deadline:
                    # standard watch semantics: the server closes the stream
                    # after timeoutSeconds; clients relist + rewatch
                    break
                ev = watch.next(timeout=1.0)
                if ev is None:
                    # keep-alive chunk boundary; loop until client disconnects
                    continue
                line = json.dumps({"type": ev.type, "object": ev.object}).encode() + b"\n"
                self.wfile.write(f"{len(line):x}\r\n".encode() + line + b"\r\n")
                self.wfile.flush()
            self.wfile.write(b"0\r\n\r\n")  # terminal chunk
            self.wfile.flush()
        except (BrokenPipeError, ConnectionResetError, OSError):
            pass
        finally:
            watch.stop()

    def do_POST(self):
        parsed = _parse_path(urlparse(self.path).path)
        if parsed is None:
            self._send_json(404, {"message": "unknown path"})
            return
        resource, namespace, _ = parsed
        try:
            obj = self._body()
            if namespace:
                obj.setdefault("metadata", {}).setdefault("namespace", namespace)
            self._send_json(201, self.api.create(resource, obj))
        except Exception as e:  # noqa: BLE001
            self._error(e)

    def do_PUT(self):
        parsed = _parse_path(urlparse(self.path).path)
        if parsed is None or not parsed[2]:
            self._send_json(404, {"message": "unknown path"})
            return
        resource, namespace, name = parsed
        try:
            obj = self._body()
            obj.setdefault("metadata", {}).setdefault("namespace", namespace)
            obj["metadata"].setdefault("name", name)
            self._send_json(200, self.api.update(resource, obj))
        except Exception as e:  # noqa: BLE001
            self._error(e)

    def do_PATCH(self):
        parsed = _parse_path(urlparse(self.path).path)
        if parsed is None or not parsed[2]:
            self._send_json(404, {"message": "unknown path"})
            return
        resource, namespace, name = parsed
        try:
            self._send_json(200, self.api.patch(resource, name, self._body(), namespace))
        except Exception as e:  # noqa: BLE001
            self._error(e)

    def do_DELETE(self):
        parsed = _parse_path(urlparse(self.path).path)
        if parsed is None or not parsed[2]:
            self._send_json(404, {"message": "unknown path"})
            return
        resource, namespace, name = parsed
        try:
            self.api.delete(resource, name, namespace)
            self._send_json(200, {"kind": "Status", "status": "Success"})
        except Exception as e:  # noqa: BLE001
            self._error(e)


class _Server(ThreadingHTTPServer):
    daemon_threads = True  # watch-stream handlers must not block shutdown


class MiniApiServer:
    def __init__(self, api: Optional[FakeApiServer] = None, port: int = 0,
                 host: str = "127.0.0.1"):
        self.api = api or FakeApiServer()
        self.httpd = _Server((host, port), _Handler)
        self.httpd.api = self.api  # type: ignore[attr-defined]
        self.port = self.httpd.server_address[1]
        self._thread: Optional[threading.Thread] = None

    def start(self) -> int:
        self._thread = threading.Thread(target=self.httpd.serve_forever, daemon=True,
                                        name="mini-apiserver")
        self._thread.start()
        return self.port

    def stop(self) -> None:
        self.httpd.shutdown()
        self.httpd.server_close()


def main() -> int:
    import argparse

    p = argparse.ArgumentParser("mini-apiserver")
    p.add_argument("--port", type=int, default=8001)
    args = p.parse_args()
    logging.basicConfig(level=logging.INFO)
    srv = MiniApiServer(port=args.port)
    srv.start()
    logger.info("mini API server on 127.0.0.1:%d", srv.port)
    threading.Event().wait()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
