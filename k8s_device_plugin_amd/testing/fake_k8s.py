"""Fake Kubernetes API server (nodes only) for labeller tests.

Supports GET/PATCH /api/v1/nodes/<name> with strategic-merge label
semantics and the watch endpoint, streaming queued events.
"""

from __future__ import annotations

import json
import queue
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Dict, Optional


class FakeK8s:
    def __init__(self, node_name: str = "node-0",
                 initial_labels: Optional[Dict[str, str]] = None):
        self.node_name = node_name
        self.labels: Dict[str, str] = dict(initial_labels or {})
        self.patches = []  # applied label patches, in order
        self.watch_requests = []  # raw query strings of watch GETs
        self.auth_headers = []  # Authorization header of every request
        self.resource_version = 100  # bumped on every mutation/event
        self.fail_next_watch_410 = False  # one-shot HTTP 410 answer
        self._events: "queue.Queue" = queue.Queue()
        self._server: Optional[ThreadingHTTPServer] = None
        self._thread: Optional[threading.Thread] = None
        self._lock = threading.Lock()

    # ---- state ----

    def node_object(self) -> dict:
        with self._lock:
            return {
                "apiVersion": "v1",
                "kind": "Node",
                "metadata": {
                    "name": self.node_name,
                    "labels": dict(self.labels),
                    "resourceVersion": str(self.resource_version),
                },
            }

    def push_event(self, evt_type: str = "ADDED") -> None:
        with self._lock:
            self.resource_version += 1
        self._events.put({"type": evt_type, "object": self.node_object()})

    def push_bookmark(self) -> None:
        """Apiserver progress bookmark (allowWatchBookmarks=true)."""
        with self._lock:
            self.resource_version += 1
        self._events.put({
            "type": "BOOKMARK",
            "object": {
                "apiVersion": "v1",
                "kind": "Node",
                "metadata": {"name": self.node_name,
                             "resourceVersion": str(self.resource_version)},
            },
        })

    def push_gone_error(self) -> None:
        """In-stream 410 Gone Status (resourceVersion too old)."""
        self._events.put({
            "type": "ERROR",
            "object": {"kind": "Status", "code": 410, "reason": "Expired"},
        })

    # ---- server ----

    def start(self) -> "FakeK8s":
        fake = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):  # silence
                pass

            def _json(self, code: int, obj: dict) -> None:
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                fake.auth_headers.append(self.headers.get("Authorization"))
                if self.path.startswith("/api/v1/nodes?watch=true"):
                    fake.watch_requests.append(self.path)
                    if fake.fail_next_watch_410:
                        fake.fail_next_watch_410 = False
                        self._json(410, {"kind": "Status", "code": 410,
                                         "reason": "Expired"})
                        return
                    self.send_response(200)
                    self.send_header("Content-Type", "application/json")
                    self.end_headers()
                    try:
                        while True:
                            evt = fake._events.get(timeout=5)
                            self.wfile.write(json.dumps(evt).encode() + b"\n")
                            self.wfile.flush()
                    except (queue.Empty, BrokenPipeError, ConnectionError):
                        return
                elif self.path == f"/api/v1/nodes/{fake.node_name}":
                    self._json(200, fake.node_object())
                else:
                    self._json(404, {"kind": "Status", "code": 404})

            def do_PATCH(self):
                fake.auth_headers.append(self.headers.get("Authorization"))
                if self.path != f"/api/v1/nodes/{fake.node_name}":
                    self._json(404, {"kind": "Status", "code": 404})
                    return
                length = int(self.headers.get("Content-Length", "0"))
                body = json.loads(self.rfile.read(length) or b"{}")
                patch = body.get("metadata", {}).get("labels", {})
                with fake._lock:
                    fake.resource_version += 1
                    fake.patches.append(dict(patch))
                    for k, v in patch.items():
                        if v is None:
                            fake.labels.pop(k, None)
                        else:
                            fake.labels[k] = v
                self._json(200, fake.node_object())

        self._server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self._thread = threading.Thread(
            target=self._server.serve_forever, daemon=True, name="fake-k8s"
        )
        self._thread.start()
        return self

    @property
    def base_url(self) -> str:
        host, port = self._server.server_address
        return f"http://{host}:{port}"

    def stop(self) -> None:
        if self._server is not None:
            self._server.shutdown()
            self._server.server_close()
            self._server = None
