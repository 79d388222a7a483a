"""Minimal Kubernetes API client for Node objects.

Replaces the reference's controller-runtime/client-go stack (reference:
cmd/k8s-node-labeller/main.go:416-472) with a small in-cluster HTTPS client:
service-account token auth, Node get/patch, and a watch stream filtered to
one node name.  Injectable base URL/session for tests (fake API server).
"""

from __future__ import annotations

import json
import logging
import os
import threading
from typing import Callable, Dict, Optional

log = logging.getLogger(__name__)

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class K8sClient:
    def __init__(
        self,
        base_url: Optional[str] = None,
        token: Optional[str] = None,
        ca_cert: Optional[str] = None,
        session=None,
    ):
        import requests

        if base_url is None:
            host = os.environ.get("KUBERNETES_SERVICE_HOST", "kubernetes.default.svc")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            base_url = f"https://{host}:{port}"
        self.base_url = base_url.rstrip("/")

        # Kubernetes bound service-account tokens rotate (~1h); the kubelet
        # refreshes the projected file in place.  client-go re-reads it per
        # request, so we must too: remember the *path* and re-read the token
        # before every request instead of pinning a static header (a pinned
        # header starts getting 401s after the first rotation and never
        # recovers until pod restart).
        self._token_path: Optional[str] = None
        self._static_token = token
        if token is None:
            token_path = os.path.join(SA_DIR, "token")
            if os.path.exists(token_path):
                self._token_path = token_path
        if ca_cert is None:
            ca_path = os.path.join(SA_DIR, "ca.crt")
            if os.path.exists(ca_path):
                ca_cert = ca_path

        self.session = session or requests.Session()
        if ca_cert:
            self.session.verify = ca_cert

    def _auth_headers(self) -> Dict[str, str]:
        """Fresh Authorization header; re-reads the projected SA token file."""
        token = self._static_token
        if token is None and self._token_path:
            try:
                with open(self._token_path) as f:
                    token = f.read().strip()
            except OSError:
                token = None
        return {"Authorization": f"Bearer {token}"} if token else {}

    def get_node(self, name: str) -> dict:
        r = self.session.get(
            f"{self.base_url}/api/v1/nodes/{name}",
            headers=self._auth_headers(),
            timeout=10,
        )
        r.raise_for_status()
        return r.json()

    def patch_node_labels(self, name: str, labels: Dict[str, Optional[str]]) -> dict:
        """Strategic-merge patch of metadata.labels; None values delete."""
        body = {"metadata": {"labels": labels}}
        headers = {"Content-Type": "application/strategic-merge-patch+json"}
        headers.update(self._auth_headers())
        r = self.session.patch(
            f"{self.base_url}/api/v1/nodes/{name}",
            data=json.dumps(body),
            headers=headers,
            timeout=10,
        )
        r.raise_for_status()
        return r.json()

    def watch_node(
        self,
        name: str,
        on_event: Callable[[str, dict], None],
        stop: threading.Event,
        timeout_seconds: int = 300,
    ) -> None:
        """Long-poll the watch API for one node; calls on_event(type, node).

        Returns when `stop` is set.  Tracks metadata.resourceVersion across
        events and resumes from it on reconnect like a client-go informer,
        so a 300 s watch timeout does not force the apiserver to replay the
        full object on every reconnect.  A 410 Gone (resourceVersion too
        old) clears the bookmark and relists once, which is the informer's
        recovery path too."""
        base = (
            f"{self.base_url}/api/v1/nodes"
            f"?watch=true&fieldSelector=metadata.name={name}"
            f"&timeoutSeconds={timeout_seconds}"
            f"&allowWatchBookmarks=true"
        )
        resource_version: Optional[str] = None
        while not stop.is_set():
            url = base
            if resource_version:
                url += f"&resourceVersion={resource_version}"
            try:
                with self.session.get(
                    url,
                    stream=True,
                    headers=self._auth_headers(),
                    timeout=timeout_seconds + 10,
                ) as r:
                    if r.status_code == 410:
                        resource_version = None
                        continue
                    r.raise_for_status()
                    for line in r.iter_lines():
                        if stop.is_set():
                            return
                        if not line:
                            continue
                        try:
                            evt = json.loads(line)
                        except ValueError:
                            continue
                        etype = evt.get("type", "")
                        obj = evt.get("object", {})
                        rv = (obj.get("metadata") or {}).get("resourceVersion")
                        if rv:
                            resource_version = rv
                        if etype == "ERROR":
                            # apiserver-sent Status (e.g. 410 inside stream)
                            resource_version = None
                            break
                        if etype == "BOOKMARK":
                            continue
                        on_event(etype, obj)
            except Exception as e:
                if stop.is_set():
                    return
                log.warning("node watch interrupted: %s; reconnecting", e)
                stop.wait(2.0)
