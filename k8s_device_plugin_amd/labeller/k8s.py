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

        if token is None:
            token_path = os.path.join(SA_DIR, "token")
            if os.path.exists(token_path):
                with open(token_path) as f:
                    token = f.read().strip()
        if ca_cert is None:
            ca_path = os.path.join(SA_DIR, "ca.crt")
            if os.path.exists(ca_path):
                ca_cert = ca_path

        self.session = session or requests.Session()
        if token:
            self.session.headers["Authorization"] = f"Bearer {token}"
        if ca_cert:
            self.session.verify = ca_cert

    def get_node(self, name: str) -> dict:
        r = self.session.get(f"{self.base_url}/api/v1/nodes/{name}", timeout=10)
        r.raise_for_status()
        return r.json()

    def patch_node_labels(self, name: str, labels: Dict[str, Optional[str]]) -> dict:
        """Strategic-merge patch of metadata.labels; None values delete."""
        body = {"metadata": {"labels": labels}}
        r = self.session.patch(
            f"{self.base_url}/api/v1/nodes/{name}",
            data=json.dumps(body),
            headers={"Content-Type": "application/strategic-merge-patch+json"},
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

        Returns when `stop` is set; reconnects on stream end like an
        informer (resourceVersion handling kept minimal: relist each
        reconnect, which is correct for this consumer — labels are
        recomputed idempotently)."""
        url = (
            f"{self.base_url}/api/v1/nodes"
            f"?watch=true&fieldSelector=metadata.name={name}"
            f"&timeoutSeconds={timeout_seconds}"
        )
        while not stop.is_set():
            try:
                with self.session.get(url, stream=True, timeout=timeout_seconds + 10) as r:
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
                        on_event(evt.get("type", ""), evt.get("object", {}))
            except Exception as e:
                if stop.is_set():
                    return
                log.warning("node watch interrupted: %s; reconnecting", e)
                stop.wait(2.0)
