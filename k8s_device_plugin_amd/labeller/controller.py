"""Node label reconciler.

Parity with the reference's controller (reference:
cmd/k8s-node-labeller/controller.go:23-55, main.go:440-469): labels are
computed once at startup, the controller reacts only to events for its own
node (DS_NODE_NAME), and a reconcile = remove all managed labels + apply
the fresh set.  Applied as one strategic-merge patch (removals as nulls)
instead of the reference's read-modify-Update, which avoids update
conflicts with other Node writers.
"""

from __future__ import annotations

import logging
import threading
from typing import Dict, Optional

from .k8s import K8sClient
from .labels import remove_old_node_labels

log = logging.getLogger(__name__)


class NodeLabelController:
    def __init__(self, client: K8sClient, node_name: str, labels: Dict[str, str]):
        self.client = client
        self.node_name = node_name
        self.labels = labels
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def update_labels(self, labels: Dict[str, str]) -> Dict[str, Optional[str]]:
        """Swap in freshly computed labels and reconcile if they changed
        (used by the optional --refresh-interval loop; the reference
        computes labels once per process lifetime, main.go:432)."""
        if labels == self.labels:
            return {}
        self.labels = labels
        return self.reconcile()

    def reconcile(self) -> Dict[str, Optional[str]]:
        """One reconcile pass; returns the patch that was applied."""
        node = self.client.get_node(self.node_name)
        current = dict(node.get("metadata", {}).get("labels") or {})
        cleaned = dict(current)
        remove_old_node_labels(cleaned)

        patch: Dict[str, Optional[str]] = {}
        for key in current:
            if key not in cleaned and key not in self.labels:
                patch[key] = None  # managed label no longer generated
        for key, value in self.labels.items():
            if current.get(key) != value:
                patch[key] = value

        if patch:
            self.client.patch_node_labels(self.node_name, patch)
            log.info("patched %d label(s) on node %s", len(patch), self.node_name)
        return patch

    def _on_event(self, evt_type: str, obj: dict) -> None:
        # Create events for our own node only (reference: main.go:442-466)
        if evt_type != "ADDED":
            return
        if obj.get("metadata", {}).get("name") != self.node_name:
            return
        try:
            self.reconcile()
        except Exception:
            log.exception("reconcile failed")

    def run(self, block: bool = True) -> None:
        self.reconcile()  # initial pass (the watch may replay it; idempotent)
        if block:
            self.client.watch_node(self.node_name, self._on_event, self._stop)
        else:
            self._thread = threading.Thread(
                target=self.client.watch_node,
                args=(self.node_name, self._on_event, self._stop),
                daemon=True,
                name="node-watch",
            )
            self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=3)
