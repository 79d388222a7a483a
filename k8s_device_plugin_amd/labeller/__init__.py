from .labels import (
    LABEL_KINDS,
    create_label_prefix,
    generate_labels,
    remove_old_node_labels,
)
from .controller import NodeLabelController
from .k8s import K8sClient

__all__ = [
    "LABEL_KINDS",
    "create_label_prefix",
    "generate_labels",
    "remove_old_node_labels",
    "NodeLabelController",
    "K8sClient",
]
