"""Label-selector association options (reference: common/association.go).

Every pod/service lookup in the reconcilers goes through these selectors so
the operator only ever lists objects it owns.
"""
from __future__ import annotations

from typing import Dict

from ..models.raycluster import RayNodeType
from ..utils import constants as C
from ..utils import names


def cluster_all_pods_selector(cluster_name: str) -> Dict[str, str]:
    return {C.RAY_CLUSTER_LABEL_KEY: names.check_label(cluster_name)}


def cluster_head_pod_selector(cluster_name: str) -> Dict[str, str]:
    return {
        C.RAY_CLUSTER_LABEL_KEY: names.check_label(cluster_name),
        C.RAY_NODE_TYPE_LABEL_KEY: RayNodeType.HEAD,
    }


def cluster_worker_pods_selector(cluster_name: str) -> Dict[str, str]:
    return {
        C.RAY_CLUSTER_LABEL_KEY: names.check_label(cluster_name),
        C.RAY_NODE_TYPE_LABEL_KEY: RayNodeType.WORKER,
    }


def cluster_group_pods_selector(cluster_name: str, group_name: str) -> Dict[str, str]:
    return {
        C.RAY_CLUSTER_LABEL_KEY: names.check_label(cluster_name),
        C.RAY_NODE_TYPE_LABEL_KEY: RayNodeType.WORKER,
        C.RAY_NODE_GROUP_LABEL_KEY: names.check_label(group_name),
    }


def originated_from_selector(owner_name: str, owner_kind: str) -> Dict[str, str]:
    return {
        C.RAY_ORIGINATED_FROM_CR_NAME_LABEL_KEY: names.check_label(owner_name),
        C.RAY_ORIGINATED_FROM_CRD_LABEL_KEY: owner_kind,
    }


def serving_pods_selector(cluster_name: str) -> Dict[str, str]:
    return {
        C.RAY_CLUSTER_LABEL_KEY: names.check_label(cluster_name),
        C.RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY: C.ENABLE_RAY_CLUSTER_SERVING_SERVICE_TRUE,
    }
