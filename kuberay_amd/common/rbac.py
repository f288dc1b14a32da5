"""Autoscaler RBAC objects (reference: common/rbac.go:30-60).

The in-pod autoscaler needs: pods get/list/watch/patch (+pods/resize patch)
and rayclusters get/patch — exactly enough to PATCH
``spec.workerGroupSpecs[i].replicas`` / ``scaleStrategy.workersToDelete``.
"""
from __future__ import annotations

from ..kube.objects import ObjectMeta, PolicyRule, Role, RoleBinding, ServiceAccount
from ..models.raycluster import RayCluster
from ..utils import constants as C
from ..utils import names


def _meta(cluster: RayCluster, name: str) -> ObjectMeta:
    return ObjectMeta(
        name=name,
        namespace=cluster.metadata.namespace or "default",
        labels={
            C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name,
            C.KUBERNETES_APPLICATION_NAME_LABEL_KEY: C.APPLICATION_NAME,
            C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME,
        },
    )


def autoscaler_service_account(cluster: RayCluster) -> ServiceAccount:
    from .pod import head_service_account_name
    name = names.check_name(head_service_account_name(cluster))
    return ServiceAccount(metadata=_meta(cluster, name))


def autoscaler_role(cluster: RayCluster) -> Role:
    name = names.check_name(cluster.metadata.name)
    return Role(
        metadata=_meta(cluster, name),
        rules=[
            PolicyRule(api_groups=[""], resources=["pods"],
                       verbs=["get", "list", "watch", "patch"]),
            PolicyRule(api_groups=[""], resources=["pods/resize"], verbs=["patch"]),
            PolicyRule(api_groups=[C.GROUP], resources=["rayclusters"],
                       verbs=["get", "patch"]),
        ],
    )


def autoscaler_role_binding(cluster: RayCluster) -> RoleBinding:
    from .pod import head_service_account_name
    name = names.check_name(cluster.metadata.name)
    sa_name = names.check_name(head_service_account_name(cluster))
    return RoleBinding(
        metadata=_meta(cluster, name),
        subjects=[{
            "kind": "ServiceAccount",
            "name": sa_name,
            "namespace": cluster.metadata.namespace or "default",
        }],
        role_ref={
            "apiGroup": "rbac.authorization.k8s.io",
            "kind": "Role",
            "name": name,
        },
    )
