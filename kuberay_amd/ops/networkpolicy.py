"""NetworkPolicy controller (reference: networkpolicy_controller.go:50-474,
feature gate RayClusterNetworkPolicy).

Builds head/worker NetworkPolicies for DenyAll / DenyAllIngress /
DenyAllEgress with always-allowed intra-cluster traffic + DNS, appends user
rules, supports per-worker-group overrides, and GCs stale policies.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional

from ..kube import objects as k8s
from ..kube.client import KubeClient
from ..kube.controller import Reconciler, Request, Result
from ..kube.events import EventRecorder, NullRecorder
from ..kube.store import AlreadyExistsError, NotFoundError
from ..models import RayCluster
from ..utils import constants as C
from ..utils import names

logger = logging.getLogger("kuberay.networkpolicy")


def _intra_cluster_peer(cluster_name: str) -> dict:
    return {"podSelector": {"matchLabels": {C.RAY_CLUSTER_LABEL_KEY: cluster_name}}}


def _dns_egress() -> dict:
    return {
        "to": [{"namespaceSelector": {}}],
        "ports": [{"protocol": "UDP", "port": 53},
                  {"protocol": "TCP", "port": 53}],
    }


def build_head_network_policy(cluster: RayCluster) -> Optional[k8s.NetworkPolicy]:
    """networkpolicy_controller.go:177 buildHeadNetworkPolicy."""
    np_cfg = cluster.spec.network_policy
    if np_cfg is None or not np_cfg.mode:
        return None
    mode = np_cfg.mode
    name = cluster.metadata.name
    ingress: List[dict] = [{"from": [_intra_cluster_peer(name)]}]
    egress: List[dict] = [{"to": [_intra_cluster_peer(name)]}, _dns_egress()]
    if np_cfg.head:
        ingress += [dict(r) for r in np_cfg.head.ingress_rules or []]
        egress += [dict(r) for r in np_cfg.head.egress_rules or []]
    policy_types = []
    if mode in ("DenyAll", "DenyAllIngress"):
        policy_types.append("Ingress")
    if mode in ("DenyAll", "DenyAllEgress"):
        policy_types.append("Egress")
    spec = {
        "podSelector": {"matchLabels": {
            C.RAY_CLUSTER_LABEL_KEY: name,
            C.RAY_NODE_TYPE_LABEL_KEY: "head"}},
        "policyTypes": policy_types,
    }
    if "Ingress" in policy_types:
        spec["ingress"] = ingress
    if "Egress" in policy_types:
        spec["egress"] = egress
    return k8s.NetworkPolicy(
        metadata=k8s.ObjectMeta(
            name=names.check_name_63(f"{name}-head"),
            namespace=cluster.metadata.namespace or "default",
            labels={C.RAY_CLUSTER_LABEL_KEY: name,
                    C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME}),
        spec=spec)


def build_worker_network_policies(cluster: RayCluster) -> List[k8s.NetworkPolicy]:
    """networkpolicy_controller.go:242 buildWorkerGroupNetworkPolicy —
    one default worker policy plus per-group overrides."""
    np_cfg = cluster.spec.network_policy
    if np_cfg is None or not np_cfg.mode:
        return []
    mode = np_cfg.mode
    name = cluster.metadata.name
    out: List[k8s.NetworkPolicy] = []
    overridden = {g.group_name for g in np_cfg.worker_groups or []}

    def build(selector_extra: Dict[str, str], rules, suffix: str):
        ingress = [{"from": [_intra_cluster_peer(name)]}]
        egress = [{"to": [_intra_cluster_peer(name)]}, _dns_egress()]
        if rules is not None:
            ingress += [dict(r) for r in rules.ingress_rules or []]
            egress += [dict(r) for r in rules.egress_rules or []]
        policy_types = []
        if mode in ("DenyAll", "DenyAllIngress"):
            policy_types.append("Ingress")
        if mode in ("DenyAll", "DenyAllEgress"):
            policy_types.append("Egress")
        selector = {C.RAY_CLUSTER_LABEL_KEY: name,
                    C.RAY_NODE_TYPE_LABEL_KEY: "worker"}
        selector.update(selector_extra)
        spec = {"podSelector": {"matchLabels": selector},
                "policyTypes": policy_types}
        if "Ingress" in policy_types:
            spec["ingress"] = ingress
        if "Egress" in policy_types:
            spec["egress"] = egress
        return k8s.NetworkPolicy(
            metadata=k8s.ObjectMeta(
                name=names.check_name_63(f"{name}-{suffix}"),
                namespace=cluster.metadata.namespace or "default",
                labels={C.RAY_CLUSTER_LABEL_KEY: name,
                        C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME}),
            spec=spec)

    out.append(build({}, np_cfg.worker, "workers"))
    for wg in np_cfg.worker_groups or []:
        out.append(build({C.RAY_NODE_GROUP_LABEL_KEY: wg.group_name}, wg,
                         f"workers-{wg.group_name}"))
    # groups referenced by overrides keep the general policy too (matchLabels
    # are additive), mirroring the reference's layering
    return out


class NetworkPolicyReconciler(Reconciler):
    def __init__(self, client: KubeClient, recorder: Optional[EventRecorder] = None):
        self.client = client
        self.recorder = recorder or NullRecorder()

    def reconcile(self, request: Request) -> Result:
        namespace, name = request
        cluster = self.client.try_get(RayCluster, namespace, name)
        if cluster is None or cluster.metadata.deletion_timestamp:
            return Result()
        desired: List[k8s.NetworkPolicy] = []
        head = build_head_network_policy(cluster)
        if head is not None:
            desired.append(head)
        desired += build_worker_network_policies(cluster)

        desired_names = {p.metadata.name for p in desired}
        existing = self.client.list(
            k8s.NetworkPolicy, namespace,
            {C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name})
        for p in existing:  # stale GC (deleteStaleNetworkPolicies :423)
            if p.metadata.name not in desired_names:
                try:
                    self.client.delete(p)
                except NotFoundError:
                    pass
        existing_by_name = {p.metadata.name: p for p in existing}
        for p in desired:
            p.metadata.owner_references = [k8s.owner_reference_for(cluster)]
            cur = existing_by_name.get(p.metadata.name)
            if cur is None:
                try:
                    self.client.create(p)
                except AlreadyExistsError:
                    pass
            elif cur.spec != p.spec:
                cur.spec = p.spec
                self.client.update(cur)
        return Result()
