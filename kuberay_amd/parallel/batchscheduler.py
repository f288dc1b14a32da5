"""Batch scheduler plugins (reference: ray-operator/controllers/ray/batchscheduler).

``BatchScheduler`` mirrors the reference interface
(batchscheduler/interface/interface.go:14-46): the RayCluster reconciler
calls ``do_batch_scheduling_on_submission`` before creating pods and
``add_metadata_to_pod`` on every pod it builds. Plugins:

* ``VolcanoBatchScheduler`` — PodGroup + queue annotations (volcano.sh),
* ``YunikornBatchScheduler`` — task-group labels/annotations,
* ``SchedulerPluginsBatchScheduler`` — kube-sigs PodGroup,
* ``XgmiGangScheduler`` — the MI355X-native default (this module's point):
  gang-schedules GPU worker groups as all-or-nothing PodGroups AND pins
  each gang inside one xGMI island via pod affinity on the island label,
  so RCCL rings never cross a PCIe hop.
"""
from __future__ import annotations

from typing import Dict, Optional

from ..kube import objects as k8s
from ..kube.client import KubeClient, RawObjectClient
from ..kube.store import AlreadyExistsError, ApiError, NotFoundError
from ..models.raycluster import RayCluster
from ..utils import constants as C
from ..utils import names
from ..utils.quantity import add_quantities, format_quantity, parse_quantity
from ..utils.resources import pod_gpu_count, worker_group_desired_replicas

# node label published by the node labeller / our topology discovery
XGMI_ISLAND_NODE_LABEL = C.XGMI_ISLAND_NODE_LABEL
POD_GROUP_ANNOTATION = "scheduling.k8s.io/group-name"
VOLCANO_POD_GROUP_ANNOTATION = "scheduling.k8s.io/group-name"
VOLCANO_QUEUE_NAME_LABEL = "volcano.sh/queue-name"
YUNIKORN_APP_ID_LABEL = "applicationId"
YUNIKORN_QUEUE_LABEL = "queue"
YUNIKORN_TASK_GROUP_NAME_ANNOTATION = "yunikorn.apache.org/task-group-name"
YUNIKORN_TASK_GROUPS_ANNOTATION = "yunikorn.apache.org/task-groups"


def pod_group_name(cluster: RayCluster) -> str:
    return names.check_name_63(f"ray-{cluster.metadata.name}-pg")


def _min_member(cluster: RayCluster) -> int:
    return 1 + sum(worker_group_desired_replicas(g) * max(g.num_of_hosts, 1)
                   for g in cluster.spec.worker_group_specs)


def _min_resources(cluster: RayCluster) -> Dict[str, str]:
    totals: Dict[str, str] = {}
    def add(template, mult=1):
        for c in template.spec.containers:
            limits = (c.resources.limits if c.resources else None) or {}
            requests = (c.resources.requests if c.resources else None) or {}
            merged = dict(requests)
            merged.update(limits)
            for key, val in merged.items():
                amount = parse_quantity(val) * mult
                totals[key] = add_quantities(totals.get(key), format_quantity(amount))
    add(cluster.spec.head_group_spec.template)
    for g in cluster.spec.worker_group_specs:
        add(g.template, worker_group_desired_replicas(g) * max(g.num_of_hosts, 1))
    return totals


class BatchScheduler:
    """interface/interface.go:14-46."""

    name: str = "default"

    def do_batch_scheduling_on_submission(self, client: KubeClient,
                                          cluster: RayCluster) -> None:
        pass

    def add_metadata_to_pod(self, client: KubeClient, cluster: RayCluster,
                            group_name: str, pod: k8s.Pod) -> None:
        pass

    def cleanup_on_completion(self, client: KubeClient, cluster: RayCluster) -> None:
        pass


class VolcanoBatchScheduler(BatchScheduler):
    """batchscheduler/volcano/volcano_scheduler.go behavioral analog."""

    name = "volcano"

    def do_batch_scheduling_on_submission(self, client, cluster) -> None:
        pg = {
            "apiVersion": "scheduling.volcano.sh/v1beta1",
            "kind": "PodGroup",
            "metadata": {
                "name": pod_group_name(cluster),
                "namespace": cluster.metadata.namespace or "default",
                "ownerReferences": [k8s.owner_reference_for(cluster).to_dict()],
            },
            "spec": {
                "minMember": _min_member(cluster),
                "minResources": _min_resources(cluster),
                **({"queue": (cluster.metadata.labels or {}).get(
                    VOLCANO_QUEUE_NAME_LABEL)}
                   if (cluster.metadata.labels or {}).get(VOLCANO_QUEUE_NAME_LABEL)
                   else {}),
            },
        }
        try:
            RawObjectClient(client).create(pg)
        except AlreadyExistsError:
            pass

    def add_metadata_to_pod(self, client, cluster, group_name, pod) -> None:
        ann = pod.metadata.ensure_annotations()
        ann[VOLCANO_POD_GROUP_ANNOTATION] = pod_group_name(cluster)
        pod.spec.scheduler_name = "volcano"
        queue = (cluster.metadata.labels or {}).get(VOLCANO_QUEUE_NAME_LABEL)
        if queue:
            pod.metadata.ensure_labels()[VOLCANO_QUEUE_NAME_LABEL] = queue


class YunikornBatchScheduler(BatchScheduler):
    """batchscheduler/yunikorn analog: task-group labels + annotations."""

    name = "yunikorn"

    def add_metadata_to_pod(self, client, cluster, group_name, pod) -> None:
        labels = pod.metadata.ensure_labels()
        labels[YUNIKORN_APP_ID_LABEL] = names.check_label(
            f"{cluster.metadata.namespace or 'default'}-{cluster.metadata.name}")
        queue = (cluster.metadata.labels or {}).get(YUNIKORN_QUEUE_LABEL)
        if queue:
            labels[YUNIKORN_QUEUE_LABEL] = queue
        ann = pod.metadata.ensure_annotations()
        ann[YUNIKORN_TASK_GROUP_NAME_ANNOTATION] = \
            f"tg-{names.check_label(group_name)}"
        pod.spec.scheduler_name = "yunikorn"


class SchedulerPluginsBatchScheduler(BatchScheduler):
    """kube-sigs scheduler-plugins PodGroup analog."""

    name = "scheduler-plugins"

    def do_batch_scheduling_on_submission(self, client, cluster) -> None:
        pg = {
            "apiVersion": "scheduling.x-k8s.io/v1alpha1",
            "kind": "PodGroup",
            "metadata": {
                "name": pod_group_name(cluster),
                "namespace": cluster.metadata.namespace or "default",
                "ownerReferences": [k8s.owner_reference_for(cluster).to_dict()],
            },
            "spec": {"minMember": _min_member(cluster),
                     "minResources": _min_resources(cluster)},
        }
        try:
            RawObjectClient(client).create(pg)
        except AlreadyExistsError:
            pass

    def add_metadata_to_pod(self, client, cluster, group_name, pod) -> None:
        pod.metadata.ensure_labels()[POD_GROUP_ANNOTATION] = pod_group_name(cluster)

    def cleanup_on_completion(self, client, cluster) -> None:
        """Delete the PodGroup when the owning workload completes
        (volcano_scheduler.go CleanupOnCompletion behavior)."""
        try:
            RawObjectClient(client).delete(
                "PodGroup", cluster.metadata.namespace or "default",
                pod_group_name(cluster),
                api_version="scheduling.x-k8s.io/v1alpha1")
        except NotFoundError:
            pass


class KaiBatchScheduler(BatchScheduler):
    """batchscheduler/kai analog: KAI's pod-grouper gang-groups pods by
    their top owner (the RayCluster), so the operator only routes pods to
    the kai-scheduler and propagates the queue label."""

    name = "kai-scheduler"
    QUEUE_LABEL = "kai.scheduler/queue"

    def add_metadata_to_pod(self, client, cluster, group_name, pod) -> None:
        pod.spec.scheduler_name = "kai-scheduler"
        queue = (cluster.metadata.labels or {}).get(self.QUEUE_LABEL)
        if queue:
            pod.metadata.ensure_labels()[self.QUEUE_LABEL] = queue


class XgmiGangScheduler(BatchScheduler):
    """MI355X-native gang scheduling with xGMI-topology affinity.

    GPU worker groups become all-or-nothing gangs (PodGroup via the
    scheduler-plugins CRD) and every pod of a gang gets pod-affinity to the
    same ``amd.com/xgmi-island`` node label, keeping each gang's RCCL ring
    on 7x153 GB/s xGMI links instead of crossing hosts or PCIe. On a single
    8xMI355X node this degenerates gracefully (one island, affinity trivially
    satisfied).
    """

    name = "xgmi-gang"

    def __init__(self, gang_cpu_groups: bool = False):
        self.inner = SchedulerPluginsBatchScheduler()
        self.gang_cpu_groups = gang_cpu_groups

    def _gpu_groups(self, cluster: RayCluster):
        for g in cluster.spec.worker_group_specs:
            try:
                if pod_gpu_count(g.template) > 0 or self.gang_cpu_groups:
                    yield g
            except (IndexError, AttributeError):
                continue

    def do_batch_scheduling_on_submission(self, client, cluster) -> None:
        if any(True for _ in self._gpu_groups(cluster)):
            self.inner.do_batch_scheduling_on_submission(client, cluster)

    def cleanup_on_completion(self, client, cluster) -> None:
        self.inner.cleanup_on_completion(client, cluster)

    # -- island scoring ----------------------------------------------------
    @staticmethod
    def _island_capacities(client) -> Dict[str, int]:
        """Per-island usable GPU capacity from the node labeller's labels.

        A split-xGMI node (``amd.com/xgmi-fully-connected: "false"``) only
        contributes its LARGEST fully-connected island: a gang ring spanning
        the split would cross PCIe, exactly what this scheduler exists to
        avoid.
        """
        capacities: Dict[str, int] = {}
        for node in RawObjectClient(client).list("Node"):
            labels = (node.get("metadata") or {}).get("labels") or {}
            island = labels.get(C.XGMI_ISLAND_NODE_LABEL)
            if not island:
                continue
            try:
                count = int(labels.get(C.AMD_GPU_COUNT_LABEL) or 0)
                if labels.get(C.XGMI_FULLY_CONNECTED_LABEL) == "false":
                    count = int(labels.get(C.XGMI_LARGEST_ISLAND_LABEL)
                                or count)
            except ValueError:
                continue
            capacities[island] = capacities.get(island, 0) + count
        return capacities

    # sentinel island value no node carries: a gang that fits no island is
    # pinned to it so ALL its pods hold Pending — refused as a unit, never
    # split across islands (a split ring would cross PCIe)
    UNSCHEDULABLE_ISLAND = "xgmi.kuberay.amd/unschedulable"

    def _best_island(self, client, cluster: RayCluster,
                     group_name: str) -> Optional[str]:
        """Best-fit: the smallest island whose usable GPU capacity covers the
        whole gang's demand, minimizing fragmentation of big islands.

        Returns None when no island data is visible (single-node dev: the
        preferred affinity still keeps gangs together), and
        UNSCHEDULABLE_ISLAND when islands ARE known but none can hold the
        gang — the gang must refuse, not split.
        """
        group = next((g for g in cluster.spec.worker_group_specs
                      if g.group_name == group_name), None)
        if group is None:
            return None
        try:
            demand = pod_gpu_count(group.template) * \
                worker_group_desired_replicas(group) * max(group.num_of_hosts, 1)
        except (IndexError, AttributeError):
            return None
        if demand <= 0:
            return None
        capacities = self._island_capacities(client)
        if not capacities:
            return None
        fits = [(cap, name) for name, cap in capacities.items()
                if cap >= demand]
        return min(fits)[1] if fits else self.UNSCHEDULABLE_ISLAND

    def add_metadata_to_pod(self, client, cluster, group_name, pod) -> None:
        is_gpu = pod_gpu_count(pod) > 0
        if not (is_gpu or self.gang_cpu_groups):
            return
        self.inner.add_metadata_to_pod(client, cluster, group_name, pod)
        # xGMI island co-location: all gang members on nodes of one island
        gang_label = names.check_label(f"{cluster.metadata.name}-{group_name}")
        pod.metadata.ensure_labels()["ray.io/xgmi-gang"] = gang_label
        affinity = pod.spec.affinity or {}
        pod_affinity = affinity.setdefault("podAffinity", {})
        terms = pod_affinity.setdefault(
            "preferredDuringSchedulingIgnoredDuringExecution", [])
        terms.append({
            "weight": 100,
            "podAffinityTerm": {
                "labelSelector": {"matchLabels": {"ray.io/xgmi-gang": gang_label}},
                "topologyKey": XGMI_ISLAND_NODE_LABEL,
            },
        })
        # Island scoring: when labelled Nodes are visible, pin the gang to the
        # best-fit island outright (the kube scheduler then only bin-packs
        # within it); without Node data the preferred affinity above still
        # keeps gang members together.
        island = self._best_island(client, cluster, group_name)
        if island is not None:
            node_affinity = affinity.setdefault("nodeAffinity", {})
            required = node_affinity.setdefault(
                "requiredDuringSchedulingIgnoredDuringExecution",
                {"nodeSelectorTerms": []})
            required["nodeSelectorTerms"].append({
                "matchExpressions": [{"key": XGMI_ISLAND_NODE_LABEL,
                                      "operator": "In",
                                      "values": [island]}]})
        pod.spec.affinity = affinity


class SchedulingRetry(RuntimeError):
    """Raised when immutable scheduling resources were deleted this pass and
    must be recreated on a later reconcile (the workqueue's backoff retries)."""


class KubernetesWASBatchScheduler(BatchScheduler):
    """Kubernetes workload-aware scheduling (scheduling.k8s.io/v1alpha2).

    Behavioral analog of batchscheduler/kubernetes-was (scheduler.go:1-81,
    v1alpha2/kubernetes_was_v1alpha2.go): the whole RayCluster (head + every
    worker group) is gang scheduled as ONE PodGroup driven by a Workload
    with a single PodGroupTemplate named ``cluster``. Opt-in per cluster via
    the ``ray.io/gang-scheduling-enabled: "true"`` label; skipped (and any
    stale resources cleaned up) when the label is absent or in-tree
    autoscaling is on. v1alpha2 scheduling resources are immutable, so a
    minCount change deletes PodGroup→Workload in dependency order and
    raises :class:`SchedulingRetry` so a later reconcile recreates them.
    """

    name = "kubernetes-was-v1alpha2"
    PROTECTION_FINALIZER = "scheduling.k8s.io/podgroup-protection"
    GROUP_VERSION = "scheduling.k8s.io/v1alpha2"
    TEMPLATE_NAME = "cluster"

    # -- naming -----------------------------------------------------------
    def _workload_name(self, cluster) -> str:
        return cluster.metadata.name

    def _pod_group_name(self, cluster) -> str:
        return names.check_name_63(f"{cluster.metadata.name}-{self.TEMPLATE_NAME}")

    # -- gating -----------------------------------------------------------
    def _skip_reason(self, cluster: RayCluster) -> Optional[str]:
        labels = cluster.metadata.labels or {}
        if labels.get(C.RAY_GANG_SCHEDULING_ENABLED, "").lower() != "true":
            return "gang scheduling not enabled on RayCluster"
        if cluster.spec.enable_in_tree_autoscaling:
            return "autoscaling is not yet supported"
        return None

    @staticmethod
    def _owned(obj: Optional[dict], cluster: RayCluster) -> bool:
        if obj is None:
            return False
        owners = (obj.get("metadata") or {}).get("ownerReferences") or []
        return any(o.get("uid") == cluster.metadata.uid and o.get("controller")
                   for o in owners)

    # -- desired state ----------------------------------------------------
    def _gang_policy(self, cluster: RayCluster) -> dict:
        return {"gang": {"minCount": _min_member(cluster)}}

    def _build(self, cluster: RayCluster):
        ns = cluster.metadata.namespace or "default"
        meta = {
            "namespace": ns,
            "labels": {C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name},
            "ownerReferences": [k8s.owner_reference_for(cluster).to_dict()],
        }
        policy = self._gang_policy(cluster)
        workload = {
            "apiVersion": self.GROUP_VERSION, "kind": "Workload",
            "metadata": {**meta, "name": self._workload_name(cluster)},
            "spec": {
                "controllerRef": {"apiGroup": C.GROUP,
                                  "kind": "RayCluster",
                                  "name": cluster.metadata.name},
                "podGroupTemplates": [{"name": self.TEMPLATE_NAME,
                                       "schedulingPolicy": policy}],
            },
        }
        pod_group = {
            "apiVersion": self.GROUP_VERSION, "kind": "PodGroup",
            "metadata": {**meta, "name": self._pod_group_name(cluster)},
            "spec": {
                "podGroupTemplateRef": {"workload": {
                    "workloadName": self._workload_name(cluster),
                    "podGroupTemplateName": self.TEMPLATE_NAME}},
                "schedulingPolicy": policy,
            },
        }
        return workload, pod_group

    # -- immutable-resource sync ------------------------------------------
    @staticmethod
    def _min_count_of(obj: Optional[dict], *path) -> Optional[int]:
        node = (obj or {}).get("spec") or {}
        for key in path:
            if isinstance(key, int):
                node = node[key] if isinstance(node, list) and \
                    len(node) > key else None
            else:
                node = node.get(key) if isinstance(node, dict) else None
            if node is None:
                return None
        return (node.get("gang") or {}).get("minCount") \
            if isinstance(node, dict) else None

    def _delete_pod_group(self, raw, pg: dict) -> None:
        """Strip the kubelet-style protection finalizer, then delete."""
        meta = pg.setdefault("metadata", {})
        fins = meta.get("finalizers") or []
        if self.PROTECTION_FINALIZER in fins:
            meta["finalizers"] = [f for f in fins
                                  if f != self.PROTECTION_FINALIZER]
            try:
                raw.update(pg)
            except ApiError:
                pass
        ns = meta.get("namespace") or "default"
        try:
            raw.delete("PodGroup", ns, meta["name"],
                       api_version=self.GROUP_VERSION)
        except ApiError:
            pass

    def do_batch_scheduling_on_submission(self, client, cluster) -> None:
        server = RawObjectClient(client)
        if self._skip_reason(cluster) is not None:
            self.cleanup_on_completion(client, cluster)
            return
        ns = cluster.metadata.namespace or "default"
        workload, pod_group = self._build(cluster)

        existing = server.try_get("Workload", ns,
                                  workload["metadata"]["name"],
                                  api_version=self.GROUP_VERSION)
        if existing is None:
            server.create(workload)
        elif not self._owned(existing, cluster):
            raise RuntimeError(
                f"Workload {ns}/{workload['metadata']['name']} exists and is "
                "not owned by this RayCluster; rename to avoid the collision")
        elif (existing["metadata"].get("deletionTimestamp")
              or self._min_count_of(existing, "podGroupTemplates", 0,
                                    "schedulingPolicy")
              != _min_member(cluster)):
            # stale/terminating Workload: drop dependent PodGroup first
            pg = server.try_get("PodGroup", ns,
                                pod_group["metadata"]["name"],
                                api_version=self.GROUP_VERSION)
            if self._owned(pg, cluster):
                self._delete_pod_group(server, pg)
            try:
                server.delete("Workload", ns, workload["metadata"]["name"],
                              api_version=self.GROUP_VERSION)
            except ApiError:
                pass
            raise SchedulingRetry(
                f"replaced stale Workload {ns}/{workload['metadata']['name']}"
                "; retrying after deletion completes")

        pg = server.try_get("PodGroup", ns, pod_group["metadata"]["name"],
                            api_version=self.GROUP_VERSION)
        if pg is None:
            server.create(pod_group)
        elif not self._owned(pg, cluster):
            raise RuntimeError(
                f"PodGroup {ns}/{pod_group['metadata']['name']} exists and is "
                "not owned by this RayCluster; rename to avoid the collision")
        elif (pg["metadata"].get("deletionTimestamp")
              or self._min_count_of(pg, "schedulingPolicy")
              != _min_member(cluster)):
            self._delete_pod_group(server, pg)
            raise SchedulingRetry(
                f"replaced stale PodGroup {ns}/{pod_group['metadata']['name']}"
                "; retrying after deletion completes")

    def add_metadata_to_pod(self, client, cluster, group_name, pod) -> None:
        if self._skip_reason(cluster) is not None:
            return
        # gang members go to the default scheduler, which understands
        # spec.schedulingGroup in WAS-enabled clusters
        pod.spec.scheduler_name = "default-scheduler"
        pod.spec.schedulingGroup = {"podGroupName": self._pod_group_name(cluster)}

    def cleanup_on_completion(self, client, cluster) -> None:
        server = RawObjectClient(client)
        ns = cluster.metadata.namespace or "default"
        pg = server.try_get("PodGroup", ns, self._pod_group_name(cluster),
                            api_version=self.GROUP_VERSION)
        if self._owned(pg, cluster):
            self._delete_pod_group(server, pg)
        wl = server.try_get("Workload", ns, self._workload_name(cluster),
                            api_version=self.GROUP_VERSION)
        if self._owned(wl, cluster):
            try:
                server.delete("Workload", ns, self._workload_name(cluster),
                              api_version=self.GROUP_VERSION)
            except ApiError:
                pass


SCHEDULERS = {
    VolcanoBatchScheduler.name: VolcanoBatchScheduler,
    YunikornBatchScheduler.name: YunikornBatchScheduler,
    SchedulerPluginsBatchScheduler.name: SchedulerPluginsBatchScheduler,
    KaiBatchScheduler.name: KaiBatchScheduler,
    KubernetesWASBatchScheduler.name: KubernetesWASBatchScheduler,
    XgmiGangScheduler.name: XgmiGangScheduler,
}


def scheduler_for(name: Optional[str]) -> Optional[BatchScheduler]:
    """schedulermanager.go:33-106 analog: select by configured name."""
    if not name:
        return None
    cls = SCHEDULERS.get(name)
    if cls is None:
        raise ValueError(f"unknown batch scheduler '{name}' "
                         f"(known: {sorted(SCHEDULERS)})")
    return cls()
