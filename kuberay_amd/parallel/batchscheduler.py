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
from ..kube.client import KubeClient
from ..kube.store import AlreadyExistsError
from ..models.raycluster import RayCluster
from ..utils import constants as C
from ..utils import names
from ..utils.quantity import add_quantities, format_quantity, parse_quantity
from ..utils.resources import pod_gpu_count, worker_group_desired_replicas

# node label published by the node labeller / our topology discovery
XGMI_ISLAND_NODE_LABEL = "amd.com/xgmi-island"
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
        server = getattr(client, "server", None)
        if server is not None:
            try:
                server.create(pg)
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
        server = getattr(client, "server", None)
        if server is not None:
            try:
                server.create(pg)
            except AlreadyExistsError:
                pass

    def add_metadata_to_pod(self, client, cluster, group_name, pod) -> None:
        pod.metadata.ensure_labels()[POD_GROUP_ANNOTATION] = pod_group_name(cluster)


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
        pod.spec.affinity = affinity


SCHEDULERS = {
    VolcanoBatchScheduler.name: VolcanoBatchScheduler,
    YunikornBatchScheduler.name: YunikornBatchScheduler,
    SchedulerPluginsBatchScheduler.name: SchedulerPluginsBatchScheduler,
    KaiBatchScheduler.name: KaiBatchScheduler,
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
