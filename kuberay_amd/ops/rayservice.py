"""RayService reconciler — active/pending two-cluster model with
zero-downtime upgrades.

Reference: ray-operator/controllers/ray/rayservice_controller.go —
cluster decision :1200-1254 + :1351-1440 (spec-hash-without-replicas),
serve config submission with cache :1551-1642 + :1896-1925, serve status
:1846-1894, head serve-label flipping :2074-2128, promotion by service
repointing :568-641 + :1927-1985, suspend :383-470, endpoints :2130-2171.

The Gateway-API incremental upgrade path (:985-1199) lives in
kuberay_amd.ops.incremental behind the RayServiceIncrementalUpgrade gate.
"""
from __future__ import annotations

import logging
import threading
import time
from typing import Callable, Dict, Optional, Tuple

from ..common import association, service as servicelib
from ..kube import objects as k8s
from ..kube.client import KubeClient
from ..kube.controller import Reconciler, Request, Result
from ..kube.events import EventRecorder, NullRecorder
from ..kube.store import AlreadyExistsError, NotFoundError, now_iso
from ..models import RayCluster, RayService
from ..models.raycluster import ClusterState
from ..models.rayservice import (
    AppStatus,
    ApplicationStatus,
    RayServiceConditionReason as Reason,
    RayServiceConditionType as Cond,
    RayServiceUpgradeType,
    ServeDeploymentStatus,
    ServiceStatus,
)
from ..utils import constants as C
from ..utils import names
from ..utils.dashboard_client import DashboardClientError
from ..utils.fake_dashboard import parse_serve_config_v2
from ..utils.hashing import hash_without_replicas_and_workers_to_delete, json_hash
from ..utils.validation import validate_rayservice_metadata, validate_rayservice_spec
from .raycluster import condition_true, set_condition

logger = logging.getLogger("kuberay.rayservice")

REQUEUE_SECONDS = 2
DEFAULT_CLUSTER_DELETION_DELAY_S = 60


class RayServiceReconciler(Reconciler):
    def __init__(self, client: KubeClient, recorder: Optional[EventRecorder] = None,
                 dashboard_factory: Optional[Callable] = None,
                 http_proxy_client=None, metrics=None,
                 cluster_deletion_delay_s: Optional[float] = None):
        self.client = client
        self.recorder = recorder or NullRecorder()
        self.dashboard_factory = dashboard_factory or self._default_dashboard
        self.http_proxy_client = http_proxy_client
        self.metrics = metrics
        self.cluster_deletion_delay_s = cluster_deletion_delay_s
        self.requeue_seconds = REQUEUE_SECONDS
        from .incremental import IncrementalUpgrader
        self.upgrader = IncrementalUpgrader(client, self.recorder)
        # serve-config cache: (namespace, service, cluster) -> config hash
        self._serve_config_cache: Dict[Tuple[str, str, str], str] = {}
        # first-unhealthy timestamps: (namespace, service, cluster) -> monotonic
        self._unhealthy_since: Dict[Tuple[str, str, str], float] = {}
        # delayed old-cluster GC: (namespace, cluster) -> not-before time
        self._pending_deletions: Dict[Tuple[str, str], float] = {}
        # goal-hash memo: uid -> (generation, hash)
        self._goal_hash_cache: Dict[str, Tuple[int, str]] = {}
        self._lock = threading.Lock()

    @staticmethod
    def _default_dashboard(url: str):
        from ..utils.dashboard_client import RayDashboardClient
        return RayDashboardClient(url)

    # ------------------------------------------------------------------
    def reconcile(self, request: Request) -> Result:
        namespace, name = request
        svc = self.client.try_get(RayService, namespace, name)
        if svc is None:
            return Result()
        if svc.spec.managed_by not in (None, "ray.io/kuberay-operator"):
            return Result()
        if svc.metadata.deletion_timestamp:
            # cleanUpServeConfigCache analog (:1896-1925)
            for key in [k for k in self._serve_config_cache
                        if k[0] == namespace and k[1] == name]:
                self._serve_config_cache.pop(key, None)
            for key in [k for k in self._unhealthy_since
                        if k[0] == namespace and k[1] == name]:
                self._unhealthy_since.pop(key, None)
            return Result()

        self._gc_old_clusters(namespace)
        self._sweep_orphan_clusters(svc)

        errs = validate_rayservice_metadata(svc.metadata) + validate_rayservice_spec(svc)
        if errs:
            self.recorder.eventf(svc, "Warning", "InvalidRayServiceSpec", "; ".join(errs))
            return Result()

        if svc.spec.suspend:
            return self._handle_suspend(svc)
        if condition_true(svc.status.conditions, Cond.SUSPENDED):
            svc.status.conditions = set_condition(
                svc.status.conditions, Cond.SUSPENDED, "False", "Resumed")

        self._reconcile_ray_cluster(svc)

        active = self._get_cluster(svc, svc.status.active_service_status.ray_cluster_name)
        pending = self._get_cluster(svc, svc.status.pending_service_status.ray_cluster_name)

        # prune serve-config cache entries for clusters that no longer exist
        # (cleanUpServeConfigCache analog — bounds memory and guarantees a
        # recreated cluster is treated as config-less)
        live_uids = {c.metadata.uid or c.metadata.name
                     for c in (active, pending) if c is not None}
        with self._lock:
            for key in [k for k in self._serve_config_cache
                        if k[0] == namespace and k[1] == name
                        and k[2] not in live_uids]:
                self._serve_config_cache.pop(key, None)

        # serve config goes to the pending cluster first, else the active one
        target, is_pending = (pending, True) if pending is not None else (active, False)
        ready = False
        if target is not None and target.status.state == ClusterState.READY:
            ready = self._reconcile_serve(svc, target, is_pending)
            self._track_health(svc, target, ready)
            if (not is_pending and not ready
                    and self._unhealthy_beyond_threshold(svc, target)):
                # reference rayservice_controller.go:1409-1440: persistently
                # unhealthy apps trigger a replacement cluster
                self._prepare_replacement_cluster(svc, target)
                pending = self._get_cluster(
                    svc, svc.status.pending_service_status.ray_cluster_name)

        if is_pending and ready:
            if self._incremental_enabled(svc) and active is not None:
                # Gateway-API weighted migration instead of instant promote
                self.upgrader.ensure_gateway_infra(svc, active, target)
                done = self.upgrader.step_traffic(svc, active, target)
                self.client.update_status(svc)
                if done:
                    self._promote(svc, target)
                    self.upgrader.cleanup(svc)
                    active, pending = target, None
                else:
                    self._reconcile_services(svc, active)
            else:
                self._promote(svc, target)
                active, pending = target, None
        elif active is not None:
            self._reconcile_services(svc, active)

        self._update_head_pod_serve_label(svc, active)
        self._calculate_status(svc, active, pending)
        return Result(requeue_after=self.requeue_seconds)

    def _incremental_enabled(self, svc: RayService) -> bool:
        from .. import features
        us = svc.spec.upgrade_strategy
        return (features.enabled("RayServiceIncrementalUpgrade")
                and us is not None
                and us.type == RayServiceUpgradeType.NEW_CLUSTER_WITH_INCREMENTAL_UPGRADE)

    # ------------------------------------------------------------------
    # cluster lifecycle
    # ------------------------------------------------------------------
    def _get_cluster(self, svc: RayService, cluster_name: Optional[str]) -> Optional[RayCluster]:
        if not cluster_name:
            return None
        return self.client.try_get(RayCluster, svc.metadata.namespace or "default",
                                   cluster_name)

    def _goal_hash(self, svc: RayService) -> str:
        # memoized per (uid, generation): the hash costs ~150 µs and the
        # 2 s requeue loop recomputes it every cycle per service
        key = svc.metadata.uid or f"{svc.metadata.namespace}/{svc.metadata.name}"
        generation = svc.metadata.generation
        if generation is not None:
            cached = self._goal_hash_cache.get(key)
            if cached is not None and cached[0] == generation:
                return cached[1]
        value = hash_without_replicas_and_workers_to_delete(
            svc.spec.ray_cluster_spec)
        if generation is not None:
            if len(self._goal_hash_cache) > 8192:
                self._goal_hash_cache.clear()
            self._goal_hash_cache[key] = (generation, value)
        return value

    def _reconcile_ray_cluster(self, svc: RayService) -> None:
        """rayservice_controller.go:1200-1254 + decision helpers :1351-1440."""
        active_name = svc.status.active_service_status.ray_cluster_name
        pending_name = svc.status.pending_service_status.ray_cluster_name
        active = self._get_cluster(svc, active_name)
        pending = self._get_cluster(svc, pending_name)
        goal = self._goal_hash(svc)

        if pending_name and pending is None:
            self._create_cluster(svc, pending_name, goal)
            return
        if pending is not None:
            if self._cluster_hash(pending) != goal:
                if active is not None and self._cluster_hash(active) == goal:
                    # spec reverted to the live cluster mid-upgrade → rollback
                    svc.status.conditions = set_condition(
                        svc.status.conditions, "RollbackInProgress", "True",
                        "SpecRevertedToActive")
                    self.upgrader.rollback(svc, pending)
                    self.client.update_status(svc)
                    return
                # spec changed while upgrading: replace the pending cluster
                self._delete_cluster_later(pending, delay=0)
                svc.status.pending_service_status.ray_cluster_name = \
                    names.ray_cluster_name_for(svc.metadata.name)
                self.client.update_status(svc)
            return

        if active is None:
            if not active_name:
                # brand new service → first cluster is "pending" until serve is up
                name = names.ray_cluster_name_for(svc.metadata.name)
                svc.status.pending_service_status.ray_cluster_name = name
                self.client.update_status(svc)
                self._create_cluster(svc, name, goal)
            else:
                self._create_cluster(svc, active_name, goal)
            return

        if self._cluster_hash(active) == goal:
            self._sync_scale_in_place(svc, active)
            return

        upgrade_type = (svc.spec.upgrade_strategy.type
                        if svc.spec.upgrade_strategy else None)
        if upgrade_type == RayServiceUpgradeType.NONE:
            return  # wait for manual intervention (reference behavior)
        # zero-downtime: prepare a pending cluster
        name = names.ray_cluster_name_for(svc.metadata.name)
        svc.status.pending_service_status.ray_cluster_name = name
        svc.status.conditions = set_condition(
            svc.status.conditions, Cond.UPGRADE_IN_PROGRESS, "True",
            "ZeroDowntimeUpgrade", "Preparing a new RayCluster")
        self.client.update_status(svc)
        self._create_cluster(svc, name, goal)

    def _cluster_hash(self, cluster: RayCluster) -> Optional[str]:
        return (cluster.metadata.annotations or {}).get(
            C.HASH_WITHOUT_REPLICAS_AND_WORKERS_TO_DELETE_KEY)

    def _sync_scale_in_place(self, svc: RayService, active: RayCluster) -> None:
        """Replicas-only changes update the live cluster (in-place)."""
        desired = svc.spec.ray_cluster_spec
        changed = False
        by_name = {g.group_name: g for g in active.spec.worker_group_specs}
        for group in desired.worker_group_specs:
            live = by_name.get(group.group_name)
            if live is None:
                active.spec.worker_group_specs.append(group.clone())
                changed = True
            elif (live.replicas != group.replicas
                  or live.min_replicas != group.min_replicas
                  or live.max_replicas != group.max_replicas):
                live.replicas = group.replicas
                live.min_replicas = group.min_replicas
                live.max_replicas = group.max_replicas
                changed = True
        if changed:
            self.client.update(active)

    def _create_cluster(self, svc: RayService, cluster_name: str, goal_hash: str) -> None:
        cluster = RayCluster(
            metadata=k8s.ObjectMeta(
                name=cluster_name,
                namespace=svc.metadata.namespace or "default",
                labels={
                    C.RAY_ORIGINATED_FROM_CR_NAME_LABEL_KEY:
                        names.check_label(svc.metadata.name),
                    C.RAY_ORIGINATED_FROM_CRD_LABEL_KEY: C.KIND_RAYSERVICE,
                },
                annotations={
                    C.HASH_WITHOUT_REPLICAS_AND_WORKERS_TO_DELETE_KEY: goal_hash,
                    C.ENABLE_SERVE_SERVICE_KEY: C.ENABLE_SERVE_SERVICE_TRUE,
                },
                owner_references=[k8s.owner_reference_for(svc)],
            ),
            spec=svc.spec.ray_cluster_spec.clone(),
        )
        try:
            self.client.create(cluster)
            self.recorder.eventf(svc, "Normal", "CreatedRayCluster",
                                 "Created RayCluster %s", cluster_name)
        except AlreadyExistsError:
            pass

    def _delete_cluster_later(self, cluster: RayCluster, delay: Optional[float] = None) -> None:
        if delay is None:
            delay = (self.cluster_deletion_delay_s
                     if self.cluster_deletion_delay_s is not None
                     else DEFAULT_CLUSTER_DELETION_DELAY_S)
        key = (cluster.metadata.namespace or "default", cluster.metadata.name)
        with self._lock:
            self._pending_deletions.setdefault(key, time.monotonic() + delay)

    def _sweep_orphan_clusters(self, svc: RayService) -> None:
        """Schedule deletion of owned RayClusters that are neither active nor
        pending (reference reconcileRayCluster's cleanUpRayClusterInstance).

        The delayed-deletion timer lives in operator memory; if the operator
        restarts inside the delay window the replaced cluster would otherwise
        leak forever. This list-based sweep rediscovers orphans from the API
        server on every reconcile, so restart loses only the remaining delay,
        never the deletion.
        """
        namespace = svc.metadata.namespace or "default"
        keep = {svc.status.active_service_status.ray_cluster_name,
                svc.status.pending_service_status.ray_cluster_name}
        owned = self.client.list(
            RayCluster, namespace,
            association.originated_from_selector(svc.metadata.name,
                                                 C.KIND_RAYSERVICE))
        for cluster in owned:
            if cluster.metadata.name in keep or cluster.metadata.deletion_timestamp:
                continue
            key = (namespace, cluster.metadata.name)
            with self._lock:
                already = key in self._pending_deletions
            if not already:
                self._delete_cluster_later(cluster)
                self.recorder.eventf(
                    svc, "Normal", "ScheduledOrphanClusterDeletion",
                    "RayCluster %s is neither active nor pending; scheduled "
                    "for deletion", cluster.metadata.name)

    def _gc_old_clusters(self, namespace: str) -> None:
        now = time.monotonic()
        with self._lock:
            due = [k for k, t in self._pending_deletions.items() if t <= now]
            for k in due:
                self._pending_deletions.pop(k)
        for ns, name in due:
            try:
                self.client.delete(RayCluster, ns, name)
            except NotFoundError:
                pass

    # ------------------------------------------------------------------
    # serve config & status
    # ------------------------------------------------------------------
    def _dashboard_for(self, svc: RayService, cluster: RayCluster):
        head_svc = names.head_service_name(C.KIND_RAYCLUSTER, cluster.spec,
                                           cluster.metadata.name)
        namespace = svc.metadata.namespace or "default"
        url = f"{head_svc}.{namespace}.svc.{names.cluster_domain_name()}:{C.DEFAULT_DASHBOARD_PORT}"
        return self.dashboard_factory(url)

    def _reconcile_serve(self, svc: RayService, cluster: RayCluster,
                         is_pending: bool) -> bool:
        """Submit config if needed + check app health. Returns True when all
        serve applications are RUNNING (rayservice_controller.go:1551-1894)."""
        namespace = svc.metadata.namespace or "default"
        config = parse_serve_config_v2(svc.spec.serve_config_v2 or "")
        config_hash = json_hash(config)
        # Keyed by cluster UID, not name: a cluster deleted and recreated
        # under the same name is a FRESH serve controller that has never
        # seen the config — a name-keyed cache would silently skip
        # resubmission (reference cleanUpServeConfigCache,
        # rayservice_controller.go:1896-1925).
        cache_key = (namespace, svc.metadata.name,
                     cluster.metadata.uid or cluster.metadata.name)
        dashboard = self._dashboard_for(svc, cluster)
        # Submission failures invalidate the cache (retry next reconcile);
        # STATUS-read failures must NOT — resubmitting a config the serve
        # controller already has resets deploy progress for nothing
        # (reference keeps the cache across getAndCheckServeStatus errors,
        # rayservice_controller.go:1551-1642).
        try:
            if self._serve_config_cache.get(cache_key) != config_hash:
                dashboard.update_serve_applications(config)
                self._serve_config_cache[cache_key] = config_hash
                self.recorder.eventf(svc, "Normal", "SubmittedServeConfig",
                                     "Submitted serve config to RayCluster %s",
                                     cluster.metadata.name)
        except DashboardClientError as e:
            self.recorder.eventf(svc, "Warning", "FailedServeConfigSubmission",
                                 str(e))
            self._serve_config_cache.pop(cache_key, None)
            return False
        try:
            details = dashboard.get_serve_applications()
        except DashboardClientError as e:
            self.recorder.eventf(svc, "Warning", "FailedServeStatusCheck",
                                 str(e))
            return False

        target_status = (svc.status.pending_service_status if is_pending
                         else svc.status.active_service_status)
        apps: Dict[str, AppStatus] = {}
        all_running = True
        declared = {a.get("name", "default")
                    for a in config.get("applications", [])} or {"default"}
        for app_name, app in (details.get("applications") or {}).items():
            deployments = {
                dname: ServeDeploymentStatus(
                    status=d.get("status"), message=d.get("message"))
                for dname, d in (app.get("deployments") or {}).items()
            }
            apps[app_name] = AppStatus(
                status=app.get("status"), message=app.get("message"),
                deployments=deployments)
            if app_name in declared and app.get("status") != ApplicationStatus.RUNNING:
                all_running = False
        missing = declared - set(apps)
        if missing:
            all_running = False
        target_status.applications = apps
        target_status.ray_cluster_status = cluster.status
        return all_running and bool(apps)

    # ------------------------------------------------------------------
    # unhealthy-cluster replacement
    # ------------------------------------------------------------------
    def _track_health(self, svc: RayService, cluster: RayCluster,
                      healthy: bool) -> None:
        key = (svc.metadata.namespace or "default", svc.metadata.name,
               cluster.metadata.name)
        if healthy:
            self._unhealthy_since.pop(key, None)
        else:
            self._unhealthy_since.setdefault(key, time.monotonic())

    def _unhealthy_beyond_threshold(self, svc: RayService,
                                    cluster: RayCluster) -> bool:
        key = (svc.metadata.namespace or "default", svc.metadata.name,
               cluster.metadata.name)
        since = self._unhealthy_since.get(key)
        if since is None:
            return False
        threshold = svc.spec.service_unhealthy_second_threshold
        if threshold is None:
            threshold = 900  # reference default
        return (time.monotonic() - since) > threshold

    def _prepare_replacement_cluster(self, svc: RayService,
                                     unhealthy: RayCluster) -> None:
        if svc.status.pending_service_status.ray_cluster_name:
            return  # replacement already underway
        name = names.ray_cluster_name_for(svc.metadata.name)
        svc.status.pending_service_status.ray_cluster_name = name
        svc.status.conditions = set_condition(
            svc.status.conditions, Cond.UPGRADE_IN_PROGRESS, "True",
            "UnhealthyClusterReplacement",
            f"Serve apps unhealthy on {unhealthy.metadata.name}; preparing a "
            "replacement RayCluster")
        self.client.update_status(svc)
        self.recorder.eventf(svc, "Warning", "UnhealthyServeApps",
                             "Preparing replacement RayCluster %s", name)
        self._create_cluster(svc, name, self._goal_hash(svc))

    # ------------------------------------------------------------------
    # services & promotion
    # ------------------------------------------------------------------
    def _reconcile_services(self, svc: RayService, cluster: RayCluster) -> None:
        """Create/repoint the RayService-owned head + serve services
        (rayservice_controller.go:568-641, :1927-1985)."""
        namespace = svc.metadata.namespace or "default"
        head = servicelib.build_head_service(
            cluster, creator_crd_type=C.KIND_RAYSERVICE, owner_name=svc.metadata.name)
        head.metadata.owner_references = [k8s.owner_reference_for(svc)]
        existing = self.client.try_get(k8s.Service, namespace, head.metadata.name)
        if existing is None:
            try:
                self.client.create(head)
            except AlreadyExistsError:
                pass
        elif (existing.spec.selector or {}).get(C.RAY_CLUSTER_LABEL_KEY) != cluster.metadata.name:
            existing.spec.selector = head.spec.selector
            self.client.update(existing)

        serve = servicelib.build_serve_service(svc, cluster)
        serve.metadata.owner_references = [k8s.owner_reference_for(svc)]
        existing = self.client.try_get(k8s.Service, namespace, serve.metadata.name)
        if existing is None:
            try:
                self.client.create(serve)
            except AlreadyExistsError:
                pass
        elif (existing.spec.selector or {}).get(C.RAY_CLUSTER_LABEL_KEY) != cluster.metadata.name:
            existing.spec.selector = serve.spec.selector
            self.client.update(existing)

    def _promote(self, svc: RayService, pending_cluster: RayCluster) -> None:
        """Pending → active: repoint services, schedule old-cluster deletion."""
        old_active = svc.status.active_service_status.ray_cluster_name
        self._reconcile_services(svc, pending_cluster)
        svc.status.active_service_status = svc.status.pending_service_status
        svc.status.pending_service_status = type(svc.status.pending_service_status)()
        svc.status.conditions = set_condition(
            svc.status.conditions, Cond.UPGRADE_IN_PROGRESS, "False",
            "UpgradeComplete", "Promoted pending RayCluster to active")
        self.client.update_status(svc)
        if old_active and old_active != pending_cluster.metadata.name:
            old = self._get_cluster(svc, old_active)
            if old is not None:
                self._delete_cluster_later(old)
                self.recorder.eventf(svc, "Normal", "ScheduledOldClusterDeletion",
                                     "RayCluster %s scheduled for deletion", old_active)

    # ------------------------------------------------------------------
    # head pod serve label
    # ------------------------------------------------------------------
    def _update_head_pod_serve_label(self, svc: RayService,
                                     active: Optional[RayCluster]) -> None:
        """rayservice_controller.go:2074-2128 — traffic-readiness flip for the
        head pod, driven by the serve proxy healthz."""
        if active is None:
            return
        namespace = svc.metadata.namespace or "default"
        pods = self.client.list(
            k8s.Pod, namespace,
            association.cluster_head_pod_selector(active.metadata.name))
        for pod in pods:
            labels = pod.metadata.labels or {}
            healthy = not svc.spec.exclude_head_pod_from_serve_svc and \
                self._head_proxy_healthy(pod)
            want = (C.ENABLE_RAY_CLUSTER_SERVING_SERVICE_TRUE if healthy
                    else C.ENABLE_RAY_CLUSTER_SERVING_SERVICE_FALSE)
            if labels.get(C.RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY) != want:
                self.client.patch(k8s.Pod, namespace, pod.metadata.name, {
                    "metadata": {"labels": {
                        C.RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY: want}}})

    def _head_proxy_healthy(self, pod: k8s.Pod) -> bool:
        if self.http_proxy_client is None:
            # no proxy client wired (in-process harness): pod readiness stands in
            from ..utils.resources import is_pod_running_and_ready
            return is_pod_running_and_ready(pod)
        if not pod.status.pod_ip:
            return False
        return self.http_proxy_client.check_proxy_healthy(pod.status.pod_ip)

    # ------------------------------------------------------------------
    # suspend
    # ------------------------------------------------------------------
    def _handle_suspend(self, svc: RayService) -> Result:
        """rayservice_controller.go:383-470."""
        namespace = svc.metadata.namespace or "default"
        deleted_any = False
        for cluster_name in (svc.status.active_service_status.ray_cluster_name,
                             svc.status.pending_service_status.ray_cluster_name):
            cluster = self._get_cluster(svc, cluster_name)
            if cluster is not None:
                try:
                    self.client.delete(cluster)
                    deleted_any = True
                except NotFoundError:
                    pass
        if deleted_any:
            svc.status.conditions = set_condition(
                svc.status.conditions, Cond.SUSPENDING, "True", "SuspendRequested")
            self.client.update_status(svc)
            return Result(requeue_after=self.requeue_seconds)
        svc.status.conditions = set_condition(
            svc.status.conditions, Cond.SUSPENDING, "False", "Suspended")
        svc.status.conditions = set_condition(
            svc.status.conditions, Cond.SUSPENDED, "True", "Suspended")
        svc.status.conditions = set_condition(
            svc.status.conditions, Cond.READY, "False", "Suspended")
        svc.status.service_status = ServiceStatus.NOT_RUNNING
        svc.status.num_serve_endpoints = 0
        self.client.update_status(svc)
        return Result()

    # ------------------------------------------------------------------
    # status
    # ------------------------------------------------------------------
    def _calculate_status(self, svc: RayService, active: Optional[RayCluster],
                          pending: Optional[RayCluster]) -> None:
        namespace = svc.metadata.namespace or "default"
        num_endpoints = 0
        if active is not None:
            serving = self.client.list(
                k8s.Pod, namespace,
                association.serving_pods_selector(active.metadata.name))
            from ..utils.resources import is_pod_running_and_ready
            num_endpoints = sum(1 for p in serving if is_pod_running_and_ready(p))
            svc.status.active_service_status.ray_cluster_status = active.status
        if pending is not None:
            svc.status.pending_service_status.ray_cluster_status = pending.status
        svc.status.num_serve_endpoints = num_endpoints
        if num_endpoints > 0:
            svc.status.conditions = set_condition(
                svc.status.conditions, Cond.READY, "True",
                Reason.NON_ZERO_SERVE_ENDPOINTS, "Serve endpoints are available")
            svc.status.service_status = ServiceStatus.RUNNING
        else:
            svc.status.conditions = set_condition(
                svc.status.conditions, Cond.READY, "False",
                Reason.ZERO_SERVE_ENDPOINTS, "No serve endpoints are available")
            svc.status.service_status = ServiceStatus.NOT_RUNNING
        svc.status.observed_generation = svc.metadata.generation
        svc.status.last_update_time = now_iso()
        try:
            self.client.update_status(svc)
        except NotFoundError:
            pass
