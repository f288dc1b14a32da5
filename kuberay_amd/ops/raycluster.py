"""RayCluster reconciler — the core control loop.

Behavioral parity with the reference reconciler
(ray-operator/controllers/ray/raycluster_controller.go): reconcile order
:359-370, reconcilePods :931-1259, shouldDeletePod :1512-1562, suspend
atomicity via Suspending/Suspended conditions, GCS-FT Redis finalizer
machine :226-352, calculateStatus :1970-2137, autoscaler RBAC :2226-2368.

MI355X-native behavior: pod construction flows through
kuberay_amd.common.pod (amd.com/gpu → --num-gpus, RCCL-over-xGMI env,
on-device readiness gate) and gang scheduling is delegated to
kuberay_amd.parallel (xGMI-topology-aware).
"""
from __future__ import annotations

import base64
import logging
import os
import secrets
import time
from typing import Dict, List, Optional

from ..common import association, gcs_ft, ingress as ingresslib, pod as podlib, rbac, service as servicelib
from ..kube import objects as k8s
from ..kube.client import KubeClient
from ..kube.controller import Reconciler, Request, Result
from ..kube.events import EventRecorder, NullRecorder
from ..kube.store import AlreadyExistsError, NotFoundError, now_iso
from ..models.raycluster import (
    ClusterState,
    RayCluster,
    RayClusterConditionReason as Reason,
    RayClusterConditionType as Cond,
    RayNodeType,
    WorkerGroupSpec,
)
from ..utils import constants as C
from ..utils import names
from ..utils import resources as res
from ..utils.hashing import hash_without_replicas_and_workers_to_delete
from ..utils.validation import (
    validate_raycluster_metadata,
    validate_raycluster_spec,
    validate_raycluster_status,
)

logger = logging.getLogger("kuberay.raycluster")


def _deletion_ts(p):
    ts = getattr(p, "deletion_timestamp", None)
    if ts is None and hasattr(p, "metadata"):
        ts = p.metadata.deletion_timestamp
    return ts or None


def set_condition(conditions: Optional[List[k8s.Condition]], cond_type: str,
                  status: str, reason: str, message: str = "") -> List[k8s.Condition]:
    """meta.SetStatusCondition analog: update-in-place, keep transition time."""
    conditions = conditions or []
    for c in conditions:
        if c.type == cond_type:
            if c.status != status:
                c.last_transition_time = now_iso()
            c.status = status
            c.reason = reason
            c.message = message
            return conditions
    conditions.append(k8s.Condition(
        type=cond_type, status=status, reason=reason, message=message,
        last_transition_time=now_iso()))
    return conditions


def get_condition(conditions, cond_type: str) -> Optional[k8s.Condition]:
    for c in conditions or []:
        if c.type == cond_type:
            return c
    return None


def condition_true(conditions, cond_type: str) -> bool:
    c = get_condition(conditions, cond_type)
    return c is not None and c.status == "True"


def find_suspend_status(cluster: RayCluster) -> Optional[str]:
    """utils.FindRayClusterSuspendStatus."""
    if condition_true(cluster.status.conditions, Cond.SUSPENDING):
        return Cond.SUSPENDING
    if condition_true(cluster.status.conditions, Cond.SUSPENDED):
        return Cond.SUSPENDED
    return None


def should_delete_pod_view(view, node_type: str) -> (bool, str):
    """View-projection variant of shouldDeletePod (same decision table)."""
    if view.phase in ("Failed", "Succeeded"):
        return True, (f"The {node_type} Pod {view.name} status is {view.phase} "
                      "which is a terminal state.")
    if view.phase == "Running" and view.ray_container_terminated:
        if view.restart_policy == "Never":
            return True, (f"Pod {view.name} Ray container terminated and "
                          "restartPolicy=Never.")
        return False, (f"Pod {view.name} Ray container terminated but will "
                       f"restart (restartPolicy={view.restart_policy}).")
    return False, f"Pod {view.name} is healthy ({view.phase})."


def should_delete_pod(pod: k8s.Pod, node_type: str) -> (bool, str):
    """raycluster_controller.go:1512 shouldDeletePod."""
    phase = pod.status.phase
    if phase in ("Failed", "Succeeded"):
        return True, (f"The {node_type} Pod {pod.metadata.name} status is {phase} "
                      "which is a terminal state.")
    terminated = _ray_container_terminated(pod)
    if phase == "Running" and terminated is not None:
        if pod.spec.restart_policy == "Never":
            return True, (f"Pod {pod.metadata.name} Ray container terminated and "
                          "restartPolicy=Never.")
        return False, (f"Pod {pod.metadata.name} Ray container terminated but will "
                       f"restart (restartPolicy={pod.spec.restart_policy}).")
    return False, f"Pod {pod.metadata.name} is healthy ({phase})."


def _ray_container_terminated(pod: k8s.Pod):
    if not pod.spec.containers:
        return None
    ray_name = pod.spec.containers[C.RAY_CONTAINER_INDEX].name
    for cs in pod.status.container_statuses or []:
        if cs.name == ray_name and cs.state and cs.state.terminated is not None:
            return cs.state.terminated
    return None


class _TypedCacheShim:
    """Expectations-cache adapter for typed clients (REST backend): contains()
    via the informer pod-view cache when present, try_get via typed verbs."""

    def __init__(self, client):
        self.client = client

    def contains(self, kind: str, namespace: str, name: str) -> bool:
        if kind == "Pod":
            cache_contains = getattr(self.client, "pod_cache_contains", None)
            if cache_contains is not None:
                return cache_contains(namespace, name)
        return self.try_get(kind, namespace, name) is not None

    def try_get(self, kind: str, namespace: str, name: str):
        from ..kube.client import model_for_kind
        model = model_for_kind(kind)
        obj = self.client.try_get(model, namespace, name)
        return obj.to_dict() if obj is not None else None


class RayClusterReconcilerOptions:
    def __init__(self):
        self.requeue_after_seconds = int(os.environ.get(
            C.RAYCLUSTER_DEFAULT_REQUEUE_SECONDS_ENV, C.RAYCLUSTER_DEFAULT_REQUEUE_SECONDS))
        self.enable_batch_scheduler = False
        self.head_sidecar_containers: List[k8s.Container] = []
        self.worker_sidecar_containers: List[k8s.Container] = []
        self.default_container_envs: List[k8s.EnvVar] = []


class RayClusterReconciler(Reconciler):
    def __init__(self, client: KubeClient, recorder: Optional[EventRecorder] = None,
                 expectations=None, batch_scheduler=None,
                 options: Optional[RayClusterReconcilerOptions] = None,
                 metrics=None):
        from .expectations import RayClusterScaleExpectations
        self.client = client
        self.recorder = recorder or NullRecorder()
        self.expectations = expectations or RayClusterScaleExpectations()
        self.batch_scheduler = batch_scheduler
        self.options = options or RayClusterReconcilerOptions()
        self.metrics = metrics
        self._provision_start: Dict[str, float] = {}
        # spec-hash memo: (uid -> (generation, hash)). The hash excludes
        # replicas/workersToDelete so it is stable within a generation, and
        # computing it costs ~150 µs — per-pod recomputation was ~15% of a
        # 2000-pod bench step.
        self._spec_hash_cache: Dict[str, tuple] = {}

    def _spec_hash(self, cluster) -> str:
        """Memoized hash_without_replicas_and_workers_to_delete[:63].
        KUBERAY_SPEC_HASH_MEMO=0 disables the memo (A/B measurement)."""
        key = cluster.metadata.uid or f"{cluster.metadata.namespace}/{cluster.metadata.name}"
        generation = cluster.metadata.generation
        if os.environ.get("KUBERAY_SPEC_HASH_MEMO", "1") != "1":
            generation = None
        if generation is not None:
            cached = self._spec_hash_cache.get(key)
            if cached is not None and cached[0] == generation:
                return cached[1]
        value = hash_without_replicas_and_workers_to_delete(cluster.spec)[:63]
        if generation is not None:
            if len(self._spec_hash_cache) > 8192:
                self._spec_hash_cache.clear()
            self._spec_hash_cache[key] = (generation, value)
        return value

    # ------------------------------------------------------------------
    def reconcile(self, request: Request) -> Result:
        namespace, name = request
        cluster = self.client.try_get(RayCluster, namespace, name)
        if cluster is None:
            self.expectations.delete(namespace, name)
            return Result()

        if cluster.spec.managed_by not in (None, "ray.io/kuberay-operator"):
            return Result()  # externally managed (e.g. multikueue)

        if cluster.metadata.deletion_timestamp:
            return self._handle_deletion(cluster)

        # status invariants (raycluster_controller.go:212-217): an invalid
        # status (Suspending+Suspended both true) means a racing writer —
        # event + requeue, never reconcile on top of it
        status_errs = validate_raycluster_status(cluster)
        if status_errs:
            self.recorder.eventf(cluster, "Warning", "InvalidRayClusterStatus",
                                 "; ".join(status_errs))
            return Result(requeue_after=2)

        # validation (invalid spec: record event, do not requeue hot)
        errs = validate_raycluster_metadata(cluster.metadata) + validate_raycluster_spec(cluster)
        if errs:
            self.recorder.eventf(cluster, "Warning", "InvalidRayClusterSpec", "; ".join(errs))
            cluster.status.reason = "; ".join(errs)
            cluster.status.state = ClusterState.FAILED
            self._update_status_if_changed(cluster)
            return Result()

        self._ensure_finalizers(cluster)

        reconcile_err: Optional[Exception] = None
        pods_complete = True
        try:
            if podlib.is_autoscaling_enabled(cluster.spec):
                self._reconcile_autoscaler_rbac(cluster)
            if cluster.spec.head_group_spec.enable_ingress:
                self._reconcile_ingress(cluster)
            if podlib.is_auth_enabled(cluster.spec):
                self._reconcile_auth_secret(cluster)
            self._reconcile_head_service(cluster)
            if any(g.num_of_hosts > 1 for g in cluster.spec.worker_group_specs):
                self._reconcile_headless_service(cluster)
            if (cluster.metadata.annotations or {}).get(
                    C.ENABLE_SERVE_SERVICE_KEY) == C.ENABLE_SERVE_SERVICE_TRUE:
                self._reconcile_serve_service(cluster)
            self._reconcile_gcs_storage_pvc(cluster)
            pods_complete = self._reconcile_pods(cluster)
        except Exception as e:  # surface via ReplicaFailure condition + requeue
            reconcile_err = e

        self._calculate_and_update_status(cluster, reconcile_err)
        if reconcile_err is not None:
            raise reconcile_err
        # suspend state machine advances through status-only writes, which
        # the event predicates deliberately ignore — requeue hot while the
        # machine is mid-flight (reference: 2s inconsistency requeue,
        # raycluster_controller.go:400-419)
        suspended_now = condition_true(cluster.status.conditions, Cond.SUSPENDED)
        suspending_now = condition_true(cluster.status.conditions, Cond.SUSPENDING)
        if (bool(cluster.spec.suspend) != suspended_now) or suspending_now \
                or not pods_complete:
            return Result(requeue_after=2)
        return Result(requeue_after=self.options.requeue_after_seconds)

    # ------------------------------------------------------------------
    # finalizers / deletion (GCS FT Redis cleanup; controller.go:226-352)
    # ------------------------------------------------------------------
    def _gcs_ft_redis_cleanup_enabled(self, cluster: RayCluster) -> bool:
        if os.environ.get(C.ENABLE_GCS_FT_REDIS_CLEANUP, "true").lower() == "false":
            return False
        opts = cluster.spec.gcs_fault_tolerance_options
        return bool(opts is not None and (opts.redis_address or opts.backend == "redis"))

    def _ensure_finalizers(self, cluster: RayCluster) -> None:
        if not self._gcs_ft_redis_cleanup_enabled(cluster):
            return
        finalizers = cluster.metadata.finalizers or []
        if C.GCS_FT_REDIS_CLEANUP_FINALIZER not in finalizers:
            cluster.metadata.finalizers = finalizers + [C.GCS_FT_REDIS_CLEANUP_FINALIZER]
            updated = self.client.update(cluster)
            cluster.metadata = updated.metadata

    def _handle_deletion(self, cluster: RayCluster) -> Result:
        finalizers = cluster.metadata.finalizers or []
        if C.GCS_FT_REDIS_CLEANUP_FINALIZER not in finalizers:
            self.expectations.delete(cluster.metadata.namespace or "default",
                                     cluster.metadata.name)
            return Result()
        namespace = cluster.metadata.namespace or "default"
        job_name = names.redis_cleanup_job_name(cluster.metadata.name)
        job = self.client.try_get(k8s.Job, namespace, job_name)
        if job is None:
            cleanup = gcs_ft.build_redis_cleanup_job(cluster)
            cleanup.metadata.owner_references = [k8s.owner_reference_for(cluster)]
            try:
                self.client.create(cleanup)
                self.recorder.eventf(cluster, "Normal", "CreatedRedisCleanupJob",
                                     "Created Redis cleanup Job %s", cleanup.metadata.name)
            except AlreadyExistsError:
                pass
            return Result(requeue_after=2)
        finished = self._job_finished(job)
        timeout = self._gcs_ft_deletion_timeout(cluster)
        expired = self._older_than(cluster.metadata.deletion_timestamp, timeout)
        if finished or expired:
            if expired and not finished:
                self.recorder.eventf(cluster, "Warning", "RedisCleanupTimeout",
                                     "Redis cleanup did not finish in %ss; removing finalizer", timeout)
            cluster.metadata.finalizers = [
                f for f in finalizers if f != C.GCS_FT_REDIS_CLEANUP_FINALIZER]
            self.client.update(cluster)
            return Result()
        return Result(requeue_after=2)

    @staticmethod
    def _job_finished(job: k8s.Job) -> bool:
        for cond in (job.status.conditions if job.status else None) or []:
            if cond.type in ("Complete", "Failed") and cond.status == "True":
                return True
        return False

    @staticmethod
    def _gcs_ft_deletion_timeout(cluster: RayCluster) -> int:
        ann = (cluster.metadata.annotations or {}).get(
            C.RAY_CLUSTER_GCS_FT_DELETION_TIMEOUT_ANNOTATION)
        try:
            return int(ann)
        except (TypeError, ValueError):
            return C.RAYCLUSTER_GCS_FT_DELETION_TIMEOUT_DEFAULT

    @staticmethod
    def _older_than(timestamp: Optional[str], seconds: int) -> bool:
        if not timestamp:
            return False
        import calendar
        try:
            t = calendar.timegm(time.strptime(timestamp, "%Y-%m-%dT%H:%M:%SZ"))
        except ValueError:
            return False
        return time.time() - t > seconds

    # ------------------------------------------------------------------
    # child objects
    # ------------------------------------------------------------------
    def _create_if_absent(self, obj, cluster: RayCluster) -> None:
        obj.metadata.owner_references = [k8s.owner_reference_for(cluster)]
        existing = self.client.try_get(type(obj), obj.metadata.namespace or "default",
                                       obj.metadata.name)
        if existing is None:
            try:
                self.client.create(obj)
            except AlreadyExistsError:
                pass

    def _reconcile_autoscaler_rbac(self, cluster: RayCluster) -> None:
        self._create_if_absent(rbac.autoscaler_service_account(cluster), cluster)
        self._create_if_absent(rbac.autoscaler_role(cluster), cluster)
        self._create_if_absent(rbac.autoscaler_role_binding(cluster), cluster)

    def _reconcile_ingress(self, cluster: RayCluster) -> None:
        self._create_if_absent(ingresslib.build_ingress_for_head_service(cluster), cluster)

    def _reconcile_auth_secret(self, cluster: RayCluster) -> None:
        opts = cluster.spec.auth_options
        if opts and opts.secret_name:
            return  # user-managed secret
        name = names.auth_secret_name(cluster.metadata.name)
        namespace = cluster.metadata.namespace or "default"
        if self.client.try_get(k8s.Secret, namespace, name) is None:
            token = secrets.token_hex(32)
            secret = k8s.Secret(
                metadata=k8s.ObjectMeta(
                    name=name, namespace=namespace,
                    labels={C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name}),
                data={C.RAY_AUTH_TOKEN_SECRET_KEY:
                      base64.b64encode(token.encode()).decode()},
            )
            self._create_if_absent(secret, cluster)

    def _reconcile_head_service(self, cluster: RayCluster) -> None:
        svc = servicelib.build_head_service(cluster)
        self._create_if_absent(svc, cluster)

    def _reconcile_headless_service(self, cluster: RayCluster) -> None:
        self._create_if_absent(servicelib.build_headless_service(cluster), cluster)

    def _reconcile_serve_service(self, cluster: RayCluster) -> None:
        self._create_if_absent(
            servicelib.build_serve_service(cluster, cluster, is_rayservice=False), cluster)

    def _reconcile_gcs_storage_pvc(self, cluster: RayCluster) -> None:
        """Embedded RocksDB GCS backend PVC (controller.go:705-817)."""
        opts = cluster.spec.gcs_fault_tolerance_options
        if not (opts and opts.backend == "embedded"):
            return
        storage = opts.storage
        if storage and storage.claim_name:
            return  # user-provided claim
        name = names.gcs_pvc_name(cluster.metadata.name)
        namespace = cluster.metadata.namespace or "default"
        if self.client.try_get(k8s.PersistentVolumeClaim, namespace, name) is not None:
            return
        pvc = k8s.PersistentVolumeClaim(
            metadata=k8s.ObjectMeta(
                name=name, namespace=namespace,
                labels={C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name}),
            spec={
                "accessModes": (storage.access_modes if storage and storage.access_modes
                                else ["ReadWriteOnce"]),
                "resources": {"requests": {"storage":
                    (storage.size if storage and storage.size else C.GCS_STORAGE_DEFAULT_SIZE)}},
                **({"storageClassName": storage.storage_class_name}
                   if storage and storage.storage_class_name else {}),
            },
        )
        # Retain policy: PVC is not owner-referenced so it survives cluster deletion
        deletion_policy = (storage.deletion_policy if storage else None) or "Delete"
        if deletion_policy == "Retain":
            existing = self.client.try_get(k8s.PersistentVolumeClaim, namespace, name)
            if existing is None:
                try:
                    self.client.create(pvc)
                except AlreadyExistsError:
                    pass
        else:
            self._create_if_absent(pvc, cluster)

    # ------------------------------------------------------------------
    # pods
    # ------------------------------------------------------------------
    def _list_cluster_pods(self, cluster: RayCluster) -> List[k8s.Pod]:
        return self.client.list(
            k8s.Pod, cluster.metadata.namespace or "default",
            association.cluster_all_pods_selector(cluster.metadata.name))

    def _list_cluster_pod_views(self, cluster: RayCluster):
        """Hot-loop projection; falls back to typed pods for clients without
        native views."""
        namespace = cluster.metadata.namespace or "default"
        selector = association.cluster_all_pods_selector(cluster.metadata.name)
        list_views = getattr(self.client, "list_pod_views", None)
        if list_views is not None:
            return list_views(namespace, selector)
        from ..kube.store import compute_pod_view
        return [compute_pod_view(p.to_dict())
                for p in self.client.list(k8s.Pod, namespace, selector)]

    def _delete_pod_by_name(self, namespace: str, name: str) -> None:
        try:
            self.client.delete(k8s.Pod, namespace, name)
        except NotFoundError:
            pass

    def _active(self, pods):
        return [p for p in pods if not _deletion_ts(p)]

    def _reconcile_pods(self, cluster: RayCluster) -> bool:
        """Returns False when work was deferred (expectations unsatisfied) —
        the caller requeues hot, mirroring the reference's 2s inconsistency
        requeue."""
        namespace = cluster.metadata.namespace or "default"
        name = cluster.metadata.name

        # ---- suspend machine (atomic via conditions; controller.go:947-990)
        suspend_status = find_suspend_status(cluster)
        if cluster.spec.suspend or suspend_status == Cond.SUSPENDING:
            if suspend_status != Cond.SUSPENDED:
                for view in self._active(self._list_cluster_pod_views(cluster)):
                    self._delete_pod_by_name(namespace, view.name)
                return True
            if cluster.spec.suspend:
                return True  # stays suspended, no pods
        if suspend_status == Cond.SUSPENDED and not cluster.spec.suspend:
            pass  # resuming: fall through to create pods

        if self.batch_scheduler is not None:
            self.batch_scheduler.do_batch_scheduling_on_submission(self.client, cluster)

        pods = self._list_cluster_pod_views(cluster)

        # UpgradeStrategy Recreate: rebuild pods whose spec-hash label is
        # stale (reference: UpgradeStrategyRecreateHashKey)
        if (cluster.spec.upgrade_strategy is not None
                and cluster.spec.upgrade_strategy.type == "Recreate"):
            current_hash = self._spec_hash(cluster)
            stale = [p for p in self._active(pods)
                     if p.labels.get(
                         C.HASH_WITHOUT_REPLICAS_AND_WORKERS_TO_DELETE_KEY)
                     not in (None, current_hash)]
            for view in stale:
                self.recorder.eventf(cluster, "Normal", "RecreatePodForUpgrade",
                                     "Deleting Pod %s (stale spec hash, "
                                     "upgradeStrategy: Recreate)", view.name)
                group = view.labels.get(C.RAY_NODE_GROUP_LABEL_KEY, "__head__")
                if view.labels.get(C.RAY_NODE_TYPE_LABEL_KEY) == RayNodeType.HEAD:
                    group = "__head__"
                self.expectations.expect_delete_pod(namespace, name, group,
                                                    view.name)
                self._delete_pod_by_name(namespace, view.name)
            if stale:
                pods = self._list_cluster_pod_views(cluster)

        head_pods = [p for p in pods
                     if p.labels.get(C.RAY_NODE_TYPE_LABEL_KEY) == RayNodeType.HEAD]

        complete = True
        # ---- head pod singleton
        if not self.expectations.is_satisfied(
                self._cache(), namespace, name, "__head__"):
            return False
        active_heads = self._active(head_pods)
        if len(active_heads) == 1:
            head = active_heads[0]
            delete, reason = should_delete_pod_view(head, RayNodeType.HEAD)
            if delete and (cluster.metadata.annotations or {}).get(
                    C.DISABLE_PROVISIONED_HEAD_RESTART_ANNOTATION_KEY) == "true" \
                    and condition_true(cluster.status.conditions, Cond.PROVISIONED):
                delete = False
            if delete:
                self.recorder.eventf(cluster, "Normal", "DeletedHeadPod", reason)
                self.expectations.expect_delete_pod(namespace, name, "__head__",
                                                    head.name)
                self._delete_pod_by_name(namespace, head.name)
        elif len(active_heads) == 0:
            self._create_head_pod(cluster)
        else:
            # too many heads: keep oldest, delete the rest
            self.recorder.eventf(cluster, "Warning", "TooManyHeadPods",
                                 "Found %d head pods; deleting extras", len(active_heads))
            for view in sorted(active_heads,
                               key=lambda p: p.creation_timestamp or "")[1:]:
                self.expectations.expect_delete_pod(namespace, name, "__head__",
                                                    view.name)
                self._delete_pod_by_name(namespace, view.name)

        # ---- worker groups
        for group in cluster.spec.worker_group_specs:
            if not self._reconcile_worker_group(cluster, group, pods):
                complete = False
        return complete

    def _cache(self):
        # the in-memory client's server doubles as the informer cache; REST
        # clients get a shim over the informer pod-view cache / typed verbs
        server = getattr(self.client, "server", None)
        if server is not None:
            return server
        return _TypedCacheShim(self.client)

    def _reconcile_worker_group(self, cluster: RayCluster, group: WorkerGroupSpec,
                                all_pods) -> bool:
        namespace = cluster.metadata.namespace or "default"
        name = cluster.metadata.name
        group_pods = [p for p in all_pods
                      if p.labels.get(C.RAY_NODE_TYPE_LABEL_KEY) == RayNodeType.WORKER
                      and p.labels.get(C.RAY_NODE_GROUP_LABEL_KEY) == group.group_name]

        # per-group suspend (controller.go:1080-1106)
        if group.suspend:
            for view in self._active(group_pods):
                self._delete_pod_by_name(namespace, view.name)
            return True

        if not self.expectations.is_satisfied(self._cache(), namespace, name, group.group_name):
            return False

        # unhealthy deletion
        for view in self._active(group_pods):
            delete, reason = should_delete_pod_view(view, RayNodeType.WORKER)
            if delete:
                self.recorder.eventf(cluster, "Normal", "DeletedWorkerPod", reason)
                self.expectations.expect_delete_pod(namespace, name, group.group_name,
                                                    view.name)
                self._delete_pod_by_name(namespace, view.name)
                group_pods = [p for p in group_pods if p.name != view.name]

        # honor ScaleStrategy.WorkersToDelete (controller.go:1135-1153)
        to_delete = set(group.scale_strategy.workers_to_delete or [])
        if to_delete:
            for view in self._active(group_pods):
                if view.name in to_delete:
                    self.expectations.expect_delete_pod(namespace, name, group.group_name,
                                                        view.name)
                    self._delete_pod_by_name(namespace, view.name)
            group_pods = [p for p in group_pods if p.name not in to_delete]

        running = self._active(group_pods)
        n_hosts = max(group.num_of_hosts, 1)
        if n_hosts > 1:
            return self._reconcile_multihost(cluster, group, running)

        desired = res.worker_group_desired_replicas(group)
        diff = desired - len(running)
        if diff > 0:
            for i in range(diff):
                self._create_worker_pod(cluster, group, replica_index=len(running) + i)
        elif diff < 0:
            # scale down without explicit WorkersToDelete: random deletion is
            # gated when autoscaling is enabled (controller.go:1222-1256)
            random_delete_enabled = os.environ.get(
                C.ENABLE_RANDOM_POD_DELETE, "").lower() == "true"
            if not podlib.is_autoscaling_enabled(cluster.spec) or random_delete_enabled:
                for view in running[:(-diff)]:
                    self.expectations.expect_delete_pod(namespace, name, group.group_name,
                                                        view.name)
                    self._delete_pod_by_name(namespace, view.name)
            # else: wait for the autoscaler to name victims via WorkersToDelete
        return True

    def _reconcile_multihost(self, cluster: RayCluster, group: WorkerGroupSpec,
                             running) -> bool:
        """Multi-host groups create/delete whole NumOfHosts replica units
        (RayMultiHostIndexing; controller.go:1287, pod.go:673-682): a replica
        that lost any host is torn down entirely and rebuilt."""
        namespace = cluster.metadata.namespace or "default"
        name = cluster.metadata.name
        n_hosts = max(group.num_of_hosts, 1)
        desired_replicas = res.worker_group_desired_replicas(group)

        by_replica = {}
        strays = []
        for view in running:
            rep = view.labels.get(C.RAY_WORKER_REPLICA_NAME_KEY)
            if rep:
                by_replica.setdefault(rep, []).append(view)
            else:
                strays.append(view)
        # pods without a replica label cannot belong to a multi-host unit
        for view in strays:
            self.expectations.expect_delete_pod(namespace, name,
                                                group.group_name, view.name)
            self._delete_pod_by_name(namespace, view.name)

        complete_replicas = []
        for rep, members in sorted(by_replica.items()):
            if len(members) == n_hosts:
                complete_replicas.append(rep)
            else:
                # partial unit (a host died or creation raced): tear it down
                for view in members:
                    self.expectations.expect_delete_pod(
                        namespace, name, group.group_name, view.name)
                    self._delete_pod_by_name(namespace, view.name)

        diff = desired_replicas - len(complete_replicas)
        if diff > 0:
            for r in range(diff):
                replica_grp = names.worker_replica_group_name(group.group_name)
                for host_idx in range(n_hosts):
                    self._create_worker_pod(
                        cluster, group,
                        replica_index=len(complete_replicas) + r,
                        host_index=host_idx, replica_grp_name=replica_grp)
        elif diff < 0:
            random_delete_enabled = os.environ.get(
                C.ENABLE_RANDOM_POD_DELETE, "").lower() == "true"
            if not podlib.is_autoscaling_enabled(cluster.spec) or random_delete_enabled:
                for rep in complete_replicas[:(-diff)]:
                    for view in by_replica[rep]:
                        self.expectations.expect_delete_pod(
                            namespace, name, group.group_name, view.name)
                        self._delete_pod_by_name(namespace, view.name)
        return True

    def _owner_crd_type(self, cluster: RayCluster) -> Optional[str]:
        return (cluster.metadata.labels or {}).get(C.RAY_ORIGINATED_FROM_CRD_LABEL_KEY)

    def _create_head_pod(self, cluster: RayCluster) -> None:
        namespace = cluster.metadata.namespace or "default"
        head_spec = cluster.spec.head_group_spec
        pod_name = names.pod_name(names.check_name(cluster.metadata.name),
                                  RayNodeType.HEAD, True)
        head_port = podlib.get_head_port(head_spec.ray_start_params)
        template = podlib.default_head_pod_template(cluster, head_spec, pod_name, head_port)
        for sidecar in self.options.head_sidecar_containers:
            template.spec.containers.append(sidecar.clone())
        fqdn = names.fqdn_service_name(cluster, namespace)
        pod = podlib.build_pod(
            template, RayNodeType.HEAD, head_spec.ray_start_params, head_port,
            podlib.is_autoscaling_enabled(cluster.spec), self._owner_crd_type(cluster),
            fqdn, self.options.default_container_envs, cluster.spec.ray_version)
        pod.metadata.owner_references = [k8s.owner_reference_for(cluster)]
        pod.metadata.labels[C.HASH_WITHOUT_REPLICAS_AND_WORKERS_TO_DELETE_KEY] = \
            self._spec_hash(cluster)
        if self.batch_scheduler is not None:
            self.batch_scheduler.add_metadata_to_pod(self.client, cluster,
                                                     "headgroup", pod)
        created = self.client.create(pod)
        self.expectations.expect_create_pod(namespace, cluster.metadata.name,
                                            "__head__", created.metadata.name)
        self.recorder.eventf(cluster, "Normal", "CreatedHeadPod",
                             "Created head Pod %s", created.metadata.name)

    def _create_worker_pod(self, cluster: RayCluster, group: WorkerGroupSpec,
                           replica_index: int = 0, host_index: int = 0,
                           replica_grp_name: Optional[str] = None) -> None:
        namespace = cluster.metadata.namespace or "default"
        pod_name = names.pod_name(
            names.check_name(f"{cluster.metadata.name}-{group.group_name}"),
            RayNodeType.WORKER, True)
        head_port = podlib.get_head_port(cluster.spec.head_group_spec.ray_start_params)
        fqdn = names.fqdn_service_name(cluster, namespace)
        replica_grp = replica_grp_name or names.worker_replica_group_name(
            group.group_name)
        template = podlib.default_worker_pod_template(
            cluster, group, pod_name, fqdn, head_port,
            replica_grp_name=replica_grp,
            replica_index=replica_index,
            num_host_index=host_index)
        for sidecar in self.options.worker_sidecar_containers:
            template.spec.containers.append(sidecar.clone())
        pod = podlib.build_pod(
            template, RayNodeType.WORKER, group.ray_start_params, head_port,
            podlib.is_autoscaling_enabled(cluster.spec), self._owner_crd_type(cluster),
            fqdn, self.options.default_container_envs, cluster.spec.ray_version)
        pod.metadata.owner_references = [k8s.owner_reference_for(cluster)]
        pod.metadata.labels[C.HASH_WITHOUT_REPLICAS_AND_WORKERS_TO_DELETE_KEY] = \
            self._spec_hash(cluster)
        if self.batch_scheduler is not None:
            self.batch_scheduler.add_metadata_to_pod(self.client, cluster,
                                                     group.group_name, pod)
        created = self.client.create(pod)
        self.expectations.expect_create_pod(namespace, cluster.metadata.name,
                                            group.group_name, created.metadata.name)
        self.recorder.eventf(cluster, "Normal", "CreatedWorkerPod",
                             "Created worker Pod %s", created.metadata.name)

    # ------------------------------------------------------------------
    # status
    # ------------------------------------------------------------------
    def _calculate_and_update_status(self, cluster: RayCluster,
                                     reconcile_err: Optional[Exception]) -> None:
        old_status = cluster.status.to_dict()
        status = cluster.status

        if reconcile_err is not None:
            status.conditions = set_condition(
                status.conditions, Cond.REPLICA_FAILURE, "True",
                "FailedCreateOrDelete", str(reconcile_err))
        else:
            status.conditions = [
                c for c in (status.conditions or []) if c.type != Cond.REPLICA_FAILURE]

        status.observed_generation = cluster.metadata.generation

        pods = self._active(self._list_cluster_pod_views(cluster))
        workers = [p for p in pods
                   if p.labels.get(C.RAY_NODE_TYPE_LABEL_KEY) == RayNodeType.WORKER]
        status.ready_worker_replicas = sum(1 for p in workers if p.ready)
        status.available_worker_replicas = sum(
            1 for p in workers if p.phase == "Running")
        status.desired_worker_replicas = res.calculate_desired_replicas(cluster)
        status.min_worker_replicas = res.calculate_min_replicas(cluster)
        status.max_worker_replicas = res.calculate_max_replicas(cluster)
        totals = res.calculate_desired_resources(cluster)
        status.desired_cpu = totals["desiredCPU"]
        status.desired_memory = totals["desiredMemory"]
        status.desired_gpu = totals["desiredGPU"]

        all_running = bool(pods) and all(
            p.phase == "Running" and p.ready for p in pods)
        if (reconcile_err is None
                and len(pods) == status.desired_worker_replicas + 1
                and all_running):
            status.state = ClusterState.READY
            status.reason = None

        # HeadPodReady condition
        head = next((p for p in pods if p.labels.get(
            C.RAY_NODE_TYPE_LABEL_KEY) == RayNodeType.HEAD), None)
        if head is None:
            status.conditions = set_condition(
                status.conditions, Cond.HEAD_POD_READY, "False",
                Reason.HEAD_POD_NOT_FOUND, "Head Pod not found")
        elif head.phase == "Running" and head.ready:
            status.conditions = set_condition(
                status.conditions, Cond.HEAD_POD_READY, "True",
                Reason.HEAD_POD_RUNNING_AND_READY, "Head Pod is running and ready")
        else:
            status.conditions = set_condition(
                status.conditions, Cond.HEAD_POD_READY, "False",
                Reason.UNKNOWN, "Head Pod is not ready")

        # RayClusterProvisioned (sticky once true, unless suspended)
        suspend_status = find_suspend_status(cluster)
        if (not condition_true(status.conditions, Cond.PROVISIONED)
                and suspend_status != Cond.SUSPENDED):
            if all_running and len(pods) == status.desired_worker_replicas + 1:
                status.conditions = set_condition(
                    status.conditions, Cond.PROVISIONED, "True",
                    Reason.ALL_POD_RUNNING_AND_READY_FIRST_TIME,
                    "All Ray Pods are ready for the first time")
            else:
                status.conditions = set_condition(
                    status.conditions, Cond.PROVISIONED, "False",
                    Reason.RAY_CLUSTER_PODS_PROVISIONING,
                    "RayCluster Pods are being provisioned for first time")

        # suspend state machine (calculateStatus switch; controller.go:2066-2115)
        if suspend_status == Cond.SUSPENDING:
            if len(pods) == 0:
                status.conditions = set_condition(
                    status.conditions, Cond.PROVISIONED, "False",
                    Reason.RAY_CLUSTER_PODS_PROVISIONING, "RayCluster has been suspended")
                status.conditions = set_condition(
                    status.conditions, Cond.SUSPENDING, "False", Cond.SUSPENDING)
                status.conditions = set_condition(
                    status.conditions, Cond.SUSPENDED, "True", Cond.SUSPENDED)
        elif suspend_status == Cond.SUSPENDED:
            if not cluster.spec.suspend:
                status.conditions = set_condition(
                    status.conditions, Cond.SUSPENDED, "False", Cond.SUSPENDED)
        else:
            status.conditions = set_condition(
                status.conditions, Cond.SUSPENDED, "False", Cond.SUSPENDED)
            status.conditions = set_condition(
                status.conditions, Cond.SUSPENDING,
                "True" if cluster.spec.suspend else "False", Cond.SUSPENDING)

        if cluster.spec.suspend and len(pods) == 0:
            status.state = ClusterState.SUSPENDED

        self._update_endpoints(cluster)
        self._update_head_info(cluster, head)

        status.last_update_time = now_iso()
        new_status = status.to_dict()
        old_state = old_status.get("state")
        if old_state != status.state:
            stt = dict(status.state_transition_times or {})
            if status.state:
                stt[status.state] = now_iso()
            status.state_transition_times = stt
            if self.metrics is not None and status.state == ClusterState.READY:
                self.metrics.observe_cluster_ready(cluster)
            new_status = status.to_dict()
        # single status write per reconcile, only on change
        if {k: v for k, v in new_status.items() if k != "lastUpdateTime"} != \
           {k: v for k, v in old_status.items() if k != "lastUpdateTime"}:
            self.client.update_status(cluster)

    def _update_endpoints(self, cluster: RayCluster) -> None:
        svc_name = names.head_service_name(C.KIND_RAYCLUSTER, cluster.spec,
                                           cluster.metadata.name)
        svc = self.client.try_get(k8s.Service, cluster.metadata.namespace or "default",
                                  svc_name)
        if svc is None:
            return
        endpoints = {}
        for port in svc.spec.ports or []:
            if port.name:
                endpoints[port.name] = str(port.node_port or port.port)
        cluster.status.endpoints = endpoints

    def _update_head_info(self, cluster: RayCluster, head) -> None:
        info = cluster.status.head
        if head is not None:
            info.pod_name = head.name
            info.pod_ip = head.pod_ip or None
        else:
            info.pod_name = None
            info.pod_ip = None
        svc_name = names.head_service_name(C.KIND_RAYCLUSTER, cluster.spec,
                                           cluster.metadata.name)
        svc = self.client.try_get(k8s.Service, cluster.metadata.namespace or "default",
                                  svc_name)
        if svc is not None:
            info.service_name = svc.metadata.name
            info.service_ip = svc.spec.cluster_ip if svc.spec.cluster_ip != "None" else info.pod_ip

    def _update_status_if_changed(self, cluster: RayCluster) -> None:
        try:
            self.client.update_status(cluster)
        except NotFoundError:
            pass
