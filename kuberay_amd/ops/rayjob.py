"""RayJob reconciler — explicit state machine over JobDeploymentStatus.

Reference: ray-operator/controllers/ray/rayjob_controller.go —
state machine :165-451, cluster get-or-create :947-1041, K8s submitter
:560-585, submitter/app status merge :1062-1232 + :296-365, suspend/retry
:366-410, deletion policies :1413-1560, backoffLimit :518-558, finalizer
StopJob :112-139.
"""
from __future__ import annotations

import calendar
import logging
import os
import time
from typing import Callable, Optional

from ..common import job as joblib
from ..kube import objects as k8s
from ..kube.client import KubeClient
from ..kube.controller import Reconciler, Request, Result
from ..kube.events import EventRecorder, NullRecorder
from ..kube.store import AlreadyExistsError, NotFoundError, now_iso
from ..models import (
    DeletionPolicyType,
    JobDeploymentStatus as JDS,
    JobFailedReason,
    JobStatus as JS,
    JobSubmissionMode as Mode,
    RayCluster,
    RayJob,
)
from ..models.raycluster import ClusterState
from ..utils import constants as C
from ..utils import names
from ..utils.dashboard_client import DashboardClientError
from ..utils.validation import (
    validate_rayjob_metadata,
    validate_rayjob_spec,
    validate_rayjob_status,
)

logger = logging.getLogger("kuberay.rayjob")

REQUEUE_SECONDS = 3


def _parse_ts(ts) -> Optional[float]:
    """ISO-8601 (operator-written) or numeric epoch (Ray dashboard job info
    reports start/end times as epoch milliseconds)."""
    if ts in (None, "", 0):
        return None
    if isinstance(ts, (int, float)):
        value = float(ts)
        return value / 1000.0 if value > 1e12 else value
    try:
        return calendar.timegm(time.strptime(ts, "%Y-%m-%dT%H:%M:%SZ"))
    except ValueError:
        try:
            value = float(ts)
            return value / 1000.0 if value > 1e12 else value
        except ValueError:
            return None


class RayJobReconciler(Reconciler):
    def __init__(self, client: KubeClient, recorder: Optional[EventRecorder] = None,
                 dashboard_factory: Optional[Callable] = None, metrics=None,
                 batch_scheduler=None):
        self.client = client
        self.recorder = recorder or NullRecorder()
        self.dashboard_factory = dashboard_factory or self._default_dashboard
        self.metrics = metrics
        self.batch_scheduler = batch_scheduler
        self.requeue_seconds = REQUEUE_SECONDS

    @staticmethod
    def _default_dashboard(url: str):
        from ..utils.dashboard_client import RayDashboardClient
        return RayDashboardClient(url)

    # ------------------------------------------------------------------
    def reconcile(self, request: Request) -> Result:
        namespace, name = request
        rayjob = self.client.try_get(RayJob, namespace, name)
        if rayjob is None:
            return Result()
        if rayjob.spec.managed_by not in (None, "ray.io/kuberay-operator"):
            return Result()

        if rayjob.metadata.deletion_timestamp:
            return self._handle_deletion(rayjob)

        status = rayjob.status.job_deployment_status
        handler = {
            JDS.NEW: self._handle_new,
            JDS.INITIALIZING: self._handle_initializing,
            JDS.WAITING: self._handle_waiting,
            JDS.RUNNING: self._handle_running,
            JDS.SUSPENDING: self._handle_suspending,
            JDS.SUSPENDED: self._handle_suspended,
            JDS.RETRYING: self._handle_retrying,
            JDS.COMPLETE: self._handle_terminal,
            JDS.FAILED: self._handle_terminal,
            JDS.VALIDATION_FAILED: lambda rj: Result(),
        }.get(status)
        if handler is None:
            logger.error("unknown JobDeploymentStatus %r for %s/%s", status, namespace, name)
            return Result()
        return handler(rayjob)

    # ------------------------------------------------------------------
    def _set_status(self, rayjob: RayJob, deployment_status: str,
                    reason: Optional[str] = None, message: Optional[str] = None) -> None:
        rayjob.status.job_deployment_status = deployment_status
        if reason is not None:
            rayjob.status.reason = reason
        if message is not None:
            rayjob.status.message = message
        rayjob.status.observed_generation = rayjob.metadata.generation
        self.client.update_status(rayjob)

    # ------------------------------------------------------------------
    def _handle_new(self, rayjob: RayJob) -> Result:
        errs = (validate_rayjob_metadata(rayjob.metadata)
                + validate_rayjob_spec(rayjob)
                + validate_rayjob_status(rayjob))
        if errs:
            self.recorder.eventf(rayjob, "Warning", "InvalidRayJobSpec", "; ".join(errs))
            self._set_status(rayjob, JDS.VALIDATION_FAILED,
                             JobFailedReason.VALIDATION_FAILED, "; ".join(errs))
            return Result()
        # finalizer for StopJob-on-delete
        finalizers = rayjob.metadata.finalizers or []
        if C.RAYJOB_STOP_JOB_FINALIZER not in finalizers:
            rayjob.metadata.finalizers = finalizers + [C.RAYJOB_STOP_JOB_FINALIZER]
            rayjob = self.client.update(rayjob)
        # init status (rayjob_controller.go:887 initRayJobStatusIfNeed)
        if not rayjob.status.job_id:
            rayjob.status.job_id = (rayjob.spec.job_id
                                    or names.ray_job_id(rayjob.metadata.name))
        if not rayjob.status.ray_cluster_name:
            if rayjob.spec.cluster_selector:
                rayjob.status.ray_cluster_name = \
                    self._resolve_selected_cluster_name(rayjob) or ""
            else:
                rayjob.status.ray_cluster_name = names.ray_cluster_name_for(
                    rayjob.metadata.name)
        rayjob.status.start_time = now_iso()
        if rayjob.spec.suspend:
            self._set_status(rayjob, JDS.SUSPENDED)
        else:
            self._set_status(rayjob, JDS.INITIALIZING)
        return Result(requeue_after=0.0)

    # ------------------------------------------------------------------
    def _handle_initializing(self, rayjob: RayJob) -> Result:
        if rayjob.spec.suspend:
            self._set_status(rayjob, JDS.SUSPENDING)
            return Result(requeue_after=0.0)
        if self._pre_running_deadline_exceeded(rayjob):
            self.recorder.eventf(rayjob, "Warning", "PreRunningDeadlineExceeded",
                                 "RayJob did not reach Running before preRunningDeadlineSeconds")
            rayjob.status.end_time = now_iso()
            self._set_status(rayjob, JDS.FAILED,
                             JobFailedReason.PRE_RUNNING_DEADLINE_EXCEEDED,
                             "The RayJob did not reach Running state before its preRunningDeadlineSeconds")
            return Result()

        cluster = self._get_or_create_cluster(rayjob)
        if cluster is None:
            return Result(requeue_after=self.requeue_seconds)
        rayjob.status.ray_cluster_status = cluster.status

        if cluster.status.state != ClusterState.READY:
            self._set_status(rayjob, rayjob.status.job_deployment_status)
            return Result(requeue_after=self.requeue_seconds)

        # cluster ready → resolve dashboard URL
        head_svc = names.head_service_name(C.KIND_RAYCLUSTER, cluster.spec,
                                           cluster.metadata.name)
        namespace = rayjob.metadata.namespace or "default"
        rayjob.status.dashboard_url = (
            f"{head_svc}.{namespace}.svc.{names.cluster_domain_name()}:"
            f"{C.DEFAULT_DASHBOARD_PORT}")

        mode = rayjob.spec.submission_mode
        if mode == Mode.K8S_JOB:
            self._create_submitter_job_if_needed(rayjob)
        elif mode == Mode.HTTP:
            try:
                dashboard = self.dashboard_factory(rayjob.status.dashboard_url)
                if dashboard.get_job_info(rayjob.status.job_id) is None:
                    dashboard.submit_job(self._build_http_submission(rayjob))
            except DashboardClientError as e:
                self.recorder.eventf(rayjob, "Warning", "FailedToSubmitJob", str(e))
                return Result(requeue_after=self.requeue_seconds)
        elif mode == Mode.SIDECAR:
            pass  # sidecar container injected into the head pod via cluster spec
        if mode == Mode.INTERACTIVE:
            # user submits out-of-band; wait for the submission-id annotation
            # (rayjob_controller.go Waiting state)
            self._set_status(rayjob, JDS.WAITING)
            return Result(requeue_after=self.requeue_seconds)

        self._set_status(rayjob, JDS.RUNNING)
        return Result(requeue_after=self.requeue_seconds)

    def _build_http_submission(self, rayjob: RayJob) -> dict:
        import yaml
        submission = {
            "entrypoint": rayjob.spec.entrypoint,
            "submission_id": rayjob.status.job_id,
        }
        if rayjob.spec.runtime_env_yaml:
            submission["runtime_env"] = yaml.safe_load(rayjob.spec.runtime_env_yaml)
        if rayjob.spec.metadata:
            submission["metadata"] = rayjob.spec.metadata
        if rayjob.spec.entrypoint_num_cpus:
            submission["entrypoint_num_cpus"] = rayjob.spec.entrypoint_num_cpus
        if rayjob.spec.entrypoint_num_gpus:
            submission["entrypoint_num_gpus"] = rayjob.spec.entrypoint_num_gpus
        return submission

    def _handle_waiting(self, rayjob: RayJob) -> Result:
        # InteractiveMode: wait for user-provided submission id annotation
        job_id = (rayjob.metadata.annotations or {}).get("ray.io/ray-job-submission-id")
        if job_id:
            rayjob.status.job_id = job_id
            self._set_status(rayjob, JDS.RUNNING)
        return Result(requeue_after=self.requeue_seconds)

    # ------------------------------------------------------------------
    def _handle_running(self, rayjob: RayJob) -> Result:
        if rayjob.spec.suspend:
            self._set_status(rayjob, JDS.SUSPENDING)
            return Result(requeue_after=0.0)
        if self._active_deadline_exceeded(rayjob):
            rayjob.status.end_time = now_iso()
            self._set_status(rayjob, JDS.FAILED, JobFailedReason.DEADLINE_EXCEEDED,
                             "The RayJob exceeded its activeDeadlineSeconds")
            return Result()

        # 1. poll the app status via the dashboard
        job_info = None
        try:
            dashboard = self.dashboard_factory(rayjob.status.dashboard_url)
            job_info = dashboard.get_job_info(rayjob.status.job_id)
            rayjob.status.job_status_check_failure_start_time = None
        except DashboardClientError:
            job_info = None
            if rayjob.status.job_status_check_failure_start_time is None:
                rayjob.status.job_status_check_failure_start_time = now_iso()
            elif self._older_than(rayjob.status.job_status_check_failure_start_time,
                                  self._status_check_timeout()):
                rayjob.status.end_time = now_iso()
                self._set_status(rayjob, JDS.FAILED,
                                 JobFailedReason.JOB_STATUS_CHECK_TIMEOUT_EXCEEDED,
                                 "Dashboard job status checks failed for too long")
                return Result()

        if job_info is not None:
            rayjob.status.job_status = job_info.get("status", rayjob.status.job_status)
            info = rayjob.status.ray_job_info
            if job_info.get("start_time"):
                info.start_time = str(job_info["start_time"])
            if job_info.get("end_time"):
                info.end_time = str(job_info["end_time"])
            if job_info.get("message") is not None:
                rayjob.status.message = job_info.get("message")

        # 2. watch the submitter (K8sJobMode)
        submitter_finished, submitter_failed = self._check_submitter(rayjob)

        job_terminal = JS.is_terminal(rayjob.status.job_status)
        mode = rayjob.spec.submission_mode

        if mode == Mode.K8S_JOB and submitter_failed and not job_terminal:
            rayjob.status.end_time = now_iso()
            rayjob.status.failed = (rayjob.status.failed or 0) + 1
            return self._fail_or_retry(rayjob, JobFailedReason.SUBMISSION_FAILED,
                                       "Submitter K8s Job failed")

        if job_terminal:
            # For K8sJobMode wait for the submitter to finish too, with a grace
            # period escape hatch (rayjob_controller.go:334-356).
            wait_submitter = mode == Mode.K8S_JOB and not submitter_finished
            if wait_submitter and not self._transition_grace_exceeded(rayjob):
                self._set_status(rayjob, JDS.RUNNING)
                return Result(requeue_after=self.requeue_seconds)
            rayjob.status.end_time = now_iso()
            if rayjob.status.job_status == JS.SUCCEEDED:
                rayjob.status.succeeded = (rayjob.status.succeeded or 0) + 1
                self._set_status(rayjob, JDS.COMPLETE)
                if self.metrics is not None:
                    self.metrics.observe_job_finished(rayjob, succeeded=True)
                return Result(requeue_after=0.0)
            rayjob.status.failed = (rayjob.status.failed or 0) + 1
            if self.metrics is not None:
                self.metrics.observe_job_finished(rayjob, succeeded=False)
            return self._fail_or_retry(
                rayjob, JobFailedReason.APP_FAILED,
                f"Ray job finished with status {rayjob.status.job_status}")

        self._set_status(rayjob, JDS.RUNNING)
        return Result(requeue_after=self.requeue_seconds)

    def _fail_or_retry(self, rayjob: RayJob, reason: str, message: str) -> Result:
        """Decide Retrying vs Failed BEFORE writing status, so the job never
        transiently reads as Failed while retries remain
        (rayjob_controller.go:518-558)."""
        backoff = rayjob.spec.backoff_limit or 0
        attempts = (rayjob.status.succeeded or 0) + (rayjob.status.failed or 0)
        if backoff > 0 and attempts <= backoff and \
                reason != JobFailedReason.DEADLINE_EXCEEDED:
            self._set_status(rayjob, JDS.RETRYING, reason, message)
        else:
            self._set_status(rayjob, JDS.FAILED, reason, message)
        return Result(requeue_after=0.0)

    # ------------------------------------------------------------------
    def _handle_suspending(self, rayjob: RayJob) -> Result:
        self._delete_cluster_resources(rayjob)
        self._delete_submitter_job(rayjob)
        rayjob.status.job_status = JS.NEW
        rayjob.status.job_deployment_status = JDS.SUSPENDED
        rayjob.status.dashboard_url = None
        rayjob.status.ray_cluster_status = type(rayjob.status.ray_cluster_status)()
        self.client.update_status(rayjob)
        return Result(requeue_after=self.requeue_seconds)

    def _handle_suspended(self, rayjob: RayJob) -> Result:
        if not rayjob.spec.suspend:
            self._set_status(rayjob, JDS.INITIALIZING)
            return Result(requeue_after=0.0)
        return Result()

    def _handle_retrying(self, rayjob: RayJob) -> Result:
        self._delete_cluster_resources(rayjob)
        self._delete_submitter_job(rayjob)
        rayjob.status.job_status = JS.NEW
        rayjob.status.dashboard_url = None
        rayjob.status.job_id = names.ray_job_id(rayjob.metadata.name)
        if not rayjob.spec.cluster_selector:
            rayjob.status.ray_cluster_name = names.ray_cluster_name_for(rayjob.metadata.name)
        self._set_status(rayjob, JDS.INITIALIZING)
        return Result(requeue_after=0.0)

    # ------------------------------------------------------------------
    def _handle_terminal(self, rayjob: RayJob) -> Result:
        """Deletion policies (rayjob_controller.go:1413-1560)."""
        if self.batch_scheduler is not None:
            # gang-scheduling artifacts (PodGroup) are no longer needed once
            # the job is terminal (BatchScheduler.CleanupOnCompletion)
            cluster = self._owned_cluster(rayjob)
            if cluster is not None:
                self.batch_scheduler.cleanup_on_completion(self.client, cluster)
        ds = rayjob.spec.deletion_strategy
        succeeded = rayjob.status.job_deployment_status == JDS.COMPLETE

        if ds is not None and ds.deletion_rules:
            requeue: Optional[float] = None
            for rule in ds.deletion_rules:
                cond = rule.condition
                if cond.job_status and cond.job_status != rayjob.status.job_status:
                    continue
                if (cond.job_deployment_status
                        and cond.job_deployment_status != rayjob.status.job_deployment_status):
                    continue
                remaining = self._ttl_remaining(rayjob, cond.ttl_seconds)
                if remaining > 0:
                    requeue = min(requeue, remaining) if requeue else remaining
                    continue
                if self._apply_deletion_policy(rayjob, rule.policy):
                    return Result()
            return Result(requeue_after=requeue) if requeue else Result()

        if ds is not None and (ds.on_success or ds.on_failure):
            block = ds.on_success if succeeded else ds.on_failure
            if block and block.policy:
                self._apply_deletion_policy(rayjob, block.policy)
            return Result()

        # legacy shutdownAfterJobFinishes + TTL
        if rayjob.spec.shutdown_after_job_finishes:
            remaining = self._ttl_remaining(rayjob, rayjob.spec.ttl_seconds_after_finished)
            if remaining > 0:
                return Result(requeue_after=remaining)
            if os.environ.get(C.DELETE_RAYJOB_CR_AFTER_JOB_FINISHES, "").lower() == "true":
                self._apply_deletion_policy(rayjob, DeletionPolicyType.DELETE_SELF)
            else:
                self._apply_deletion_policy(rayjob, DeletionPolicyType.DELETE_CLUSTER)
                self._delete_submitter_job(rayjob)
        return Result()

    def _apply_deletion_policy(self, rayjob: RayJob, policy: str) -> bool:
        namespace = rayjob.metadata.namespace or "default"
        if policy == DeletionPolicyType.DELETE_NONE:
            return False
        if policy == DeletionPolicyType.DELETE_SELF:
            self._remove_finalizer(rayjob)
            try:
                self.client.delete(RayJob, namespace, rayjob.metadata.name)
            except NotFoundError:
                pass
            return True
        if policy == DeletionPolicyType.DELETE_CLUSTER:
            self._delete_cluster_resources(rayjob)
            return False
        if policy == DeletionPolicyType.DELETE_WORKERS:
            cluster = self._owned_cluster(rayjob)
            if cluster is not None:
                changed = False
                for group in cluster.spec.worker_group_specs:
                    if not group.suspend:
                        group.suspend = True
                        changed = True
                if changed:
                    self.client.update(cluster)
            return False
        return False

    # ------------------------------------------------------------------
    def _handle_deletion(self, rayjob: RayJob) -> Result:
        finalizers = rayjob.metadata.finalizers or []
        if C.RAYJOB_STOP_JOB_FINALIZER not in finalizers:
            return Result()
        # StopJob on delete (rayjob_controller.go:112-139)
        if rayjob.status.job_id and not JS.is_terminal(rayjob.status.job_status) \
                and rayjob.status.dashboard_url:
            try:
                self.dashboard_factory(rayjob.status.dashboard_url).stop_job(
                    rayjob.status.job_id)
            except DashboardClientError:
                pass
        self._remove_finalizer(rayjob)
        return Result()

    def _remove_finalizer(self, rayjob: RayJob) -> None:
        finalizers = rayjob.metadata.finalizers or []
        if C.RAYJOB_STOP_JOB_FINALIZER in finalizers:
            fresh = self.client.try_get(RayJob, rayjob.metadata.namespace or "default",
                                        rayjob.metadata.name)
            if fresh is not None:
                fresh.metadata.finalizers = [
                    f for f in (fresh.metadata.finalizers or [])
                    if f != C.RAYJOB_STOP_JOB_FINALIZER]
                self.client.update(fresh)
                rayjob.metadata.finalizers = fresh.metadata.finalizers

    # ------------------------------------------------------------------
    # cluster & submitter management
    # ------------------------------------------------------------------
    def _resolve_selected_cluster_name(self, rayjob: RayJob) -> Optional[str]:
        """Name of the cluster a clusterSelector points at.

        The ``ray.io/cluster`` selector key IS the cluster name — resolve it
        directly and independently of whether that RayCluster exists yet
        (reference getRayClusterNameFromSelector semantics), falling back to
        a live selector lookup for arbitrary label selectors.
        """
        selector = rayjob.spec.cluster_selector or {}
        direct = selector.get(C.RAY_CLUSTER_LABEL_KEY)
        if direct:
            return direct
        matches = self.client.list(RayCluster,
                                   rayjob.metadata.namespace or "default",
                                   selector)
        return matches[0].metadata.name if matches else None

    def _owned_cluster(self, rayjob: RayJob) -> Optional[RayCluster]:
        if not rayjob.status.ray_cluster_name and rayjob.spec.cluster_selector:
            # Re-resolve each reconcile: the selected cluster may have been
            # created after the RayJob (round-1 bug left the job stuck in
            # Initializing forever when the selector matched nothing at
            # creation time).
            name = self._resolve_selected_cluster_name(rayjob)
            if name:
                rayjob.status.ray_cluster_name = name
        if not rayjob.status.ray_cluster_name:
            return None
        return self.client.try_get(RayCluster, rayjob.metadata.namespace or "default",
                                   rayjob.status.ray_cluster_name)

    def _get_or_create_cluster(self, rayjob: RayJob) -> Optional[RayCluster]:
        """rayjob_controller.go:947 getOrCreateRayClusterInstance."""
        namespace = rayjob.metadata.namespace or "default"
        cluster = self._owned_cluster(rayjob)
        if cluster is not None:
            return cluster
        if rayjob.spec.cluster_selector:
            return None  # selected cluster vanished (or not created yet); wait
        spec = rayjob.spec.ray_cluster_spec.clone()
        if rayjob.spec.submission_mode == Mode.SIDECAR:
            self._inject_sidecar_submitter(rayjob, spec)
        cluster = RayCluster(
            metadata=k8s.ObjectMeta(
                name=rayjob.status.ray_cluster_name,
                namespace=namespace,
                labels={
                    C.RAY_ORIGINATED_FROM_CR_NAME_LABEL_KEY:
                        names.check_label(rayjob.metadata.name),
                    C.RAY_ORIGINATED_FROM_CRD_LABEL_KEY: C.KIND_RAYJOB,
                },
                annotations=dict(rayjob.metadata.annotations or {}) or None,
                owner_references=[k8s.owner_reference_for(rayjob)],
            ),
            spec=spec,
        )
        if rayjob.spec.submission_mode == Mode.SIDECAR:
            cluster.metadata.ensure_annotations()[
                C.DISABLE_PROVISIONED_HEAD_RESTART_ANNOTATION_KEY] = "true"
        try:
            self.client.create(cluster)
            self.recorder.eventf(rayjob, "Normal", "CreatedRayCluster",
                                 "Created RayCluster %s", cluster.metadata.name)
        except AlreadyExistsError:
            pass
        return self._owned_cluster(rayjob)

    def _inject_sidecar_submitter(self, rayjob: RayJob, cluster_spec) -> None:
        cmd = joblib.build_job_submit_command(rayjob, Mode.SIDECAR)
        container = joblib.default_submitter_container(rayjob.spec.ray_cluster_spec)
        container.command = ["/bin/bash", "-c", "--"]
        container.args = [" ".join(cmd)]
        container.set_env_if_absent(C.RAY_JOB_SUBMISSION_ID, rayjob.status.job_id or "")
        cluster_spec.head_group_spec.template.spec.containers.append(container)

    def _submitter_job_name(self, rayjob: RayJob) -> str:
        return names.submitter_job_name(rayjob.metadata.name)

    def _create_submitter_job_if_needed(self, rayjob: RayJob) -> None:
        namespace = rayjob.metadata.namespace or "default"
        existing = self.client.try_get(k8s.Job, namespace, self._submitter_job_name(rayjob))
        if existing is not None:
            return
        job = joblib.build_submitter_job(rayjob)
        job.metadata.owner_references = [k8s.owner_reference_for(rayjob)]
        try:
            self.client.create(job)
            self.recorder.eventf(rayjob, "Normal", "CreatedSubmitterJob",
                                 "Created submitter K8s Job %s", job.metadata.name)
        except AlreadyExistsError:
            pass

    def _check_submitter(self, rayjob: RayJob) -> (bool, bool):
        """Returns (finished, failed) for the submitter K8s Job."""
        if rayjob.spec.submission_mode != Mode.K8S_JOB:
            return True, False
        namespace = rayjob.metadata.namespace or "default"
        job = self.client.try_get(k8s.Job, namespace, self._submitter_job_name(rayjob))
        if job is None or job.status is None:
            return False, False
        for cond in job.status.conditions or []:
            if cond.status != "True":
                continue
            if cond.type == "Complete":
                return True, False
            if cond.type in ("Failed", "FailureTarget"):
                return True, True
        return False, False

    def _delete_cluster_resources(self, rayjob: RayJob) -> None:
        cluster = self._owned_cluster(rayjob)
        if cluster is not None and not rayjob.spec.cluster_selector:
            try:
                self.client.delete(cluster)
                self.recorder.eventf(rayjob, "Normal", "DeletedRayCluster",
                                     "Deleted RayCluster %s", cluster.metadata.name)
            except NotFoundError:
                pass

    def _delete_submitter_job(self, rayjob: RayJob) -> None:
        namespace = rayjob.metadata.namespace or "default"
        try:
            self.client.delete(k8s.Job, namespace, self._submitter_job_name(rayjob))
        except NotFoundError:
            pass

    # ------------------------------------------------------------------
    # deadlines
    # ------------------------------------------------------------------
    @staticmethod
    def _older_than(ts: Optional[str], seconds: float) -> bool:
        t = _parse_ts(ts)
        return t is not None and (time.time() - t) > seconds

    def _active_deadline_exceeded(self, rayjob: RayJob) -> bool:
        if not rayjob.spec.active_deadline_seconds:
            return False
        return self._older_than(rayjob.status.start_time,
                                rayjob.spec.active_deadline_seconds)

    def _pre_running_deadline_exceeded(self, rayjob: RayJob) -> bool:
        if not rayjob.spec.pre_running_deadline_seconds:
            return False
        return self._older_than(rayjob.status.start_time,
                                rayjob.spec.pre_running_deadline_seconds)

    def _transition_grace_exceeded(self, rayjob: RayJob) -> bool:
        grace = int(os.environ.get(
            C.RAYJOB_DEPLOYMENT_STATUS_TRANSITION_GRACE_PERIOD_SECONDS,
            C.DEFAULT_RAYJOB_DEPLOYMENT_STATUS_TRANSITION_GRACE_PERIOD_SECONDS))
        end = rayjob.status.ray_job_info.end_time or rayjob.status.start_time
        return self._older_than(end, grace)

    @staticmethod
    def _status_check_timeout() -> int:
        return int(os.environ.get(C.RAYJOB_STATUS_CHECK_TIMEOUT_SECONDS,
                                  C.DEFAULT_RAYJOB_STATUS_CHECK_TIMEOUT_SECONDS))

    def _ttl_remaining(self, rayjob: RayJob, ttl_seconds: int) -> float:
        if not ttl_seconds:
            return 0.0
        end = _parse_ts(rayjob.status.end_time)
        if end is None:
            return float(ttl_seconds)
        return max(0.0, ttl_seconds - (time.time() - end))
