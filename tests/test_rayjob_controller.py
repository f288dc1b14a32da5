"""RayJob state-machine tests (reference analog: rayjob_controller_test.go +
rayjob_controller_suspended_test.go)."""
import time

import pytest

from kuberay_amd.models import RayCluster, RayJob
from kuberay_amd.testing import simple_raycluster
from kuberay_amd.utils import constants as C


def make_rayjob(name="job1", **spec_overrides):
    spec = {
        "entrypoint": "python script.py",
        "rayClusterSpec": simple_raycluster("x", workers=1).spec.to_dict(),
    }
    spec.update(spec_overrides)
    return RayJob.from_dict({
        "apiVersion": "ray.io/v1", "kind": "RayJob",
        "metadata": {"name": name, "namespace": "default"},
        "spec": spec,
    })


def job_of(cp, name="job1"):
    return cp.client.get(RayJob, "default", name)


def wait_deployment_status(cp, name, status, timeout=20):
    return cp.wait_for(
        lambda: job_of(cp, name).status.job_deployment_status == status, timeout)


class TestHappyPath:
    def test_k8s_job_mode_completes(self, control_plane):
        control_plane.client.create(make_rayjob())
        assert wait_deployment_status(control_plane, "job1", "Complete")
        job = job_of(control_plane)
        assert job.status.job_status == "SUCCEEDED"
        assert job.status.job_id.startswith("job1-")
        assert job.status.start_time and job.status.end_time
        assert job.status.succeeded == 1
        # submitter K8s Job was created
        assert control_plane.server.try_get("Job", "default", "job1") is not None

    def test_cluster_created_with_owner_labels(self, control_plane):
        control_plane.client.create(make_rayjob())
        assert control_plane.wait_for(
            lambda: job_of(control_plane).status.ray_cluster_name)
        name = job_of(control_plane).status.ray_cluster_name
        assert control_plane.wait_for(
            lambda: control_plane.server.try_get("RayCluster", "default", name))
        rc = control_plane.server.get("RayCluster", "default", name)
        labels = rc["metadata"]["labels"]
        assert labels[C.RAY_ORIGINATED_FROM_CR_NAME_LABEL_KEY] == "job1"
        assert labels[C.RAY_ORIGINATED_FROM_CRD_LABEL_KEY] == "RayJob"

    def test_shutdown_after_job_finishes(self, control_plane):
        control_plane.client.create(make_rayjob(shutdownAfterJobFinishes=True))
        assert wait_deployment_status(control_plane, "job1", "Complete")
        assert control_plane.wait_for(
            lambda: control_plane.server.count("RayCluster") == 0)

    def test_ttl_seconds_after_finished(self, control_plane):
        import time as _t
        control_plane.client.create(make_rayjob(
            shutdownAfterJobFinishes=True, ttlSecondsAfterFinished=5))
        assert wait_deployment_status(control_plane, "job1", "Complete")
        t_complete = _t.monotonic()
        # cluster still exists within the TTL window (checked well inside it)
        if _t.monotonic() - t_complete < 2.0:
            assert control_plane.server.count("RayCluster") == 1
        assert control_plane.wait_for(
            lambda: control_plane.server.count("RayCluster") == 0, timeout=15)

    def test_http_mode(self, control_plane):
        control_plane.client.create(make_rayjob(submissionMode="HTTPMode"))
        assert wait_deployment_status(control_plane, "job1", "Complete")
        # no submitter K8s Job in HTTP mode
        assert control_plane.server.try_get("Job", "default", "job1") is None

    def test_sidecar_mode_injects_submitter_container(self, control_plane):
        control_plane.client.create(make_rayjob(submissionMode="SidecarMode"))
        assert control_plane.wait_for(
            lambda: control_plane.server.count("RayCluster") == 1)
        rc = control_plane.server.list("RayCluster")[0]
        containers = rc["spec"]["headGroupSpec"]["template"]["spec"]["containers"]
        assert any(c["name"] == "ray-job-submitter" for c in containers)


class TestFailure:
    def test_app_failure_marks_failed(self, control_plane):
        control_plane.dashboard.get_job_info_mock = lambda job_id: {
            "submission_id": job_id, "status": "FAILED", "message": "boom"}
        control_plane.client.create(make_rayjob())
        assert wait_deployment_status(control_plane, "job1", "Failed")
        job = job_of(control_plane)
        assert job.status.reason == "AppFailed"
        assert job.status.failed == 1

    def test_backoff_limit_retries(self, control_plane):
        calls = {"n": 0}
        def mock(job_id):
            calls["n"] += 1
            return {"submission_id": job_id, "status": "FAILED"}
        control_plane.dashboard.get_job_info_mock = mock
        control_plane.client.create(make_rayjob(backoffLimit=1))
        assert wait_deployment_status(control_plane, "job1", "Failed", timeout=30)
        job = job_of(control_plane)
        assert job.status.failed == 2  # initial attempt + 1 retry

    def test_validation_failure(self, control_plane):
        bad = make_rayjob()
        bad.spec.entrypoint = None
        control_plane.client.create(bad)
        assert wait_deployment_status(control_plane, "job1", "ValidationFailed")

    def test_active_deadline(self, control_plane):
        control_plane.dashboard.get_job_info_mock = lambda job_id: {
            "submission_id": job_id, "status": "RUNNING"}
        control_plane.client.create(make_rayjob(activeDeadlineSeconds=1))
        assert wait_deployment_status(control_plane, "job1", "Failed", timeout=30)
        assert job_of(control_plane).status.reason == "DeadlineExceeded"


class TestSuspend:
    def test_created_suspended(self, control_plane):
        control_plane.client.create(make_rayjob(
            suspend=True, shutdownAfterJobFinishes=True))
        assert wait_deployment_status(control_plane, "job1", "Suspended")
        assert control_plane.server.count("RayCluster") == 0

    def test_suspend_running_job_tears_down_cluster(self, control_plane):
        control_plane.dashboard.get_job_info_mock = lambda job_id: {
            "submission_id": job_id, "status": "RUNNING"}
        control_plane.client.create(make_rayjob())
        assert wait_deployment_status(control_plane, "job1", "Running")
        job = job_of(control_plane)
        job.spec.suspend = True
        control_plane.client.update(job)
        assert wait_deployment_status(control_plane, "job1", "Suspended")
        assert control_plane.wait_for(
            lambda: control_plane.server.count("RayCluster") == 0)

    def test_resume_creates_fresh_cluster(self, control_plane):
        control_plane.client.create(make_rayjob(
            suspend=True, shutdownAfterJobFinishes=True))
        assert wait_deployment_status(control_plane, "job1", "Suspended")
        job = job_of(control_plane)
        job.spec.suspend = False
        control_plane.client.update(job)
        assert wait_deployment_status(control_plane, "job1", "Complete", timeout=30)


class TestDeletionPolicies:
    def test_deletion_rules_delete_self(self, control_plane):
        control_plane.client.create(make_rayjob(deletionStrategy={
            "deletionRules": [{"policy": "DeleteSelf",
                               "condition": {"jobStatus": "SUCCEEDED"}}]}))
        assert control_plane.wait_for(
            lambda: control_plane.client.try_get(RayJob, "default", "job1") is None,
            timeout=30)

    def test_deletion_rules_delete_workers(self, control_plane):
        control_plane.client.create(make_rayjob(deletionStrategy={
            "deletionRules": [{"policy": "DeleteWorkers",
                               "condition": {"jobStatus": "SUCCEEDED"}}]}))
        assert wait_deployment_status(control_plane, "job1", "Complete")
        def workers_suspended():
            clusters = control_plane.server.list("RayCluster")
            return clusters and all(
                g.get("suspend") for c in clusters
                for g in c["spec"].get("workerGroupSpecs", []))
        assert control_plane.wait_for(workers_suspended)

    def test_legacy_on_success_delete_cluster(self, control_plane):
        control_plane.client.create(make_rayjob(deletionStrategy={
            "onSuccess": {"policy": "DeleteCluster"},
            "onFailure": {"policy": "DeleteNone"}}))
        assert wait_deployment_status(control_plane, "job1", "Complete")
        assert control_plane.wait_for(
            lambda: control_plane.server.count("RayCluster") == 0)


class TestClusterSelector:
    def test_uses_existing_cluster(self, control_plane):
        existing = simple_raycluster("shared", workers=1)
        existing.metadata.labels = {"pool": "a"}
        control_plane.client.create(existing)
        assert control_plane.wait_cluster_state("default", "shared", "ready")
        control_plane.client.create(make_rayjob(
            clusterSelector={"pool": "a"}, rayClusterSpec=None))
        assert wait_deployment_status(control_plane, "job1", "Complete")
        job = job_of(control_plane)
        assert job.status.ray_cluster_name == "shared"
        # selected cluster must never be deleted by the job
        assert control_plane.server.try_get("RayCluster", "default", "shared")


    def test_selector_created_before_cluster_exists(self, control_plane):
        """A RayJob whose clusterSelector matches nothing at creation time
        must pick up the cluster once it appears — not sit in Initializing
        forever (round-1 bug: the selector was resolved exactly once)."""
        control_plane.client.create(make_rayjob(
            clusterSelector={C.RAY_CLUSTER_LABEL_KEY: "late"},
            rayClusterSpec=None))
        assert wait_deployment_status(control_plane, "job1", "Initializing")
        # ray.io/cluster key resolves directly even before the cluster exists
        assert control_plane.wait_for(
            lambda: job_of(control_plane).status.ray_cluster_name == "late")
        time.sleep(0.3)
        assert job_of(control_plane).status.job_deployment_status == "Initializing"
        # now the cluster arrives
        control_plane.client.create(simple_raycluster("late", workers=1))
        assert wait_deployment_status(control_plane, "job1", "Complete",
                                      timeout=30)

    def test_label_selector_rechecked_each_reconcile(self, control_plane):
        """Arbitrary label selectors (no ray.io/cluster key) are re-run every
        reconcile until a cluster matches."""
        control_plane.client.create(make_rayjob(
            clusterSelector={"pool": "b"}, rayClusterSpec=None))
        assert wait_deployment_status(control_plane, "job1", "Initializing")
        time.sleep(0.3)
        assert not job_of(control_plane).status.ray_cluster_name
        late = simple_raycluster("pool-b", workers=1)
        late.metadata.labels = {"pool": "b"}
        control_plane.client.create(late)
        assert wait_deployment_status(control_plane, "job1", "Complete",
                                      timeout=30)
        assert job_of(control_plane).status.ray_cluster_name == "pool-b"


class TestInteractiveMode:
    def test_waits_for_submission_id_annotation(self, control_plane):
        job = make_rayjob(submissionMode="InteractiveMode", entrypoint=None)
        control_plane.client.create(job)
        assert wait_deployment_status(control_plane, "job1", "Waiting")
        # user submits out-of-band and annotates with the submission id
        j = job_of(control_plane)
        control_plane.dashboard.set_job_status("user-sub-1", "_pinned")
        control_plane.dashboard.jobs["user-sub-1"]["status"] = "SUCCEEDED"
        j.metadata.annotations = {"ray.io/ray-job-submission-id": "user-sub-1"}
        control_plane.client.update(j)
        assert wait_deployment_status(control_plane, "job1", "Complete")
        assert job_of(control_plane).status.job_id == "user-sub-1"


class TestCronEndToEnd:
    def test_cron_fires_job_through_full_stack(self, control_plane):
        import datetime as dt
        from kuberay_amd.models import RayCronJob
        # controllable clock anchored at real now (the CR's creationTimestamp
        # is real wall time and is the first schedule basis)
        real_now = dt.datetime.utcnow()
        fake_now = {"t": real_now}
        control_plane.raycronjob_reconciler.now_fn = lambda: fake_now["t"]
        control_plane.client.create(RayCronJob.from_dict({
            "apiVersion": "ray.io/v1", "kind": "RayCronJob",
            "metadata": {"name": "nightly", "namespace": "default"},
            "spec": {"schedule": "* * * * *",
                     "jobTemplate": make_rayjob().spec.to_dict()}}))
        import time as _t
        _t.sleep(0.3)
        assert control_plane.server.count("RayJob") == 0  # not due yet
        fake_now["t"] = real_now + dt.timedelta(minutes=2)
        # force a reconcile via annotation touch
        control_plane.client.patch(RayCronJob, "default", "nightly",
                                   {"metadata": {"annotations": {"touch": "1"}}})
        def fired():
            jobs = control_plane.server.list("RayJob")
            return len(jobs) == 1 and jobs[0]["metadata"]["labels"][
                "ray.io/cronjob-name"] == "nightly"
        assert control_plane.wait_for(fired, timeout=10)
        # the spawned RayJob runs to completion through the whole stack
        job_name = control_plane.server.list("RayJob")[0]["metadata"]["name"]
        assert wait_deployment_status(control_plane, job_name, "Complete",
                                      timeout=30)
        from kuberay_amd.models import RayCronJob as RCJ
        cron = control_plane.client.get(RCJ, "default", "nightly")
        assert cron.status.last_schedule_time  # recorded the fire time


class TestCronTimeZone:
    """spec.timeZone: cron wall-clock fields interpret in the IANA zone
    (CronJob semantics) while stored timestamps stay UTC."""

    def _reconcile(self, now_utc, tz, last=None):
        import datetime as dt
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.models import RayCronJob
        from kuberay_amd.ops.raycronjob import RayCronJobReconciler
        client = InMemoryClient()
        cron = client.create(RayCronJob.from_dict({
            "apiVersion": "ray.io/v1", "kind": "RayCronJob",
            "metadata": {"name": "tzcron", "namespace": "default"},
            "spec": {"schedule": "0 9 * * *", "timeZone": tz,
                     "jobTemplate": make_rayjob().spec.to_dict()}}))
        if last:
            cron.status.last_schedule_time = last
            client.update_status(cron)
        rec = RayCronJobReconciler(client, now_fn=lambda: now_utc)
        result = rec.reconcile(("default", "tzcron"))
        return client, result

    def test_9am_new_york_fires_at_13_utc_in_summer(self):
        import datetime as dt
        # EDT (UTC-4): 9:00 America/New_York == 13:00 UTC. At 12:59 UTC the
        # job must NOT fire; at 13:01 it must.
        client, result = self._reconcile(
            dt.datetime(2026, 7, 1, 12, 59), "America/New_York",
            last="2026-06-30T13:00:00Z")
        assert client.server.count("RayJob") == 0
        assert result.requeue_after is not None
        client, _ = self._reconcile(
            dt.datetime(2026, 7, 1, 13, 1), "America/New_York",
            last="2026-06-30T13:00:00Z")
        assert client.server.count("RayJob") == 1
        # the recorded lastScheduleTime is the UTC fire instant
        cron = client.server.get("RayCronJob", "default", "tzcron")
        assert cron["status"]["lastScheduleTime"] == "2026-07-01T13:00:00Z"

    def test_utc_default_unchanged(self):
        import datetime as dt
        client, _ = self._reconcile(dt.datetime(2026, 7, 1, 9, 1), None,
                                    last="2026-06-30T09:00:00Z")
        assert client.server.count("RayJob") == 1

    def test_invalid_timezone_rejected(self):
        from kuberay_amd.models import RayCronJob
        from kuberay_amd.utils.validation import validate_raycronjob_spec
        cron = RayCronJob.from_dict({
            "apiVersion": "ray.io/v1", "kind": "RayCronJob",
            "metadata": {"name": "bad", "namespace": "default"},
            "spec": {"schedule": "0 9 * * *", "timeZone": "Mars/Olympus",
                     "jobTemplate": make_rayjob().spec.to_dict()}})
        errs = validate_raycronjob_spec(cron)
        assert any("timeZone" in e for e in errs)


class TestSubmitterPodTemplate:
    """e2erayjobsubmitter analog: a user-supplied submitterPodTemplate is
    honored — custom image kept, custom command NOT overwritten by the
    generated `ray job submit` pipeline (job.go GetSubmitterTemplate)."""

    def test_custom_template_used_and_command_preserved(self, control_plane):
        cp = control_plane
        cp.client.create(make_rayjob(submitterPodTemplate={
            "spec": {"containers": [{
                "name": "custom-submitter",
                "image": "mycorp/ray-submitter:1.2",
                "command": ["python", "/opt/submit.py"],
                "resources": {"limits": {"cpu": "2", "memory": "2Gi"}}}],
                "restartPolicy": "Never"}}))
        assert control_plane.wait_for(
            lambda: cp.server.try_get("Job", "default", "job1") is not None)
        job = cp.server.get("Job", "default", "job1")
        container = job["spec"]["template"]["spec"]["containers"][0]
        assert container["image"] == "mycorp/ray-submitter:1.2"
        assert container["command"] == ["python", "/opt/submit.py"]
        # env the submitter needs is still injected
        env = {e["name"] for e in container.get("env", [])}
        assert "RAY_DASHBOARD_ADDRESS" in env
        assert "RAY_JOB_SUBMISSION_ID" in env
        assert wait_deployment_status(cp, "job1", "Complete", timeout=30)

    def test_default_template_uses_cluster_image(self, control_plane):
        cp = control_plane
        cp.client.create(make_rayjob())
        assert cp.wait_for(
            lambda: cp.server.try_get("Job", "default", "job1") is not None)
        job = cp.server.get("Job", "default", "job1")
        container = job["spec"]["template"]["spec"]["containers"][0]
        head_image = make_rayjob().spec.ray_cluster_spec.head_group_spec \
            .template.spec.containers[0].image
        assert container["image"] == head_image
        assert "ray job submit" in " ".join(container.get("args", []))


class TestCronSuspendAndCatchup:
    """raycronjob_controller.go analogs not yet covered: suspend gates
    firing; a long gap collapses to ONE catch-up job (latest missed tick),
    not one per missed minute."""

    def _cron(self, client, now_fn, **spec):
        from kuberay_amd.models import RayCronJob
        from kuberay_amd.ops.raycronjob import RayCronJobReconciler
        body = {"schedule": "* * * * *",
                "jobTemplate": make_rayjob().spec.to_dict()}
        body.update(spec)
        cron = client.create(RayCronJob.from_dict({
            "apiVersion": "ray.io/v1", "kind": "RayCronJob",
            "metadata": {"name": "cj", "namespace": "default"},
            "spec": body}))
        return cron, RayCronJobReconciler(client, now_fn=now_fn)

    def test_suspended_cron_never_fires(self):
        import datetime as dt

        from kuberay_amd.kube.client import InMemoryClient
        client = InMemoryClient()
        now = dt.datetime.utcnow() + dt.timedelta(minutes=10)
        _, rec = self._cron(client, lambda: now, suspend=True)
        rec.reconcile(("default", "cj"))
        assert client.server.count("RayJob") == 0

    def test_missed_ticks_collapse_to_latest(self):
        import datetime as dt

        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.models import RayCronJob
        client = InMemoryClient()
        # operator was down for 30 minutes: exactly ONE catch-up RayJob
        now = dt.datetime.utcnow() + dt.timedelta(minutes=30)
        _, rec = self._cron(client, lambda: now)
        rec.reconcile(("default", "cj"))
        assert client.server.count("RayJob") == 1
        status = client.get(RayCronJob, "default", "cj").status
        assert status.last_schedule_time
        # the recorded tick is the LATEST missed one (within a minute of now)
        fired = dt.datetime.strptime(status.last_schedule_time,
                                     "%Y-%m-%dT%H:%M:%SZ")
        assert (now - fired).total_seconds() <= 120

    def test_resume_after_suspend_fires_fresh(self):
        import datetime as dt

        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.models import RayCronJob
        client = InMemoryClient()
        now = {"t": dt.datetime.utcnow() + dt.timedelta(minutes=5)}
        cron, rec = self._cron(client, lambda: now["t"], suspend=True)
        rec.reconcile(("default", "cj"))
        assert client.server.count("RayJob") == 0

        def resume(obj):
            obj.spec.suspend = False
        client.update_with_retry(RayCronJob, "default", "cj", resume)
        now["t"] += dt.timedelta(minutes=2)
        rec.reconcile(("default", "cj"))
        assert client.server.count("RayJob") == 1


class TestSuspendResumeStorm:
    def test_random_suspend_toggles_always_recover(self, control_plane):
        """Adversarial lifecycle fuzz: random suspend/resume flips while
        the job progresses must always converge — job lands in a valid
        state, no reconciler exception, no leaked cluster after the final
        suspend."""
        import random
        rng = random.Random(7)
        control_plane.dashboard.get_job_info_mock = lambda job_id: {
            "submission_id": job_id, "status": "RUNNING"}
        control_plane.client.create(make_rayjob(
            shutdownAfterJobFinishes=True))
        assert wait_deployment_status(control_plane, "job1", "Running")
        from kuberay_amd.models import RayJob
        for _ in range(6):
            want = rng.random() < 0.5
            control_plane.client.update_with_retry(
                RayJob, "default", "job1",
                lambda j, w=want: setattr(j.spec, "suspend", w))
            time.sleep(rng.uniform(0.05, 0.3))
        # final state: suspend and verify full teardown
        control_plane.client.update_with_retry(
            RayJob, "default", "job1",
            lambda j: setattr(j.spec, "suspend", True))
        assert wait_deployment_status(control_plane, "job1", "Suspended",
                                      timeout=30)
        assert control_plane.wait_for(
            lambda: control_plane.server.count("RayCluster") == 0,
            timeout=30)
        # and resume one last time: it must come back to life
        control_plane.client.update_with_retry(
            RayJob, "default", "job1",
            lambda j: setattr(j.spec, "suspend", False))
        assert control_plane.wait_for(
            lambda: job_of(control_plane).status.job_deployment_status
            in ("Initializing", "Running", "Complete"), timeout=30)
