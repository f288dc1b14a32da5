"""RayJob submitter construction (reference: common/job.go).

The submitter shell pipeline is idempotent against retries
(job.go:119-209): wait for dashboard GCS health, then
``ray job status || ray job submit --no-wait``, then ``ray job logs
--follow``.
"""
from __future__ import annotations

import json
import shlex
from typing import List, Optional

import yaml

from ..kube.objects import (
    Container,
    Job,
    JobSpec,
    ObjectMeta,
    PodTemplateSpec,
    ResourceRequirements,
)
from ..models.rayjob import JobSubmissionMode, RayJob
from ..utils import constants as C
from ..utils import names
from ..utils.resources import find_container_port

SUBMITTER_CONTAINER_NAME = "ray-job-submitter"
GCS_HEALTH_CHECK_TIMEOUT_SECONDS = 5
BASE_PYTHON_HEALTH_COMMAND = (
    'python -c "import urllib.request, sys; '
    "sys.exit(0 if b'success' in urllib.request.urlopen('%s', timeout=%d).read() else 1)\""
)


def _runtime_env_json(rayjob: RayJob) -> Optional[str]:
    if not rayjob.spec.runtime_env_yaml:
        return None
    data = yaml.safe_load(rayjob.spec.runtime_env_yaml)
    if not data:
        return None
    return json.dumps(data, sort_keys=True)


def build_job_submit_command(rayjob: RayJob, submission_mode: str) -> List[str]:
    """job.go:90 BuildJobSubmitCommand."""
    if submission_mode == JobSubmissionMode.SIDECAR:
        head_container = (rayjob.spec.ray_cluster_spec.head_group_spec
                          .template.spec.containers[C.RAY_CONTAINER_INDEX])
        port = find_container_port(head_container, C.DASHBOARD_PORT_NAME,
                                   C.DEFAULT_DASHBOARD_PORT)
        address = f"http://127.0.0.1:{port}"
        health_url = f"http://localhost:{port}/{C.RAY_DASHBOARD_GCS_HEALTH_PATH}"
        # sidecar containers never retry unless the SidecarSubmitterRestart
        # gate makes them restartable (job.go:160-165)
        from .. import features
        needs_status_check = features.enabled("SidecarSubmitterRestart")
    elif submission_mode == JobSubmissionMode.K8S_JOB:
        address = rayjob.status.dashboard_url or ""
        if not address.startswith("http://"):
            address = "http://" + address
        health_url = f"{address}/{C.RAY_DASHBOARD_GCS_HEALTH_PATH}"
        needs_status_check = True
    else:
        raise ValueError(f"unsupported submission mode for submit command: {submission_mode}")

    job_id = rayjob.status.job_id or ""
    entrypoint = (rayjob.spec.entrypoint or "").strip()

    health_cmd = BASE_PYTHON_HEALTH_COMMAND % (health_url, GCS_HEALTH_CHECK_TIMEOUT_SECONDS)
    cmd: List[str] = [
        "until", health_cmd, ">/dev/null", "2>&1", ";",
        "do", "echo", shlex.quote(f"Waiting for Ray Dashboard GCS to become healthy at {address} ..."),
        ";", "sleep", "2", ";", "done", ";",
    ]

    status_cmd = ["ray", "job", "status", "--address", address, job_id, ">/dev/null", "2>&1"]
    submit_cmd = ["ray", "job", "submit", "--address", address]
    follow_cmd = ["ray", "job", "logs", "--address", address, "--follow", job_id]

    if needs_status_check:
        cmd += ["if", "!", *status_cmd, ";", "then"]
    cmd += submit_cmd
    if needs_status_check:
        cmd.append("--no-wait")

    runtime_env = _runtime_env_json(rayjob)
    if runtime_env:
        cmd += ["--runtime-env-json", shlex.quote(runtime_env)]
    if rayjob.spec.metadata:
        cmd += ["--metadata-json", shlex.quote(json.dumps(rayjob.spec.metadata, sort_keys=True))]
    if job_id:
        cmd += ["--submission-id", job_id]
    if rayjob.spec.entrypoint_num_cpus and rayjob.spec.entrypoint_num_cpus > 0:
        cmd += ["--entrypoint-num-cpus", f"{rayjob.spec.entrypoint_num_cpus:f}"]
    if rayjob.spec.entrypoint_num_gpus and rayjob.spec.entrypoint_num_gpus > 0:
        cmd += ["--entrypoint-num-gpus", f"{rayjob.spec.entrypoint_num_gpus:f}"]
    if rayjob.spec.entrypoint_resources:
        cmd += ["--entrypoint-resources", shlex.quote(rayjob.spec.entrypoint_resources)]
    cmd += ["--", entrypoint, ";"]
    if needs_status_check:
        cmd += ["fi", ";", *follow_cmd]
    return cmd


def default_submitter_container(ray_cluster_spec) -> Container:
    """job.go:230 GetDefaultSubmitterContainer — head image for version parity."""
    image = None
    if ray_cluster_spec is not None:
        image = (ray_cluster_spec.head_group_spec.template.spec
                 .containers[C.RAY_CONTAINER_INDEX].image)
    return Container(
        name=SUBMITTER_CONTAINER_NAME,
        image=image,
        resources=ResourceRequirements(
            limits={"cpu": "1", "memory": "1Gi"},
            requests={"cpu": "500m", "memory": "200Mi"},
        ),
    )


def get_submitter_template(rayjob: RayJob) -> PodTemplateSpec:
    """job.go:215 GetSubmitterTemplate."""
    if rayjob.spec.submitter_pod_template is not None:
        return rayjob.spec.submitter_pod_template.clone()
    return PodTemplateSpec.model_validate({
        "spec": {
            "containers": [default_submitter_container(rayjob.spec.ray_cluster_spec).to_dict()],
            "restartPolicy": "Never",
        }
    })


def build_submitter_job(rayjob: RayJob) -> Job:
    """rayjob_controller.go:560-585 createK8sJobIfNeed → the batch Job spec."""
    template = get_submitter_template(rayjob)
    cmd = build_job_submit_command(rayjob, JobSubmissionMode.K8S_JOB)
    container = template.spec.containers[0]
    if not container.command:
        container.command = ["/bin/bash", "-c", "--"]
        container.args = [" ".join(cmd)]
    container.set_env_if_absent(C.RAY_DASHBOARD_ADDRESS, rayjob.status.dashboard_url or "")
    container.set_env_if_absent(C.RAY_JOB_SUBMISSION_ID, rayjob.status.job_id or "")
    template.spec.restart_policy = template.spec.restart_policy or "Never"

    labels = template.metadata.ensure_labels()
    labels.update({
        C.RAY_ORIGINATED_FROM_CR_NAME_LABEL_KEY: names.check_label(rayjob.metadata.name),
        C.RAY_ORIGINATED_FROM_CRD_LABEL_KEY: C.KIND_RAYJOB,
        C.KUBERNETES_APPLICATION_NAME_LABEL_KEY: C.APPLICATION_NAME,
        C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME,
    })

    backoff = 2
    if rayjob.spec.submitter_config and rayjob.spec.submitter_config.backoff_limit is not None:
        backoff = rayjob.spec.submitter_config.backoff_limit

    return Job(
        metadata=ObjectMeta(
            name=names.submitter_job_name(rayjob.metadata.name),
            namespace=rayjob.metadata.namespace or "default",
            labels=dict(labels),
        ),
        spec=JobSpec(template=template, backoff_limit=backoff),
    )
