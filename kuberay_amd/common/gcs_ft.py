"""GCS fault-tolerance helpers: Redis cleanup Job (reference: common/gcs_ft.go).

When a GCS-FT RayCluster is deleted, its Redis external-storage namespace
must be wiped; the operator runs a one-shot cleanup Job (finalizer machine in
raycluster_controller.go:226-352).
"""
from __future__ import annotations

import copy

from ..kube.objects import Job, JobSpec, ObjectMeta, PodTemplateSpec
from ..models.raycluster import RayCluster, RayNodeType
from ..utils import constants as C
from ..utils import names


def build_redis_cleanup_job(cluster: RayCluster) -> Job:
    """gcs_ft.go BuildRedisCleanupJob — reuses the head template's image/env
    so credentials and network policy match the head pod."""
    template: PodTemplateSpec = cluster.spec.head_group_spec.template.clone()
    container = template.spec.containers[C.RAY_CONTAINER_INDEX]
    container.name = "redis-cleanup"
    container.command = ["/bin/bash", "-c", "--"]
    container.args = [
        "python -c "
        "\"from ray._private.gcs_utils import cleanup_redis_storage; "
        "from urllib.parse import urlparse; import os, sys; "
        "redis_address = os.getenv('RAY_REDIS_ADDRESS', '').split(',')[0]; "
        "redis_address = redis_address if '://' in redis_address else 'redis://' + redis_address; "
        "parsed = urlparse(redis_address); "
        "sys.exit(1) if not cleanup_redis_storage("
        "host=parsed.hostname, port=parsed.port, "
        "password=os.getenv('REDIS_PASSWORD', parsed.password or ''), "
        "use_ssl=parsed.scheme=='rediss', "
        "storage_namespace=os.getenv('RAY_external_storage_namespace'), "
        "username=os.getenv('REDIS_USERNAME', parsed.username or None)) else None\""
    ]
    # cleanup pod needs only the env, not ports/probes/lifecycle
    container.ports = None
    container.liveness_probe = None
    container.readiness_probe = None
    container.lifecycle = None
    container.resources = copy.deepcopy(container.resources)

    opts = cluster.spec.gcs_fault_tolerance_options
    if opts:
        if opts.redis_address:
            container.set_env_if_absent(C.RAY_REDIS_ADDRESS, opts.redis_address)
        ns = (opts.external_storage_namespace or
              (cluster.metadata.annotations or {}).get(C.RAY_EXTERNAL_STORAGE_NS_ANNOTATION_KEY)
              or cluster.metadata.uid or "")
        container.set_env_if_absent(C.RAY_EXTERNAL_STORAGE_NS, ns)

    template.metadata.name = None
    template.metadata.generate_name = None
    template.metadata.labels = {
        C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name,
        C.RAY_NODE_TYPE_LABEL_KEY: RayNodeType.REDIS_CLEANUP,
        C.KUBERNETES_APPLICATION_NAME_LABEL_KEY: C.APPLICATION_NAME,
        C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME,
    }
    template.spec.restart_policy = "Never"
    template.spec.init_containers = None

    return Job(
        metadata=ObjectMeta(
            name=names.redis_cleanup_job_name(cluster.metadata.name),
            namespace=cluster.metadata.namespace or "default",
            labels=dict(template.metadata.labels),
        ),
        spec=JobSpec(template=template, backoff_limit=0,
                     ttl_seconds_after_finished=300),
    )
