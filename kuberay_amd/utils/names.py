"""Name generation & truncation (reference: utils/util.go:195-400).

Kubernetes object names are capped at 63 chars; the reference reserves
space for the "-head-"/"-worker-" middle plus a 5-char random suffix and
truncates from the FRONT (keeping the distinguishing tail). Same semantics
here so that multi-CR setups name-collide (or not) identically.
"""
from __future__ import annotations

import os
import random

from . import constants as C

_RAND_ALPHABET = "bcdfghjklmnpqrstvwxz2456789"  # k8s rand.String alphabet


def rand_suffix(n: int = 5) -> str:
    return "".join(random.choice(_RAND_ALPHABET) for _ in range(n))


def check_name(s: str) -> str:
    """util.go:221 CheckName — cap at 50 chars (63 - 8 - 5), fix leading char."""
    max_length = 50
    if len(s) > max_length:
        offset = len(s) - max_length
        s = s[offset:]
    if s and (s[0].isdigit() or not s[0].isalnum()):
        s = "r" + s[1:]
    return s


def check_label(s: str) -> str:
    """util.go:251 CheckLabel — cap at 63, fix leading char."""
    max_length = 63
    if len(s) > max_length:
        s = s[len(s) - max_length:]
    if s and (s[0].isdigit() or not s[0].isalnum()):
        s = "r" + s[1:]
    return s


def trim_job_name(job_name: str) -> str:
    return check_label(job_name)


def pod_name(prefix: str, node_type: str, is_generate_name: bool) -> str:
    """util.go:203 PodName."""
    max_prefix_length = 50
    pod_prefix = prefix if len(prefix) <= max_prefix_length else prefix[:max_prefix_length]
    result = (pod_prefix + C.DASH + node_type).lower()
    if is_generate_name:
        result += C.DASH
    return result


def head_service_name(crd_type: str, cluster_spec, owner_name: str) -> str:
    """util.go:316 GenerateHeadServiceName.

    RayService/RayJob-owned head services are always ``{owner}-head-svc``;
    RayCluster ones honor a user-provided ``headGroupSpec.headService`` name.
    """
    default = f"{owner_name}-{C.HEAD_NODE}-svc"
    if crd_type in (C.KIND_RAYSERVICE, C.KIND_RAYJOB):
        return check_name_63(default)
    if crd_type == C.KIND_RAYCLUSTER:
        hs = getattr(cluster_spec.head_group_spec, "head_service", None)
        if hs is not None and hs.metadata.name:
            return hs.metadata.name
        return check_name_63(default)
    raise ValueError(f"unknown CRD type: {crd_type}")


def check_name_63(s: str) -> str:
    if len(s) > 63:
        s = s[len(s) - 63:]
    if s and (s[0].isdigit() or not s[0].isalnum()):
        s = "r" + s[1:]
    return s


def cluster_domain_name() -> str:
    return os.environ.get("CLUSTER_DOMAIN", "cluster.local")


def fqdn_service_name(cluster, namespace: str) -> str:
    """util.go:332 GenerateFQDNServiceName."""
    svc = head_service_name(C.KIND_RAYCLUSTER, cluster.spec, cluster.metadata.name)
    return f"{svc}.{namespace}.svc.{cluster_domain_name()}"


def extract_ray_ip_from_fqdn(fqdn: str) -> str:
    return fqdn.split(".")[0]


def serve_service_name(service_name: str) -> str:
    return check_name_63(f"{service_name}-serve-svc")


def serve_service_label(service_name: str) -> str:
    return check_label(f"{service_name}-serve")


def headless_service_name(cluster_name: str) -> str:
    return check_name_63(f"{cluster_name}-{C.HEADLESS_SERVICE_SUFFIX}")


def ingress_name(cluster_name: str) -> str:
    return f"{cluster_name}-{C.HEAD_NODE}-ingress"


def ray_cluster_name_for(service_or_job_name: str) -> str:
    return f"{service_or_job_name}-{rand_suffix()}"


def ray_job_id(rayjob_name: str) -> str:
    return f"{rayjob_name}-{rand_suffix()}"


def worker_replica_group_name(worker_group_name: str) -> str:
    return f"{worker_group_name}-{rand_suffix()}"


def identifier(cluster_name: str, node_type: str) -> str:
    """util.go:385 GenerateIdentifier (the ray.io/identifier label value)."""
    return f"{cluster_name}-{node_type}"


def submitter_job_name(rayjob_name: str) -> str:
    return trim_job_name(rayjob_name)


def autoscaler_service_account_name(cluster_name: str) -> str:
    return check_name(cluster_name)


def auth_secret_name(cluster_name: str) -> str:
    return check_name_63(f"{cluster_name}-auth-token")


def gcs_pvc_name(cluster_name: str) -> str:
    return check_name_63(f"{cluster_name}{C.GCS_STORAGE_PVC_SUFFIX}")


def redis_cleanup_job_name(cluster_name: str) -> str:
    return trim_job_name(f"{cluster_name}-redis-cleanup")
