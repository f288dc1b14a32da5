"""APIServer v1 object converters (reference: apiserver/pkg/util/cluster.go +
pkg/model/converter.go).

The v1 HTTP API speaks simplified ``api.Cluster``-shaped JSON; these
converters expand it into full CRs and back. MI355X scoping: the default
accelerator resource is ``amd.com/gpu`` (the reference defaults to
nvidia.com/gpu at apiserver/pkg/util/cluster.go:257,:554 — dropped).
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

from ..kube.objects import ConfigMap, ObjectMeta
from ..models import RayCluster, RayJob, RayService
from ..utils import constants as C

COMPUTE_TEMPLATE_LABEL = "ray.io/compute-template"


# ---------------------------------------------------------------------------
# compute templates <-> ConfigMaps (resource_manager.go:317-395)
# ---------------------------------------------------------------------------

def compute_template_to_configmap(namespace: str, template: Dict[str, Any]) -> ConfigMap:
    data = {
        "name": template["name"],
        "namespace": namespace,
        "cpu": str(template.get("cpu", 1)),
        "memory": str(template.get("memory", 1)),
        "gpu": str(template.get("gpu", 0)),
        # MI355X: accelerator key is amd.com/gpu unless caller overrides
        "gpu_accelerator": template.get("gpuAccelerator", C.AMD_GPU_RESOURCE_NAME),
    }
    if template.get("tolerations"):
        import json
        data["tolerations"] = json.dumps(template["tolerations"])
    return ConfigMap(
        metadata=ObjectMeta(name=template["name"], namespace=namespace,
                            labels={COMPUTE_TEMPLATE_LABEL: template["name"]}),
        data=data,
    )


def configmap_to_compute_template(cm: ConfigMap) -> Dict[str, Any]:
    data = cm.data or {}
    out = {
        "name": data.get("name", cm.metadata.name),
        "namespace": data.get("namespace", cm.metadata.namespace),
        "cpu": int(float(data.get("cpu", "1"))),
        "memory": int(float(data.get("memory", "1"))),
        "gpu": int(float(data.get("gpu", "0"))),
        "gpuAccelerator": data.get("gpu_accelerator", C.AMD_GPU_RESOURCE_NAME),
    }
    if data.get("tolerations"):
        import json
        out["tolerations"] = json.loads(data["tolerations"])
    return out


def _container_from_template(name: str, image: str, template: Dict[str, Any],
                             env: Optional[Dict[str, str]] = None) -> Dict[str, Any]:
    limits: Dict[str, Any] = {
        "cpu": str(template.get("cpu", 1)),
        "memory": f"{template.get('memory', 1)}Gi",
    }
    if template.get("gpu"):
        limits[template.get("gpuAccelerator", C.AMD_GPU_RESOURCE_NAME)] = \
            str(template["gpu"])
    container = {
        "name": name,
        "image": image,
        "resources": {"limits": dict(limits), "requests": dict(limits)},
    }
    if env:
        container["env"] = [{"name": k, "value": v} for k, v in env.items()]
    return container


# ---------------------------------------------------------------------------
# api.Cluster <-> RayCluster CR
# ---------------------------------------------------------------------------

def api_cluster_to_raycluster(namespace: str, cluster: Dict[str, Any],
                              templates: Dict[str, Dict[str, Any]]) -> RayCluster:
    def resolved(group: Dict[str, Any], default: Dict[str, Any]):
        """Compute template resolved + inline cpu/memory/gpu overrides (the
        dashboard UI submits inline resources without a template)."""
        t = dict(templates.get(group.get("computeTemplate", ""), default))
        for key in ("cpu", "memory", "gpu"):
            if group.get(key) is not None:
                t[key] = group[key]
        return t

    spec = cluster.get("clusterSpec", {})
    head = spec.get("headGroupSpec", {})
    head_template = resolved(head, {"cpu": 1, "memory": 2})
    head_pod_spec: Dict[str, Any] = {"containers": [
        _container_from_template("ray-head", head.get("image")
                                 or C.DEFAULT_RAY_ROCM_IMAGE,
                                 head_template,
                                 head.get("environment"))]}
    if head_template.get("tolerations"):
        head_pod_spec["tolerations"] = head_template["tolerations"]
    head_group = {
        "serviceType": head.get("serviceType"),
        "rayStartParams": dict(head.get("rayStartParams") or {}),
        "template": {"spec": head_pod_spec},
    }
    worker_groups = []
    for wg in spec.get("workerGroupSpec", []) or []:
        t = resolved(wg, {"cpu": 1, "memory": 1})
        worker_groups.append({
            "groupName": wg.get("groupName", "worker-group"),
            "replicas": wg.get("replicas", 1),
            "minReplicas": wg.get("minReplicas", 0),
            "maxReplicas": wg.get("maxReplicas", wg.get("replicas", 1)),
            "rayStartParams": dict(wg.get("rayStartParams") or {}),
            "template": {"spec": {
                "containers": [
                    _container_from_template("ray-worker", wg.get("image")
                                             or C.DEFAULT_RAY_ROCM_IMAGE, t,
                                             wg.get("environment"))],
                **({"tolerations": t["tolerations"]}
                   if t.get("tolerations") else {}),
            }},
        })
    annotations = dict(cluster.get("annotations") or {})
    labels = dict(cluster.get("labels") or {})
    if cluster.get("user"):
        labels["ray.io/user"] = str(cluster["user"]).replace("@", "-")
    return RayCluster.from_dict({
        "apiVersion": C.API_VERSION,
        "kind": C.KIND_RAYCLUSTER,
        "metadata": {"name": cluster["name"], "namespace": namespace,
                     "labels": labels or None,
                     "annotations": annotations or None},
        "spec": {
            "rayVersion": cluster.get("version"),
            "headGroupSpec": head_group,
            "workerGroupSpecs": worker_groups,
            **({"enableInTreeAutoscaling": True}
               if spec.get("enableInTreeAutoscaling") else {}),
        },
    })


def raycluster_to_api_cluster(rc: RayCluster) -> Dict[str, Any]:
    head = rc.spec.head_group_spec
    head_container = head.template.spec.containers[0] if head.template.spec.containers else None
    out: Dict[str, Any] = {
        "name": rc.metadata.name,
        "namespace": rc.metadata.namespace,
        "version": rc.spec.ray_version,
        "createdAt": rc.metadata.creation_timestamp,
        "clusterState": rc.status.state or "",
        "clusterSpec": {
            "headGroupSpec": {
                "serviceType": head.service_type,
                "rayStartParams": head.ray_start_params,
                "image": head_container.image if head_container else None,
            },
            "workerGroupSpec": [
                {
                    "groupName": g.group_name,
                    "replicas": g.replicas,
                    "minReplicas": g.min_replicas,
                    "maxReplicas": g.max_replicas,
                    "rayStartParams": g.ray_start_params,
                    "image": (g.template.spec.containers[0].image
                              if g.template.spec.containers else None),
                }
                for g in rc.spec.worker_group_specs
            ],
        },
        "events": [],
        "serviceEndpoint": dict(rc.status.endpoints or {}),
    }
    if rc.metadata.labels and rc.metadata.labels.get("ray.io/user"):
        out["user"] = rc.metadata.labels["ray.io/user"]
    return out


# ---------------------------------------------------------------------------
# api.RayJob / api.RayService
# ---------------------------------------------------------------------------

def api_job_to_rayjob(namespace: str, job: Dict[str, Any],
                      templates: Dict[str, Dict[str, Any]]) -> RayJob:
    spec: Dict[str, Any] = {
        "entrypoint": job.get("entrypoint"),
        "shutdownAfterJobFinishes": job.get("shutdownAfterJobFinishes", True),
        "ttlSecondsAfterFinished": job.get("ttlSecondsAfterFinished", 0),
        "submissionMode": job.get("submissionMode", "K8sJobMode"),
    }
    if job.get("runtimeEnv"):
        spec["runtimeEnvYAML"] = job["runtimeEnv"]
    if job.get("metadata"):
        spec["metadata"] = job["metadata"]
    if job.get("clusterSelector"):
        spec["clusterSelector"] = job["clusterSelector"]
    elif job.get("clusterSpec") is not None:
        rc = api_cluster_to_raycluster(
            namespace, {"name": job["name"], "clusterSpec": job["clusterSpec"]},
            templates)
        spec["rayClusterSpec"] = rc.spec.to_dict()
    return RayJob.from_dict({
        "apiVersion": C.API_VERSION, "kind": C.KIND_RAYJOB,
        "metadata": {"name": job["name"], "namespace": namespace},
        "spec": spec,
    })


def rayjob_to_api_job(job: RayJob) -> Dict[str, Any]:
    return {
        "name": job.metadata.name,
        "namespace": job.metadata.namespace,
        "entrypoint": job.spec.entrypoint,
        "jobStatus": job.status.job_status,
        "jobDeploymentStatus": job.status.job_deployment_status,
        "message": job.status.message,
        "rayClusterName": job.status.ray_cluster_name,
        "createdAt": job.metadata.creation_timestamp,
        "startTime": job.status.start_time,
        "endTime": job.status.end_time,
    }


def api_service_to_rayservice(namespace: str, svc: Dict[str, Any],
                              templates: Dict[str, Dict[str, Any]]) -> RayService:
    rc = api_cluster_to_raycluster(
        namespace, {"name": svc["name"], "clusterSpec": svc.get("clusterSpec", {})},
        templates)
    return RayService.from_dict({
        "apiVersion": C.API_VERSION, "kind": C.KIND_RAYSERVICE,
        "metadata": {"name": svc["name"], "namespace": namespace},
        "spec": {
            "serveConfigV2": svc.get("serveConfig_V2") or svc.get("serveConfigV2"),
            "rayClusterConfig": rc.spec.to_dict(),
        },
    })


def rayservice_to_api_service(svc: RayService) -> Dict[str, Any]:
    return {
        "name": svc.metadata.name,
        "namespace": svc.metadata.namespace,
        "serveConfigV2": svc.spec.serve_config_v2,
        "serviceStatus": svc.status.service_status,
        "numServeEndpoints": svc.status.num_serve_endpoints,
        "activeRayClusterName": svc.status.active_service_status.ray_cluster_name,
        "pendingRayClusterName": svc.status.pending_service_status.ray_cluster_name,
        "createdAt": svc.metadata.creation_timestamp,
    }
