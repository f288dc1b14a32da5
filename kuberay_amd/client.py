"""User-facing python client (reference: clients/python-client —
RayClusterApi + kuberay_cluster_builder.Director/ClusterBuilder).

``RayClusterApi`` speaks the ray.io/v1 CRD surface against any KubeClient
(in-memory, HTTP, or REST); the ``Director`` builds opinionated MI355X
cluster presets (small/medium/large map to 1/4/8 amd.com/gpu workers on the
single 8xMI355X node).
"""
from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from .kube.client import KubeClient
from .kube.httpclient import HttpKubeClient
from .models import RayCluster, RayJob, RayService
from .utils import constants as C


class RayClusterApi:
    """clients/python-client kuberay_cluster_api analog."""

    def __init__(self, client: Optional[KubeClient] = None,
                 server_url: Optional[str] = None):
        if client is None:
            client = HttpKubeClient(server_url or "http://127.0.0.1:8888")
        self.client = client

    # -- clusters ------------------------------------------------------
    def create_ray_cluster(self, body: Dict[str, Any]) -> Dict[str, Any]:
        return self.client.create(RayCluster.from_dict(body)).to_dict()

    def get_ray_cluster(self, name: str, k8s_namespace: str = "default") -> Optional[Dict[str, Any]]:
        obj = self.client.try_get(RayCluster, k8s_namespace, name)
        return obj.to_dict() if obj else None

    def list_ray_clusters(self, k8s_namespace: str = "default",
                          label_selector: Optional[Dict[str, str]] = None) -> List[Dict[str, Any]]:
        return [o.to_dict() for o in self.client.list(RayCluster, k8s_namespace,
                                                      label_selector)]

    def patch_ray_cluster(self, name: str, ray_patch: Dict[str, Any],
                          k8s_namespace: str = "default") -> Dict[str, Any]:
        return self.client.patch(RayCluster, k8s_namespace, name, ray_patch).to_dict()

    def delete_ray_cluster(self, name: str, k8s_namespace: str = "default") -> None:
        self.client.delete(RayCluster, k8s_namespace, name)

    def wait_until_ray_cluster_running(self, name: str,
                                       k8s_namespace: str = "default",
                                       timeout: int = 300,
                                       delay_between_attempts: float = 2.0) -> bool:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            obj = self.client.try_get(RayCluster, k8s_namespace, name)
            if obj is not None and obj.status.state == "ready":
                return True
            time.sleep(delay_between_attempts)
        return False

    # -- jobs / services ----------------------------------------------
    def submit_ray_job(self, body: Dict[str, Any]) -> Dict[str, Any]:
        return self.client.create(RayJob.from_dict(body)).to_dict()

    def get_ray_job(self, name: str, k8s_namespace: str = "default") -> Optional[Dict[str, Any]]:
        obj = self.client.try_get(RayJob, k8s_namespace, name)
        return obj.to_dict() if obj else None

    def create_ray_service(self, body: Dict[str, Any]) -> Dict[str, Any]:
        return self.client.create(RayService.from_dict(body)).to_dict()


class ClusterBuilder:
    """kuberay_cluster_builder.ClusterBuilder analog (fluent)."""

    def __init__(self):
        self._name = "ray-cluster"
        self._namespace = "default"
        self._image = C.DEFAULT_RAY_ROCM_IMAGE
        self._head = {"cpu": "2", "memory": "4Gi"}
        self._groups: List[Dict[str, Any]] = []
        self._labels: Dict[str, str] = {}
        self._autoscaler = False

    def build_meta(self, name: str, k8s_namespace: str = "default",
                   labels: Optional[Dict[str, str]] = None) -> "ClusterBuilder":
        self._name = name
        self._namespace = k8s_namespace
        self._labels = labels or {}
        return self

    def build_head(self, cpu: str = "2", memory: str = "4Gi",
                   image: Optional[str] = None) -> "ClusterBuilder":
        self._head = {"cpu": cpu, "memory": memory}
        if image:
            self._image = image
        return self

    def build_worker(self, group_name: str = "default-group", replicas: int = 1,
                     min_replicas: int = 0, max_replicas: int = 8,
                     cpu: str = "4", memory: str = "8Gi", gpu: int = 0,
                     image: Optional[str] = None) -> "ClusterBuilder":
        limits = {"cpu": cpu, "memory": memory}
        if gpu:
            limits[C.AMD_GPU_RESOURCE_NAME] = str(gpu)
        self._groups.append({
            "groupName": group_name, "replicas": replicas,
            "minReplicas": min_replicas, "maxReplicas": max_replicas,
            "rayStartParams": {},
            "template": {"spec": {"containers": [{
                "name": "ray-worker", "image": image or self._image,
                "resources": {"limits": limits, "requests": dict(limits)}}]}},
        })
        return self

    def enable_autoscaling(self) -> "ClusterBuilder":
        self._autoscaler = True
        return self

    def get_cluster(self) -> Dict[str, Any]:
        spec: Dict[str, Any] = {
            "rayVersion": "2.46.0",
            "headGroupSpec": {
                "rayStartParams": {},
                "template": {"spec": {"containers": [{
                    "name": "ray-head", "image": self._image,
                    "resources": {"limits": dict(self._head),
                                  "requests": dict(self._head)}}]}},
            },
            "workerGroupSpecs": self._groups or [],
        }
        if self._autoscaler:
            spec["enableInTreeAutoscaling"] = True
        return {"apiVersion": C.API_VERSION, "kind": C.KIND_RAYCLUSTER,
                "metadata": {"name": self._name, "namespace": self._namespace,
                             **({"labels": self._labels} if self._labels else {})},
                "spec": spec}


class Director:
    """Small/medium/large MI355X presets (python-client Director analog)."""

    def __init__(self, api: Optional[RayClusterApi] = None):
        self.api = api

    def build_small_cluster(self, name: str, k8s_namespace: str = "default") -> Dict[str, Any]:
        return (ClusterBuilder().build_meta(name, k8s_namespace)
                .build_head()
                .build_worker(replicas=1, gpu=1, cpu="16", memory="64Gi")
                .get_cluster())

    def build_medium_cluster(self, name: str, k8s_namespace: str = "default") -> Dict[str, Any]:
        return (ClusterBuilder().build_meta(name, k8s_namespace)
                .build_head(cpu="4", memory="8Gi")
                .build_worker(replicas=4, gpu=1, cpu="16", memory="128Gi")
                .get_cluster())

    def build_large_cluster(self, name: str, k8s_namespace: str = "default") -> Dict[str, Any]:
        # one full 8xMI355X node
        return (ClusterBuilder().build_meta(name, k8s_namespace)
                .build_head(cpu="8", memory="16Gi")
                .build_worker(replicas=8, gpu=1, cpu="24", memory="256Gi")
                .get_cluster())

    def create(self, cluster: Dict[str, Any]) -> Dict[str, Any]:
        if self.api is None:
            raise RuntimeError("Director needs a RayClusterApi to create clusters")
        return self.api.create_ray_cluster(cluster)
