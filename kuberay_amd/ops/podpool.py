"""Warm pod pool (reference: podpool/ — experimental virtual-kubelet
pre-warmed pods, 337 LoC).

Operator-side analog without a virtual kubelet: keep N warm standby pods
per (image, resource-shape) template; the RayCluster reconciler can adopt a
warm pod by relabeling instead of cold-starting one (saving image pull +
ray runtime init on scale-up bursts). Opt-in via PodPoolManager wiring.
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional

from ..kube import objects as k8s
from ..kube.client import KubeClient
from ..kube.store import NotFoundError

WARM_POD_LABEL = "ray.io/warm-pod"
WARM_POOL_LABEL = "ray.io/warm-pool"


class PodPoolManager:
    def __init__(self, client: KubeClient, namespace: str = "default"):
        self.client = client
        self.namespace = namespace
        self._lock = threading.Lock()
        # pool name -> (template dict, target size)
        self._pools: Dict[str, tuple] = {}

    def define_pool(self, name: str, template: Dict, size: int) -> None:
        with self._lock:
            self._pools[name] = (template, size)

    def _pool_pods(self, pool: str) -> List[k8s.Pod]:
        return self.client.list(k8s.Pod, self.namespace,
                                {WARM_POOL_LABEL: pool})

    def reconcile(self) -> None:
        """Top up every pool to its target size."""
        with self._lock:
            pools = dict(self._pools)
        for pool, (template, size) in pools.items():
            pods = [p for p in self._pool_pods(pool)
                    if not p.metadata.deletion_timestamp]
            warm = [p for p in pods
                    if (p.metadata.labels or {}).get(WARM_POD_LABEL) == "true"]
            for _ in range(size - len(warm)):
                pod = k8s.Pod.from_dict(template)
                pod.metadata.generate_name = f"warm-{pool}-"
                pod.metadata.name = None
                pod.metadata.namespace = self.namespace
                labels = pod.metadata.ensure_labels()
                labels[WARM_POD_LABEL] = "true"
                labels[WARM_POOL_LABEL] = pool
                self.client.create(pod)

    def adopt(self, pool: str, new_labels: Dict[str, str]) -> Optional[str]:
        """Hand a warm pod to a consumer: relabel it out of the pool.
        Returns the pod name or None when the pool is dry."""
        for pod in self._pool_pods(pool):
            labels = pod.metadata.labels or {}
            if labels.get(WARM_POD_LABEL) != "true" or pod.metadata.deletion_timestamp:
                continue
            patch = {"metadata": {"labels": {
                WARM_POD_LABEL: "adopted", **new_labels}}}
            try:
                self.client.patch(k8s.Pod, self.namespace, pod.metadata.name, patch)
                return pod.metadata.name
            except NotFoundError:
                continue
        return None
