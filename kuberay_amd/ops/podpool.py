"""Warm pod pool (reference: podpool/ — experimental virtual-kubelet
pre-warmed pods; the reference's PodLifecycleHandler bodies are TODO stubs,
so behavior here is designed from its README contract: requested pods are
served from a warmed-up pool, skipping scheduling + image pull + volume
prep).

Two layers:

* ``PodPoolManager`` — operator-side: keep N warm standby pods per
  (image, resource-shape) template; consumers adopt a warm pod by
  relabeling instead of cold-starting one.
* ``VirtualKubeletPodPool`` — the virtual-kubelet analog
  (podpool/cmd/main.go + manager/manager.go): registers a virtual Node
  with pooled capacity, heartbeats it, and BINDS pods that target the
  pool (``ray.io/warm-pod-pool`` nodeSelector/label) to warm pods
  instantly — the pod goes Running without waiting for a kubelet.
"""
from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional

from ..kube import objects as k8s
from ..kube.client import KubeClient
from ..kube.store import NotFoundError, now_iso

WARM_POD_LABEL = "ray.io/warm-pod"
WARM_POOL_LABEL = "ray.io/warm-pool"
POOL_SELECTOR_LABEL = "ray.io/warm-pod-pool"
VIRTUAL_NODE_ROLE = "kuberay.amd/pod-pool"


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


class VirtualKubeletPodPool:
    """Virtual-kubelet registration + instant pod binding
    (podpool/cmd/main.go runVirtualKubelet + manager.CachePodManager).

    Registers a Node named ``node_name`` whose capacity is the pools'
    aggregate shape, heartbeats it (NodeProvider Ping/NotifyNodeStatus
    analog), and watches for Pods labeled ``ray.io/warm-pod-pool: <pool>``:
    each is bound immediately — phase Running, ready condition, the warm
    pod's identity recorded — and the consumed warm pod is released. The
    consumer pod skips scheduling, image pull and volume prep, which is
    the pool's entire point (podpool/README.md:1-12).
    """

    def __init__(self, client: KubeClient, manager: PodPoolManager,
                 node_name: str = "kuberay-pod-pool",
                 heartbeat_s: float = 10.0):
        self.client = client
        self.manager = manager
        self.node_name = node_name
        self.heartbeat_s = heartbeat_s
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []

    # -- node registration (runVirtualKubelet node.NewNodeController) ---
    def register_node(self) -> None:
        from ..kube.client import RawObjectClient
        server = RawObjectClient(self.client)
        capacity = {"pods": str(sum(size for _, size in
                                    self.manager._pools.values()))}
        node = {
            "apiVersion": "v1", "kind": "Node",
            "metadata": {
                "name": self.node_name,
                "labels": {
                    "type": "virtual-kubelet",
                    "kubernetes.io/role": VIRTUAL_NODE_ROLE,
                    "kubernetes.io/os": "linux",
                    "kubernetes.io/arch": "amd64",
                }},
            "spec": {"taints": [{
                "key": "virtual-kubelet.io/provider",
                "value": "kuberay-pod-pool", "effect": "NoSchedule"}]},
            "status": {"capacity": capacity,
                       "conditions": [{"type": "Ready", "status": "True",
                                       "lastHeartbeatTime": now_iso()}]},
        }
        existing = server.try_get("Node", "default", self.node_name)
        if existing is None:
            server.create(node)
        else:
            existing["status"] = node["status"]
            server.update(existing)

    def heartbeat(self) -> None:
        from ..kube.client import RawObjectClient
        server = RawObjectClient(self.client)
        node = server.try_get("Node", "default", self.node_name)
        if node is None:
            self.register_node()
            return
        node.setdefault("status", {})["conditions"] = [{
            "type": "Ready", "status": "True",
            "lastHeartbeatTime": now_iso()}]
        server.update(node)

    # -- pod binding (CreatePod/GetPodStatus/NotifyPods analog) -----------
    def bind_pending_pods(self) -> int:
        """Bind every unbound pod that targets a pool. Returns bound count."""
        bound = 0
        pods = self.client.list(k8s.Pod, self.manager.namespace)
        for pod in pods:
            labels = pod.metadata.labels or {}
            pool = labels.get(POOL_SELECTOR_LABEL)
            if not pool or labels.get(WARM_POD_LABEL):
                continue
            if (pod.status.phase or "Pending") != "Pending":
                continue
            warm = self.manager.adopt(pool, {"ray.io/warm-consumer":
                                             pod.metadata.name or ""})
            if warm is None:
                continue  # pool dry: pod stays Pending (real-kubelet path)
            try:
                self.client.patch(
                    k8s.Pod, self.manager.namespace, pod.metadata.name,
                    {"metadata": {"annotations": {
                        "ray.io/warm-pod-source": warm,
                        "kubernetes.io/hostname": self.node_name}}})
                self.client.patch(
                    k8s.Pod, self.manager.namespace, pod.metadata.name,
                    {"phase": "Running", "podIP": "10.255.0.1",
                     "startTime": now_iso(),
                     "containerStatuses": [
                         {"name": c.name, "ready": True,
                          "state": {"running": {}}}
                         for c in pod.spec.containers or []],
                     "conditions": [{"type": "Ready", "status": "True"}]},
                    subresource="status")
                bound += 1
                # warm pod consumed: remove it and let reconcile refill
                try:
                    self.client.delete(k8s.Pod, self.manager.namespace, warm)
                except NotFoundError:
                    pass
            except NotFoundError:
                continue
        if bound:
            self.manager.reconcile()  # refill consumed capacity
        return bound

    # -- lifecycle ---------------------------------------------------------
    def start(self) -> None:
        self.manager.reconcile()
        self.register_node()

        def hb_loop():
            while not self._stop.wait(self.heartbeat_s):
                try:
                    self.heartbeat()
                except Exception:
                    pass

        def bind_loop():
            server = getattr(self.client, "server", None)
            watcher = server.watch({"Pod"}) if server is not None else None
            while not self._stop.is_set():
                if watcher is not None:
                    ev = watcher.next(timeout=0.2)
                    if ev is None:
                        continue
                else:
                    self._stop.wait(0.2)
                try:
                    self.bind_pending_pods()
                except Exception:
                    pass
            if watcher is not None:
                watcher.stop()

        for fn, name in ((hb_loop, "podpool-heartbeat"),
                         (bind_loop, "podpool-bind")):
            t = threading.Thread(target=fn, name=name, daemon=True)
            t.start()
            self._threads.append(t)

    def stop(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=2)
