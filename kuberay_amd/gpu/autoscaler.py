"""MI355X autoscaler decision loop.

The BASELINE north star replaces the nvidia-smi-free world of the reference
(where autoscaling is delegated entirely to Ray's in-pod sidecar, §3.4 of
SURVEY.md) with an operator-owned loop that reads rocm-smi utilisation and
HBM occupancy (288 GB per GPU) and drives scale through the SAME CR contract
the sidecar uses: PATCH ``spec.workerGroupSpecs[i].replicas`` and
``scaleStrategy.workersToDelete``. Both autoscalers therefore compose: the
in-tree sidecar reacts to Ray scheduler load, this loop reacts to device
telemetry. It is enabled per-cluster via the
``ray.io/amd-gpu-autoscaler: "true"`` annotation.

Policy (per GPU worker group, clamped to [minReplicas, maxReplicas]):
  * scale UP by 1 when avg GPU utilisation > ``up_util_pct`` OR max HBM
    occupancy > ``up_hbm_fraction`` for ``up_stable_s`` seconds
    (the node is saturated — more workers spread the actors),
  * scale DOWN by naming the newest worker in WorkersToDelete when avg
    utilisation < ``down_util_pct`` AND HBM < ``down_hbm_fraction`` for
    ``idle_timeout_s`` seconds (never random-delete: the operator honors
    the same no-random-delete gate as the reference),
  * cooldown between decisions.
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Tuple

from ..common import association
from ..kube import objects as k8s
from ..kube.client import KubeClient
from ..models import RayCluster
from ..utils.resources import pod_gpu_count

AMD_AUTOSCALER_ANNOTATION = "ray.io/amd-gpu-autoscaler"


@dataclass
class AutoscalerPolicy:
    up_util_pct: float = 70.0
    up_hbm_fraction: float = 0.85
    up_stable_s: float = 10.0
    down_util_pct: float = 15.0
    down_hbm_fraction: float = 0.30
    idle_timeout_s: float = 60.0
    cooldown_s: float = 30.0


@dataclass
class GroupState:
    high_since: Optional[float] = None
    low_since: Optional[float] = None
    last_action_ts: float = 0.0


class MI355XAutoscaler:
    def __init__(self, client: KubeClient,
                 telemetry: Optional[Callable[[], Dict]] = None,
                 policy: Optional[AutoscalerPolicy] = None,
                 clock: Callable[[], float] = time.monotonic,
                 recorder=None, metrics=None, node_name: str = ""):
        import os
        from .rocm_smi import node_gpu_summary
        self.client = client
        self.telemetry = telemetry or node_gpu_summary
        self.policy = policy or AutoscalerPolicy()
        self.clock = clock
        self.recorder = recorder
        self.metrics = metrics
        self.node_name = node_name or os.environ.get("NODE_NAME", "local")
        self._states: Dict[Tuple[str, str, str], GroupState] = {}
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    # ------------------------------------------------------------------
    def step(self) -> List[str]:
        """One evaluation pass over all opted-in clusters. Returns a list of
        human-readable decisions (for logs/tests)."""
        decisions: List[str] = []
        try:
            summary = self.telemetry()
            if self.metrics is not None:
                self.metrics.observe_gpu_health(self.node_name, True)
        except Exception:
            if self.metrics is not None:
                self.metrics.observe_gpu_health(self.node_name, False)
            return decisions
        from ..kube.store import ApiError
        for cluster in self.client.list(RayCluster):
            ann = cluster.metadata.annotations or {}
            if ann.get(AMD_AUTOSCALER_ANNOTATION, "").lower() != "true":
                continue
            if cluster.spec.suspend:
                # a suspended cluster has no pods; node telemetry cannot
                # be attributed to it and scaling it would fight resume
                continue
            try:
                decisions.extend(self._evaluate_cluster(cluster, summary))
            except ApiError:
                # optimistic-concurrency loss against the operator/users —
                # next tick re-evaluates from fresh state
                continue
        return decisions

    def _evaluate_cluster(self, cluster: RayCluster, summary: Dict) -> List[str]:
        decisions = []
        now = self.clock()
        p = self.policy
        namespace = cluster.metadata.namespace or "default"
        high = (summary.get("avg_utilization_pct", 0) > p.up_util_pct
                or summary.get("max_vram_used_fraction", 0) > p.up_hbm_fraction)
        low = (summary.get("avg_utilization_pct", 0) < p.down_util_pct
               and summary.get("max_vram_used_fraction", 0) < p.down_hbm_fraction)

        changed = False
        for i, group in enumerate(cluster.spec.worker_group_specs):
            if not self._is_gpu_group(group):
                continue
            key = (namespace, cluster.metadata.name, group.group_name)
            state = self._states.setdefault(key, GroupState())
            replicas = group.replicas if group.replicas is not None else (group.min_replicas or 0)
            min_r = group.min_replicas or 0
            max_r = group.max_replicas if group.max_replicas is not None else 2**31 - 1

            if high:
                if state.high_since is None:
                    state.high_since = now
            else:
                state.high_since = None
            if low:
                if state.low_since is None:
                    state.low_since = now
            else:
                state.low_since = None
            in_cooldown = now - state.last_action_ts < p.cooldown_s

            if (high and not in_cooldown and replicas < max_r
                    and state.high_since is not None
                    and now - state.high_since >= p.up_stable_s):
                group.replicas = replicas + 1
                state.last_action_ts = now
                state.high_since = None
                changed = True
                decisions.append(
                    f"scale-up {cluster.metadata.name}/{group.group_name} "
                    f"{replicas}->{replicas + 1} "
                    f"(util={summary.get('avg_utilization_pct', 0):.0f}% "
                    f"hbm={summary.get('max_vram_used_fraction', 0):.2f})")
            elif (low and not in_cooldown and replicas > min_r
                    and state.low_since is not None
                    and now - state.low_since >= p.idle_timeout_s):
                victim = self._pick_victim(cluster, group)
                group.replicas = replicas - 1
                if victim:
                    wtd = set(group.scale_strategy.workers_to_delete or [])
                    wtd.add(victim)
                    group.scale_strategy.workers_to_delete = sorted(wtd)
                state.last_action_ts = now
                state.low_since = None
                changed = True
                decisions.append(
                    f"scale-down {cluster.metadata.name}/{group.group_name} "
                    f"{replicas}->{replicas - 1} victim={victim}")

        if changed:
            self.client.update(cluster)
            if self.recorder is not None:
                for d in decisions:
                    self.recorder.eventf(cluster, "Normal", "MI355XAutoscale", d)
        return decisions

    @staticmethod
    def _is_gpu_group(group) -> bool:
        try:
            return pod_gpu_count(group.template) > 0
        except (IndexError, AttributeError):
            return False

    def _pick_victim(self, cluster: RayCluster, group) -> Optional[str]:
        """Newest worker first (least likely to hold long-lived actors)."""
        namespace = cluster.metadata.namespace or "default"
        pods = self.client.list(
            k8s.Pod, namespace,
            association.cluster_group_pods_selector(cluster.metadata.name,
                                                    group.group_name))
        live = [p for p in pods if not p.metadata.deletion_timestamp]
        if not live:
            return None
        live.sort(key=lambda p: p.metadata.creation_timestamp or "", reverse=True)
        return live[0].metadata.name

    # ------------------------------------------------------------------
    def start(self, interval_s: float = 5.0) -> None:
        def loop():
            while not self._stop.is_set():
                try:
                    self.step()
                except Exception:
                    pass
                self._stop.wait(interval_s)
        self._thread = threading.Thread(target=loop, name="mi355x-autoscaler",
                                        daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
