"""RayService incremental upgrade — Gateway API weighted traffic migration.

Reference: rayservice_controller.go :899-983 (Gateway), :985-1199 +
:1644-1843 (HTTPRoute weights + TargetCapacity stepping), :2308-2413
(rollback). Feature gate: RayServiceIncrementalUpgrade; requires
``spec.upgradeStrategy.type: NewClusterWithIncrementalUpgrade`` plus
``clusterUpgradeOptions`` (gatewayClassName, stepSizePercent,
intervalSeconds, maxSurgePercent).

Flow, once the pending cluster's serve apps are RUNNING:
  1. per-cluster serve services exist for active + pending,
  2. a Gateway (``{svc}-gateway``) and an HTTPRoute (``{svc}-route``) carry
     weighted backendRefs [active: 100-w, pending: w],
  3. every intervalSeconds, w += stepSizePercent; the pending cluster's
     serve target_capacity follows w (capped by maxSurgePercent headroom),
     mirrored into status (targetCapacity / trafficRoutedPercent /
     lastTrafficMigratedTime),
  4. at w == 100 the service promotes (head/serve services repoint, old
     cluster GC'd after the deletion delay),
  5. rollback: if the goal hash returns to the active cluster's hash
     mid-upgrade, the pending cluster is dropped and weights reset.
"""
from __future__ import annotations

import time
from typing import Any, Dict, Optional

from ..kube import objects as k8s
from ..kube.store import AlreadyExistsError, NotFoundError, now_iso
from ..models import RayCluster, RayService
from ..models.rayservice import RayServiceConditionType as Cond
from ..utils import constants as C
from ..utils import names


def gateway_name(svc: RayService) -> str:
    return names.check_name_63(f"{svc.metadata.name}-gateway")


def route_name(svc: RayService) -> str:
    return names.check_name_63(f"{svc.metadata.name}-route")


def per_cluster_serve_service_name(cluster_name: str) -> str:
    return names.serve_service_name(cluster_name)


def upgrade_options(svc: RayService):
    us = svc.spec.upgrade_strategy
    return us.cluster_upgrade_options if us else None


def _condition_true(conditions, cond_type: str) -> bool:
    return any(c.get("type") == cond_type and c.get("status") == "True"
               for c in conditions or [])


def is_gateway_ready(gw: Optional[Dict[str, Any]]) -> bool:
    """util.go:874-883 — Gateway must be Accepted AND Programmed."""
    if gw is None:
        return False
    conditions = (gw.get("status") or {}).get("conditions")
    return (_condition_true(conditions, "Accepted")
            and _condition_true(conditions, "Programmed"))


def is_http_route_ready(gw: Optional[Dict[str, Any]],
                        route: Optional[Dict[str, Any]]) -> bool:
    """util.go:896-915 — the route's parent-status entry for OUR Gateway
    must be Accepted with ResolvedRefs before traffic weights advance."""
    if gw is None or route is None:
        return False
    gw_name = (gw.get("metadata") or {}).get("name")
    gw_ns = (gw.get("metadata") or {}).get("namespace")
    for parent in (route.get("status") or {}).get("parents") or []:
        ref = parent.get("parentRef") or {}
        if ref.get("name") != gw_name:
            continue
        if ref.get("namespace") and ref["namespace"] != gw_ns:
            continue
        if (_condition_true(parent.get("conditions"), "Accepted")
                and _condition_true(parent.get("conditions"), "ResolvedRefs")):
            return True
    return False


# Gateway/HTTPRoute have no typed model; the shared raw-object seam
# covers both backends (kube/client.py)
from ..kube.client import RawObjectClient as _RawObjects  # noqa: E402


class IncrementalUpgrader:
    """Composable helper driven by RayServiceReconciler."""

    def __init__(self, client, recorder, clock=time.monotonic):
        self.client = client
        self.recorder = recorder
        self.clock = clock
        self.raw = _RawObjects(client)

    # ------------------------------------------------------------------
    def ensure_gateway_infra(self, svc: RayService, active: RayCluster,
                             pending: RayCluster) -> None:
        """Create per-cluster serve services + Gateway + HTTPRoute."""
        from ..common import service as servicelib
        namespace = svc.metadata.namespace or "default"
        opts = upgrade_options(svc)

        for cluster in (active, pending):
            serve = servicelib.build_serve_service(cluster, cluster,
                                                   is_rayservice=False)
            serve.metadata.owner_references = [k8s.owner_reference_for(svc)]
            if self.client.try_get(k8s.Service, namespace,
                                   serve.metadata.name) is None:
                try:
                    self.client.create(serve)
                except AlreadyExistsError:
                    pass

        gw = {
            "apiVersion": "gateway.networking.k8s.io/v1",
            "kind": "Gateway",
            "metadata": {"name": gateway_name(svc), "namespace": namespace,
                         "ownerReferences": [k8s.owner_reference_for(svc).to_dict()]},
            "spec": {
                "gatewayClassName": opts.gateway_class_name if opts else "istio",
                "listeners": [{"name": "http", "protocol": "HTTP", "port": 80,
                               "allowedRoutes": {"namespaces": {"from": "Same"}}}],
            },
        }
        if self.raw.try_get("Gateway", namespace, gateway_name(svc)) is None:
            try:
                self.raw.create(gw)
            except AlreadyExistsError:
                pass
        if self.raw.try_get("HTTPRoute", namespace, route_name(svc)) is None:
            try:
                self.raw.create(self._route(svc, active.metadata.name,
                                            pending.metadata.name, 0))
            except AlreadyExistsError:
                pass

    def _route(self, svc: RayService, active_name: str, pending_name: str,
               pending_weight: int) -> Dict[str, Any]:
        namespace = svc.metadata.namespace or "default"
        return {
            "apiVersion": "gateway.networking.k8s.io/v1",
            "kind": "HTTPRoute",
            "metadata": {"name": route_name(svc), "namespace": namespace,
                         "ownerReferences": [k8s.owner_reference_for(svc).to_dict()]},
            "spec": {
                "parentRefs": [{"name": gateway_name(svc)}],
                "rules": [{
                    "backendRefs": [
                        {"name": per_cluster_serve_service_name(active_name),
                         "port": C.DEFAULT_SERVING_PORT,
                         "weight": 100 - pending_weight},
                        {"name": per_cluster_serve_service_name(pending_name),
                         "port": C.DEFAULT_SERVING_PORT,
                         "weight": pending_weight},
                    ],
                }],
            },
        }

    # ------------------------------------------------------------------
    def step_traffic(self, svc: RayService, active: RayCluster,
                     pending: RayCluster) -> bool:
        """Advance the weighted migration. Returns True when the pending
        cluster carries 100% and the service should promote."""
        namespace = svc.metadata.namespace or "default"
        gw = self.raw.try_get("Gateway", namespace, gateway_name(svc))
        route = self.raw.try_get("HTTPRoute", namespace, route_name(svc))
        if not is_gateway_ready(gw) or not is_http_route_ready(gw, route):
            # hold until the gateway controller accepts + programs the route
            # (rayservice_controller.go:1657-1667)
            return False

        opts = upgrade_options(svc)
        step = (opts.step_size_percent if opts and opts.step_size_percent
                else 25)
        interval = (opts.interval_seconds if opts and opts.interval_seconds
                    is not None else 30)
        pstatus = svc.status.pending_service_status
        current = pstatus.traffic_routed_percent or 0

        last = pstatus.last_traffic_migrated_time
        if last is not None and current > 0:
            import calendar
            try:
                t_last = calendar.timegm(time.strptime(last, "%Y-%m-%dT%H:%M:%SZ"))
                if time.time() - t_last < interval:
                    return False  # hold until the interval elapses
            except ValueError:
                pass

        new_weight = min(100, current + step)
        # TargetCapacity follows traffic, bounded by maxSurge headroom
        max_surge = (opts.max_surge_percent if opts and opts.max_surge_percent
                     is not None else 100)
        pstatus.target_capacity = min(100, max(new_weight,
                                               min(current + max_surge, 100)))
        pstatus.traffic_routed_percent = new_weight
        pstatus.last_traffic_migrated_time = now_iso()
        svc.status.active_service_status.traffic_routed_percent = 100 - new_weight
        svc.status.active_service_status.target_capacity = 100 - new_weight \
            if new_weight == 100 else 100

        try:
            self.raw.patch(
                "HTTPRoute", svc.metadata.namespace or "default",
                route_name(svc),
                {"spec": self._route(svc, active.metadata.name,
                                     pending.metadata.name,
                                     new_weight)["spec"]})
        except NotFoundError:
            pass
        self.recorder.eventf(svc, "Normal", "TrafficMigrated",
                             "Routed %d%% of traffic to RayCluster %s",
                             new_weight, pending.metadata.name)
        return new_weight >= 100

    # ------------------------------------------------------------------
    def rollback(self, svc: RayService, pending: Optional[RayCluster]) -> None:
        """rayservice_controller.go:2308 — drop the pending cluster, restore
        100% of traffic to the active one."""
        namespace = svc.metadata.namespace or "default"
        if pending is not None:
            try:
                self.client.delete(pending)
            except NotFoundError:
                pass
        active_name = svc.status.active_service_status.ray_cluster_name
        if active_name:
            try:
                self.raw.patch(
                    "HTTPRoute", namespace, route_name(svc),
                    {"spec": self._route(svc, active_name, active_name, 0)["spec"]})
            except NotFoundError:
                pass
        pstatus = svc.status.pending_service_status
        pstatus.ray_cluster_name = None
        pstatus.traffic_routed_percent = None
        pstatus.target_capacity = None
        svc.status.conditions = _set_condition(
            svc.status.conditions, Cond.ROLLBACK_IN_PROGRESS, "False",
            "RollbackComplete")
        svc.status.conditions = _set_condition(
            svc.status.conditions, Cond.UPGRADE_IN_PROGRESS, "False",
            "RolledBack")
        self.recorder.eventf(svc, "Normal", "UpgradeRolledBack",
                             "Incremental upgrade rolled back to %s", active_name)

    def cleanup(self, svc: RayService) -> None:
        """Remove Gateway/HTTPRoute after promotion completes."""
        namespace = svc.metadata.namespace or "default"
        for kind, name in (("HTTPRoute", route_name(svc)),
                           ("Gateway", gateway_name(svc))):
            try:
                self.raw.delete(kind, namespace, name)
            except NotFoundError:
                pass


def _set_condition(conditions, cond_type, status, reason, message=""):
    from .raycluster import set_condition
    return set_condition(conditions, cond_type, status, reason, message)
