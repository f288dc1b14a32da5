"""Reconcilers (the controllers/ray analog)."""
from .expectations import FakeScaleExpectations, RayClusterScaleExpectations  # noqa: F401
from .raycluster import RayClusterReconciler, RayClusterReconcilerOptions  # noqa: F401
from .raycronjob import RayCronJobReconciler  # noqa: F401
from .rayjob import RayJobReconciler  # noqa: F401
from .rayservice import RayServiceReconciler  # noqa: F401
