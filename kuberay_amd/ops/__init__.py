"""Reconcilers (the controllers/ray analog): RayCluster, RayJob, RayService,
RayCronJob, NetworkPolicy, plus scale expectations."""
