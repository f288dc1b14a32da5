"""Kubernetes machinery: typed objects, client seam, in-memory apiserver
(envtest analog), workqueue, controller runtime, simulated kubelet.

Submodule imports are lazy to break the models<->client import cycle.
"""
_LAZY = {
    "InMemoryClient": ".client", "KubeClient": ".client",
    "Controller": ".controller", "Manager": ".controller",
    "Reconciler": ".controller", "Request": ".controller", "Result": ".controller",
    "EventRecorder": ".events", "NullRecorder": ".events", "StoreRecorder": ".events",
    "SimKubelet": ".kubelet",
    "AlreadyExistsError": ".store", "ApiError": ".store", "ConflictError": ".store",
    "InMemoryApiServer": ".store", "NotFoundError": ".store",
}


def __getattr__(name):
    if name in _LAZY:
        import importlib
        mod = importlib.import_module(_LAZY[name], __name__)
        return getattr(mod, name)
    raise AttributeError(name)
