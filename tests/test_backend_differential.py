"""Differential property test: the native C++ backend must be
observationally identical to the python backend under arbitrary operation
sequences (hypothesis-driven)."""
import pytest

pytest.importorskip("hypothesis")

from hypothesis import given, settings, strategies as st

from kuberay_amd.kube.store import InMemoryApiServer, PyBackend
from kuberay_amd.kube.store import AlreadyExistsError, ApiError, ConflictError, NotFoundError

try:
    from kuberay_amd.kube.native import NativeBackend
except ImportError:
    pytest.skip("native engine not built", allow_module_level=True)


NAMES = ["a", "b", "c"]
KINDS = ["Pod", "RayCluster"]
LABEL_VALUES = ["x", "y"]

op_strategy = st.one_of(
    st.tuples(st.just("create"), st.sampled_from(KINDS), st.sampled_from(NAMES),
              st.sampled_from(LABEL_VALUES)),
    st.tuples(st.just("update_spec"), st.sampled_from(KINDS),
              st.sampled_from(NAMES), st.integers(0, 5)),
    st.tuples(st.just("patch_status"), st.sampled_from(KINDS),
              st.sampled_from(NAMES),
              st.sampled_from(["Running", "Failed", "Pending"])),
    st.tuples(st.just("delete"), st.sampled_from(KINDS), st.sampled_from(NAMES)),
    st.tuples(st.just("add_finalizer"), st.sampled_from(KINDS),
              st.sampled_from(NAMES)),
    st.tuples(st.just("clear_finalizers"), st.sampled_from(KINDS),
              st.sampled_from(NAMES)),
)


def apply_op(server: InMemoryApiServer, op):
    """Apply one op; return a comparable observation."""
    try:
        if op[0] == "create":
            _, kind, name, lv = op
            obj = {"kind": kind,
                   "metadata": {"name": name, "namespace": "d",
                                "labels": {"g": lv}},
                   "spec": {"containers": [{"name": "ray"}]} if kind == "Pod"
                   else {"v": 0}}
            out = server.create(obj)
            return ("created", out["metadata"]["generation"])
        if op[0] == "update_spec":
            _, kind, name, v = op
            cur = server.get(kind, "d", name)
            cur["spec"] = dict(cur.get("spec") or {}, v=v)
            out = server.update(cur)
            return ("updated", out["metadata"]["generation"], out["spec"].get("v"))
        if op[0] == "patch_status":
            _, kind, name, phase = op
            out = server.patch_merge(kind, "d", name,
                                     {"status": {"phase": phase}},
                                     subresource="status")
            return ("patched", out["status"].get("phase"),
                    out["metadata"]["generation"])
        if op[0] == "delete":
            _, kind, name = op
            server.delete(kind, "d", name)
            return ("deleted",)
        if op[0] == "add_finalizer":
            _, kind, name = op
            cur = server.get(kind, "d", name)
            cur["metadata"]["finalizers"] = ["f"]
            server.update(cur)
            return ("finalized",)
        if op[0] == "clear_finalizers":
            _, kind, name = op
            cur = server.get(kind, "d", name)
            cur["metadata"]["finalizers"] = []
            server.update(cur)
            return ("unfinalized",)
    except NotFoundError:
        return ("not_found",)
    except AlreadyExistsError:
        return ("exists",)
    except ConflictError:
        return ("conflict",)
    except ApiError as e:
        return ("api_error", e.code)
    raise AssertionError(f"unknown op {op}")


def observe(server: InMemoryApiServer):
    """Full observable state (uids/rvs/timestamps excluded — they are
    legitimately run-specific; structure, counts, views are not)."""
    state = {}
    for kind in KINDS:
        objs = server.list(kind, "d")
        state[kind] = [
            (o["metadata"]["name"],
             sorted((o["metadata"].get("labels") or {}).items()),
             o["metadata"].get("generation"),
             bool(o["metadata"].get("deletionTimestamp")),
             tuple(o["metadata"].get("finalizers") or []),
             (o.get("status") or {}).get("phase"),
             (o.get("spec") or {}).get("v"))
            for o in objs
        ]
    state["views"] = [
        (v.name, v.phase, sorted(v.labels.items()))
        for v in server.list_pod_views("d")
    ]
    state["counts"] = {k: server.count(k) for k in KINDS}
    state["selector"] = [
        o["metadata"]["name"] for o in server.list("Pod", "d", {"g": "x"})]
    return state


@settings(max_examples=200, deadline=None)
@given(st.lists(op_strategy, min_size=1, max_size=30))
def test_native_backend_matches_python_backend(ops):
    py = InMemoryApiServer(backend=PyBackend())
    native = InMemoryApiServer(backend=NativeBackend())
    for op in ops:
        r1 = apply_op(py, op)
        r2 = apply_op(native, op)
        assert r1 == r2, (op, r1, r2)
    assert observe(py) == observe(native)
