"""Property tests for the apiserver patch machinery (RFC 6902 json-patch,
strategic-merge-patch, RFC 7386 merge-patch) — hypothesis-driven analogs
of apimachinery's patch fuzzers."""
import copy

import pytest

pytest.importorskip("hypothesis")

from hypothesis import given, settings, strategies as st

from kuberay_amd.kube.httpserver import apply_json_patch, strategic_merge

KEYS = st.sampled_from(["a", "b", "name", "spec", "items", "replicas"])
SCALARS = st.one_of(st.integers(-5, 5), st.text(max_size=4), st.booleans(),
                    st.none())


def docs(depth=3):
    if depth == 0:
        return SCALARS
    return st.one_of(
        SCALARS,
        st.lists(docs(depth - 1), max_size=3),
        st.dictionaries(KEYS, docs(depth - 1), max_size=3))


DICT_DOCS = st.dictionaries(KEYS, docs(), min_size=0, max_size=4)


class TestJsonPatch:
    @settings(max_examples=80, deadline=None)
    @given(doc=DICT_DOCS, key=KEYS, value=docs())
    def test_add_then_remove_roundtrips(self, doc, key, value):
        before = copy.deepcopy(doc)
        added = apply_json_patch(doc, [{"op": "add", "path": f"/{key}",
                                        "value": value}])
        assert added[key] == value
        if key in before:
            # removed slot previously occupied: add acts as replace, remove
            # deletes — original value is gone (spec semantics), but the
            # patch input must not be mutated
            assert doc == before
        else:
            removed = apply_json_patch(added, [{"op": "remove",
                                                "path": f"/{key}"}])
            assert removed == before

    @settings(max_examples=80, deadline=None)
    @given(doc=DICT_DOCS)
    def test_copy_then_test_passes(self, doc):
        if not doc:
            return
        key = sorted(doc)[0]
        out = apply_json_patch(doc, [
            {"op": "copy", "from": f"/{key}", "path": "/copied"},
            {"op": "test", "path": "/copied", "value": doc[key]}])
        assert out["copied"] == doc[key]

    @settings(max_examples=80, deadline=None)
    @given(doc=DICT_DOCS)
    def test_input_never_mutated(self, doc):
        snapshot = copy.deepcopy(doc)
        try:
            apply_json_patch(doc, [{"op": "add", "path": "/x", "value": 1},
                                   {"op": "remove", "path": "/nope"}])
        except Exception:
            pass
        assert doc == snapshot


class TestStrategicMerge:
    @settings(max_examples=80, deadline=None)
    @given(doc=DICT_DOCS, patch=DICT_DOCS)
    def test_idempotent(self, doc, patch):
        once = strategic_merge(copy.deepcopy(doc), patch)
        twice = strategic_merge(copy.deepcopy(once), patch)
        assert once == twice

    @settings(max_examples=80, deadline=None)
    @given(doc=DICT_DOCS, patch=st.dictionaries(KEYS, SCALARS, max_size=3))
    def test_scalar_leaves_win(self, doc, patch):
        out = strategic_merge(copy.deepcopy(doc), patch)
        for k, v in patch.items():
            if v is None:
                assert k not in out  # null deletes, RFC 7386 style
            else:
                assert out[k] == v

    def test_merge_key_containers(self):
        doc = {"containers": [{"name": "a", "image": "x"},
                              {"name": "b", "image": "y"}]}
        out = strategic_merge(doc, {"containers": [{"name": "b",
                                                    "image": "z"}]})
        assert [c["image"] for c in out["containers"]] == ["x", "z"]

    def test_patch_delete_directive(self):
        doc = {"containers": [{"name": "a"}, {"name": "b"}]}
        out = strategic_merge(doc, {"containers": [
            {"name": "a", "$patch": "delete"}]})
        assert [c["name"] for c in out["containers"]] == ["b"]


class TestWorkqueueCoalescing:
    """Property tests for the delayed-entry coalescing fix: no lost keys,
    and memory bounded by DISTINCT keys regardless of requeue volume."""

    @settings(max_examples=25, deadline=None)
    @given(ops=st.lists(
        st.tuples(st.sampled_from(["k1", "k2", "k3"]),
                  st.floats(min_value=0.0005, max_value=0.004)),
        min_size=1, max_size=40))
    def test_every_key_delivered_and_heap_bounded(self, ops):
        from kuberay_amd.kube.workqueue import RateLimitingQueue
        q = RateLimitingQueue()
        keys = {k for k, _ in ops}
        for k, d in ops:
            q.add_after(k, d)
            # the RSS invariant: one live deadline per key, ever
            assert len(q._delayed_next) <= len(keys)
        delivered = set()
        import time as _t
        deadline = _t.monotonic() + 5
        while delivered != keys and _t.monotonic() < deadline:
            item = q.get(timeout=0.5)
            if item is None:
                continue
            delivered.add(item)
            q.done(item)
        assert delivered == keys
        # after full drain no delayed state remains
        assert len(q._delayed_next) == 0

    def test_requeue_while_processing_redelivers(self):
        from kuberay_amd.kube.workqueue import RateLimitingQueue
        q = RateLimitingQueue()
        q.add("x")
        item = q.get(timeout=1)
        assert item == "x"
        q.add_after("x", 0.001)   # fires while x is processing → dirty
        import time as _t
        _t.sleep(0.05)
        q.get(timeout=0.05)       # drains delayed heap into dirty set
        q.done("x")
        assert q.get(timeout=1) == "x"  # dirty redelivery


class TestWatchResumeExactness:
    """Resuming from ANY rv inside the retained window yields exactly the
    suffix of events after that rv, in order (the two round-2 watch bugs —
    replay-before-register and delete-rv-reuse — were both violations of
    this property)."""

    @settings(max_examples=30, deadline=None)
    @given(ops=st.lists(
        st.tuples(st.sampled_from(["create", "update", "delete"]),
                  st.sampled_from(["o1", "o2", "o3", "o4"])),
        min_size=1, max_size=60))
    def test_suffix_exactness(self, ops):
        from kuberay_amd.kube.store import InMemoryApiServer
        server = InMemoryApiServer()
        log = []  # (rv_after_op, etype, name)
        live = set()
        for op, name in ops:
            if op == "create" and name not in live:
                out = server.create({"kind": "ConfigMap", "apiVersion": "v1",
                                     "metadata": {"name": name}, "data": {}})
                live.add(name)
                log.append((int(out["metadata"]["resourceVersion"]),
                            "ADDED", name))
            elif op == "update" and name in live:
                cur = server.get("ConfigMap", "default", name)
                cur["data"] = {"n": str(len(log))}
                out = server.update(cur)
                log.append((int(out["metadata"]["resourceVersion"]),
                            "MODIFIED", name))
            elif op == "delete" and name in live:
                server.delete("ConfigMap", "default", name)
                live.discard(name)
                log.append((server.current_rv, "DELETED", name))
        # resume from every recorded rv plus 0: suffix must match the log
        checkpoints = [0] + [rv for rv, _, _ in log]
        for rv in checkpoints:
            got = server.events_since(rv, kinds={"ConfigMap"})
            assert got is not None
            expect = [(etype, name) for (erv, etype, name) in log
                      if erv > rv]
            assert [(e, o["metadata"]["name"]) for e, o in got] == expect, rv

    def test_rv_strictly_increases_per_event(self):
        from kuberay_amd.kube.store import InMemoryApiServer
        server = InMemoryApiServer()
        server.create({"kind": "ConfigMap", "apiVersion": "v1",
                       "metadata": {"name": "a"}})
        rv1 = server.current_rv
        server.delete("ConfigMap", "default", "a")
        assert server.current_rv > rv1  # DELETED must own a fresh rv


class TestNamingAndHashProperties:
    @settings(max_examples=60, deadline=None)
    @given(name=st.text(alphabet=st.characters(
        whitelist_categories=("Ll", "Nd"), whitelist_characters="-."),
        min_size=1, max_size=120))
    def test_check_name_bounds_and_determinism(self, name):
        from kuberay_amd.utils.names import check_name, check_name_63
        out = check_name(name)
        assert len(out) <= 63 and check_name(name) == out
        assert len(check_name_63(name)) <= 63

    @settings(max_examples=40, deadline=None)
    @given(replicas=st.integers(0, 50), to_delete=st.lists(
        st.sampled_from(["p1", "p2", "p3"]), max_size=3))
    def test_spec_hash_ignores_scale_churn(self, replicas, to_delete):
        """Upgrade detection must not fire on autoscaler activity:
        replicas and workersToDelete never affect the spec hash
        (reference GenerateHashWithoutReplicasAndWorkersToDelete)."""
        from kuberay_amd.testing import simple_raycluster
        from kuberay_amd.utils.hashing import (
            hash_without_replicas_and_workers_to_delete)
        base = simple_raycluster("h", workers=1)
        h0 = hash_without_replicas_and_workers_to_delete(base.spec)
        c = simple_raycluster("h", workers=1)
        g = c.spec.worker_group_specs[0]
        g.replicas = replicas
        g.max_replicas = max(replicas, g.max_replicas or 0)
        if to_delete:
            g.scale_strategy.workers_to_delete = to_delete
        assert hash_without_replicas_and_workers_to_delete(c.spec) == h0
        # but a real spec change (image) must change it
        c.spec.worker_group_specs[0].template.spec.containers[0].image = \
            "other:tag"
        assert hash_without_replicas_and_workers_to_delete(c.spec) != h0


class TestClientDifferential:
    """The same verb sequence through InMemoryClient and through
    RestClient-over-the-facade must converge to the same object tree
    (modulo resourceVersions, which depend on server-internal counters)."""

    @settings(max_examples=15, deadline=None)
    @given(ops=st.lists(st.tuples(
        st.sampled_from(["create", "update", "patch", "delete"]),
        st.sampled_from(["c1", "c2", "c3"]),
        st.dictionaries(st.sampled_from(["x", "y", "z"]),
                        st.text(max_size=3), max_size=2)),
        min_size=1, max_size=25))
    def test_configmap_sequences_converge(self, ops):
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.kube.httpserver import KubeApiFacade
        from kuberay_amd.kube.objects import ConfigMap
        from kuberay_amd.kube.rest import RestClient
        from kuberay_amd.kube.store import (AlreadyExistsError, ApiError,
                                            NotFoundError)
        mem = InMemoryClient()
        facade = KubeApiFacade().start()
        try:
            rest = RestClient(base_url=facade.url)
            for client in (mem, rest):
                for op, name, data in ops:
                    try:
                        if op == "create":
                            client.create(ConfigMap.from_dict(
                                {"kind": "ConfigMap", "apiVersion": "v1",
                                 "metadata": {"name": name}, "data": data}))
                        elif op == "update":
                            cur = client.get(ConfigMap, "default", name)
                            cur.data = data
                            client.update(cur)
                        elif op == "patch":
                            client.patch(ConfigMap, "default", name,
                                         {"data": data})
                        elif op == "delete":
                            client.delete(ConfigMap, "default", name)
                    except (NotFoundError, AlreadyExistsError, ApiError):
                        pass
            def snapshot(client):
                out = {}
                for cm in client.list(ConfigMap, "default"):
                    out[cm.metadata.name] = cm.data
                return out
            assert snapshot(mem) == snapshot(rest)
        finally:
            facade.stop()


class TestOwnerGCCascade:
    """Random ownership forests: deleting a root garbage-collects exactly
    its transitive dependents — nothing orphaned, nothing over-deleted
    (kube garbage-collector background-cascade semantics)."""

    @settings(max_examples=25, deadline=None)
    @given(parents=st.lists(st.integers(0, 9), min_size=10, max_size=10),
           root=st.integers(0, 9))
    def test_cascade_exactness(self, parents, root):
        from kuberay_amd.kube.store import InMemoryApiServer
        server = InMemoryApiServer()
        uids = {}
        # node i is owned by parents[i] when parents[i] < i (forest, no
        # cycles); otherwise it is a root
        for i in range(10):
            refs = []
            p = parents[i]
            if p < i:
                refs = [{"apiVersion": "v1", "kind": "ConfigMap",
                         "name": f"n{p}", "uid": uids[p]}]
            out = server.create({
                "kind": "ConfigMap", "apiVersion": "v1",
                "metadata": {"name": f"n{i}",
                             "ownerReferences": refs or None}})
            uids[i] = out["metadata"]["uid"]
        reach = {root}
        changed = True
        while changed:
            changed = False
            for i in range(10):
                if parents[i] < i and parents[i] in reach and i not in reach:
                    reach.add(i)
                    changed = True
        server.delete("ConfigMap", "default", f"n{root}")
        left = {o["metadata"]["name"]
                for o in server.list("ConfigMap", "default")}
        assert left == {f"n{i}" for i in range(10) if i not in reach}


class TestQuantityRoundTrip:
    @settings(max_examples=80, deadline=None)
    @given(n=st.integers(0, 10**12),
           suffix=st.sampled_from(["", "m", "k", "M", "G", "Ki", "Mi", "Gi"]))
    def test_format_parse_roundtrip(self, n, suffix):
        from kuberay_amd.utils.quantity import format_quantity, parse_quantity
        v = parse_quantity(f"{n}{suffix}")
        assert parse_quantity(format_quantity(v)) == v


class TestStoreMergePatch:
    @settings(max_examples=60, deadline=None)
    @given(patch=DICT_DOCS)
    def test_store_merge_patch_idempotent_and_null_free(self, patch):
        from kuberay_amd.kube.store import InMemoryApiServer
        server = InMemoryApiServer()
        server.create({"kind": "ConfigMap", "apiVersion": "v1",
                       "metadata": {"name": "m"}, "data": {}})
        once = server.patch_merge("ConfigMap", "default", "m",
                                  {"data": dict(patch)})
        twice = server.patch_merge("ConfigMap", "default", "m",
                                   {"data": dict(patch)})
        assert once["data"] == twice["data"]

        def no_nulls(node):
            if isinstance(node, dict):
                return all(v is not None and no_nulls(v)
                           for v in node.values())
            return True
        assert no_nulls(once["data"])


class TestUnknownFieldPreservation:
    def test_exotic_and_future_fields_survive_round_trip(self):
        """The operator must never wipe fields it doesn't model when it
        writes a CR back (kubectl-applied topologySpreadConstraints,
        lifecycle hooks, future API fields)."""
        from kuberay_amd.models import RayCluster
        from kuberay_amd.testing import simple_raycluster
        d = simple_raycluster("u", workers=1).to_dict()
        ps = d["spec"]["headGroupSpec"]["template"]["spec"]
        ps["topologySpreadConstraints"] = [
            {"maxSkew": 1, "topologyKey": "zone",
             "whenUnsatisfiable": "DoNotSchedule"}]
        ps["containers"][0]["lifecycle"] = {
            "preStop": {"exec": {"command": ["/bin/sleep", "5"]}}}
        d["spec"]["workerGroupSpecs"][0]["template"]["metadata"] = {
            "annotations": {"custom.io/x": "1"}}
        d["spec"]["someFutureField"] = {"a": 1}
        rt = RayCluster.from_dict(d).to_dict()
        hs = rt["spec"]["headGroupSpec"]["template"]["spec"]
        assert hs["topologySpreadConstraints"][0]["topologyKey"] == "zone"
        assert hs["containers"][0]["lifecycle"]["preStop"]["exec"][
            "command"] == ["/bin/sleep", "5"]
        assert rt["spec"]["workerGroupSpecs"][0]["template"]["metadata"][
            "annotations"] == {"custom.io/x": "1"}
        assert rt["spec"]["someFutureField"] == {"a": 1}


class TestSnapshotWatchSemantics:
    def test_resume_below_restore_point_gets_410(self, tmp_path):
        """A watcher holding a pre-restart rv cannot be silently given a
        partial replay after snapshot restore — it must see Gone and
        re-list (events between its rv and the restore point are lost)."""
        from kuberay_amd.kube.snapshot import load_snapshot, save_snapshot
        from kuberay_amd.kube.store import InMemoryApiServer
        s1 = InMemoryApiServer()
        for i in range(5):
            s1.create({"kind": "ConfigMap", "apiVersion": "v1",
                       "metadata": {"name": f"c{i}"}})
        path = str(tmp_path / "state.jsonl")
        save_snapshot(s1, path)
        s2 = InMemoryApiServer()
        load_snapshot(s2, path)
        assert s2.events_since(1) is None          # pre-restart rv: 410
        assert s2.events_since(s2.current_rv) == []  # at restore point: ok
        out = s2.create({"kind": "ConfigMap", "apiVersion": "v1",
                         "metadata": {"name": "after"}})
        got = s2.events_since(s2.current_rv - 1)
        assert [o["metadata"]["name"] for _, o in got] == ["after"]
        assert int(out["metadata"]["resourceVersion"]) > 5  # rv monotone


class TestSelectorDifferential:
    @settings(max_examples=10, deadline=None)
    @given(labelings=st.lists(
        st.dictionaries(st.sampled_from(["app", "tier", "env"]),
                        st.sampled_from(["a", "b", "c"]), max_size=3),
        min_size=4, max_size=8),
        sel=st.dictionaries(st.sampled_from(["app", "tier"]),
                            st.sampled_from(["a", "b"]), min_size=1,
                            max_size=2))
    def test_label_selector_lists_match(self, labelings, sel):
        """Label-selector list results agree between the in-memory client
        and REST-over-facade for random labelings and selectors."""
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.kube.httpserver import KubeApiFacade
        from kuberay_amd.kube.objects import ConfigMap
        from kuberay_amd.kube.rest import RestClient
        mem = InMemoryClient()
        facade = KubeApiFacade().start()
        try:
            rest = RestClient(base_url=facade.url)
            for client in (mem, rest):
                for i, labels in enumerate(labelings):
                    client.create(ConfigMap.from_dict(
                        {"kind": "ConfigMap", "apiVersion": "v1",
                         "metadata": {"name": f"s{i}", "labels": labels}}))
            a = {c.metadata.name
                 for c in mem.list(ConfigMap, "default", sel)}
            b = {c.metadata.name
                 for c in rest.list(ConfigMap, "default", sel)}
            assert a == b
            expect = {f"s{i}" for i, lb in enumerate(labelings)
                      if all(lb.get(k) == v for k, v in sel.items())}
            assert a == expect
        finally:
            facade.stop()
