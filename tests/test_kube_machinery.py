"""In-memory apiserver / workqueue / expectations / cron unit tests."""
import threading
import time

import pytest

from kuberay_amd.kube.store import (
    AlreadyExistsError,
    ConflictError,
    InMemoryApiServer,
    NotFoundError,
)
from kuberay_amd.kube.workqueue import RateLimitingQueue
from kuberay_amd.ops.expectations import FakeScaleExpectations, RayClusterScaleExpectations
from kuberay_amd.utils.cron import parse_cron
import datetime as dt


def mk(kind, name, namespace="default", labels=None, **extra):
    obj = {"apiVersion": "v1", "kind": kind,
           "metadata": {"name": name, "namespace": namespace}}
    if labels:
        obj["metadata"]["labels"] = labels
    obj.update(extra)
    return obj


class TestStore:
    def test_create_get_uid_rv(self):
        s = InMemoryApiServer()
        out = s.create(mk("Pod", "p1"))
        assert out["metadata"]["uid"]
        assert out["metadata"]["resourceVersion"] == "1"
        assert s.get("Pod", "default", "p1")["metadata"]["name"] == "p1"

    def test_duplicate_create_conflicts(self):
        s = InMemoryApiServer()
        s.create(mk("Pod", "p1"))
        with pytest.raises(AlreadyExistsError):
            s.create(mk("Pod", "p1"))

    def test_generate_name(self):
        s = InMemoryApiServer()
        out = s.create({"kind": "Pod", "metadata": {"generateName": "x-"}})
        assert out["metadata"]["name"].startswith("x-")

    def test_optimistic_concurrency(self):
        s = InMemoryApiServer()
        a = s.create(mk("Pod", "p1"))
        b = s.get("Pod", "default", "p1")
        a["spec"] = {"x": 1}
        s.update(a)
        b["spec"] = {"x": 2}
        with pytest.raises(ConflictError):
            s.update(b)

    def test_generation_bumps_on_spec_change_only(self):
        s = InMemoryApiServer()
        obj = s.create(mk("RayCluster", "c1", spec={"a": 1}))
        assert obj["metadata"]["generation"] == 1
        obj["spec"] = {"a": 2}
        obj = s.update(obj)
        assert obj["metadata"]["generation"] == 2
        obj["status"] = {"s": 1}
        obj2 = s.update(obj, subresource="status")
        assert obj2["metadata"]["generation"] == 2
        assert obj2["status"] == {"s": 1}

    def test_status_subresource_does_not_touch_spec(self):
        s = InMemoryApiServer()
        obj = s.create(mk("RayCluster", "c1", spec={"a": 1}))
        stale = dict(obj)
        stale["spec"] = {"a": 999}
        stale["status"] = {"ok": True}
        out = s.update(stale, subresource="status")
        assert out["spec"] == {"a": 1}
        assert out["status"] == {"ok": True}

    def test_label_selector_list(self):
        s = InMemoryApiServer()
        s.create(mk("Pod", "a", labels={"g": "1", "t": "w"}))
        s.create(mk("Pod", "b", labels={"g": "2", "t": "w"}))
        s.create(mk("Pod", "c", labels={"g": "1", "t": "h"}))
        assert len(s.list("Pod", label_selector={"g": "1"})) == 2
        assert len(s.list("Pod", label_selector={"g": "1", "t": "w"})) == 1
        assert len(s.list("Pod")) == 3

    def test_label_index_updated_on_update(self):
        s = InMemoryApiServer()
        obj = s.create(mk("Pod", "a", labels={"g": "1"}))
        obj["metadata"]["labels"] = {"g": "2"}
        s.update(obj)
        assert len(s.list("Pod", label_selector={"g": "1"})) == 0
        assert len(s.list("Pod", label_selector={"g": "2"})) == 1

    def test_finalizer_blocks_deletion(self):
        s = InMemoryApiServer()
        obj = mk("RayCluster", "c1")
        obj["metadata"]["finalizers"] = ["f1"]
        s.create(obj)
        s.delete("RayCluster", "default", "c1")
        cur = s.get("RayCluster", "default", "c1")
        assert cur["metadata"]["deletionTimestamp"]
        cur["metadata"]["finalizers"] = []
        s.update(cur)
        assert s.try_get("RayCluster", "default", "c1") is None

    def test_owner_gc_cascade(self):
        s = InMemoryApiServer()
        owner = s.create(mk("RayCluster", "c1"))
        child = mk("Pod", "p1")
        child["metadata"]["ownerReferences"] = [{
            "kind": "RayCluster", "name": "c1", "uid": owner["metadata"]["uid"]}]
        s.create(child)
        s.delete("RayCluster", "default", "c1")
        assert s.try_get("Pod", "default", "p1") is None

    def test_watch_events(self):
        s = InMemoryApiServer()
        w = s.watch({"Pod"})
        s.create(mk("Pod", "p1"))
        s.create(mk("Service", "s1"))
        ev = w.next(timeout=1)
        assert ev[0] == "ADDED" and ev[1]["metadata"]["name"] == "p1"
        assert w.next(timeout=0.1) is None  # Service filtered out
        w.stop()

    def test_patch_merge(self):
        s = InMemoryApiServer()
        s.create(mk("Pod", "p1", labels={"a": "1"}))
        s.patch_merge("Pod", "default", "p1", {"metadata": {"labels": {"b": "2"}}})
        assert s.get("Pod", "default", "p1")["metadata"]["labels"] == {"a": "1", "b": "2"}


class TestWorkqueue:
    def test_dedup_while_queued(self):
        q = RateLimitingQueue()
        q.add("a"); q.add("a"); q.add("b")
        assert len(q) == 2

    def test_dirty_requeue_while_processing(self):
        q = RateLimitingQueue()
        q.add("a")
        item = q.get()
        q.add("a")  # re-added while processing -> dirty
        q.done(item)
        assert q.get(timeout=0.5) == "a"

    def test_add_after(self):
        q = RateLimitingQueue()
        q.add_after("a", 0.1)
        assert q.get(timeout=0.02) is None
        assert q.get(timeout=1.0) == "a"

    def test_add_after_coalesces_per_key(self):
        """A key requeued every reconcile must hold ONE live heap entry —
        an unbounded delayed heap was the round-1 soak's RSS growth."""
        q = RateLimitingQueue()
        for _ in range(10_000):
            q.add_after("k", 300.0)
        assert len(q._delayed) == 1
        assert len(q) == 1

    def test_add_after_earlier_deadline_wins(self):
        q = RateLimitingQueue()
        q.add_after("k", 300.0)
        q.add_after("k", 0.05)  # sooner: must fire at the sooner deadline
        assert q.get(timeout=1.0) == "k"
        q.done("k")
        # the stale 300 s entry must not fire the key a second time
        assert q.get(timeout=0.3) is None

    def test_rate_limited_backoff_grows(self):
        q = RateLimitingQueue(base_delay=0.01, max_delay=1.0)
        t0 = time.monotonic()
        q.add_rate_limited("a")
        assert q.get(timeout=2.0) == "a"
        q.done("a")
        q.add_rate_limited("a")
        assert q.get(timeout=2.0) == "a"
        assert time.monotonic() - t0 >= 0.02


class TestExpectations:
    def test_create_expectation_blocks_until_observed(self):
        s = InMemoryApiServer()
        e = RayClusterScaleExpectations()
        e.expect_create_pod("default", "c1", "g1", "pod-x")
        assert not e.is_satisfied(s, "default", "c1", "g1")
        s.create(mk("Pod", "pod-x"))
        assert e.is_satisfied(s, "default", "c1", "g1")

    def test_delete_expectation(self):
        s = InMemoryApiServer()
        s.create(mk("Pod", "pod-x"))
        e = RayClusterScaleExpectations()
        e.expect_delete_pod("default", "c1", "g1", "pod-x")
        assert not e.is_satisfied(s, "default", "c1", "g1")
        s.delete("Pod", "default", "pod-x")
        assert e.is_satisfied(s, "default", "c1", "g1")

    def test_timeout_unblocks(self):
        s = InMemoryApiServer()
        e = RayClusterScaleExpectations(timeout_s=0.05)
        e.expect_create_pod("default", "c1", "g1", "never-created")
        time.sleep(0.1)
        assert e.is_satisfied(s, "default", "c1", "g1")

    def test_fake_always_satisfied(self):
        e = FakeScaleExpectations()
        e.expect_create_pod("d", "c", "g", "p")
        assert e.is_satisfied(None, "d", "c", "g")


class TestCron:
    def test_parse_basic(self):
        sched = parse_cron("*/5 * * * *")
        nxt = sched.next_after(dt.datetime(2026, 1, 1, 0, 1))
        assert nxt == dt.datetime(2026, 1, 1, 0, 5)

    def test_daily(self):
        sched = parse_cron("30 2 * * *")
        nxt = sched.next_after(dt.datetime(2026, 1, 1, 3, 0))
        assert nxt == dt.datetime(2026, 1, 2, 2, 30)

    def test_macro(self):
        sched = parse_cron("@hourly")
        nxt = sched.next_after(dt.datetime(2026, 1, 1, 0, 30))
        assert nxt == dt.datetime(2026, 1, 1, 1, 0)

    def test_invalid(self):
        with pytest.raises(ValueError):
            parse_cron("61 * * * *")
        with pytest.raises(ValueError):
            parse_cron("* * *")

    def test_dow(self):
        sched = parse_cron("0 0 * * 1")  # Mondays
        nxt = sched.next_after(dt.datetime(2026, 1, 1))  # Thursday
        assert nxt == dt.datetime(2026, 1, 5)


class TestSnapshot:
    def test_save_and_restore_resumes_state(self, tmp_path):
        from kuberay_amd.kube.snapshot import load_snapshot, save_snapshot
        from kuberay_amd.testing import ControlPlane, simple_raycluster
        from kuberay_amd.models import RayCluster

        path = str(tmp_path / "state.jsonl")
        cp = ControlPlane(kubelet_delay=0.0, poll_seconds=0.05)
        cp.start()
        try:
            cp.client.create(simple_raycluster("snap", workers=1))
            assert cp.wait_cluster_state("default", "snap", "ready")
            n = save_snapshot(cp.server, path)
            assert n >= 3  # CR + head svc + pods...
        finally:
            cp.stop()

        # "restart" the operator: fresh control plane restored from disk
        cp2 = ControlPlane(kubelet_delay=0.0, poll_seconds=0.05)
        restored = load_snapshot(cp2.server, path)
        assert restored == n
        cp2.start()
        try:
            rc = cp2.client.get(RayCluster, "default", "snap")
            assert rc.status.state == "ready"
            assert rc.metadata.uid  # identity preserved
            # the resumed control plane still reconciles: scale up works
            rc.spec.worker_group_specs[0].replicas = 2
            cp2.client.update(rc)
            assert cp2.wait_for(
                lambda: cp2.client.get(RayCluster, "default", "snap")
                .status.ready_worker_replicas == 2, timeout=15)
        finally:
            cp2.stop()

    def test_snapshot_loop(self, tmp_path):
        import os
        from kuberay_amd.kube.snapshot import SnapshotLoop
        from kuberay_amd.kube.store import InMemoryApiServer
        server = InMemoryApiServer()
        server.create({"kind": "ConfigMap", "metadata": {"name": "x"}})
        loop = SnapshotLoop(server, str(tmp_path / "s.jsonl"), interval_s=0.05)
        loop.start()
        import time
        time.sleep(0.15)
        loop.stop()
        assert os.path.exists(tmp_path / "s.jsonl")


class TestLeaderElection:
    def test_single_candidate_acquires(self):
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.kube.leaderelection import LeaderElector
        client = InMemoryClient()
        started = []
        e = LeaderElector(client, identity="a",
                          on_started_leading=lambda: started.append("a"))
        assert e.try_acquire_or_renew() is True
        assert started == ["a"]
        # renewal keeps leadership
        assert e.try_acquire_or_renew() is True

    def test_second_candidate_waits_then_takes_over(self):
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.kube.leaderelection import LeaderElector
        client = InMemoryClient()
        a = LeaderElector(client, identity="a", lease_duration_s=0.2)
        b = LeaderElector(client, identity="b", lease_duration_s=0.2)
        assert a.try_acquire_or_renew() is True
        assert b.try_acquire_or_renew() is False
        time.sleep(1.1)  # let the lease expire (timestamps have 1s granularity)
        assert b.try_acquire_or_renew() is True
        # a notices it lost on its next round (lease now held by b)
        assert a.try_acquire_or_renew() is False
        assert a.is_leader is False

    def test_graceful_release_on_stop(self):
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.kube.leaderelection import LeaderElector
        client = InMemoryClient()
        stopped = []
        a = LeaderElector(client, identity="a", renew_period_s=0.05,
                          on_stopped_leading=lambda: stopped.append(1))
        a.start()
        time.sleep(0.2)
        assert a.is_leader
        a.stop()
        assert stopped == [1]
        b = LeaderElector(client, identity="b")
        assert b.try_acquire_or_renew() is True  # released lease is free


class TestLeaderElectionOverRest:
    """The production '--backend kubernetes' lease path. Regression for the
    round-1 split-brain: lease takeover used a merge patch with no
    resourceVersion precondition, so two standbys racing on an expired lease
    could both become leader."""

    def _facade(self):
        from kuberay_amd.kube.httpserver import KubeApiFacade
        return KubeApiFacade().start()

    def test_acquire_renew_and_blocked_standby(self):
        from kuberay_amd.kube.leaderelection import LeaderElector
        from kuberay_amd.kube.rest import RestClient
        facade = self._facade()
        try:
            a = LeaderElector(RestClient(base_url=facade.url), identity="a")
            b = LeaderElector(RestClient(base_url=facade.url), identity="b")
            assert a.try_acquire_or_renew() is True
            assert a.try_acquire_or_renew() is True  # renewal
            assert b.try_acquire_or_renew() is False
        finally:
            facade.stop()

    def test_expired_lease_takeover_race_elects_exactly_one(self):
        """Both standbys read the same expired lease revision; only the first
        write may win — the second must get 409 and stand down."""
        import pytest as _pytest

        from kuberay_amd.kube.leaderelection import LEASE_KIND, LeaderElector, _LeaseStore
        from kuberay_amd.kube.rest import RestClient
        from kuberay_amd.kube.store import ConflictError
        facade = self._facade()
        try:
            # dead leader: lease exists but renewTime is ancient
            facade.store.create({
                "apiVersion": "coordination.k8s.io/v1", "kind": LEASE_KIND,
                "metadata": {"name": "op-lease", "namespace": "ray-system"},
                "spec": {"holderIdentity": "dead",
                         "leaseDurationSeconds": 1,
                         "renewTime": "2000-01-01T00:00:00.000000Z",
                         "leaseTransitions": 1}})
            sa = _LeaseStore(RestClient(base_url=facade.url))
            sb = _LeaseStore(RestClient(base_url=facade.url))
            lease_a = sa.get("ray-system", "op-lease")
            lease_b = sb.get("ray-system", "op-lease")  # same resourceVersion
            lease_a["spec"]["holderIdentity"] = "standby-a"
            sa.update(lease_a)  # first write wins
            lease_b["spec"]["holderIdentity"] = "standby-b"
            with _pytest.raises(ConflictError):
                sb.update(lease_b)  # stale RV must NOT take the lease
            # and the full elector treats that conflict as a lost election
            b = LeaderElector(RestClient(base_url=facade.url),
                              identity="standby-b", lease_name="op-lease",
                              namespace="ray-system")
            got = facade.store.get(LEASE_KIND, "ray-system", "op-lease")
            assert got["spec"]["holderIdentity"] == "standby-a"
        finally:
            facade.stop()

    def test_elector_conflict_on_takeover_stands_down(self):
        """End-to-end through try_acquire_or_renew: a racing write between a
        standby's read and write yields is_leader False, not split-brain."""
        from kuberay_amd.kube.leaderelection import LEASE_KIND, LeaderElector
        from kuberay_amd.kube.rest import RestClient
        facade = self._facade()
        try:
            facade.store.create({
                "apiVersion": "coordination.k8s.io/v1", "kind": LEASE_KIND,
                "metadata": {"name": "op-lease", "namespace": "ray-system"},
                "spec": {"holderIdentity": "dead",
                         "leaseDurationSeconds": 1,
                         "renewTime": "2000-01-01T00:00:00.000000Z",
                         "leaseTransitions": 1}})
            b = LeaderElector(RestClient(base_url=facade.url),
                              identity="standby-b", lease_name="op-lease",
                              namespace="ray-system")
            real_get = b.store.get

            def racing_get(ns, name):
                lease = real_get(ns, name)
                # another standby sneaks its write in between read and write
                fresh = dict(facade.store.get(LEASE_KIND, ns, name))
                fresh["spec"] = dict(fresh["spec"],
                                     holderIdentity="standby-a",
                                     renewTime="2099-01-01T00:00:00.000000Z")
                facade.store.update(fresh)
                return lease

            b.store.get = racing_get
            assert b.try_acquire_or_renew() is False
            assert b.is_leader is False
            got = facade.store.get(LEASE_KIND, "ray-system", "op-lease")
            assert got["spec"]["holderIdentity"] == "standby-a"
        finally:
            facade.stop()


class TestEventRecorder:
    def test_events_aggregate_by_reason(self):
        from kuberay_amd.kube.events import StoreRecorder
        from kuberay_amd.testing import simple_raycluster
        server = InMemoryApiServer()
        rec = StoreRecorder(server)
        obj = simple_raycluster("evt")
        obj.metadata.uid = "u1"
        for _ in range(3):
            rec.eventf(obj, "Normal", "CreatedWorkerPod", "Created worker Pod")
        events = server.list("Event")
        assert len(events) == 1
        assert events[0]["count"] == 3
        assert events[0]["involvedObject"]["name"] == "evt"

    def test_distinct_messages_create_distinct_events(self):
        from kuberay_amd.kube.events import StoreRecorder
        from kuberay_amd.testing import simple_raycluster
        server = InMemoryApiServer()
        rec = StoreRecorder(server)
        obj = simple_raycluster("evt")
        rec.eventf(obj, "Normal", "CreatedWorkerPod", "Created worker Pod %s", "a")
        rec.eventf(obj, "Normal", "CreatedWorkerPod", "Created worker Pod %s", "b")
        assert len(server.list("Event")) == 2


class TestPodPool:
    """Warm pod pool (reference: podpool/ virtual-kubelet warm pods)."""

    TEMPLATE = {"apiVersion": "v1", "kind": "Pod",
                "metadata": {},
                "spec": {"containers": [{
                    "name": "ray-worker", "image": "rocm/ray:2.46.0",
                    "resources": {"limits": {"amd.com/gpu": "1"}}}]}}

    def _mgr(self):
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.ops.podpool import PodPoolManager
        client = InMemoryClient()
        return PodPoolManager(client), client

    def test_reconcile_tops_up_to_target(self):
        mgr, client = self._mgr()
        mgr.define_pool("mi355x", self.TEMPLATE, size=3)
        mgr.reconcile()
        pods = client.server.list("Pod", "default")
        assert len(pods) == 3
        assert all(p["metadata"]["labels"]["ray.io/warm-pod"] == "true"
                   for p in pods)
        mgr.reconcile()  # idempotent
        assert len(client.server.list("Pod", "default")) == 3

    def test_adopt_relabels_and_pool_refills(self):
        mgr, client = self._mgr()
        mgr.define_pool("mi355x", self.TEMPLATE, size=2)
        mgr.reconcile()
        name = mgr.adopt("mi355x", {"ray.io/cluster": "c1",
                                    "ray.io/node-type": "worker"})
        assert name is not None
        pod = client.server.get("Pod", "default", name)
        assert pod["metadata"]["labels"]["ray.io/warm-pod"] == "adopted"
        assert pod["metadata"]["labels"]["ray.io/cluster"] == "c1"
        # top-up replaces the adopted pod
        mgr.reconcile()
        warm = [p for p in client.server.list("Pod", "default")
                if p["metadata"]["labels"].get("ray.io/warm-pod") == "true"]
        assert len(warm) == 2

    def test_adopt_from_dry_pool_returns_none(self):
        mgr, _ = self._mgr()
        mgr.define_pool("empty", self.TEMPLATE, size=0)
        mgr.reconcile()
        assert mgr.adopt("empty", {"ray.io/cluster": "c1"}) is None


class TestVirtualKubeletPodPool:
    """Virtual-kubelet layer (reference podpool/cmd/main.go contract):
    node registration + heartbeat + instant binding from the warm pool."""

    TEMPLATE = TestPodPool.TEMPLATE

    def _stack(self):
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.ops.podpool import PodPoolManager, VirtualKubeletPodPool
        client = InMemoryClient()
        mgr = PodPoolManager(client)
        mgr.define_pool("mi355x", self.TEMPLATE, size=2)
        vk = VirtualKubeletPodPool(client, mgr, heartbeat_s=0.05)
        return client, mgr, vk

    def test_node_registered_with_taint_and_capacity(self):
        client, mgr, vk = self._stack()
        mgr.reconcile()
        vk.register_node()
        node = client.server.get("Node", "default", "kuberay-pod-pool")
        assert node["metadata"]["labels"]["type"] == "virtual-kubelet"
        assert node["spec"]["taints"][0]["key"] == \
            "virtual-kubelet.io/provider"
        assert node["status"]["capacity"]["pods"] == "2"

    def test_heartbeat_refreshes_ready_condition(self):
        import time as _t
        client, _, vk = self._stack()
        vk.register_node()
        first = client.server.get("Node", "default", "kuberay-pod-pool")[
            "status"]["conditions"][0]["lastHeartbeatTime"]
        _t.sleep(1.1)  # now_iso has 1s granularity
        vk.heartbeat()
        second = client.server.get("Node", "default", "kuberay-pod-pool")[
            "status"]["conditions"][0]["lastHeartbeatTime"]
        assert second >= first

    def test_pool_targeted_pod_binds_instantly(self):
        client, mgr, vk = self._stack()
        mgr.reconcile()
        consumer = {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": "worker-0", "namespace": "default",
                         "labels": {"ray.io/warm-pod-pool": "mi355x",
                                    "ray.io/cluster": "c1"}},
            "spec": {"containers": [{"name": "ray-worker",
                                     "image": "rocm/ray:2.46.0"}]}}
        client.server.create(consumer)
        assert vk.bind_pending_pods() == 1
        pod = client.server.get("Pod", "default", "worker-0")
        assert pod["status"]["phase"] == "Running"
        assert pod["status"]["containerStatuses"][0]["ready"] is True
        assert pod["metadata"]["annotations"]["ray.io/warm-pod-source"]
        # consumed warm pod removed, pool refilled to target
        warm = [p for p in client.server.list("Pod", "default")
                if (p["metadata"].get("labels") or {})
                .get("ray.io/warm-pod") == "true"]
        assert len(warm) == 2

    def test_dry_pool_leaves_pod_pending(self):
        client, mgr, vk = self._stack()
        mgr.define_pool("mi355x", self.TEMPLATE, size=0)
        mgr.reconcile()
        client.server.create({
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": "worker-0", "namespace": "default",
                         "labels": {"ray.io/warm-pod-pool": "mi355x"}},
            "spec": {"containers": [{"name": "w", "image": "i"}]}})
        assert vk.bind_pending_pods() == 0
        pod = client.server.get("Pod", "default", "worker-0")
        assert pod.get("status", {}).get("phase") in (None, "Pending")

    def test_watch_driven_bind_loop(self):
        import time as _t
        client, mgr, vk = self._stack()
        vk.start()
        try:
            client.server.create({
                "apiVersion": "v1", "kind": "Pod",
                "metadata": {"name": "late-worker", "namespace": "default",
                             "labels": {"ray.io/warm-pod-pool": "mi355x"}},
                "spec": {"containers": [{"name": "w", "image": "i"}]}})
            deadline = _t.monotonic() + 5
            phase = None
            while _t.monotonic() < deadline:
                phase = client.server.get("Pod", "default", "late-worker") \
                    .get("status", {}).get("phase")
                if phase == "Running":
                    break
                _t.sleep(0.05)
            assert phase == "Running"
        finally:
            vk.stop()


class TestShardedLeases:
    def test_shards_hold_distinct_leases_concurrently(self):
        """Two operator shards elect leaders on independent Lease names."""
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.kube.leaderelection import LeaderElector
        client = InMemoryClient()
        s0 = LeaderElector(client, identity="proc-0",
                           lease_name="kuberay-amd-operator-shard-0")
        s1 = LeaderElector(client, identity="proc-1",
                           lease_name="kuberay-amd-operator-shard-1")
        assert s0.try_acquire_or_renew() is True
        assert s1.try_acquire_or_renew() is True  # no contention across shards
        # a standby for shard 0 still blocks on that shard's lease
        standby = LeaderElector(client, identity="proc-0b",
                                lease_name="kuberay-amd-operator-shard-0")
        assert standby.try_acquire_or_renew() is False
