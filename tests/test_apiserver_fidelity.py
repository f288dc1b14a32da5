"""Real-apiserver semantic fidelity of the REST stack.

The reference proves its controllers against envtest (a real
kube-apiserver). This image has no kube-apiserver binary, so the next-best
tier is making the HTTP facade semantically faithful — watch resumption by
resourceVersion, 410 Gone + re-list, watch bookmarks, chunked lists with
continue tokens, RFC 6902 json-patch, strategic-merge-patch, PUT
optimistic-concurrency — and running the controller stacks through
kube/rest.py against it (see also tests/test_rest_e2e.py).
"""
import json
import threading
import time

import pytest

from kuberay_amd.kube.httpserver import (
    JsonPatchTestFailed,
    KubeApiFacade,
    apply_json_patch,
    strategic_merge,
)
from kuberay_amd.kube.kubelet import SimKubelet
from kuberay_amd.kube.rest import RestApiServerAdapter, RestClient
from kuberay_amd.kube.store import GoneError, InMemoryApiServer
from kuberay_amd.models import RayCluster, RayJob, RayService
from kuberay_amd.testing import simple_raycluster


@pytest.fixture()
def facade():
    f = KubeApiFacade().start()
    yield f
    f.stop()


def _cm(name, ns="default", **data):
    return {"apiVersion": "v1", "kind": "ConfigMap",
            "metadata": {"name": name, "namespace": ns},
            "data": data or {"k": "v"}}


class TestWatchResume:
    def test_watch_from_rv_replays_missed_events(self, facade):
        client = RestClient(base_url=facade.url)
        client.raw_create(_cm("before"))
        _, rv = client.raw_list_with_rv("ConfigMap")
        # events AFTER the list
        client.raw_create(_cm("after-1"))
        client.raw_create(_cm("after-2"))
        got = []
        for event_type, obj in client.raw_watch_stream("ConfigMap", rv):
            got.append((event_type, obj["metadata"]["name"]))
            if len(got) == 2:
                break
        assert got == [("ADDED", "after-1"), ("ADDED", "after-2")]

    def test_watch_from_current_rv_gets_only_new(self, facade):
        client = RestClient(base_url=facade.url)
        client.raw_create(_cm("old"))
        _, rv = client.raw_list_with_rv("ConfigMap")
        received = []
        done = threading.Event()

        def consume():
            for event_type, obj in client.raw_watch_stream("ConfigMap", rv):
                received.append(obj["metadata"]["name"])
                done.set()
                return
        t = threading.Thread(target=consume, daemon=True)
        t.start()
        time.sleep(0.2)
        client.raw_create(_cm("new"))
        assert done.wait(5)
        assert received == ["new"]  # "old" was before the list rv

    def test_too_old_rv_gets_410(self, facade):
        client = RestClient(base_url=facade.url)
        # churn past the retained-history window
        limit = InMemoryApiServer.EVENT_HISTORY_LIMIT
        for i in range(limit + 10):
            facade.store.create(_cm(f"churn-{i:05d}"))
        with pytest.raises(GoneError):
            for _ in client.raw_watch_stream("ConfigMap", "1"):
                break

    def test_bookmarks_advance_rv_without_traffic(self, facade):
        client = RestClient(base_url=facade.url)
        client.raw_create(_cm("seed"))
        _, rv = client.raw_list_with_rv("ConfigMap")
        got = {}
        done = threading.Event()

        def consume():
            for event_type, obj in client.raw_watch_stream(
                    "ConfigMap", rv, allow_bookmarks=True):
                if event_type == "BOOKMARK":
                    got["rv"] = obj["metadata"]["resourceVersion"]
                    done.set()
                    return
        t = threading.Thread(target=consume, daemon=True)
        t.start()
        assert done.wait(5), "no BOOKMARK within the idle window"
        assert int(got["rv"]) >= int(rv)

    def test_adapter_recovers_from_410_compaction(self, facade):
        """Informer disconnects, the server compacts its event history,
        reconnect with the stale rv → 410 → automatic re-list; new objects
        still flow."""
        port = int(facade.url.rsplit(":", 1)[1])
        adapter = RestApiServerAdapter(
            rest_client=RestClient(base_url=facade.url))
        watcher = adapter.watch(["ConfigMap"])
        try:
            facade.store.create(_cm("first"))
            seen = set()
            deadline = time.monotonic() + 10
            while "first" not in seen and time.monotonic() < deadline:
                ev = watcher.next(timeout=0.2)
                if ev:
                    seen.add(ev[1]["metadata"]["name"])
            assert "first" in seen
            # apiserver goes away; massive churn happens while we're gone
            facade.stop()
            limit = InMemoryApiServer.EVENT_HISTORY_LIMIT
            for i in range(limit + 10):
                facade.store.create(_cm(f"churn-{i:05d}"))
            facade.store.create(_cm("final"))
            facade2 = KubeApiFacade(facade.store, port=port).start()
            try:
                deadline = time.monotonic() + 20
                while "final" not in seen and time.monotonic() < deadline:
                    ev = watcher.next(timeout=0.2)
                    if ev:
                        seen.add(ev[1]["metadata"]["name"])
                assert "final" in seen, \
                    "informer did not recover after 410/compaction"
            finally:
                facade2.stop()
        finally:
            adapter.stop()


class TestChunkedLists:
    def test_pagination_with_continue_tokens(self, facade):
        client = RestClient(base_url=facade.url)
        for i in range(5):
            client.raw_create(_cm(f"page-{i}"))
        http = client._http
        seen = []
        token = None
        pages = 0
        while True:
            params = {"limit": "2"}
            if token:
                params["continue"] = token
            body = http.get("/api/v1/namespaces/default/configmaps",
                            params=params).json()
            seen += [o["metadata"]["name"] for o in body["items"]]
            pages += 1
            token = body["metadata"].get("continue")
            if not token:
                break
        assert pages == 3
        assert sorted(seen) == [f"page-{i}" for i in range(5)]

    def test_invalid_continue_token_is_400(self, facade):
        client = RestClient(base_url=facade.url)
        resp = client._http.get("/api/v1/namespaces/default/configmaps",
                                params={"limit": "2",
                                        "continue": "not-a-token!"})
        assert resp.status_code == 400

    def test_list_carries_resource_version(self, facade):
        client = RestClient(base_url=facade.url)
        client.raw_create(_cm("x"))
        body = client._http.get(
            "/api/v1/namespaces/default/configmaps").json()
        assert int(body["metadata"]["resourceVersion"]) >= 1


class TestJsonPatch:
    def test_pure_ops(self):
        doc = {"a": {"b": 1}, "list": [1, 2]}
        out = apply_json_patch(doc, [
            {"op": "replace", "path": "/a/b", "value": 2},
            {"op": "add", "path": "/list/-", "value": 3},
            {"op": "add", "path": "/c", "value": "new"},
            {"op": "remove", "path": "/list/0"},
            {"op": "test", "path": "/a/b", "value": 2},
            {"op": "move", "from": "/c", "path": "/d"},
            {"op": "copy", "from": "/a/b", "path": "/e"},
        ])
        assert out == {"a": {"b": 2}, "list": [2, 3], "d": "new", "e": 2}
        assert doc["a"]["b"] == 1  # original untouched

    def test_test_op_failure(self):
        with pytest.raises(JsonPatchTestFailed):
            apply_json_patch({"a": 1}, [{"op": "test", "path": "/a",
                                         "value": 2}])

    def test_json_patch_over_http(self, facade):
        client = RestClient(base_url=facade.url)
        client.raw_create(_cm("jp", k="v"))
        resp = client._http.patch(
            "/api/v1/namespaces/default/configmaps/jp",
            content=json.dumps([
                {"op": "replace", "path": "/data/k", "value": "patched"}]),
            headers={"Content-Type": "application/json-patch+json"})
        assert resp.status_code == 200
        assert facade.store.get("ConfigMap", "default", "jp")["data"]["k"] \
            == "patched"

    def test_json_patch_test_failure_is_409(self, facade):
        client = RestClient(base_url=facade.url)
        client.raw_create(_cm("jp2", k="v"))
        resp = client._http.patch(
            "/api/v1/namespaces/default/configmaps/jp2",
            content=json.dumps([
                {"op": "test", "path": "/data/k", "value": "other"},
                {"op": "replace", "path": "/data/k", "value": "nope"}]),
            headers={"Content-Type": "application/json-patch+json"})
        assert resp.status_code == 409
        assert facade.store.get("ConfigMap", "default", "jp2")["data"]["k"] \
            == "v"


class TestStrategicMerge:
    def test_containers_merge_by_name(self):
        current = {"spec": {"containers": [
            {"name": "ray", "image": "a", "env": [{"name": "X", "value": "1"}]},
            {"name": "sidecar", "image": "s"}]}}
        patch = {"spec": {"containers": [
            {"name": "ray", "image": "b"}]}}
        out = strategic_merge(current, patch)
        by_name = {c["name"]: c for c in out["spec"]["containers"]}
        assert by_name["ray"]["image"] == "b"
        assert by_name["ray"]["env"] == [{"name": "X", "value": "1"}]
        assert "sidecar" in by_name  # NOT replaced away (≠ merge patch)

    def test_patch_delete_directive(self):
        current = {"spec": {"containers": [{"name": "a"}, {"name": "b"}]}}
        patch = {"spec": {"containers": [{"name": "a",
                                          "$patch": "delete"}]}}
        out = strategic_merge(current, patch)
        assert [c["name"] for c in out["spec"]["containers"]] == ["b"]

    def test_plain_list_replaces(self):
        out = strategic_merge({"args": ["a"]}, {"args": ["b"]})
        assert out == {"args": ["b"]}

    def test_strategic_merge_over_http_vs_merge_patch(self, facade):
        """The semantic difference the controller relies on: merge patch
        REPLACES the containers list; strategic merge UPDATES in place."""
        client = RestClient(base_url=facade.url)
        pod = {"apiVersion": "v1", "kind": "Pod",
               "metadata": {"name": "p1", "namespace": "default"},
               "spec": {"containers": [{"name": "ray", "image": "a"},
                                       {"name": "sidecar", "image": "s"}]}}
        client.raw_create(pod)
        resp = client._http.patch(
            "/api/v1/namespaces/default/pods/p1",
            content=json.dumps({"spec": {"containers": [
                {"name": "ray", "image": "b"}]}}),
            headers={
                "Content-Type": "application/strategic-merge-patch+json"})
        assert resp.status_code == 200
        got = facade.store.get("Pod", "default", "p1")
        assert [c["name"] for c in got["spec"]["containers"]] == \
            ["ray", "sidecar"]
        assert got["spec"]["containers"][0]["image"] == "b"


class TestConflictStorm:
    def test_concurrent_read_modify_write_all_land(self, facade):
        """10 threads × 10 increments against one object through the REST
        backend: update_with_retry must absorb every 409."""
        client = RestClient(base_url=facade.url)
        client.create(simple_raycluster("storm", workers=0))
        threads, errors = [], []

        def worker():
            local = RestClient(base_url=facade.url)
            for _ in range(10):
                try:
                    def bump(rc):
                        rc.spec.worker_group_specs[0].max_replicas = \
                            (rc.spec.worker_group_specs[0].max_replicas
                             or 0) + 1
                    local.update_with_retry(RayCluster, "default", "storm",
                                            bump, attempts=100)
                except Exception as e:  # noqa: BLE001 — collected for assert
                    errors.append(e)

        # give the cluster one worker group to bump
        def seed(rc):
            from kuberay_amd.models.raycluster import WorkerGroupSpec
            rc.spec.worker_group_specs = [WorkerGroupSpec.from_dict({
                "groupName": "g", "replicas": 0, "minReplicas": 0,
                "maxReplicas": 0, "rayStartParams": {},
                "template": {"spec": {"containers": [
                    {"name": "w", "image": "i"}]}}})]
        client.update_with_retry(RayCluster, "default", "storm", seed)
        for _ in range(10):
            t = threading.Thread(target=worker)
            t.start()
            threads.append(t)
        for t in threads:
            t.join(timeout=60)
        assert not errors, errors[:3]
        final = client.get(RayCluster, "default", "storm")
        assert final.spec.worker_group_specs[0].max_replicas == 100


class TestControllersOverRest:
    """The RayJob and RayService controller stacks end-to-end through
    RestClient against the HTTP facade (test_rest_e2e.py covers
    RayCluster) — the 'controller scenarios pass against the real REST
    surface' criterion."""

    def _stack(self, facade):
        from kuberay_amd.kube.controller import Controller, Manager
        from kuberay_amd.ops.raycluster import (
            RayClusterReconciler,
            RayClusterReconcilerOptions,
        )
        from kuberay_amd.ops.rayjob import RayJobReconciler
        from kuberay_amd.ops.rayservice import RayServiceReconciler
        from kuberay_amd.utils.fake_dashboard import FakeRayDashboardClient

        adapter = RestApiServerAdapter(
            rest_client=RestClient(base_url=facade.url))
        client = adapter.client()
        dashboard = FakeRayDashboardClient()
        options = RayClusterReconcilerOptions()
        options.requeue_after_seconds = 300
        manager = Manager(adapter)
        manager.add_controller(Controller(
            "raycluster", "RayCluster",
            RayClusterReconciler(client, options=options),
            owned_kinds=["Pod", "Service", "Secret",
                         "PersistentVolumeClaim", "Job"], workers=2))
        job_rec = RayJobReconciler(client,
                                   dashboard_factory=lambda url: dashboard)
        job_rec.requeue_seconds = 0.1
        manager.add_controller(Controller(
            "rayjob", "RayJob", job_rec,
            owned_kinds=["RayCluster", "Job"], workers=2))
        svc_rec = RayServiceReconciler(
            client, dashboard_factory=lambda url: dashboard)
        svc_rec.requeue_seconds = 0.1
        manager.add_controller(Controller(
            "rayservice", "RayService", svc_rec,
            owned_kinds=["RayCluster", "Service"], workers=2))
        kubelet = SimKubelet(facade.store, startup_delay=0.01,
                             job_runtime=0.2)
        return adapter, client, manager, kubelet

    def test_rayjob_completes_over_rest(self, facade):
        adapter, client, manager, kubelet = self._stack(facade)
        manager.start()
        kubelet.start()
        try:
            client.create(RayJob.from_dict({
                "apiVersion": "ray.io/v1", "kind": "RayJob",
                "metadata": {"name": "rj", "namespace": "default"},
                "spec": {"entrypoint": "python t.py",
                         "rayClusterSpec":
                             simple_raycluster("x").spec.to_dict()}}))
            deadline = time.monotonic() + 40
            status = None
            while time.monotonic() < deadline:
                job = client.try_get(RayJob, "default", "rj")
                status = job.status.job_deployment_status if job else None
                if status == "Complete":
                    break
                time.sleep(0.1)
            assert status == "Complete"
        finally:
            kubelet.stop()
            manager.stop()
            adapter.stop()

    def test_rayservice_ready_over_rest(self, facade):
        adapter, client, manager, kubelet = self._stack(facade)
        manager.start()
        kubelet.start()
        try:
            client.create(RayService.from_dict({
                "apiVersion": "ray.io/v1", "kind": "RayService",
                "metadata": {"name": "rs", "namespace": "default"},
                "spec": {"serveConfigV2":
                             "applications:\n- name: a\n",
                         "rayClusterConfig":
                             simple_raycluster("x").spec.to_dict()}}))
            deadline = time.monotonic() + 40
            ready = False
            while time.monotonic() < deadline:
                svc = client.try_get(RayService, "default", "rs")
                ready = bool(svc and svc.condition_true("Ready"))
                if ready:
                    break
                time.sleep(0.1)
            assert ready
        finally:
            kubelet.stop()
            manager.stop()
            adapter.stop()


@pytest.mark.timeout(180)
def test_sharded_operator_topology_end_to_end():
    """Two REAL operator processes (--shards 2) against the HTTP facade:
    every shard elects its own lease, owns its hash-split of the CR space,
    and the whole space converges (benchmark/perf-tests/sharded.py at CI
    size). Also the regression test for the facade watch-registration race
    (events between history read and watcher attach were lost)."""
    import json
    import os
    import subprocess
    import sys
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "benchmark/perf-tests/sharded.py",
         "--shards", "2", "--clusters", "24", "--timeout", "90"],
        capture_output=True, text=True, cwd=here, timeout=170)
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-800:])
    result = json.loads(out.stdout)
    assert result["all_ready"] is True
    assert result["all_shards_active"] is True


class TestDeleteRevision:
    """A delete is its own revision: a watcher resuming from the rv at
    which it saw the object's LAST update must still receive the DELETED
    event (kube-apiserver semantics; regression for rv-reuse on delete)."""

    def test_resumed_watch_sees_deletion(self, facade):
        client = RestClient(base_url=facade.url)
        client.raw_create(_cm("doomed"))
        _, rv = client.raw_list_with_rv("ConfigMap")  # saw the object
        facade.store.delete("ConfigMap", "default", "doomed")
        got = []
        for event_type, obj in client.raw_watch_stream("ConfigMap", rv):
            got.append((event_type, obj["metadata"]["name"]))
            break
        assert got == [("DELETED", "doomed")]


class TestSchedulersOverRest:
    """Gang schedulers must work on the REST backend too (they used to
    silently no-op without the in-memory `server` attribute)."""

    def test_podgroup_created_and_cleaned_over_rest(self, facade):
        from kuberay_amd.parallel.batchscheduler import XgmiGangScheduler
        client = RestClient(base_url=facade.url)
        cluster = client.create(simple_raycluster("restgang", workers=2,
                                                  gpus_per_worker=1))
        sched = XgmiGangScheduler()
        sched.do_batch_scheduling_on_submission(client, cluster)
        pgs = facade.store.list("PodGroup")
        assert len(pgs) == 1
        assert pgs[0]["spec"]["minMember"] == 3
        sched.cleanup_on_completion(client, cluster)
        assert facade.store.list("PodGroup") == []

    def test_island_pinning_reads_nodes_over_rest(self, facade):
        from kuberay_amd.common import pod as podlib
        from kuberay_amd.parallel.batchscheduler import XgmiGangScheduler
        client = RestClient(base_url=facade.url)
        facade.store.create({"kind": "Node", "apiVersion": "v1",
                             "metadata": {"name": "n1", "labels": {
                                 "amd.com/gpu.count": "8",
                                 "amd.com/xgmi-island": "n1-island0"}}})
        cluster = client.create(simple_raycluster("restpin", workers=2,
                                                  gpus_per_worker=2))
        group = cluster.spec.worker_group_specs[0]
        fqdn = "x.default.svc.cluster.local"
        t = podlib.default_worker_pod_template(cluster, group, "p-", fqdn,
                                               "6379")
        pod = podlib.build_pod(t, "worker", group.ray_start_params, "6379",
                               False, None, fqdn)
        XgmiGangScheduler().add_metadata_to_pod(client, cluster,
                                                "default-group", pod)
        terms = pod.spec.affinity["nodeAffinity"][
            "requiredDuringSchedulingIgnoredDuringExecution"][
            "nodeSelectorTerms"]
        assert terms[0]["matchExpressions"][0]["values"] == ["n1-island0"]

    def test_cert_manager_issuance_over_rest(self, facade):
        from kuberay_amd.ops.mtls import MTLSReconciler
        client = RestClient(base_url=facade.url)
        client.create(simple_raycluster("restmtls",
                                        tlsOptions={"enabled": True}))
        r = MTLSReconciler(client, mode="cert-manager")
        r.reconcile(("default", "restmtls"))
        issuers = facade.store.list("Issuer")
        certs = facade.store.list("Certificate")
        assert {i["metadata"]["name"] for i in issuers} == {
            "ray-selfsigned-issuer-restmtls", "ray-ca-issuer-restmtls"}
        assert len(certs) == 3
        # second reconcile is idempotent
        r.reconcile(("default", "restmtls"))
        assert len(facade.store.list("Certificate")) == 3

    def test_dynamic_kind_watch_over_rest(self, facade):
        """PodGroup (a CRD kind outside the static resource map) is
        watchable over REST once the facade has learned it."""
        import threading
        client = RestClient(base_url=facade.url)
        client.raw_create({"apiVersion": "scheduling.x-k8s.io/v1alpha1",
                           "kind": "PodGroup",
                           "metadata": {"name": "w0", "namespace": "default"},
                           "spec": {"minMember": 2}})
        seen = []
        def consume():
            for etype, obj in client.raw_watch_stream(
                    "PodGroup", "0",
                    api_version="scheduling.x-k8s.io/v1alpha1"):
                seen.append((etype, obj["metadata"]["name"]))
                if len(seen) >= 2:
                    break
        t = threading.Thread(target=consume, daemon=True)
        t.start()
        import time as _t
        _t.sleep(0.3)
        client.raw_create({"apiVersion": "scheduling.x-k8s.io/v1alpha1",
                           "kind": "PodGroup",
                           "metadata": {"name": "w1", "namespace": "default"},
                           "spec": {"minMember": 3}})
        t.join(timeout=10)
        assert ("ADDED", "w0") in seen and ("ADDED", "w1") in seen


def test_rest_op_latency_has_no_delayed_ack_stall(facade):
    """100 sequential REST gets must beat 3 s by a wide margin. With the
    Nagle/delayed-ACK interaction (no TCP_NODELAY) each op stalls ~40 ms
    and this takes >4 s; post-fix it is ~0.1 s. Generous bound so loaded
    CI never flakes, tight enough to catch the 40 ms/op failure mode."""
    import time as _t
    client = RestClient(base_url=facade.url)
    facade.store.create({"kind": "ConfigMap", "apiVersion": "v1",
                         "metadata": {"name": "lat"}, "data": {}})
    t0 = _t.perf_counter()
    for _ in range(100):
        client.raw_try_get("ConfigMap", "default", "lat")
    assert _t.perf_counter() - t0 < 3.0


def test_was_scheduler_over_rest(facade):
    """The Workload-API (kubernetes-was) scheduler manages Workload +
    PodGroup over the REST backend, including the protection-finalizer
    strip on cleanup."""
    from kuberay_amd.parallel.batchscheduler import KubernetesWASBatchScheduler
    from kuberay_amd.utils import constants as C
    client = RestClient(base_url=facade.url)
    c = simple_raycluster("wasrest", workers=2, gpus_per_worker=1)
    c.metadata.labels = {**(c.metadata.labels or {}),
                         C.RAY_GANG_SCHEDULING_ENABLED: "true"}
    cluster = client.create(c)
    sched = KubernetesWASBatchScheduler()
    sched.do_batch_scheduling_on_submission(client, cluster)
    wls = facade.store.list("Workload")
    pgs = facade.store.list("PodGroup")
    assert len(wls) == 1 and len(pgs) == 1
    sched.cleanup_on_completion(client, cluster)
    assert facade.store.list("Workload") == []
    assert facade.store.list("PodGroup") == []


def test_podpool_vk_node_registers_over_rest(facade):
    from kuberay_amd.ops.podpool import PodPoolManager, VirtualKubeletPodPool
    client = RestClient(base_url=facade.url)
    mgr = PodPoolManager(client)
    vk = VirtualKubeletPodPool(client, mgr, node_name="vk-rest")
    vk.register_node()
    node = facade.store.try_get("Node", "default", "vk-rest")
    assert node is not None
    assert node["metadata"]["labels"]["type"] == "virtual-kubelet"
    vk.heartbeat()
    assert facade.store.try_get("Node", "default", "vk-rest")[
        "status"]["conditions"][0]["status"] == "True"


def test_incremental_upgrade_gateway_over_rest(facade):
    """Gateway/HTTPRoute management and weight stepping run over the REST
    backend (production topology for the Gateway-API migration)."""
    import yaml as _yaml
    from kuberay_amd.kube.events import NullRecorder
    from kuberay_amd.models import RayService
    from kuberay_amd.ops.incremental import IncrementalUpgrader
    client = RestClient(base_url=facade.url)
    cluster_spec = simple_raycluster("x", workers=1).spec.to_dict()
    cluster_spec["enableInTreeAutoscaling"] = True
    svc = client.create(RayService.from_dict({
        "apiVersion": "ray.io/v1", "kind": "RayService",
        "metadata": {"name": "irsvc", "namespace": "default"},
        "spec": {"serveConfigV2": _yaml.safe_dump(
                     {"applications": [{"name": "a", "import_path": "m:x"}]}),
                 "rayClusterConfig": cluster_spec,
                 "upgradeStrategy": {
                     "type": "NewClusterWithIncrementalUpgrade",
                     "clusterUpgradeOptions": {
                         "gatewayClassName": "istio",
                         "stepSizePercent": 50, "intervalSeconds": 0,
                         "maxSurgePercent": 100}}}}))
    active = client.create(simple_raycluster("ir-act", workers=1))
    pending = client.create(simple_raycluster("ir-pen", workers=1))
    up = IncrementalUpgrader(client, NullRecorder())
    up.ensure_gateway_infra(svc, active, pending)
    assert facade.store.try_get("Gateway", "default",
                                "irsvc-gateway") is not None
    # no gateway controller status yet -> step holds
    assert up.step_traffic(svc, active, pending) is False
    facade.store.patch_merge("Gateway", "default", "irsvc-gateway",
        {"status": {"conditions": [
            {"type": "Accepted", "status": "True"},
            {"type": "Programmed", "status": "True"}]}},
        subresource="status")
    facade.store.patch_merge("HTTPRoute", "default", "irsvc-route",
        {"status": {"parents": [{
            "parentRef": {"name": "irsvc-gateway"},
            "conditions": [{"type": "Accepted", "status": "True"},
                           {"type": "ResolvedRefs", "status": "True"}]}]}},
        subresource="status")
    assert up.step_traffic(svc, active, pending) is False   # 50
    assert up.step_traffic(svc, active, pending) is True    # 100
    route = facade.store.get("HTTPRoute", "default", "irsvc-route")
    weights = {r["name"]: r["weight"]
               for r in route["spec"]["rules"][0]["backendRefs"]}
    assert any(w == 100 for w in weights.values())


def test_reflector_relist_purges_deleted_pods(facade):
    """Objects deleted while a watch is DOWN must leave the informer
    cache on re-list (client-go Reflector Replace semantics): ghost pods
    would make reconcilers refuse to recreate. The watch is genuinely
    severed here — the stream raises GoneError while the pod is deleted
    behind its back, so only the Replace() path can purge it."""
    from kuberay_amd.kube.rest import RestApiServerAdapter
    from kuberay_amd.kube.store import GoneError as _Gone
    client = RestClient(base_url=facade.url)
    adapter = RestApiServerAdapter(rest_client=client)
    facade.store.create({"kind": "Pod", "apiVersion": "v1",
                         "metadata": {"name": "ghost", "labels": {}},
                         "spec": {}, "status": {"phase": "Running"}})
    import threading as _th
    severed = _th.Event()
    deleted = _th.Event()
    real_stream = client.raw_watch_stream

    def breaking_stream(kind, rv=None, allow_bookmarks=False, **kw):
        if not severed.is_set():
            severed.set()
            deleted.wait(timeout=10)   # pod is deleted while "down"
            raise _Gone("watch severed by test")
        return real_stream(kind, rv, allow_bookmarks=allow_bookmarks, **kw)

    client.raw_watch_stream = breaking_stream
    w = adapter.watch({"Pod"})
    assert severed.wait(timeout=10)
    # first list seeded the cache before the stream call
    assert adapter._client.pod_cache_contains("default", "ghost")
    facade.store.delete("Pod", "default", "ghost")
    deleted.set()
    events = []
    deadline = time.time() + 10
    while time.time() < deadline:
        ev = w.next(timeout=0.2)
        if ev and ev[0] == "DELETED" and \
                ev[1]["metadata"]["name"] == "ghost":
            events.append(ev)
            break
    assert events, "no synthetic DELETED delivered on re-list"
    assert not adapter._client.pod_cache_contains("default", "ghost")
    adapter.stop()


class TestPaginationSnapshots:
    def test_pages_are_churn_consistent(self, facade):
        """Continue tokens pin a snapshot: mutations between page fetches
        never cause skips or duplicates (etcd-revision semantics)."""
        import httpx
        for i in range(30):
            facade.store.create(_cm(f"pg{i:02d}"))
        base = facade.url + "/api/v1/namespaces/default/configmaps"
        r1 = httpx.get(base, params={"limit": 10}).json()
        assert len(r1["items"]) == 10
        token = r1["metadata"]["continue"]
        # churn between pages: delete one already-served and one
        # yet-to-be-served item, create new ones
        facade.store.delete("ConfigMap", "default", "pg03")
        facade.store.delete("ConfigMap", "default", "pg25")
        facade.store.create(_cm("zz-new"))
        r2 = httpx.get(base, params={"limit": 10, "continue": token}).json()
        token = r2["metadata"]["continue"]
        r3 = httpx.get(base, params={"limit": 10, "continue": token}).json()
        names = [o["metadata"]["name"] for r in (r1, r2, r3)
                 for o in r["items"]]
        assert names == [f"pg{i:02d}" for i in range(30)]
        assert "continue" not in r3["metadata"]
        # all three pages report the snapshot's rv
        assert (r1["metadata"]["resourceVersion"]
                == r3["metadata"]["resourceVersion"])

    def test_expired_token_answers_410(self, facade):
        import base64 as b64
        import httpx
        for i in range(5):
            facade.store.create(_cm(f"ex{i}"))
        base = facade.url + "/api/v1/namespaces/default/configmaps"
        bogus = b64.b64encode(b"deadbeef0000:2").decode()
        r = httpx.get(base, params={"limit": 2, "continue": bogus})
        assert r.status_code == 410
        assert r.json()["reason"] == "Expired"


def test_arbitrary_crd_kind_full_round_trip(facade):
    """Any CRD kind works over REST with zero registration: paths derive
    from apiVersion, the facade learns the kind from the POST body."""
    from kuberay_amd.kube.client import RawObjectClient
    client = RestClient(base_url=facade.url)
    raw = RawObjectClient(client)
    av = "argoproj.io/v1alpha1"
    raw.create({"apiVersion": av, "kind": "Workflow",
                "metadata": {"name": "wf1", "namespace": "default"},
                "spec": {"entrypoint": "main"}})
    got = raw.try_get("Workflow", "default", "wf1", api_version=av)
    assert got["spec"]["entrypoint"] == "main"
    raw.patch("Workflow", "default", "wf1", {"spec": {"parallelism": 3}},
              api_version=av)
    assert raw.try_get("Workflow", "default", "wf1",
                       api_version=av)["spec"]["parallelism"] == 3
    assert len(raw.list("Workflow", "default", api_version=av)) == 1
    raw.delete("Workflow", "default", "wf1", api_version=av)
    assert raw.try_get("Workflow", "default", "wf1",
                       api_version=av) is None


class TestFacadeRobustness:
    """Malformed requests get structured Status errors, never 500s."""

    def test_malformed_json_body(self, facade):
        import httpx
        r = httpx.post(facade.url + "/api/v1/namespaces/default/configmaps",
                       content=b"{not json", headers={
                           "Content-Type": "application/json"})
        assert r.status_code in (400, 422), r.status_code

    def test_unknown_paths_404(self, facade):
        import httpx
        for path in ("/", "/api", "/api/v2/zzz", "/apis/x",
                     "/api/v1/namespaces/default/configmaps/a/b/c"):
            r = httpx.get(facade.url + path)
            assert r.status_code == 404, path

    def test_delete_nonexistent_is_status_404(self, facade):
        import httpx
        r = httpx.delete(facade.url +
                         "/api/v1/namespaces/default/configmaps/nope")
        assert r.status_code == 404
        assert r.json()["kind"] == "Status"

    def test_put_with_stale_rv_conflicts(self, facade):
        import httpx
        facade.store.create(_cm("rvx"))
        cur = facade.store.get("ConfigMap", "default", "rvx")
        facade.store.update({**cur, "data": {"v": "2"}})
        stale = dict(cur)
        stale["data"] = {"v": "stale"}
        r = httpx.put(facade.url +
                      "/api/v1/namespaces/default/configmaps/rvx",
                      json=stale)
        assert r.status_code == 409

    def test_bad_json_patch_is_422(self, facade):
        import httpx
        facade.store.create(_cm("jp"))
        r = httpx.patch(facade.url +
                        "/api/v1/namespaces/default/configmaps/jp",
                        json=[{"op": "frobnicate", "path": "/x"}],
                        headers={"Content-Type":
                                 "application/json-patch+json"})
        assert r.status_code == 422

    def test_watch_unknown_rv_format_starts_from_now(self, facade):
        import httpx
        # non-numeric rv: facade treats it as 0/now rather than crashing
        with httpx.stream(
                "GET", facade.url + "/api/v1/namespaces/default/configmaps",
                params={"watch": "true", "resourceVersion": "abc"},
                timeout=2) as resp:
            assert resp.status_code in (200, 400, 410)


def test_snapshot_table_bounded_and_watch_ordered(facade):
    """Pagination snapshots are bounded (old tokens expire as 410 once
    evicted) and watch events always arrive in resourceVersion order."""
    import httpx
    for i in range(80):
        facade.store.create(_cm(f"bnd{i:03d}"))
    base = facade.url + "/api/v1/namespaces/default/configmaps"
    # open 70 paginated lists without consuming them: table caps at 64
    tokens = []
    for _ in range(70):
        r = httpx.get(base, params={"limit": 10}).json()
        tokens.append(r["metadata"]["continue"])
    snaps = getattr(facade._httpd, "_page_snaps", {})
    assert len(snaps) <= 64
    r = httpx.get(base, params={"limit": 10, "continue": tokens[0]})
    assert r.status_code == 410  # evicted token expires, never mis-pages
    # watch ordering: rv strictly increases over a live stream
    client = RestClient(base_url=facade.url)
    rv0 = str(facade.store.current_rv)
    seen = []
    import threading
    def consume():
        for _, obj in client.raw_watch_stream("ConfigMap", rv0):
            seen.append(int(obj["metadata"]["resourceVersion"]))
            if len(seen) >= 10:
                break
    t = threading.Thread(target=consume, daemon=True)
    t.start()
    time.sleep(0.2)
    for i in range(10):
        facade.store.create(_cm(f"ord{i}"))
    t.join(timeout=10)
    assert seen == sorted(seen) and len(seen) == 10
