"""History server tests: storage, collector, event replay, HTTP API
(reference analogs: historyserver unit tests with fake storage)."""
import json

import pytest
from fastapi.testclient import TestClient

import kuberay_amd.features as features
from kuberay_amd.historyserver.collector import Collector, build_collector_container
from kuberay_amd.historyserver.eventserver import SessionState, load_session
from kuberay_amd.historyserver.server import create_history_app
from kuberay_amd.historyserver.storage import (
    LocalStorage,
    MemoryStorage,
    compress,
    decompress,
    storage_for,
)

EVENTS = [
    {"event_type": "DRIVER_JOB_DEFINITION_EVENT",
     "data": {"job_id": "j1", "entrypoint": "python t.py"}},
    {"event_type": "DRIVER_JOB_LIFECYCLE_EVENT",
     "data": {"job_id": "j1", "state": "RUNNING", "start_time": 100}},
    {"event_type": "TASK_DEFINITION_EVENT",
     "data": {"task_id": "t1", "name": "f", "job_id": "j1"}},
    {"event_type": "TASK_LIFECYCLE_EVENT",
     "data": {"task_id": "t1", "state": "FINISHED", "timestamp": 105}},
    {"event_type": "ACTOR_DEFINITION_EVENT",
     "data": {"actor_id": "a1", "class_name": "Worker"}},
    {"event_type": "ACTOR_LIFECYCLE_EVENT",
     "data": {"actor_id": "a1", "state": "ALIVE"}},
    {"event_type": "NODE_DEFINITION_EVENT",
     "data": {"node_id": "n1", "node_ip": "10.0.0.1"}},
    {"event_type": "TASK_PROFILE_EVENT",
     "data": {"event_name": "f", "start_time": 100.0, "end_time": 100.5,
              "node_ip_address": "10.0.0.1", "component_id": "w1"}},
    {"event_type": "DRIVER_JOB_LIFECYCLE_EVENT",
     "data": {"job_id": "j1", "state": "SUCCEEDED", "end_time": 110}},
]


class TestStorage:
    def test_local_roundtrip(self, tmp_path):
        s = LocalStorage(str(tmp_path))
        s.write("a/b/c.txt", b"hello")
        assert s.read("a/b/c.txt") == b"hello"
        assert s.exists("a/b/c.txt")
        s.append("a/b/log.txt", b"x")
        s.append("a/b/log.txt", b"y")
        assert s.read("a/b/log.txt") == b"xy"
        assert s.list("a") == ["a/b/c.txt", "a/b/log.txt"]

    def test_local_path_escape_rejected(self, tmp_path):
        s = LocalStorage(str(tmp_path))
        with pytest.raises(ValueError):
            s.write("../../etc/evil", b"x")

    def test_compression(self):
        data = b"x" * 10000
        c = compress(data)
        assert len(c) < 200
        assert decompress(c) == data

    def test_backend_selection(self, tmp_path):
        assert isinstance(storage_for("local", root=str(tmp_path)), LocalStorage)
        assert isinstance(storage_for("memory"), MemoryStorage)
        with pytest.raises(RuntimeError):
            storage_for("s3").read("x")
        with pytest.raises(ValueError):
            storage_for("nope")


class TestCollector:
    def test_push_and_replay(self):
        storage = MemoryStorage()
        batches = [EVENTS[:4], EVENTS[4:]]
        it = iter(batches)
        collector = Collector(storage, "c1", namespace="ns1",
                              fetch_events=lambda: next(it, []),
                              fetch_logs=lambda: {"raylet.out": "log line"})
        assert collector.push_once() == 4
        assert collector.push_once() == 5
        state = load_session(storage, "ns1/c1/session-1")
        assert state.jobs["j1"]["status"] == "SUCCEEDED"
        assert state.tasks["t1"]["state"] == "FINISHED"
        assert state.actors["a1"]["state"] == "ALIVE"
        assert state.nodes["n1"]["node_ip"] == "10.0.0.1"
        assert len(state.timeline()) == 1

    def test_collector_container_shape(self):
        from kuberay_amd.models.raycluster import CollectorOptions
        c = build_collector_container(CollectorOptions(), "head", "c1", "ns1",
                                      "c1-head-svc.ns1.svc.cluster.local")
        env = {e.name: e.value for e in c.env if e.value is not None}
        assert env["OWNER_NAME"] == "c1"
        assert env["RAY_ROLE"] == "head"
        assert env["STORAGE_BACKEND"] == "local"

    def test_sidecar_injection_gated(self):
        from kuberay_amd.common import pod as podlib
        from kuberay_amd.testing import simple_raycluster
        cluster = simple_raycluster("demo", historyServerOptions={
            "collectorOptions": {}})
        t = podlib.default_head_pod_template(cluster, cluster.spec.head_group_spec,
                                             "demo-head-", "6379")
        assert "history-collector" not in [c.name for c in t.spec.containers]
        features.set_gate("RayClusterHistoryServer", True)
        try:
            cluster2 = simple_raycluster("demo2", historyServerOptions={
                "collectorOptions": {}})
            t2 = podlib.default_head_pod_template(
                cluster2, cluster2.spec.head_group_spec, "demo2-head-", "6379")
            names = [c.name for c in t2.spec.containers]
            assert "history-collector" in names
            # collector defaults to the ray image
            collector = next(c for c in t2.spec.containers
                             if c.name == "history-collector")
            assert collector.image == t2.spec.containers[0].image
        finally:
            features.reset()


class TestHistoryServerApi:
    @pytest.fixture()
    def app(self):
        storage = MemoryStorage()
        collector = Collector(storage, "c1", namespace="ns1",
                              fetch_events=lambda: EVENTS,
                              fetch_logs=lambda: {"raylet.out": "hello log"})
        collector.push_once()
        return TestClient(create_history_app(storage))

    def test_sessions(self, app):
        r = app.get("/api/sessions")
        assert r.json()["sessions"] == ["ns1/c1/session-1"]

    def test_jobs_tasks_actors_nodes(self, app):
        base = "/api/sessions/ns1/c1/session-1"
        assert app.get(f"{base}/jobs").json()["data"]["jobs"][0]["job_id"] == "j1"
        assert len(app.get(f"{base}/tasks").json()["data"]["tasks"]) == 1
        assert len(app.get(f"{base}/actors").json()["data"]["actors"]) == 1
        assert len(app.get(f"{base}/nodes").json()["data"]["nodes"]) == 1

    def test_timeline_trace_format(self, app):
        tl = app.get("/api/sessions/ns1/c1/session-1/timeline").json()
        assert tl[0]["ph"] == "X"
        assert tl[0]["dur"] == pytest.approx(0.5e6)

    def test_logs(self, app):
        r = app.get("/api/sessions/ns1/c1/session-1/logs/raylet.out")
        assert r.json()["logs"] == "hello log"
        assert app.get("/api/sessions/ns1/c1/session-1/logs/nope").status_code == 404


class TestHistoryServerListOptions:
    """Ray state-API list options (router.go getTasks/getTaskSummarize):
    filter triples, limit, detail endpoints, log listing/pagination."""

    @pytest.fixture()
    def app(self):
        storage = MemoryStorage()
        events = list(EVENTS) + [
            {"event_type": "TASK_DEFINITION_EVENT",
             "data": {"task_id": "t2", "name": "g", "job_id": "j1"}},
            {"event_type": "TASK_LIFECYCLE_EVENT",
             "data": {"task_id": "t2", "state": "RUNNING",
                      "timestamp": 106}},
            {"event_type": "TASK_DEFINITION_EVENT",
             "data": {"task_id": "t3", "name": "f", "job_id": "j2"}},
        ]
        collector = Collector(storage, "c1", namespace="ns1",
                              fetch_events=lambda: events,
                              fetch_logs=lambda: {
                                  "raylet.out": "l1\nl2\nl3\nl4\nl5\n",
                                  "gcs.out": "g"})
        collector.push_once()
        return TestClient(create_history_app(storage))

    BASE = "/api/sessions/ns1/c1/session-1"

    def test_filter_equals(self, app):
        r = app.get(f"{self.BASE}/tasks", params={
            "filter_keys": "job_id", "filter_values": "j1"})
        data = r.json()["data"]
        assert {t["task_id"] for t in data["tasks"]} == {"t1", "t2"}
        assert data["num_after_truncation"] == 3
        assert data["num_filtered"] == 2

    def test_filter_not_equals_and_limit(self, app):
        r = app.get(f"{self.BASE}/tasks", params=[
            ("filter_keys", "state"), ("filter_predicates", "!="),
            ("filter_values", "FINISHED"), ("limit", "1")])
        data = r.json()["data"]
        assert len(data["tasks"]) == 1
        assert data["num_filtered"] == 2  # t2 RUNNING + t3 (no state)

    def test_bad_filter_rejected(self, app):
        assert app.get(f"{self.BASE}/tasks", params=[
            ("filter_keys", "a"), ("filter_keys", "b"),
            ("filter_values", "1")]).status_code == 400
        assert app.get(f"{self.BASE}/tasks", params={
            "filter_keys": "a", "filter_predicates": ">",
            "filter_values": "1"}).status_code == 400

    def test_task_summarize(self, app):
        r = app.get(f"{self.BASE}/tasks/summarize").json()
        by_name = {e["func_or_class_name"]: e for e in
                   r["data"]["summary"]}
        assert by_name["f"]["state_counts"]["FINISHED"] == 1
        assert by_name["g"]["state_counts"]["RUNNING"] == 1
        assert r["data"]["total_tasks"] == 3

    def test_detail_endpoints(self, app):
        assert app.get(f"{self.BASE}/tasks/t1").json()["data"]["detail"][
            "task_id"] == "t1"
        assert app.get(f"{self.BASE}/jobs/j1").json()["data"]["detail"][
            "job_id"] == "j1"
        assert app.get(f"{self.BASE}/nodes/n1").json()["data"]["detail"][
            "node_id"] == "n1"
        assert app.get(f"{self.BASE}/tasks/zz").status_code == 404

    def test_log_listing_and_pagination(self, app):
        names = app.get(f"{self.BASE}/logs").json()["data"]["logs"]
        assert names == ["gcs.out", "raylet.out"]
        # tail
        r = app.get(f"{self.BASE}/logs/raylet.out", params={"lines": 2}).json()
        assert r["logs"] == "l4\nl5\n" and r["total_lines"] == 5
        # window
        r = app.get(f"{self.BASE}/logs/raylet.out",
                    params={"offset": 1, "lines": 2}).json()
        assert r["logs"] == "l2\nl3\n"


class TestClusterStatus:
    """cluster_status.go analog: autoscaler-style replayed summary."""

    @pytest.fixture()
    def app(self):
        storage = MemoryStorage()
        events = list(EVENTS) + [
            {"event_type": "NODE_DEFINITION_EVENT",
             "data": {"node_id": "n2", "node_ip": "10.0.0.2"}},
            {"event_type": "NODE_LIFECYCLE_EVENT",
             "data": {"node_id": "n2", "state": "DEAD"}},
            {"event_type": "TASK_DEFINITION_EVENT",
             "data": {"task_id": "tp", "name": "pending_f", "job_id": "j1"}},
            {"event_type": "TASK_LIFECYCLE_EVENT",
             "data": {"task_id": "tp", "state": "PENDING_NODE_ASSIGNMENT"}},
        ]
        collector = Collector(storage, "c1", namespace="ns1",
                              fetch_events=lambda: events,
                              fetch_logs=lambda: {})
        collector.push_once()
        return TestClient(create_history_app(storage))

    def test_status_counts_and_text(self, app):
        r = app.get("/api/sessions/ns1/c1/session-1/cluster_status").json()
        cs = r["data"]["clusterStatus"]
        assert cs["activeNodes"] == 1          # n1 alive, n2 dead
        assert cs["failedNodes"] == ["n2"]
        assert cs["pendingDemands"] == [{"resources": {"CPU": 1},
                                         "count": 1}]
        assert "Active: 1 node(s)" in cs["text"]
        assert "10.0.0.2" in cs["text"]

    def test_monitoring_health_stubs(self, app):
        assert app.get("/api/grafana_health").json()["result"] is False
        assert app.get("/api/prometheus_health").json()["result"] is False


def test_corrupt_event_batch_skipped():
    """A corrupt gzip batch must not abort replay of the whole session."""
    from kuberay_amd.historyserver.eventserver import load_session
    from kuberay_amd.historyserver.storage import MemoryStorage, compress
    storage = MemoryStorage()
    good = b'{"event_type": "DRIVER_JOB_DEFINITION_EVENT", "data": {"job_id": "j1"}}\n'
    storage.write("ns1/c1/s1/events/000.jsonl.gz", compress(good))
    storage.write("ns1/c1/s1/events/001.jsonl.gz", b"\x1f\x8bnot-gzip-data")
    storage.write("ns1/c1/s1/events/002.jsonl.gz", compress(
        b'{"event_type": "DRIVER_JOB_DEFINITION_EVENT", "data": {"job_id": "j2"}}\n'))
    state = load_session(storage, "ns1/c1/s1")
    assert set(state.jobs) == {"j1", "j2"}  # batch 001 skipped, not fatal
