"""History server tests: storage, collector, event replay, HTTP API
(reference analogs: historyserver unit tests with fake storage;
eventserver.go state machines, timeline.go, log_event_reader.go)."""
import json

import pytest
from fastapi.testclient import TestClient

import kuberay_amd.features as features
from kuberay_amd.historyserver.collector import Collector, build_collector_container
from kuberay_amd.historyserver.eventserver import (
    SessionState,
    decode_event_file_bytes,
    extract_actor_id_from_task_id,
    load_session,
    merge_state_transitions,
    normalize_id,
)
from kuberay_amd.historyserver.server import create_history_app
from kuberay_amd.historyserver.storage import (
    LocalStorage,
    MemoryStorage,
    compress,
    decompress,
    storage_for,
)

# Reference-shaped export events (Ray export-event envelope; see
# eventserver.go storeEvent). IDs are plain strings that normalize_id keeps.
EVENTS = [
    {"eventType": "DRIVER_JOB_DEFINITION_EVENT",
     "driverJobDefinitionEvent": {"jobId": "0b", "entrypoint": "python t.py",
                                  "submissionId": "raysubmit_1"}},
    {"eventType": "DRIVER_JOB_LIFECYCLE_EVENT",
     "driverJobLifecycleEvent": {"jobId": "0b", "stateTransitions": [
         {"state": "CREATED", "timestamp": "2026-01-01T00:00:10Z"},
         {"state": "RUNNING", "timestamp": "2026-01-01T00:00:11Z"}]}},
    {"eventType": "TASK_DEFINITION_EVENT",
     "taskDefinitionEvent": {"taskId": "t1", "taskAttempt": 0,
                             "funcOrClassName": "f", "jobId": "0b"}},
    {"eventType": "TASK_LIFECYCLE_EVENT",
     "taskLifecycleEvent": {"taskId": "t1", "taskAttempt": 0,
                            "jobId": "0b", "stateTransitions": [
                                {"state": "PENDING_ARGS_AVAIL",
                                 "timestamp": "2026-01-01T00:00:12Z"},
                                {"state": "RUNNING",
                                 "timestamp": "2026-01-01T00:00:13Z"},
                                {"state": "FINISHED",
                                 "timestamp": "2026-01-01T00:00:15Z"}]}},
    {"eventType": "ACTOR_DEFINITION_EVENT",
     "actorDefinitionEvent": {"actorId": "a1", "className": "Worker",
                              "jobId": "0b"}},
    {"eventType": "ACTOR_LIFECYCLE_EVENT",
     "actorLifecycleEvent": {"actorId": "a1", "stateTransitions": [
         {"state": "ALIVE", "timestamp": "2026-01-01T00:00:12Z",
          "nodeId": "n1", "workerId": "w1"}]}},
    {"eventType": "NODE_DEFINITION_EVENT",
     "nodeDefinitionEvent": {"nodeId": "n1", "nodeIpAddress": "10.0.0.1"}},
    {"eventType": "NODE_LIFECYCLE_EVENT",
     "nodeLifecycleEvent": {"nodeId": "n1", "stateTransitions": [
         {"state": "ALIVE", "timestamp": "2026-01-01T00:00:01Z",
          "resources": {"CPU": 8, "GPU": 8}}]}},
    {"eventType": "TASK_PROFILE_EVENT",
     "taskProfileEvents": {
         "taskId": "t1", "jobId": "0b", "attemptNumber": 0,
         "profileEvents": {
             "componentId": "w1", "componentType": "worker",
             "nodeIpAddress": "10.0.0.1",
             "events": [{"eventName": "task::f",
                         "startTime": "100000000000",
                         "endTime": "100500000000",
                         "extraData": "{\"name\": \"f\"}"}]}}},
    {"eventType": "DRIVER_JOB_LIFECYCLE_EVENT",
     "driverJobLifecycleEvent": {"jobId": "0b", "stateTransitions": [
         {"state": "FINISHED", "timestamp": "2026-01-01T00:00:20Z"}]}},
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


class TestStorageConformance:
    """Interface conformance suite (storage/interface.go:10-17): every
    backend — including future S3/GCS/Azure/OSS ones — must pass these.
    Add a new backend by adding a row to BACKENDS."""

    BACKENDS = ["memory", "local"]

    @pytest.fixture(params=BACKENDS)
    def store(self, request, tmp_path):
        if request.param == "local":
            return LocalStorage(str(tmp_path))
        return MemoryStorage()

    def test_write_read_roundtrip(self, store):
        store.write("p/x.bin", b"\x00\x01binary\xff")
        assert store.read("p/x.bin") == b"\x00\x01binary\xff"

    def test_overwrite_replaces(self, store):
        store.write("p/x", b"one")
        store.write("p/x", b"two")
        assert store.read("p/x") == b"two"

    def test_append_accumulates(self, store):
        store.append("p/log", b"a")
        store.append("p/log", b"b")
        assert store.read("p/log") == b"ab"

    def test_exists(self, store):
        assert not store.exists("nope")
        store.write("yes", b"1")
        assert store.exists("yes")

    def test_read_missing_raises(self, store):
        with pytest.raises((FileNotFoundError, OSError)):
            store.read("missing/file")

    def test_list_prefix_sorted(self, store):
        store.write("pfx/b", b"1")
        store.write("pfx/a", b"2")
        store.write("other/c", b"3")
        listed = store.list("pfx")
        assert listed == sorted(listed)
        assert all(p.startswith("pfx") for p in listed)
        assert len(listed) == 2

    def test_list_empty_prefix_returns_all(self, store):
        store.write("a/1", b"x")
        store.write("b/2", b"y")
        assert len(store.list("")) >= 2

    def test_nested_paths(self, store):
        store.write("a/b/c/d/e.jsonl.gz", b"deep")
        assert store.read("a/b/c/d/e.jsonl.gz") == b"deep"
        assert "a/b/c/d/e.jsonl.gz" in store.list("a/b")


class TestIdNormalization:
    """utils.ConvertBase64ToHex analog."""

    def test_base64_to_hex(self):
        import base64
        raw = bytes(range(16))
        assert normalize_id(base64.b64encode(raw).decode()) == raw.hex()

    def test_hex_kept(self):
        assert normalize_id("deadbeef") == "deadbeef"
        assert normalize_id("DEADBEEF") == "deadbeef"

    def test_invalid_kept_verbatim(self):
        assert normalize_id("t1") == "t1"
        assert normalize_id("") == ""
        assert normalize_id(None) == ""

    def test_actor_id_extraction(self):
        # TaskID = 8B unique + 12B actor-unique + 4B job (timeline.go:231)
        tid = "00" * 8 + "ab" * 12 + "cd" * 4
        assert extract_actor_id_from_task_id(tid) == "ab" * 12 + "cd" * 4
        # all-Fs actor portion = no actor
        tid = "00" * 8 + "ff" * 12 + "cd" * 4
        assert extract_actor_id_from_task_id(tid) == ""
        assert extract_actor_id_from_task_id("short") == ""


class TestStateTransitionMerge:
    def test_dedup_and_sort(self):
        a = [{"state": "RUNNING", "timestamp": "2026-01-01T00:00:02Z"}]
        b = [{"state": "RUNNING", "timestamp": "2026-01-01T00:00:02Z"},
             {"state": "PENDING_ARGS_AVAIL",
              "timestamp": "2026-01-01T00:00:01Z"}]
        merged = merge_state_transitions(a, b)
        assert [t["state"] for t in merged] == ["PENDING_ARGS_AVAIL",
                                                "RUNNING"]


class TestTaskStateMachine:
    """eventserver.go:658-884 — attempts, out-of-order arrival, derived
    times, log-info merge."""

    def test_attempts_are_distinct_rows(self):
        s = SessionState()
        for attempt in (0, 1):
            s.apply({"eventType": "TASK_LIFECYCLE_EVENT",
                     "taskLifecycleEvent": {
                         "taskId": "t1", "taskAttempt": attempt,
                         "stateTransitions": [
                             {"state": "RUNNING",
                              "timestamp": f"2026-01-01T00:00:0{attempt+1}Z"}]}})
        assert set(s.tasks) == {"t1:0", "t1:1"}

    def test_lifecycle_before_definition_preserved(self):
        """Definition arriving AFTER lifecycle must not clobber state."""
        s = SessionState()
        s.apply({"eventType": "TASK_LIFECYCLE_EVENT",
                 "taskLifecycleEvent": {
                     "taskId": "t1", "taskAttempt": 0, "nodeId": "n1",
                     "workerPid": 42, "stateTransitions": [
                         {"state": "RUNNING",
                          "timestamp": "2026-01-01T00:00:01Z"}]}})
        s.apply({"eventType": "TASK_DEFINITION_EVENT",
                 "taskDefinitionEvent": {"taskId": "t1", "taskAttempt": 0,
                                         "funcOrClassName": "g"}})
        t = s.tasks["t1:0"]
        assert t["state"] == "RUNNING"
        assert t["nodeId"] == "n1"
        assert t["workerPid"] == 42
        assert t["funcOrClassName"] == "g"

    def test_derived_times(self):
        s = SessionState()
        s.apply(EVENTS[3])
        t = s.tasks["t1:0"]
        assert t["creationTime"] == "2026-01-01T00:00:12Z"
        assert t["startTime"] == "2026-01-01T00:00:13Z"
        assert t["endTime"] == "2026-01-01T00:00:15Z"
        assert t["state"] == "FINISHED"

    def test_duplicate_transitions_deduped(self):
        s = SessionState()
        s.apply(EVENTS[3])
        s.apply(EVENTS[3])  # replay the same batch twice
        assert len(s.tasks["t1:0"]["stateTransitions"]) == 3

    def test_actor_task_definition_sets_type(self):
        s = SessionState()
        s.apply({"eventType": "ACTOR_TASK_DEFINITION_EVENT",
                 "actorTaskDefinitionEvent": {"taskId": "t9",
                                              "actorId": "a1"}})
        assert s.tasks["t9:0"]["taskType"] == "ACTOR_TASK"

    def test_task_log_info_partial_merge(self):
        """mergeTaskLogStream: log-start and log-end arrive separately."""
        s = SessionState()
        s.apply({"eventType": "TASK_LIFECYCLE_EVENT",
                 "taskLifecycleEvent": {
                     "taskId": "t1",
                     "taskLogInfo": {"stdoutFile": "/logs/out.log",
                                     "stdoutStart": 10}}})
        s.apply({"eventType": "TASK_LIFECYCLE_EVENT",
                 "taskLifecycleEvent": {
                     "taskId": "t1",
                     "taskLogInfo": {"stdoutEnd": 99}}})
        info = s.tasks["t1:0"]["taskLogInfo"]
        assert info["stdoutFile"] == "/logs/out.log"
        assert info["stdoutStart"] == 10
        assert info["stdoutEnd"] == 99


class TestActorStateMachine:
    """eventserver.go:146-344 — address from ALIVE, restarts, death cause."""

    def _alive_then_dead(self):
        s = SessionState()
        s.apply({"eventType": "ACTOR_DEFINITION_EVENT",
                 "actorDefinitionEvent": {"actorId": "a1",
                                          "className": "Counter"}})
        s.apply({"eventType": "ACTOR_LIFECYCLE_EVENT",
                 "actorLifecycleEvent": {"actorId": "a1",
                                         "stateTransitions": [
                     {"state": "ALIVE", "timestamp": "2026-01-01T00:00:01Z",
                      "nodeId": "n1", "workerId": "w1",
                      "reprName": "Counter(idx=1)"},
                     {"state": "RESTARTING",
                      "timestamp": "2026-01-01T00:00:02Z"},
                     {"state": "ALIVE", "timestamp": "2026-01-01T00:00:03Z",
                      "nodeId": "n2", "workerId": "w2"},
                     {"state": "DEAD", "timestamp": "2026-01-01T00:00:04Z",
                      "deathCause": {"actorDiedErrorContext": {
                          "pid": 314, "nodeIpAddress": "10.0.0.2",
                          "errorMessage": "oom-killed"}}}]}})
        return s.actors["a1"]

    def test_full_lifecycle(self):
        a = self._alive_then_dead()
        assert a["state"] == "DEAD"
        assert a["className"] == "Counter"
        assert a["numRestarts"] == 1
        assert a["startTime"] == "2026-01-01T00:00:01Z"
        assert a["endTime"] == "2026-01-01T00:00:04Z"
        # address tracks the LAST ALIVE transition
        assert a["address"]["nodeId"] == "n2"
        assert a["address"]["workerId"] == "w2"
        # death cause parsed
        assert a["pid"] == 314
        assert a["address"]["ipAddress"] == "10.0.0.2"
        assert a["exitDetails"] == "oom-killed"

    def test_repr_name_tracked(self):
        s = SessionState()
        s.apply({"eventType": "ACTOR_LIFECYCLE_EVENT",
                 "actorLifecycleEvent": {"actorId": "a2",
                                         "stateTransitions": [
                     {"state": "ALIVE", "timestamp": "2026-01-01T00:00:01Z",
                      "reprName": "Worker(rank=3)"}]}})
        assert s.actors["a2"]["reprName"] == "Worker(rank=3)"

    def test_definition_after_lifecycle_preserves_state(self):
        s = SessionState()
        s.apply({"eventType": "ACTOR_LIFECYCLE_EVENT",
                 "actorLifecycleEvent": {"actorId": "a3",
                                         "stateTransitions": [
                     {"state": "ALIVE",
                      "timestamp": "2026-01-01T00:00:01Z"}]}})
        s.apply({"eventType": "ACTOR_DEFINITION_EVENT",
                 "actorDefinitionEvent": {"actorId": "a3",
                                          "className": "Late"}})
        assert s.actors["a3"]["state"] == "ALIVE"
        assert s.actors["a3"]["className"] == "Late"


class TestNodeStateMachine:
    def test_alive_resources_and_death_info(self):
        s = SessionState()
        s.apply({"eventType": "NODE_DEFINITION_EVENT",
                 "nodeDefinitionEvent": {"nodeId": "n1",
                                         "nodeIpAddress": "10.0.0.1",
                                         "labels": {"gpu": "mi355x"}}})
        s.apply({"eventType": "NODE_LIFECYCLE_EVENT",
                 "nodeLifecycleEvent": {"nodeId": "n1", "stateTransitions": [
                     {"state": "ALIVE", "timestamp": "2026-01-01T00:00:01Z",
                      "resources": {"CPU": 16, "GPU": 8}},
                     {"state": "DEAD", "timestamp": "2026-01-01T00:01:00Z",
                      "deathInfo": {"reason": "EXPECTED_TERMINATION",
                                    "reasonMessage": "scale-down"}}]}})
        n = s.nodes["n1"]
        assert n["state"] == "DEAD"
        assert n["resources"] == {"CPU": 16, "GPU": 8}
        assert n["deathInfo"]["reason"] == "EXPECTED_TERMINATION"
        assert n["endTime"] == "2026-01-01T00:01:00Z"
        assert n["labels"] == {"gpu": "mi355x"}


class TestJobStateMachine:
    def test_created_running_finished(self):
        s = SessionState()
        s.apply(EVENTS[0])
        s.apply(EVENTS[1])
        s.apply(EVENTS[9])
        j = s.jobs["0b"]
        assert j["state"] == "FINISHED"
        assert j["startTime"] == "2026-01-01T00:00:10Z"
        assert j["endTime"] == "2026-01-01T00:00:20Z"
        assert j["entrypoint"] == "python t.py"
        assert j["submissionId"] == "raysubmit_1"


class TestDecodeEventFile:
    def test_jsonl(self):
        raw = b'{"a": 1}\n\n{"b": 2}\nnot-json\n{"c": 3}\n'
        out = decode_event_file_bytes("f.jsonl", raw)
        assert len(out) == 3  # malformed line skipped, not fatal

    def test_legacy_json_array(self):
        out = decode_event_file_bytes("f", b'  [{"a": 1}, {"b": 2}]')
        assert len(out) == 2

    def test_empty(self):
        assert decode_event_file_bytes("f", b"  \n ") == []


class TestCollector:
    def test_push_and_replay(self):
        storage = MemoryStorage()
        batches = [EVENTS[:4], EVENTS[4:]]
        it = iter(batches)
        collector = Collector(storage, "c1", namespace="ns1",
                              fetch_events=lambda: next(it, []),
                              fetch_logs=lambda: {"raylet.out": "log line"})
        assert collector.push_once() == 4
        assert collector.push_once() == 6
        state = load_session(storage, "ns1/c1/session-1")
        assert state.jobs["0b"]["state"] == "FINISHED"
        assert state.tasks["t1:0"]["state"] == "FINISHED"
        assert state.actors["a1"]["state"] == "ALIVE"
        assert state.nodes["n1"]["nodeIpAddress"] == "10.0.0.1"
        # timeline: 1 process meta + 1 thread meta + 1 slice
        tl = state.timeline()
        assert len(tl) == 3

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
        from kuberay_amd.testing import simple_raycluster as src
        cluster = src("demo", historyServerOptions={"collectorOptions": {}})
        t = podlib.default_head_pod_template(cluster, cluster.spec.head_group_spec,
                                             "demo-head-", "6379")
        assert "history-collector" not in [c.name for c in t.spec.containers]


class TestEventCollector:
    """Disk-first receiver (eventcollector.go): categorization, rotation,
    upload, crash resume, drain, disk pressure."""

    def _collector(self, tmp_path, storage=None, **kw):
        from kuberay_amd.historyserver.collector import EventCollector
        return EventCollector(storage or MemoryStorage(), "c1",
                              namespace="ns1", node_id="n1",
                              data_dir=str(tmp_path), **kw)

    def test_categorize(self):
        from kuberay_amd.historyserver.collector import categorize
        assert categorize({"eventType": "NODE_LIFECYCLE_EVENT",
                           "nodeLifecycleEvent": {}}) == "node-events"
        assert categorize({"eventType": "TASK_LIFECYCLE_EVENT",
                           "taskLifecycleEvent": {"jobId": "0b"}}) == "job/0b"
        # unsafe job id falls back to the node category
        assert categorize({"eventType": "TASK_LIFECYCLE_EVENT",
                           "taskLifecycleEvent": {"jobId": "../../etc"}}) \
            == "node-events"

    def test_persist_rotate_upload_replay(self, tmp_path):
        storage = MemoryStorage()
        c = self._collector(tmp_path, storage)
        assert c.persist_events(EVENTS) == len(EVENTS)
        c.stop()  # drain: rotate + upload everything
        uploaded = storage.list("ns1/c1/session-1/events")
        assert uploaded, "nothing uploaded on drain"
        assert any("/job/0b/" in p for p in uploaded)
        assert any("/node-events/" in p for p in uploaded)
        assert all(p.endswith(".jsonl.gz") for p in uploaded)
        # the whole session replays from what the collector wrote
        state = load_session(storage, "ns1/c1/session-1")
        assert state.jobs["0b"]["state"] == "FINISHED"
        assert state.tasks["t1:0"]["state"] == "FINISHED"
        assert state.nodes["n1"]["state"] == "ALIVE"

    def test_size_rotation(self, tmp_path):
        storage = MemoryStorage()
        c = self._collector(tmp_path, storage, max_file_bytes=200)
        big = {"eventType": "DRIVER_JOB_DEFINITION_EVENT",
               "driverJobDefinitionEvent": {"jobId": "0b",
                                            "entrypoint": "x" * 300}}
        c.persist_events([big])  # exceeds max_file_bytes -> rotated inline
        assert storage.list("ns1/c1/session-1/events")
        c.stop()

    def test_rejects_while_draining(self, tmp_path):
        c = self._collector(tmp_path)
        c.stop()
        with pytest.raises(RuntimeError):
            c.persist_events([{"eventType": "NODE_DEFINITION_EVENT"}])

    def test_crash_resume(self, tmp_path):
        """Files a dead collector left on disk upload on next start."""
        storage = MemoryStorage()
        c1 = self._collector(tmp_path, storage)
        c1.persist_events(EVENTS[:3])
        # crash: no stop(), active file remains on disk
        assert not storage.list("ns1/c1/session-1/events")
        c2 = self._collector(tmp_path, storage)
        c2.start()
        try:
            assert storage.list("ns1/c1/session-1/events"), \
                "pending file not resumed"
        finally:
            c2.stop()

    def test_disk_pressure_drops(self, tmp_path):
        c = self._collector(tmp_path, max_disk_bytes=1)
        c.persist_events([{"eventType": "NODE_DEFINITION_EVENT",
                           "nodeDefinitionEvent": {"nodeId": "n1"}}] * 3)
        assert c.events_dropped >= 2

    def test_http_receiver(self, tmp_path):
        from kuberay_amd.historyserver.collector import create_receiver_app
        storage = MemoryStorage()
        c = self._collector(tmp_path, storage)
        app = TestClient(create_receiver_app(c))
        r = app.post("/v1/events", json=EVENTS)
        assert r.status_code == 200
        assert r.json()["accepted"] == len(EVENTS)
        c.stop()
        r = app.post("/v1/events", json=[{}])
        assert r.status_code == 503
        assert r.headers["Retry-After"] == "5"


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
        assert app.get(f"{base}/jobs").json()["data"]["jobs"][0]["jobId"] == "0b"
        assert len(app.get(f"{base}/tasks").json()["data"]["tasks"]) == 1
        assert len(app.get(f"{base}/actors").json()["data"]["actors"]) == 1
        assert len(app.get(f"{base}/nodes").json()["data"]["nodes"]) == 1

    def test_timeline_trace_format(self, app):
        tl = app.get("/api/sessions/ns1/c1/session-1/timeline").json()
        slices = [e for e in tl if e["ph"] == "X"]
        metas = [e for e in tl if e["ph"] == "M"]
        assert len(slices) == 1
        assert slices[0]["dur"] == pytest.approx(0.5e6)  # 0.5 s in µs
        assert {m["name"] for m in metas} == {"process_name", "thread_name"}

    def test_tasks_timeline_job_filter(self, app):
        base = "/api/sessions/ns1/c1/session-1"
        assert app.get(f"{base}/tasks/timeline",
                       params={"job_id": "0b"}).json()
        assert app.get(f"{base}/tasks/timeline",
                       params={"job_id": "zz"}).json() == []

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
            {"eventType": "TASK_DEFINITION_EVENT",
             "taskDefinitionEvent": {"taskId": "t2", "funcOrClassName": "g",
                                     "jobId": "0b"}},
            {"eventType": "TASK_LIFECYCLE_EVENT",
             "taskLifecycleEvent": {"taskId": "t2", "stateTransitions": [
                 {"state": "RUNNING", "timestamp": "2026-01-01T00:00:16Z"}]}},
            {"eventType": "TASK_DEFINITION_EVENT",
             "taskDefinitionEvent": {"taskId": "t3", "funcOrClassName": "f",
                                     "jobId": "0c"}},
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
            "filter_keys": "jobId", "filter_values": "0b"})
        data = r.json()["data"]
        assert {t["taskId"] for t in data["tasks"]} == {"t1", "t2"}
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
            "taskId"] == "t1"
        assert app.get(f"{self.BASE}/jobs/0b").json()["data"]["detail"][
            "jobId"] == "0b"
        assert app.get(f"{self.BASE}/nodes/n1").json()["data"]["detail"][
            "nodeId"] == "n1"
        assert app.get(f"{self.BASE}/actors/a1").json()["data"]["detail"][
            "actorId"] == "a1"
        assert app.get(f"{self.BASE}/tasks/zz").status_code == 404

    def test_task_detail_groups_attempts(self, app):
        detail = app.get(f"{self.BASE}/tasks/t1").json()["data"]["detail"]
        assert [a["taskAttempt"] for a in detail["attempts"]] == [0]

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
            {"eventType": "NODE_DEFINITION_EVENT",
             "nodeDefinitionEvent": {"nodeId": "n2",
                                     "nodeIpAddress": "10.0.0.2"}},
            {"eventType": "NODE_LIFECYCLE_EVENT",
             "nodeLifecycleEvent": {"nodeId": "n2", "stateTransitions": [
                 {"state": "DEAD", "timestamp": "2026-01-01T00:00:30Z"}]}},
            {"eventType": "TASK_DEFINITION_EVENT",
             "taskDefinitionEvent": {"taskId": "tp",
                                     "funcOrClassName": "pending_f",
                                     "jobId": "0b"}},
            {"eventType": "TASK_LIFECYCLE_EVENT",
             "taskLifecycleEvent": {"taskId": "tp", "stateTransitions": [
                 {"state": "PENDING_NODE_ASSIGNMENT",
                  "timestamp": "2026-01-01T00:00:31Z"}]}},
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


class TestLogEvents:
    """log_event_reader.go analog: logs/{node}/events/event_*.log →
    /events API grouped by job."""

    def test_log_events_replayed_and_served(self):
        storage = MemoryStorage()
        lines = [
            {"event_id": "e1", "source_type": "GCS", "severity": "INFO",
             "message": "job started", "timestamp": 100,
             "custom_fields": {"job_id": "0b"}},
            {"event_id": "e2", "source_type": "RAYLET", "severity": "ERROR",
             "message": "worker died", "timestamp": 101,
             "custom_fields": {"job_id": "0b"}},
            {"event_id": "e3", "source_type": "GCS", "severity": "INFO",
             "message": "no job", "timestamp": 102},
        ]
        payload = "\n".join(json.dumps(l) for l in lines).encode()
        storage.write("ns1/c1/s1/logs/n1/events/event_GCS.log", payload)
        state = load_session(storage, "ns1/c1/s1")
        assert len(state.log_events["0b"]) == 2
        assert state.log_events["0b"][1]["severity"] == "ERROR"
        assert len(state.log_events["global"]) == 1

        app = TestClient(create_history_app(storage))
        r = app.get("/api/sessions/ns1/c1/s1/events",
                    params={"job_id": "0b"}).json()
        assert len(r["data"]["events"]["0b"]) == 2


class TestGoldenSession:
    """VERDICT r1 item 4 done-criterion: a golden session (events file
    checked in) replays to golden API responses incl. timeline."""

    FIXTURE = "tests/data/golden_session_events.jsonl"
    GOLDEN = "tests/data/golden_session_expected.json"

    @pytest.fixture()
    def app(self):
        import os
        storage = MemoryStorage()
        here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        with open(os.path.join(here, self.FIXTURE), "rb") as f:
            storage.write("ns1/golden/session-1/events/000.jsonl", f.read())
        return TestClient(create_history_app(storage))

    def _golden(self):
        import os
        here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        with open(os.path.join(here, self.GOLDEN)) as f:
            return json.load(f)

    def test_replay_matches_golden(self, app):
        golden = self._golden()
        base = "/api/sessions/ns1/golden/session-1"
        for endpoint, want in golden.items():
            got = app.get(f"{base}/{endpoint}").json()
            assert got == want, f"golden mismatch for /{endpoint}"


def test_corrupt_event_batch_skipped():
    """A corrupt gzip batch must not abort replay of the whole session."""
    storage = MemoryStorage()
    good = (b'{"eventType": "DRIVER_JOB_DEFINITION_EVENT", '
            b'"driverJobDefinitionEvent": {"jobId": "0b"}}\n')
    storage.write("ns1/c1/s1/events/000.jsonl.gz", compress(good))
    storage.write("ns1/c1/s1/events/001.jsonl.gz", b"\x1f\x8bnot-gzip-data")
    storage.write("ns1/c1/s1/events/002.jsonl.gz", compress(
        b'{"eventType": "DRIVER_JOB_DEFINITION_EVENT", '
        b'"driverJobDefinitionEvent": {"jobId": "0c"}}\n'))
    state = load_session(storage, "ns1/c1/s1")
    assert set(state.jobs) == {"0b", "0c"}  # batch 001 skipped, not fatal
