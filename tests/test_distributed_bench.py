"""Multi-process bench-path test: world_size=2 over gloo on CPU (the same
aggregation code the driver exercises at N=1..8 over RCCL on the GPU node).
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class TestBenchSingleProcess:
    def test_bench_contract_json(self):
        out = subprocess.run(
            [sys.executable, "bench.py", "--clusters", "20", "--steps", "1",
             "--warmup", "0", "--workers-per-cluster", "1"],
            capture_output=True, text=True, cwd=REPO, timeout=300)
        assert out.returncode == 0, out.stderr[-2000:]
        line = out.stdout.strip().splitlines()[-1]
        d = json.loads(line)
        assert d["metric"] == "rayclusters_to_ready_per_sec"
        assert d["value"] > 0
        assert d["higher_is_better"] is True
        assert d["scaling"] == "weak"
        assert d["n_gpus"] == 1
        assert d["config"]["p50_cr_to_ready_s"] > 0
        assert d["vs_baseline"] > 0


@pytest.mark.timeout(600)
class TestBenchDistributed:
    def test_world2_gloo(self):
        """Two ranks, gloo backend, 127.0.0.1 rendezvous — mirrors the
        driver's torch.distributed.run launch."""
        out = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", "29517", "bench.py", "--clusters", "15",
             "--steps", "1", "--warmup", "0", "--workers-per-cluster", "1"],
            capture_output=True, text=True, cwd=REPO, timeout=540,
            env={**os.environ, "MASTER_ADDR": "127.0.0.1"})
        assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
        lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
        assert len(lines) == 1, "exactly one JSON line from rank 0"
        d = json.loads(lines[0])
        assert d["n_gpus"] == 2
        # whole-job aggregate: 2 ranks x 15 clusters
        assert d["config"]["global_batch"] == 30
        assert d["config"]["parallelism"] == "dp2"

    def test_world8_gloo(self):
        """Eight ranks with tiny per-rank work — pre-proves rendezvous and
        MAX/SUM aggregation at the driver's full N=8 scaling point so
        `bench.py --gpus 8` just works when the driver gets an 8-GPU node."""
        out = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
             "--master-port", "29521", "bench.py", "--clusters", "4",
             "--steps", "1", "--warmup", "0", "--workers-per-cluster", "1",
             "--controller-workers", "2"],
            capture_output=True, text=True, cwd=REPO, timeout=540,
            env={**os.environ, "MASTER_ADDR": "127.0.0.1"})
        assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
        lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
        assert len(lines) == 1
        d = json.loads(lines[0])
        assert d["n_gpus"] == 8
        assert d["config"]["global_batch"] == 32  # 8 ranks x 4 clusters
        assert d["config"]["parallelism"] == "dp8"

    def test_world4_gloo(self):
        """Four ranks — same shape the driver uses for the N=4 scaling
        point; validates rendezvous + MAX/SUM aggregation at higher fan-out."""
        out = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
             "--master-port", "29519", "bench.py", "--clusters", "8",
             "--steps", "1", "--warmup", "0", "--workers-per-cluster", "1"],
            capture_output=True, text=True, cwd=REPO, timeout=540,
            env={**os.environ, "MASTER_ADDR": "127.0.0.1"})
        assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
        lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
        assert len(lines) == 1
        d = json.loads(lines[0])
        assert d["n_gpus"] == 4
        assert d["config"]["global_batch"] == 32  # 4 ranks x 8 clusters
        assert d["config"]["parallelism"] == "dp4"
