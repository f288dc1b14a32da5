"""Operator configuration (reference: ray-operator/apis/config/v1alpha1/
configuration_types.go:18-90 + defaults.go:8-13 + main.go:80-143).

Flags mirror into a versioned Configuration object loadable from a YAML
file (``--config``); explicit flags win over the file.
"""
from __future__ import annotations

import argparse
from typing import List, Optional

import yaml
from .kube.objects import K8sModel


class Configuration(K8sModel):
    api_version: str = "config.ray.io/v1alpha1"
    kind: str = "Configuration"

    metrics_addr: str = ":8080"
    probe_addr: str = ":8082"
    enable_leader_election: bool = True
    leader_election_namespace: str = ""
    leader_lease_seconds: float = 15.0
    # horizontal sharding: run N operator processes against one cluster,
    # each owning CRs whose crc32(ns/name) %% shards == shard_index; every
    # shard elects its own leader Lease (kuberay-amd-operator-shard-I)
    shards: int = 1
    shard_index: int = 0
    reconcile_concurrency: int = 4          # reference default 1; Python
                                            # reconcilers are pure cache fns
    watch_namespaces: Optional[List[str]] = None
    log_file: Optional[str] = None
    log_file_encoder: str = "json"
    log_stdout_encoder: str = "console"
    batch_scheduler: Optional[str] = None   # volcano|yunikorn|scheduler-plugins|xgmi-gang
    enable_batch_scheduler: bool = False
    head_sidecar_containers: Optional[List[dict]] = None
    worker_sidecar_containers: Optional[List[dict]] = None
    feature_gates: str = ""
    qps: float = 100.0
    burst: int = 200
    enable_metrics: bool = True
    # MI355X-native knobs
    enable_mi355x_autoscaler: bool = True
    mi355x_autoscaler_interval_s: float = 5.0
    # backend: "memory" (self-contained control plane, tests/bench/local) or
    # "kubernetes" (real cluster via REST)
    backend: str = "memory"
    kubeconfig: Optional[str] = None
    # memory backend durability: JSONL snapshot path ("" = off)
    state_file: str = ""
    state_snapshot_interval_s: float = 30.0
    # memory backend: serve the kube-API facade on this port (0 = off) so
    # kubectl-style tooling / kray --server can target the control plane
    api_port: int = 0


def load_config(argv: Optional[List[str]] = None) -> Configuration:
    parser = argparse.ArgumentParser(prog="kuberay-amd-operator")
    parser.add_argument("--config", help="YAML Configuration file")
    parser.add_argument("--metrics-addr")
    parser.add_argument("--probe-addr")
    parser.add_argument("--reconcile-concurrency", type=int)
    parser.add_argument("--watch-namespace", action="append", dest="watch_namespaces")
    parser.add_argument("--snapshot-interval", type=float,
                        dest="state_snapshot_interval_s")
    parser.add_argument("--shards", type=int)
    parser.add_argument("--shard-index", type=int)
    parser.add_argument("--batch-scheduler")
    parser.add_argument("--enable-batch-scheduler", action="store_true", default=None)
    parser.add_argument("--feature-gates", default=None)
    parser.add_argument("--backend", choices=["memory", "kubernetes"])
    parser.add_argument("--kubeconfig")
    parser.add_argument("--log-file")
    parser.add_argument("--state-file")
    parser.add_argument("--api-port", type=int, default=None)
    parser.add_argument("--leader-lease-seconds", type=float, default=None)
    parser.add_argument("--no-metrics", action="store_true", default=None)
    args = parser.parse_args(argv)

    data = {}
    if args.config:
        with open(args.config) as f:
            data = yaml.safe_load(f) or {}
    cfg = Configuration.from_dict(data)

    for flag, attr in [
        ("metrics_addr", "metrics_addr"), ("probe_addr", "probe_addr"),
        ("reconcile_concurrency", "reconcile_concurrency"),
        ("shards", "shards"), ("shard_index", "shard_index"),
        ("state_snapshot_interval_s", "state_snapshot_interval_s"),
        ("watch_namespaces", "watch_namespaces"),
        ("batch_scheduler", "batch_scheduler"),
        ("enable_batch_scheduler", "enable_batch_scheduler"),
        ("feature_gates", "feature_gates"),
        ("backend", "backend"), ("kubeconfig", "kubeconfig"),
        ("log_file", "log_file"), ("state_file", "state_file"),
        ("api_port", "api_port"),
        ("leader_lease_seconds", "leader_lease_seconds"),
    ]:
        val = getattr(args, flag, None)
        if val is not None:
            setattr(cfg, attr, val)
    if args.no_metrics:
        cfg.enable_metrics = False
    return cfg
