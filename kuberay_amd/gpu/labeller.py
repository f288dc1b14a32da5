"""Node labeller: publishes xGMI topology as node labels.

The AMD device plugin ships a node labeller for device properties; this one
adds what gang scheduling needs (kuberay_amd/parallel XgmiGangScheduler's
``amd.com/xgmi-island`` topologyKey): each node is labelled with its island
signature so PodGroups land inside one fully-connected xGMI island.

Run as a DaemonSet (``python -m kuberay_amd.gpu.labeller --node $NODE_NAME``)
or in-process for tests.
"""
from __future__ import annotations

import os
from typing import Dict, Optional

from ..utils.constants import (AMD_GPU_COUNT_LABEL as GPU_COUNT_LABEL,
                               XGMI_FULLY_CONNECTED_LABEL,
                               XGMI_ISLAND_NODE_LABEL,
                               XGMI_LARGEST_ISLAND_LABEL)
from . import topology


def compute_node_labels(topo: Optional[topology.XgmiTopology],
                        node_name: str) -> Dict[str, str]:
    """Labels for this node given its discovered topology."""
    if topo is None or topo.num_gpus == 0:
        return {}
    islands = topo.islands()
    # single-island nodes (the 8xMI355X case) are one schedulable unit; a
    # split node advertises its largest island size so gangs can avoid it
    largest = max((len(i) for i in islands), default=0)
    return {
        XGMI_ISLAND_NODE_LABEL: f"{node_name}-island0",
        GPU_COUNT_LABEL: str(topo.num_gpus),
        XGMI_FULLY_CONNECTED_LABEL: "true" if topo.fully_connected() else "false",
        XGMI_LARGEST_ISLAND_LABEL: str(largest),
    }


def label_node(client, node_name: str,
               topo: Optional[topology.XgmiTopology] = None) -> Dict[str, str]:
    """Discover topology (if not given) and patch the Node's labels."""
    if topo is None:
        topo = topology.discover()
    labels = compute_node_labels(topo, node_name)
    if not labels:
        return {}
    patch = {"metadata": {"labels": labels}}
    raw_patch = getattr(client, "raw_patch", None)
    if raw_patch is not None:
        raw_patch("Node", "", node_name, patch)  # Nodes are cluster-scoped
    else:
        server = getattr(client, "server", None)
        if server is not None:
            server.patch_merge("Node", "default", node_name, patch)
    return labels


def main(argv=None) -> int:
    import argparse
    import json

    parser = argparse.ArgumentParser(prog="kuberay-amd-node-labeller")
    parser.add_argument("--node", default=os.environ.get("NODE_NAME", ""))
    parser.add_argument("--dry-run", action="store_true")
    args = parser.parse_args(argv)
    topo = topology.discover()
    labels = compute_node_labels(topo, args.node or "unknown-node")
    print(json.dumps(labels, indent=2))
    if not args.dry_run and args.node:
        from ..kube.rest import RestClient
        label_node(RestClient(), args.node, topo)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
