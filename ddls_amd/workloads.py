"""Synthetic computation-graph workloads in pipedream profile format.

The reference trains on pipedream-profiled graphs ("small_graphs" set,
``env_dev.yaml:39-104``) which are not distributed with the repo.  This module
deterministically generates pipedream-format ``.txt`` profiles with the same
structure (mostly-chain DAGs with skip connections, forward/backward compute
times, activation/parameter sizes) and cost scales chosen so the SiP-ML rule
(quantum 0.01) yields partition degrees spanning 1..16.

``python -m ddls_amd.workloads <out_dir>`` writes the default 5-model set.
"""
from __future__ import annotations

import os
from typing import List, Optional, Tuple

import numpy as np

DEFAULT_MODELS = {
    # name: (num_fwd_nodes, num_skip_edges, compute_scale, seed)
    "synth_alexnet": (14, 0, 1.0, 1),
    "synth_vgg": (22, 0, 1.6, 2),
    "synth_resnext": (30, 6, 0.7, 3),
    "synth_gnmt": (24, 3, 1.2, 4),
    "synth_squeezenet": (10, 2, 0.5, 5),
}


def generate_model(name: str,
                   num_nodes: int,
                   num_skips: int = 0,
                   compute_scale: float = 1.0,
                   seed: int = 0) -> Tuple[List[Tuple[str, dict]], List[Tuple[str, str]]]:
    """Return (nodes, edges) in the shape of graphs.parse_pipedream_txt output."""
    rng = np.random.RandomState(seed)
    nodes, edges = [], []
    for i in range(1, num_nodes + 1):
        fwd = float(rng.uniform(0.004, 0.16)) * compute_scale
        bwd = 2.0 * fwd
        activation = float(rng.uniform(2e6, 4e8))
        parameter = float(rng.uniform(1e6, 2e8))
        nodes.append((str(i), {
            "type": f"Op{i}",
            "forward": round(fwd, 6),
            "backward": round(bwd, 6),
            "activation": round(activation, 1),
            "parameter": round(parameter, 1),
        }))
    for i in range(1, num_nodes):
        edges.append((str(i), str(i + 1)))
    # skip connections (keep DAG: src < dst)
    for _ in range(num_skips):
        src = int(rng.randint(1, num_nodes - 1))
        dst = int(rng.randint(src + 2, num_nodes + 1)) if src + 2 <= num_nodes else num_nodes
        if (str(src), str(dst)) not in edges and src < dst:
            edges.append((str(src), str(dst)))
    return nodes, edges


def write_pipedream_txt(path: str, nodes, edges):
    with open(path, "w") as f:
        for nid, ft in nodes:
            f.write(
                f"node{nid} -- {ft['type']}() -- "
                f"forward_compute_time={ft['forward']}, "
                f"backward_compute_time={ft['backward']}, "
                f"activation_size={ft['activation']}, "
                f"parameter_size={ft['parameter']}\n")
        for u, v in edges:
            f.write(f"node{u} -- node{v}\n")


def generate_default_set(out_dir: str):
    os.makedirs(out_dir, exist_ok=True)
    for name, (n, skips, scale, seed) in DEFAULT_MODELS.items():
        nodes, edges = generate_model(name, n, skips, scale, seed)
        write_pipedream_txt(os.path.join(out_dir, f"{name}.txt"), nodes, edges)


def default_data_dir() -> str:
    return os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                        "data", "small_graphs")


def ensure_default_set(out_dir: Optional[str] = None) -> str:
    out_dir = out_dir or default_data_dir()
    if not os.path.isdir(out_dir) or not os.listdir(out_dir):
        generate_default_set(out_dir)
    return out_dir


if __name__ == "__main__":
    import sys
    target = sys.argv[1] if len(sys.argv) > 1 else default_data_dir()
    generate_default_set(target)
    print(f"wrote synthetic pipedream profiles to {target}")


MEDIUM_MODELS = {
    # larger graphs near the 150-node observation cap (after fwd+bwd
    # mirroring a 70-fwd-node model has 140 obs nodes): stresses padding,
    # block placement and the batched lookahead at scale
    "synth_bert": (64, 8, 1.4, 11),
    "synth_resnet152": (75, 15, 0.9, 12),
    "synth_wideresnet": (70, 5, 1.8, 13),
}


def generate_medium_set(out_dir: str):
    os.makedirs(out_dir, exist_ok=True)
    for name, (n, skips, scale, seed) in MEDIUM_MODELS.items():
        nodes, edges = generate_model(name, n, skips, scale, seed)
        write_pipedream_txt(os.path.join(out_dir, f"{name}.txt"), nodes, edges)


def ensure_medium_set(out_dir: Optional[str] = None) -> str:
    out_dir = out_dir or os.path.join(
        os.path.dirname(default_data_dir()), "medium_graphs")
    if not os.path.isdir(out_dir) or not os.listdir(out_dir):
        generate_medium_set(out_dir)
    return out_dir
