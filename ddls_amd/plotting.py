"""Plotting utilities (reference ``ddls/plotting/plotting.py``: computation
graph rendering + paper-style metric charts).  Pure matplotlib over the flat
CompGraph structure (no networkx/graphviz dependency); all functions return
the figure so callers can save or show.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict, List, Optional, Sequence

import numpy as np

from .graphs import BWD, CompGraph


def _require_matplotlib():
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        return plt
    except ImportError as e:  # pragma: no cover
        raise ImportError("matplotlib is required for plotting") from e


def plot_computation_graph(graph: CompGraph, figsize=(10, 8),
                           label_nodes: bool = True,
                           node_size_by_memory: bool = True):
    """Layered DAG rendering: x = BFS depth, y = index within layer; forward
    ops blue, backward ops red; edge width ~ log(size)."""
    plt = _require_matplotlib()
    depth = graph.node_depths()
    layers = defaultdict(list)
    for i in range(graph.n):
        layers[int(depth[i])].append(i)
    pos = {}
    for d, nodes in layers.items():
        for k, i in enumerate(nodes):
            pos[i] = (d, k - (len(nodes) - 1) / 2)

    fig, ax = plt.subplots(figsize=figsize)
    for e in range(graph.m):
        u, v = int(graph.src[e]), int(graph.dst[e])
        x0, y0 = pos[u]
        x1, y1 = pos[v]
        lw = 0.5 + 0.3 * np.log10(1 + graph.size[e])
        ax.annotate("", xy=(x1, y1), xytext=(x0, y0),
                    arrowprops=dict(arrowstyle="-|>", lw=min(lw, 3),
                                    color="gray", alpha=0.5))
    xs = [pos[i][0] for i in range(graph.n)]
    ys = [pos[i][1] for i in range(graph.n)]
    if node_size_by_memory and graph.memory_cost.max() > 0:
        sizes = 60 + 240 * graph.memory_cost / graph.memory_cost.max()
    else:
        sizes = np.full(graph.n, 120.0)
    colors = ["#d62728" if graph.pass_type[i] == BWD else "#1f77b4"
              for i in range(graph.n)]
    ax.scatter(xs, ys, s=sizes, c=colors, zorder=3, edgecolors="black",
               linewidths=0.5)
    if label_nodes:
        for i in range(graph.n):
            ax.annotate(graph.names[i], pos[i], fontsize=7, ha="center",
                        va="center", zorder=4)
    ax.set_xlabel("depth")
    ax.set_yticks([])
    ax.set_title(f"{graph.model}: {graph.n} ops, {graph.m} deps")
    fig.tight_layout()
    return fig


def plot_training_curves(results_log: Dict[str, List], metrics: Optional[Sequence[str]] = None,
                         figsize=(12, 6)):
    """Grid of training-metric curves from a Launcher results log."""
    plt = _require_matplotlib()
    if metrics is None:
        metrics = [k for k in ("mean_reward", "episode_reward_mean", "kl",
                               "entropy", "total_loss", "blocking_rate_mean")
                   if k in results_log and len(results_log[k]) > 0]
    n = max(1, len(metrics))
    ncols = min(3, n)
    nrows = (n + ncols - 1) // ncols
    fig, axes = plt.subplots(nrows, ncols, figsize=figsize, squeeze=False)
    for k, metric in enumerate(metrics):
        ax = axes[k // ncols][k % ncols]
        ax.plot(results_log[metric])
        ax.set_title(metric)
        ax.set_xlabel("epoch")
    fig.tight_layout()
    return fig


def plot_episode_stats_comparison(actor_to_stats: Dict[str, dict],
                                  metrics: Sequence[str] = (
                                      "blocking_rate", "acceptance_rate",
                                      "mean_job_completion_time"),
                                  figsize=(12, 4)):
    """Bar chart comparing actors on episode-level metrics (paper figures)."""
    plt = _require_matplotlib()
    fig, axes = plt.subplots(1, len(metrics), figsize=figsize, squeeze=False)
    actors = list(actor_to_stats.keys())
    for k, metric in enumerate(metrics):
        ax = axes[0][k]
        vals = [actor_to_stats[a].get(metric) or 0 for a in actors]
        ax.bar(range(len(actors)), vals)
        ax.set_xticks(range(len(actors)))
        ax.set_xticklabels(actors, rotation=45, ha="right", fontsize=8)
        ax.set_title(metric)
    fig.tight_layout()
    return fig


def computation_graph_to_dot(graph: CompGraph) -> str:
    """Graphviz-dot export of a computation graph (usable without matplotlib;
    reference renders via nx.draw, ``plotting/plotting.py:95-115``)."""
    lines = [f'digraph "{graph.model}" {{', "  rankdir=LR;"]
    for i in range(graph.n):
        color = "indianred" if graph.pass_type[i] == BWD else "steelblue"
        lines.append(
            f'  "{graph.names[i]}" [style=filled, fillcolor={color}, '
            f'label="{graph.names[i]}\\nmem={float(graph.memory_cost[i]):.3g}"];')
    for e in range(graph.m):
        u, v = graph.names[int(graph.src[e])], graph.names[int(graph.dst[e])]
        lines.append(f'  "{u}" -> "{v}" [label="{float(graph.size[e]):.3g}"];')
    lines.append("}")
    return "\n".join(lines)


def plot_metric_distributions(actor_to_values: Dict[str, Sequence[float]],
                              metric_name: str = "job_completion_time",
                              bins: int = 30, cdf: bool = True,
                              figsize=(10, 4)):
    """Histogram + CDF of a per-job metric across actors (paper-style
    seaborn hist/ecdf figures, reference ``plotting/plotting.py``)."""
    plt = _require_matplotlib()
    fig, axes = plt.subplots(1, 2 if cdf else 1, figsize=figsize,
                             squeeze=False)
    for actor, vals in actor_to_values.items():
        vals = np.asarray(list(vals), dtype=float)
        if len(vals) == 0:
            continue
        axes[0][0].hist(vals, bins=bins, alpha=0.5, label=actor)
        if cdf:
            xs = np.sort(vals)
            axes[0][1].plot(xs, np.arange(1, len(xs) + 1) / len(xs),
                            label=actor)
    axes[0][0].set_xlabel(metric_name)
    axes[0][0].set_ylabel("count")
    axes[0][0].legend(fontsize=8)
    if cdf:
        axes[0][1].set_xlabel(metric_name)
        axes[0][1].set_ylabel("CDF")
        axes[0][1].legend(fontsize=8)
    fig.tight_layout()
    return fig


def plot_cluster_timeline(steps_log: Dict[str, Sequence[float]],
                          metrics: Sequence[str] = (
                              "num_jobs_running", "num_mounted_ops",
                              "mean_num_active_workers"),
                          figsize=(12, 4)):
    """Per-cluster-step utilisation curves from the cluster ``steps_log``."""
    plt = _require_matplotlib()
    avail = [m for m in metrics if m in steps_log and len(steps_log[m]) > 0]
    n = max(1, len(avail))
    fig, axes = plt.subplots(1, n, figsize=figsize, squeeze=False)
    for k, metric in enumerate(avail):
        axes[0][k].plot(steps_log[metric])
        axes[0][k].set_title(metric)
        axes[0][k].set_xlabel("cluster step")
    fig.tight_layout()
    return fig
