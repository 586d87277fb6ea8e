"""Results analysis utilities.

Reference: ``ddls/environments/ramp_cluster/utils.py`` — the RLlib callback
that forwards cluster step/episode stats into custom metrics (:25-73), the
custom eval function (:75-127) and the wandb results loaders (:129-473).
This rebuild has no wandb; results live in the Logger's gzip-pkl/sqlite files
and these helpers load, summarise and tabulate them.
"""
from __future__ import annotations

import csv
import glob
import json
import os
from typing import Dict, List, Sequence

import numpy as np


def harvest_episode_stats(env) -> dict:
    """Forward a finished env's cluster episode stats into a flat metrics dict
    (the reference callback's on_episode_end behaviour)."""
    from ..cluster.environment import RampClusterEnvironment
    stats = env.cluster.episode_stats
    out = {}
    for key in RampClusterEnvironment.episode_metrics():
        if key in stats:
            val = stats[key]
            if isinstance(val, list):
                out[f"mean_{key}"] = float(np.mean(val)) if val else None
            else:
                out[key] = val
    for key in (RampClusterEnvironment.episode_completion_metrics()
                | RampClusterEnvironment.episode_blocked_metrics()):
        val = stats.get(key)
        if isinstance(val, list) and val:
            out[f"mean_{key}"] = float(np.mean(val))
    return out


def load_training_run(run_dir: str) -> Dict[str, List]:
    """Load a Launcher training log (train_log.pkl / .sqlite) from a run dir."""
    from .logger import Logger
    for sqlite in (False, True):
        try:
            return Logger(run_dir, use_sqlite_database=sqlite).load("train_log")
        except FileNotFoundError:
            continue
    raise FileNotFoundError(f"no train_log in {run_dir}")


def summarise_runs(base_dir: str) -> List[dict]:
    """Summarise every run under base_dir: final/best reward, epochs, etc."""
    rows = []
    for run_dir in sorted(glob.glob(os.path.join(base_dir, "*"))):
        if not os.path.isdir(run_dir):
            continue
        try:
            log = load_training_run(run_dir)
        except (FileNotFoundError, Exception):
            continue
        row = {"run": os.path.basename(run_dir)}
        rewards = log.get("episode_reward_mean") or log.get("mean_reward") or []
        if rewards:
            row["final_reward"] = rewards[-1]
            row["best_reward"] = max(rewards)
        row["epochs"] = len(log.get("epoch_counter", []))
        rows.append(row)
    return rows


def write_metrics_table(rows: Sequence[dict], path: str):
    """Persist a metric table as CSV + JSON (the wandb-tables stand-in)."""
    if not rows:
        return
    keys = sorted({k for r in rows for k in r})
    with open(path + ".csv", "w", newline="") as f:
        w = csv.DictWriter(f, fieldnames=keys)
        w.writeheader()
        for r in rows:
            w.writerow(r)
    with open(path + ".json", "w") as f:
        json.dump(list(rows), f, indent=2, default=str)
