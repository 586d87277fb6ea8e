#!/usr/bin/env python3
"""Hyperparameter sweep driver.

Replaces the reference's ``scripts/run_wandb_sweep.py`` (tmux-window-parallel
wandb sweep agents): a grid/random sweep over dotted config overrides, each
trial a subprocess of ``scripts/train.py``, results summarised into a
CSV/JSON table (``ddls_amd.runtime.metrics``).

Sweep spec (YAML):
    method: grid | random
    num_trials: 8            # random only
    parallel: 2              # concurrent trials
    base_overrides:          # applied to every trial
      - num_epochs=100
    parameters:
      algo.lr: [1e-4, 2.785e-4, 1e-3]
      algo.entropy_coeff: [0.001, 0.003]
"""
import argparse
import itertools
import json
import os
import random
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import yaml

from ddls_amd.runtime.metrics import summarise_runs, write_metrics_table


def trials_from_spec(spec: dict):
    params = spec["parameters"]
    keys = list(params.keys())
    if spec.get("method", "grid") == "grid":
        for combo in itertools.product(*[params[k] for k in keys]):
            yield dict(zip(keys, combo))
    else:
        for _ in range(spec.get("num_trials", 8)):
            yield {k: random.choice(params[k]) for k in keys}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("sweep_spec", help="YAML sweep spec")
    ap.add_argument("--config", default=os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "configs", "train_config.yaml"))
    ap.add_argument("--out-dir", default="runs/sweep")
    args = ap.parse_args()

    with open(args.sweep_spec) as f:
        spec = yaml.safe_load(f)
    os.makedirs(args.out_dir, exist_ok=True)

    trials = list(trials_from_spec(spec))
    parallel = spec.get("parallel", 1)
    # divide CPU threads across concurrent trials: with OMP_NUM_THREADS
    # unset every torch process spins ncpu OpenMP threads and parallel
    # trials livelock on spin-waits (measured 2 trials on 8 cores: 244 s
    # vs 4.4 s with the split)
    child_env = dict(os.environ)
    if "OMP_NUM_THREADS" not in child_env:
        child_env["OMP_NUM_THREADS"] = str(
            max(1, (os.cpu_count() or 1) // max(parallel, 1)))
    running, idx = [], 0
    while idx < len(trials) or running:
        while idx < len(trials) and len(running) < parallel:
            overrides = trials[idx]
            name = f"trial_{idx:03d}"
            cli = [str(o) for o in spec.get("base_overrides", [])]
            cli += [f"{k}={v}" for k, v in overrides.items()]
            cli += [f"experiment.name={name}",
                    f"experiment.path_to_save={args.out_dir}"]
            log = open(os.path.join(args.out_dir, f"{name}.log"), "w")
            p = subprocess.Popen(
                [sys.executable, os.path.join(os.path.dirname(__file__),
                                              "train.py"),
                 "--config", args.config] + cli,
                stdout=log, stderr=subprocess.STDOUT, env=child_env)
            with open(os.path.join(args.out_dir, f"{name}.json"), "w") as f:
                json.dump(overrides, f)
            running.append((name, p, log))
            idx += 1
        name, p, log = running.pop(0)
        rc = p.wait()
        log.close()
        print(f"{name}: exit {rc}", flush=True)

    rows = summarise_runs(args.out_dir)
    for row in rows:
        trial_json = os.path.join(args.out_dir, f"{row['run']}.json")
        if os.path.exists(trial_json):
            with open(trial_json) as f:
                row.update(json.load(f))
    write_metrics_table(rows, os.path.join(args.out_dir, "sweep_results"))
    print(json.dumps(rows, indent=2, default=str))


if __name__ == "__main__":
    main()
