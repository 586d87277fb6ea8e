"""YAML config groups with merging and dotted overrides (hydra-lite).

Reference config system: hydra + OmegaConf groups under
``scripts/ramp_job_partitioning_configs/`` merged in
``train_rllib_from_config.py:46-64``, with ``_target_`` class instantiation
and dotted-key overrides (wandb sweeps).  This rebuild implements the same
surface with yaml + a small merger: a top-level config names its groups
(``defaults``), each group file lives at ``<config_dir>/<group>/<name>.yaml``,
and CLI overrides use ``a.b.c=value``.
"""
from __future__ import annotations

import copy
import os
from typing import Any, Dict, List, Optional

import yaml


def load_yaml(path: str) -> dict:
    with open(path) as f:
        return yaml.safe_load(f) or {}


def deep_merge(base: dict, override: dict) -> dict:
    out = copy.deepcopy(base)
    for k, v in override.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = deep_merge(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


def _parse_value(s: str) -> Any:
    try:
        v = yaml.safe_load(s)
    except yaml.YAMLError:
        return s
    # YAML 1.1 leaves exponent-only floats ("1e-4") as strings; overrides
    # like algo.lr=1e-4 must land as numbers
    if isinstance(v, str):
        try:
            return float(v)
        except ValueError:
            return v
    return v


def apply_dotted_overrides(cfg: dict, overrides: List[str]) -> dict:
    cfg = copy.deepcopy(cfg)
    for ov in overrides:
        if "=" not in ov:
            raise ValueError(f"override '{ov}' must be key.path=value")
        key, _, raw = ov.partition("=")
        node = cfg
        parts = key.split(".")
        for p in parts[:-1]:
            node = node.setdefault(p, {})
        node[parts[-1]] = _parse_value(raw)
    return cfg


def load_config(config_path: str,
                overrides: Optional[List[str]] = None) -> dict:
    """Load a top-level config, resolving its ``defaults`` group list
    (each entry ``{group: name}`` pulls ``<dir>/<group>/<name>.yaml`` into
    ``cfg[group]``), then apply dotted overrides."""
    config_dir = os.path.dirname(os.path.abspath(config_path))
    cfg = load_yaml(config_path)
    defaults = cfg.pop("defaults", [])
    resolved: Dict[str, Any] = {}
    for entry in defaults:
        if isinstance(entry, dict):
            for group, name in entry.items():
                group_cfg = load_yaml(os.path.join(config_dir, group,
                                                   f"{name}.yaml"))
                resolved[group] = group_cfg
        else:
            resolved = deep_merge(resolved, load_yaml(
                os.path.join(config_dir, f"{entry}.yaml")))
    cfg = deep_merge(resolved, cfg)
    if overrides:
        # hydra-style GROUP overrides first: "env_config=env_1024worker"
        # swaps in <config_dir>/env_config/env_1024worker.yaml
        dotted = []
        for ov in overrides:
            key, _, raw = ov.partition("=")
            group_file = os.path.join(config_dir, key, f"{raw}.yaml")
            if "." not in key and "=" in ov and os.path.isfile(group_file):
                cfg[key] = load_yaml(group_file)
            else:
                dotted.append(ov)
        cfg = apply_dotted_overrides(cfg, dotted)
    return cfg


# ---------------------------------------------------------------------------
# factories
# ---------------------------------------------------------------------------

def build_env_from_config(cfg: dict):
    """Build a RampJobPartitioningEnvironment from cfg['env_config']."""
    from ..envs import RampJobPartitioningEnvironment
    env_cfg = copy.deepcopy(cfg["env_config"])
    jobs_config = env_cfg.get("jobs_config", {})
    if jobs_config.get("path_to_files") in (None, "default", "synthetic"):
        from ..workloads import ensure_default_set
        jobs_config["path_to_files"] = ensure_default_set()
    elif jobs_config.get("path_to_files") == "medium":
        from ..workloads import ensure_medium_set
        jobs_config["path_to_files"] = ensure_medium_set()
    return RampJobPartitioningEnvironment(
        topology_config=env_cfg["topology_config"],
        node_config=env_cfg["node_config"],
        jobs_config=jobs_config,
        max_partitions_per_op=env_cfg.get("max_partitions_per_op"),
        min_op_run_time_quantum=env_cfg.get("min_op_run_time_quantum", 0.01),
        pad_obs_kwargs=env_cfg.get("pad_obs_kwargs"),
        reward_function=env_cfg.get("reward_function",
                                    "lookahead_job_completion_time"),
        reward_function_kwargs=env_cfg.get("reward_function_kwargs"),
        max_simulation_run_time=env_cfg.get("max_simulation_run_time"),
        job_queue_capacity=env_cfg.get("job_queue_capacity", 10))


def build_trainer_from_config(cfg: dict, device=None):
    import torch

    from ..models.gnn import GNNPolicy
    from ..rl.ppo import PPOConfig, PPOTrainer
    from ..rl.subproc_env import SubprocVectorEnv

    algo = cfg.get("algo", {})
    model_cfg = cfg.get("model", {})
    loop_cfg = cfg.get("epoch_loop", {})
    n_envs = loop_cfg.get("num_envs", 8)
    n_workers = loop_cfg.get("num_env_workers") or max(
        1, min(n_envs, (os.cpu_count() or 2) - 1))
    seed = cfg.get("seed", 0)

    def env_fn():
        env = build_env_from_config(cfg)
        env.reuse_jobs_generator = True
        return env

    # env backend: "engine" = vectorised engine (GPU kernel on cuda,
    # CpuEngine mirror on cpu), "subproc" = fork-based env workers,
    # "auto" (default) = engine on cuda, subprocess on cpu (historical
    # default; the engine supports 1-channel RAMP + the non-throughput
    # rewards — unsupported configs raise with a clear message)
    backend = loop_cfg.get("env_backend", "auto")
    use_cuda = (device is not None and str(device) != "cpu"
                and torch.cuda.is_available())
    use_engine = (backend == "engine"
                  or (backend == "auto" and use_cuda))
    if use_engine:
        from ..rl.engine_env import EngineVectorEnv
        try:
            venv = EngineVectorEnv(env_fn, num_envs=n_envs,
                                   device=device if use_cuda
                                   else torch.device("cpu"),
                                   base_seed=seed)
        except ValueError:
            # engine-unsupported config (throughput reward, multi-channel,
            # non-RAMP): explicit request surfaces the error, auto falls
            # back to the subprocess path
            if backend == "engine":
                raise
            use_engine = False
    if not use_engine:
        # fork env workers before any HIP context exists in this process
        venv = SubprocVectorEnv(env_fn, num_envs=n_envs,
                                num_workers=n_workers, base_seed=seed)
    num_actions = cfg.get("env_config", {}).get("max_partitions_per_op", 16) + 1
    if not use_engine and loop_cfg.get("precompute_lookaheads", True):
        from ..cluster.batched_lookahead import precompute_lookahead_memos
        scratch = build_env_from_config(cfg)
        scratch.reset(seed=seed)
        memo_l, memo_i = precompute_lookahead_memos(
            scratch, device=device if use_cuda else "cpu")
        del scratch
        venv.preload_memos(memo_l, memo_i)
    torch.manual_seed(seed)
    policy = GNNPolicy(num_actions=num_actions,
                       config=model_cfg.get("custom_model_config"))
    if algo.get("name", "ppo") == "es":
        from ..rl.es import ESConfig, ESTrainer
        es_cfg = ESConfig(
            noise_stdev=algo.get("noise_stdev", 0.02),
            stepsize=algo.get("stepsize", 0.01),
            l2_coeff=algo.get("l2_coeff", 0.005),
            perturbation_pairs=algo.get("perturbation_pairs", 16),
            fragment_steps=algo.get("fragment_steps", 16))
        return ESTrainer(venv, policy, es_cfg, device=device)
    if algo.get("name", "ppo") == "dqn":
        from ..rl.dqn import DQNConfig, DQNTrainer
        dq = DQNConfig(
            gamma=algo.get("gamma", 0.999), lr=algo.get("lr", 4.121e-7),
            n_step=algo.get("n_step", 3),
            double_q=algo.get("double_q", True),
            dueling=algo.get("dueling", True),
            target_network_update_freq=algo.get(
                "target_network_update_freq", 100000),
            v_min=algo.get("v_min", -1000.0),
            v_max=algo.get("v_max", 1000.0),
            train_batch_size=algo.get("train_batch_size", 4096),
            sgd_minibatch_size=algo.get("sgd_minibatch_size", 256),
            num_sgd_iter=algo.get("num_sgd_iter", 4))
        return DQNTrainer(venv, policy, dq, device=device)
    if algo.get("name", "ppo") == "pg":
        from ..rl.pg import PGConfig, PGTrainer
        pg_cfg = PGConfig(lr=algo.get("lr", 4e-4),
                          gamma=algo.get("gamma", 0.99),
                          train_batch_size=algo.get("train_batch_size", 200),
                          grad_clip=algo.get("grad_clip"))
        return PGTrainer(venv, policy, pg_cfg, device=device)
    if algo.get("name", "ppo") == "impala":
        from ..rl.impala import ImpalaConfig, ImpalaTrainer
        imp_cfg = ImpalaConfig(
            lr=algo.get("lr", 5e-4),
            gamma=algo.get("gamma", 0.997),
            vf_loss_coeff=algo.get("vf_loss_coeff", 0.5),
            entropy_coeff=algo.get("entropy_coeff", 0.01),
            grad_clip=algo.get("grad_clip", 40.0),
            rho_clip=algo.get("vtrace_clip_rho_threshold", 1.0),
            pg_rho_clip=algo.get("vtrace_clip_pg_rho_threshold", 1.0),
            c_clip=algo.get("vtrace_clip_c_threshold", 1.0),
            train_batch_size=algo.get("train_batch_size", 4000),
            vtrace_drop_last_ts=algo.get("vtrace_drop_last_ts", True))
        return ImpalaTrainer(venv, policy, imp_cfg, device=device)
    ppo_cfg = PPOConfig(
        lr=algo.get("lr", 2.785e-4),
        gamma=algo.get("gamma", 0.997),
        lambda_=algo.get("lambda", 1.0),
        clip_param=algo.get("clip_param", 0.18),
        entropy_coeff=algo.get("entropy_coeff", 3e-3),
        kl_coeff=algo.get("kl_coeff", 0.01),
        kl_target=algo.get("kl_target", 1e-3),
        vf_clip_param=algo.get("vf_clip_param", 128.8),
        grad_clip=algo.get("grad_clip", 1.5),
        sgd_minibatch_size=algo.get("sgd_minibatch_size", 128),
        train_batch_size=algo.get("train_batch_size", 4000),
        num_sgd_iter=algo.get("num_sgd_iter", 50))
    return PPOTrainer(venv, policy, ppo_cfg, device=device)
