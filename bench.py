#!/usr/bin/env python3
"""Flagship benchmark: PPO training of the PAC-ML GNN partitioner on the
32-worker RAMP cluster simulator (BASELINE.json metric: PPO env-steps/sec,
whole node, + mean simulated JCT).

One process per GPU (torchrun for N>1), fused gradient all-reduce with RCCL
over xGMI.  A "step" is one PPO iteration at the reference's tuned operating
point (algo/ppo.yaml:26-54 + BASELINE configs[2]): 256 vectorised envs/rank,
16 rollout steps -> train batch 4096/rank, 50 SGD epochs x minibatch 128.

On GPU the envs are GPU-resident: the whole vectorised env step runs as one
batched HIP kernel (cluster/gpu_engine.py) and the rollout
policy uses a per-model GNN embedding cache (rl/engine_env.py).  --cpu-envs
falls back to the round-1 subprocess-worker path.

Synthetic workload: deterministic pipedream-format job graphs (the
reference's "small_graphs" profile set is not distributed); random-init GNN
weights.
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch


def build_env_fn(seed_offset: int = 0, device_type: str = "A100",
                 lookahead_memo=None, init_details_memo=None):
    from ddls_amd.envs import RampJobPartitioningEnvironment
    from ddls_amd.workloads import ensure_default_set
    data = ensure_default_set()
    worker_cls = ("ddls_amd.devices.A100" if device_type == "A100"
                  else "ddls_amd.devices.MI355X")

    def make():
        return RampJobPartitioningEnvironment(
            lookahead_memo_preload=lookahead_memo,
            init_details_memo_preload=init_details_memo,
            reuse_jobs_generator=True,
            topology_config={"type": "ramp", "kwargs": {
                "num_communication_groups": 4,
                "num_racks_per_communication_group": 4,
                "num_servers_per_rack": 2,
                "num_channels": 1,
                "total_node_bandwidth": 1.6e12,
                "intra_gpu_propagation_latency": 50e-9,
                "worker_io_latency": 100e-9}},
            node_config={"type_1": {"num_nodes": 32, "workers_config": [
                {"num_workers": 1, "worker": worker_cls}]}},
            jobs_config={"path_to_files": data,
                         "replication_factor": 1000,
                         "job_sampling_mode": "remove_and_repeat",
                         "job_interarrival_time_dist": {
                             "_target_": "ddls_amd.distributions.Fixed",
                             "val": 1000},
                         "max_acceptable_job_completion_time_frac_dist": {
                             "_target_": "ddls_amd.distributions.Uniform",
                             "min_val": 0.1, "max_val": 1, "decimals": 2},
                         "num_training_steps": 50},
            max_partitions_per_op=16,
            min_op_run_time_quantum=0.01,
            pad_obs_kwargs=None,
            max_simulation_run_time=1e6)
    return make


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=6)
    ap.add_argument("--warmup", type=int, default=3)
    # reference-tuned operating point (algo/ppo.yaml + BASELINE configs[2]):
    # 256 envs x 16 rollout steps = 4096 train batch, 50 SGD epochs, mb 128
    ap.add_argument("--envs-per-rank", type=int, default=256)
    ap.add_argument("--rollout-steps-per-env", type=int, default=16)
    ap.add_argument("--num-sgd-iter", type=int, default=50)
    ap.add_argument("--sgd-minibatch-size", type=int, default=128)
    ap.add_argument("--cpu-envs", action="store_true",
                    help="round-1 CPU subprocess env workers instead of the "
                         "GPU-resident engine")
    ap.add_argument("--env-workers", type=int, default=0,
                    help="(cpu-envs) worker processes per rank (0 = auto)")
    ap.add_argument("--no-precompute", action="store_true",
                    help="(cpu-envs) skip the batched GPU lookahead memo "
                         "precompute")
    args = ap.parse_args()

    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.parallel import (get_world_size,
                                   init_distributed_from_env, is_distributed)
    from ddls_amd.rl.ppo import PPOConfig, PPOTrainer

    rank = init_distributed_from_env()
    world_size = get_world_size()
    use_cuda = torch.cuda.is_available()
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if use_cuda:
        # single-GPU rehearsal of the dp path (gloo backend) maps every rank
        # onto the available device(s)
        local_rank = local_rank % max(torch.cuda.device_count(), 1)
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    torch.manual_seed(0)  # identical init weights on every rank
    policy = GNNPolicy(num_actions=17)

    env_fn = build_env_fn()
    # engine rollouts everywhere (GPU kernel on cuda, CpuEngine mirror on
    # cpu); --cpu-envs selects the round-1 subprocess-worker path
    use_engine = not args.cpu_envs
    jct_fn = None
    if use_engine:
        from ddls_amd.rl.engine_env import EngineVectorEnv
        policy = policy.to(device)
        venv = EngineVectorEnv(env_fn, num_envs=args.envs_per_rank,
                               device=device,
                               base_seed=1 + 100000 * rank, verbose=rank == 0)
        jct_fn = venv.jct_running_stats
    else:
        from ddls_amd.rl.subproc_env import SubprocVectorEnv
        n_workers = args.env_workers
        if n_workers <= 0:
            n_workers = max(1, min(args.envs_per_rank,
                                   (os.cpu_count() or 2)
                                   // max(world_size, 1) - 1))
        # fork env workers BEFORE any HIP context exists in this process
        venv = SubprocVectorEnv(env_fn, num_envs=args.envs_per_rank,
                                num_workers=n_workers,
                                base_seed=1 + 100000 * rank)
        if not args.no_precompute:
            from ddls_amd.cluster.batched_lookahead import \
                precompute_lookahead_memos
            scratch = env_fn()
            scratch.reset(seed=0)
            memo_l, memo_i = precompute_lookahead_memos(
                scratch, device=device if use_cuda else "cpu")
            del scratch
            venv.preload_memos(memo_l, memo_i)
        if hasattr(venv, "warm"):
            venv.warm(96)

    per_step_env_steps = args.envs_per_rank * args.rollout_steps_per_env
    cfg = PPOConfig(train_batch_size=per_step_env_steps,
                    sgd_minibatch_size=args.sgd_minibatch_size,
                    num_sgd_iter=args.num_sgd_iter)
    trainer = PPOTrainer(venv, policy, cfg, device=device)

    def barrier_sync():
        if is_distributed():
            torch.distributed.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        trainer.train(num_steps=args.rollout_steps_per_env)

    jct_samples = []
    jct0 = jct_fn() if jct_fn else None
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        st = trainer.train(num_steps=args.rollout_steps_per_env)
        if "mean_job_completion_time" in st:
            jct_samples.append(st["mean_job_completion_time"])
        if rank == 0:
            print(f"[bench] iter {st['iteration']}: rollout "
                  f"{st['rollout_time_s']:.3f}s update {st['update_time_s']:.3f}s "
                  f"hipgraph_mb={st.get('hipgraph_minibatches', 0)} "
                  f"captures={st.get('hipgraph_captures', 0)}",
                  file=sys.stderr, flush=True)
    barrier_sync()
    elapsed = time.perf_counter() - t0
    if rank == 0 and getattr(trainer, "_stepper", None) is not None:
        sp = trainer._stepper
        print(f"[bench] hipgraph: broken={sp.broken} captures={sp.capture_count} "
              f"flat_adam={getattr(sp, 'flat_p', None) is not None} "
              f"fused_loss={getattr(sp, '_fused_loss', False)} "
              f"repairs={getattr(sp, 'repair_count', None)} "
              f"err={sp.last_error}", file=sys.stderr, flush=True)

    # mean simulated JCT over jobs completed INSIDE the timed window
    # (BASELINE metric second half); fall back to full-episode means
    mean_jct = None
    if jct_fn is not None:
        c1, s1 = jct_fn()
        c0, s0 = jct0
        if c1 > c0:
            mean_jct = (s1 - s0) / (c1 - c0)
    if mean_jct is None and jct_samples:
        mean_jct = float(np.mean(jct_samples))

    # max over ranks
    if is_distributed():
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_cuda else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    venv_close = getattr(venv, "close", None)
    total_env_steps = args.steps * per_step_env_steps * world_size
    value = total_env_steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(json.dumps({
            "metric": "ppo_env_steps_per_sec",
            "value": value,
            "unit": "env_steps/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "pacml_gnn_msg32_hidden64_2rounds",
                "global_batch": per_step_env_steps * world_size,
                "seq_len": 150,
                "parallelism": f"dp{world_size}",
                "ramp": "4x4x2_32workers",
                "envs_per_rank": args.envs_per_rank,
                "env_engine": (("gpu_resident" if use_cuda else "cpu_engine")
                               if use_engine else "cpu_subproc"),
                "num_sgd_iter": args.num_sgd_iter,
                "sgd_minibatch_size": args.sgd_minibatch_size,
                "mean_simulated_jct": mean_jct,
            },
        }))
    if venv_close is not None:
        venv_close()


if __name__ == "__main__":
    main()
