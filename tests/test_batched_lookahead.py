"""Batched lookahead engine tests: CPU memo-precompute parity + HIP kernel
exactness (gpu-marked)."""
import copy

import numpy as np
import pytest

from tests.conftest import make_env


def _uniform_frac():
    return {"_target_": "ddls_amd.distributions.Uniform",
            "min_val": 0.1, "max_val": 1.0, "decimals": 2}


def test_precompute_memo_matches_cold_run(tiny_model_files):
    """An env preloaded with CPU-precomputed memos must produce exactly the
    same episode as a cold env."""
    from ddls_amd.cluster.batched_lookahead import precompute_lookahead_memos

    env_cold = make_env(tiny_model_files, replication=4, frac_dist=_uniform_frac())
    env_cold.reset(seed=7)

    scratch = make_env(tiny_model_files, replication=4, frac_dist=_uniform_frac())
    scratch.reset(seed=7)
    lookahead_memo, init_memo = precompute_lookahead_memos(scratch, device="cpu")
    assert ("tiny", 2) in lookahead_memo
    assert ("tiny", 1) in lookahead_memo

    env_pre = make_env(tiny_model_files, replication=4, frac_dist=_uniform_frac())
    env_pre.lookahead_memo_preload = lookahead_memo
    env_pre.init_details_memo_preload = init_memo
    env_pre.reset(seed=7)
    # preloaded memo table must already be populated
    assert 2 in env_pre.cluster.job_model_to_max_num_partitions_to_lookahead["tiny"]

    def run_episode(env):
        obs = env.reset(seed=11)
        rewards, done = [], False
        while not done:
            valid = obs["action_set"][obs["action_mask"].astype(bool)]
            obs, r, done, _ = env.step(int(valid[-1]))
            rewards.append(r)
        return rewards

    # sequential episodes (the envs share the global RNG stream)
    rewards_c = run_episode(env_cold)
    rewards_p = run_episode(env_pre)
    assert rewards_c == pytest.approx(rewards_p, rel=1e-12)
    sc, sp = env_cold.cluster.episode_stats, env_pre.cluster.episode_stats
    assert sc["num_jobs_completed"] == sp["num_jobs_completed"]
    assert sc["job_completion_time"] == pytest.approx(sp["job_completion_time"])
    assert (sc["jobs_completed_mean_mounted_worker_utilisation_frac"]
            == pytest.approx(sp["jobs_completed_mean_mounted_worker_utilisation_frac"]))


def _prepare_jobs(tiny_model_files, actions=(2, 4)):
    """Build mounted partitioned jobs via the precompute pipeline pieces."""
    from ddls_amd.agents.partitioners import sip_ml_num_partitions
    from ddls_amd.cluster.actions import OpPartition
    from ddls_amd.cluster.batched_lookahead import mount_job_for_lookahead
    from ddls_amd.graphs import FWD

    env = make_env(tiny_model_files, replication=1)
    env.reset(seed=0)
    cluster = env.cluster
    job = next(iter(cluster.job_queue.jobs.values()))
    g = job.graph
    cc = g.compute_cost[cluster.device_type]
    jobs = []
    for action in actions:
        pa = {}
        for i in range(g.n):
            if g.pass_type[i] != FWD:
                continue
            num = sip_ml_num_partitions(float(cc[i]), 0.01, action)
            pa[g.names[i]] = num
            pa[g.names[int(g.counterpart[i])]] = num
        op = OpPartition({job.job_id: pa}, cluster=cluster)
        placement = env.op_placer.get(op_partition=op, cluster=cluster)
        sched = env.op_scheduler.get(op_partition=op, op_placement=placement,
                                     cluster=cluster)
        dp = env.dep_placer.get(op_partition=op, op_placement=placement,
                                cluster=cluster)
        ds = env.dep_scheduler.get(op_partition=op, dep_placement=dp,
                                   cluster=cluster)
        pjob = op.partitioned_jobs[job.job_id]
        assert mount_job_for_lookahead(cluster, pjob,
                                       placement.action[job.job_id], sched, dp, ds)
        jobs.append(pjob)
    return jobs


def test_batched_cpu_path_matches_tick_loop(tiny_model_files):
    from ddls_amd.cluster.batched_lookahead import run_lookahead_batch
    from ddls_amd.cluster.lookahead import active_time_sum, run_lookahead_ticks

    jobs = _prepare_jobs(tiny_model_files)
    jobs_copy = [copy.deepcopy(j) for j in jobs]
    batch_results = run_lookahead_batch(jobs, device="cpu")
    for job, (t, comp, comm, active) in zip(jobs_copy, batch_results):
        t2, comp2, comm2, tm = run_lookahead_ticks(job)
        assert t == pytest.approx(t2, rel=0, abs=0)
        assert comp == comp2 and comm == comm2
        assert active == pytest.approx(active_time_sum(tm))


@pytest.mark.gpu
def test_hip_lookahead_matches_cpu_exactly(tiny_model_files):
    """The HIP batched kernel must reproduce the CPU tick loop bitwise
    (same fp64 operations in the same per-env order)."""
    import torch
    from ddls_amd.cluster.batched_lookahead import run_lookahead_batch
    from ddls_amd.cluster.lookahead import active_time_sum, run_lookahead_ticks

    jobs = _prepare_jobs(tiny_model_files, actions=(1, 2, 4, 8, 16))
    jobs_cpu = [copy.deepcopy(j) for j in jobs]
    gpu_results = run_lookahead_batch(jobs, device=torch.device("cuda:0"))
    for job, (t, comp, comm, active) in zip(jobs_cpu, gpu_results):
        t2, comp2, comm2, tm = run_lookahead_ticks(job)
        assert t == t2, (t, t2)                      # bitwise
        assert comp == comp2 and comm == comm2
        assert active == active_time_sum(tm)


@pytest.mark.gpu
def test_gpu_precompute_memo(tiny_model_files):
    import torch
    from ddls_amd.cluster.batched_lookahead import precompute_lookahead_memos

    scratch = make_env(tiny_model_files, replication=2)
    scratch.reset(seed=3)
    memo_gpu, _ = precompute_lookahead_memos(scratch, device=torch.device("cuda:0"))

    scratch2 = make_env(tiny_model_files, replication=2)
    scratch2.reset(seed=3)
    memo_cpu, _ = precompute_lookahead_memos(scratch2, device="cpu")

    assert set(memo_gpu) == set(memo_cpu)
    for k in memo_cpu:
        for a, b in zip(memo_gpu[k][:3], memo_cpu[k][:3]):
            assert a == b, (k, memo_gpu[k], memo_cpu[k])
