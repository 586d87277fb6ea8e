"""Parity tests for the vectorised env engine (cluster/vec_engine.py).

Contract: CPU mirror == RampJobPartitioningEnvironment step-for-step
(obs / reward / done, f64 exact), so the HIP kernel (tested separately on
GPU against the mirror) is transitively equivalent to the reference env.
"""

import numpy as np
import pytest

from ddls_amd.cluster.vec_engine import (EngineState, ST_MISS, ST_OK,
                                         build_episode_stats,
                                         compile_engine_spec, cpu_step_env,
                                         drain_episode_schedule)
from ddls_amd.envs import RampJobPartitioningEnvironment


def make_env(jobs_dir, mode="remove", replication=3, max_sim=1e6,
             interarrival=40, num_training_steps=10, reward_kwargs=None,
             reward="lookahead_job_completion_time"):
    return RampJobPartitioningEnvironment(
        topology_config={"type": "ramp", "kwargs": {
            "num_communication_groups": 4,
            "num_racks_per_communication_group": 4,
            "num_servers_per_rack": 2,
            "num_channels": 1,
            "total_node_bandwidth": 1.6e12,
            "intra_gpu_propagation_latency": 50e-9,
            "worker_io_latency": 100e-9}},
        node_config={"type_1": {"num_nodes": 32, "workers_config": [
            {"num_workers": 1, "worker": "ddls_amd.devices.A100"}]}},
        jobs_config={"path_to_files": jobs_dir,
                     "replication_factor": replication,
                     "job_sampling_mode": mode,
                     "job_interarrival_time_dist": {
                         "_target_": "ddls_amd.distributions.Fixed",
                         "val": interarrival},
                     "max_acceptable_job_completion_time_frac_dist": {
                         "_target_": "ddls_amd.distributions.Uniform",
                         "min_val": 0.1, "max_val": 1, "decimals": 2},
                     "num_training_steps": num_training_steps},
        max_partitions_per_op=16,
        min_op_run_time_quantum=0.01,
        pad_obs_kwargs=None,
        reward_function=reward,
        reward_function_kwargs=reward_kwargs,
        max_simulation_run_time=max_sim)


@pytest.fixture
def multi_model_files(tmp_path):
    """Three small models with different sizes/costs."""
    from ddls_amd.workloads import generate_model, write_pipedream_txt
    d = tmp_path / "jobs"
    d.mkdir()
    for name, (nn, sk, sc, seed) in {
            "m_small": (5, 0, 0.5, 11),
            "m_mid": (8, 2, 1.0, 12),
            "m_big": (11, 3, 1.6, 13)}.items():
        nodes, edges = generate_model(name, nn, sk, sc, seed)
        write_pipedream_txt(str(d / f"{name}.txt"), nodes, edges)
    return str(d)


def scripted_action(mask, t):
    """Deterministic action choice both sides can reproduce: cycle valid
    actions, visiting 0 (don't-place) every 7th step."""
    valid = [a for a in range(len(mask)) if mask[a]]
    if t % 7 == 3:
        return 0
    return valid[t % len(valid)]


def _memo_preload_for_env(spec):
    """spec.memo (keyed (model_id, degree)) -> the env's preload format."""
    out = {}
    for (mid, deg), (jct, comm, comp, act_sum) in spec.memo.items():
        out[(spec.models[mid].name, deg)] = (jct, comm, comp,
                                             {"active_time_sum": act_sum})
    return out


def run_parity(jobs_dir, seed, steps, mode="remove", replication=3,
               max_sim=1e6, interarrival=40, reward_kwargs=None,
               reward="lookahead_job_completion_time",
               action_fn=scripted_action):
    # spec compiled from a scratch env (empty-cluster pipeline runs)
    env_c = make_env(jobs_dir, mode, replication, max_sim, interarrival,
                     reward_kwargs=reward_kwargs, reward=reward)
    env_c.reset(seed=12345)
    spec = compile_engine_spec(env_c)
    memo_preload = _memo_preload_for_env(spec)

    # mirror schedule FIRST: draining re-seeds the global RNG, and the real
    # env below must run its episode on a pristine stream from `seed`
    gen = env_c.cluster.jobs_generator
    sched = drain_episode_schedule(gen, spec, seed=seed)
    st = EngineState(spec, B=1, n_jobs_cap=sched.n + 2)
    st.reset_env(spec, 0, sched)

    # real env, canonical memo preloaded (the production configuration)
    env = make_env(jobs_dir, mode, replication, max_sim, interarrival,
                   reward_kwargs=reward_kwargs, reward=reward)
    env.lookahead_memo_preload = memo_preload
    obs = env.reset(seed=seed)

    for t in range(steps):
        # observation parity (graph features = 17 features ++ mask)
        real_gf = obs["graph_features"]
        mine_gf = np.concatenate([st.obs_gf[0], st.obs_mask[0]])
        np.testing.assert_array_equal(real_gf, mine_gf,
                                      err_msg=f"graph_features step {t}")
        np.testing.assert_array_equal(
            obs["action_mask"].astype(np.float32), st.obs_mask[0],
            err_msg=f"action_mask step {t}")
        mid = int(st.obs_model[0])
        np.testing.assert_array_equal(obs["node_features"],
                                      spec.models[mid].node_features)
        np.testing.assert_array_equal(obs["edge_features"],
                                      spec.models[mid].edge_features)

        a = action_fn(obs["action_mask"], t)
        obs, reward, done, _ = env.step(int(a))
        status = cpu_step_env(spec, st, 0, sched, int(a))
        assert status == ST_OK
        assert st.reward[0] == reward, (
            f"step {t}: reward {st.reward[0]} != {reward}")
        assert bool(st.done[0]) == bool(done), f"step {t}: done mismatch"
        if done:
            return env, spec, sched, st, t + 1
    return env, spec, sched, st, steps


def test_mirror_matches_real_env_finite_pool(multi_model_files):
    """Full episode to pool exhaustion: obs/reward/done parity each step +
    episode stats parity at the end."""
    env, spec, sched, st, n = run_parity(multi_model_files, seed=7,
                                         steps=500, mode="remove",
                                         replication=4, interarrival=40)
    assert st.done[0], "episode should have ended"
    es_real = env.cluster.episode_stats
    es_mine = build_episode_stats(spec, sched, st, 0)
    for key in ("num_jobs_arrived", "num_jobs_completed", "num_jobs_blocked",
                "blocking_rate", "acceptance_rate", "episode_time"):
        assert es_mine[key] == es_real[key], key
    for key in ("job_completion_time", "job_completion_time_speedup",
                "jobs_completed_max_partitions_per_op",
                "jobs_completed_num_mounted_workers",
                "jobs_completed_num_mounted_channels",
                "jobs_completed_max_acceptable_job_completion_time",
                "jobs_blocked_job_sequential_completion_time",
                "job_communication_overhead_time"):
        np.testing.assert_allclose(es_mine[key], es_real[key], rtol=0,
                                   atol=0, err_msg=key)
    # info-processed totals telescope exactly
    for key in ("compute_info_processed", "dep_info_processed",
                "flow_info_processed", "cluster_info_processed",
                "demand_total_info_processed"):
        np.testing.assert_allclose(es_mine[key], float(np.sum(es_real[key])),
                                   rtol=1e-12, err_msg=key)
    np.testing.assert_allclose(es_mine["mean_load_rate"],
                               es_real["mean_load_rate"], rtol=1e-12)


def test_mirror_matches_real_env_infinite_pool(multi_model_files):
    """remove_and_repeat pool bounded by max_simulation_run_time, congested
    cluster (short interarrival -> occupancy-dependent placement search)."""
    env, spec, sched, st, n = run_parity(multi_model_files, seed=3,
                                         steps=400, mode="remove_and_repeat",
                                         replication=2, max_sim=4000,
                                         interarrival=15)
    assert n < 400, "episode should end at max_sim"
    es_real = env.cluster.episode_stats
    es_mine = build_episode_stats(spec, sched, st, 0)
    for key in ("num_jobs_arrived", "num_jobs_completed", "num_jobs_blocked"):
        assert es_mine[key] == es_real[key], key
    np.testing.assert_allclose(es_mine["job_completion_time"],
                               es_real["job_completion_time"], rtol=0, atol=0)


def test_mirror_multiple_seeds(multi_model_files):
    for seed in (1, 42, 1799):
        run_parity(multi_model_files, seed=seed, steps=60,
                   mode="remove_and_repeat", replication=2, max_sim=3000,
                   interarrival=25)


def test_mirror_log_transform_reward(multi_model_files):
    run_parity(multi_model_files, seed=5, steps=50, mode="remove_and_repeat",
               replication=2, max_sim=2500, interarrival=30,
               reward_kwargs={"transform_with_log": True,
                              "fail_reward_factor": 2.0})


def test_memo_miss_protocol(multi_model_files):
    """With a partially-warm memo, the step reports ST_MISS without mutating
    state; inserting the entry and re-stepping matches the warm path."""
    env_c = make_env(multi_model_files)
    env_c.reset(seed=12345)
    spec = compile_engine_spec(env_c)
    gen = env_c.cluster.jobs_generator
    sched = drain_episode_schedule(gen, spec, seed=9)

    cold = {}          # empty memo: every placement probe misses
    st = EngineState(spec, B=1, n_jobs_cap=sched.n + 2)
    st.reset_env(spec, 0, sched)
    st_warm = EngineState(spec, B=1, n_jobs_cap=sched.n + 2)
    st_warm.reset_env(spec, 0, sched)

    for t in range(30):
        mask = st.obs_mask[0]
        a = scripted_action(mask, t)
        t_before = float(st.t[0])
        status = cpu_step_env(spec, st, 0, sched, int(a), memo=cold)
        if status == ST_MISS:
            assert st.t[0] == t_before, "miss must not mutate state"
            # service the miss from the compiled table, then re-step
            mid = int(sched.model_id[st.queued[0]])
            deg = int(spec.models[mid].action_to_degree[a])
            cold[(mid, deg)] = spec.memo[(mid, deg)]
            status = cpu_step_env(spec, st, 0, sched, int(a), memo=cold)
        assert status == ST_OK
        status_w = cpu_step_env(spec, st_warm, 0, sched, int(a))
        assert status_w == ST_OK
        assert st.reward[0] == st_warm.reward[0]
        assert st.t[0] == st_warm.t[0]
        np.testing.assert_array_equal(st.obs_gf[0], st_warm.obs_gf[0])
        if st.done[0]:
            break


# ---------------------------------------------------------------------------
# GPU: HIP kernel vs CPU mirror (bitwise)
# ---------------------------------------------------------------------------

def _lockstep_engines(multi_model_files, B, steps, seeds, preload=True,
                      mode="remove_and_repeat", max_sim=4000, interarrival=15):
    import torch
    from ddls_amd.cluster.gpu_engine import GpuEngine

    env_c = make_env(multi_model_files, mode, 2, max_sim, interarrival)
    env_c.reset(seed=12345)
    spec = compile_engine_spec(env_c)
    gen = env_c.cluster.jobs_generator

    scheds = [drain_episode_schedule(gen, spec, seed=s) for s in seeds]
    cap = max(s.n for s in scheds) + 8
    st = EngineState(spec, B=B, n_jobs_cap=cap)
    eng = GpuEngine(spec, B=B, device=torch.device("cuda:0"),
                    preload_memo=preload)
    episode_counter = [0] * B
    for b in range(B):
        st.reset_env(spec, b, scheds[b])
        eng.reset_env(b, scheds[b])

    n_episodes_done = 0
    for t in range(steps):
        actions = np.zeros(B, dtype=np.int64)
        for b in range(B):
            actions[b] = scripted_action(st.obs_mask[b], t + b)
        # mirror
        for b in range(B):
            status = cpu_step_env(spec, st, b, scheds[b], int(actions[b]))
            assert status == ST_OK
        # kernel (services memo misses internally)
        eng.step(torch.as_tensor(actions, device="cuda:0"))
        status = eng.T["status"].cpu().numpy()
        assert (status == ST_OK).all(), status

        # bitwise compare: obs, reward, done, core state
        np.testing.assert_array_equal(eng.T["obs_gf"].cpu().numpy(),
                                      st.obs_gf, err_msg=f"obs_gf step {t}")
        np.testing.assert_array_equal(eng.T["obs_mask"].cpu().numpy(),
                                      st.obs_mask)
        np.testing.assert_array_equal(eng.T["reward"].cpu().numpy(),
                                      st.reward, err_msg=f"reward step {t}")
        np.testing.assert_array_equal(
            eng.T["done"].cpu().numpy().astype(bool), st.done)
        np.testing.assert_array_equal(eng.T["t"].cpu().numpy(), st.t)
        np.testing.assert_array_equal(eng.T["occ"].cpu().numpy(),
                                      st.occ.view(np.int64))
        np.testing.assert_array_equal(eng.T["n_running"].cpu().numpy(),
                                      st.n_running)
        np.testing.assert_array_equal(eng.T["arr_ptr"].cpu().numpy(),
                                      st.arr_ptr)
        # episode service
        for b in range(B):
            if st.done[b]:
                n_episodes_done += 1
                es_cpu = build_episode_stats(spec, scheds[b], st, b)
                es_gpu = eng.episode_stats(b)
                for key in ("num_jobs_arrived", "num_jobs_completed",
                            "num_jobs_blocked", "episode_return"):
                    assert es_cpu[key] == es_gpu[key], key
                np.testing.assert_array_equal(es_cpu["job_completion_time"],
                                              es_gpu["job_completion_time"])
                episode_counter[b] += 1
                sched = drain_episode_schedule(
                    gen, spec, seed=seeds[b] + 1000 * episode_counter[b])
                scheds[b] = sched
                st.reset_env(spec, b, sched)
                eng.reset_env(b, sched)
    return n_episodes_done


@pytest.mark.gpu
def test_gpu_kernel_matches_mirror(multi_model_files):
    """Whole-episode bitwise f64/f32 parity: HIP kernel == CPU mirror,
    including auto-reset across episode boundaries."""
    n_done = _lockstep_engines(multi_model_files, B=4, steps=260,
                               seeds=[1, 7, 42, 1799], max_sim=2500)
    assert n_done >= 4, "expected an episode completion per env"


@pytest.mark.gpu
def test_gpu_kernel_memo_miss_path(multi_model_files):
    """Cold device hash table: misses are serviced (insert + relaunch) and
    results stay bitwise-identical to the mirror (warm dict)."""
    _lockstep_engines(multi_model_files, B=2, steps=60, seeds=[3, 5],
                      preload=False)


def test_mirror_matches_real_env_1024_workers(multi_model_files):
    """configs[3]: 1024-worker RAMP — mirror vs real env over a short
    congested prefix (the block search runs over a (16,16,4) mesh)."""
    from ddls_amd.envs import RampJobPartitioningEnvironment

    def mk():
        return RampJobPartitioningEnvironment(
            topology_config={"type": "ramp", "kwargs": {
                "num_communication_groups": 16,
                "num_racks_per_communication_group": 16,
                "num_servers_per_rack": 4,
                "num_channels": 1,
                "total_node_bandwidth": 1.6e12,
                "intra_gpu_propagation_latency": 50e-9,
                "worker_io_latency": 100e-9}},
            node_config={"type_1": {"num_nodes": 1024, "workers_config": [
                {"num_workers": 1, "worker": "ddls_amd.devices.A100"}]}},
            jobs_config={"path_to_files": multi_model_files,
                         "replication_factor": 2,
                         "job_sampling_mode": "remove_and_repeat",
                         "job_interarrival_time_dist": {
                             "_target_": "ddls_amd.distributions.Exponential",
                             "mean": 20},
                         "max_acceptable_job_completion_time_frac_dist": {
                             "_target_": "ddls_amd.distributions.Uniform",
                             "min_val": 0.1, "max_val": 1, "decimals": 2},
                         "num_training_steps": 10},
            max_partitions_per_op=16,
            min_op_run_time_quantum=0.01,
            pad_obs_kwargs=None,
            max_simulation_run_time=2000)

    env_c = mk()
    env_c.reset(seed=12345)
    spec = compile_engine_spec(env_c)
    assert spec.W == 1024
    memo_preload = _memo_preload_for_env(spec)
    gen = env_c.cluster.jobs_generator
    sched = drain_episode_schedule(gen, spec, seed=5)
    st = EngineState(spec, B=1, n_jobs_cap=sched.n + 2)
    st.reset_env(spec, 0, sched)
    env = mk()
    env.lookahead_memo_preload = memo_preload
    obs = env.reset(seed=5)
    for t in range(15):
        np.testing.assert_array_equal(
            obs["action_mask"].astype(np.float32), st.obs_mask[0])
        a = scripted_action(obs["action_mask"], t)
        obs, reward, done, _ = env.step(int(a))
        assert cpu_step_env(spec, st, 0, sched, int(a)) == ST_OK
        assert st.reward[0] == reward
        assert bool(st.done[0]) == bool(done)
        if done:
            break


def test_mirror_parity_random_action_streams(multi_model_files):
    """Property-style sweep: the mirror must match the real env bitwise
    under RANDOM valid-action streams (not just the scripted cycle), across
    episode seeds and arrival intensities — broadens coverage of placement/
    queue interleavings the scripted pattern cannot reach."""
    cases = [(101, 5001, 15), (202, 5002, 40), (303, 5003, 90),
             (404, 5004, 25), (505, 5005, 60)]
    for ep_seed, act_seed, interarrival in cases:
        rng = np.random.RandomState(act_seed)

        def rand_action(mask, t, rng=rng):
            valid = [a for a in range(len(mask)) if mask[a]]
            if rng.rand() < 0.15:
                return 0
            return int(valid[rng.randint(len(valid))])

        env, spec, sched, st_, n = run_parity(
            multi_model_files, seed=ep_seed, steps=30,
            interarrival=interarrival, action_fn=rand_action)
        assert n > 0


def test_mirror_job_acceptance_reward(multi_model_files):
    """JobAcceptance (+success/-fail on accept/block, reference
    ``job_acceptance.py:9-33``): mirror reward bitwise-equal to the real
    env over a full episode, including blocked steps."""
    env, spec, sched, st_, n = run_parity(
        multi_model_files, seed=11, steps=40, interarrival=15,
        reward="job_acceptance",
        reward_kwargs={"fail_reward": -2.0, "success_reward": 0.5})
    assert n > 0


def test_mirror_multi_objective_reward(multi_model_files):
    """MultiObjectiveJCTBlocking (weighted -JCT + acceptance, reference
    ``multi_objective_jct_blocking.py:9-90``): mirror reward bitwise-equal
    to the real env."""
    env, spec, sched, st_, n = run_parity(
        multi_model_files, seed=13, steps=40, interarrival=20,
        reward="multi_objective_jct_blocking",
        reward_kwargs={"jct_weight": 0.7, "blocking_weight": 0.3,
                       "transform_with_log": True,
                       "fail_reward_factor": 2.0})
    assert n > 0


@pytest.mark.parametrize("reward", ["mean_compute_throughput",
                                    "mean_cluster_throughput",
                                    "mean_demand_total_throughput"])
def test_mirror_throughput_rewards(multi_model_files, reward):
    """Throughput reward family (reference mean_compute_throughput.py:9-57,
    mean_cluster_throughput.py, mean_demand_total_throughput.py): the
    mirror accrues per-tick info_processed with the env's exact f64
    summation order, so the first-cluster-step info/step_time reward is
    bitwise-equal over whole episodes."""
    env, spec, sched, st_, n = run_parity(
        multi_model_files, seed=17, steps=60, interarrival=25,
        replication=4, reward=reward)
    assert n > 0


def test_cpu_engine_capacity_guard(multi_model_files):
    """CpuEngine refuses a schedule longer than its job-log capacity with a
    clear error instead of corrupting state."""
    from ddls_amd.cluster.vec_engine import CpuEngine

    env_c = make_env(multi_model_files)
    env_c.reset(seed=1)
    spec = compile_engine_spec(env_c)
    gen = env_c.cluster.jobs_generator
    sched = drain_episode_schedule(gen, spec, seed=2)
    eng = CpuEngine(spec, B=1, n_jobs_cap=4)
    if sched.n + 1 > 4:
        with pytest.raises(ValueError, match="capacity"):
            eng.reset_env(0, sched)
    big = CpuEngine(spec, B=1, n_jobs_cap=sched.n + 8)
    big.reset_env(0, sched)
    assert int(big.T["obs_model"][0]) >= 0
