"""Golden-trace tests of the cluster simulator + partitioning env.

These replace the reference's absent test suite with
hand-computed tick sequences.
"""
import numpy as np
import pytest

from tests.conftest import make_env


def test_sequential_jct_golden(tiny_model_files):
    """action=1 (no partitioning): all ops on one worker, every dep same-node
    -> JCT = (sum of compute costs) * num_training_steps."""
    env = make_env(tiny_model_files, replication=1, num_training_steps=10)
    obs = env.reset(seed=0)
    assert obs["action_mask"][1]
    obs, reward, done, info = env.step(1)
    seq_per_step = 0.02 + 0.03 + 0.04 + 0.06
    expected_jct = seq_per_step * 10
    # one job total -> it ran to completion
    assert done
    stats = env.cluster.episode_stats
    assert stats["num_jobs_completed"] == 1
    assert stats["job_completion_time"][0] == pytest.approx(expected_jct)
    assert stats["job_completion_time_speedup"][0] == pytest.approx(1.0)
    # reward = -lookahead JCT
    assert reward == pytest.approx(-expected_jct)


def test_partitioned_faster_than_sequential(tiny_model_files):
    env = make_env(tiny_model_files, replication=1, num_training_steps=10)
    env.reset(seed=0)
    obs, reward, done, info = env.step(2)
    stats = env.cluster.episode_stats
    assert stats["num_jobs_completed"] == 1
    jct = stats["job_completion_time"][0]
    seq = (0.02 + 0.03 + 0.04 + 0.06) * 10
    # compute halves; comm overhead added; must be faster than sequential
    # but slower than the pure-compute lower bound
    assert jct < seq
    assert jct > seq / 2
    assert stats["jobs_completed_num_mounted_workers"][0] == 2


def test_partitioned_jct_golden(tiny_model_files):
    """Hand-computed lookahead for degree-2 partitioning of the tiny chain.

    Both ops split over the same 2 workers (parent-collective co-location), so
    every fwd/bwd cross edge is priced as a symmetric collective; sync pairs
    likewise.  Lookahead ticks: fwd split compute 0.01+0.015, bwd 0.03+0.02,
    plus the collective times of groups that span 2 servers.
    """
    from ddls_amd.cluster.comm_model import calc_ramp_all_reduce_time
    env = make_env(tiny_model_files, replication=1, num_training_steps=1)
    env.reset(seed=0)
    obs, reward, done, info = env.step(2)
    stats = env.cluster.episode_stats
    assert stats["num_jobs_completed"] == 1
    jct = stats["job_completion_time"][0]

    topo = env.cluster.topology
    placement = None  # recover placement geometry from the completed job
    job = list(env.cluster.jobs_completed.values())[0]
    workers = sorted(job.details["mounted_workers"])
    assert len(workers) == 2
    coords = [topo.coords[topo.worker_to_node[w]] for w in workers]
    cgs = len({c for c, r, s in coords})
    racks = len({r for c, r, s in coords})
    nodes = len({s for c, r, s in coords})

    def coll(msg):
        return calc_ramp_all_reduce_time(
            message_size=msg, node_ids=nodes, racks=racks, cgs=cgs,
            x=topo.num_communication_groups, DATA_RATE=topo.channel_bandwidth,
            latency=topo.intra_gpu_propagation_latency,
            IO_latency=topo.worker_io_latency)

    mem = {"1": 1.5e8, "2": 3e8, "3": 3e8, "4": 1.5e8}
    # groups (each spans both workers):
    # fwd out-edges of op1 splits: (1a->2a),(1a->2b),(1b->2a),(1b->2b):
    #   created during op2's split as in-edges -> size mem(1a)/2 = mem1/4
    t_fwd1 = coll(4 * mem["1"] / 4)
    # fwd out-edges of op2 splits = join edges (2i->3j), in-edges of 3 splits:
    #   size mem(2i)/2 = mem2/4
    t_fwd2 = coll(4 * mem["2"] / 4)
    # bwd in-edges of op2's bwd (3) splits = join edges again -> priced with
    # t_fwd2's group; bwd in-edges of op1's bwd (4) splits: (3i->4j) out-edges
    # of 3 -> size mem(4j)/2 = mem4/4
    t_bwd1 = coll(4 * mem["4"] / 4)
    # sync pairs: (3a<->3b) msg = 2 * mem3/2; (4a<->4b) msg = 2 * mem4/2
    t_sync3 = coll(2 * mem["3"] / 2)
    t_sync4 = coll(2 * mem["4"] / 2)

    compute = 0.01 + 0.015 + 0.03 + 0.02
    # The op-2 sync collective (3a<->3b) becomes ready together with the
    # backward cross flows (3i->4j); both tick in parallel, and the sync
    # remainder then hides under op 4a/4b's 0.02 compute (flows tick in
    # parallel with ops).  Serial critical path:
    #   0.01 -> t_fwd1 -> 0.015 -> t_fwd2 -> 0.03 -> t_bwd1 ->
    #   max(0.02, t_sync3 - t_bwd1) -> t_sync4
    assert t_sync3 - t_bwd1 < 0.02  # precondition of this trace
    expected = compute + t_fwd1 + t_fwd2 + t_bwd1 + t_sync4
    assert jct == pytest.approx(expected, rel=1e-9)
    # and the fully-serial sum over-counts exactly t_sync3
    assert jct == pytest.approx(
        compute + t_fwd1 + t_fwd2 + t_bwd1 + t_sync3 + t_sync4 - t_sync3,
        rel=1e-9)


def test_blocking_on_max_acceptable_jct(tiny_model_files):
    """frac=0.5 and action=1 -> sequential JCT > 0.5*seq -> job blocked."""
    env = make_env(tiny_model_files, replication=1,
                   frac_dist={"_target_": "ddls_amd.distributions.Fixed",
                              "val": 0.5})
    env.reset(seed=0)
    obs, reward, done, info = env.step(1)
    stats = env.cluster.episode_stats
    assert stats["num_jobs_blocked"] == 1
    assert stats["num_jobs_completed"] == 0
    # fail reward = -seq JCT
    seq = (0.02 + 0.03 + 0.04 + 0.06) * 10
    assert reward == pytest.approx(-seq)


def test_action_zero_blocks(tiny_model_files):
    env = make_env(tiny_model_files, replication=1)
    env.reset(seed=0)
    obs, reward, done, info = env.step(0)
    assert env.cluster.episode_stats["num_jobs_blocked"] == 1


def test_lookahead_memo_reused(tiny_model_files):
    env = make_env(tiny_model_files, replication=3)
    env.reset(seed=0)
    env.step(2)
    memo = env.cluster.job_model_to_max_num_partitions_to_lookahead
    assert "tiny" in memo and 2 in memo["tiny"]
    jct0 = memo["tiny"][2][0]
    env.step(2)  # second job of same model+degree -> memo hit
    assert memo["tiny"][2][0] == jct0
    stats = env.cluster.episode_stats
    assert stats["job_completion_time"][0] == pytest.approx(
        stats["job_completion_time"][1])


def test_seeded_determinism(tiny_model_files):
    rewards = []
    for _ in range(2):
        env = make_env(tiny_model_files, replication=3, frac_dist={
            "_target_": "ddls_amd.distributions.Uniform",
            "min_val": 0.1, "max_val": 1.0, "decimals": 2})
        obs = env.reset(seed=123)
        rs, done = [], False
        while not done:
            valid = obs["action_set"][obs["action_mask"].astype(bool)]
            a = int(valid[-1])
            obs, r, done, _ = env.step(a)
            rs.append(r)
        rewards.append(rs)
    assert rewards[0] == rewards[1]


def test_action_mask_shape_and_validity(tiny_model_files):
    env = make_env(tiny_model_files)
    obs = env.reset(seed=0)
    assert len(obs["action_set"]) == 17
    assert obs["action_mask"][0] == 1  # action 0 always valid
    mask = obs["action_mask"].astype(bool)
    # odd actions > 1 invalid
    for a in (3, 5, 7, 9, 11, 13, 15):
        assert not mask[a]
    # 1 and even actions <= 16 valid on the empty 32-worker 4x4x2 ramp
    for a in (1, 2, 4, 8, 16):
        assert mask[a]


def test_placement_shaping_env(tiny_model_files):
    """The shaping env variant: agent picks a (c,r,s) meta-block shape."""
    from ddls_amd.envs.ramp_job_placement_shaping import (
        RampJobPlacementShapingEnvironment)
    import numpy as np
    env = RampJobPlacementShapingEnvironment(
        topology_config={"type": "ramp", "kwargs": {
            "num_communication_groups": 4,
            "num_racks_per_communication_group": 4,
            "num_servers_per_rack": 2, "num_channels": 1,
            "total_node_bandwidth": 1.6e12,
            "intra_gpu_propagation_latency": 50e-9,
            "worker_io_latency": 100e-9}},
        node_config={"type_1": {"num_nodes": 32, "workers_config": [
            {"num_workers": 1, "worker": "ddls_amd.devices.A100"}]}},
        jobs_config={"path_to_files": tiny_model_files,
                     "replication_factor": 3,
                     "job_sampling_mode": "remove",
                     "job_interarrival_time_dist": {
                         "_target_": "ddls_amd.distributions.Fixed",
                         "val": 1000},
                     "max_acceptable_job_completion_time_frac_dist": {
                         "_target_": "ddls_amd.distributions.Fixed", "val": 1.0},
                     "num_training_steps": 5,
                     "max_partitions_per_op_in_observation": 2},
        op_partitioner_kwargs={"min_op_run_time_quantum": 0.01},
        max_simulation_run_time=1e6)
    obs = env.reset(seed=0)
    assert len(obs["action_set"]) == 4 * 4 * 2 + 1
    assert obs["action_mask"][0]
    done, steps, completed = False, 0, 0
    while not done and steps < 10:
        valid = np.flatnonzero(obs["action_mask"])
        a = int(valid[-1])
        obs, r, done, _ = env.step(a)
        steps += 1
    stats = env.cluster.episode_stats
    assert stats["num_jobs_completed"] + stats["num_jobs_blocked"] == 3


def test_shaper_agents(tiny_model_files):
    from ddls_amd.agents.shapers import (RampFirstFitJobPlacementShaper,
                                         RampRandomJobPlacementShaper)
    from ddls_amd.agents.partitioners import SipMlOpPartitioner
    env = make_env(tiny_model_files, replication=1)
    env.reset(seed=0)
    op = SipMlOpPartitioner(min_op_run_time_quantum=0.01).get(
        env.cluster, max_partitions_per_op=2)
    for shaper in (RampFirstFitJobPlacementShaper(), RampRandomJobPlacementShaper()):
        shape = shaper.get(op, env.cluster)
        assert len(shape.action) == 1
        c, r, s = next(iter(shape.action.values()))
        assert c * r * s >= 2


def test_legacy_cluster_env(tiny_model_files):
    """Legacy dynamic per-tick sim: jobs run tick-by-tick, no comm model."""
    from ddls_amd.agents.job_managers import RandomJobPlacer, SRPTJobScheduler
    from ddls_amd.cluster.legacy_environment import ClusterEnvironment
    from ddls_amd.utils import seed_everything
    seed_everything(0)
    env = ClusterEnvironment(
        topology_config={"type": "torus", "kwargs": {
            "x_dims": 4, "y_dims": 1, "z_dims": 1, "num_channels": 1}},
        node_config={"type_1": {"num_nodes": 4, "workers_config": [
            {"num_workers": 1, "worker": "ddls_amd.devices.A100"}]}})
    env.reset(jobs_config={
        "path_to_files": tiny_model_files,
        "replication_factor": 2,
        "job_sampling_mode": "remove",
        "job_interarrival_time_dist": {
            "_target_": "ddls_amd.distributions.Fixed", "val": 0.1},
        "num_training_steps": 3}, seed=0)
    placer, scheduler = RandomJobPlacer(), SRPTJobScheduler()
    done, guard = False, 0
    while not done and guard < 50:
        placement = placer.get(env)
        schedule = scheduler.get(placement, env)
        _, _, _, done, _ = env.step({"job_placement": placement,
                                     "job_schedule": schedule})
        guard += 1
    assert env.episode_stats["num_jobs_completed"] == 2
    # per-training-step sequential lower bound: each job needs >= 3 steps of
    # its critical path; with no comm overhead and enough workers, JCT >=
    # num_training_steps * critical path
    jcts = env.episode_stats["job_completion_time"]
    assert all(j >= 3 * (0.02 + 0.03 + 0.06 + 0.04) - 1e-9 for j in jcts)


def test_job_placing_env(tiny_model_files):
    from ddls_amd.envs.job_placing import JobPlacingAllNodesEnvironment
    from ddls_amd.utils import seed_everything
    seed_everything(0)
    env = JobPlacingAllNodesEnvironment(
        topology_config={"type": "torus", "kwargs": {
            "x_dims": 4, "y_dims": 1, "z_dims": 1, "num_channels": 1}},
        node_config={"type_1": {"num_nodes": 4, "workers_config": [
            {"num_workers": 1, "worker": "ddls_amd.devices.A100"}]}},
        jobs_config={"path_to_files": tiny_model_files,
                     "replication_factor": 2, "job_sampling_mode": "remove",
                     "job_interarrival_time_dist": {
                         "_target_": "ddls_amd.distributions.Fixed", "val": 1.0},
                     "num_training_steps": 2})
    obs = env.reset(seed=0)
    assert obs.shape == (6,)
    done, guard = False, 0
    while not done and guard < 20:
        obs, r, done, _ = env.step(2)
        guard += 1
    assert env.cluster.episode_stats["num_jobs_completed"] == 2


def test_job_managers_extra(tiny_model_files):
    from ddls_amd.agents.job_managers import (AllReduceJobCommunicator,
                                              SRPTJobPrioritiser)
    env = make_env(tiny_model_files, replication=2)
    env.reset(seed=0)
    jobs = list(env.cluster.job_queue.jobs.values())
    pri = SRPTJobPrioritiser().prioritise(jobs)
    assert len(pri) == len(jobs)
    comm = AllReduceJobCommunicator()
    assert comm.all_reduce_time(1e9, 1) == 0.0
    t2, t8 = comm.all_reduce_time(1e9, 2), comm.all_reduce_time(1e9, 8)
    assert t2 > 0 and t8 > t2 * 0.5  # per-link bound: weak n-dependence


def test_pipeline_cache_trace_equal(tiny_model_files):
    """Cached empty-cluster pipeline must reproduce the uncached episode
    exactly (rewards + episode stats)."""
    def run(cache):
        env = make_env(tiny_model_files, replication=6, frac_dist={
            "_target_": "ddls_amd.distributions.Uniform",
            "min_val": 0.1, "max_val": 1.0, "decimals": 2})
        env.cache_pipeline = cache
        obs = env.reset(seed=42)
        rewards, done = [], False
        actions = [1, 2, 4, 2, 16, 8]
        i = 0
        while not done:
            a = actions[i % len(actions)]
            if not obs["action_mask"][a]:
                a = 0
            obs, r, done, _ = env.step(a)
            rewards.append(r)
            i += 1
        return rewards, dict(env.cluster.episode_stats)

    r1, s1 = run(cache=False)
    r2, s2 = run(cache=True)
    assert r1 == pytest.approx(r2, rel=1e-12)
    assert s1["num_jobs_completed"] == s2["num_jobs_completed"]
    assert s1["num_jobs_blocked"] == s2["num_jobs_blocked"]
    assert s1["job_completion_time"] == pytest.approx(s2["job_completion_time"])


def _make_shaping_env(tiny_model_files):
    from ddls_amd.envs.ramp_job_placement_shaping import (
        RampJobPlacementShapingEnvironment)
    return RampJobPlacementShapingEnvironment(
        topology_config={"type": "ramp", "kwargs": {
            "num_communication_groups": 4,
            "num_racks_per_communication_group": 4,
            "num_servers_per_rack": 2, "num_channels": 1,
            "total_node_bandwidth": 1.6e12,
            "intra_gpu_propagation_latency": 50e-9,
            "worker_io_latency": 100e-9}},
        node_config={"type_1": {"num_nodes": 32, "workers_config": [
            {"num_workers": 1, "worker": "ddls_amd.devices.A100"}]}},
        jobs_config={"path_to_files": tiny_model_files,
                     "replication_factor": 3,
                     "job_sampling_mode": "remove",
                     "job_interarrival_time_dist": {
                         "_target_": "ddls_amd.distributions.Fixed",
                         "val": 1000},
                     "max_acceptable_job_completion_time_frac_dist": {
                         "_target_": "ddls_amd.distributions.Fixed", "val": 1.0},
                     "num_training_steps": 5,
                     "max_partitions_per_op_in_observation": 2},
        op_partitioner_kwargs={"min_op_run_time_quantum": 0.01},
        max_simulation_run_time=1e6)


def test_shaping_env_baseline_agents(tiny_model_files):
    """first/last/random shape pickers run episodes on the shaping env
    (reference ramp_job_placement_shaping/agents/, 3 baseline agents)."""
    from ddls_amd.envs.ramp_job_placement_shaping import (
        SHAPING_AGENTS, RampJobPlacementShapingEnvironment)
    from ddls_amd.utils import seed_everything
    for name, cls in SHAPING_AGENTS.items():
        seed_everything(4)
        env = _make_shaping_env(tiny_model_files)
        agent = cls()
        obs = env.reset(seed=4)
        done, steps = False, 0
        while not done and steps < 25:
            obs, r, done, _ = env.step(agent.compute_action(obs))
            steps += 1
        assert env.cluster.episode_stats["num_jobs_arrived"] >= 1, name
