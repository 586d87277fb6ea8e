"""GPU tests for the GPU-resident rollout path (rl/engine_env.py):
embedding-cache forward == full flat GNN forward, and end-to-end PPO
iterations on the engine."""
import numpy as np
import pytest
import torch

from tests.test_vec_engine import make_env, multi_model_files  # noqa: F401


@pytest.mark.gpu
def test_embedding_cache_forward_matches_flat(multi_model_files):
    """The per-model GNN embedding cache + tiny head must reproduce
    GNNPolicy.forward_flat exactly (same weights, same obs)."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.engine_env import EngineVectorEnv
    from ddls_amd.rl.rollout import collate

    dev = torch.device("cuda:0")
    torch.manual_seed(3)
    policy = GNNPolicy(num_actions=17).to(dev)
    venv = EngineVectorEnv(
        lambda: make_env(multi_model_files, "remove_and_repeat", 2, 5000, 20),
        num_envs=8, device=dev, base_seed=5)

    # run a few engine steps so obs vary across envs
    data = venv.rollout(policy, steps=5)
    obs_list = venv._obs_list()
    inputs = collate(obs_list, dev)
    with torch.no_grad():
        ref_logits, ref_values = policy.forward_flat(
            inputs["batch"], inputs["graph_features"], inputs["action_mask"])
        from ddls_amd.models.gnn import graph_mean
        node_emb = policy.gnn(venv._models_batch)
        model_emb = graph_mean(node_emb, venv._models_batch)
        gf = venv.eng.T["obs_gf"]
        mask = venv.eng.T["obs_mask"]
        gfull = torch.cat([gf, mask], dim=1)
        logits, values = venv._head_forward(
            policy, model_emb, venv.eng.T["obs_model"].long(), gfull, mask)
    np.testing.assert_allclose(logits.cpu().numpy(), ref_logits.cpu().numpy(),
                               rtol=0, atol=2e-5)
    np.testing.assert_allclose(values.cpu().numpy(), ref_values.cpu().numpy(),
                               rtol=0, atol=2e-5)


@pytest.mark.gpu
def test_ppo_on_engine_env(multi_model_files):
    """Two PPO iterations end-to-end on the GPU-resident envs: finite stats,
    captured SGD path, trajectory shapes."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.engine_env import EngineVectorEnv
    from ddls_amd.rl.ppo import PPOConfig, PPOTrainer

    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    policy = GNNPolicy(num_actions=17).to(dev)
    venv = EngineVectorEnv(
        lambda: make_env(multi_model_files, "remove_and_repeat", 2, 3000, 15),
        num_envs=16, device=dev, base_seed=11)
    cfg = PPOConfig(train_batch_size=16 * 8, sgd_minibatch_size=32,
                    num_sgd_iter=4)
    trainer = PPOTrainer(venv, policy, cfg, device=dev)
    for _ in range(2):
        stats = trainer.train(num_steps=8)
        assert np.isfinite(stats["total_loss"])
        assert np.isfinite(stats["mean_reward"])
    assert trainer.total_env_steps == 2 * 16 * 8
    # episodes complete (max_sim 3000 / interarrival 15 = 200 arrivals) only
    # on long runs; here just check the stats plumbing exists
    venv.drain_episode_stats()
    c, s = venv.jct_running_stats()
    assert c > 0 and np.isfinite(s)


@pytest.mark.gpu
def test_cached_models_sgd_matches_eager(multi_model_files):
    """Cached-models captured SGD (GNN once per model, grads summed via
    index_select) vs the eager per-sample path: same trajectories, updated
    params must agree to fp-reordering tolerance."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.engine_env import EngineVectorEnv
    from ddls_amd.rl.ppo import PPOConfig, PPOTrainer

    dev = torch.device("cuda:0")

    def build(use_graphs):
        torch.manual_seed(0)
        policy = GNNPolicy(num_actions=17).to(dev)
        venv = EngineVectorEnv(
            lambda: make_env(multi_model_files, "remove_and_repeat", 2,
                             3000, 15),
            num_envs=16, device=dev, base_seed=21)
        cfg = PPOConfig(train_batch_size=16 * 8, sgd_minibatch_size=32,
                        num_sgd_iter=2, use_hip_graphs=use_graphs)
        return PPOTrainer(venv, policy, cfg, device=dev)

    t_cap = build(True)
    t_eag = build(False)
    torch.manual_seed(1234)
    s_cap = t_cap.train(num_steps=8)
    torch.manual_seed(1234)
    s_eag = t_eag.train(num_steps=8)
    assert s_cap.get("hipgraph_minibatches", 0) > 0, "captured path not used"
    assert np.isclose(s_cap["mean_reward"], s_eag["mean_reward"]), \
        "trajectories diverged (rollout should be identical)"
    for (k1, p1), (k2, p2) in zip(t_cap.policy.state_dict().items(),
                                  t_eag.policy.state_dict().items()):
        np.testing.assert_allclose(p1.cpu().numpy(), p2.cpu().numpy(),
                                   rtol=0, atol=5e-4, err_msg=k1)


@pytest.mark.gpu
def test_fused_cached_step_grads_match_autograd(multi_model_files):
    """cached_step_fwd/bwd (whole-net analytic backward into flat_g) vs the
    torch-autograd cached-models path on one identical minibatch: loss stats
    and every flat gradient must agree."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.engine_env import EngineVectorEnv
    from ddls_amd.rl.graph_step import CapturedSGDStep, _FusedCachedEngine
    from ddls_amd.rl.ppo import PPOConfig

    dev = torch.device("cuda:0")
    torch.manual_seed(4)
    policy = GNNPolicy(num_actions=17).to(dev)
    venv = EngineVectorEnv(
        lambda: make_env(multi_model_files, "remove_and_repeat", 2, 4000, 20),
        num_envs=8, device=dev, base_seed=31)
    data = venv.rollout(policy, steps=8)   # 64 samples
    obs = data["obs"]

    cfg = PPOConfig(sgd_minibatch_size=32, num_sgd_iter=1)
    opt = torch.optim.Adam(policy.parameters(), lr=cfg.lr)
    st = CapturedSGDStep(policy, opt, cfg, dev,
                         models_batch=venv._models_batch)
    assert _FusedCachedEngine.eligible(policy, st._ext)
    st._alloc(0, 0)
    st._flatten_params()
    st.set_kl_coeff(cfg.kl_coeff)
    fused = _FusedCachedEngine(st, venv._models_batch)

    # stage one minibatch
    idx = np.arange(32)
    mb = [obs[i] for i in idx]
    rng = np.random.RandomState(0)
    actions = data["actions"].reshape(-1)[idx]
    old_logp = data["logp"].reshape(-1)[idx].astype(np.float32)
    adv = rng.randn(32).astype(np.float32)
    vtarg = rng.randn(32).astype(np.float32)
    st.d["model_ids"].copy_(torch.as_tensor(
        [o.model_id for o in mb], dtype=torch.int64))
    st.d["gf"].copy_(torch.as_tensor(np.stack(
        [o.graph_features for o in mb])))
    st.d["mask"].copy_(torch.as_tensor(np.stack(
        [o.action_mask for o in mb])))
    st.d["actions"].copy_(torch.as_tensor(actions))
    st.d["old_logp"].copy_(torch.as_tensor(old_logp))
    st.d["adv"].copy_(torch.as_tensor(adv))
    st.d["vtarg"].copy_(torch.as_tensor(vtarg))

    # fused kernels
    st.reset_stats()
    fused.run()
    torch.cuda.synchronize()
    g_fused = st.flat_g.detach().clone()
    stats_fused = st.stats_acc.detach().clone()

    # torch-autograd reference (the stepper's non-fused cached branch)
    st._fused_cached = None
    st.flat_g.zero_()
    st.reset_stats()
    st._body_fwd_bwd()
    torch.cuda.synchronize()
    g_ref = st.flat_g.detach().clone()
    stats_ref = st.stats_acc.detach().clone()

    np.testing.assert_allclose(stats_fused.cpu().numpy(),
                               stats_ref.cpu().numpy(), rtol=1e-4,
                               atol=1e-5, err_msg="loss stats")
    gf_, gr_ = g_fused.cpu().numpy(), g_ref.cpu().numpy()
    scale = np.abs(gr_).max()
    np.testing.assert_allclose(gf_, gr_, rtol=0, atol=max(scale, 1.0) * 2e-5,
                               err_msg="flat gradients")


@pytest.mark.gpu
def test_bench_dp2_gloo_single_gpu_rehearsal(tmp_path):
    """World-size-2 rehearsal of the data-parallel bench on ONE GPU (gloo
    backend, host-staged flat all-reduce; the real 8-GPU run uses RCCL with
    the same code path).  Validates the engine + split-capture stepper under
    torch.distributed end to end."""
    import json
    import os
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update({"DDLS_AMD_DIST_BACKEND": "gloo",
                "MASTER_ADDR": "127.0.0.1"})
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29731", "bench.py", "--gpus", "2",
           "--steps", "2", "--warmup", "1", "--envs-per-rank", "16",
           "--rollout-steps-per-env", "4", "--num-sgd-iter", "2",
           "--sgd-minibatch-size", "32"]
    out = subprocess.run(cmd, cwd=root, env=env, capture_output=True,
                         text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2
    assert rec["config"]["env_engine"] == "gpu_resident"
    assert rec["value"] > 0


def test_ppo_on_cpu_engine_env(multi_model_files):
    """The CpuEngine backend (vec_engine.CpuEngine — the parity-tested
    mirror as a vectorised engine): EngineVectorEnv on device='cpu' runs
    PPO end-to-end with the same trainer-facing surface as the GPU engine,
    replacing subprocess env workers on CPU-only machines."""
    from ddls_amd.cluster.vec_engine import CpuEngine
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.engine_env import EngineVectorEnv
    from ddls_amd.rl.ppo import PPOConfig, PPOTrainer

    dev = torch.device("cpu")
    torch.manual_seed(0)
    policy = GNNPolicy(num_actions=17)
    venv = EngineVectorEnv(
        lambda: make_env(multi_model_files, "remove_and_repeat", 2, 3000, 15),
        num_envs=8, device=dev, base_seed=11)
    assert isinstance(venv.eng, CpuEngine)
    cfg = PPOConfig(train_batch_size=8 * 8, sgd_minibatch_size=32,
                    num_sgd_iter=3)
    trainer = PPOTrainer(venv, policy, cfg, device=dev)
    for _ in range(2):
        stats = trainer.train(num_steps=8)
        assert np.isfinite(stats["total_loss"])
        assert np.isfinite(stats["mean_reward"])
    assert trainer.total_env_steps == 2 * 8 * 8
    venv.drain_episode_stats()
    c, s = venv.jct_running_stats()
    assert c > 0 and np.isfinite(s)


def test_cpu_engine_episode_stats_roundtrip(multi_model_files):
    """CpuEngine full episode: completed-episode stats drain through
    EngineVectorEnv and carry the exact episode-stats vocabulary."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.engine_env import EngineVectorEnv

    torch.manual_seed(1)
    policy = GNNPolicy(num_actions=17)
    venv = EngineVectorEnv(
        lambda: make_env(multi_model_files, "remove", 2, 2500, 15),
        num_envs=4, device=torch.device("cpu"), base_seed=3)
    total = 0
    for _ in range(40):
        venv.rollout(policy, steps=4)
        stats = venv.drain_episode_stats()
        total += len(stats)
        for es in stats:
            assert es["num_jobs_arrived"] > 0
            assert 0.0 <= es["blocking_rate"] <= 1.0
        if total >= 2:
            break
    assert total >= 1, "at least one episode should complete"


def test_bench_dp2_cpu_engine_rehearsal():
    """World-size-2 bench on CPU (gloo + CpuEngine): the same distributed
    code path (barriers, fused flat all-reduce, MAX-over-ranks timing,
    rank-0 single JSON line) the driver runs at 8 GPUs — exercised HERE
    without any GPU."""
    import json
    import os
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update({"DDLS_AMD_DIST_BACKEND": "gloo",
                "MASTER_ADDR": "127.0.0.1"})
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29741", "bench.py", "--gpus", "2",
           "--steps", "1", "--warmup", "0", "--envs-per-rank", "8",
           "--rollout-steps-per-env", "2", "--num-sgd-iter", "1",
           "--sgd-minibatch-size", "16"]
    out = subprocess.run(cmd, cwd=root, env=env, capture_output=True,
                         text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["env_engine"] == "cpu_engine"
    assert rec["value"] > 0
    assert rec["config"]["mean_simulated_jct"] is not None
