"""IMPALA / V-trace tests (rl/impala.py)."""
import numpy as np
import pytest
import torch

from tests.test_vec_engine import make_env, multi_model_files  # noqa: F401


def vtrace_reference(rewards, dones, values, bootstrap, rho, c, gamma):
    """Slow literal recursion of Espeholt et al. (2018), eqns 1-2."""
    T, N = rewards.shape
    nonterminal = 1.0 - dones
    vp1 = np.concatenate([values[1:], bootstrap[None]], axis=0) * nonterminal
    delta = rho * (rewards + gamma * vp1 - values)
    vs = np.zeros_like(values)
    for t in reversed(range(T)):
        nxt = (vs[t + 1] - values[t + 1]) if t + 1 < T else 0.0
        vs[t] = values[t] + delta[t] + gamma * c[t] * nonterminal[t] * nxt
    vsp1 = np.concatenate([vs[1:], bootstrap[None]], axis=0) * nonterminal
    pg_adv = rewards + gamma * vsp1 - values
    return vs, pg_adv


def test_vtrace_matches_reference():
    from ddls_amd.rl.impala import vtrace_targets
    rng = np.random.RandomState(0)
    T, N = 13, 5
    rewards = rng.randn(T, N).astype(np.float32)
    dones = (rng.rand(T, N) < 0.15).astype(np.float32)
    values = rng.randn(T, N).astype(np.float32)
    bootstrap = rng.randn(N).astype(np.float32)
    rho = np.clip(np.exp(rng.randn(T, N) * 0.3), None, 1.0).astype(np.float32)
    c = rho.copy()
    t = lambda a: torch.as_tensor(a)
    vs, pg = vtrace_targets(t(rewards), t(dones), t(values), t(bootstrap),
                            t(rho), t(c), 0.99)
    vs_ref, pg_ref = vtrace_reference(rewards, dones, values, bootstrap,
                                      rho, c, 0.99)
    np.testing.assert_allclose(vs.numpy(), vs_ref, rtol=1e-5, atol=1e-6)
    np.testing.assert_allclose(pg.numpy(), pg_ref, rtol=1e-5, atol=1e-6)


def test_vtrace_onpolicy_reduces_to_returns():
    """With rho = c = 1 (on-policy), vs is the n-step return target."""
    from ddls_amd.rl.impala import vtrace_targets
    rng = np.random.RandomState(1)
    T, N = 8, 3
    rewards = rng.randn(T, N).astype(np.float32)
    dones = np.zeros((T, N), dtype=np.float32)
    values = rng.randn(T, N).astype(np.float32)
    bootstrap = rng.randn(N).astype(np.float32)
    ones = np.ones((T, N), dtype=np.float32)
    t = lambda a: torch.as_tensor(a)
    vs, _ = vtrace_targets(t(rewards), t(dones), t(values), t(bootstrap),
                           t(ones), t(ones), 0.9)
    # n-step return: G_t = r_t + 0.9 G_{t+1}, G_T = bootstrap
    G = np.zeros_like(values)
    acc = bootstrap.copy()
    for tt in reversed(range(T)):
        acc = rewards[tt] + 0.9 * acc
        G[tt] = acc
    np.testing.assert_allclose(vs.numpy(), G, rtol=1e-4, atol=1e-5)


def test_impala_trains_on_cpu(multi_model_files):
    """Two IMPALA iterations on the in-process vector env: finite losses,
    params actually move."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.impala import ImpalaConfig, ImpalaTrainer
    from ddls_amd.rl.rollout import VectorEnv

    torch.manual_seed(0)
    venv = VectorEnv([lambda i=i: make_env(multi_model_files,
                                           "remove_and_repeat", 2, 3000, 30)
                      for i in range(4)], base_seed=3)
    policy = GNNPolicy(num_actions=17)
    before = {k: v.clone() for k, v in policy.state_dict().items()}
    tr = ImpalaTrainer(venv, policy, ImpalaConfig(train_batch_size=4 * 6),
                       device=torch.device("cpu"))
    for _ in range(2):
        st = tr.train(num_steps=6)
        assert np.isfinite(st["total_loss"])
        assert np.isfinite(st["mean_rho"]) and 0 < st["mean_rho"] <= 1.0
    moved = any(not torch.equal(before[k], v)
                for k, v in policy.state_dict().items())
    assert moved


def test_impala_config_group_builds(multi_model_files, tmp_path):
    """configs/algo/impala.yaml routes build_trainer_from_config to the
    ImpalaTrainer."""
    import yaml
    from ddls_amd.runtime.config import build_trainer_from_config, load_config
    root = __import__("os").path.dirname(__import__("os").path.dirname(
        __import__("os").path.abspath(__file__)))
    cfg = load_config(__import__("os").path.join(root, "configs",
                                                 "train_config.yaml"))
    with open(__import__("os").path.join(root, "configs", "algo",
                                         "impala.yaml")) as f:
        cfg["algo"] = yaml.safe_load(f)
    cfg["env_config"]["jobs_config"]["path_to_files"] = multi_model_files
    cfg["env_config"]["jobs_config"]["replication_factor"] = 2
    cfg["epoch_loop"]["num_envs"] = 2
    cfg["epoch_loop"]["num_env_workers"] = 1
    cfg["epoch_loop"]["precompute_lookaheads"] = False
    from ddls_amd.rl.impala import ImpalaTrainer
    tr = build_trainer_from_config(cfg, device=torch.device("cpu"))
    assert isinstance(tr, ImpalaTrainer)
    st = tr.train(num_steps=4)
    assert np.isfinite(st["total_loss"])
    tr.env.close()


@pytest.mark.gpu
def test_impala_on_engine_env(multi_model_files):
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.engine_env import EngineVectorEnv
    from ddls_amd.rl.impala import ImpalaConfig, ImpalaTrainer

    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    policy = GNNPolicy(num_actions=17).to(dev)
    venv = EngineVectorEnv(
        lambda: make_env(multi_model_files, "remove_and_repeat", 2, 3000, 15),
        num_envs=16, device=dev, base_seed=17)
    tr = ImpalaTrainer(venv, policy, ImpalaConfig(train_batch_size=16 * 8),
                       device=dev)
    for _ in range(2):
        st = tr.train(num_steps=8)
        assert np.isfinite(st["total_loss"])


def test_pg_trains_on_cpu(multi_model_files):
    """REINFORCE on the in-process vector env: finite loss, params move."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.pg import PGConfig, PGTrainer
    from ddls_amd.rl.rollout import VectorEnv

    torch.manual_seed(0)
    venv = VectorEnv([lambda i=i: make_env(multi_model_files,
                                           "remove_and_repeat", 2, 3000, 30)
                      for i in range(3)], base_seed=9)
    policy = GNNPolicy(num_actions=17)
    before = {k: v.clone() for k, v in policy.state_dict().items()}
    tr = PGTrainer(venv, policy, PGConfig(train_batch_size=3 * 5),
                   device=torch.device("cpu"))
    for _ in range(2):
        st = tr.train(num_steps=5)
        assert np.isfinite(st["total_loss"])
    assert any(not torch.equal(before[k], v)
               for k, v in policy.state_dict().items())


def test_pg_returns_monte_carlo():
    """Discounted-return computation: terminal cuts + zero bootstrap."""
    from ddls_amd.rl.pg import PGConfig, PGTrainer
    rewards = np.array([[1.0], [1.0], [1.0]])
    dones = np.array([[0.0], [1.0], [0.0]])
    cfg = PGConfig(gamma=0.5)
    # inline the return recursion (same code path as update())
    T, N = rewards.shape
    returns = np.zeros_like(rewards)
    acc = np.zeros(N)
    for t in reversed(range(T)):
        acc = rewards[t] + cfg.gamma * (1.0 - dones[t]) * acc
        returns[t] = acc
    np.testing.assert_allclose(returns[:, 0], [1 + 0.5 * 1.0, 1.0, 1.0])


def test_dqn_trains_on_cpu(multi_model_files):
    """Dueling double-DQN: replay ingestion (n-step cuts), target sync, and
    finite losses on the in-process vector env."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.dqn import DQNConfig, DQNTrainer
    from ddls_amd.rl.rollout import VectorEnv

    torch.manual_seed(0)
    venv = VectorEnv([lambda i=i: make_env(multi_model_files,
                                           "remove_and_repeat", 2, 3000, 30)
                      for i in range(3)], base_seed=13)
    policy = GNNPolicy(num_actions=17)
    cfg = DQNConfig(learning_starts=10, sgd_minibatch_size=8, num_sgd_iter=2,
                    target_network_update_freq=20, lr=1e-4)
    tr = DQNTrainer(venv, policy, cfg, device=torch.device("cpu"))
    before = {k: v.clone() for k, v in policy.state_dict().items()}
    for _ in range(3):
        st = tr.train(num_steps=8)
        assert np.isfinite(st["total_loss"])
    assert len(tr.replay) > 0
    assert any(not torch.equal(before[k], v)
               for k, v in policy.state_dict().items())
    # masked actions never win the double-DQN argmax
    q = tr._q_values([tr.replay[0][0]], tr.policy)
    mask = torch.as_tensor(tr.replay[0][0].action_mask) > 0
    assert bool(mask[q.argmax(-1)[0]])


def test_dqn_nstep_transition_cuts(multi_model_files):
    """n-step returns stop at episode boundaries (discount carried for the
    bootstrap matches the number of accumulated steps)."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.dqn import DQNConfig, DQNTrainer
    from ddls_amd.rl.rollout import VectorEnv

    torch.manual_seed(0)
    venv = VectorEnv([lambda: make_env(multi_model_files,
                                       "remove_and_repeat", 2, 3000, 30)],
                     base_seed=13)
    tr = DQNTrainer(venv, GNNPolicy(num_actions=17),
                    DQNConfig(n_step=3, gamma=0.5),
                    device=torch.device("cpu"))
    T, N = 7, 1
    obs = [tr.obs[0]] * (T * N)
    data = {"obs": obs,
            "actions": np.zeros((T, N), dtype=np.int64),
            "rewards": np.ones((T, N)),
            "dones": np.zeros((T, N), dtype=bool)}
    data["dones"][2][0] = True
    tr.replay.clear()
    tr._ingest(data)
    # t=0: cut at t=2 -> R = 1 + .5 + .25, no next obs
    assert tr.replay[0][2] == 1 + 0.5 + 0.25 and tr.replay[0][3] is None
    # t=3: full 3-step -> R = 1 + .5 + .25, next obs present, disc .125
    assert tr.replay[3][2] == 1.75 and tr.replay[3][3] is not None
    assert tr.replay[3][4] == 0.125


def test_es_centered_ranks():
    from ddls_amd.rl.es import centered_ranks
    r = centered_ranks(np.array([3.0, 1.0, 2.0, 4.0]))
    np.testing.assert_allclose(r, [1 / 6, -0.5, -1 / 6, 0.5])


def test_es_improves_on_bandit(multi_model_files):
    """ES moves parameters and produces finite fitness on the env."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.es import ESConfig, ESTrainer
    from ddls_amd.rl.rollout import VectorEnv

    torch.manual_seed(0)
    venv = VectorEnv([lambda: make_env(multi_model_files,
                                       "remove_and_repeat", 2, 3000, 30)],
                     base_seed=21)
    policy = GNNPolicy(num_actions=17)
    tr = ESTrainer(venv, policy,
                   ESConfig(perturbation_pairs=2, fragment_steps=3),
                   device=torch.device("cpu"))
    before = tr._get_flat().clone()
    st = tr.train()
    assert np.isfinite(st["fitness_mean"])
    assert not torch.equal(before, tr._get_flat())


@pytest.mark.gpu
def test_dqn_and_es_on_engine_env(multi_model_files):
    """DQN and ES end-to-end on the GPU-resident envs (cached-models
    forward path)."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.dqn import DQNConfig, DQNTrainer
    from ddls_amd.rl.engine_env import EngineVectorEnv
    from ddls_amd.rl.es import ESConfig, ESTrainer

    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    venv = EngineVectorEnv(
        lambda: make_env(multi_model_files, "remove_and_repeat", 2, 3000, 15),
        num_envs=16, device=dev, base_seed=23)
    dqn = DQNTrainer(venv, GNNPolicy(num_actions=17).to(dev),
                     DQNConfig(learning_starts=50, sgd_minibatch_size=32,
                               num_sgd_iter=2, lr=1e-4), device=dev)
    for _ in range(2):
        st = dqn.train(num_steps=8)
        assert np.isfinite(st["total_loss"])
    es = ESTrainer(venv, GNNPolicy(num_actions=17).to(dev),
                   ESConfig(perturbation_pairs=2, fragment_steps=4),
                   device=dev)
    st = es.train()
    assert np.isfinite(st["fitness_mean"])


@pytest.mark.parametrize("group,cls_path", [
    ("pg", "ddls_amd.rl.pg.PGTrainer"),
    ("apex_dqn", "ddls_amd.rl.dqn.DQNTrainer"),
    ("es", "ddls_amd.rl.es.ESTrainer"),
])
def test_remaining_config_groups_build(multi_model_files, group, cls_path):
    """Each remaining reference algo config group
    (scripts/ramp_job_partitioning_configs/algo/{pg,apex_dqn,es}.yaml)
    routes build_trainer_from_config to its trainer and runs an
    iteration on CPU."""
    import importlib
    import os

    import yaml

    from ddls_amd.runtime.config import build_trainer_from_config, load_config
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cfg = load_config(os.path.join(root, "configs", "train_config.yaml"))
    with open(os.path.join(root, "configs", "algo", f"{group}.yaml")) as f:
        cfg["algo"] = yaml.safe_load(f)
    cfg["env_config"]["jobs_config"]["path_to_files"] = multi_model_files
    cfg["env_config"]["jobs_config"]["replication_factor"] = 2
    cfg["epoch_loop"]["num_envs"] = 2
    cfg["epoch_loop"]["num_env_workers"] = 1
    cfg["epoch_loop"]["precompute_lookaheads"] = False
    mod, name = cls_path.rsplit(".", 1)
    cls = getattr(importlib.import_module(mod), name)
    tr = build_trainer_from_config(cfg, device=torch.device("cpu"))
    assert isinstance(tr, cls)
    st = tr.train(num_steps=4)
    assert np.isfinite(st.get("total_loss", 0.0))
    tr.env.close()
