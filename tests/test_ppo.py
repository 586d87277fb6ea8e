"""PPO trainer tests (CPU) + multi-process DP all-reduce test (gloo)."""
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

from tests.conftest import make_env


def _make_trainer(jobs_dir, n_envs=2, train_batch=16, minibatch=8, iters=2):
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.ppo import PPOConfig, PPOTrainer
    from ddls_amd.rl.rollout import VectorEnv

    def fn():
        return make_env(jobs_dir, replication=50,
                        frac_dist={"_target_": "ddls_amd.distributions.Uniform",
                                   "min_val": 0.1, "max_val": 1.0,
                                   "decimals": 2})
    venv = VectorEnv([fn for _ in range(n_envs)], base_seed=0)
    torch.manual_seed(0)
    policy = GNNPolicy(num_actions=17)
    cfg = PPOConfig(train_batch_size=train_batch, sgd_minibatch_size=minibatch,
                    num_sgd_iter=iters)
    return PPOTrainer(venv, policy, cfg, device=torch.device("cpu"))


def test_ppo_iteration_runs_and_is_finite(tiny_model_files):
    trainer = _make_trainer(tiny_model_files)
    stats = trainer.train()
    assert np.isfinite(stats["total_loss"])
    assert np.isfinite(stats["kl"])
    assert stats["env_steps_this_iter"] == 16
    assert trainer.total_env_steps == 16


def test_ppo_gae_shapes(tiny_model_files):
    trainer = _make_trainer(tiny_model_files)
    batch = trainer.collect_rollout(num_steps=5)
    assert len(batch["actions"]) == 10  # 5 steps x 2 envs
    assert batch["advantages"].shape == (10,)
    assert np.isfinite(batch["advantages"]).all()
    assert np.isfinite(batch["value_targets"]).all()


def test_ppo_actions_respect_mask(tiny_model_files):
    trainer = _make_trainer(tiny_model_files)
    for _ in range(3):
        obs = trainer.obs
        actions, logp, values, logits = trainer._policy_step(obs)
        for i, o in enumerate(obs):
            assert o.action_mask[actions[i]] == 1, \
                f"sampled masked action {actions[i]}"
        trainer.obs, _, _ = trainer.env.step(actions)


def test_ppo_checkpoint_roundtrip(tiny_model_files, tmp_path):
    trainer = _make_trainer(tiny_model_files)
    trainer.train()
    state = trainer.state_dict()
    trainer2 = _make_trainer(tiny_model_files)
    trainer2.load_state_dict(state)
    for p1, p2 in zip(trainer.policy.parameters(), trainer2.policy.parameters()):
        assert torch.equal(p1, p2)
    assert trainer2.iteration == trainer.iteration


_DIST_SCRIPT = r"""
import os, sys
import torch, torch.distributed as dist
sys.path.insert(0, os.environ["DDLS_REPO"])
from ddls_amd.parallel import init_distributed_from_env, all_reduce_gradients, get_world_size

rank = init_distributed_from_env()
torch.manual_seed(0)
m = torch.nn.Linear(4, 4)
x = torch.full((2, 4), float(rank + 1))
m(x).sum().backward()
all_reduce_gradients(m.parameters())
# mean of grads for inputs filled with 1 and 2 -> equals grads for input 1.5
expected = torch.full((4,), 2 * 1.5)
assert torch.allclose(m.weight.grad, expected.repeat(4, 1)), m.weight.grad
print(f"RANK{rank}_OK")
"""


def test_dp_all_reduce_gloo(tmp_path):
    """world_size=2 gradient all-reduce over gloo (the CPU stand-in for RCCL)."""
    script = tmp_path / "dist_worker.py"
    script.write_text(_DIST_SCRIPT)
    env = dict(os.environ)
    env["DDLS_REPO"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = "29617"
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, str(script)],
                                      env=e, stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=120)[0].decode() for p in procs]
    for rank, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, out
        assert f"RANK{rank}_OK" in out


def test_subproc_vector_env(tiny_model_files):
    from ddls_amd.rl.subproc_env import SubprocVectorEnv

    def fn():
        return make_env(tiny_model_files, replication=50)

    venv = SubprocVectorEnv(fn, num_envs=4, num_workers=2, base_seed=0)
    try:
        obs = venv.reset()
        assert len(obs) == 4
        for _ in range(3):
            actions = np.array([int(o.action_mask.argmax()) for o in obs])
            obs, rewards, dones = venv.step(actions)
            assert len(obs) == 4 and rewards.shape == (4,)
            assert np.isfinite(rewards).all()
    finally:
        venv.close()


def test_worker_side_rollouts(tiny_model_files):
    """RLlib-style worker-side rollout path: trajectories have the right
    shapes, masked actions only, finite GAE."""
    import torch
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.ppo import PPOConfig, PPOTrainer
    from ddls_amd.rl.subproc_env import SubprocVectorEnv

    def fn():
        return make_env(tiny_model_files, replication=50)

    venv = SubprocVectorEnv(fn, num_envs=4, num_workers=2, base_seed=0)
    try:
        torch.manual_seed(0)
        policy = GNNPolicy(num_actions=17)
        trainer = PPOTrainer(venv, policy,
                             PPOConfig(train_batch_size=24,
                                       sgd_minibatch_size=8, num_sgd_iter=2),
                             device=torch.device("cpu"))
        batch = trainer.collect_rollout(num_steps=6)
        assert len(batch["actions"]) == 24
        assert len(batch["obs"]) == 24
        assert np.isfinite(batch["advantages"]).all()
        stats = trainer.update(batch)
        assert np.isfinite(stats["total_loss"])
        full = trainer.train(num_steps=6)
        assert full["env_steps_this_iter"] == 24
    finally:
        venv.close()


def _random_compact_obs(rng, num_actions=17):
    from ddls_amd.rl.rollout import CompactObs
    n = int(rng.randint(4, 20))
    m = int(rng.randint(3, 3 * n))
    mask = np.ones(num_actions, dtype=np.float32)
    mask[rng.randint(1, num_actions)] = 0
    return CompactObs(
        node_features=rng.rand(n, 5).astype(np.float32),
        edge_features=rng.rand(m, 2).astype(np.float32),
        edges_src=rng.randint(0, n, size=m).astype(np.int64),
        edges_dst=rng.randint(0, n, size=m).astype(np.int64),
        graph_features=rng.rand(17 + num_actions).astype(np.float32),
        action_mask=mask)


@pytest.mark.gpu
def test_hipgraph_captured_sgd_matches_eager():
    """One hipGraph-captured SGD minibatch step == the eager step (same data,
    same init) within atomics tolerance; a second step at a different
    minibatch geometry re-captures and stays finite."""
    import torch.nn.functional as F
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.graph_step import CapturedSGDStep
    from ddls_amd.rl.ppo import PPOConfig
    from ddls_amd.rl.rollout import collate

    device = torch.device("cuda:0")
    B = 8
    cfg = PPOConfig(sgd_minibatch_size=B, num_sgd_iter=1)
    rng = np.random.RandomState(0)
    mb_obs = [_random_compact_obs(rng) for _ in range(B)]
    actions = rng.randint(0, 1, size=B).astype(np.int64)  # action 0 valid
    old_logp = rng.randn(B).astype(np.float32) * 0.1 - 2.0
    adv = rng.randn(B).astype(np.float32)
    vtarg = rng.randn(B).astype(np.float32)

    def make():
        torch.manual_seed(3)
        pol = GNNPolicy(num_actions=17).to(device)
        opt = torch.optim.Adam(pol.parameters(), lr=cfg.lr)
        return pol, opt

    # eager reference step (the exact update() math)
    pol_e, opt_e = make()
    inputs = collate(mb_obs, device)
    logits, values = pol_e.forward_flat(inputs["batch"],
                                        inputs["graph_features"],
                                        inputs["action_mask"])
    logp_all = F.log_softmax(logits, dim=-1)
    logp = logp_all.gather(1, torch.as_tensor(actions, device=device)
                           .unsqueeze(1)).squeeze(1)
    ratio = torch.exp(logp - torch.as_tensor(old_logp, device=device))
    adv_t = torch.as_tensor(adv, device=device)
    surr = torch.min(ratio * adv_t,
                     torch.clamp(ratio, 1 - cfg.clip_param,
                                 1 + cfg.clip_param) * adv_t)
    policy_loss = -surr.mean()
    kl = (torch.as_tensor(old_logp, device=device) - logp).mean()
    vf_loss = torch.clamp(
        (values - torch.as_tensor(vtarg, device=device)) ** 2,
        0, cfg.vf_clip_param).mean()
    entropy = (-(logp_all.exp() * logp_all).sum(-1)).mean()
    loss = (policy_loss + cfg.kl_coeff * kl + cfg.vf_loss_coeff * vf_loss
            - cfg.entropy_coeff * entropy)
    opt_e.zero_grad()
    loss.backward()
    torch.nn.utils.clip_grad_norm_(pol_e.parameters(), cfg.grad_clip)
    opt_e.step()

    # captured step
    pol_c, opt_c = make()
    stepper = CapturedSGDStep(pol_c, opt_c, cfg, device)
    stepper.set_kl_coeff(cfg.kl_coeff)
    assert stepper.step(mb_obs, actions, old_logp, adv, vtarg), \
        "hipGraph capture failed (fell back)"
    torch.cuda.synchronize()
    assert not stepper.broken

    acc = stepper.stats_acc.cpu().numpy()
    assert np.isfinite(acc).all()
    assert acc[0] == pytest.approx(policy_loss.item(), abs=1e-3)
    assert acc[4] == pytest.approx(loss.item(), abs=1e-3)
    for (n1, p1), (n2, p2) in zip(pol_e.named_parameters(),
                                  pol_c.named_parameters()):
        assert torch.allclose(p1, p2, rtol=1e-3, atol=1e-5), n1

    # different geometry -> growth + re-capture, still sane
    mb2 = [_random_compact_obs(rng) for _ in range(B - 1)]
    mb2.append(_random_compact_obs(np.random.RandomState(99)))
    big = _random_compact_obs(np.random.RandomState(7))
    big.node_features = np.random.rand(60, 5).astype(np.float32)
    big.edge_features = np.random.rand(150, 2).astype(np.float32)
    big.edges_src = np.random.randint(0, 60, 150).astype(np.int64)
    big.edges_dst = np.random.randint(0, 60, 150).astype(np.int64)
    mb2[0] = big
    assert stepper.step(mb2, actions, old_logp, adv, vtarg)
    torch.cuda.synchronize()
    for _n, p in pol_c.named_parameters():
        assert torch.isfinite(p).all()

    # partial minibatch falls back
    assert not stepper.step(mb2[:B - 1], actions[:B - 1], old_logp[:B - 1],
                            adv[:B - 1], vtarg[:B - 1])


def test_shared_memory_rollout_matches_pickled(tiny_model_files):
    """The fork-shared weight buffer path must produce trajectories identical
    to pickling the state_dict to every worker."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.subproc_env import SubprocVectorEnv
    from tests.conftest import make_env as _mk

    def env_fn():
        return _mk(tiny_model_files, replication=50, num_training_steps=10)

    torch.manual_seed(0)
    policy = GNNPolicy(17)
    v1 = SubprocVectorEnv(env_fn, num_envs=4, num_workers=2, base_seed=3)
    a1 = v1.rollout(policy, 5)
    b1 = v1.rollout(policy, 5)
    assert v1._shm_views is not None  # shm path actually engaged
    v1.close()
    v2 = SubprocVectorEnv(env_fn, num_envs=4, num_workers=2, base_seed=3)
    v2._shm_buf = None  # force pickled fallback
    a2 = v2.rollout(policy, 5)
    b2 = v2.rollout(policy, 5)
    v2.close()
    for k in ("actions", "logp", "values", "rewards", "dones"):
        assert np.array_equal(a1[k], a2[k]), k
        assert np.array_equal(b1[k], b2[k]), k


_DIST_CAPTURE_SCRIPT = r"""
import os, sys
import numpy as np
import torch
sys.path.insert(0, os.environ["DDLS_REPO"])
from ddls_amd.parallel import init_distributed_from_env
from ddls_amd.models.gnn import GNNPolicy
from ddls_amd.rl.graph_step import CapturedSGDStep
from ddls_amd.rl.ppo import PPOConfig
from ddls_amd.rl.rollout import CompactObs

rank = init_distributed_from_env()
device = torch.device("cuda:0")
torch.manual_seed(3)
pol = GNNPolicy(num_actions=17).to(device)
opt = torch.optim.Adam(pol.parameters(), lr=1e-3, foreach=True)
cfg = PPOConfig(sgd_minibatch_size=4)
rng = np.random.RandomState(rank)  # rank-DIFFERENT data

def obs():
    n = int(rng.randint(4, 12)); m = int(rng.randint(3, 2 * n))
    return CompactObs(
        node_features=rng.rand(n, 5).astype(np.float32),
        edge_features=rng.rand(m, 2).astype(np.float32),
        edges_src=rng.randint(0, n, m).astype(np.int64),
        edges_dst=rng.randint(0, n, m).astype(np.int64),
        graph_features=rng.rand(34).astype(np.float32),
        action_mask=np.ones(17, dtype=np.float32))

stepper = CapturedSGDStep(pol, opt, cfg, device)
stepper.set_kl_coeff(cfg.kl_coeff)
for it in range(3):
    mb = [obs() for _ in range(4)]
    ok = stepper.step(mb, np.zeros(4, dtype=np.int64),
                      rng.randn(4).astype(np.float32) * 0.1 - 2,
                      rng.randn(4).astype(np.float32),
                      rng.randn(4).astype(np.float32))
    assert ok, "capture failed"
torch.cuda.synchronize()
assert stepper.graph_opt is not None  # split capture engaged
# after the all-reduce both ranks must hold IDENTICAL params
import torch.distributed as dist
for name, p in pol.named_parameters():
    other = p.detach().clone()
    dist.broadcast(other, src=0)
    assert torch.allclose(p.detach(), other, rtol=1e-5, atol=1e-7), name
print(f"RANK{rank}_CAPTURE_OK captures={stepper.capture_count}")
"""


@pytest.mark.gpu
def test_distributed_split_capture_one_gpu():
    """world_size=2 on ONE GPU over gloo: the distributed captured step
    (fwd+bwd graph | eager all-reduce | clip+Adam graph) keeps ranks in
    lockstep — rehearses the round-end multi-GPU path without RCCL."""
    import tempfile
    with tempfile.TemporaryDirectory() as td:
        script = os.path.join(td, "dist_cap.py")
        with open(script, "w") as f:
            f.write(_DIST_CAPTURE_SCRIPT)
        env = dict(os.environ)
        env["DDLS_REPO"] = os.path.dirname(os.path.dirname(
            os.path.abspath(__file__)))
        env["MASTER_ADDR"] = "127.0.0.1"
        env["MASTER_PORT"] = "29641"
        env["DDLS_AMD_DIST_BACKEND"] = "gloo"
        procs = []
        try:
            for rank in range(2):
                e = dict(env, RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK="0")
                procs.append(subprocess.Popen([sys.executable, script], env=e,
                                              stdout=subprocess.PIPE,
                                              stderr=subprocess.STDOUT))
            outs = [p.communicate(timeout=240)[0].decode() for p in procs]
        finally:
            # never leave stragglers holding the GPU / rendezvous port
            for p in procs:
                if p.poll() is None:
                    p.kill()
                    p.wait(timeout=30)
        for rank, (p, out) in enumerate(zip(procs, outs)):
            assert p.returncode == 0, out
            assert f"RANK{rank}_CAPTURE_OK" in out


@pytest.mark.gpu
def test_fused_ppo_loss_matches_torch():
    """ops/hip/ppo_loss.hip fwd + analytic bwd vs the torch autograd chain."""
    import torch.nn.functional as F
    from ddls_amd import ops as hip_ops
    from ddls_amd.rl.graph_step import _PPOLossFn
    ext = hip_ops.get_extension(required=True)
    assert hasattr(ext, "ppo_loss_fwd")
    dev = "cuda:0"
    torch.manual_seed(11)
    B, A = 96, 17
    clip, vf_clip, vf_coef, ent_coef, kl_coef = 0.18, 128.8, 1.0, 3e-3, 0.01
    logits0 = torch.randn(B, A, device=dev) * 2
    logits0[torch.rand(B, A, device=dev) < 0.2] = torch.finfo(torch.float32).min
    values0 = torch.randn(B, device=dev) * 5
    actions = torch.zeros(B, dtype=torch.int64, device=dev)
    old_logp = torch.randn(B, device=dev) * 0.3 - 2
    adv = torch.randn(B, device=dev) * 3
    vtarg = torch.randn(B, device=dev) * 8  # some rows exceed vf_clip

    # torch reference
    lt = logits0.clone().requires_grad_(True)
    vt = values0.clone().requires_grad_(True)
    lp_all = F.log_softmax(lt, dim=-1)
    logp = lp_all.gather(1, actions.unsqueeze(1)).squeeze(1)
    ratio = torch.exp(logp - old_logp)
    surr = torch.min(ratio * adv,
                     torch.clamp(ratio, 1 - clip, 1 + clip) * adv)
    pl = -surr.mean()
    kl = (old_logp - logp).mean()
    vf = torch.clamp((vt - vtarg) ** 2, 0, vf_clip).mean()
    ent = (-(lp_all.exp() * lp_all).sum(-1)).mean()
    loss_ref = pl + kl_coef * kl + vf_coef * vf - ent_coef * ent
    loss_ref.backward()

    # fused
    lf = logits0.clone().requires_grad_(True)
    vf_ = values0.clone().requires_grad_(True)
    kl_t = torch.full((), kl_coef, device=dev)
    loss_f, stats = _PPOLossFn.apply(lf, vf_, actions, old_logp, adv, vtarg,
                                     kl_t, clip, vf_clip, vf_coef, ent_coef)
    loss_f.backward()
    torch.cuda.synchronize()

    assert loss_f.item() == pytest.approx(loss_ref.item(), abs=2e-4)
    s = stats.cpu().numpy()
    for got, want in zip(s, (pl, vf, kl, ent, loss_ref)):
        assert got == pytest.approx(want.item(), abs=2e-4)
    assert torch.allclose(lf.grad, lt.grad, atol=1e-5), \
        (lf.grad - lt.grad).abs().max().item()
    assert torch.allclose(vf_.grad, vt.grad, atol=1e-6)


@pytest.mark.gpu
def test_flat_adam_state_checkpoint_roundtrip():
    """Flat Adam state syncs back into the torch optimizer for checkpoints
    and reloads into a fresh stepper."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.graph_step import CapturedSGDStep
    from ddls_amd.rl.ppo import PPOConfig

    device = torch.device("cuda:0")
    B = 8
    cfg = PPOConfig(sgd_minibatch_size=B)
    rng = np.random.RandomState(1)
    mb = [_random_compact_obs(rng) for _ in range(B)]
    acts = np.zeros(B, dtype=np.int64)
    olp = (rng.randn(B) * 0.1 - 2).astype(np.float32)
    adv = rng.randn(B).astype(np.float32)
    vt = rng.randn(B).astype(np.float32)

    torch.manual_seed(3)
    pol = GNNPolicy(num_actions=17).to(device)
    opt = torch.optim.Adam(pol.parameters(), lr=cfg.lr, foreach=True)
    st = CapturedSGDStep(pol, opt, cfg, device)
    st.set_kl_coeff(cfg.kl_coeff)
    for _ in range(3):
        assert st.step(mb, acts, olp, adv, vt)
    torch.cuda.synchronize()
    st.sync_state_to_optimizer()
    p0 = next(iter(pol.parameters()))
    state = opt.state[p0]
    assert float(state["step"]) == 3.0
    assert torch.allclose(state["exp_avg"].reshape(-1).to(st.flat_m.device),
                          st.flat_m[:p0.numel()])

    # reload path: a fresh stepper adopts the synced torch state
    st2 = CapturedSGDStep(pol, opt, cfg, device)
    st2.set_kl_coeff(cfg.kl_coeff)
    assert st2.step(mb, acts, olp, adv, vt)
    torch.cuda.synchronize()
    assert float(st2.step_t.item()) == 4.0  # continued from step 3


def test_launcher_checkpoints_and_best_tracking(tiny_model_files, tmp_path):
    """Launcher: checkpoint at epoch 0 + every evaluation_interval, best
    checkpoint tracked by eval episode return, results log populated
    (reference launcher.py:97-184, rllib_epoch_loop.py:144-230)."""
    from ddls_amd.models.gnn import GNNPolicy
    from ddls_amd.rl.ppo import PPOConfig, PPOTrainer
    from ddls_amd.rl.rollout import VectorEnv
    from ddls_amd.runtime.loops import EpochLoop, Launcher

    torch.manual_seed(0)
    venv = VectorEnv([lambda: make_env(tiny_model_files, replication=50)
                      for _ in range(2)], base_seed=3)
    trainer = PPOTrainer(venv, GNNPolicy(num_actions=17),
                         PPOConfig(train_batch_size=8, sgd_minibatch_size=8,
                                   num_sgd_iter=1),
                         device=torch.device("cpu"))
    evals = iter([{"episode_return": -100.0}, {"episode_return": -50.0}])
    launcher = Launcher(EpochLoop(trainer), num_epochs=4,
                        evaluation_interval=2, eval_fn=lambda: next(evals),
                        path_to_save=str(tmp_path / "run"), verbose=False)
    launcher.run()
    ckpts = sorted((tmp_path / "run" / "checkpoints").iterdir())
    assert [c.name for c in ckpts] == ["checkpoint_000000", "checkpoint_000002",
                                       "checkpoint_000004"]
    assert launcher.best_eval_return == -50.0
    assert launcher.best_checkpoint.endswith("checkpoint_000004/checkpoint-4")
    assert len(launcher.results_log["epoch_counter"]) == 4
    assert launcher.results_log["eval_episode_return"] == [-100.0, -50.0]
