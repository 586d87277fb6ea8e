"""Bisect GPU-only training-path features via their disable flags."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from bench import build_env_fn
from ddls_amd.cluster.batched_lookahead import precompute_lookahead_memos
from ddls_amd.models.gnn import GNNPolicy
from ddls_amd.rl.ppo import PPOConfig, PPOTrainer
from ddls_amd.rl.subproc_env import SubprocVectorEnv

env_fn = build_env_fn()
venv = SubprocVectorEnv(env_fn, num_envs=64, num_workers=64, base_seed=1)
scratch = env_fn(); scratch.reset(seed=0)
ml, mi = precompute_lookahead_memos(scratch, device="cuda:0")
venv.preload_memos(ml, mi)

FLAGS = ("DDLS_AMD_DISABLE_FLAT_ADAM", "DDLS_AMD_DISABLE_FUSED_LOSS",
         "DDLS_AMD_DISABLE_MFMA", "DDLS_AMD_DISABLE_MFMA_BWD")

def run(tag, on, use_graphs=True, iters=20, sgd_iters=8, noprefetch=False):
    for k in FLAGS:
        os.environ[k] = "1" if k in on else "0"
    os.environ["DDLS_AMD_DISABLE_PREFETCH"] = "1" if noprefetch else "0"
    torch.manual_seed(0)
    policy = GNNPolicy(num_actions=17)
    tr = PPOTrainer(venv, policy,
                    PPOConfig(train_batch_size=1024, sgd_minibatch_size=128,
                              num_sgd_iter=sgd_iters, use_hip_graphs=use_graphs),
                    device=torch.device("cuda:0"))
    ents, dps, kls = [], [], []
    st = {}
    prev = torch.cat([p.detach().reshape(-1).clone()
                      for p in policy.parameters()])
    for i in range(iters):
        st = tr.train(num_steps=16)
        cur = torch.cat([p.detach().reshape(-1).clone()
                         for p in policy.parameters()])
        dps.append(round(float((cur - prev).norm()), 4))
        prev = cur
        ents.append(round(st["entropy"], 3))
        kls.append(round(st["kl"], 5))
        sp = tr._stepper
        grad_linked = None
        if sp is not None and sp.flat_p is not None:
            base = sp.flat_g.data_ptr()
            end = base + sp.flat_g.numel() * 4
            grad_linked = all(
                p.grad is not None and base <= p.grad.data_ptr() < end
                for p in policy.parameters())
        print(f"  iter {i+1}: dP={dps[-1]} caps="
              f"{getattr(sp, 'capture_count', None)} "
              f"grad_linked={grad_linked}", flush=True)
    print(f"{tag:18s} ent={ents[::4]} dP={dps[::4]} kl={kls[::4]} "
          f"reward={st['mean_reward']:.1f}", flush=True)

run("all-on", (), iters=12)
run("no-mfma-bwd", ("DDLS_AMD_DISABLE_MFMA_BWD",), iters=12)
venv.close()
