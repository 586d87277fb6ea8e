"""Probe inside frozen training: is the captured graph broken or the data?"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from bench import build_env_fn
from ddls_amd.cluster.batched_lookahead import precompute_lookahead_memos
from ddls_amd.models.gnn import GNNPolicy
from ddls_amd.rl.ppo import PPOConfig, PPOTrainer
from ddls_amd.rl.rollout import CompactObs, collate
from ddls_amd.rl.subproc_env import SubprocVectorEnv

env_fn = build_env_fn()
venv = SubprocVectorEnv(env_fn, num_envs=64, num_workers=64, base_seed=1)
scratch = env_fn(); scratch.reset(seed=0)
ml, mi = precompute_lookahead_memos(scratch, device="cuda:0")
venv.preload_memos(ml, mi)
dev = torch.device("cuda:0")
torch.manual_seed(0)
policy = GNNPolicy(num_actions=17)
tr = PPOTrainer(venv, policy,
                PPOConfig(train_batch_size=1024, sgd_minibatch_size=128,
                          num_sgd_iter=8), device=dev)
rng = np.random.RandomState(7)
def synth():
    n = int(rng.randint(20, 60)); m = int(rng.randint(30, 120))
    return CompactObs(rng.rand(n,5).astype(np.float32), rng.rand(m,2).astype(np.float32),
                      rng.randint(0,n,m).astype(np.int64), rng.randint(0,n,m).astype(np.int64),
                      rng.rand(34).astype(np.float32), np.ones(17, dtype=np.float32))
smb = [synth() for _ in range(128)]
sacts = np.zeros(128, dtype=np.int64)
solp = (rng.randn(128)*0.1-2).astype(np.float32)
sadv = rng.randn(128).astype(np.float32)
svt = rng.randn(128).astype(np.float32)

def flat():
    return torch.cat([p.detach().reshape(-1).clone() for p in policy.parameters()])

last_batch = {}
orig_collect = tr.collect_rollout
def collect(*a, **k):
    b = orig_collect(*a, **k)
    last_batch.update(b)
    return b
tr.collect_rollout = collect

prev = flat()
for i in range(5):
    st = tr.train(num_steps=16)
    cur = flat(); dp = float((cur-prev).norm()); prev = cur
    sp = tr._stepper
    # probe A: captured step with synthetic data
    p0 = flat()
    ok = sp.step(smb, sacts, solp, sadv, svt); torch.cuda.synchronize()
    dpa = float((flat()-p0).norm())
    # probe B: EAGER fwd/bwd on a real minibatch from the last rollout
    obs = last_batch["obs"][:128]
    inputs = collate(obs, dev)
    for p in policy.parameters():
        if p.grad is not None:
            with torch.no_grad(): p.grad.zero_()
    logits, values = policy.forward_flat(inputs["batch"], inputs["graph_features"], inputs["action_mask"])
    loss = logits.sum() * 1e-3 + values.sum() * 1e-3
    loss.backward()
    gnorm = float(torch.cat([p.grad.reshape(-1) for p in policy.parameters()]).norm())
    # restore params (probe A moved them)
    prev = flat()
    print(f"iter {i+1}: dP={dp:.4f} caps={sp.capture_count} "
          f"probeA_captured_synth_dP={dpa:.5f} probeB_eager_real_gnorm={gnorm:.5f} "
          f"kl={st['kl']:.5f} ent={st['entropy']:.3f}", flush=True)
venv.close()
