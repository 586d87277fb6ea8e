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
for i in range(4):
    st = tr.train(num_steps=16)
    cur = flat(); dp = float((cur-prev).norm()); prev = cur
    print(f"iter {i+1}: dP={dp:.4f} kl={st['kl']:.5f}", flush=True)

sp = tr._stepper
print("state: |m|", float(sp.flat_m.norm()), "|v|", float(sp.flat_v.norm()),
      "step", float(sp.step_t.item()), "|g|", float(sp.flat_g.norm()), flush=True)

# one captured replay with real data, stats inspected
obs = last_batch["obs"][:128]
acts = np.asarray(last_batch["actions"][:128], dtype=np.int64)
olp = np.asarray(last_batch["logp"][:128], dtype=np.float32)
a = np.asarray(last_batch["advantages"][:128], dtype=np.float32)
a = (a - a.mean()) / max(a.std(), 1e-4)
vt = np.asarray(last_batch["value_targets"][:128], dtype=np.float32)
sp.reset_stats()
p0 = flat()
assert sp.step(obs, acts, olp, a, vt); torch.cuda.synchronize()
print("captured replay: dP", float((flat()-p0).norm()),
      "stats", sp.stats_acc.cpu().numpy().round(4),
      "|g| after", float(sp.flat_g.norm()), flush=True)

# eager body on the same (already-filled) buffers
p0 = flat()
sp._body_fwd_bwd(); torch.cuda.synchronize()
print("eager body: |g|", float(sp.flat_g.norm()), flush=True)
sp._body_opt(); torch.cuda.synchronize()
print("eager body dP:", float((flat()-p0).norm()), flush=True)
venv.close()
