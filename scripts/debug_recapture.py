"""Does the SECOND capture produce zero grads? (mfma-bwd + flat adam)"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from ddls_amd.models.gnn import GNNPolicy
from ddls_amd.rl.graph_step import CapturedSGDStep
from ddls_amd.rl.ppo import PPOConfig
from ddls_amd.rl.rollout import CompactObs

device = torch.device("cuda:0")
torch.manual_seed(3)
pol = GNNPolicy(num_actions=17).to(device)
opt = torch.optim.Adam(pol.parameters(), lr=2.785e-4, foreach=True)
cfg = PPOConfig(sgd_minibatch_size=8)
rng = np.random.RandomState(0)

def obs(n=None):
    n = n or int(rng.randint(6, 14))
    m = int(rng.randint(4, 2 * n))
    return CompactObs(
        node_features=rng.rand(n, 5).astype(np.float32),
        edge_features=rng.rand(m, 2).astype(np.float32),
        edges_src=rng.randint(0, n, m).astype(np.int64),
        edges_dst=rng.randint(0, n, m).astype(np.int64),
        graph_features=rng.rand(34).astype(np.float32),
        action_mask=np.ones(17, dtype=np.float32))

mb = [obs() for _ in range(8)]
acts = np.zeros(8, dtype=np.int64)
olp = (rng.randn(8) * 0.1 - 2).astype(np.float32)
adv = rng.randn(8).astype(np.float32)
vt = rng.randn(8).astype(np.float32)

st = CapturedSGDStep(pol, opt, cfg, device)
st.set_kl_coeff(cfg.kl_coeff)

def flat():
    return torch.cat([p.detach().reshape(-1).clone() for p in pol.parameters()])

prev = flat()
for i in range(3):
    assert st.step(mb, acts, olp, adv, vt)
    torch.cuda.synchronize()
    cur = flat(); print(f"cap1 step {i}: dP={float((cur-prev).norm()):.5f} "
                        f"stats={st.stats_acc.cpu().numpy().round(3)}")
    st.reset_stats(); prev = cur

# force re-capture with bigger capacity
big = [obs(30) for _ in range(8)]
assert st.ensure_capacity(8 * 40, 8 * 80)
print("recaptured: caps =", st.capture_count)
for i in range(3):
    assert st.step(mb, acts, olp, adv, vt)
    torch.cuda.synchronize()
    cur = flat(); print(f"cap2 step {i}: dP={float((cur-prev).norm()):.5f} "
                        f"stats={st.stats_acc.cpu().numpy().round(3)}")
    st.reset_stats(); prev = cur
