"""MFMA bwd at training scale: eager parity + captured steps."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from ddls_amd import ops as hip_ops
from ddls_amd.models.gnn import GNNPolicy
from ddls_amd.rl.graph_step import CapturedSGDStep
from ddls_amd.rl.ppo import PPOConfig
from ddls_amd.rl.rollout import CompactObs

ext = hip_ops.get_extension(required=True)
dev = torch.device("cuda:0")
torch.manual_seed(6)
for (N, E, OUT) in ():
    hn = torch.rand(N, 16, device=dev)
    he = torch.rand(E, 16, device=dev)
    src = torch.randint(0, N, (E,), device=dev)
    dst = torch.randint(0, N, (E,), device=dev)
    counts = torch.bincount(dst, minlength=N)
    indptr = torch.zeros(N + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, 0, out=indptr[1:])
    ln_g = torch.rand(32, device=dev) + 0.5
    ln_b = torch.rand(32, device=dev) - 0.5
    Wr = torch.randn(OUT, 32, device=dev) / 4
    br = torch.randn(OUT, device=dev)
    order = torch.argsort(dst, stable=True)
    _o, r_e, r_s = ext.message_reduce_train(hn, he, src, order, indptr,
                                            ln_g, ln_b, Wr, br)
    gout = torch.randn(N, OUT, device=dev)
    ref = ext.message_reduce_bwd(hn, he, src, dst, indptr, ln_g, ln_b, Wr,
                                 r_e, r_s, gout)
    got = ext.message_reduce_bwd_mfma(hn, he, src, dst, indptr, ln_g, ln_b,
                                      Wr, r_e, r_s, gout)
    for nm, a, b in zip(("ghn","ghe","gWr","gbr","glng","glnb"), ref, got):
        md = (a - b).abs().max().item()
        ok = "OK " if md < 1e-2 * max(1, a.abs().max().item()) else "BAD"
        print(f"N={N} E={E} OUT={OUT} {nm}: maxdiff={md:.5f} "
              f"|ref|max={a.abs().max().item():.4f} {ok}", flush=True)

# captured steps at training scale
torch.manual_seed(3)
pol = GNNPolicy(num_actions=17).to(dev)
opt = torch.optim.Adam(pol.parameters(), lr=2.785e-4, foreach=True)
cfg = PPOConfig(sgd_minibatch_size=128)
rng = np.random.RandomState(0)
def obs():
    n = int(rng.randint(40, 90)); m = int(rng.randint(60, 160))
    return CompactObs(
        node_features=rng.rand(n, 5).astype(np.float32),
        edge_features=rng.rand(m, 2).astype(np.float32),
        edges_src=rng.randint(0, n, m).astype(np.int64),
        edges_dst=rng.randint(0, n, m).astype(np.int64),
        graph_features=rng.rand(34).astype(np.float32),
        action_mask=np.ones(17, dtype=np.float32))
mb = [obs() for _ in range(128)]
acts = np.zeros(128, dtype=np.int64)
olp = (rng.randn(128)*0.1 - 2).astype(np.float32)
adv = rng.randn(128).astype(np.float32)
vt = rng.randn(128).astype(np.float32)
st = CapturedSGDStep(pol, opt, cfg, dev)
st.set_kl_coeff(cfg.kl_coeff)
def flat(): return torch.cat([p.detach().reshape(-1).clone() for p in pol.parameters()])
prev = flat()
for i in range(3):
    assert st.step(mb, acts, olp, adv, vt); torch.cuda.synchronize()
    cur = flat(); print(f"bigcap1 step {i}: dP={float((cur-prev).norm()):.5f}"); prev = cur
assert st.ensure_capacity(128*120, 128*200)
print("recaptured caps =", st.capture_count)
for i in range(3):
    assert st.step(mb, acts, olp, adv, vt); torch.cuda.synchronize()
    cur = flat(); print(f"bigcap2 step {i}: dP={float((cur-prev).norm()):.5f}"); prev = cur
