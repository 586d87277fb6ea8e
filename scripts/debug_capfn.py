"""Capture message_reduce_bwd_mfma alone; check replay output stability."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from ddls_amd import ops as hip_ops
ext = hip_ops.get_extension(required=True)
dev = torch.device("cuda:0")
torch.manual_seed(6)
N, E, OUT = 7400, 14000, 64
hn = torch.rand(N, 16, device=dev); he = torch.rand(E, 16, device=dev)
src = torch.randint(0, N, (E,), device=dev)
dst = torch.randint(0, N, (E,), device=dev)
counts = torch.bincount(dst, minlength=N)
indptr = torch.zeros(N + 1, dtype=torch.int64, device=dev)
torch.cumsum(counts, 0, out=indptr[1:])
ln_g = torch.rand(32, device=dev) + 0.5; ln_b = torch.rand(32, device=dev) - 0.5
Wr = torch.randn(OUT, 32, device=dev) / 4; br = torch.randn(OUT, device=dev)
order = torch.argsort(dst, stable=True)
_o, r_e, r_s = ext.message_reduce_train(hn, he, src, order, indptr, ln_g, ln_b, Wr, br)
gout = torch.randn(N, OUT, device=dev)

args = (hn, he, src, dst, indptr, ln_g, ln_b, Wr, r_e, r_s, gout)
ref = [t.clone() for t in ext.message_reduce_bwd_mfma(*args)]

s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    for _ in range(3):
        outs = ext.message_reduce_bwd_mfma(*args)
torch.cuda.current_stream().wait_stream(s); torch.cuda.synchronize()
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    outs = ext.message_reduce_bwd_mfma(*args)
def check(tag):
    torch.cuda.synchronize()
    for nm, a, b in zip(("ghn","ghe","gWr","gbr","glng","glnb"), ref, outs):
        md = (a - b).abs().max().item()
        print(f"{tag} {nm}: maxdiff={md:.6f} |out|={b.abs().max().item():.4f}", flush=True)
g.replay(); check("replay1")
for _ in range(200): g.replay()
check("replay201")
