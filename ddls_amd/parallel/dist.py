"""Data-parallel utilities: RCCL (torch.distributed "nccl" backend on ROCm)
gradient all-reduce over xGMI, one process per GPU.

The reference has no torch.distributed at all (single RLlib learner GPU +
Ray CPU actors); DP scaling over the 8 MI355X GPUs of a
node is new first-class functionality here.  The policy is ~1e5 params, so the
all-reduce is latency-bound on xGMI: a single fused flat bucket per step is
the right shape (not many small per-tensor reduces).
"""
from __future__ import annotations

import os
from typing import Iterable, Optional

import torch
import torch.distributed as dist


def init_distributed_from_env(device: Optional[torch.device] = None) -> int:
    """Initialise torch.distributed from torchrun env vars; returns rank.

    Backend: "nccl" (RCCL) when CUDA/HIP devices are visible, else gloo.
    No-op (rank 0) when WORLD_SIZE is absent or 1.
    """
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1:
        return 0
    if dist.is_initialized():
        return dist.get_rank()
    backend = os.environ.get(
        "DDLS_AMD_DIST_BACKEND",
        "nccl" if torch.cuda.is_available() else "gloo")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(backend=backend)
    if torch.cuda.is_available():
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        # modulo lets a multi-rank job rehearse on fewer GPUs (gloo backend,
        # e.g. the world-size-2 single-GPU smoke test)
        torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))
    return dist.get_rank()


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


@torch.no_grad()
def all_reduce_gradients(params: Iterable[torch.nn.Parameter]):
    """Mean-all-reduce all gradients as ONE fused flat bucket (latency-bound
    collective on xGMI for small models)."""
    if not is_distributed():
        return
    grads = [p.grad for p in params if p.grad is not None]
    if not grads:
        return
    flat = torch._utils._flatten_dense_tensors(grads)
    if flat.is_cuda and dist.get_backend() == "gloo":
        # gloo CUDA support varies by build; stage through host
        host = flat.cpu()
        dist.all_reduce(host, op=dist.ReduceOp.SUM)
        flat = host.to(flat.device)
    else:
        dist.all_reduce(flat, op=dist.ReduceOp.SUM)
    flat /= get_world_size()
    for g, synced in zip(grads, torch._utils._unflatten_dense_tensors(flat, grads)):
        g.copy_(synced)


@torch.no_grad()
def all_reduce_scalar(x: float, op: str = "sum") -> float:
    if not is_distributed():
        return x
    t = torch.tensor([x], dtype=torch.float64)
    if torch.cuda.is_available():
        t = t.cuda()
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    if op == "mean":
        t /= get_world_size()
    return float(t.item())


def pick_least_used_gpu() -> int:
    """Index of the GPU with the most free memory (reference
    ``ml_models/utils.py:134-159`` scraped nvidia-smi; rocm equivalent via
    torch.cuda.mem_get_info).  Prefer torchrun LOCAL_RANK when present."""
    if "LOCAL_RANK" in os.environ:
        return int(os.environ["LOCAL_RANK"])
    if not torch.cuda.is_available():
        return 0
    best, best_free = 0, -1
    for i in range(torch.cuda.device_count()):
        free, _total = torch.cuda.mem_get_info(i)
        if free > best_free:
            best, best_free = i, free
    return best
