"""Small utilities (reference ``ddls/utils.py``: Stopwatch:485, Sampler:50,
seed_stochastic_modules_globally:20-47)."""
from __future__ import annotations

import copy
import random
from typing import List, Sequence

import numpy as np


class Stopwatch:
    def __init__(self):
        self.reset()

    def reset(self):
        self._time = 0.0

    def tick(self, tick=1):
        self._time += tick

    def time(self):
        return self._time


class Sampler:
    """Pool sampler with replace/remove/remove_and_repeat modes.

    Copy-on-sample: the pool holds prototype jobs and ``sample()`` hands out a
    private deep copy with its id re-based so ids stay unique across refills
    (reference ``utils.py:95-105`` re-bases at refill; copying lazily makes
    pool resets O(1) instead of O(pool)).
    """

    def __init__(self, pool: list, sampling_mode: str, shuffle: bool = False,
                 automatically_change_ids: bool = True):
        self.original_pool = list(pool)
        self.sampling_mode = sampling_mode
        self.shuffle = shuffle
        self.automatically_change_ids = automatically_change_ids
        self.reset_counter = 0
        self.reset()

    def _draw(self):
        """One RNG draw + index bookkeeping; returns (proto_idx, base_id).
        Shared by ``sample`` (clones the datum) and ``sample_ref`` (no clone:
        the vectorised env engine drains whole-episode job schedules without
        paying per-draw Job construction)."""
        idx = np.random.randint(low=0, high=len(self.sample_indices))
        proto_idx = self.sample_indices[idx]
        # capture the base BEFORE the mode handling: the pool-exhausting draw
        # of remove_and_repeat triggers reset(), which advances _base_ids —
        # the reference bakes ids into sample_pool at reset, so the popped
        # datum must keep the OLD cycle's id (utils.py:69-105)
        base = self._base_ids[proto_idx]
        if self.sampling_mode == "replace":
            pass
        elif self.sampling_mode == "remove":
            self.sample_indices.pop(idx)
        elif self.sampling_mode == "remove_and_repeat":
            self.sample_indices.pop(idx)
            if len(self.sample_indices) == 0:
                self.reset()
        else:
            raise ValueError(f"Unrecognised sampling_mode {self.sampling_mode}")
        return proto_idx, base

    def sample_ref(self):
        """RNG-identical to ``sample`` but returns the pool PROTOTYPE without
        cloning (read-only use)."""
        proto_idx, _base = self._draw()
        return self.original_pool[proto_idx]

    def sample(self):
        proto_idx, base = self._draw()
        proto = self.original_pool[proto_idx]
        # pool prototypes are pristine (never run), so the fast field-level
        # clone is equivalent to deepcopy at a fraction of the cost
        job = proto.clone() if hasattr(proto, "clone") else copy.deepcopy(proto)
        if self.automatically_change_ids:
            job.job_id = int(base + proto.job_id)
        return job

    def __len__(self):
        return len(self.sample_indices)

    def reset(self):
        n = len(self.original_pool)
        self.sample_indices = list(range(n))
        # re-base ids per refill generation (reference semantics)
        self._base_ids = [n * self.reset_counter] * n
        if self.shuffle:
            random.shuffle(self.sample_indices)
        self.reset_counter += 1


def seed_everything(seed: int = 0, seed_torch: bool = True):
    np.random.seed(seed)
    random.seed(seed)
    if seed_torch:
        try:
            import torch
            torch.manual_seed(seed)
            if torch.cuda.is_available():
                torch.cuda.manual_seed_all(seed)
        except ImportError:
            pass


def flatten_list(t: Sequence[Sequence]) -> List:
    return [item for sub in t for item in sub]
