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

    On pool reset, job ids are re-based so they stay unique across refills
    (reference ``utils.py:95-105``).
    """

    def __init__(self, pool: list, sampling_mode: str, shuffle: bool = False,
                 automatically_change_ids: bool = True):
        self.original_pool = pool
        self.sampling_mode = sampling_mode
        self.shuffle = shuffle
        self.automatically_change_ids = automatically_change_ids
        self.reset_counter = 0
        self.reset()

    def sample(self):
        idx = np.random.randint(low=0, high=len(self.sample_pool))
        datum = self.sample_pool[idx]
        if self.sampling_mode == "replace":
            # hand out a private copy so cluster state never aliases the pool
            datum = copy.deepcopy(datum)
        elif self.sampling_mode == "remove":
            self.sample_pool.pop(idx)
        elif self.sampling_mode == "remove_and_repeat":
            self.sample_pool.pop(idx)
            if len(self.sample_pool) == 0:
                self.reset()
        else:
            raise ValueError(f"Unrecognised sampling_mode {self.sampling_mode}")
        return datum

    def __len__(self):
        return len(self.sample_pool)

    def reset(self):
        # one deepcopy call so graphs shared between jobs stay shared
        # (CompGraph/GraphImmutableDetails define __deepcopy__ -> self)
        self.sample_pool = copy.deepcopy(list(self.original_pool))
        if self.automatically_change_ids:
            base_id = len(self.original_pool) * self.reset_counter
            for job in self.sample_pool:
                job.job_id = int(base_id + job.job_id)
        if self.shuffle:
            random.shuffle(self.sample_pool)
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
