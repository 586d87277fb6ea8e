"""RLlib-layout-compatible checkpointing.

Reference: ``ddls/checkpointers/checkpointer.py:3-18`` +
``rllib_epoch_loop.py:251-252`` — checkpoints live at
``<save_dir>/checkpoints/checkpoint_00000N/checkpoint-N`` (pickled state),
written at epoch 0 and every evaluation_interval epochs, restored by eval
loops via the checkpoint path.
"""
from __future__ import annotations

import os
import pickle
from typing import Optional


class Checkpointer:
    def __init__(self, path_to_save: str):
        self.checkpoint_dir = os.path.join(path_to_save, "checkpoints")
        os.makedirs(self.checkpoint_dir, exist_ok=True)

    def write(self, state: dict, index: Optional[int] = None) -> str:
        if index is None:
            index = len(os.listdir(self.checkpoint_dir)) + 1
        d = os.path.join(self.checkpoint_dir, f"checkpoint_{index:06d}")
        os.makedirs(d, exist_ok=True)
        path = os.path.join(d, f"checkpoint-{index}")
        with open(path, "wb") as f:
            pickle.dump(state, f)
        return path

    @staticmethod
    def read(checkpoint_path: str) -> dict:
        with open(checkpoint_path, "rb") as f:
            return pickle.load(f)

    def latest(self) -> Optional[str]:
        entries = sorted(os.listdir(self.checkpoint_dir))
        if not entries:
            return None
        last = entries[-1]
        index = int(last.split("_")[-1])
        return os.path.join(self.checkpoint_dir, last, f"checkpoint-{index}")
