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

    def _indices(self):
        """Existing checkpoint indices, ignoring stray files."""
        out = []
        for e in os.listdir(self.checkpoint_dir):
            if e.startswith("checkpoint_"):
                try:
                    out.append(int(e.split("_")[-1]))
                except ValueError:
                    pass
        return sorted(out)

    def write(self, state: dict, index: Optional[int] = None) -> str:
        if index is None:
            existing = self._indices()
            index = (existing[-1] + 1) if existing else 1
        d = os.path.join(self.checkpoint_dir, f"checkpoint_{index:06d}")
        os.makedirs(d, exist_ok=True)
        path = os.path.join(d, f"checkpoint-{index}")
        with open(path, "wb") as f:
            pickle.dump(state, f)
        return path

    @staticmethod
    def read(checkpoint_path: str) -> dict:
        """Load a checkpoint; GPU-written tensors map to CPU automatically
        when no GPU is visible (RLlib restore is likewise device-portable)."""
        import torch
        with open(checkpoint_path, "rb") as f:
            if torch.cuda.is_available():
                return pickle.load(f)
            return _CpuUnpickler(f).load()

    def latest(self) -> Optional[str]:
        existing = self._indices()
        if not existing:
            return None
        index = existing[-1]
        return os.path.join(self.checkpoint_dir, f"checkpoint_{index:06d}",
                            f"checkpoint-{index}")


class _CpuUnpickler(pickle.Unpickler):
    """Unpickler that remaps torch storages to CPU (plain-pickle checkpoints
    carry the saving device inside ``torch.storage._load_from_bytes``)."""

    def find_class(self, module, name):
        if module == "torch.storage" and name == "_load_from_bytes":
            import io

            import torch
            return lambda b: torch.load(io.BytesIO(b), map_location="cpu",
                                        weights_only=False)
        return super().find_class(module, name)
