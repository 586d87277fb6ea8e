"""Vanilla policy gradient (REINFORCE) — the reference's third algorithm
config group (``scripts/.../algo/pg.yaml`` -> ray.rllib.agents.pg.PGTrainer,
which uses Monte-Carlo discounted returns with no critic and no GAE:
compute_advantages(use_gae=False, use_critic=False), zero bootstrap at
fragment truncation).  Shares the rollout machinery with PPO/IMPALA."""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

import numpy as np
import torch

from ..parallel import all_reduce_gradients
from .impala import ImpalaTrainer


@dataclass
class PGConfig:
    # RLlib PGConfig defaults
    lr: float = 4e-4
    gamma: float = 0.99
    train_batch_size: int = 200
    grad_clip: Optional[float] = None


class PGTrainer(ImpalaTrainer):
    """REINFORCE: loss = -mean(logp(a_t) * G_t).  Reuses ImpalaTrainer's
    rollout collection and differentiable batch forward."""

    def __init__(self, vector_env, policy, config: Optional[PGConfig] = None,
                 device: Optional[torch.device] = None):
        cfg = config or PGConfig()
        super().__init__(vector_env, policy, None, device=device)
        self.config = cfg
        self.optimizer = torch.optim.Adam(self.policy.parameters(),
                                          lr=cfg.lr, foreach=True)

    def update(self, data: Dict) -> Dict[str, float]:
        cfg = self.config
        dev = self.device
        T, N = data["rewards"].shape
        rewards = np.asarray(data["rewards"], dtype=np.float64)
        dones = np.asarray(data["dones"], dtype=np.float64)
        # Monte-Carlo discounted returns, zero bootstrap at truncation
        returns = np.zeros_like(rewards)
        acc = np.zeros(N)
        for t in reversed(range(T)):
            acc = rewards[t] + cfg.gamma * (1.0 - dones[t]) * acc
            returns[t] = acc
        actions = torch.as_tensor(np.asarray(data["actions"]).reshape(-1),
                                  device=dev)
        ret_t = torch.as_tensor(returns.reshape(-1).astype(np.float32),
                                device=dev)
        logits, _values = self._forward_batch(data["obs"])
        dist = torch.distributions.Categorical(logits=logits)
        logp = dist.log_prob(actions)
        loss = -(logp * ret_t).mean()
        self.optimizer.zero_grad(set_to_none=True)
        loss.backward()
        all_reduce_gradients(self.policy.parameters())
        if cfg.grad_clip is not None:
            torch.nn.utils.clip_grad_norm_(self.policy.parameters(),
                                           cfg.grad_clip)
        self.optimizer.step()
        return {"policy_loss": float(loss.item()),
                "total_loss": float(loss.item()),
                "entropy": float(dist.entropy().mean().item()),
                "mean_return": float(ret_t.mean().item())}

    def state_dict(self) -> Dict:
        out = super().state_dict()
        out["algo"] = "pg"
        return out
