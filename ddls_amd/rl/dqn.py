"""DQN-family trainer: dueling double-DQN with n-step targets and a replay
buffer — the reference's APEX-DQN config group
(``scripts/.../algo/apex_dqn.yaml``: gamma 0.999, lr 4.121e-7, dueling,
double_q, n_step 3, target_network_update_freq 100000, hiddens [256]).

Mapping onto this stack:
- The GNNPolicy already has the dueling decomposition: Q(s, a) =
  V(s) + A(s, a) - mean_a A(s, a), with A = policy_branch logits and
  V = value_branch (both fed by the shared GNN embedding; RLlib's dueling
  head has the same structure over ``hiddens: [256]``).
- Exploration is Boltzmann over Q (Categorical(logits=Q)) so collection
  reuses the SAME rollout machinery as PPO/IMPALA (engine or subprocess
  workers); APEX's epsilon-greedy-per-worker and its distributed
  prioritized replay shards are replaced by a uniform in-memory buffer —
  divergences from APEX, not from DQN.
- Unlike the reference (which DISABLES action masking for DQN as a
  workaround for an RLlib crash, apex_dqn.yaml "TEMP HACK"), masking works
  here: invalid actions carry -inf Q and are never selected.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional

import numpy as np
import torch
import torch.nn.functional as F

from ..parallel import all_reduce_gradients
from .impala import ImpalaTrainer
from .rollout import CompactObs


@dataclass
class DQNConfig:
    # reference algo/apex_dqn.yaml tuned values
    gamma: float = 0.999
    lr: float = 4.121e-7
    n_step: int = 3
    double_q: bool = True
    dueling: bool = True
    target_network_update_freq: int = 100000   # env steps between syncs
    v_min: float = -1000.0
    v_max: float = 1000.0
    train_batch_size: int = 4096               # transitions per update
    sgd_minibatch_size: int = 256
    num_sgd_iter: int = 4
    replay_capacity: int = 200_000
    learning_starts: int = 2_000
    grad_clip: Optional[float] = 40.0


class DQNTrainer(ImpalaTrainer):
    """Same trainer-facing interface as PPOTrainer/ImpalaTrainer."""

    def __init__(self, vector_env, policy, config: Optional[DQNConfig] = None,
                 device: Optional[torch.device] = None):
        cfg = config or DQNConfig()
        super().__init__(vector_env, policy, None, device=device)
        self.config = cfg
        self.optimizer = torch.optim.Adam(self.policy.parameters(),
                                          lr=cfg.lr, foreach=True)
        import copy
        self.target = copy.deepcopy(self.policy).to(self.device)
        self.target.eval()
        self._steps_since_sync = 0
        # list-based ring buffer (deque indexing is O(n))
        self.replay: List = []
        self._replay_ptr = 0

    # Q(s, .) with the dueling recombination over masked actions
    def _q_values(self, obs: List[CompactObs], net):
        logits, values = self._forward_batch_with(obs, net)
        if self.config.dueling:
            # masked actions carry clamp(log(0)) = f32-min in `logits`
            # (forward_flat); exclude them from the advantage mean and keep
            # them at f32-min in Q so argmax/gather never select them
            valid = logits > -1e30
            adv = torch.where(valid, logits, torch.zeros_like(logits))
            mean_adv = adv.sum(-1) / valid.sum(-1).clamp(min=1)
            q = values.unsqueeze(-1) + logits - mean_adv.unsqueeze(-1)
            q = torch.where(valid, q, logits)
            return q
        return logits

    def _forward_batch_with(self, obs, net):
        policy = self.policy
        try:
            self.policy = net
            return self._forward_batch(obs)
        finally:
            self.policy = policy

    # ------------------------------------------------------------------
    def _ingest(self, data: Dict):
        """Turn a [T, N] rollout fragment into n-step transitions."""
        cfg = self.config
        T, N = data["rewards"].shape
        obs = data["obs"]               # t-major flat list [T*N]
        rewards = data["rewards"]
        dones = data["dones"]
        n = cfg.n_step
        for t in range(T - n):
            for b in range(N):
                R, discount, cut = 0.0, 1.0, False
                for i in range(n):
                    R += discount * rewards[t + i][b]
                    discount *= cfg.gamma
                    if dones[t + i][b]:
                        cut = True
                        break
                item = (obs[t * N + b], int(data["actions"][t][b]),
                        float(R),
                        obs[(t + i + 1) * N + b] if not cut else None,
                        float(discount))
                if len(self.replay) < cfg.replay_capacity:
                    self.replay.append(item)
                else:
                    self.replay[self._replay_ptr] = item
                    self._replay_ptr = (self._replay_ptr + 1) \
                        % cfg.replay_capacity

    def update(self, data: Dict) -> Dict[str, float]:
        cfg = self.config
        self._ingest(data)
        if len(self.replay) < cfg.learning_starts:
            return {"total_loss": 0.0, "q_mean": 0.0, "replay": len(self.replay)}
        rng = np.random.RandomState(self.iteration)
        stats = {"total_loss": 0.0, "q_mean": 0.0}
        n_updates = 0
        for _ in range(cfg.num_sgd_iter):
            idx = rng.randint(0, len(self.replay), size=cfg.sgd_minibatch_size)
            batch = [self.replay[i] for i in idx]
            s = [b[0] for b in batch]
            a = torch.as_tensor([b[1] for b in batch], device=self.device)
            r = torch.as_tensor([b[2] for b in batch], device=self.device,
                                dtype=torch.float32)
            disc = torch.as_tensor([b[4] for b in batch], device=self.device,
                                   dtype=torch.float32)
            has_next = [b[3] is not None for b in batch]
            sp = [b[3] for b in batch if b[3] is not None]

            with torch.no_grad():
                boot = torch.zeros(len(batch), device=self.device)
                if sp:
                    q_next_t = self._q_values(sp, self.target)
                    if cfg.double_q:
                        q_next_o = self._q_values(sp, self.policy)
                        a_star = q_next_o.argmax(-1)
                        vals = q_next_t.gather(
                            1, a_star.unsqueeze(1)).squeeze(1)
                    else:
                        vals = q_next_t.max(-1).values
                    boot[torch.as_tensor(has_next,
                                         device=self.device)] = vals
                target = torch.clamp(r + disc * boot, cfg.v_min, cfg.v_max)
            q = self._q_values(s, self.policy)
            q_a = q.gather(1, a.unsqueeze(1)).squeeze(1)
            loss = F.smooth_l1_loss(q_a, target)
            self.optimizer.zero_grad(set_to_none=True)
            loss.backward()
            all_reduce_gradients(self.policy.parameters())
            if cfg.grad_clip is not None:
                torch.nn.utils.clip_grad_norm_(self.policy.parameters(),
                                               cfg.grad_clip)
            self.optimizer.step()
            stats["total_loss"] += float(loss.item())
            stats["q_mean"] += float(q_a.mean().item())
            n_updates += 1
        for k in ("total_loss", "q_mean"):
            stats[k] /= max(n_updates, 1)
        stats["replay"] = len(self.replay)
        # target sync on an env-step cadence (reference: 100k tuned)
        self._steps_since_sync += int(np.prod(data["rewards"].shape))
        if self._steps_since_sync >= cfg.target_network_update_freq:
            self.target.load_state_dict(self.policy.state_dict())
            self._steps_since_sync = 0
        return stats

    # ------------------------------------------------------------------
    def state_dict(self) -> Dict:
        out = super().state_dict()
        out["target"] = self.target.state_dict()
        out["algo"] = "dqn"
        return out

    def load_state_dict(self, state: Dict):
        super().load_state_dict(state)
        if "target" in state:
            self.target.load_state_dict(state["target"])
