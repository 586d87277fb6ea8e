"""Synchronous IMPALA (V-trace) for the PAC-ML GNN policy.

Second training algorithm sharing the PPO stack's rollout/learner machinery
(reference trains the env under RLlib IMPALA too:
``scripts/ramp_job_partitioning_configs/algo/impala.yaml`` —
vtrace rho/c clips 1.0, vtrace_drop_last_ts, num_sgd_iter 1, adam,
grad_clip 40, vf_loss_coeff 0.5, entropy_coeff 0.01).

Synchronous variant: rollouts are collected with the current weights (no
actor lag), then ONE SGD pass over the train batch computes V-trace targets
from the post-collection forward — the importance ratios rho correct any
within-batch staleness exactly as RLlib's learner does.  Distributed: same
one-process-per-GPU fused gradient all-reduce as PPO.

V-trace (Espeholt et al. 2018):
  delta_t = rho_t (r_t + gamma V_{t+1} - V_t)
  vs_t    = V_t + delta_t + gamma c_t (vs_{t+1} - V_{t+1})
  pg_adv  = pg_rho_t (r_t + gamma vs_{t+1} - V_t)
with rho_t = min(rho_clip, pi/mu), c_t = min(c_clip, pi/mu).
"""
from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Dict, List, Optional

import numpy as np
import torch

from ..parallel import all_reduce_gradients
from .rollout import CompactObs, collate


@dataclass
class ImpalaConfig:
    # reference algo/impala.yaml values
    lr: float = 5e-4
    gamma: float = 0.997
    vf_loss_coeff: float = 0.5
    entropy_coeff: float = 0.01
    grad_clip: float = 40.0
    rho_clip: float = 1.0
    pg_rho_clip: float = 1.0
    c_clip: float = 1.0
    train_batch_size: int = 4000
    vtrace_drop_last_ts: bool = True


def vtrace_targets(rewards, dones, values, bootstrap, rho, c, gamma):
    """All inputs [T, N] torch tensors (rho/c already clipped); returns
    (vs [T, N], pg_adv [T, N])."""
    T = rewards.shape[0]
    nonterminal = 1.0 - dones
    values_tp1 = torch.cat([values[1:], bootstrap.unsqueeze(0)], dim=0)
    values_tp1 = values_tp1 * nonterminal        # V(x_{t+1}) = 0 past a reset
    delta = rho * (rewards + gamma * values_tp1 - values)
    vs_minus_v = torch.zeros_like(values)
    acc = torch.zeros_like(values[0])
    for t in reversed(range(T)):
        acc = delta[t] + gamma * c[t] * nonterminal[t] * acc
        vs_minus_v[t] = acc
    vs = values + vs_minus_v
    vs_tp1 = torch.cat([vs[1:], bootstrap.unsqueeze(0)], dim=0) * nonterminal
    pg_adv = rewards + gamma * vs_tp1 - values
    return vs, pg_adv


class ImpalaTrainer:
    """Same trainer-facing interface as PPOTrainer (train/state_dict/...)."""

    def __init__(self, vector_env, policy, config: Optional[ImpalaConfig] = None,
                 device: Optional[torch.device] = None):
        self.env = vector_env
        self.config = config or ImpalaConfig()
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu")
        self.policy = policy.to(self.device)
        self.optimizer = torch.optim.Adam(self.policy.parameters(),
                                          lr=self.config.lr, foreach=True)
        self.obs = self.env.reset()
        self.total_env_steps = 0
        self.iteration = 0

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _policy_step(self, obs_list: List[CompactObs]):
        inputs = collate(obs_list, self.device)
        logits, values = self.policy.forward_flat(
            inputs["batch"], inputs["graph_features"], inputs["action_mask"])
        dist = torch.distributions.Categorical(logits=logits)
        actions = dist.sample()
        return (actions.cpu().numpy(), dist.log_prob(actions).cpu().numpy(),
                values.cpu().numpy())

    def collect_rollout(self, num_steps: Optional[int] = None) -> Dict:
        n_envs = len(self.env)
        steps = num_steps if num_steps is not None else max(
            1, self.config.train_batch_size // n_envs)
        if hasattr(self.env, "rollout"):
            data = self.env.rollout(self.policy, steps)
            self.total_env_steps += steps * n_envs
            return data
        obs_buf, act_buf, logp_buf, val_buf, rew_buf, done_buf = \
            [], [], [], [], [], []
        for _ in range(steps):
            actions, logp, values = self._policy_step(self.obs)
            obs_buf.append(list(self.obs))
            act_buf.append(actions)
            logp_buf.append(logp)
            val_buf.append(values)
            self.obs, rewards, dones = self.env.step(actions)
            rew_buf.append(rewards)
            done_buf.append(dones)
            self.total_env_steps += n_envs
        _, _, bootstrap = self._policy_step(self.obs)
        return {
            "obs": [o for row in obs_buf for o in row],
            "actions": np.stack(act_buf),
            "logp": np.stack(logp_buf),
            "values": np.stack(val_buf),
            "rewards": np.stack(rew_buf),
            "dones": np.stack(done_buf),
            "bootstrap_values": bootstrap,
        }

    # ------------------------------------------------------------------
    def _forward_batch(self, obs: List[CompactObs]):
        """Differentiable forward over the flat obs list; uses the
        cached-models path when the env serves per-model static obs."""
        mb_static = getattr(self.env, "_models_batch", None)
        if mb_static is not None and getattr(obs[0], "model_id", -1) >= 0:
            from ..models.gnn import graph_mean
            node_emb = self.policy.gnn(mb_static)
            model_emb = graph_mean(node_emb, mb_static)
            mids = torch.as_tensor([o.model_id for o in obs],
                                   device=self.device)
            gfull = torch.as_tensor(
                np.stack([o.graph_features for o in obs]), device=self.device)
            mask = torch.as_tensor(np.stack([o.action_mask for o in obs]),
                                   device=self.device)
            graph_emb = self.policy.graph_module(gfull)
            final = torch.cat([model_emb[mids], graph_emb], dim=-1)
            logits = self.policy.policy_branch(final)
            values = self.policy.value_branch(final).squeeze(-1)
            if self.policy.config["apply_action_mask"]:
                logits = logits + torch.clamp(
                    torch.log(mask), min=torch.finfo(torch.float32).min)
            return logits, values
        inputs = collate(obs, self.device)
        return self.policy.forward_flat(
            inputs["batch"], inputs["graph_features"], inputs["action_mask"])

    def update(self, data: Dict) -> Dict[str, float]:
        cfg = self.config
        dev = self.device
        T, N = data["rewards"].shape
        t64 = lambda a: torch.as_tensor(np.asarray(a, dtype=np.float32),
                                        device=dev)
        rewards = t64(data["rewards"])
        dones = t64(data["dones"].astype(np.float32))
        behavior_logp = t64(data["logp"])
        actions = torch.as_tensor(np.asarray(data["actions"]).reshape(-1),
                                  device=dev)

        logits, values = self._forward_batch(data["obs"])   # [T*N, A], [T*N]
        dist = torch.distributions.Categorical(logits=logits)
        new_logp = dist.log_prob(actions).reshape(T, N)
        values_tn = values.reshape(T, N)
        with torch.no_grad():
            log_rho = new_logp - behavior_logp
            rho = torch.clamp(torch.exp(log_rho), max=cfg.rho_clip)
            pg_rho = torch.clamp(torch.exp(log_rho), max=cfg.pg_rho_clip)
            c = torch.clamp(torch.exp(log_rho), max=cfg.c_clip)
            bootstrap = t64(data["bootstrap_values"])
            vs, pg_adv_raw = vtrace_targets(rewards, dones,
                                            values_tn.detach(), bootstrap,
                                            rho, c, cfg.gamma)
            pg_adv = pg_rho * pg_adv_raw

        keep = slice(0, T - 1) if (cfg.vtrace_drop_last_ts and T > 1) \
            else slice(0, T)
        policy_loss = -(pg_adv[keep] * new_logp[keep]).mean()
        vf_loss = 0.5 * ((vs[keep].detach() - values_tn[keep]) ** 2).mean()
        entropy = dist.entropy().reshape(T, N)[keep].mean()
        loss = (policy_loss + cfg.vf_loss_coeff * vf_loss
                - cfg.entropy_coeff * entropy)

        self.optimizer.zero_grad(set_to_none=True)
        loss.backward()
        all_reduce_gradients(self.policy.parameters())
        if cfg.grad_clip is not None:
            torch.nn.utils.clip_grad_norm_(self.policy.parameters(),
                                           cfg.grad_clip)
        self.optimizer.step()
        return {
            "policy_loss": float(policy_loss.item()),
            "vf_loss": float(vf_loss.item()),
            "entropy": float(entropy.item()),
            "total_loss": float(loss.item()),
            "mean_rho": float(rho.mean().item()),
        }

    # ------------------------------------------------------------------
    def train(self, num_steps: Optional[int] = None) -> Dict[str, float]:
        t0 = time.perf_counter()
        data = self.collect_rollout(num_steps)
        t1 = time.perf_counter()
        stats = self.update(data)
        t2 = time.perf_counter()
        self.iteration += 1
        stats.update({
            "iteration": self.iteration,
            "env_steps_this_iter": int(np.prod(data["rewards"].shape)),
            "total_env_steps": self.total_env_steps,
            "rollout_time_s": t1 - t0,
            "update_time_s": t2 - t1,
            "mean_reward": float(np.mean(data["rewards"])),
        })
        episode_stats = self.env.drain_episode_stats()
        if episode_stats:
            stats["episode_reward_mean"] = float(np.mean(
                [s["episode_return"] for s in episode_stats]))
            stats["blocking_rate_mean"] = float(np.mean(
                [s.get("blocking_rate", 0) for s in episode_stats]))
            jcts = [np.mean(s["job_completion_time"]) for s in episode_stats
                    if len(s.get("job_completion_time", [])) > 0]
            if jcts:
                stats["mean_job_completion_time"] = float(np.mean(jcts))
        return stats

    # ------------------------------------------------------------------
    def state_dict(self) -> Dict:
        return {"policy": self.policy.state_dict(),
                "optimizer": self.optimizer.state_dict(),
                "iteration": self.iteration,
                "total_env_steps": self.total_env_steps,
                "algo": "impala"}

    def load_state_dict(self, state: Dict):
        self.policy.load_state_dict(state["policy"])
        self.optimizer.load_state_dict(state["optimizer"])
        self.iteration = state.get("iteration", 0)
        self.total_env_steps = state.get("total_env_steps", 0)
