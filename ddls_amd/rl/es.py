"""Evolution Strategies — the reference's fifth algorithm config group
(``scripts/.../algo/es.yaml`` -> ray.rllib.agents.es.ESTrainer, OpenAI-ES:
antithetic Gaussian perturbations, centered-rank fitness shaping, l2 decay;
reference hparams noise_stdev 0.02, stepsize 0.01, l2_coeff 0.005).

Synchronous variant: perturbation pairs are evaluated sequentially on the
(shared) vectorised env as fixed-length fragments whose mean per-step reward
is the fitness (the engine makes fragment evaluation cheap); RLlib's
distributed noise table + full-episode returns are replaced accordingly —
an honest simplification of APEX-style actor fan-out, not of the ES math.
"""
from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Dict, Optional

import numpy as np
import torch


@dataclass
class ESConfig:
    # reference algo/es.yaml values
    noise_stdev: float = 0.02
    stepsize: float = 0.01
    l2_coeff: float = 0.005
    perturbation_pairs: int = 16
    fragment_steps: int = 16


def centered_ranks(x: np.ndarray) -> np.ndarray:
    """RLlib es_utils.compute_centered_ranks: ranks scaled to [-0.5, 0.5]."""
    ranks = np.empty(len(x), dtype=np.float64)
    ranks[x.argsort()] = np.arange(len(x))
    return ranks / (len(x) - 1) - 0.5


class ESTrainer:
    """Same trainer-facing interface as the gradient trainers."""

    def __init__(self, vector_env, policy, config: Optional[ESConfig] = None,
                 device: Optional[torch.device] = None):
        self.env = vector_env
        self.config = config or ESConfig()
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu")
        self.policy = policy.to(self.device)
        self.obs = self.env.reset()
        self.total_env_steps = 0
        self.iteration = 0
        self._params = [p for p in self.policy.parameters()]
        self._numel = sum(p.numel() for p in self._params)

    def _get_flat(self) -> torch.Tensor:
        return torch.cat([p.detach().reshape(-1) for p in self._params])

    @torch.no_grad()
    def _set_flat(self, flat: torch.Tensor):
        off = 0
        for p in self._params:
            n = p.numel()
            p.copy_(flat[off:off + n].view(p.shape))
            off += n

    @torch.no_grad()
    def _fitness(self, steps: int) -> float:
        data = self.env.rollout(self.policy, steps) \
            if hasattr(self.env, "rollout") else self._inline_rollout(steps)
        self.total_env_steps += int(np.prod(np.asarray(
            data["rewards"]).shape))
        return float(np.mean(data["rewards"]))

    def _inline_rollout(self, steps: int):
        from .rollout import collate
        rews = []
        for _ in range(steps):
            inputs = collate(self.obs, self.device)
            logits, _ = self.policy.forward_flat(
                inputs["batch"], inputs["graph_features"],
                inputs["action_mask"])
            actions = torch.distributions.Categorical(
                logits=logits).sample().cpu().numpy()
            self.obs, r, _dones = self.env.step(actions)
            rews.append(r)
        return {"rewards": np.stack(rews)}

    def train(self, num_steps: Optional[int] = None) -> Dict[str, float]:
        cfg = self.config
        steps = num_steps or cfg.fragment_steps
        t0 = time.perf_counter()
        theta = self._get_flat()
        rng = np.random.RandomState(self.iteration)
        eps = []
        fitness = []
        for _k in range(cfg.perturbation_pairs):
            e = torch.as_tensor(
                rng.randn(self._numel).astype(np.float32),
                device=theta.device)
            eps.append(e)
            for sign in (1.0, -1.0):
                self._set_flat(theta + sign * cfg.noise_stdev * e)
                fitness.append(self._fitness(steps))
        ranks = centered_ranks(np.asarray(fitness))
        grad = torch.zeros_like(theta)
        for k, e in enumerate(eps):
            grad += float(ranks[2 * k] - ranks[2 * k + 1]) * e
        grad /= (2 * cfg.perturbation_pairs * cfg.noise_stdev)
        theta = theta + cfg.stepsize * grad - cfg.stepsize * cfg.l2_coeff * theta
        self._set_flat(theta)
        self.iteration += 1
        stats = {
            "iteration": self.iteration,
            "fitness_mean": float(np.mean(fitness)),
            "fitness_max": float(np.max(fitness)),
            "total_loss": -float(np.mean(fitness)),   # for generic loops
            "mean_reward": float(np.mean(fitness)),
            "total_env_steps": self.total_env_steps,
            "update_time_s": time.perf_counter() - t0,
        }
        episode_stats = getattr(self.env, "drain_episode_stats", lambda: [])()
        if episode_stats:
            stats["episode_reward_mean"] = float(np.mean(
                [s["episode_return"] for s in episode_stats]))
        return stats

    def state_dict(self) -> Dict:
        return {"policy": self.policy.state_dict(),
                "iteration": self.iteration,
                "total_env_steps": self.total_env_steps, "algo": "es"}

    def load_state_dict(self, state: Dict):
        self.policy.load_state_dict(state["policy"])
        self.iteration = state.get("iteration", 0)
        self.total_env_steps = state.get("total_env_steps", 0)
