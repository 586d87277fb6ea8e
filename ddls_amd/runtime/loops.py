"""Training/eval loops + launcher.

Reference: ``ddls/loops/rllib_epoch_loop.py:34`` (epoch = one trainer.train()),
``rllib_eval_loop.py:11`` (checkpoint rollout eval), ``eval_loop.py:14``
(heuristic actor eval), ``ddls/launchers/launcher.py:17`` (run-until loop with
checkpoint/log cadence and best-checkpoint tracking).
"""
from __future__ import annotations

import time
from collections import defaultdict
from typing import Callable, Dict, List, Optional

import numpy as np

from .checkpointer import Checkpointer
from .logger import Logger


class EpochLoop:
    """One epoch = one PPO iteration (rollout + SGD update)."""

    def __init__(self, trainer):
        self.trainer = trainer
        self.best_eval_reward = -float("inf")
        self.best_checkpoint_paths: List[str] = []

    def run(self) -> Dict:
        t0 = time.perf_counter()
        stats = self.trainer.train()
        stats["epoch_time_s"] = time.perf_counter() - t0
        return stats

    def state_dict(self):
        return self.trainer.state_dict()


class EvalLoop:
    """Roll a heuristic actor (or a policy actor) through full episodes and
    harvest the cluster's episode stats (reference ``eval_loop.py:26-129``)."""

    def __init__(self, actor, env, max_steps: int = 100000):
        self.actor = actor
        self.env = env
        self.max_steps = max_steps

    def run(self, seed: Optional[int] = None, verbose: bool = False) -> Dict:
        obs = self.env.reset(seed=seed)
        done, steps = False, 0
        rewards = []
        t0 = time.perf_counter()
        while not done and steps < self.max_steps:
            job_to_place = next(iter(self.env.cluster.job_queue.jobs.values()))
            action = self.actor.compute_action(obs, job_to_place=job_to_place)
            obs, reward, done, info = self.env.step(int(action))
            rewards.append(reward)
            steps += 1
        results = {
            "episode_stats": dict(self.env.cluster.episode_stats),
            "num_actor_steps": steps,
            "episode_return": float(np.sum(rewards)),
            "mean_reward": float(np.mean(rewards)) if rewards else 0.0,
            "run_time": time.perf_counter() - t0,
        }
        es = results["episode_stats"]
        for key in ("job_completion_time", "job_completion_time_speedup"):
            vals = es.get(key, [])
            results[f"mean_{key}"] = float(np.mean(vals)) if len(vals) else None
        results["blocking_rate"] = es.get("blocking_rate", None)
        results["acceptance_rate"] = es.get("acceptance_rate", None)
        return results


class PolicyActor:
    """Greedy/stochastic actor wrapping a trained GNNPolicy for EvalLoop."""

    def __init__(self, policy, device=None, deterministic: bool = True,
                 name: str = "gnn_policy"):
        import torch
        self.torch = torch
        self.policy = policy
        self.device = device or torch.device("cpu")
        self.policy.to(self.device)
        self.deterministic = deterministic
        self.name = name

    def compute_action(self, obs, *args, **kwargs):
        import torch
        from ..rl.rollout import CompactObs, collate
        with torch.no_grad():
            inputs = collate([CompactObs.from_obs(obs)], self.device)
            logits, _ = self.policy.forward_flat(
                inputs["batch"], inputs["graph_features"], inputs["action_mask"])
            if self.deterministic:
                return int(torch.argmax(logits, dim=-1).item())
            return int(torch.distributions.Categorical(logits=logits)
                       .sample().item())

    @classmethod
    def from_checkpoint(cls, checkpoint_path: str, num_actions: int = 17,
                        config: Optional[dict] = None, **kwargs):
        from ..models.gnn import GNNPolicy
        state = Checkpointer.read(checkpoint_path)
        state = state.get("trainer", state)   # launcher full-state layout
        policy = GNNPolicy(num_actions=num_actions, config=config)
        policy.load_state_dict(state["policy"])
        return cls(policy, **kwargs)


class Launcher:
    """Run epochs until a budget is hit; checkpoint + log on a cadence and
    track the best checkpoint by eval episode return
    (reference ``launcher.py:97-184``, ``rllib_epoch_loop.py:144-230``)."""

    def __init__(self,
                 epoch_loop: EpochLoop,
                 num_epochs: Optional[int] = None,
                 max_actor_steps: Optional[int] = None,
                 evaluation_interval: int = 10,
                 eval_fn: Optional[Callable[[], Dict]] = None,
                 path_to_save: Optional[str] = None,
                 use_sqlite_database: bool = False,
                 external_logger=None,
                 verbose: bool = True):
        self.epoch_loop = epoch_loop
        self.num_epochs = num_epochs
        self.max_actor_steps = max_actor_steps
        self.evaluation_interval = evaluation_interval
        self.eval_fn = eval_fn
        self.logger = (Logger(path_to_save, use_sqlite_database=use_sqlite_database)
                       if path_to_save else None)
        self.checkpointer = Checkpointer(path_to_save) if path_to_save else None
        self.external_logger = external_logger   # runtime.logger.ExternalLogger
        self.verbose = verbose
        self.results_log = defaultdict(list)
        self.best_eval_return = -float("inf")
        self.best_checkpoint: Optional[str] = None
        self.epoch = 0

    # ------------------------------------------------------------------
    # full-state resume (reference launcher.py:97-184
    # restarts from the RLlib checkpoint — here the launcher's own
    # bookkeeping rides along in the same checkpoint file)
    def _full_state(self) -> Dict:
        return {
            "trainer": self.epoch_loop.state_dict(),
            "launcher": {
                "epoch": self.epoch,
                "results_log": {k: list(v)
                                for k, v in self.results_log.items()},
                "best_eval_return": self.best_eval_return,
                "best_checkpoint": self.best_checkpoint,
            },
        }

    def _write_checkpoint(self, index: int) -> Optional[str]:
        if self.checkpointer is None:
            return None
        return self.checkpointer.write(self._full_state(), index=index)

    def resume(self, checkpoint_path: Optional[str] = None) -> bool:
        """Restore trainer + launcher bookkeeping from a checkpoint (the
        latest one by default).  Returns True if a checkpoint was loaded."""
        if checkpoint_path is None:
            if self.checkpointer is None:
                return False
            checkpoint_path = self.checkpointer.latest()
            if checkpoint_path is None:
                return False
        state = Checkpointer.read(checkpoint_path)
        trainer_state = state.get("trainer", state)   # old flat layout too
        self.epoch_loop.trainer.load_state_dict(trainer_state)
        ls = state.get("launcher")
        if ls is not None:
            self.epoch = int(ls.get("epoch", 0))
            self.results_log = defaultdict(list)
            for k, v in ls.get("results_log", {}).items():
                self.results_log[k] = list(v)
            self.best_eval_return = ls.get("best_eval_return", -float("inf"))
            self.best_checkpoint = ls.get("best_checkpoint")
        return True

    def _should_stop(self, epoch: int, total_steps: int) -> bool:
        if self.num_epochs is not None and epoch >= self.num_epochs:
            return True
        if self.max_actor_steps is not None and total_steps >= self.max_actor_steps:
            return True
        return False

    def run(self) -> Dict:
        epoch = self.epoch
        if self.checkpointer is not None and epoch == 0:
            self._write_checkpoint(index=0)
        while not self._should_stop(epoch, getattr(self.epoch_loop.trainer,
                                                   "total_env_steps", 0)):
            stats = self.epoch_loop.run()
            epoch += 1
            self.epoch = epoch
            for k, v in stats.items():
                if isinstance(v, (int, float, np.floating, np.integer)):
                    self.results_log[k].append(float(v))
            self.results_log["epoch_counter"].append(epoch)
            if self.external_logger is not None:
                self.external_logger.log(stats, step=epoch)

            if epoch % self.evaluation_interval == 0 or self._should_stop(
                    epoch, getattr(self.epoch_loop.trainer, "total_env_steps", 0)):
                if self.eval_fn is not None:
                    eval_results = self.eval_fn()
                    ret = eval_results.get("episode_return", None)
                    self.results_log["eval_episode_return"].append(ret)
                    if ret is not None and ret > self.best_eval_return:
                        self.best_eval_return = ret
                        self.best_checkpoint = self._write_checkpoint(epoch)
                    else:
                        self._write_checkpoint(epoch)
                else:
                    self._write_checkpoint(epoch)
                if self.logger is not None:
                    self.logger.write({"train_log": dict(self.results_log)})
            if self.verbose:
                msg = (f"epoch {epoch}: loss={stats.get('total_loss', 0):.4f} "
                       f"kl={stats.get('kl', 0):.5f} "
                       f"reward={stats.get('mean_reward', 0):.3f}")
                if "episode_reward_mean" in stats:
                    msg += f" ep_ret={stats['episode_reward_mean']:.2f}"
                print(msg, flush=True)
        if self.logger is not None:
            self.logger.write({"train_log": dict(self.results_log)}, block=True)
        return dict(self.results_log)
