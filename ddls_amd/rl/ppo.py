"""From-scratch PPO for the PAC-ML GNN policy.

Replaces the reference's RLlib PPOTrainer (``loops/rllib_epoch_loop.py:81,235``;
tuned hparams from ``scripts/.../algo/ppo.yaml:16-56``): GAE, clipped
surrogate, adaptive KL coefficient, clipped value loss, entropy bonus,
minibatch SGD with shuffling, advantage standardisation.  Data parallelism:
one process per GPU, fused gradient all-reduce with RCCL over xGMI
(ddls_amd.parallel).
"""
from __future__ import annotations

import os
import time
from dataclasses import dataclass
from typing import Dict, List, Optional

import numpy as np
import torch
import torch.nn.functional as F

from ..models.gnn import GNNPolicy
from ..parallel import all_reduce_gradients, get_rank
from .rollout import CompactObs, VectorEnv, collate


@dataclass
class PPOConfig:
    # tuned reference values (algo/ppo.yaml)
    lr: float = 2.785e-4
    gamma: float = 0.997
    lambda_: float = 1.0
    clip_param: float = 0.18
    entropy_coeff: float = 3e-3
    kl_coeff: float = 0.01
    kl_target: float = 1e-3
    vf_clip_param: float = 128.8
    vf_loss_coeff: float = 1.0
    grad_clip: float = 1.5
    sgd_minibatch_size: int = 128
    train_batch_size: int = 4000
    num_sgd_iter: int = 50
    # capture the SGD minibatch step in a hipGraph on GPU (graph_step.py);
    # eager fallback on CPU / partial minibatches / capture failure
    use_hip_graphs: bool = True


class PPOTrainer:
    def __init__(self,
                 vector_env: VectorEnv,
                 policy: GNNPolicy,
                 config: Optional[PPOConfig] = None,
                 device: Optional[torch.device] = None):
        self.env = vector_env
        self.config = config or PPOConfig()
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu")
        self.policy = policy.to(self.device)
        # foreach: batched multi-tensor Adam kernels (single-tensor capturable
        # Adam shows up as ~73 tiny per-param Div launches per graph replay)
        self.optimizer = torch.optim.Adam(self.policy.parameters(),
                                          lr=self.config.lr, foreach=True)
        self.kl_coeff = self.config.kl_coeff
        self.obs = self.env.reset()
        self.total_env_steps = 0
        self.iteration = 0
        self._stepper = None  # lazy CapturedSGDStep (GPU only)
        self._mb_executor = None  # prefetch thread for captured minibatches

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _policy_step(self, obs_list: List[CompactObs]):
        inputs = collate(obs_list, self.device)
        logits, values = self._forward_flat(inputs)
        dist = torch.distributions.Categorical(logits=logits)
        actions = dist.sample()
        logp = dist.log_prob(actions)
        return (actions.cpu().numpy(), logp.cpu().numpy(),
                values.cpu().numpy(), logits.cpu().numpy())

    def _forward_flat(self, inputs):
        """Policy forward on a pre-collated flat batch."""
        return self.policy.forward_flat(inputs["batch"],
                                        inputs["graph_features"],
                                        inputs["action_mask"])

    # ------------------------------------------------------------------
    def collect_rollout(self, num_steps: Optional[int] = None) -> Dict[str, np.ndarray]:
        """Collect num_steps per-rank env steps across the vectorised envs."""
        cfg = self.config
        n_envs = len(self.env)
        steps = num_steps if num_steps is not None else max(
            1, cfg.train_batch_size // n_envs)

        if hasattr(self.env, "rollout"):
            # RLlib-style worker-side rollouts: policy copies live in the env
            # workers; one IPC round trip per iteration
            data = self.env.rollout(self.policy, steps)
            self.total_env_steps += steps * n_envs
            rewards, dones = data["rewards"], data["dones"]
            values, logps, actions = data["values"], data["logp"], data["actions"]
            bootstrap_values = data["bootstrap_values"]
            T = rewards.shape[0]
            adv = np.zeros_like(rewards)
            lastgaelam = np.zeros(n_envs)
            for t in reversed(range(T)):
                next_values = bootstrap_values if t == T - 1 else values[t + 1]
                nonterminal = 1.0 - dones[t].astype(np.float64)
                delta = (rewards[t] + cfg.gamma * next_values * nonterminal
                         - values[t])
                lastgaelam = (delta + cfg.gamma * cfg.lambda_ * nonterminal
                              * lastgaelam)
                adv[t] = lastgaelam
            out = {
                "obs": data["obs"],
                "actions": actions.reshape(-1),
                "logp": logps.reshape(-1),
                "advantages": adv.reshape(-1),
                "value_targets": (adv + values).reshape(-1),
                "values": values.reshape(-1),
                "rewards": rewards.reshape(-1),
            }
            if "lp_all" in data:
                lp = np.asarray(data["lp_all"])
                out["lp_all"] = lp.reshape(-1, lp.shape[-1])
            return out

        obs_buf: List[CompactObs] = []
        act_buf, logp_buf, lp_all_buf = [], [], []
        rew_buf, done_buf, val_buf = [], [], []
        for _ in range(steps):
            actions, logp, values, logits = self._policy_step(self.obs)
            x = logits - logits.max(axis=-1, keepdims=True)
            lp_all_buf.append(
                (x - np.log(np.exp(x).sum(axis=-1, keepdims=True)))
                .astype(np.float32))
            obs_buf.extend(self.obs)
            act_buf.append(actions)
            logp_buf.append(logp)
            val_buf.append(values)
            self.obs, rewards, dones = self.env.step(actions)
            rew_buf.append(rewards)
            done_buf.append(dones)
            self.total_env_steps += n_envs
        with torch.no_grad():
            _, _, bootstrap_values, _ = self._policy_step(self.obs)

        # [T, N] arrays
        rewards = np.stack(rew_buf)
        dones = np.stack(done_buf)
        values = np.stack(val_buf)
        logps = np.stack(logp_buf)
        actions = np.stack(act_buf)

        # GAE(lambda) per env column
        T = rewards.shape[0]
        adv = np.zeros_like(rewards)
        lastgaelam = np.zeros(n_envs)
        for t in reversed(range(T)):
            next_values = bootstrap_values if t == T - 1 else values[t + 1]
            nonterminal = 1.0 - dones[t].astype(np.float64)
            delta = rewards[t] + cfg.gamma * next_values * nonterminal - values[t]
            lastgaelam = delta + cfg.gamma * cfg.lambda_ * nonterminal * lastgaelam
            adv[t] = lastgaelam
        value_targets = adv + values

        lp_all = np.stack(lp_all_buf)
        return {
            "obs": obs_buf,
            "actions": actions.reshape(-1),
            "logp": logps.reshape(-1),
            "advantages": adv.reshape(-1),
            "value_targets": value_targets.reshape(-1),
            "values": values.reshape(-1),
            "rewards": rewards.reshape(-1),
            "lp_all": lp_all.reshape(-1, lp_all.shape[-1]),
        }

    # ------------------------------------------------------------------
    def update(self, batch: Dict) -> Dict[str, float]:
        cfg = self.config
        n = len(batch["actions"])
        adv = batch["advantages"]
        # RLlib standardizes advantages over the train batch
        adv = (adv - adv.mean()) / max(adv.std(), 1e-4)

        actions = torch.as_tensor(batch["actions"], device=self.device)
        old_logp = torch.as_tensor(batch["logp"], device=self.device,
                                   dtype=torch.float32)
        advantages = torch.as_tensor(adv, device=self.device,
                                     dtype=torch.float32)
        value_targets = torch.as_tensor(batch["value_targets"],
                                        device=self.device, dtype=torch.float32)
        old_values = torch.as_tensor(batch["values"], device=self.device,
                                     dtype=torch.float32)

        stats = {"policy_loss": 0.0, "vf_loss": 0.0, "kl": 0.0, "entropy": 0.0,
                 "total_loss": 0.0}
        num_updates = 0
        num_captured = 0

        # hipGraph-captured minibatch step (GPU): one replay per minibatch
        # instead of ~120 host-dispatched kernel launches
        stepper = None
        if (cfg.use_hip_graphs and self.device.type == "cuda"
                and torch.cuda.is_available()):
            if self._stepper is None:
                from .graph_step import CapturedSGDStep
                # cached-models mode when the env serves per-model static
                # obs (vectorised engine): GNN once per model per minibatch
                mb_static = getattr(self.env, "_models_batch", None)
                if mb_static is not None and batch["obs"] \
                        and getattr(batch["obs"][0], "model_id", -1) < 0:
                    mb_static = None
                self._stepper = CapturedSGDStep(self.policy, self.optimizer,
                                                cfg, self.device,
                                                models_batch=mb_static)
            stepper = self._stepper
            stepper.reset_stats()
            stepper.set_kl_coeff(self.kl_coeff)
        actions_np = np.asarray(batch["actions"], dtype=np.int64)
        logp_np = np.asarray(batch["logp"], dtype=np.float32)
        adv_np = np.asarray(adv, dtype=np.float32)
        vtarg_np = np.asarray(batch["value_targets"], dtype=np.float32)

        rng = np.random.RandomState(self.iteration + 1234 * get_rank())
        B = cfg.sgd_minibatch_size

        captured = False
        if stepper is not None and not stepper.broken and n >= B and n % B == 0:
            idx_lists = []
            for _ in range(cfg.num_sgd_iter):
                perm = rng.permutation(n)
                for start in range(0, n, B):
                    idx_lists.append(perm[start:start + B])
            if stepper.models_batch is not None:
                # cached-models mode: fixed shapes, single capture
                captured = stepper.ensure_capacity(0, 0)
            else:
                # the permutations are fixed up front, so capacity can be
                # sized to the LARGEST ACTUAL minibatch (sum of 128 sample
                # sizes concentrates near B x mean — sizing by B x
                # max-sample would nearly double the padded kernel work)
                ns_arr = np.array([len(o.node_features)
                                   for o in batch["obs"]], dtype=np.int64)
                es_arr = np.array([len(o.edges_src) for o in batch["obs"]],
                                  dtype=np.int64)
                need_n = max(int(ns_arr[idx].sum()) for idx in idx_lists)
                need_e = max(int(es_arr[idx].sum()) for idx in idx_lists)
                captured = stepper.ensure_capacity(int(need_n * 1.05) + 8,
                                                   int(need_e * 1.05) + 8)
            if not captured:  # replay the same permutation stream eagerly
                rng = np.random.RandomState(self.iteration + 1234 * get_rank())

        if captured and os.environ.get("DDLS_AMD_DISABLE_PREFETCH", "0") == "1":
            for idx in idx_lists:
                ok = self._stepper.step([batch["obs"][i] for i in idx],
                                        actions_np[idx], logp_np[idx],
                                        adv_np[idx], vtarg_np[idx])
                assert ok
                num_updates += 1
                num_captured += 1
            rng = None
            captured = False

        if captured:
            # captured fast loop with CPU prefetch: a worker thread stages
            # minibatch k+1 into the spare pinned set while the GPU replays
            # minibatch k (the double-buffer events make this safe)
            def mb_data(idx):
                return ([batch["obs"][i] for i in idx], actions_np[idx],
                        logp_np[idx], adv_np[idx], vtarg_np[idx])

            if self._mb_executor is None:
                from concurrent.futures import ThreadPoolExecutor
                self._mb_executor = ThreadPoolExecutor(max_workers=1)
            ex = self._mb_executor
            fut = ex.submit(stepper.prepare, *mb_data(idx_lists[0]))
            for i in range(len(idx_lists)):
                j = fut.result()
                if i + 1 < len(idx_lists):
                    fut = ex.submit(stepper.prepare, *mb_data(idx_lists[i + 1]))
                stepper.commit(j)
                num_updates += 1
                num_captured += 1
            rng = None  # permutations consumed identically to the eager loop

        if rng is not None:
            for _ in range(cfg.num_sgd_iter):
                perm = rng.permutation(n)
                for start in range(0, n, cfg.sgd_minibatch_size):
                    idx = perm[start:start + cfg.sgd_minibatch_size]
                    if len(idx) < 2:
                        continue
                    mb_obs = [batch["obs"][i] for i in idx]
                    if stepper is not None and stepper.step(
                            mb_obs, actions_np[idx], logp_np[idx], adv_np[idx],
                            vtarg_np[idx]):
                        num_updates += 1
                        num_captured += 1
                        continue
                    inputs = collate(mb_obs, self.device)
                    logits, values = self._forward_flat(inputs)
                    dist = torch.distributions.Categorical(logits=logits)
                    idx_t = torch.as_tensor(idx, device=self.device)
                    logp = dist.log_prob(actions[idx_t])
                    ratio = torch.exp(logp - old_logp[idx_t])
                    mb_adv = advantages[idx_t]
                    surr = torch.min(
                        ratio * mb_adv,
                        torch.clamp(ratio, 1 - cfg.clip_param,
                                    1 + cfg.clip_param) * mb_adv)
                    policy_loss = -surr.mean()

                    # KL(old || new) sample estimate (RLlib uses action logp kl)
                    kl = (old_logp[idx_t] - logp).mean()

                    vf_err = (values - value_targets[idx_t]) ** 2
                    vf_loss = torch.clamp(vf_err, 0, cfg.vf_clip_param).mean()

                    entropy = dist.entropy().mean()

                    loss = (policy_loss + self.kl_coeff * kl
                            + cfg.vf_loss_coeff * vf_loss
                            - cfg.entropy_coeff * entropy)

                    self.optimizer.zero_grad(set_to_none=True)
                    loss.backward()
                    all_reduce_gradients(self.policy.parameters())
                    if cfg.grad_clip is not None:
                        torch.nn.utils.clip_grad_norm_(self.policy.parameters(),
                                                       cfg.grad_clip)
                    self.optimizer.step()

                    stats["policy_loss"] += policy_loss.detach().item()
                    stats["vf_loss"] += vf_loss.detach().item()
                    stats["kl"] += kl.detach().item()
                    stats["entropy"] += entropy.detach().item()
                    stats["total_loss"] += loss.detach().item()
                    num_updates += 1

        if num_captured:
            acc = self._stepper.stats_acc.cpu().numpy()  # one sync per update
            for k, v in zip(("policy_loss", "vf_loss", "kl", "entropy",
                             "total_loss"), acc):
                stats[k] += float(v)
        for k in stats:
            stats[k] /= max(num_updates, 1)

        # adaptive KL coefficient (RLlib rule) driven by the ANALYTIC
        # KL(old || new) over the full old/new distributions (ADVICE r01:
        # the taken-action sample estimate is noisy and can go negative)
        kl_for_adapt = stats["kl"]
        if batch.get("lp_all") is not None:
            try:
                kl_an = self._analytic_kl(batch)
                stats["kl_analytic"] = kl_an
                kl_for_adapt = kl_an
            except Exception:
                pass
        if kl_for_adapt > 2.0 * cfg.kl_target:
            self.kl_coeff *= 1.5
        elif kl_for_adapt < 0.5 * cfg.kl_target:
            self.kl_coeff *= 0.5
        stats["kl_coeff"] = self.kl_coeff
        if stepper is not None:
            stats["hipgraph_minibatches"] = num_captured
            stats["hipgraph_captures"] = stepper.capture_count
        return stats

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _analytic_kl(self, batch: Dict) -> float:
        """KL(old || new) = E_old[sum_a p_old (lp_old - lp_new)] over the
        train batch, with the post-update policy."""
        obs = batch["obs"]
        old_lp = torch.as_tensor(batch["lp_all"], device=self.device)
        mb_static = getattr(self.env, "_models_batch", None)
        if mb_static is not None and getattr(obs[0], "model_id", -1) >= 0:
            # cached-models forward: GNN once over the model graphs
            from ..models.gnn import graph_mean
            node_emb = self.policy.gnn(mb_static)
            model_emb = graph_mean(node_emb, mb_static)
            mids = torch.as_tensor([o.model_id for o in obs],
                                   device=self.device)
            gfull = torch.as_tensor(
                np.stack([o.graph_features for o in obs]),
                device=self.device)
            mask = torch.as_tensor(np.stack([o.action_mask for o in obs]),
                                   device=self.device)
            graph_emb = self.policy.graph_module(gfull)
            final = torch.cat([model_emb[mids], graph_emb], dim=-1)
            logits = self.policy.policy_branch(final)
            if self.policy.config["apply_action_mask"]:
                logits = logits + torch.clamp(
                    torch.log(mask), min=torch.finfo(torch.float32).min)
            new_lp = F.log_softmax(logits, dim=-1)
        else:
            chunks = []
            for start in range(0, len(obs), 1024):
                inputs = collate(obs[start:start + 1024], self.device)
                logits, _ = self._forward_flat(inputs)
                chunks.append(F.log_softmax(logits, dim=-1))
            new_lp = torch.cat(chunks)
        p_old = torch.exp(old_lp)
        kl = (p_old * (old_lp - new_lp)).sum(-1).mean()
        return float(kl.item())

    # ------------------------------------------------------------------
    def train(self, num_steps: Optional[int] = None) -> Dict[str, float]:
        """One PPO iteration: collect rollout + SGD update."""
        t0 = time.perf_counter()
        batch = self.collect_rollout(num_steps)
        t1 = time.perf_counter()
        stats = self.update(batch)
        t2 = time.perf_counter()
        self.iteration += 1
        stats.update({
            "iteration": self.iteration,
            "env_steps_this_iter": len(batch["actions"]),
            "total_env_steps": self.total_env_steps,
            "rollout_time_s": t1 - t0,
            "update_time_s": t2 - t1,
            "mean_reward": float(np.mean(batch["rewards"])),
        })
        episode_stats = self.env.drain_episode_stats()
        if episode_stats:
            stats["episode_reward_mean"] = float(np.mean(
                [s["episode_return"] for s in episode_stats]))
            stats["blocking_rate_mean"] = float(np.mean(
                [s.get("blocking_rate", 0) for s in episode_stats]))
            jcts = [np.mean(s["job_completion_time"]) for s in episode_stats
                    if isinstance(s.get("job_completion_time"), list)
                    and len(s["job_completion_time"]) > 0]
            if jcts:
                stats["mean_job_completion_time"] = float(np.mean(jcts))
        return stats

    # ------------------------------------------------------------------
    def state_dict(self) -> Dict:
        if self._stepper is not None:
            # the captured step keeps Adam state in flat buffers; mirror it
            # back so checkpoints stay torch-Adam compatible
            self._stepper.sync_state_to_optimizer()
        return {
            "policy": self.policy.state_dict(),
            "optimizer": self.optimizer.state_dict(),
            "kl_coeff": self.kl_coeff,
            "iteration": self.iteration,
            "total_env_steps": self.total_env_steps,
        }

    def load_state_dict(self, state: Dict):
        self.policy.load_state_dict(state["policy"])
        self.optimizer.load_state_dict(state["optimizer"])
        # optimizer state tensors were replaced; any captured graph holds
        # stale addresses, so force a re-capture
        self._stepper = None
        self.kl_coeff = state.get("kl_coeff", self.config.kl_coeff)
        self.iteration = state.get("iteration", 0)
        self.total_env_steps = state.get("total_env_steps", 0)
