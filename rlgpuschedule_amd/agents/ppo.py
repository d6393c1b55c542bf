"""PPO (ratio-clip) and A2C learners over collected rollouts.

Torch-autograd reference path; the MI355X fast path swaps the policy
forward/backward and the GAE scan for hand-written HIP kernels while using
the same update equations (tests compare the two).
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F

from ..models.policy import evaluate_actions
from ..ops.gae_ref import gae_advantages
from .rollout import RolloutBuffer


def compute_gae(buf: RolloutBuffer, gamma: float, lam: float):
    """HIP segmented-scan kernel on GPU, torch reference on CPU."""
    if buf.reward.is_cuda:
        from ..ops.nn import gae_scan
        return gae_scan(buf.reward, buf.value, buf.done, buf.last_value,
                        gamma, lam)
    return gae_advantages(buf.reward, buf.value, buf.done, buf.last_value,
                          gamma, lam)


@dataclass
class PPOConfig:
    lr: float = 3e-4
    gamma: float = 0.99
    lam: float = 0.95
    clip: float = 0.2
    epochs: int = 4
    minibatches: int = 4
    vf_coef: float = 0.5
    ent_coef: float = 0.01
    max_grad_norm: float = 0.5
    norm_adv: bool = True
    norm_reward: bool = True
    # KL-to-behavior penalty (k3 estimator: ratio-1-log ratio, always >=0).
    # 0.0 = vanilla PPO. Non-zero counteracts the late-training entropy
    # collapse documented in docs/PERF.md (policy drifts far off-behavior
    # between graph-captured updates).
    kl_coef: float = 0.0
    # Entropy floor (round 2): samples whose per-row policy entropy is
    # below ent_floor (nats) get an EXTRA entropy bonus of ent_floor_coef
    # on top of ent_coef — exploration is forced only where it collapsed.
    ent_floor: float = 0.0
    ent_floor_coef: float = 0.0


class PPOLearner:
    def __init__(self, policy, cfg: PPOConfig, grad_sync=None):
        self.policy = policy
        self.cfg = cfg
        self.opt = torch.optim.Adam(policy.parameters(), lr=cfg.lr, eps=1e-5)
        self.grad_sync = grad_sync  # parallel/dist.py hook (RCCL all-reduce)

    def load_optim_state(self, sd):
        if isinstance(sd, dict) and sd.get("hip_adam"):
            import warnings
            warnings.warn("checkpoint carries HipAdam state but this is the "
                          "torch learner; optimizer state not restored")
            return
        self.opt.load_state_dict(sd)

    def update(self, buf: RolloutBuffer) -> dict:
        cfg = self.cfg
        if cfg.norm_reward:
            if not hasattr(self, "_rnorm"):
                from .rewardnorm import RewardNormalizer
                self._rnorm = RewardNormalizer(cfg.gamma, buf.reward.device)
            self._rnorm(buf.reward, buf.done)
        adv, ret = compute_gae(buf, cfg.gamma, cfg.lam)
        stats = {"pi_loss": 0.0, "v_loss": 0.0, "entropy": 0.0, "clipfrac": 0.0}
        nup = 0
        if self.policy.is_recurrent:
            nup = self._update_recurrent(buf, adv, ret, stats)
        else:
            nup = self._update_ff(buf, adv, ret, stats)
        for k in stats:
            stats[k] /= max(nup, 1)
        stats["mean_reward"] = buf.reward.mean().item()
        return stats

    def _loss(self, logp, entropy, value, old_logp, old_value, adv, ret, stats):
        cfg = self.cfg
        ratio = (logp - old_logp).exp()
        s1 = ratio * adv
        s2 = ratio.clamp(1.0 - cfg.clip, 1.0 + cfg.clip) * adv
        pi_loss = -torch.min(s1, s2).mean()
        if cfg.kl_coef:
            pi_loss = pi_loss + cfg.kl_coef * (ratio - 1.0
                                               - (logp - old_logp)).mean()
        v_clip = old_value + (value - old_value).clamp(-cfg.clip, cfg.clip)
        v_loss = 0.5 * torch.max((value - ret) ** 2, (v_clip - ret) ** 2).mean()
        ent = entropy.mean()
        loss = pi_loss + cfg.vf_coef * v_loss - cfg.ent_coef * ent
        if getattr(cfg, "ent_floor_coef", 0.0):
            # per-row hinge: extra bonus only where entropy collapsed
            # (matches the fused ppo_head_kernel; the hinge gate is
            # treated as a constant wrt the gradient, like the kernel)
            gate = (entropy.detach() < cfg.ent_floor).float()
            loss = loss - cfg.ent_floor_coef * (gate * entropy).mean()
        stats["pi_loss"] += pi_loss.item()
        stats["v_loss"] += v_loss.item()
        stats["entropy"] += ent.item()
        stats["clipfrac"] += ((ratio - 1.0).abs() > cfg.clip).float().mean().item()
        return loss

    def _step_opt(self, loss):
        self.opt.zero_grad(set_to_none=True)
        loss.backward()
        if self.grad_sync is not None:
            self.grad_sync(self.policy)
        torch.nn.utils.clip_grad_norm_(self.policy.parameters(),
                                       self.cfg.max_grad_norm)
        self.opt.step()

    def _update_ff(self, buf, adv, ret, stats) -> int:
        cfg = self.cfg
        T, E = buf.reward.shape
        B = T * E
        obs = buf.obs.reshape(B, -1)
        mask = buf.mask.reshape(B, -1)
        acts = buf.actions.reshape(B)
        old_logp = buf.logp.reshape(B)
        old_v = buf.value.reshape(B)
        adv = adv.reshape(B)
        ret = ret.reshape(B)
        if cfg.norm_adv:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        nup = 0
        mb = B // cfg.minibatches
        for _ in range(cfg.epochs):
            perm = torch.randperm(B, device=obs.device)
            for i in range(cfg.minibatches):
                ix = perm[i * mb:(i + 1) * mb]
                logits, value, _ = self.policy(obs[ix], mask[ix], None)
                logp, entropy = evaluate_actions(logits, acts[ix])
                loss = self._loss(logp, entropy, value, old_logp[ix],
                                  old_v[ix], adv[ix], ret[ix], stats)
                self._step_opt(loss)
                nup += 1
        return nup

    def _update_recurrent(self, buf, adv, ret, stats) -> int:
        """Minibatch over ENVS; replay the T-step sequence with stored h0,
        resetting hidden state at stored done flags."""
        cfg = self.cfg
        T, E = buf.reward.shape
        if cfg.norm_adv:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        nup = 0
        mbe = max(E // cfg.minibatches, 1)
        for _ in range(cfg.epochs):
            perm = torch.randperm(E, device=buf.obs.device)
            for i in range(cfg.minibatches):
                ix = perm[i * mbe:(i + 1) * mbe]
                hx = buf.h0[ix]
                logps, ents, values = [], [], []
                for t in range(T):
                    logits, value, hx = self.policy(buf.obs[t, ix],
                                                    buf.mask[t, ix], hx)
                    lp, ent = evaluate_actions(logits, buf.actions[t, ix])
                    logps.append(lp)
                    ents.append(ent)
                    values.append(value)
                    hx = hx * (1.0 - buf.done[t, ix].to(hx.dtype)).unsqueeze(-1)
                logp = torch.stack(logps)
                entropy = torch.stack(ents)
                value = torch.stack(values)
                loss = self._loss(logp, entropy, value, buf.logp[:, ix],
                                  buf.value[:, ix], adv[:, ix], ret[:, ix],
                                  stats)
                self._step_opt(loss)
                nup += 1
        return nup


@dataclass
class A2CConfig:
    lr: float = 7e-4
    gamma: float = 0.99
    lam: float = 1.0       # A2C uses plain returns (lam=1 GAE)
    vf_coef: float = 0.5
    ent_coef: float = 0.01
    max_grad_norm: float = 0.5
    norm_reward: bool = True


class A2CLearner:
    """Synchronous advantage actor-critic: one gradient step per rollout."""

    def __init__(self, policy, cfg: A2CConfig, grad_sync=None):
        self.policy = policy
        self.cfg = cfg
        self.opt = torch.optim.RMSprop(policy.parameters(), lr=cfg.lr,
                                       alpha=0.99, eps=1e-5)
        self.grad_sync = grad_sync

    load_optim_state = PPOLearner.load_optim_state

    def update(self, buf: RolloutBuffer) -> dict:
        cfg = self.cfg
        if cfg.norm_reward:
            if not hasattr(self, "_rnorm"):
                from .rewardnorm import RewardNormalizer
                self._rnorm = RewardNormalizer(cfg.gamma, buf.reward.device)
            self._rnorm(buf.reward, buf.done)
        adv, ret = compute_gae(buf, cfg.gamma, cfg.lam)
        T, E = buf.reward.shape
        if self.policy.is_recurrent:
            hx = buf.h0
            logps, ents, values = [], [], []
            for t in range(T):
                logits, value, hx = self.policy(buf.obs[t], buf.mask[t], hx)
                lp, ent = evaluate_actions(logits, buf.actions[t])
                logps.append(lp); ents.append(ent); values.append(value)
                hx = hx * (1.0 - buf.done[t].to(hx.dtype)).unsqueeze(-1)
            logp, entropy, value = (torch.stack(logps), torch.stack(ents),
                                    torch.stack(values))
        else:
            B = T * E
            logits, value, _ = self.policy(buf.obs.reshape(B, -1),
                                           buf.mask.reshape(B, -1), None)
            logp, entropy = evaluate_actions(logits, buf.actions.reshape(B))
            value = value.reshape(T, E)
            logp = logp.reshape(T, E)
            entropy = entropy.reshape(T, E)
        pi_loss = -(adv.detach() * logp).mean()
        v_loss = 0.5 * F.mse_loss(value, ret)
        ent = entropy.mean()
        loss = pi_loss + cfg.vf_coef * v_loss - cfg.ent_coef * ent
        self.opt.zero_grad(set_to_none=True)
        loss.backward()
        if self.grad_sync is not None:
            self.grad_sync(self.policy)
        torch.nn.utils.clip_grad_norm_(self.policy.parameters(),
                                       cfg.max_grad_norm)
        self.opt.step()
        return {"pi_loss": pi_loss.item(), "v_loss": v_loss.item(),
                "entropy": ent.item(), "mean_reward": buf.reward.mean().item()}
