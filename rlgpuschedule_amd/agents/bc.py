"""Behavior-cloning warm start from the SJF heuristic (round 2).

Motivation (docs/PERF.md round-2 quality section): on properly-sized
held-out pools the SJF-FIFO mean-JCT gap is <1%, and PPO from random
init reliably converges to FIFO-level behavior without rediscovering
SJF's shortest-job priority. The fix is classic: clone SJF into the
policy first (its action is a pure function of the observation slot
features), regress the value head on the SJF rollouts' discounted
returns, THEN let PPO fine-tune with the KL-to-behavior anchor —
starting from the heuristic it must beat instead of from noise.

Works on both HipVectorEnv (cuda tensors) and CpuVectorEnv (numpy); the
cloning itself is plain torch autograd (offline, not a hot path).
"""
from __future__ import annotations

import numpy as np
import torch
import torch.nn.functional as F


def sjf_actions(obs: torch.Tensor, mask: torch.Tensor, K: int
                ) -> torch.Tensor:
    """Vectorized SjfAgent (agents/baselines.py): among mask-valid slots
    [0,K), pick min work feature (obs[:, 4+5k+3]); first index on ties
    (torch argmin = first min, matching np.argmin); else NOOP (=K)."""
    E = obs.shape[0]
    work = obs[:, 4:4 + 5 * K].view(E, K, 5)[:, :, 3]
    valid = mask[:, :K].bool()
    w = torch.where(valid, work, torch.full_like(work, torch.inf))
    amin = w.argmin(dim=1)
    any_valid = valid.any(dim=1)
    return torch.where(any_valid, amin,
                       torch.full_like(amin, K))


@torch.no_grad()
def collect_sjf_rollouts(env, T: int, n_rollouts: int, device,
                         gamma: float = 0.99):
    """Roll the env with SJF actions; returns flat (obs, mask, actions,
    returns) tensors on `device`. Returns are discounted reward-to-go
    with 0 bootstrap at each rollout end (value-head warmup targets)."""
    K = env.spec.queue_visible
    is_torch = hasattr(env, "b")   # HipVectorEnv (device buffers)
    if is_torch:
        env.reset()
        obs = env.b["obs"].clone()
        mask = env.action_masks().clone()
    else:
        obs = torch.as_tensor(env.reset(), dtype=torch.float32)
        mask = torch.as_tensor(env.action_masks()).to(torch.uint8)
    O, M, A_, R_, D_ = [], [], [], [], []
    for _ in range(n_rollouts):
        for _t in range(T):
            a = sjf_actions(obs.to(device), mask.to(device), K)
            O.append(obs.to(device).clone())
            M.append(mask.to(device).clone())
            A_.append(a)
            if is_torch:
                o2, r, d, info = env.step(a)
                obs, mask = o2.clone(), info["action_mask"].clone()
                R_.append(r.to(device).clone())
                D_.append(d.to(device).clone())
            else:
                o2, r, d, info = env.step(a.cpu().numpy())
                obs = torch.as_tensor(o2, dtype=torch.float32)
                mask = torch.as_tensor(info["action_mask"]).to(torch.uint8)
                R_.append(torch.as_tensor(r, dtype=torch.float32,
                                          device=device))
                D_.append(torch.as_tensor(np.asarray(d, np.uint8),
                                          device=device))
    Tn = len(O)
    rew = torch.stack(R_)            # [Tn,E]
    done = torch.stack(D_).float()
    ret = torch.zeros_like(rew)
    acc = torch.zeros_like(rew[0])
    for t in range(Tn - 1, -1, -1):
        if (t + 1) % T == 0:
            acc.zero_()              # rollout boundary: 0 bootstrap
        acc = rew[t] + gamma * acc * (1.0 - done[t])
        ret[t] = acc
    return (torch.cat([o for o in O]), torch.cat([m for m in M]),
            torch.cat(A_), ret.reshape(-1))


def pretrain_bc(policy, env, T: int = 64, n_rollouts: int = 8,
                epochs: int = 4, batch: int = 65536, lr: float = 1e-3,
                gamma: float = 0.99, val_coef: float = 0.5,
                ent_coef: float = 0.0,
                device: str | torch.device = "cuda", log_fn=None) -> dict:
    """Clone SJF into `policy` (cross-entropy on masked logits) and warm
    up the value head (MSE on discounted returns). Returns final losses.

    ent_coef>0 clones SOFTLY: the argmax still matches SJF but the
    distribution keeps entropy, so a subsequent PPO fine-tune starts
    with live exploration instead of a deterministic policy whose
    entropy gradient has already vanished."""
    device = torch.device(device)
    obs, mask, acts, ret = collect_sjf_rollouts(env, T, n_rollouts, device,
                                                gamma)
    n = obs.shape[0]
    opt = torch.optim.Adam(policy.parameters(), lr=lr)
    stats = {}
    for ep in range(epochs):
        perm = torch.randperm(n, device=device)
        for i in range(0, n, batch):
            ix = perm[i:i + batch]
            logits, value, _ = policy(obs[ix], mask[ix], None)
            ce = F.cross_entropy(logits, acts[ix])
            vloss = F.mse_loss(value, ret[ix])
            loss = ce + val_coef * vloss
            if ent_coef:
                lp = F.log_softmax(logits, dim=-1)
                ent = -(lp.exp() * lp.clamp_min(-30.0)).sum(-1).mean()
                loss = loss - ent_coef * ent
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
        stats = {"bc_ce": float(ce.item()), "bc_vloss": float(vloss.item()),
                 "bc_samples": n}
        if log_fn:
            log_fn(f"[bc] epoch {ep}: ce={stats['bc_ce']:.4f} "
                   f"v={stats['bc_vloss']:.4f}")
    return stats
