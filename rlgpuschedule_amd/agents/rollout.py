"""Rollout collection over a VectorEnv (CPU or HIP) with a torch policy."""
from __future__ import annotations

import numpy as np
import torch

from ..models.policy import sample_categorical


class TorchEnvAdapter:
    """Presents CpuVectorEnv through the torch-tensor VectorEnv protocol."""

    def __init__(self, venv, device="cpu"):
        self.venv = venv
        self.spec = venv.spec
        self.num_envs = venv.num_envs
        self.device = torch.device(device)

    def reset(self):
        return torch.from_numpy(self.venv.reset()).to(self.device)

    def action_masks(self):
        return torch.from_numpy(self.venv.action_masks()).to(self.device)

    def step(self, actions: torch.Tensor):
        obs, rew, done, info = self.venv.step(actions.cpu().numpy())
        return (torch.from_numpy(obs).to(self.device),
                torch.from_numpy(rew).to(self.device),
                torch.from_numpy(done.astype(np.uint8)).to(self.device),
                {"action_mask": torch.from_numpy(
                    info["action_mask"].astype(np.uint8)).to(self.device)})


class RolloutBuffer:
    def __init__(self, T: int, E: int, obs_dim: int, num_actions: int,
                 device, hidden: int = 0):
        self.T, self.E = T, E
        dev = device
        self.obs = torch.zeros(T, E, obs_dim, device=dev)
        self.mask = torch.zeros(T, E, num_actions, dtype=torch.uint8, device=dev)
        self.actions = torch.zeros(T, E, dtype=torch.int64, device=dev)
        self.logp = torch.zeros(T, E, device=dev)
        self.value = torch.zeros(T, E, device=dev)
        self.reward = torch.zeros(T, E, device=dev)
        self.done = torch.zeros(T, E, dtype=torch.uint8, device=dev)
        self.last_value = torch.zeros(E, device=dev)
        # recurrent-policy support: hidden state at rollout start
        self.h0 = torch.zeros(E, hidden, device=dev) if hidden else None


class Collector:
    """Steps the env with the policy for T steps per collect() call.

    Keeps (obs, mask, hx) across calls so rollouts are contiguous.
    """

    def __init__(self, env, policy, T: int, device, seed: int = 0):
        self.env = env
        self.policy = policy
        self.T = T
        self.device = torch.device(device)
        self.gen = torch.Generator(device=self.device)
        self.gen.manual_seed(seed)
        self.obs = env.reset().to(torch.float32)
        self.mask = env.action_masks()
        self.hx = policy.initial_state(env.num_envs, self.device)
        hidden = policy.hidden if policy.is_recurrent else 0
        self.buf = RolloutBuffer(T, env.num_envs, env.spec.obs_dim,
                                 env.spec.num_actions, self.device, hidden)

    @torch.no_grad()
    def collect(self) -> RolloutBuffer:
        buf = self.buf
        if buf.h0 is not None:
            buf.h0.copy_(self.hx)
        for t in range(self.T):
            buf.obs[t].copy_(self.obs)
            buf.mask[t].copy_(self.mask)
            logits, value, hx_next = self.policy(self.obs, self.mask, self.hx)
            u = torch.rand(self.env.num_envs, device=self.device,
                           generator=self.gen)
            a, lp, _ = sample_categorical(logits, u)
            buf.actions[t] = a
            buf.logp[t] = lp
            buf.value[t] = value
            obs, rew, done, info = self.env.step(a)
            buf.reward[t].copy_(rew)
            buf.done[t].copy_(done)
            self.obs = obs.to(torch.float32)
            self.mask = info["action_mask"]
            if hx_next is not None:
                self.hx = hx_next * (1.0 - done.to(hx_next.dtype)).unsqueeze(-1)
        _, last_v, _ = self.policy(self.obs, self.mask, self.hx)
        buf.last_value.copy_(last_v)
        return buf
