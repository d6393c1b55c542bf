"""Fully on-device rollout engine: fused actor kernel + HIP env step.

Per rollout step exactly TWO kernel launches (actor_mlp, env_step), all
outputs written straight into the rollout buffer slices (no copies). The
whole T-step loop is captured in a hipGraph (torch.cuda.CUDAGraph) and
replayed once per iteration — rollout cost becomes pure kernel time.
"""
from __future__ import annotations

import torch

from ..models.hip_actor import HipActorMLP
from .rollout import RolloutBuffer


class GpuRolloutEngine:
    def __init__(self, env, policy, T: int, seed: int = 0,
                 use_graph: bool = True):
        self.env = env
        self.policy = policy
        self.T = T
        self.device = torch.device("cuda")
        self.actor = HipActorMLP(policy)
        E = env.num_envs
        spec = env.spec
        self.buf = RolloutBuffer(T, E, spec.obs_dim, spec.num_actions,
                                 self.device)
        self.u_all = torch.empty(T, E, device=self.device)
        self.gen = torch.Generator(device=self.device)
        self.gen.manual_seed(seed)
        self.obs_last = torch.empty(E, spec.obs_dim, device=self.device)
        self.mask_last = torch.empty(E, spec.num_actions, dtype=torch.uint8,
                                     device=self.device)
        self.rew_last = torch.empty(E, device=self.device)
        self.done_last = torch.empty(E, dtype=torch.uint8, device=self.device)
        self.use_graph = use_graph
        self._graph = None
        # initial reset: env writes obs/mask into its default buffers;
        # obs_last/mask_last always hold the NEXT rollout's start state
        env.reset()
        self.obs_last.copy_(env.b["obs"])
        self.mask_last.copy_(env.b["mask"])

    def _rollout_body(self):
        buf, env, T = self.buf, self.env, self.T
        for t in range(T):
            self.actor.act(buf.obs[t], buf.mask[t], self.u_all[t],
                           buf.actions[t], buf.logp[t], buf.value[t])
            last = t == T - 1
            env.step(
                buf.actions[t],
                obs_out=self.obs_last if last else buf.obs[t + 1],
                mask_out=self.mask_last if last else buf.mask[t + 1],
                rew_out=buf.reward[t],
                done_out=buf.done[t],
            )
        self.actor.values(self.obs_last, buf.last_value, self.mask_last)

    def collect(self) -> RolloutBuffer:
        self.actor.refresh()
        self.u_all.uniform_(generator=self.gen)
        # rollout starts from the state the previous one ended in; the t=0
        # slices must stay untouched afterwards (the learner reads them)
        self.buf.obs[0].copy_(self.obs_last)
        self.buf.mask[0].copy_(self.mask_last)
        if self.use_graph:
            if self._graph is None:
                # warm once on a side stream, then capture
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    self._rollout_body()
                torch.cuda.current_stream().wait_stream(s)
                # warmup advanced the env: re-chain the replay's start state
                self.buf.obs[0].copy_(self.obs_last)
                self.buf.mask[0].copy_(self.mask_last)
                torch.cuda.synchronize()
                self._graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(self._graph):
                    self._rollout_body()
            self._graph.replay()
        else:
            self._rollout_body()
        return self.buf
