"""Fully on-device rollout engine: fused actor kernel + HIP env step.

Per rollout step exactly TWO kernel launches (actor_mlp, env_step), all
outputs written straight into the rollout buffer slices (no copies). The
T-step loop is captured in a hipGraph (one graph per buffer) and replayed
once per iteration — rollout cost becomes pure kernel time.

`num_bufs=2` double-buffers rollouts for the IMPALA async actor/learner
(agents/impala.py): the actor stream fills buf[i] while the learner stream
trains on buf[i^1]. Env state chains across rollouts via obs_last/mask_last.
"""
from __future__ import annotations

import torch

from ..models.hip_actor import HipActorGRU, HipActorMLP
from .rollout import RolloutBuffer


class GpuRolloutEngine:
    def __init__(self, env, policy, T: int, seed: int = 0,
                 use_graph: bool = True, num_bufs: int = 1):
        self.env = env
        self.policy = policy
        self.T = T
        self.device = torch.device("cuda")
        # one behavior-policy mirror per buffer: lets an async learner
        # refresh mirror i (after its update) while a rollout on the other
        # parity is still reading mirror i^1 (agents/impala.py)
        acls = HipActorGRU if policy.is_recurrent else HipActorMLP
        self.actors = [acls(policy) for _ in range(num_bufs)]
        self.actor = self.actors[0]
        E = env.num_envs
        spec = env.spec
        hid = policy.hidden if policy.is_recurrent else 0
        self.bufs = [RolloutBuffer(T, E, spec.obs_dim, spec.num_actions,
                                   self.device, hidden=hid)
                     for _ in range(num_bufs)]
        if policy.is_recurrent:
            self.hx = torch.zeros(E, policy.hidden, device=self.device)
            self.hx_scratch = torch.empty_like(self.hx)
            self.done_last = torch.zeros(E, dtype=torch.uint8,
                                         device=self.device)
        self.buf = self.bufs[0]
        self.u_all = torch.empty(T, E, device=self.device)
        self.gen = torch.Generator(device=self.device)
        self.gen.manual_seed(seed)
        self.obs_last = torch.empty(E, spec.obs_dim, device=self.device)
        self.mask_last = torch.empty(E, spec.num_actions, dtype=torch.uint8,
                                     device=self.device)
        self.use_graph = use_graph
        self._graphs: dict[int, torch.cuda.CUDAGraph] = {}
        # initial reset: obs_last/mask_last always hold the NEXT rollout's
        # start state
        env.reset()
        self.obs_last.copy_(env.b["obs"])
        self.mask_last.copy_(env.b["mask"])

    def _rollout_body(self, buf: RolloutBuffer, actor):
        env, T = self.env, self.T
        recurrent = self.policy.is_recurrent
        if recurrent:
            # learner replays from h0 (already reset where done_last)
            buf.h0.copy_(self.hx
                         * (1.0 - self.done_last.float()).unsqueeze(-1))
        for t in range(T):
            if recurrent:
                dprev = self.done_last if t == 0 else buf.done[t - 1]
                actor.act(buf.obs[t], buf.mask[t], self.u_all[t], dprev,
                          self.hx, self.hx, buf.actions[t], buf.logp[t],
                          buf.value[t])
            else:
                actor.act(buf.obs[t], buf.mask[t], self.u_all[t],
                          buf.actions[t], buf.logp[t], buf.value[t])
            last = t == T - 1
            env.step(
                buf.actions[t],
                obs_out=self.obs_last if last else buf.obs[t + 1],
                mask_out=self.mask_last if last else buf.mask[t + 1],
                rew_out=buf.reward[t],
                done_out=buf.done[t],
            )
        if recurrent:
            self.done_last.copy_(buf.done[T - 1])
            actor.values(self.obs_last, self.mask_last, self.done_last,
                         self.hx, self.hx_scratch, buf.last_value)
        else:
            actor.values(self.obs_last, buf.last_value, self.mask_last)

    def collect(self, idx: int = 0, refresh: bool = True) -> RolloutBuffer:
        """Run one T-step rollout into bufs[idx] on the CURRENT stream.

        The t=0 slices stay valid afterwards (the learner trains on them).
        `refresh=False` keeps the behavior policy stale (IMPALA).
        """
        buf = self.bufs[idx]
        actor = self.actors[idx]
        if refresh:
            actor.refresh()
        self.u_all.uniform_(generator=self.gen)
        buf.obs[0].copy_(self.obs_last)
        buf.mask[0].copy_(self.mask_last)
        if self.use_graph:
            if idx not in self._graphs:
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    self._rollout_body(buf, actor)   # warmup
                torch.cuda.current_stream().wait_stream(s)
                # warmup advanced the env: re-chain the replay's start state
                buf.obs[0].copy_(self.obs_last)
                buf.mask[0].copy_(self.mask_last)
                torch.cuda.synchronize()
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self._rollout_body(buf, actor)
                self._graphs[idx] = g
            self._graphs[idx].replay()
        else:
            self._rollout_body(buf, actor)
        self.buf = buf
        return buf
