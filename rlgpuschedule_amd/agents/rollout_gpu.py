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

from ..models.hip_actor import (HipActorGRU, HipActorGRUBf16, HipActorMLP,
                                HipActorMLPBf16)
from .rollout import RolloutBuffer


class GpuRolloutEngine:
    def __init__(self, env, policy, T: int, seed: int = 0,
                 use_graph: bool = True, num_bufs: int = 1,
                 actor_dtype: str = "fp32", rollout: str = "twokernel"):
        """`env` may be a LIST of HipVectorEnvs (shards): each shard rolls
        out on its own HIP stream so shard A's env_step overlaps shard B's
        actor kernel (the per-step chain is serial within a shard). The
        rollout buffer covers all shards (shard s owns env-rows
        [off_s, off_{s+1}) — contiguous slices, zero-copy for the learner).
        """
        self.envs = env if isinstance(env, (list, tuple)) else [env]
        self.env = self.envs[0]
        self.policy = policy
        self.T = T
        self.device = torch.device("cuda")
        # rollout="mega": the whole T-step rollout is ONE fused kernel
        # (csrc/rollout_mega.hip) — bit-identical outputs to the
        # two-kernel graph path (tests/test_mega_rollout.py). MLP + fp32
        # actor + single shard only.
        self.rollout_mode = rollout
        if rollout == "mega":
            assert not policy.is_recurrent, "mega rollout: MLP policy only"
            assert actor_dtype == "fp32", "mega rollout: fp32 actor only"
            assert len(self.envs) == 1, "mega rollout: single shard only"
        # one behavior-policy mirror per buffer: lets an async learner
        # refresh mirror i (after its update) while a rollout on the other
        # parity is still reading mirror i^1 (agents/impala.py)
        if policy.is_recurrent:
            acls = HipActorGRUBf16 if actor_dtype == "bf16" else HipActorGRU
        else:
            acls = HipActorMLPBf16 if actor_dtype == "bf16" else HipActorMLP
        self.actors = [acls(policy) for _ in range(num_bufs)]
        self.actor = self.actors[0]
        E = sum(e.num_envs for e in self.envs)
        self.offsets = []
        off = 0
        for e in self.envs:
            self.offsets.append((off, off + e.num_envs))
            off += e.num_envs
        self._streams = [torch.cuda.Stream() for _ in self.envs] \
            if len(self.envs) > 1 else None
        self._join = torch.cuda.Event() if len(self.envs) > 1 else None
        spec = self.env.spec
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
        for sh, (o0, o1) in zip(self.envs, self.offsets):
            sh.reset()
            self.obs_last[o0:o1].copy_(sh.b["obs"])
            self.mask_last[o0:o1].copy_(sh.b["mask"])

    def _rollout_body(self, buf: RolloutBuffer, actor, shard: int = 0):
        env = self.envs[shard]
        o0, o1 = self.offsets[shard]
        T = self.T
        recurrent = self.policy.is_recurrent
        if recurrent:
            # learner replays from h0 (already reset where done_last)
            buf.h0[o0:o1].copy_(
                self.hx[o0:o1]
                * (1.0 - self.done_last[o0:o1].float()).unsqueeze(-1))
        for t in range(T):
            if recurrent:
                dprev = (self.done_last[o0:o1] if t == 0
                         else buf.done[t - 1, o0:o1])
                actor.act(buf.obs[t, o0:o1], buf.mask[t, o0:o1],
                          self.u_all[t, o0:o1], dprev,
                          self.hx[o0:o1], self.hx[o0:o1],
                          buf.actions[t, o0:o1], buf.logp[t, o0:o1],
                          buf.value[t, o0:o1])
            else:
                actor.act(buf.obs[t, o0:o1], buf.mask[t, o0:o1],
                          self.u_all[t, o0:o1], buf.actions[t, o0:o1],
                          buf.logp[t, o0:o1], buf.value[t, o0:o1])
            last = t == T - 1
            env.step(
                buf.actions[t, o0:o1],
                obs_out=self.obs_last[o0:o1] if last else buf.obs[t + 1, o0:o1],
                mask_out=(self.mask_last[o0:o1] if last
                          else buf.mask[t + 1, o0:o1]),
                rew_out=buf.reward[t, o0:o1],
                done_out=buf.done[t, o0:o1],
            )
        if recurrent:
            self.done_last[o0:o1].copy_(buf.done[T - 1, o0:o1])
            actor.values(self.obs_last[o0:o1], self.mask_last[o0:o1],
                         self.done_last[o0:o1], self.hx[o0:o1],
                         self.hx_scratch[o0:o1], buf.last_value[o0:o1])
        else:
            actor.values(self.obs_last[o0:o1], buf.last_value[o0:o1],
                         self.mask_last[o0:o1])

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
        if self.rollout_mode == "mega":
            from ..ops.mega import mega_rollout
            mega_rollout(self.env, actor, buf, self.u_all,
                         self.obs_last, self.mask_last,
                         torch.cuda.current_stream().cuda_stream)
            self.buf = buf
            return buf
        nsh = len(self.envs)
        if self.use_graph:
            for shard in range(nsh):
                key = (idx, shard)
                if key not in self._graphs:
                    s = torch.cuda.Stream()
                    s.wait_stream(torch.cuda.current_stream())
                    with torch.cuda.stream(s):
                        self._rollout_body(buf, actor, shard)   # warmup
                    torch.cuda.current_stream().wait_stream(s)
                    # warmup advanced the env: re-chain the start state
                    o0, o1 = self.offsets[shard]
                    buf.obs[0, o0:o1].copy_(self.obs_last[o0:o1])
                    buf.mask[0, o0:o1].copy_(self.mask_last[o0:o1])
                    torch.cuda.synchronize()
                    g = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(g):
                        self._rollout_body(buf, actor, shard)
                    self._graphs[key] = g
            if nsh == 1:
                self._graphs[(idx, 0)].replay()
            else:
                cur = torch.cuda.current_stream()
                for shard in range(nsh):
                    ss = self._streams[shard]
                    ss.wait_stream(cur)
                    with torch.cuda.stream(ss):
                        self._graphs[(idx, shard)].replay()
                for ss in self._streams:
                    cur.wait_stream(ss)
        else:
            for shard in range(nsh):
                self._rollout_body(buf, actor, shard)
        self.buf = buf
        return buf
