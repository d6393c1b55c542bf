"""HipVectorEnv — thousands of simulated clusters stepping on one MI355X.

Same VectorEnv protocol as CpuVectorEnv (gym_api.py) but fully on-device:
state lives in torch CUDA tensors, step() is one launch of the CDNA4 kernel
(csrc/env_step.hip), auto-reset happens in-kernel. No host sync per step;
outputs are device tensors (views of persistent buffers unless per-step
out tensors are passed, e.g. rollout slices).
"""
from __future__ import annotations

import numpy as np
import torch

from ..ops.envstep import EnvStepBinding
from ..traces.synthetic import JobTrace, make_pool
from ..utils.spaces import Box, Discrete
from .spec import EnvSpec


class HipVectorEnv:
    def __init__(self, spec: EnvSpec, num_envs: int, trace_kind: str = "philly",
                 jobs_per_env: int = 128, seed: int = 0,
                 device: str = "cuda", pool: JobTrace | None = None,
                 job_lo: np.ndarray | None = None,
                 job_hi: np.ndarray | None = None, debug_skip: int = 0,
                 **trace_kw):
        if not torch.cuda.is_available():
            raise RuntimeError("HipVectorEnv needs a GPU (use CpuVectorEnv here)")
        assert spec.queue_visible <= 63, "kernel assumes K <= 63 (one lane/slot)"
        assert spec.max_span <= 8, "RLS_MAX_SPAN in env_common.h is 8"
        assert spec.num_nodes <= 128, "RLS_MAX_NODES in env_step.hip is 128"
        self.spec = spec
        self.num_envs = E = num_envs
        self.device = torch.device(device)
        if pool is None:
            pool, job_lo, job_hi = make_pool(
                trace_kind, E, jobs_per_env, seed, spec.max_job_gpus, **trace_kw)
        self.pool = pool
        dev = self.device
        N, P, R, S = spec.num_nodes, spec.queue_cap, spec.run_cap, spec.max_span
        A, OD = spec.num_actions, spec.obs_dim

        t = lambda arr: torch.from_numpy(np.ascontiguousarray(arr)).to(dev)
        z = lambda shape, dt: torch.zeros(shape, dtype=dt, device=dev)
        self.b = {
            "trace_arr": t(pool.arrival_us),
            "trace_gpus": t(pool.gpus),
            "trace_work": t(pool.work_us),
            "trace_ming": t(pool.min_gpus),
            "job_lo": t(job_lo), "job_hi": t(job_hi),
            "base": t(pool.arrival_us[job_lo]),
            "clock": z(E, torch.int64), "cursor": z(E, torch.int64),
            "node_free": z((E, N), torch.int32),
            "pend_job": z((E, P), torch.int32),
            "pend_override": z((E, P), torch.int64),
            "pend_count": z(E, torch.int32),
            "run_valid": z((E, R), torch.int32),
            "run_job": z((E, R), torch.int32),
            "run_gpus": z((E, R), torch.int32),
            "run_finish": z((E, R), torch.int64),
            "run_start": z((E, R), torch.int64),
            "run_alloc": z((E, R, S), torch.int32),
            "run_count": z(E, torch.int32),
            "decisions": z(E, torch.int32),
            "completed": z(E, torch.int32),
            "jct_sum": z(E, torch.int64),
            "reward_sum": z(E, torch.float32),
            "actions": z(E, torch.int64),
            "obs": z((E, OD), torch.float32),
            "rewards": z(E, torch.float32),
            "done": z(E, torch.uint8),
            "mask": z((E, A), torch.uint8),
            "ep_count": z(E, torch.int32),
            "ep_return": z(E, torch.float32),
            "ep_jct_sum": z(E, torch.int64),
            "ep_completed": z(E, torch.int64),
            "ep_decisions": z(E, torch.int64),
        }
        params = np.array([
            E, N, spec.gpus_per_node, spec.queue_visible, P, R, S, A, OD,
            spec.idle_tick_us, spec.time_scale_us, spec.horizon,
            1 if spec.gandiva else 0,
            1 if spec.elastic else 0,
            int(debug_skip),
            int(spec.obs_version),
            1 if spec.elastic_resize else 0,
        ], dtype=np.int64)
        self.binding = EnvStepBinding(
            params, {k: v.data_ptr() for k, v in self.b.items()})
        self.observation_space = Box(-np.inf, np.inf, (OD,), np.float32)
        self.action_space = Discrete(A)

    def _stream(self) -> int:
        return torch.cuda.current_stream(self.device).cuda_stream

    def reset(self) -> torch.Tensor:
        self.binding.reset(self._stream())
        return self.b["obs"]

    def step(self, actions: torch.Tensor, obs_out: torch.Tensor | None = None,
             rew_out: torch.Tensor | None = None,
             done_out: torch.Tensor | None = None,
             mask_out: torch.Tensor | None = None):
        """One batched transition. `actions` int64 [E] on device.

        Optional out tensors redirect kernel outputs (e.g. rollout-buffer
        slices) — they must be contiguous CUDA tensors of the right shape.
        """
        ov = {}
        if actions.data_ptr() != self.b["actions"].data_ptr():
            ov["actions"] = actions.contiguous().data_ptr()
        if obs_out is not None:
            ov["obs"] = obs_out.data_ptr()
        if rew_out is not None:
            ov["rewards"] = rew_out.data_ptr()
        if done_out is not None:
            ov["done"] = done_out.data_ptr()
        if mask_out is not None:
            ov["mask"] = mask_out.data_ptr()
        self.binding.step(self._stream(), **ov)
        obs = obs_out if obs_out is not None else self.b["obs"]
        rew = rew_out if rew_out is not None else self.b["rewards"]
        done = done_out if done_out is not None else self.b["done"]
        mask = mask_out if mask_out is not None else self.b["mask"]
        return obs, rew, done, {"action_mask": mask}

    def action_masks(self) -> torch.Tensor:
        return self.b["mask"]

    # -- episode statistics (host sync) ------------------------------------
    def episode_stats(self) -> dict:
        c = int(self.b["ep_count"].sum().item())
        comp = int(self.b["ep_completed"].sum().item())
        return {
            "episodes": c,
            "mean_return": (self.b["ep_return"].sum().item() / c) if c else 0.0,
            "mean_jct_us": (self.b["ep_jct_sum"].sum().item() / comp) if comp else 0.0,
            "completed": comp,
            "decisions": int(self.b["ep_decisions"].sum().item()),
        }

    def state_fingerprints(self) -> list[dict]:
        """Host copies of integer state for differential tests (syncs)."""
        h = {k: self.b[k].cpu().numpy() for k in
             ("clock", "cursor", "pend_count", "run_count", "node_free",
              "pend_job", "completed", "jct_sum", "decisions")}
        out = []
        for e in range(self.num_envs):
            pc = int(h["pend_count"][e])
            out.append({
                "clock": int(h["clock"][e]),
                "cursor": int(h["cursor"][e]),
                "pend_count": pc,
                "run_count": int(h["run_count"][e]),
                "node_free": h["node_free"][e].copy(),
                "pend_job": h["pend_job"][e][:pc].astype(np.int64),
                "completed": int(h["completed"][e]),
                "jct_sum_us": int(h["jct_sum"][e]),
                "decisions": int(h["decisions"][e]),
            })
        return out

    def validate_state(self) -> None:
        """Invariant checker (a data-race in the wave-lockstep kernel or a
        bad auto-reset shows up here): GPU-accounting consistency across
        the whole batch. Host-syncs; call from tests/debug only."""
        b, s = self.b, self.spec
        nf = b["node_free"].cpu()
        assert (nf >= 0).all() and (nf <= s.gpus_per_node).all(), "node_free out of range"
        rv = (b["run_valid"].cpu() != 0)
        rc = b["run_count"].cpu()
        assert (rv.sum(1).to(rc.dtype) == rc).all(), "run_count != #valid slots"
        pc = b["pend_count"].cpu()
        assert (pc >= 0).all() and (pc <= s.queue_cap).all(), "pend_count out of range"
        alloc = (b["run_alloc"].cpu() & 0xFFFF) * rv.unsqueeze(-1)
        used = alloc.sum((1, 2))
        free_tot = nf.sum(1)
        assert (used + free_tot == s.total_gpus).all(), "GPU accounting leak"
        galloc = b["run_gpus"].cpu() * rv
        assert (galloc.sum(1) == used).all(), "alloc pairs != allocated gpus"
