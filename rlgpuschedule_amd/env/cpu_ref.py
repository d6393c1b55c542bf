"""Pure-Python reference implementation of the cluster-scheduling env.

This is the ORACLE: the batched HIP env (csrc/env_step.hip) is differential-
tested against this implementation for bit-exact integer state and
atol=1e-5 observations. Semantics: docs/ENV_SPEC.md. Keep this file simple
and obviously-correct rather than fast; the fast paths are the HIP kernel
and (for CPU-only configs) the vectorized numpy env.
"""
from __future__ import annotations

import math

import numpy as np

from ..traces.synthetic import JobTrace
from .spec import EnvSpec

NO_EVENT = np.int64(0x7FFFFFFFFFFFFFFF)


def _lognorm(x_us: int) -> np.float32:
    # float32 math end-to-end, mirrored by the HIP kernel (log1pf).
    return np.float32(math.log1p(np.float32(x_us) / np.float32(1e6)) / np.float32(10.0))


class ClusterEnvRef:
    """Single-environment reference implementation (integer-tick DES)."""

    def __init__(self, spec: EnvSpec, trace: JobTrace, job_lo: int = 0,
                 job_hi: int | None = None):
        self.spec = spec
        self.trace = trace
        self.job_lo = int(job_lo)
        self.job_hi = int(len(trace) if job_hi is None else job_hi)
        assert self.job_lo < self.job_hi <= len(trace)
        trace.validate(spec.max_job_gpus, self.job_lo, self.job_hi)
        self.base = int(trace.arrival_us[self.job_lo])
        self.reset()

    # -- helpers -----------------------------------------------------------
    def _arrival(self, j: int) -> int:
        return int(self.trace.arrival_us[j]) - self.base

    def _rem_gpuus(self, j: int, pend_pos: int | None = None) -> int:
        """Remaining work in GPU-us (ENV_SPEC v3; exact for rigid jobs)."""
        if pend_pos is not None and self.pend_override[pend_pos] > 0:
            return int(self.pend_override[pend_pos])
        return int(self.trace.work_us[j]) * int(self.trace.gpus[j])

    def _pend_work_us(self, pos: int) -> int:
        """Obs work feature: remaining wall-time at max demand."""
        j = int(self.pend_job[pos])
        o = int(self.pend_override[pos])
        g = int(self.trace.gpus[j])
        if o > 0:
            return -(-o // g)   # ceil
        return int(self.trace.work_us[j])

    def first_fit(self, g: int) -> list[tuple[int, int]] | None:
        """Deterministic first-fit: returns alloc pairs or None."""
        pairs: list[tuple[int, int]] = []
        left = g
        for n in range(self.spec.num_nodes):
            free = int(self.node_free[n])
            if free > 0:
                take = min(free, left)
                if len(pairs) == self.spec.max_span:
                    return None
                pairs.append((n, take))
                left -= take
                if left == 0:
                    return pairs
        return None

    def fits_cap(self) -> int:
        """Largest g for which first_fit(g) succeeds = the free capacity
        of the FIRST max_span nonzero-free nodes in index order
        (first-fit always consumes exactly that node prefix, so
        first_fit(g) is not None  <=>  g <= fits_cap() for g >= 1).
        Lets the K per-slot fits bits cost one O(N) pass instead of K."""
        cap = 0
        used = 0
        for n in range(self.spec.num_nodes):
            f = int(self.node_free[n])
            if f > 0:
                cap += f
                used += 1
                if used == self.spec.max_span:
                    break
        return cap

    def _next_event(self) -> int:
        t = NO_EVENT
        for r in range(self.spec.run_cap):
            if self.run_valid[r] and self.run_finish[r] < t:
                t = self.run_finish[r]
        if self.cursor < self.job_hi:
            t = min(t, self._arrival(self.cursor))
        return int(t)

    def _admit(self) -> None:
        while (self.cursor < self.job_hi
               and self._arrival(self.cursor) <= self.clock
               and self.pend_count < self.spec.queue_cap):
            self.pend_job[self.pend_count] = self.cursor
            self.pend_override[self.pend_count] = 0
            self.pend_count += 1
            self.cursor += 1

    # -- API ----------------------------------------------------------------
    def reset(self) -> np.ndarray:
        s = self.spec
        self.clock = 0
        self.cursor = self.job_lo
        self.node_free = np.full(s.num_nodes, s.gpus_per_node, np.int32)
        self.pend_job = np.zeros(s.queue_cap, np.int64)
        self.pend_override = np.zeros(s.queue_cap, np.int64)  # preempted remaining work
        self.pend_count = 0
        self.run_valid = np.zeros(s.run_cap, bool)
        self.run_job = np.zeros(s.run_cap, np.int64)
        self.run_gpus = np.zeros(s.run_cap, np.int32)
        self.run_finish = np.zeros(s.run_cap, np.int64)
        self.run_start = np.zeros(s.run_cap, np.int64)
        self.run_alloc: list[list[tuple[int, int]]] = [[] for _ in range(s.run_cap)]
        self.run_count = 0
        self.decisions = 0
        self.completed = 0
        self.jct_sum_us = 0
        self.reward_sum = np.float32(0.0)
        self._admit()
        return self.observe()

    def _free_total(self) -> int:
        return int(self.node_free.sum())

    def _try_fit(self, j: int) -> tuple[int, list] | None:
        """Descending-g placement scan (g == min == max for rigid jobs)."""
        g_max = int(self.trace.gpus[j])
        g_min = int(self.trace.min_gpus[j]) if self.spec.elastic else g_max
        for g in range(g_max, g_min - 1, -1):
            pairs = self.first_fit(g)
            if pairs is not None:
                return g, pairs
        return None

    def _place(self, pend_pos: int) -> bool:
        j = int(self.pend_job[pend_pos])
        fit = self._try_fit(j)
        if fit is None:
            return False
        g, pairs = fit
        for n, c in pairs:
            self.node_free[n] -= c
        slot = next(r for r in range(self.spec.run_cap) if not self.run_valid[r])
        rem = self._rem_gpuus(j, pend_pos)
        self.run_valid[slot] = True
        self.run_job[slot] = j
        self.run_gpus[slot] = g   # ALLOCATED gpus (== demand for rigid)
        self.run_start[slot] = self.clock
        self.run_finish[slot] = self.clock + (-(-rem // g))
        self.run_alloc[slot] = pairs
        self.run_count += 1
        # remove pending position
        for p in range(pend_pos, self.pend_count - 1):
            self.pend_job[p] = self.pend_job[p + 1]
            self.pend_override[p] = self.pend_override[p + 1]
        self.pend_count -= 1
        return True

    def _release(self, slot: int) -> None:
        for n, c in self.run_alloc[slot]:
            self.node_free[n] += c
        self.run_valid[slot] = False
        self.run_alloc[slot] = []
        self.run_count -= 1

    def _advance(self) -> np.float32:
        t_next = self._next_event()
        if t_next == NO_EVENT:
            t_next = self.clock + self.spec.idle_tick_us
        dt = t_next - self.clock
        n_sys = self.pend_count + self.run_count
        reward = np.float32(-(np.float32(n_sys) * np.float32(dt))
                            / np.float32(self.spec.time_scale_us))
        self.clock = t_next
        for r in range(self.spec.run_cap):
            if self.run_valid[r] and self.run_finish[r] <= self.clock:
                j = int(self.run_job[r])
                self.completed += 1
                self.jct_sum_us += self.clock - self._arrival(j)
                self._release(r)
        self._admit()
        return reward

    def _migrate(self) -> bool:
        """Gandiva MIGRATE: re-place the widest-span running job. False if
        nothing is running (degrade to NOOP)."""
        if self.run_count == 0:
            return False
        spans = [(len(self.run_alloc[r]), r) for r in range(self.spec.run_cap)
                 if self.run_valid[r]]
        span, slot = max(spans, key=lambda t: (t[0], -t[1]))
        # free, then first-fit re-place (may land identically)
        pairs_old = self.run_alloc[slot]
        for n, c in pairs_old:
            self.node_free[n] += c
        j = int(self.run_job[slot])
        g_old = int(self.run_gpus[slot])
        fit = self._try_fit(j)
        if fit is None:
            # span-constrained first-fit can fail even after freeing (the
            # freed cluster may be more fragmented than at placement time):
            # restore and degrade to NOOP.
            for n, c in pairs_old:
                self.node_free[n] -= c
            return False
        g_new, pairs_new = fit
        for n, c in pairs_new:
            self.node_free[n] -= c
        self.run_alloc[slot] = pairs_new
        if g_new != g_old:  # elastic re-placement changes speed
            rem = max(1, int(self.run_finish[slot]) - self.clock) * g_old
            self.run_gpus[slot] = g_new
            self.run_finish[slot] = self.clock + (-(-rem // g_new))
        return True

    def _timeslice(self) -> bool:
        """Gandiva TIMESLICE: preempt the longest-elapsed running job to the
        pending tail. False if nothing running or pending full."""
        if self.run_count == 0 or self.pend_count >= self.spec.queue_cap:
            return False
        el = [(self.clock - int(self.run_start[r]), -r) for r in range(self.spec.run_cap)
              if self.run_valid[r]]
        _, negr = max(el)
        slot = -negr
        remaining = (max(1, int(self.run_finish[slot]) - self.clock)
                     * int(self.run_gpus[slot]))  # GPU-us (ENV_SPEC v3)
        self.pend_job[self.pend_count] = self.run_job[slot]
        self.pend_override[self.pend_count] = remaining
        self.pend_count += 1
        self._release(slot)
        return True

    def _shrink(self) -> bool:
        """ENV_SPEC v5 SHRINK: halve the widest running elastic job
        (largest allocated g with g//2 >= min_gpus; tie -> lowest slot).
        Frees GPUs for the queue at the cost of slower progress. False
        (degrade to NOOP) if no candidate or the re-placement fails."""
        best_g, slot = -1, -1
        for r in range(self.spec.run_cap):
            if not self.run_valid[r]:
                continue
            g = int(self.run_gpus[r])
            j = int(self.run_job[r])
            if g >= 2 and g // 2 >= int(self.trace.min_gpus[j]):
                if g > best_g:
                    best_g, slot = g, r
        if slot < 0:
            return False
        g_old = best_g
        g_new = g_old // 2
        rem = max(1, int(self.run_finish[slot]) - self.clock) * g_old
        pairs_old = self.run_alloc[slot]
        for n, c in pairs_old:
            self.node_free[n] += c
        pairs_new = self.first_fit(g_new)
        if pairs_new is None:   # span-constrained fit can fail: restore
            for n, c in pairs_old:
                self.node_free[n] -= c
            return False
        for n, c in pairs_new:
            self.node_free[n] -= c
        self.run_alloc[slot] = pairs_new
        self.run_gpus[slot] = g_new
        self.run_finish[slot] = self.clock + (-(-rem // g_new))
        return True

    def _expand(self) -> bool:
        """ENV_SPEC v5 EXPAND: grow the most-shrunk running job (largest
        demand deficit gpus - g; tie -> lowest slot) toward its demand,
        doubling at most (g_target = min(gpus, 2g)), taking the largest
        g' in (g, g_target] that fits. False if no candidate / no fit."""
        best_d, slot = 0, -1
        for r in range(self.spec.run_cap):
            if not self.run_valid[r]:
                continue
            d = int(self.trace.gpus[int(self.run_job[r])]) - int(self.run_gpus[r])
            if d > best_d:
                best_d, slot = d, r
        if slot < 0:
            return False
        g_old = int(self.run_gpus[slot])
        j = int(self.run_job[slot])
        g_target = min(int(self.trace.gpus[j]), 2 * g_old)
        rem = max(1, int(self.run_finish[slot]) - self.clock) * g_old
        pairs_old = self.run_alloc[slot]
        for n, c in pairs_old:
            self.node_free[n] += c
        g_new, pairs_new = -1, None
        for gt in range(g_target, g_old, -1):
            pairs_new = self.first_fit(gt)
            if pairs_new is not None:
                g_new = gt
                break
        if g_new < 0:
            for n, c in pairs_old:
                self.node_free[n] -= c
            return False
        for n, c in pairs_new:
            self.node_free[n] -= c
        self.run_alloc[slot] = pairs_new
        self.run_gpus[slot] = g_new
        self.run_finish[slot] = self.clock + (-(-rem // g_new))
        return True

    def step(self, action: int) -> tuple[np.ndarray, np.float32, bool, dict]:
        s = self.spec
        reward = np.float32(0.0)
        a = int(action)
        did = False
        if a < s.queue_visible:
            if a < self.pend_count:
                did = self._place(a)
        elif s.gandiva and a == s.migrate_action:
            did = self._migrate()
        elif s.gandiva and a == s.timeslice_action:
            did = self._timeslice()
        elif s.elastic_resize and a == s.shrink_action:
            did = self._shrink()
        elif s.elastic_resize and a == s.expand_action:
            did = self._expand()
        if not did:
            reward = self._advance()
        self.decisions += 1
        self.reward_sum = np.float32(self.reward_sum + reward)
        done = ((self.cursor == self.job_hi and self.pend_count == 0
                 and self.run_count == 0)
                or self.decisions >= s.horizon)
        info = {"action_mask": self.action_mask()}
        if done:
            info["episode"] = {
                "return": float(self.reward_sum),
                "decisions": self.decisions,
                "completed": self.completed,
                "mean_jct_us": (self.jct_sum_us / self.completed
                                if self.completed else 0.0),
            }
        return self.observe(), reward, done, info

    def action_mask(self) -> np.ndarray:
        s = self.spec
        m = np.zeros(s.num_actions, bool)
        cap = self.fits_cap()
        for k in range(min(s.queue_visible, self.pend_count)):
            j = int(self.pend_job[k])
            g_fit = (int(self.trace.min_gpus[j]) if s.elastic
                     else int(self.trace.gpus[j]))
            m[k] = g_fit <= cap
        m[s.noop_action] = True
        if s.gandiva:
            m[s.migrate_action] = self.run_count > 0
            m[s.timeslice_action] = (self.run_count > 0
                                     and self.pend_count < s.queue_cap)
        if s.elastic_resize:
            can_s = can_e = False
            for r in range(s.run_cap):
                if not self.run_valid[r]:
                    continue
                g = int(self.run_gpus[r])
                j = int(self.run_job[r])
                if g >= 2 and g // 2 >= int(self.trace.min_gpus[j]):
                    can_s = True
                if g < int(self.trace.gpus[j]):
                    can_e = True
            m[s.shrink_action] = can_s
            m[s.expand_action] = can_e
        return m

    def observe(self) -> np.ndarray:
        s = self.spec
        obs = np.zeros(s.obs_dim, np.float32)
        C = s.total_gpus
        obs[0] = np.float32(self._free_total()) / np.float32(C)
        obs[1] = np.float32(self.pend_count) / np.float32(s.queue_cap)
        obs[2] = np.float32(self.run_count) / np.float32(s.run_cap)
        t_next = self._next_event()
        obs[3] = _lognorm(t_next - self.clock) if t_next != NO_EVENT else np.float32(0)
        cap = self.fits_cap()
        for k in range(min(s.queue_visible, self.pend_count)):
            j = int(self.pend_job[k])
            g = int(self.trace.gpus[j])
            g_fit = int(self.trace.min_gpus[j]) if s.elastic else g
            b = 4 + 5 * k
            obs[b + 0] = 1.0
            obs[b + 1] = np.float32(g) / np.float32(C)
            obs[b + 2] = _lognorm(self.clock - self._arrival(j))
            obs[b + 3] = _lognorm(self._pend_work_us(k))
            obs[b + 4] = np.float32(1.0 if g_fit <= cap else 0.0)
        if s.obs_version >= 2:
            e = 4 + 5 * s.queue_visible
            gpn = s.gpus_per_node
            for n in range(s.num_nodes):
                idx = min(7, (int(self.node_free[n]) * 8) // gpn)
                obs[e + idx] += np.float32(1.0) / np.float32(s.num_nodes)
            obs[e + 8] = (np.float32(max(0, self.pend_count - s.queue_visible))
                          / np.float32(s.queue_cap))
            obs[e + 9] = (np.float32(int(self.node_free.max()))
                          / np.float32(gpn))
            span = max(1, self.job_hi - self.job_lo)
            obs[e + 10] = np.float32(self.cursor - self.job_lo) / np.float32(span)
            obs[e + 11] = np.float32(self.decisions) / np.float32(s.horizon)
        return obs

    # integer-state snapshot for differential tests against the HIP env
    def state_fingerprint(self) -> dict:
        return {
            "clock": self.clock,
            "cursor": self.cursor,
            "pend_count": self.pend_count,
            "run_count": self.run_count,
            "node_free": self.node_free.copy(),
            "pend_job": self.pend_job[: self.pend_count].copy(),
            "completed": self.completed,
            "jct_sum_us": self.jct_sum_us,
            "decisions": self.decisions,
        }

    def validate_invariants(self) -> None:
        """Accounting invariants (mirrors HipVectorEnv.validate_state)."""
        s = self.spec
        assert (self.node_free >= 0).all(), "negative node_free"
        assert (self.node_free <= s.gpus_per_node).all(), "node_free > G"
        assert self.run_count == int(self.run_valid.sum()), "run_count drift"
        assert 0 <= self.pend_count <= s.queue_cap
        used = sum(c for r in range(s.run_cap) if self.run_valid[r]
                   for _, c in self.run_alloc[r])
        assert used + int(self.node_free.sum()) == s.total_gpus, "GPU leak"
        galloc = sum(int(self.run_gpus[r]) for r in range(s.run_cap)
                     if self.run_valid[r])
        assert galloc == used, "alloc pairs != allocated gpus"
        for r in range(s.run_cap):
            if self.run_valid[r]:
                assert self.run_finish[r] > self.clock or \
                    self.run_finish[r] >= self.clock, "stale running job"
