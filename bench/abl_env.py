"""Env-kernel phase ablation (GPU): attribute the per-step cost.

debug_skip bits: 1=per-slot obs/fits, 2=whole obs/mask phase,
4=obs[3] next_event. Random mask-valid actions via a cheap proxy (mix of
schedule-0 and noop) so placement/advance paths both run.
"""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from rlgpuschedule_amd.env.hip_env import HipVectorEnv  # noqa: E402
from rlgpuschedule_amd.env.spec import cluster64  # noqa: E402


def run(skip, E=8192, steps=300):
    spec = cluster64(horizon=1_000_000)
    env = HipVectorEnv(spec, E, trace_kind="philly", jobs_per_env=256,
                       seed=1, debug_skip=skip)
    env.reset()
    g = torch.Generator(device="cuda").manual_seed(0)
    # 50/50 schedule-slot-0 / noop (slot0 invalid -> degrades to noop)
    acts = torch.where(torch.rand(E, device="cuda", generator=g) < 0.5,
                       torch.zeros(E, dtype=torch.int64, device="cuda"),
                       torch.full((E,), spec.noop_action, dtype=torch.int64,
                                  device="cuda"))
    for _ in range(20):
        env.step(acts)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        env.step(acts)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1e6


def run_env(env, acts, steps=200):
    import time
    for _ in range(20):
        env.step(acts)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        env.step(acts)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1e6


if __name__ == "__main__":
    # within-probe interleaved A/B: shared envs, repeated rounds
    E = 8192
    spec = cluster64(horizon=1_000_000)
    variants = [(0, "full"), (1, "no slot obs/fits"), (4, "no next_event"),
                (5, "neither"), (2, "no obs/mask")]
    envs = {s: HipVectorEnv(spec, E, trace_kind="philly", jobs_per_env=256,
                            seed=1, debug_skip=s) for s, _ in variants}
    g = torch.Generator(device="cuda").manual_seed(0)
    acts = torch.where(torch.rand(E, device="cuda", generator=g) < 0.5,
                       torch.zeros(E, dtype=torch.int64, device="cuda"),
                       torch.full((E,), spec.noop_action, dtype=torch.int64,
                                  device="cuda"))
    for e in envs.values():
        e.reset()
    import collections
    res = collections.defaultdict(list)
    for rnd in range(3):
        for s, name in variants:
            res[s].append(run_env(envs[s], acts))
    for s, name in variants:
        v = sorted(res[s])
        print(f"skip={s} ({name}): median {v[1]:7.1f} us  all={[round(x,1) for x in v]}")
