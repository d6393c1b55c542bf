"""End-to-end training driver wiring env + policy + learner + DP.

The five named configs mirror BASELINE.json's config list.
"""
from __future__ import annotations

from dataclasses import dataclass, field

import torch

from .agents.ppo import A2CConfig, A2CLearner, PPOConfig, PPOLearner
from .agents.rollout import Collector, TorchEnvAdapter
from .env.gym_api import CpuVectorEnv
from .env.spec import EnvSpec, cluster16, cluster64, cluster256, cluster1024
from .models.policy import GRUPolicy, MLPPolicy
from .parallel.dist import FlatGradAllreduce, broadcast_params, init_from_env


@dataclass
class TrainConfig:
    name: str = "c2_ppo_mlp"
    spec: EnvSpec = field(default_factory=cluster64)
    num_envs: int = 1024          # per GPU
    rollout_T: int = 64
    algo: str = "ppo"             # ppo | a2c | impala
    policy: str = "mlp"           # mlp | gru
    hidden: int = 256
    trace_kind: str = "philly"
    jobs_per_env: int = 256
    seed: int = 0
    iterations: int = 100
    device: str = "auto"
    ppo: PPOConfig = field(default_factory=PPOConfig)
    a2c: A2CConfig = field(default_factory=A2CConfig)
    use_hip_env: bool = True      # False -> CPU reference env
    trace_kw: dict = field(default_factory=dict)
    actor_dtype: str = "bf16"     # rollout policy compute (fp32 | bf16)
    update_dtype: str = "fp32"    # learner GEMMs (fp32 exact | bf16 mixed)
    # SJF behavior-cloning warm start (agents/bc.py; round-2 quality
    # recipe): 0 = off; >0 = clone SJF from this many rollouts before RL
    bc_rollouts: int = 0
    bc_epochs: int = 4
    bc_lr: float = 1e-3
    bc_ent: float = 0.03          # soft-clone entropy (keeps exploration)


CONFIGS = {
    # 16-GPU simulated cluster, random-policy agent on CPU Gym env
    "c1_cpu_random": TrainConfig(
        name="c1_cpu_random", spec=cluster16(), num_envs=8, rollout_T=32,
        algo="random", policy="mlp", use_hip_env=False, jobs_per_env=64),
    # 64-GPU Philly-trace sim, PPO MLP, 256 envs, 1 GPU
    "c2_ppo_mlp": TrainConfig(
        name="c2_ppo_mlp", spec=cluster64(), num_envs=256, rollout_T=64,
        algo="ppo", policy="mlp"),
    # 256-GPU sim, PPO GRU, 2048 envs, DP=8
    "c3_ppo_gru": TrainConfig(
        name="c3_ppo_gru", spec=cluster256(), num_envs=2048, rollout_T=64,
        algo="ppo", policy="gru"),
    # Gandiva actions + elastic jobs, A2C
    "c4_gandiva_a2c": TrainConfig(
        name="c4_gandiva_a2c", spec=cluster64(gandiva=True, elastic=True,
                       elastic_resize=True),
        num_envs=1024, rollout_T=64, algo="a2c", policy="mlp",
        trace_kw={"elastic": True, "fail_frac": 0.1}),
    # 1024-GPU Poisson trace, IMPALA-style async actor/learner
    "c5_impala": TrainConfig(
        name="c5_impala", spec=cluster1024(), num_envs=2048, rollout_T=64,
        algo="impala", policy="mlp", trace_kind="poisson"),
}


def resolve_device(cfg: TrainConfig) -> torch.device:
    if cfg.device != "auto":
        return torch.device(cfg.device)
    return torch.device("cuda" if torch.cuda.is_available() else "cpu")


def make_env(cfg: TrainConfig, device: torch.device, rank: int = 0):
    seed = cfg.seed * 7919 + rank
    if cfg.use_hip_env and device.type == "cuda":
        from .env.hip_env import HipVectorEnv
        return HipVectorEnv(cfg.spec, cfg.num_envs, trace_kind=cfg.trace_kind,
                            jobs_per_env=cfg.jobs_per_env, seed=seed,
                            **cfg.trace_kw)
    venv = CpuVectorEnv(cfg.spec, cfg.num_envs, trace_kind=cfg.trace_kind,
                        jobs_per_env=cfg.jobs_per_env, seed=seed,
                        **cfg.trace_kw)
    return TorchEnvAdapter(venv, device)


def make_policy(cfg: TrainConfig, device: torch.device):
    cls = GRUPolicy if cfg.policy == "gru" else MLPPolicy
    pol = cls(cfg.spec.obs_dim, cfg.spec.num_actions, cfg.hidden)
    return pol.to(device)


def make_learner(cfg: TrainConfig, policy, grad_sync=None):
    if cfg.algo == "ppo":
        return PPOLearner(policy, cfg.ppo, grad_sync)
    if cfg.algo == "a2c":
        return A2CLearner(policy, cfg.a2c, grad_sync)
    raise ValueError(f"unknown algo {cfg.algo}")


def train(cfg: TrainConfig, iterations: int | None = None,
          log_fn=print, out_dir: str | None = None, save_every: int = 0,
          resume: str | None = None, callback=None,
          solo: bool = False) -> dict:
    """Synchronous trainer (ppo/a2c). IMPALA lives in agents/impala.py.
    callback(it, policy, stats), if given, runs on rank 0 after every
    update (it must not mutate the policy) — used by agents/selection.py
    for eval-based checkpoint selection.
    solo=True trains independently even inside an initialized process
    group (no param broadcast, no grad all-reduce) — used by the striped
    multi-seed sweep where each rank trains DIFFERENT seeds and matched
    collective sequences across ranks cannot be guaranteed."""
    rank, world, local = init_from_env()
    if solo:
        rank, world = 0, 1
    device = resolve_device(cfg)
    if device.type == "cuda":
        device = torch.device(f"cuda:{local}")
    torch.manual_seed(cfg.seed * 31 + rank)
    if cfg.algo == "impala":
        if callback is not None or solo:
            raise ValueError("algo='impala' does not support callback/solo "
                             "(selection pipelines run on the sync trainer)")
        from .agents.impala import train_impala
        return train_impala(cfg, iterations=iterations, log_fn=log_fn,
                            out_dir=out_dir, save_every=save_every,
                            resume=resume)
    env = make_env(cfg, device, rank)
    policy = make_policy(cfg, device)
    if not solo:
        broadcast_params(policy)
    if cfg.bc_rollouts and cfg.algo in ("ppo", "a2c"):
        # SJF warm start on a separate env pool; resume (below) overrides
        from dataclasses import replace as _replace

        from .agents.bc import pretrain_bc
        bc_env = make_env(_replace(cfg, num_envs=min(cfg.num_envs, 1024)),
                          device, rank + 977)
        if hasattr(bc_env, "venv"):
            bc_env = bc_env.venv   # bc drives CpuVectorEnv directly
        pretrain_bc(policy, bc_env, T=cfg.rollout_T,
                    n_rollouts=cfg.bc_rollouts, epochs=cfg.bc_epochs,
                    lr=cfg.bc_lr, ent_coef=cfg.bc_ent, device=device)
        if not solo:
            broadcast_params(policy)   # ranks cloned from different data
    grad_sync = FlatGradAllreduce(policy) if world > 1 else None
    on_gpu_fast = device.type == "cuda" and cfg.use_hip_env
    if on_gpu_fast:
        from .agents.rollout_gpu import GpuRolloutEngine
        collector = GpuRolloutEngine(env, policy, cfg.rollout_T,
                                     seed=cfg.seed * 101 + rank,
                                     actor_dtype=cfg.actor_dtype)
    else:
        collector = Collector(env, policy, cfg.rollout_T, device,
                              seed=cfg.seed * 101 + rank)
    if cfg.algo == "random":
        learner = None
    elif (on_gpu_fast and cfg.algo == "ppo" and not policy.is_recurrent):
        from .agents.ppo_hip import HipPPOLearner
        learner = HipPPOLearner(policy, cfg.ppo, grad_sync,
                                dtype=cfg.update_dtype)
    elif on_gpu_fast and cfg.algo == "ppo" and policy.is_recurrent:
        from .agents.ppo_gru_hip import HipGRUPPOLearner
        learner = HipGRUPPOLearner(policy, cfg.ppo, grad_sync,
                                   dtype=cfg.update_dtype)
    elif (on_gpu_fast and cfg.algo == "a2c" and not policy.is_recurrent):
        from .agents.a2c_hip import HipA2CLearner
        learner = HipA2CLearner(policy, cfg.a2c, grad_sync,
                                dtype=cfg.update_dtype)
    else:
        learner = make_learner(cfg, policy, grad_sync)
    start_it = 0
    if resume:
        from .ckpt.io import apply_optim_state, load_policy
        # Do NOT pass the optimizer here: the HIP learners create HipAdam
        # lazily in _alloc, so the state is routed via load_optim_state
        # and applied once the real optimizer exists.
        meta = load_policy(resume, policy, map_location=str(device))
        apply_optim_state(learner, meta.get("optim_state"))
        start_it = int(meta.get("meta", {}).get("iteration", 0))
        if rank == 0 and log_fn:
            log_fn(f"[{cfg.name}] resumed from {resume} at iter {start_it}")
    logger = None
    if out_dir is not None and rank == 0:
        from .metrics.logger import MetricsLogger
        logger = MetricsLogger(out_dir, echo=False,
                               env_steps_per_iter=cfg.num_envs * cfg.rollout_T)
    iters = iterations if iterations is not None else cfg.iterations
    stats = {}
    from .metrics.logger import check_finite
    if (callback is not None and rank == 0 and cfg.bc_rollouts
            and start_it == 0):
        callback(-1, policy, {})   # BC warm start as a selection candidate
    for it in range(start_it, start_it + iters):
        buf = collector.collect()
        if learner is not None:
            stats = learner.update(buf)
            check_finite(stats, where=f"iter {it}")
        if callback is not None and rank == 0:
            callback(it, policy, stats)
        if logger is not None:
            logger.log(stats, iteration=it)
        if rank == 0 and log_fn and (it % 10 == 0 or it == start_it + iters - 1):
            log_fn(f"[{cfg.name}] iter {it}: " +
                   " ".join(f"{k}={v:.4f}" for k, v in stats.items()))
        if (save_every and rank == 0 and out_dir is not None
                and learner is not None
                and ((it + 1) % save_every == 0 or it == start_it + iters - 1)):
            from .ckpt.io import save_policy
            save_policy(f"{out_dir}/policy_{it + 1:06d}.pt", policy,
                        spec=cfg.spec, optimizer=getattr(learner, "opt", None),
                        meta={"iteration": it + 1, "config": cfg.name})
    out = dict(stats)
    out["_policy"] = policy   # for selection/eval pipelines; not serialized
    if hasattr(env, "episode_stats"):
        out["episode_stats"] = env.episode_stats()
    elif hasattr(env, "venv"):
        st = env.venv.stats
        out["episode_stats"] = {
            "episodes": st.count, "mean_return": st.mean_return(),
            "mean_jct_us": st.mean_jct_us(), "completed": st.completed_sum,
        }
    return out
