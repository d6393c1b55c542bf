"""IMPALA-style async actor/learner with V-trace (BASELINE config 5).

MI355X-native design: no RPC framework. On each GPU, the ACTOR is the
graph-replayed rollout engine running on its own HIP stream with a stale
behavior-policy mirror (HipActorMLP, refreshed every `behavior_sync`
updates), double-buffered; the LEARNER consumes the previously-completed
buffer on the default stream, correcting off-policyness with the V-trace
HIP scan (csrc/nn_kernels.hip). Events order buffer reuse. Across the 8
GPUs of a node, learners are data-parallel over RCCL (FlatGradAllreduce).

CPU fallback (tests): sequential actor/learner with an explicitly stale
policy copy, torch V-trace reference.
"""
from __future__ import annotations

import copy
from dataclasses import dataclass

import torch

from ..models.policy import evaluate_actions
from ..ops.gae_ref import vtrace_reference
from .rollout import Collector, RolloutBuffer


@dataclass
class ImpalaConfig:
    lr: float = 3e-4
    gamma: float = 0.99
    lam: float = 1.0
    rho_bar: float = 1.0
    c_bar: float = 1.0
    vf_coef: float = 0.5
    ent_coef: float = 0.01
    max_grad_norm: float = 0.5
    behavior_sync: int = 1   # refresh behavior policy every N updates
    norm_reward: bool = True
    # Ablation switch: False disables the V-trace
    # off-policy correction — targets become plain GAE computed as if the
    # rollout were on-policy (no rho/c importance clipping). Used by
    # bench/impala_staleness.py to show the correction earns its keep as
    # behavior_sync (staleness) grows.
    use_vtrace: bool = True
    # Ablation aid: run the torch VtraceLearner even where the HIP one
    # would be picked, so both ablation arms share one implementation.
    force_torch_learner: bool = False


class VtraceLearner:
    """Computes the IMPALA loss on a rollout collected by a (possibly
    stale) behavior policy whose log-probs are stored in buf.logp."""

    def __init__(self, policy, cfg: ImpalaConfig, grad_sync=None):
        self.policy = policy
        self.cfg = cfg
        self.grad_sync = grad_sync
        self.opt = torch.optim.Adam(policy.parameters(), lr=cfg.lr, eps=1e-5)

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
        T, E = buf.reward.shape
        B = T * E
        logits, values, _ = self.policy(buf.obs.reshape(B, -1),
                                        buf.mask.reshape(B, -1), None)
        lp_cur, entropy = evaluate_actions(logits, buf.actions.reshape(B))
        values = values.reshape(T, E)
        lp_cur = lp_cur.reshape(T, E)
        with torch.no_grad():
            v_det = values.detach()
            if not cfg.use_vtrace:
                # ablation: pretend the rollout is on-policy (plain GAE)
                from ..ops.gae_ref import gae_advantages
                pg_adv, ret = gae_advantages(
                    buf.reward, v_det, buf.done, buf.last_value,
                    cfg.gamma, cfg.lam)
                vs = ret
            elif buf.reward.is_cuda:
                from ..ops.nn import vtrace_scan
                vs, pg_adv = vtrace_scan(
                    buf.reward, v_det, buf.done, buf.last_value,
                    lp_cur.detach(), buf.logp, cfg.gamma, cfg.rho_bar,
                    cfg.c_bar, cfg.lam)
            else:
                vs, pg_adv = vtrace_reference(
                    buf.reward, v_det, buf.done, buf.last_value,
                    lp_cur.detach(), buf.logp, cfg.gamma, cfg.rho_bar,
                    cfg.c_bar, cfg.lam)
        pi_loss = -(pg_adv * lp_cur).mean()
        v_loss = 0.5 * ((values - vs) ** 2).mean()
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


def train_impala(cfg, iterations=None, log_fn=print, out_dir=None,
                 save_every=0, resume=None, icfg: ImpalaConfig | None = None
                 ) -> dict:
    """Async actor/learner loop (cfg is a trainer.TrainConfig).

    out_dir/save_every/resume mirror trainer.train() so the c5_impala
    config gets checkpointing, metrics logging and resume too (ADVICE r1:
    these silently no-op'd before)."""
    from ..parallel.dist import (FlatGradAllreduce, broadcast_params,
                                 init_from_env)
    from ..trainer import make_env, make_policy, resolve_device
    rank, world, local = init_from_env()
    device = resolve_device(cfg)
    if device.type == "cuda":
        device = torch.device(f"cuda:{local}")
    torch.manual_seed(cfg.seed * 31 + rank)
    icfg = icfg if icfg is not None else ImpalaConfig()
    env = make_env(cfg, device, rank)
    policy = make_policy(cfg, device)
    broadcast_params(policy)
    grad_sync = FlatGradAllreduce(policy) if world > 1 else None
    if (device.type == "cuda" and not policy.is_recurrent
            and cfg.use_hip_env and icfg.use_vtrace
            and not icfg.force_torch_learner):
        from .impala_hip import HipVtraceLearner
        learner = HipVtraceLearner(policy, icfg, grad_sync)
    else:
        # torch learner (also carries the use_vtrace=False ablation path)
        learner = VtraceLearner(policy, icfg, grad_sync)
    start_it = 0
    if resume:
        from ..ckpt.io import apply_optim_state, load_policy
        meta = load_policy(resume, policy, map_location=str(device))
        apply_optim_state(learner, meta.get("optim_state"))
        start_it = int(meta.get("meta", {}).get("iteration", 0))
        if rank == 0 and log_fn:
            log_fn(f"[{cfg.name}] resumed from {resume} at iter {start_it}")
    logger = None
    if out_dir is not None and rank == 0:
        from ..metrics.logger import MetricsLogger
        logger = MetricsLogger(out_dir, echo=False,
                               env_steps_per_iter=cfg.num_envs * cfg.rollout_T)

    def after_update(u, stats, last):
        """Rank-0 bookkeeping after completed update index u (0-based,
        resume-global)."""
        if logger is not None:
            logger.log(stats, iteration=u)
        if (save_every and rank == 0 and out_dir is not None
                and ((u + 1) % save_every == 0 or last)):
            from ..ckpt.io import save_policy
            save_policy(f"{out_dir}/policy_{u + 1:06d}.pt", policy,
                        spec=cfg.spec, optimizer=getattr(learner, "opt", None),
                        meta={"iteration": u + 1, "config": cfg.name})

    iters = iterations if iterations is not None else cfg.iterations
    stats = {}

    if device.type == "cuda":
        from .rollout_gpu import GpuRolloutEngine
        engine = GpuRolloutEngine(env, policy, cfg.rollout_T,
                                  seed=cfg.seed * 101 + rank, num_bufs=2)
        s_act = torch.cuda.Stream(device)
        ev_collect = [torch.cuda.Event() for _ in range(2)]
        ev_update = [torch.cuda.Event() for _ in range(2)]
        ev_theta = [torch.cuda.Event() for _ in range(2)]
        for e in ev_update + ev_theta:
            e.record()  # buffers free; mirrors initialized at engine init
        # Pipeline: the actor fills buf[i%2] with behavior mirror[i%2] on
        # s_act while the learner trains on the other buffer on the default
        # stream. After update j the learner refreshes mirror[(j)%2] (its
        # next reader is rollout j+2) — staleness 1, no cross-stream races.
        for it in range(iters + 1):
            idx = it % 2
            if it < iters:
                with torch.cuda.stream(s_act):
                    cur = torch.cuda.current_stream()
                    cur.wait_event(ev_update[idx])  # buffer free
                    cur.wait_event(ev_theta[idx])   # mirror refreshed
                    engine.collect(idx, refresh=False)
                    ev_collect[idx].record()
            if it > 0:
                prev = (it - 1) % 2
                torch.cuda.current_stream().wait_event(ev_collect[prev])
                stats = learner.update(engine.bufs[prev])
                ev_update[prev].record()
                if (it - 1) % icfg.behavior_sync == 0:
                    engine.actors[prev].refresh()
                ev_theta[prev].record()
                if rank == 0 and log_fn and (it % 10 == 1 or it == iters):
                    log_fn(f"[impala] update {it}: " +
                           " ".join(f"{k}={v:.4f}" for k, v in stats.items()))
                after_update(start_it + it - 1, stats, it == iters)
        out = dict(stats)
        out["episode_stats"] = env.episode_stats()
        return out

    # CPU fallback: explicit stale behavior copy, sequential
    behavior = copy.deepcopy(policy)
    collector = Collector(env, behavior, cfg.rollout_T, device,
                          seed=cfg.seed * 101 + rank)
    for it in range(start_it, start_it + iters):
        if it % icfg.behavior_sync == 0:
            behavior.load_state_dict(policy.state_dict())
        buf = collector.collect()
        stats = learner.update(buf)
        if rank == 0 and log_fn and (it % 10 == 0 or it == start_it + iters - 1):
            log_fn(f"[impala-cpu] iter {it}: " +
                   " ".join(f"{k}={v:.4f}" for k, v in stats.items()))
        after_update(it, stats, it == start_it + iters - 1)
    out = dict(stats)
    if hasattr(env, "venv"):
        st = env.venv.stats
        out["episode_stats"] = {
            "episodes": st.count, "mean_return": st.mean_return(),
            "mean_jct_us": st.mean_jct_us(), "completed": st.completed_sum,
        }
    return out
