"""Multi-seed training with unbiased eval-based checkpoint selection.

The deployment pipeline proven in round-1 experiments (docs/PERF.md):
PPO training on the scheduling sim is high-variance (atomic-accumulation
nondeterminism makes late training chaotic), so the reliable recipe is
  1. train N seeds with the KL-to-behavior penalty (PPOConfig.kl_coef)
     so no seed collapses into a degenerate never-schedule policy;
  2. during each run, periodically greedy-evaluate on a SELECTION trace
     pool and keep the best-JCT snapshot;
  3. pick the winning seed by its selection-pool JCT and report/deploy
     its held-out-pool JCT (selection and held-out pools use disjoint
     seeds, so the reported number carries no selection bias).

A seed sweep is embarrassingly parallel: under torch.distributed.run
(one rank per GPU) train_multi_seed stripes the seed list across ranks
(rank r trains seeds[r::world]) and all-gathers the per-seed results,
so `train-sweep --seeds 0 1 2 3 4 5 6 7` on an 8-GPU node trains all
eight seeds concurrently with no inter-rank communication during
training.
"""
from __future__ import annotations

import dataclasses

import torch

from .evaluate import evaluate_policy

# disjoint pool seeds: selection pool != held-out report pool
SELECT_POOL_SEED = 555
HELDOUT_POOL_SEED = 991


class BestCheckpointSelector:
    """trainer.train callback: every `every` iters, greedy-evaluate on the
    selection pool and snapshot the policy if mean JCT improved.

    A degenerate never-schedule policy completes 0 jobs and reports
    mean_jct_us == 0 — the `completed > 0` guard keeps it from winning.
    """

    def __init__(self, spec, every: int, device: str | None = None,
                 **eval_kw):
        self.spec = spec
        self.every = every
        self.device = device
        self.eval_kw = dict(eval_kw)
        self.eval_kw.setdefault("seed", SELECT_POOL_SEED)
        self.eval_kw.setdefault("num_envs", 8)
        self.eval_kw.setdefault("min_episodes", 8)
        # bound the cost of evaluating a degenerate policy (its episodes
        # never finish, so the eval would otherwise run to max_steps)
        self.eval_kw.setdefault("max_steps", 30_000)
        self.best_jct = float("inf")
        self.state = None
        self.at = -1

    @torch.no_grad()
    def __call__(self, it, policy, stats):
        if not self.every or (it + 1) % self.every:
            return
        st = evaluate_policy(self.spec, policy, greedy=True,
                             device=self.device, **self.eval_kw)
        if st["completed"] > 0 and st["mean_jct_us"] < self.best_jct:
            self.best_jct = st["mean_jct_us"]
            self.state = {k: v.detach().clone()
                          for k, v in policy.state_dict().items()}
            self.at = it + 1


def train_multi_seed(cfg, seeds, iterations: int | None = None,
                     select_every: int = 200, select_kw: dict | None = None,
                     eval_kw: dict | None = None, log_fn=None,
                     save_best: str | None = None) -> dict:
    """Run `cfg` once per seed with checkpoint selection; return per-seed
    held-out results plus the winning seed's state_dict.

    The winner is chosen by SELECTION-pool JCT (never by the held-out
    number), so `best_heldout_jct_us` is an unbiased estimate of deployed
    quality. cfg.ppo.kl_coef > 0 is strongly recommended (see module doc).
    """
    import torch.distributed as dist

    from ..trainer import train

    rank, world = 0, 1
    if dist.is_initialized():
        rank, world = dist.get_rank(), dist.get_world_size()
    my_seeds = list(seeds)[rank::world] if world > 1 else list(seeds)

    device = ("cuda" if torch.cuda.is_available() and cfg.use_hip_env
              else "cpu")
    base_kw = dict(jobs_per_env=cfg.jobs_per_env,
                   trace_kind=cfg.trace_kind, **cfg.trace_kw)
    skw = {**base_kw, **(select_kw or {})}
    ekw = {**base_kw, "seed": HELDOUT_POOL_SEED, "num_envs": 16,
           "min_episodes": 16, **(eval_kw or {})}
    runs = []
    best = None
    for sd in my_seeds:
        c = dataclasses.replace(cfg, seed=int(sd))
        sel = BestCheckpointSelector(cfg.spec, select_every, device=device,
                                     **skw)
        out = train(c, iterations, log_fn=None, callback=sel, solo=True)
        policy = out.pop("_policy")
        if sel.state is not None:
            policy.load_state_dict(sel.state)
        held = evaluate_policy(cfg.spec, policy, greedy=True, device=device,
                               **ekw)
        rec = {"seed": int(sd), "selected_at": sel.at,
               "selection_jct_us": (sel.best_jct if sel.state is not None
                                    else None),  # None: selection disabled
               "heldout_jct_us": held["mean_jct_us"],
               "heldout_episodes": held["episodes"]}
        runs.append(rec)
        if log_fn:
            log_fn(f"[sweep] seed {sd}: selection "
                   f"{sel.best_jct / 1e6:.1f}s @it{sel.at}, held-out "
                   f"{held['mean_jct_us'] / 1e6:.1f}s")
        if sel.state is not None and (best is None
                                      or sel.best_jct < best["jct"]):
            best = {"jct": sel.best_jct, "seed": int(sd),
                    "state": sel.state, "heldout": held["mean_jct_us"]}
    if world > 1:
        # gather per-seed records and each rank's best candidate
        gathered = [None] * world
        dist.all_gather_object(gathered, {
            "runs": runs,
            "best": None if best is None else
            {k: best[k] for k in ("jct", "seed", "heldout")}})
        runs = [r for g in gathered for r in g["runs"]]
        runs.sort(key=lambda r: r["seed"])
        cands = [g["best"] for g in gathered if g["best"] is not None]
        if cands:
            # every rank computes the same winner + source rank, and every
            # rank joins the broadcast (collective — no conditional paths)
            win = min(cands, key=lambda c: (c["jct"], c["seed"]))
            src = next(i for i, g in enumerate(gathered)
                       if g["best"] is not None
                       and g["best"]["seed"] == win["seed"])
            holder = [None]
            if rank == src:
                holder = [{k: v.cpu() for k, v in best["state"].items()}]
            dist.broadcast_object_list(holder, src=src)
            best = {**win, "state": {
                k: v.to(device) for k, v in holder[0].items()}}
        else:
            best = None
    res = {"runs": runs}
    if best is not None:
        res.update(best_seed=best["seed"],
                   best_selection_jct_us=best["jct"],
                   best_heldout_jct_us=best["heldout"],
                   best_state=best["state"])
        if save_best:
            from ..ckpt.io import save_policy
            from ..trainer import make_policy
            pol = make_policy(cfg, torch.device(device))
            pol.load_state_dict(best["state"])
            save_policy(save_best, pol, spec=cfg.spec,
                        meta={"sweep_seeds": [int(s) for s in seeds],
                              "best_seed": best["seed"],
                              "selection_jct_us": best["jct"]})
    return res
