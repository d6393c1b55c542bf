"""Multi-seed JCT quality sweep with eval-based checkpoint selection.

The honest statistical claim for docs/PERF.md: distribution of final
(held-out, seed-991) greedy/sampled JCT over several training seeds,
each run selecting its best checkpoint on a DISJOINT selection pool
(seed 555). Baselines evaluated once on the same held-out pool.

Round-2 knobs: --ent-floor/--ent-floor-coef (per-row hinge
entropy bonus in the fused loss), --obs-version 2, --elastic (elastic
jobs + SHRINK/EXPAND actions), longer --iters.
"""
import argparse
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
sys.path.insert(0, str(Path(__file__).resolve().parent))

from train_eval_jct import run  # noqa: E402

if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--seeds", type=int, nargs="+",
                    default=[7, 21, 42, 1234])
    ap.add_argument("--iters", type=int, default=3000)
    ap.add_argument("--envs", type=int, default=2048)
    ap.add_argument("--hidden", type=int, default=512)
    ap.add_argument("--select-every", type=int, default=200)
    ap.add_argument("--select-envs", type=int, default=8)
    ap.add_argument("--select-eps", type=int, default=8)
    ap.add_argument("--phase2-iters", type=int, default=0,
                    help="extra iters at --phase2-lr after the main run "
                         "(fresh learner/capture)")
    ap.add_argument("--phase2-lr", type=float, default=1e-4)
    ap.add_argument("--eval-envs", type=int, default=16)
    ap.add_argument("--eval-eps", type=int, default=16)
    ap.add_argument("--bc-rollouts", type=int, default=0,
                    help=">0: behavior-clone SJF first (agents/bc.py)")
    ap.add_argument("--bc-epochs", type=int, default=4)
    ap.add_argument("--bc-lr", type=float, default=1e-3)
    ap.add_argument("--bc-ent", type=float, default=0.0)
    ap.add_argument("--kl", type=float, default=0.0)
    ap.add_argument("--ent-floor", type=float, default=0.0)
    ap.add_argument("--ent-floor-coef", type=float, default=0.0)
    ap.add_argument("--obs-version", type=int, default=1, choices=[1, 2])
    ap.add_argument("--elastic", action="store_true")
    ap.add_argument("--out", default=None, help="also write JSON here")
    ap.add_argument("--save-dir", default=None,
                    help="save each seed's selected policy checkpoint here")
    a = ap.parse_args()
    out = {"params": vars(a), "runs": []}
    for i, sd in enumerate(a.seeds):
        ck = (f"{a.save_dir}/seed{sd}.pt" if a.save_dir else None)
        if ck:
            Path(a.save_dir).mkdir(parents=True, exist_ok=True)
        r = run(save_ckpt=ck,
                iters=a.iters, envs=a.envs, hidden=a.hidden, seed=sd,
                select_every=a.select_every, dtype="bf16", kl_coef=a.kl,
                ent_floor=a.ent_floor, ent_floor_coef=a.ent_floor_coef,
                obs_version=a.obs_version, elastic=a.elastic,
                select_envs=a.select_envs, select_eps=a.select_eps,
                phase2_iters=a.phase2_iters, phase2_lr=a.phase2_lr,
                eval_envs=a.eval_envs, eval_eps=a.eval_eps,
                bc_rollouts=a.bc_rollouts, bc_epochs=a.bc_epochs,
                bc_lr=a.bc_lr, bc_ent=a.bc_ent)
        rec = {"seed": sd, "train_s": r["train_s"],
               "bc_ce": r.get("bc", {}).get("bc_ce"),
               "selected_at": r.get("selected_at"),
               "selection_jct_us": r.get("selection_jct_us"),
               "final_entropy": r["final"].get("entropy"),
               "greedy_jct_s": round(r["ppo_greedy"]["mean_jct_us"] / 1e6, 1),
               "sample_jct_s": round(r["ppo_sample"]["mean_jct_us"] / 1e6, 1),
               # eval integrity: a policy must not "win" by starving hard
               # windows out of the completed-episode statistics
               "greedy_episodes": r["ppo_greedy"]["episodes"],
               "greedy_completed": r["ppo_greedy"]["completed"]}
        if i == 0:
            for b in ("sjf", "fifo", "random"):
                out[b + "_jct_s"] = round(r[b]["mean_jct_us"] / 1e6, 1)
                out[b + "_episodes"] = r[b]["episodes"]
                out[b + "_completed"] = r[b]["completed"]
        out["runs"].append(rec)
        print("RUN", json.dumps(rec), flush=True)
    gj = sorted(r["greedy_jct_s"] for r in out["runs"])
    out["greedy_median_jct_s"] = gj[len(gj) // 2] if len(gj) % 2 else \
        0.5 * (gj[len(gj) // 2 - 1] + gj[len(gj) // 2])
    print("SWEEP", json.dumps(out))
    if a.out:
        Path(a.out).parent.mkdir(parents=True, exist_ok=True)
        Path(a.out).write_text(json.dumps(out, indent=1))
