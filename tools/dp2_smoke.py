"""dp2-on-one-GPU RCCL smoke.

Launch: python -m torch.distributed.run --standalone --nnodes=1
          --nproc-per-node 2 --master-addr 127.0.0.1 tools/dp2_smoke.py

Two ranks share cuda:0 (NCCL/RCCL does not officially support >1 rank per
device; this probes whether the runtime permits it for our collective
set). Each rank runs the REAL DP code path: broadcast_params +
HipPPOLearner with the fused flat all-reduce inside HipAdam, 3 update
iterations, then verifies parameters agree bitwise across ranks.

Outcome is printed as one JSON line per rank: {"rank", "ok", "err"}; the
caller records the result either way (a documented refusal is an
acceptable answer per the verdict).
"""
import json
import os
import sys
import traceback
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    import torch
    import torch.distributed as dist
    rank = int(os.environ["RANK"])
    ok, err = False, ""
    try:
        # both ranks on the SAME device (1-GPU box)
        os.environ["LOCAL_RANK"] = "0"
        torch.cuda.set_device(0)
        dist.init_process_group("nccl")
        from rlgpuschedule_amd.agents.ppo import PPOConfig
        from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
        from rlgpuschedule_amd.agents.rollout_gpu import GpuRolloutEngine
        from rlgpuschedule_amd.env.hip_env import HipVectorEnv
        from rlgpuschedule_amd.env.spec import cluster64
        from rlgpuschedule_amd.models.policy import MLPPolicy
        from rlgpuschedule_amd.parallel.dist import (FlatGradAllreduce,
                                                     broadcast_params)
        torch.manual_seed(100 + rank)  # different init; broadcast aligns
        spec = cluster64(horizon=100_000)
        env = HipVectorEnv(spec, 128, trace_kind="philly", jobs_per_env=32,
                           seed=11 + rank)
        policy = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
        broadcast_params(policy)
        sync = FlatGradAllreduce(policy)
        learner = HipPPOLearner(policy, PPOConfig(epochs=1, minibatches=2),
                                grad_sync=sync, use_graph=True)
        engine = GpuRolloutEngine(env, policy, 16, seed=5 + rank)
        for _ in range(3):
            learner.update(engine.collect())
        torch.cuda.synchronize()
        # params must agree across ranks after synced updates
        flat = torch.cat([p.detach().reshape(-1) for p in
                          policy.parameters()])
        mx = flat.clone()
        dist.all_reduce(mx, op=dist.ReduceOp.MAX)
        mn = flat.clone()
        dist.all_reduce(mn, op=dist.ReduceOp.MIN)
        diff = float((mx - mn).abs().max().item())
        assert diff == 0.0, f"rank divergence {diff}"
        ok = True
    except Exception as e:  # noqa: BLE001
        err = f"{type(e).__name__}: {e}\n{traceback.format_exc(limit=3)}"
    print(json.dumps({"rank": rank, "ok": ok, "err": err[:500]}),
          flush=True)
    if ok:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
