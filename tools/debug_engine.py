"""Debug: find where rollout-engine actions become invalid under masks."""
import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch

from rlgpuschedule_amd.agents.rollout_gpu import GpuRolloutEngine
from rlgpuschedule_amd.env.hip_env import HipVectorEnv
from rlgpuschedule_amd.env.spec import cluster64
from rlgpuschedule_amd.models.policy import MLPPolicy


def run(use_graph):
    torch.manual_seed(0)
    spec = cluster64(horizon=100_000)
    env = HipVectorEnv(spec, 256, trace_kind="philly", jobs_per_env=64, seed=5)
    policy = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    eng = GpuRolloutEngine(env, policy, T=32, seed=0, use_graph=use_graph)
    bad_total = 0
    for it in range(3):
        buf = eng.collect()
        torch.cuda.synchronize()
        for t in range(32):
            chosen = buf.mask[t].gather(1, buf.actions[t].unsqueeze(1)).squeeze(1)
            bad = (chosen == 0).nonzero().flatten()
            if len(bad):
                bad_total += len(bad)
                e = int(bad[0])
                print(f"graph={use_graph} it={it} t={t} nbad={len(bad)} env={e} "
                      f"act={int(buf.actions[t,e])} mask={buf.mask[t,e].tolist()} "
                      f"done[t-1]={int(buf.done[t-1,e]) if t>0 else -1} "
                      f"done[t]={int(buf.done[t,e])} "
                      f"logp={float(buf.logp[t,e]):.4f} "
                      f"u={float(eng.u_all[t,e]):.6f}")
                if bad_total > 6:
                    return
    print(f"graph={use_graph}: bad_total={bad_total}")


if __name__ == "__main__":
    run(False)
    run(True)
