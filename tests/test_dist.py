"""Multi-process data-parallel tests on gloo (world_size=2, CPU).

Validates the exact code path bench.py/trainer use for RCCL on MI355X:
broadcast_params + FlatGradAllreduce, via torch.multiprocessing spawn with
MASTER_ADDR=127.0.0.1.
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from rlgpuschedule_amd.env.spec import cluster16


def _worker_allreduce(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    from rlgpuschedule_amd.models.policy import MLPPolicy
    from rlgpuschedule_amd.parallel.dist import (FlatGradAllreduce,
                                                 broadcast_params,
                                                 init_from_env)
    r, w, _ = init_from_env()
    assert w == world
    torch.manual_seed(100 + rank)  # different init per rank
    pol = MLPPolicy(10, 5, 16)
    broadcast_params(pol)
    # after broadcast all ranks share rank0 weights
    sync = FlatGradAllreduce(pol)
    # rank-dependent grads -> mean across ranks
    loss = (pol.pi.weight * (rank + 1.0)).sum()
    loss.backward()
    sync()
    g = pol.pi.weight.grad.clone()
    expect = (sum(i + 1.0 for i in range(world)) / world)
    q.put((rank, float(g.mean().item()), float(expect),
           float(pol.pi.weight.detach().mean().item())))
    dist.destroy_process_group()


def _worker_train(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    import dataclasses
    from rlgpuschedule_amd.trainer import CONFIGS, train
    cfg = dataclasses.replace(
        CONFIGS["c2_ppo_mlp"], num_envs=2, jobs_per_env=8, rollout_T=8,
        use_hip_env=False, device="cpu", hidden=16)
    out = train(cfg, iterations=2, log_fn=None)
    # params must be identical across ranks after synced updates
    from rlgpuschedule_amd.models.policy import MLPPolicy  # noqa
    import torch.distributed as d
    import hashlib
    q.put((rank, out["pi_loss"]))
    d.destroy_process_group()


def _spawn(fn, world=2):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = int(np.random.default_rng().integers(20000, 40000))
    ps = [ctx.Process(target=fn, args=(r, world, port, q)) for r in range(world)]
    for p in ps:
        p.start()
    res = [q.get() for _ in range(world)]
    for p in ps:
        p.join(120)
        assert p.exitcode == 0
    return sorted(res)


def test_flat_grad_allreduce_gloo():
    res = _spawn(_worker_allreduce)
    w0 = None
    for rank, gmean, expect, wmean in res:
        base = 1.0  # grad of (w * c).sum() wrt w is c
        assert abs(gmean - expect * base) < 1e-6
        if w0 is None:
            w0 = wmean
        assert abs(wmean - w0) < 1e-7  # broadcast made weights identical


def test_distributed_ppo_training_gloo():
    res = _spawn(_worker_train)
    assert len(res) == 2
    for rank, loss in res:
        assert np.isfinite(loss)


def _worker_agree(rank, world, port, q):
    import os
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as d
    d.init_process_group("gloo", rank=rank, world_size=world)
    from rlgpuschedule_amd.parallel.dist import agree_all
    # mixed votes -> everyone must get False; unanimous True -> True
    r1 = agree_all(rank == 0)
    r2 = agree_all(True)
    q.put((rank, r1, r2))
    d.destroy_process_group()


def test_agree_all_gloo():
    res = _spawn(_worker_agree)
    for rank, r1, r2 in res:
        assert r1 is False    # one rank voted False -> all False
        assert r2 is True


def _worker_sweep(rank, world, port, q):
    import os
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    import dataclasses

    import torch.distributed as d
    d.init_process_group("gloo", rank=rank, world_size=world)
    from rlgpuschedule_amd.agents.selection import train_multi_seed
    from rlgpuschedule_amd.trainer import CONFIGS
    cfg = dataclasses.replace(
        CONFIGS["c2_ppo_mlp"], num_envs=4, jobs_per_env=8, rollout_T=8,
        use_hip_env=False, device="cpu", hidden=16)
    cfg = dataclasses.replace(
        cfg, ppo=dataclasses.replace(cfg.ppo, epochs=1, minibatches=1))
    res = train_multi_seed(
        cfg, [1, 2, 3], iterations=4, select_every=2,
        select_kw=dict(num_envs=2, min_episodes=2, max_steps=3000),
        eval_kw=dict(num_envs=2, min_episodes=2, max_steps=3000))
    # note: trainer.train calls init_from_env which must not re-init
    q.put((rank, sorted(r["seed"] for r in res["runs"]), res["best_seed"],
           float(sum(v.sum() for v in res["best_state"].values()))))
    d.destroy_process_group()


def test_multi_seed_sweep_striped_across_ranks_gloo():
    """3 seeds striped over 2 gloo ranks: every rank sees all 3 results,
    agrees on the winner, and holds the identical winning state."""
    res = _spawn(_worker_sweep)
    (r0, seeds0, best0, sum0), (r1, seeds1, best1, sum1) = res
    assert seeds0 == seeds1 == [1, 2, 3]
    assert best0 == best1
    assert abs(sum0 - sum1) < 1e-6
