"""CLI + checkpoint/resume + metrics-logging integration (CPU)."""
import dataclasses
import json

import numpy as np

from rlgpuschedule_amd.cli import main as cli_main
from rlgpuschedule_amd.trainer import CONFIGS, train


def test_cli_configs_lists(capsys):
    assert cli_main(["configs"]) == 0
    out = capsys.readouterr().out
    for name in CONFIGS:
        assert name in out


def test_cli_train_tiny(tmp_path, capsys):
    rc = cli_main([
        "train", "--config", "c1_cpu_random", "--iterations", "2",
        "--envs", "2", "--jobs-per-env", "8", "--out", str(tmp_path),
    ])
    assert rc == 0


def test_train_save_resume_roundtrip(tmp_path):
    cfg = dataclasses.replace(
        CONFIGS["c2_ppo_mlp"], num_envs=2, jobs_per_env=8, rollout_T=8,
        use_hip_env=False, device="cpu", hidden=16)
    out = train(cfg, iterations=3, log_fn=None, out_dir=str(tmp_path),
                save_every=3)
    ck = tmp_path / "policy_000003.pt"
    assert ck.exists()
    # metrics written
    lines = [json.loads(l) for l in
             (tmp_path / "metrics.jsonl").read_text().splitlines()]
    assert len(lines) == 3 and lines[0]["iter"] == 0
    assert "env_steps_per_sec" in lines[0]
    # resume continues iteration numbering
    out2 = train(cfg, iterations=2, log_fn=None, out_dir=str(tmp_path),
                 save_every=0, resume=str(ck))
    assert np.isfinite(out2["pi_loss"])


def test_nan_guard_raises():
    from rlgpuschedule_amd.metrics.logger import check_finite
    import pytest
    with pytest.raises(FloatingPointError):
        check_finite({"pi_loss": float("nan")})
    check_finite({"pi_loss": 1.0})


def test_cli_eval_baseline(capsys):
    rc = cli_main(["eval", "--config", "c1_cpu_random", "--baseline", "sjf",
                   "--envs", "2", "--episodes", "2", "--jobs-per-env", "12"])
    assert rc == 0
    out = json.loads(capsys.readouterr().out.strip().splitlines()[-1])
    assert out["episodes"] >= 2 and out["mean_jct_us"] > 0


def test_cli_eval_policy_cpu(capsys, tmp_path):
    import dataclasses
    from rlgpuschedule_amd.ckpt.io import save_policy
    from rlgpuschedule_amd.trainer import make_policy, resolve_device
    import torch
    cfg = CONFIGS["c1_cpu_random"]  # hidden must match what eval builds
    pol = make_policy(cfg, torch.device("cpu"))
    save_policy(tmp_path / "p.pt", pol, spec=cfg.spec)
    rc = cli_main(["eval", "--config", "c1_cpu_random", "--ckpt",
                   str(tmp_path / "p.pt"), "--envs", "2", "--episodes", "2",
                   "--jobs-per-env", "10"])
    assert rc == 0


def test_quickstart_example_runs(tmp_path, monkeypatch):
    import subprocess
    import sys
    from pathlib import Path
    root = Path(__file__).resolve().parent.parent
    r = subprocess.run([sys.executable, str(root / "examples/quickstart.py"),
                        "--iters", "2"], cwd=tmp_path, capture_output=True,
                       text=True, timeout=900)
    assert r.returncode == 0, r.stderr[-800:]
    assert "ppo: mean JCT" in r.stdout
