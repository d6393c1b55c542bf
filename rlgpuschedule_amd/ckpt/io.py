"""Policy checkpoint save/load.

Native container: safetensors (+ a JSON sidecar for spec/metadata), plus a
single-file torch.save fallback (.pt). The reference repo's checkpoint
format could not be inspected (its mount was empty), so this
defines the rlsched-v1 format; a compatibility reader can be added once the
reference format is known.
"""
from __future__ import annotations

import json
from dataclasses import asdict
from pathlib import Path

import torch

FORMAT = "rlsched-v1"


def save_policy(path: str | Path, policy: torch.nn.Module, spec=None,
                optimizer=None, meta: dict | None = None):
    path = Path(path)
    payload = {
        "format": FORMAT,
        "policy_class": type(policy).__name__,
        "model_state": policy.state_dict(),
        "meta": meta or {},
    }
    if spec is not None:
        payload["env_spec"] = asdict(spec)
    if optimizer is not None:
        payload["optim_state"] = optimizer.state_dict()
    if path.suffix == ".safetensors":
        from safetensors.torch import save_file
        tensors = {k: v.detach().cpu().contiguous()
                   for k, v in policy.state_dict().items()}
        side = {
            "format": FORMAT,
            "policy_class": type(policy).__name__,
            "env_spec": asdict(spec) if spec is not None else None,
            "meta": meta or {},
        }
        save_file(tensors, str(path), metadata={"rlsched": json.dumps(side)})
    else:
        torch.save(payload, str(path))


def load_policy(path: str | Path, policy: torch.nn.Module | None = None,
                optimizer=None, map_location="cpu") -> dict:
    """Loads into `policy` (if given) and returns the metadata dict."""
    path = Path(path)
    if path.suffix == ".safetensors":
        from safetensors.torch import load_file
        import safetensors
        tensors = load_file(str(path))
        with safetensors.safe_open(str(path), framework="pt") as f:
            side = json.loads((f.metadata() or {}).get("rlsched", "{}"))
        if policy is not None:
            policy.load_state_dict(tensors)
        return side
    payload = torch.load(str(path), map_location=map_location,
                         weights_only=False)
    assert payload.get("format") == FORMAT, f"unknown format {payload.get('format')}"
    if policy is not None:
        policy.load_state_dict(payload["model_state"])
    if optimizer is not None and "optim_state" in payload:
        optimizer.load_state_dict(payload["optim_state"])
    return {k: v for k, v in payload.items() if k != "model_state"}
