"""Policy checkpoint save/load.

Native container: safetensors (+ a JSON sidecar for spec/metadata), plus a
single-file torch.save fallback (.pt). The reference repo's checkpoint
format could not be inspected (its mount was empty), so this
defines the rlsched-v1 format; a compatibility reader can be added once the
reference format is known.

Both containers persist optimizer state (torch optimizers AND HipAdam):
in .safetensors the optimizer tensors are stored under `__optim__/...`
keys with the non-tensor skeleton in the JSON sidecar; in .pt the raw
state_dict rides in the payload. Loading defaults to
`torch.load(weights_only=True)`; pass allow_pickle=True only for trusted
legacy files.
"""
from __future__ import annotations

import json
import warnings
from dataclasses import asdict
from pathlib import Path

import torch

FORMAT = "rlsched-v1"

_OPT_PREFIX = "__optim__"


def _flatten_state(obj, prefix: str, tensors: dict):
    """Replace tensors in a nested container with placeholder refs,
    collecting them into `tensors`; returns a JSON-able skeleton.
    Preserves int dict keys (torch optimizer state uses them)."""
    if isinstance(obj, torch.Tensor):
        tensors[prefix] = obj.detach().cpu().contiguous()
        return {"__tensor__": prefix}
    if isinstance(obj, dict):
        items = []
        for k, v in obj.items():
            kk = str(k)
            items.append([kk, "int" if isinstance(k, int) else "str",
                          _flatten_state(v, f"{prefix}/{kk}", tensors)])
        return {"__dict__": items}
    if isinstance(obj, (list, tuple)):
        return {"__list__": [_flatten_state(v, f"{prefix}/{i}", tensors)
                             for i, v in enumerate(obj)]}
    return {"__val__": obj}


def _unflatten_state(skel, tensors: dict):
    if "__tensor__" in skel:
        return tensors[skel["__tensor__"]]
    if "__dict__" in skel:
        return {(int(k) if kt == "int" else k): _unflatten_state(v, tensors)
                for k, kt, v in skel["__dict__"]}
    if "__list__" in skel:
        return [_unflatten_state(v, tensors) for v in skel["__list__"]]
    return skel["__val__"]


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
        if optimizer is not None:
            opt_tensors: dict = {}
            side["optim_skel"] = _flatten_state(
                optimizer.state_dict(), _OPT_PREFIX, opt_tensors)
            tensors.update(opt_tensors)
        save_file(tensors, str(path), metadata={"rlsched": json.dumps(side)})
    else:
        torch.save(payload, str(path))


def load_policy(path: str | Path, policy: torch.nn.Module | None = None,
                optimizer=None, map_location="cpu",
                allow_pickle: bool = False) -> dict:
    """Loads into `policy` (if given) and returns the metadata dict.

    The returned dict includes "optim_state" when the checkpoint carries
    optimizer state — prefer feeding that to `apply_optim_state(learner,
    sd)` over passing `optimizer=` here, so learners that create their
    optimizer lazily (the HIP fast paths) can defer the restore.
    """
    path = Path(path)
    if path.suffix == ".safetensors":
        from safetensors.torch import load_file
        import safetensors
        tensors = load_file(str(path))
        with safetensors.safe_open(str(path), framework="pt") as f:
            side = json.loads((f.metadata() or {}).get("rlsched", "{}"))
        model_t = {k: v for k, v in tensors.items()
                   if not k.startswith(_OPT_PREFIX)}
        if policy is not None:
            policy.load_state_dict(model_t)
        skel = side.pop("optim_skel", None)
        if skel is not None:
            side["optim_state"] = _unflatten_state(skel, tensors)
            if optimizer is not None:
                optimizer.load_state_dict(side["optim_state"])
        elif optimizer is not None:
            warnings.warn(f"{path}: no optimizer state in checkpoint; "
                          "optimizer left untouched")
        return side
    payload = torch.load(str(path), map_location=map_location,
                         weights_only=not allow_pickle)
    assert payload.get("format") == FORMAT, f"unknown format {payload.get('format')}"
    if policy is not None:
        policy.load_state_dict(payload["model_state"])
    if optimizer is not None and "optim_state" in payload:
        optimizer.load_state_dict(payload["optim_state"])
    return {k: v for k, v in payload.items() if k != "model_state"}


def resolve_pending_opt_state(learner):
    """Called by the HIP learners at the end of _alloc(): applies an
    optimizer state stashed by load_optim_state() now that the real
    optimizer (HipAdam, created lazily) exists. ADVICE r1: applying the
    state to the placeholder torch Adam before _alloc crashed, and a
    compatible state would have been silently discarded at the swap."""
    sd = getattr(learner, "_pending_opt_state", None)
    if sd is None:
        return
    learner._pending_opt_state = None
    opt = learner.opt
    is_hip_state = isinstance(sd, dict) and sd.get("hip_adam")
    is_hip_opt = type(opt).__name__ == "HipAdam"
    if is_hip_state != is_hip_opt:
        warnings.warn("checkpointed optimizer state kind (HipAdam vs "
                      "torch) does not match this learner's optimizer; "
                      "optimizer state not restored")
        return
    opt.load_state_dict(sd)


def apply_optim_state(learner, sd) -> bool:
    """Route checkpointed optimizer state to a learner, handling lazy
    optimizers (HIP learners create HipAdam on first update) and
    cross-path mismatches (HipAdam state vs torch optimizer and vice
    versa). Returns True if the state was applied or stashed."""
    if sd is None or learner is None:
        return False
    if hasattr(learner, "load_optim_state"):
        learner.load_optim_state(sd)
        return True
    opt = getattr(learner, "opt", None)
    if opt is None:
        warnings.warn("learner has no optimizer yet and no "
                      "load_optim_state(); optimizer state dropped")
        return False
    is_hip_state = isinstance(sd, dict) and sd.get("hip_adam")
    is_hip_opt = type(opt).__name__ == "HipAdam"
    if is_hip_state != is_hip_opt:
        warnings.warn("optimizer-state kind mismatch (HipAdam vs torch); "
                      "optimizer state not restored")
        return False
    opt.load_state_dict(sd)
    return True
