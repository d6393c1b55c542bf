"""Foreign-checkpoint compatibility scaffold.

The north star requires reading the REFERENCE framework's policy
checkpoints, but its format has been unobservable for two rounds (empty
mount — SURVEY.md §0/§0.1). This module makes the eventual reader a
pure data-entry exercise: once the reference format is known, describe
it as a `CompatSpec` (tensor-name map + per-tensor transforms +
container hints) and `load_foreign` does the rest. A synthetic spec
(`EXAMPLE_TF_STYLE`) exercises the machinery in tests today.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from pathlib import Path
from typing import Callable

import numpy as np
import torch


@dataclass
class CompatSpec:
    """Declarative description of a foreign checkpoint format."""
    name: str
    # foreign tensor name -> our state_dict key (models/policy.py names)
    name_map: dict[str, str]
    # our key -> transform applied to the foreign array (e.g. transpose:
    # TF dense kernels are [in,out], torch Linear weights are [out,in])
    transforms: dict[str, Callable[[np.ndarray], np.ndarray]] = \
        field(default_factory=dict)
    # container loader: path -> {foreign_name: np.ndarray}
    reader: Callable[[Path], dict[str, np.ndarray]] | None = None


def _read_npz(path: Path) -> dict[str, np.ndarray]:
    with np.load(path) as z:
        return {k: z[k] for k in z.files}


def _read_pt(path: Path) -> dict[str, np.ndarray]:
    obj = torch.load(str(path), map_location="cpu", weights_only=True)
    if isinstance(obj, dict) and "state_dict" in obj:
        obj = obj["state_dict"]
    return {k: (v.numpy() if torch.is_tensor(v) else np.asarray(v))
            for k, v in obj.items()}


_READERS = {".npz": _read_npz, ".pt": _read_pt, ".pth": _read_pt}


def load_foreign(path: str | Path, policy: torch.nn.Module,
                 spec: CompatSpec, strict: bool = True) -> dict:
    """Load a foreign-format checkpoint into `policy` via `spec`.

    Returns {"mapped": [...], "skipped_foreign": [...], "missing": [...]}.
    strict=True raises if any of the policy's parameters has no mapped
    source tensor.
    """
    path = Path(path)
    reader = spec.reader or _READERS.get(path.suffix)
    if reader is None:
        raise ValueError(f"no reader for {path.suffix} (give spec.reader)")
    foreign = reader(path)
    target = policy.state_dict()
    out = {k: v.clone() for k, v in target.items()}
    mapped, skipped = [], []
    for fname, arr in foreign.items():
        ours = spec.name_map.get(fname)
        if ours is None:
            skipped.append(fname)
            continue
        if ours not in target:
            raise KeyError(f"{spec.name}: map target {ours!r} not in the "
                           f"policy state_dict")
        a = np.asarray(arr)
        tf = spec.transforms.get(ours)
        if tf is not None:
            a = tf(a)
        t = torch.as_tensor(a, dtype=target[ours].dtype)
        if tuple(t.shape) != tuple(target[ours].shape):
            raise ValueError(
                f"{spec.name}: {fname} -> {ours}: shape {tuple(t.shape)} "
                f"!= expected {tuple(target[ours].shape)}")
        out[ours] = t
        mapped.append(ours)
    missing = [k for k in target if k not in mapped]
    if strict and missing:
        raise ValueError(f"{spec.name}: unmapped policy tensors: {missing}")
    policy.load_state_dict(out)
    return {"mapped": mapped, "skipped_foreign": skipped,
            "missing": missing}


# Example spec exercising the machinery: a TF1-style flat-name format
# (this lineage's public repos are mostly TF1) with [in,out] dense
# kernels. NOT the reference's real format — replace name_map/transforms
# when the reference becomes readable.
def _t(a: np.ndarray) -> np.ndarray:
    return a.T


def example_tf_style_spec(layers: int = 2) -> CompatSpec:
    nm = {}
    tf = {}
    for i in range(layers):
        nm[f"policy/dense_{i}/kernel"] = f"trunk.{i}.weight"
        nm[f"policy/dense_{i}/bias"] = f"trunk.{i}.bias"
        tf[f"trunk.{i}.weight"] = _t
    nm["policy/logits/kernel"] = "pi.weight"
    nm["policy/logits/bias"] = "pi.bias"
    nm["policy/value/kernel"] = "v.weight"
    nm["policy/value/bias"] = "v.bias"
    tf["pi.weight"] = _t
    tf["v.weight"] = _t
    return CompatSpec(name="example-tf-style", name_map=nm, transforms=tf)
