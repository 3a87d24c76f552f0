"""Policy deployment helpers: save/load trained policies as safetensors
(weights) + a JSON sidecar (architecture string), so serving code does not
need pickle or the training framework.

A green-field addition — the reference's deployment story is pickled
modules (examples/scripts/rl_enjoy.py); safetensors files are
language-agnostic and safe to load from untrusted sources.
"""

import json
import os
from typing import Optional, Union

import torch
from torch import nn

__all__ = ["save_policy", "load_policy"]


def save_policy(policy: nn.Module, path: str, *, metadata: Optional[dict] = None) -> str:
    """Write `policy`'s weights to `<path>` (safetensors) and its repr plus
    any user metadata to `<path>.json`. Returns the weights path."""
    from safetensors.torch import save_file

    state = {k: v.detach().cpu().contiguous() for k, v in policy.state_dict().items()}
    if not path.endswith(".safetensors"):
        path = path + ".safetensors"
    save_file(state, path)
    sidecar = {
        "format": "evotorch_amd.policy.v1",
        "repr": repr(policy),
        "metadata": metadata or {},
    }
    with open(path + ".json", "w") as f:
        json.dump(sidecar, f, indent=1)
    return path


def load_policy(path: str, module: nn.Module, *, device: Union[str, torch.device] = "cpu") -> nn.Module:
    """Load weights saved by `save_policy` into a freshly constructed
    `module` of the same architecture (e.g. another `problem.to_policy(...)`
    output, or the same `str_to_net` string)."""
    from safetensors.torch import load_file

    if not path.endswith(".safetensors"):
        path = path + ".safetensors"
    if not os.path.exists(path):
        raise FileNotFoundError(path)
    state = load_file(path, device=str(device))
    module.load_state_dict(state)
    module.requires_grad_(False)
    return module.to(device)
