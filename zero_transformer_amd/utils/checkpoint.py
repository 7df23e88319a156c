"""Two-stream checkpointing with keep=5, mirroring the reference
(main_zero.py:58-139): `params_<step>.pt` (full fp32 model state_dict in the
torch_compatability .pth key layout) and `optimizer_<step>.pt` (gathered Adam
moments + step + RNG state). Rank 0 writes after collective gathers.
"""

from __future__ import annotations

import os
import re
from typing import Dict, Optional, Tuple

import torch

KEEP = 5


def _prune(workdir: str, prefix: str, keep: int = KEEP) -> None:
    pat = re.compile(rf"{re.escape(prefix)}(\d+)\.pt$")
    found = []
    for f in os.listdir(workdir):
        m = pat.match(f)
        if m:
            found.append((int(m.group(1)), f))
    for _, f in sorted(found)[:-keep]:
        try:
            os.remove(os.path.join(workdir, f))
        except OSError:
            pass


def latest_step(workdir: str, prefix: str = "params_") -> Optional[int]:
    if not os.path.isdir(workdir):
        return None
    pat = re.compile(rf"{re.escape(prefix)}(\d+)\.pt$")
    steps = [int(m.group(1)) for f in os.listdir(workdir) if (m := pat.match(f))]
    return max(steps) if steps else None


def save_checkpoint_params(workdir: str, step: int, param_sd: Dict[str, torch.Tensor]) -> str:
    os.makedirs(workdir, exist_ok=True)
    path = os.path.join(workdir, f"params_{step}.pt")
    torch.save(param_sd, path)
    _prune(workdir, "params_")
    return path


def save_checkpoint_optimizer(workdir: str, step: int, opt_state: Dict) -> str:
    os.makedirs(workdir, exist_ok=True)
    path = os.path.join(workdir, f"optimizer_{step}.pt")
    torch.save(opt_state, path)
    _prune(workdir, "optimizer_")
    return path


def restore_checkpoint(workdir: str, step: Optional[int] = None) -> Tuple[Dict, Dict, int]:
    """Load (params_sd, optimizer_state, step) for the given or latest step."""
    if step is None:
        step = latest_step(workdir)
    if step is None:
        raise FileNotFoundError(f"no checkpoints under {workdir}")
    params = torch.load(os.path.join(workdir, f"params_{step}.pt"), map_location="cpu", weights_only=True)
    opt = torch.load(os.path.join(workdir, f"optimizer_{step}.pt"), map_location="cpu", weights_only=True)
    return params, opt, step
