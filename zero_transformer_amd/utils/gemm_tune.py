"""hipBLASLt GEMM algorithm selection via PyTorch TunableOp.

The reference leaves GEMM selection to XLA; on MI355X the plain-GEMM path is
hipBLASLt through PyTorch, whose default heuristic picks ~1.1-1.2 PF/s
kernels for our (32k, 2048/6144/8192/50304) training shapes. TunableOp
benchmarks every available algorithm per shape once and records the winner.

Workflow:
  1. `python tools/gemm_tune.py` on a GPU box sweeps the flagship training
     shapes and writes profiles/tunableop_gfx950.csv (committed).
  2. Training/bench entry points call `enable()` which loads the committed
     results (tuning itself stays off, so startup cost is zero).
Opt out with ZTA_TUNABLEOP=0.
"""

from __future__ import annotations

import os

_REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
RESULTS = os.path.join(_REPO, "profiles", "tunableop_gfx950.csv")


def enable(tuning: bool = False, filename: str | None = None) -> bool:
    """Enable TunableOp; load committed results unless `tuning` is set.

    Returns True if enabled. No-op (False) when ZTA_TUNABLEOP=0, when torch
    lacks the API, or when loading and no results file exists.
    """
    if os.environ.get("ZTA_TUNABLEOP", "1") == "0":
        return False
    import torch

    if not hasattr(torch.cuda, "tunable"):
        return False
    fn = filename or RESULTS
    if not tuning and not os.path.exists(fn):
        return False
    t = torch.cuda.tunable
    t.enable(True)
    t.tuning_enable(tuning)
    # %d is replaced by the local rank / instance id by TunableOp itself
    t.set_filename(fn, insert_device_ordinal=False)
    if not tuning:
        try:
            t.read_file(fn)
        except Exception:
            pass
    return True


# Note: TunableOp persists results to the configured filename automatically
# at process shutdown when tuning was enabled; there is no explicit write API
# in this torch build.
