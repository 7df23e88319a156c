"""Warm-start depth doubling (Gopher §G.3.3), reference src/utils/extend_params.py.

Takes a trained N-block model state_dict and produces a 2N-block state_dict
where trained block i populates new blocks 2i and 2i+1; wte / final norm are
copied through. Operates on the .pth key layout (blocks.{i}.*).
"""

from __future__ import annotations

import re
from typing import Dict

import torch


def create_mapping(n_layers_in: int) -> Dict[int, int]:
    """new block index -> source block index (each source feeds 2 blocks)."""
    return {j: j // 2 for j in range(2 * n_layers_in)}


def extend_params(
    sd: Dict[str, torch.Tensor], n_layers_in: int
) -> Dict[str, torch.Tensor]:
    mapping = create_mapping(n_layers_in)
    out: Dict[str, torch.Tensor] = {}
    pat = re.compile(r"^blocks\.(\d+)\.(.+)$")
    for k, v in sd.items():
        if not pat.match(k):
            out[k] = v.clone()
    for new_idx, src_idx in mapping.items():
        for k, v in sd.items():
            m = pat.match(k)
            if m and int(m.group(1)) == src_idx:
                out[f"blocks.{new_idx}.{m.group(2)}"] = v.clone()
    return out
