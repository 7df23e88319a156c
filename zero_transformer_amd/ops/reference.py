"""Pure-PyTorch fp32-reference implementations of every fused HIP op.

These serve two purposes:
  1. CPU execution path (tests, gloo-based distributed plumbing tests).
  2. Numerics references for HIP-kernel parity tests (tests/test_ops_gpu.py):
     each HIP kernel is compared against the plain fp32 PyTorch op here.

Semantics mirror the JAX reference:
  - fp32 softmax in attention (reference src/models/layers.py:167-173 — bf16
    softmax caused a documented model failure, logs/580.md:94-98).
  - tanh-approximate GELU (flax nn.gelu default, reference layers.py:68).
  - one-hot-free shifted cross entropy in fp32 (reference src/utils/losses.py
    computes one-hot x log_softmax; we fuse the gather instead).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn.functional as F


def alibi_slopes(num_heads: int) -> torch.Tensor:
    """Standard ALiBi per-head slopes (Press et al., Train Short Test Long).

    Matches reference src/models/layers.py:17-30 (get_slopes).
    """

    def pow2_slopes(n: int):
        start = 2.0 ** (-(2.0 ** -(math.log2(n) - 3)))
        return [start * (start ** i) for i in range(n)]

    if math.log2(num_heads).is_integer():
        s = pow2_slopes(num_heads)
    else:
        p = 2 ** math.floor(math.log2(num_heads))
        extra = pow2_slopes(2 * p)[0::2][: num_heads - p]
        s = pow2_slopes(p) + extra
    return torch.tensor(s, dtype=torch.float32)


def attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    slopes: Optional[torch.Tensor] = None,
    dropout_p: float = 0.0,
    training: bool = False,
) -> torch.Tensor:
    """Causal multi-head attention with optional ALiBi bias, fp32 softmax.

    q, k, v: (B, H, T, D). Returns (B, H, T, D) in q.dtype.

    ALiBi bias for query i, key j (j <= i): slope_h * (j - i)  (<= 0).
    The reference applies the shift-invariant single-row form
    (layers.py:33-44); per-row they differ by a constant, so softmax output
    is identical.
    """
    B, H, T, D = q.shape
    scores = (q.float() @ k.float().transpose(-1, -2)) / math.sqrt(D)
    if slopes is not None:
        pos = torch.arange(T, device=q.device, dtype=torch.float32)
        rel = pos.view(1, T) - pos.view(T, 1)  # (j - i), negative below diag
        scores = scores + slopes.to(q.device).float().view(1, H, 1, 1) * rel.view(1, 1, T, T)
    causal = torch.ones(T, T, dtype=torch.bool, device=q.device).tril()
    scores = scores.masked_fill(~causal.view(1, 1, T, T), torch.finfo(torch.float32).min)
    probs = F.softmax(scores, dim=-1)
    if dropout_p > 0.0 and training:
        probs = F.dropout(probs, p=dropout_p)
    out = probs.to(v.dtype) @ v
    return out.to(q.dtype)


def drop_mask(seed: int, B: int, H: int, T: int, p: float) -> torch.Tensor:
    """Regenerate the attention-dropout keep mask of the HIP kernels.

    Exact Python mirror of csrc/common.h drop_bits32 (murmur3-finalizer
    mix32; 4x 8-bit thresholds per hash over key groups of 4). Returns a
    bool (B, H, T, T) tensor: True = keep. Used by GPU parity tests to
    compare kernel backward against autograd with the explicit mask.
    """
    import numpy as np

    thr = int(p * 256.0 + 0.5)

    def mix32(x):
        x = x.astype(np.uint32, copy=True)
        x ^= x >> np.uint32(16)
        x *= np.uint32(0x85EBCA6B)
        x ^= x >> np.uint32(13)
        x *= np.uint32(0xC2B2AE35)
        x ^= x >> np.uint32(16)
        return x

    bh = np.arange(B * H, dtype=np.uint32)
    qi = np.arange(T, dtype=np.uint32)
    kg = np.arange((T + 3) // 4, dtype=np.uint32)
    bhT_qi = (bh[:, None] * np.uint32(T) + qi[None, :]).reshape(-1, 1)  # (BH*T, 1)
    with np.errstate(over="ignore"):
        x = (
            np.uint32(seed)
            ^ (bhT_qi * np.uint32(0x9E3779B9))
            ^ (kg[None, :] * np.uint32(0x85EBCA6B))
        )
        bits = mix32(x)  # (BH*T, ceil(T/4))
    bytes_ = np.stack([(bits >> np.uint32(8 * e)) & np.uint32(0xFF) for e in range(4)], axis=-1)
    keep = (bytes_ >= np.uint32(thr)).reshape(B * H * T, -1)[:, :T]
    return torch.from_numpy(keep.reshape(B, H, T, T).copy())


def residual_drop_mask(seed: int, n: int, p: float) -> torch.Tensor:
    """Mirror of csrc/residual.hip's element mask: element i keeps iff byte
    (i & 3) of drop_bits32(seed, i4 >> 16, i4 & 0xffff) >= thr, i4 = i >> 2."""
    import numpy as np

    thr = int(p * 256.0 + 0.5)

    def mix32(x):
        x = x.astype(np.uint32, copy=True)
        x ^= x >> np.uint32(16)
        x *= np.uint32(0x85EBCA6B)
        x ^= x >> np.uint32(13)
        x *= np.uint32(0xC2B2AE35)
        x ^= x >> np.uint32(16)
        return x

    i4 = np.arange((n + 3) // 4, dtype=np.uint64)
    hi = (i4 >> np.uint64(16)).astype(np.uint32)
    lo = (i4 & np.uint64(0xFFFF)).astype(np.uint32)
    with np.errstate(over="ignore"):
        bits = mix32(
            np.uint32(seed) ^ (hi * np.uint32(0x9E3779B9)) ^ (lo * np.uint32(0x85EBCA6B))
        )
    by = np.stack([(bits >> np.uint32(8 * e)) & np.uint32(0xFF) for e in range(4)], -1)
    keep = (by >= np.uint32(thr)).reshape(-1)[:n]
    return torch.from_numpy(keep.copy())


def drop_inv_keep(p: float) -> float:
    """Rescale factor matching the kernels' realized 8-bit threshold."""
    thr = int(p * 256.0 + 0.5)
    return 256.0 / (256.0 - thr) if thr else 1.0


def attention_with_mask(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    slopes: Optional[torch.Tensor],
    keep: torch.Tensor,
    inv_keep: float,
) -> torch.Tensor:
    """Reference attention with an explicit dropout keep-mask on the probs
    (denominator keeps the full softmax, matching the fused kernel)."""
    B, H, T, D = q.shape
    scores = (q.float() @ k.float().transpose(-1, -2)) / math.sqrt(D)
    if slopes is not None:
        pos = torch.arange(T, device=q.device, dtype=torch.float32)
        rel = pos.view(1, T) - pos.view(T, 1)
        scores = scores + slopes.to(q.device).float().view(1, H, 1, 1) * rel.view(1, 1, T, T)
    causal = torch.ones(T, T, dtype=torch.bool, device=q.device).tril()
    scores = scores.masked_fill(~causal.view(1, 1, T, T), torch.finfo(torch.float32).min)
    probs = F.softmax(scores, dim=-1)
    probs = probs * keep.to(probs.dtype).to(probs.device) * inv_keep
    out = probs.to(v.dtype) @ v
    return out.to(q.dtype)


def layer_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    """Bias-free LayerNorm in fp32, cast back to input dtype.

    Matches flax nn.LayerNorm(use_bias=False) (reference src/models/GPT.py:42).
    """
    y = F.layer_norm(x.float(), (x.shape[-1],), weight=weight.float(), bias=None, eps=eps)
    return y.to(x.dtype)


def gelu(x: torch.Tensor) -> torch.Tensor:
    """tanh-approximate GELU (flax default, reference layers.py:68)."""
    return F.gelu(x, approximate="tanh")


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    """Mean cross entropy over all positions, fp32 log-softmax.

    logits: (N, V), targets: (N,) int64. Equivalent to the reference's
    one-hot * log_softmax mean (src/utils/losses.py:10-23) without
    materializing the one-hot.
    """
    return F.cross_entropy(logits.float(), targets, ignore_index=-1)


def adamw_update(
    param_f32: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    step: int,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    clip_value: float = 1.0,
    grad_scale: float = 1.0,
) -> None:
    """In-place AdamW on an fp32 master shard, reference semantics.

    Order matches the reference optax chain (main_zero.py:160-168):
    grad * grad_scale (the 1/accum divide of xmap_train_functions.py:81),
    element-wise clip of the gradient to +-clip_value (optax.clip(1.0) —
    NOT global-norm clipping), then scale_by_adam(b2=0.95), then masked
    weight decay, then -lr scaling.  Decoupled decay: update includes
    weight_decay * param (optax.add_decayed_weights semantics).
    """
    g = grad.float() * grad_scale
    if clip_value:
        g.clamp_(-clip_value, clip_value)
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    m_hat = exp_avg / bc1
    v_hat = exp_avg_sq / bc2
    update = m_hat / (v_hat.sqrt() + eps)
    if weight_decay:
        update = update + weight_decay * param_f32
    param_f32.add_(update, alpha=-lr)
