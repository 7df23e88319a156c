"""Attention kernel microbenchmark: TFLOP/s for fwd and bwd at the flagship
shape. Causal flops counted as 2*2*B*H*(T^2/2)*D per GEMM pair."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from zero_transformer_amd import ops
from zero_transformer_amd.ops import reference


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--B", type=int, default=8)
    p.add_argument("--H", type=int, default=16)
    p.add_argument("--T", type=int, default=2048)
    p.add_argument("--D", type=int, default=128)
    p.add_argument("--dropout", type=float, default=0.0)
    args = p.parse_args()
    B, H, T, D = args.B, args.H, args.T, args.D
    dev = torch.device("cuda")
    torch.manual_seed(0)
    qkv = torch.randn(B, T, 3 * H * D, device=dev).to(torch.bfloat16)
    slopes = reference.alibi_slopes(H).to(dev)
    ext = ops.hip_ops()

    fwd_flops = 2 * 2 * B * H * (T * T / 2) * D  # QK^T + PV, causal half
    t = bench(lambda: ext.attn_fwd(qkv, slopes, H, args.dropout, 7))
    print(f"fwd : {t*1e3:8.3f} ms  {fwd_flops/t/1e12:7.1f} TF/s")

    o, lse = ext.attn_fwd(qkv, slopes, H, args.dropout, 7)
    do = torch.randn_like(o)
    bwd_flops = fwd_flops * 2.5  # 5 GEMMs vs 2
    t = bench(lambda: ext.attn_bwd(do, qkv, slopes, o, lse, H, args.dropout, 7))
    print(f"bwd : {t*1e3:8.3f} ms  {bwd_flops/t/1e12:7.1f} TF/s")


if __name__ == "__main__":
    main()
