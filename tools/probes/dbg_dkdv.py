"""Debug the dkdv kernel: tiny shapes, row-wise diff maps vs autograd."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))

import torch

from zero_transformer_amd import ops as O
from zero_transformer_amd.ops import reference

dev = torch.device("cuda", 0)
torch.manual_seed(0)


def run(B, H, T, D):
    q, k, v = (torch.randn(B, H, T, D, device=dev).to(torch.bfloat16) for _ in range(3))
    slopes = reference.alibi_slopes(H).to(dev)
    ext = O.hip_ops()
    C = H * D
    qkv = torch.cat([t.transpose(1, 2).reshape(B, T, C) for t in (q, k, v)], -1).contiguous()
    o, lse = ext.attn_fwd(qkv, slopes, H, 0.0, 0)
    do = torch.randn_like(o)
    (dqkv,) = ext.attn_bwd(do, qkv, slopes, o, lse, H, 0.0, 0)
    dq, dk, dv = (t.view(B, T, H, D).transpose(1, 2).contiguous() for t in dqkv.split(C, dim=-1))
    o = o.view(B, T, H, D).transpose(1, 2).contiguous()
    do = do.view(B, T, H, D).transpose(1, 2)

    qr, kr, vr = (t.detach().float().requires_grad_(True) for t in (q, k, v))
    ref = reference.attention(qr, kr, vr, slopes)
    ref.backward(do.float())
    print(f"== B{B} H{H} T{T} D{D}")
    od = (o.float() - ref).abs()
    print(f"o : rel-max {od.max().item()/(ref.abs().max().item()+1e-6):.4f}  "
          f"bad rows: {(od[0,0].amax(dim=1) > 0.03).nonzero().flatten().tolist()[:20]}")
    # lse check
    sc = (qr.detach() @ kr.detach().transpose(-1, -2)) / (D ** 0.5)
    pos = torch.arange(T, device=dev, dtype=torch.float32)
    rel = pos.view(1, T) - pos.view(T, 1)
    sc = sc + slopes.float().view(1, H, 1, 1) * rel.view(1, 1, T, T)
    sc = sc.masked_fill(~torch.ones(T, T, dtype=torch.bool, device=dev).tril().view(1, 1, T, T),
                        float("-inf"))
    lse_ref = torch.logsumexp(sc, dim=-1)
    ld = (lse - lse_ref).abs()
    print(f"lse: max-abs {ld.max().item():.5f}")
    for got, want, name in [(dq, qr.grad, "dq"), (dk, kr.grad, "dk"), (dv, vr.grad, "dv")]:
        d = (got.float() - want).abs()
        sc = want.abs().max().item() + 1e-6
        print(f"{name}: rel-max {d.max().item()/sc:.4f}  ", end="")
        rowdiff = d[0, 0].amax(dim=1)  # per-row max
        bad = (rowdiff / sc > 0.05).nonzero().flatten().tolist()
        print(f"bad rows: {bad[:20]}{'...' if len(bad) > 20 else ''}")
        if name in ("dk", "dv") and bad:
            # column structure of first bad row
            r0 = bad[0]
            cd = d[0, 0, r0] / sc
            badc = (cd > 0.05).nonzero().flatten().tolist()
            print(f"   row {r0}: bad cols {badc[:24]}{'...' if len(badc) > 24 else ''}")
            print(f"   got[{r0},0:8] ", got[0, 0, r0, :8].float().tolist())
            print(f"   want[{r0},0:8]", want[0, 0, r0, :8].tolist())


run(1, 1, 32, 128)
run(1, 1, 64, 128)
run(1, 1, 256, 128)
run(1, 2, 96, 64)
