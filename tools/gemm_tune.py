"""Tune hipBLASLt GEMM algorithms for the training shapes (run on a GPU box).

Runs TunableOp in tuning mode over a couple of train steps of each model
config (the full fwd+bwd GEMM shape set: qkv/attn-out/mlp/lm-head and their
gradients), then writes profiles/tunableop_gfx950.csv.

    python tools/gemm_tune.py [--models 1_3b_2048 760m] [--batch 8]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from zero_transformer_amd.utils import gemm_tune


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--models", nargs="+", default=["1_3b_2048"])
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--steps", type=int, default=2)
    p.add_argument("--duration", type=int, default=100, help="ms per candidate algo")
    p.add_argument("--iters", type=int, default=0, help="max tuning iterations (0: default)")
    args = p.parse_args()
    assert torch.cuda.is_available()
    assert gemm_tune.enable(tuning=True), "TunableOp unavailable"
    torch.cuda.tunable.set_max_tuning_duration(args.duration)
    if args.iters:
        torch.cuda.tunable.set_max_tuning_iterations(args.iters)

    from zero_transformer_amd.models import model_getter
    from zero_transformer_amd.parallel.zero import ZeRO1Optimizer
    from zero_transformer_amd.training.trainer import TrainEngine

    dev = torch.device("cuda", 0)
    for name in args.models:
        model, cfg = model_getter(name, return_cfg=True)
        model = model.to(dev)
        opt = ZeRO1Optimizer(
            list(model.named_parameters()), lr=1e-4, param_dtype=torch.bfloat16
        )
        eng = TrainEngine(model, opt, 1, int(cfg.block_size), dev)
        batch = torch.randint(0, cfg.vocab_size, (args.batch, int(cfg.block_size)), device=dev)
        for _ in range(args.steps):
            eng.train_step(batch)
        torch.cuda.synchronize()
        del model, opt, eng, batch
        torch.cuda.empty_cache()
        print(f"tuned {name}")
    # TunableOp writes the results file automatically at process shutdown
    os.makedirs(os.path.dirname(gemm_tune.RESULTS), exist_ok=True)
    print("results will be written to", gemm_tune.RESULTS)


if __name__ == "__main__":
    main()
