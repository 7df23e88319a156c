"""Tune hipBLASLt algorithms for the skinny decode GEMV shapes (GPU box).

Runs a few static-cache decode steps of the inference model under TunableOp
tuning for batch 1 and 16; results append to profiles/tunableop_gfx950.csv.

    python tools/tune_decode.py [--model-size 1_3b]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from zero_transformer_amd.utils import gemm_tune


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model-size", default="1_3b")
    args = p.parse_args()
    assert torch.cuda.is_available()
    assert gemm_tune.enable(tuning=True), "TunableOp unavailable"
    torch.cuda.tunable.set_max_tuning_duration(50)

    from zero_transformer_amd.models.inference import generate_fast, model_getter

    dev = torch.device("cuda", 0)
    model = model_getter(args.model_size).to(dev).half().eval()
    for b in (1, 16):
        idx = torch.randint(0, model.vocab_size, (b, 64), device=dev)
        generate_fast(model, idx, 4, use_graph=False)  # tuning + capture don't mix
    torch.cuda.synchronize()
    print("tuned decode shapes; results flush to", gemm_tune.RESULTS, "at exit")


if __name__ == "__main__":
    main()
