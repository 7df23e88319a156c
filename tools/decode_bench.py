"""KV-cached decode throughput of the inference model (serving evidence).

    python tools/decode_bench.py --model-size 1_3b --tokens 128 --batch 1
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from zero_transformer_amd.models.inference import model_getter
from zero_transformer_amd.utils import gemm_tune


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model-size", default="1_3b")
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--prompt-len", type=int, default=128)
    p.add_argument("--tokens", type=int, default=128)
    p.add_argument("--dtype", default="fp16", choices=["fp16", "bf16"])
    p.add_argument("--fast", action="store_true",
                   help="static KV cache + hipGraph-captured decode step")
    args = p.parse_args()
    gemm_tune.enable()  # committed hipBLASLt tunings (incl. decode shapes)
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    model = model_getter(args.model_size).to(dev)
    model = (model.half() if args.dtype == "fp16" else model.to(torch.bfloat16)).eval()

    idx = torch.randint(0, model.vocab_size, (args.batch, args.prompt_len), device=dev)

    @torch.no_grad()
    def run():
        logits, presents = model(idx, use_cache=True)
        nxt = logits[:, -1:].argmax(-1)
        t0 = None
        for i in range(args.tokens):
            if i == 8:  # warm
                torch.cuda.synchronize()
                t0 = time.perf_counter()
            logits, presents = model(nxt, use_cache=True, past_states=presents)
            nxt = logits[:, -1:].argmax(-1)
        torch.cuda.synchronize()
        return (args.tokens - 8) / (time.perf_counter() - t0)

    if args.fast:
        from zero_transformer_amd.models.inference import generate_fast

        @torch.no_grad()
        def run():
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            generate_fast(model, idx, args.tokens)
            torch.cuda.synchronize()
            return args.tokens / (time.perf_counter() - t0)

    run()  # warmup pass
    tps = run()
    mode = "fast(graph)" if args.fast else "dynamic"
    print(f"{args.model_size} {args.dtype} batch {args.batch} {mode}: "
          f"{tps * args.batch:.1f} tokens/s ({1e3 / tps:.2f} ms/token)")


if __name__ == "__main__":
    main()
