"""Node health check: RCCL all-reduce smoke test over all visible GPUs.

MI355X-native equivalent of the reference's TPU pod health script
(src/utils/pod_test.py:18-34 — a psum over global + local devices to detect
lost cores). Run with:

    torchrun --standalone --nproc-per-node <n_gpus> -m zero_transformer_amd.utils.gpu_health
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist


def main() -> None:
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    use_gpu = torch.cuda.is_available()
    backend = "nccl" if use_gpu else "gloo"
    if world > 1 or "MASTER_ADDR" in os.environ:
        dist.init_process_group(backend)
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0))) if use_gpu else torch.device("cpu")
    if use_gpu:
        torch.cuda.set_device(device)
    x = torch.ones(world if world else 1, device=device)
    if dist.is_initialized():
        dist.all_reduce(x)
    torch.cuda.synchronize() if use_gpu else None
    if rank == 0:
        print("world size:", world)
        print("device:", device, torch.cuda.get_device_name(device) if use_gpu else "")
        print("all_reduce result (expect all == world):", x.tolist())
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
