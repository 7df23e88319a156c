"""zero_transformer_amd — MI355X-native ZeRO-1 GPT training framework.

A from-scratch AMD MI355X (gfx950 / CDNA4) framework with the capabilities of
the JAX/Flax reference (fattorib/ZeRO-transformer): GPT-2-style decoder-only
transformers with ALiBi attention, ZeRO stage-1 optimizer-state sharding over
RCCL/xGMI, bf16 activations with fp32 master params, and hand-written HIP
kernels (MFMA + LDS tiling) for the hot ops.

Layout:
    models/    GPT model, KV-cached inference model, model_getter
    ops/       HIP/CDNA4 kernels + CPU reference implementations
    parallel/  ZeRO-1 engine: bucketed reduce-scatter / all-gather over RCCL
    training/  trainer loop helpers
    utils/     config, LR schedule, data, checkpointing, metrics
"""

__version__ = "0.1.0"
