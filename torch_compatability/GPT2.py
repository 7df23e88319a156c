"""Compatibility surface for the reference's torch_compatability/GPT2.py:
re-exports the KV-cached inference model and its model_getter. External
users (e.g. lm-eval-harness wiring) import `torch_compatability.GPT2`.
"""

from zero_transformer_amd.models.inference import (  # noqa: F401
    GPT2,
    InferenceAttention,
    InferenceBlock,
    InferenceMLP,
    model_getter,
)

# Reference-name aliases (torch_compatability/GPT2.py:49,84,248)
MLPBlock = InferenceMLP
ALiBi = InferenceAttention
GPT2Block = InferenceBlock
