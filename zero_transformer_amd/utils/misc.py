"""Small training utilities (reference src/training/training_utils.py)."""

from __future__ import annotations


def compute_tokens_seen(absolute_step: int, max_context: int) -> int:
    """Tokens consumed after `absolute_step` steps at fixed packed context
    (reference training_utils.py:32-34; the packed batch always carries
    max_context tokens per sequence regardless of the train_context reshape).
    """
    return absolute_step * max_context
