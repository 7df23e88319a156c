"""torch_compatability contract tests, mirroring the reference's
test_flax_conversion.py (round-trip) and test_torch_models.py (KV cache)."""

import os
import tempfile

import pytest
import torch

from torch_compatability.GPT2 import GPT2, model_getter
from torch_compatability.convert_to_torch import match_and_save
from zero_transformer_amd.models import GPT
from zero_transformer_amd.parallel.zero import ZeRO1Optimizer
from zero_transformer_amd.utils.config import DotDict
from zero_transformer_amd.utils import checkpoint as ckpt

CFG = DotDict(
    embedding_dim=64, vocab_size=256, num_head=4, block_size=32,
    dropout=0.0, N=2, alibi_attn=True,
)


def test_conversion_roundtrip():
    """Trainer checkpoint -> inference .pth -> identical tensors + identical
    logits between training model (eval) and inference model."""
    torch.manual_seed(0)
    train_model = GPT(CFG).eval()
    opt = ZeRO1Optimizer(list(train_model.named_parameters()), lr=1e-3)
    with tempfile.TemporaryDirectory() as td:
        sd = opt.full_param_state_dict()
        path = ckpt.save_checkpoint_params(td, 1, sd)
        inf_model = model_getter("test", config_path="torch_compatability/model_config.yaml")
        out = os.path.join(td, "torch_test.pth")
        match_and_save(inf_model, path, out)
        # every mapped tensor identical
        reloaded = torch.load(out, map_location="cpu", weights_only=True)
        for k, v in train_model.state_dict().items():
            assert torch.allclose(v.float(), reloaded[k], atol=1e-6), k
        # logits parity
        idx = torch.randint(0, 256, (2, 16))
        with torch.no_grad():
            lt = train_model(idx)
            li = inf_model.eval()(idx)
        assert torch.allclose(lt.float(), li.float(), atol=1e-4)


def test_kv_cache_growth():
    """KV cache shapes grow across decode steps (reference
    test_torch_models.py:42-91,111-160)."""
    model = model_getter("test", config_path="torch_compatability/model_config.yaml").eval()
    idx = torch.randint(0, 256, (1, 8))
    with torch.no_grad():
        logits, states = model(idx, use_cache=True)
        assert len(states) == model.N
        assert states[0][0].shape == (1, 4, 8, 16)  # (B, H, T, D)
        nxt = logits[:, -1:].argmax(-1)
        logits2, states2 = model(nxt, use_cache=True, past_states=states)
        assert states2[0][0].shape == (1, 4, 9, 16)
        assert logits2.shape == (1, 1, 256)


def test_cached_matches_uncached_decode():
    """Greedy decode with KV cache == full re-forward decode."""
    torch.manual_seed(1)
    model = model_getter("test", config_path="torch_compatability/model_config.yaml").eval()
    idx = torch.randint(0, 256, (1, 8))
    # cached
    with torch.no_grad():
        logits, states = model(idx, use_cache=True)
        toks = [int(logits[0, -1].argmax())]
        cur = torch.tensor([[toks[-1]]])
        for _ in range(4):
            logits, states = model(cur, use_cache=True, past_states=states)
            toks.append(int(logits[0, -1].argmax()))
            cur = torch.tensor([[toks[-1]]])
        # uncached
        full = model.generate(idx, max_new_tokens=5)
    assert toks == full[0, 8:].tolist()


def test_gpt2_forward_with_labels():
    model = model_getter("test", config_path="torch_compatability/model_config.yaml")
    idx = torch.randint(0, 256, (2, 16))
    logits, loss = model(idx, labels=idx)
    assert logits.shape == (2, 16, 256)
    assert loss.dim() == 0 and torch.isfinite(loss)


def test_factory_errors():
    with pytest.raises(AssertionError):
        model_getter("nope", config_path="torch_compatability/model_config.yaml")


def test_extend_params_depth_doubling():
    from zero_transformer_amd.utils.extend_params import create_mapping, extend_params

    torch.manual_seed(2)
    model = GPT(CFG)
    sd = {k: v.float() for k, v in model.state_dict().items()}
    out = extend_params(sd, CFG.N)
    assert create_mapping(2) == {0: 0, 1: 0, 2: 1, 3: 1}
    big = GPT(DotDict(dict(CFG, N=4)))
    big.load_state_dict(out)
    for j in range(4):
        src = j // 2
        assert torch.allclose(
            out[f"blocks.{j}.attn.query.weight"], sd[f"blocks.{src}.attn.query.weight"]
        )
    assert torch.allclose(out["wte.weight"], sd["wte.weight"])
