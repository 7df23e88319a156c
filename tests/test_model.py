"""Model component tests, mirroring the reference's test strategy
(tests/test_model_components.py, test_model_factory.py)."""

import math

import pytest
import torch

from zero_transformer_amd.models import GPT, Block, CausalSelfAttention, MLP, model_getter
from zero_transformer_amd.models.gpt import LayerNorm
from zero_transformer_amd.ops import reference
from zero_transformer_amd.utils.config import DotDict

CFG = DotDict(
    embedding_dim=64,
    vocab_size=256,
    num_head=4,
    block_size=32,
    dropout=0.1,
    N=2,
    alibi_attn=True,
)


def test_mlp_shape():
    m = MLP(CFG)
    x = torch.randn(2, 32, 64)
    assert m(x).shape == (2, 32, 64)


@pytest.mark.parametrize("alibi", [True, False])
def test_attention_shape(alibi):
    cfg = DotDict(dict(CFG, alibi_attn=alibi))
    attn = CausalSelfAttention(cfg)
    x = torch.randn(2, 32, 64)
    assert attn(x).shape == (2, 32, 64)


def test_block_shape():
    blk = Block(CFG)
    x = torch.randn(2, 32, 64)
    assert blk(x).shape == (2, 32, 64)


def test_transformer_logits_shape():
    model = GPT(CFG)
    idx = torch.randint(0, 256, (2, 32))
    logits = model(idx)
    assert logits.shape == (2, 32, 256)


def test_transformer_loss_matches_external_ce():
    """Model-internal loss == externally computed shifted CE
    (reference tests/test_model_components.py:232-262)."""
    model = GPT(CFG).eval()
    idx = torch.randint(0, 256, (2, 32))
    with torch.no_grad():
        logits, loss = model(idx, labels=idx)
        ext = reference.cross_entropy(
            logits[..., :-1, :].reshape(-1, 256), idx[..., 1:].reshape(-1)
        )
    assert torch.allclose(loss, ext, atol=1e-5)


def test_bf16_forward():
    model = GPT(CFG).to(torch.bfloat16).eval()
    idx = torch.randint(0, 256, (2, 32))
    with torch.no_grad():
        logits = model(idx)
    assert logits.dtype == torch.bfloat16
    assert torch.isfinite(logits.float()).all()


def test_causality():
    """Changing a future token must not change earlier logits."""
    model = GPT(DotDict(dict(CFG, dropout=0.0))).eval()
    a = torch.randint(0, 256, (1, 32))
    b = a.clone()
    b[0, -1] = (b[0, -1] + 1) % 256
    with torch.no_grad():
        la, lb = model(a), model(b)
    assert torch.allclose(la[0, :-1], lb[0, :-1], atol=1e-5)
    assert not torch.allclose(la[0, -1], lb[0, -1], atol=1e-5)


def test_weight_tying():
    model = GPT(CFG)
    assert model.lm_head.weight.data_ptr() == model.wte.weight.data_ptr()
    sd = model.state_dict()
    assert "lm_head.weight" in sd and "wte.weight" in sd


def test_init_scaling():
    model = GPT(DotDict(dict(CFG, N=8)))
    resid = model.blocks[0].attn.fc_resid.weight
    base = model.blocks[0].attn.qkv_w
    # residual-out init std is 0.02/sqrt(2N)
    assert resid.std().item() < base.std().item()
    assert abs(resid.std().item() - 0.02 / math.sqrt(16)) < 2e-3


def test_layernorm_no_bias():
    ln = LayerNorm(64)
    assert not hasattr(ln, "bias") or ln.bias is None
    x = torch.randn(4, 64)
    y = ln(x)
    assert torch.allclose(y.mean(-1), torch.zeros(4), atol=1e-5)


def test_model_factory():
    model, cfg = model_getter("test", config_path="conf/model_config.yaml", return_cfg=True)
    assert model.N == cfg.N == 2
    with pytest.raises(AssertionError):
        model_getter("nonexistent", config_path="conf/model_config.yaml")
    with pytest.raises(AssertionError):
        model_getter("test", config_path="conf/model_config.yaml", dtype=torch.int8)


def test_pth_contract_keys():
    """State dict follows the torch_compatability .pth key layout
    (reference flax_to_pytorch.py:10-35,96-114)."""
    model = GPT(CFG)
    keys = set(model.state_dict().keys())
    expect = {"wte.weight", "norm.weight", "lm_head.weight"}
    for i in range(CFG.N):
        expect |= {
            f"blocks.{i}.attn.query.weight",
            f"blocks.{i}.attn.key.weight",
            f"blocks.{i}.attn.value.weight",
            f"blocks.{i}.attn.fc_resid.weight",
            f"blocks.{i}.mlp.fc1.weight",
            f"blocks.{i}.mlp.fc_resid.weight",
            f"blocks.{i}.ln1.weight",
            f"blocks.{i}.ln2.weight",
        }
    assert expect == keys
