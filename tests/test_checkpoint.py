"""Weight checkpointing roundtrip (models/io.py, safetensors)."""

import os

import pytest
import torch

from senweaver_amd.models import LlamaModel, load_weights, save_weights, tiny_debug, tiny_moe


def _out(model, seed=3):
    toks = torch.randint(0, model.config.vocab_size, (1, 32),
                         generator=torch.Generator().manual_seed(seed))
    return model.prefill(toks)


def test_roundtrip_bf16(tmp_path):
    a = LlamaModel(tiny_debug(), device="cpu", seed=1)
    b = LlamaModel(tiny_debug(), device="cpu", seed=2)
    assert not torch.equal(_out(a), _out(b))
    p = str(tmp_path / "a.safetensors")
    n = save_weights(a, p)
    assert n > 4 and os.path.getsize(p) > 1000
    assert load_weights(b, p) == n
    torch.testing.assert_close(_out(a), _out(b))


def test_roundtrip_moe_fp8(tmp_path):
    a = LlamaModel(tiny_moe(), device="cpu", seed=5, quant="fp8")
    b = LlamaModel(tiny_moe(), device="cpu", seed=6, quant="fp8")
    p = str(tmp_path / "m.safetensors")
    save_weights(a, p)
    load_weights(b, p)
    torch.testing.assert_close(_out(a), _out(b))


def test_mismatch_raises(tmp_path):
    a = LlamaModel(tiny_debug(), device="cpu", seed=1)
    p = str(tmp_path / "a.safetensors")
    save_weights(a, p)
    with pytest.raises(ValueError, match="quant"):
        load_weights(LlamaModel(tiny_debug(), device="cpu", quant="fp8"), p)
    with pytest.raises(ValueError, match="model"):
        load_weights(LlamaModel(tiny_moe(), device="cpu"), p)


def test_tied_embeddings_stay_tied(tmp_path):
    import dataclasses
    cfg = dataclasses.replace(tiny_debug(), tie_embeddings=True)
    a = LlamaModel(cfg, device="cpu", seed=1)
    b = LlamaModel(cfg, device="cpu", seed=9)
    p = str(tmp_path / "t.safetensors")
    save_weights(a, p)
    load_weights(b, p)
    assert b.lm_head is b.embed  # the tie survives an in-place restore
    torch.testing.assert_close(_out(a), _out(b))
