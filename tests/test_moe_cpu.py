"""MoE (Mixtral-style) grouped-GEMM path — CPU tests with the tiny model."""

import pytest
import torch

from senweaver_amd import ops
from senweaver_amd.models import tiny_moe
from senweaver_amd.models.llama import LlamaModel


def test_grouped_gemm_cpu_matches_per_expert():
    torch.manual_seed(0)
    E, K, N = 3, 128, 256
    seg = [0, 5, 5, 12]  # expert 1 empty
    a = torch.randn(12 + 128, K, dtype=torch.bfloat16)
    w = torch.randn(E, N, K, dtype=torch.bfloat16)
    c = ops.grouped_gemm_bt(a, w, seg)
    for e in range(E):
        s, t = seg[e], seg[e + 1]
        if t > s:
            ref = (a[s:t].float() @ w[e].float().t()).to(torch.bfloat16)
            torch.testing.assert_close(c[s:t].float(), ref.float(), atol=0.1, rtol=3e-2)


def moe_ffn_naive(model, h):
    """Per-token loop reference for the routed FFN (fp32 weights math kept
    at the same bf16 points as the kernel path)."""
    c = model.config
    L = model.layers[0]
    out = torch.zeros(h.shape[0], c.hidden_size, dtype=torch.float32)
    logits = h @ L["router"].t()
    probs = torch.softmax(logits.float(), dim=-1)
    topw, topi = probs.topk(c.num_experts_per_tok, dim=-1)
    topw = topw / topw.sum(-1, keepdim=True)
    for t in range(h.shape[0]):
        for j in range(c.num_experts_per_tok):
            e = int(topi[t, j])
            gateup = (h[t:t + 1].float() @ L["w13"][e].float().t()).to(torch.bfloat16)
            inter = c.intermediate_size
            g, u = gateup[..., :inter].float(), gateup[..., inter:].float()
            act = (torch.nn.functional.silu(g) * u).to(torch.bfloat16)
            down = (act.float() @ L["w2"][e].float().t()).to(torch.bfloat16)
            out[t] += float(topw[t, j]) * down[0].float()
    return out.to(torch.bfloat16)


def test_moe_ffn_matches_naive():
    torch.manual_seed(1)
    model = LlamaModel(tiny_moe(), device="cpu")
    h = torch.randn(9, 256, dtype=torch.bfloat16)
    got = model._moe_ffn(h, model.layers[0])
    want = moe_ffn_naive(model, h)
    torch.testing.assert_close(got.float(), want.float(), atol=5e-2, rtol=5e-2)


def test_moe_model_forward():
    model = LlamaModel(tiny_moe(), device="cpu")
    tokens = torch.randint(0, 512, (2, 64))
    hidden = model.prefill(tokens)
    assert hidden.shape == (2, 64, 256)
    assert torch.isfinite(hidden.float()).all()
    logits = model.logits(hidden[:, -1])
    assert logits.shape == (2, 512)


def test_moe_deterministic():
    m1 = LlamaModel(tiny_moe(), device="cpu", seed=3)
    m2 = LlamaModel(tiny_moe(), device="cpu", seed=3)
    tokens = torch.randint(0, 512, (1, 64))
    h1 = m1.prefill(tokens)
    h2 = m2.prefill(tokens)
    assert torch.equal(h1, h2)
