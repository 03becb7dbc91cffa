"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32 reference.

All inputs are random (never zero-filled / symmetric — transpose-detecting per
the CDNA4 methodology rules).  Run on an MI355X with `pytest -m gpu`.
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from senweaver_amd import ops
from senweaver_amd.ops import reference as ref


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    return torch.device("cuda:0")


def test_extension_is_native(dev):
    # the HIP extension must load on a GPU box — no silent eager fallback
    assert ops.extension_loaded(), "HIP extension failed to build/load"
    import senweaver_amd.ops.build as b
    import os
    assert os.path.exists(os.path.join(b.BUILD_DIR, b.EXT_NAME + ".so"))


def test_rmsnorm(dev):
    x = torch.randn(512, 4096, dtype=torch.bfloat16, device=dev)
    w = torch.randn(4096, dtype=torch.bfloat16, device=dev)
    y = ops.rmsnorm(x, w, 1e-5)
    y_ref = ref.rmsnorm_ref(x, w, 1e-5)
    torch.testing.assert_close(y.float(), y_ref.float(), atol=2e-2, rtol=2e-2)


def test_fused_add_rmsnorm(dev):
    x = torch.randn(256, 4096, dtype=torch.bfloat16, device=dev)
    res = torch.randn(256, 4096, dtype=torch.bfloat16, device=dev)
    w = torch.randn(4096, dtype=torch.bfloat16, device=dev)
    res_ref_in = res.clone()
    y = ops.fused_add_rmsnorm(x, res, w, 1e-5)
    y_ref, r_ref = ref.fused_add_rmsnorm_ref(x, res_ref_in, w, 1e-5)
    torch.testing.assert_close(res.float(), r_ref.float(), atol=1e-2, rtol=1e-2)
    torch.testing.assert_close(y.float(), y_ref.float(), atol=2e-2, rtol=2e-2)


def test_rope(dev):
    T, Hq, Hk, D = 333, 8, 2, 128
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(T, Hk, D, dtype=torch.bfloat16, device=dev)
    cos_sin = ops.rope_tables(4096, D).to(dev)
    pos = torch.randint(0, 4096, (T,), dtype=torch.int32, device=dev)
    q_ref = ref.rope_ref(q, cos_sin, pos)
    k_ref = ref.rope_ref(k, cos_sin, pos)
    ops.rope_inplace(q, k, cos_sin, pos)
    torch.testing.assert_close(q.float(), q_ref.float(), atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(k.float(), k_ref.float(), atol=2e-2, rtol=2e-2)


def test_swiglu(dev):
    x = torch.randn(1000, 2 * 1024, dtype=torch.bfloat16, device=dev)
    y = ops.swiglu(x)
    y_ref = ref.swiglu_ref(x)
    torch.testing.assert_close(y.float(), y_ref.float(), atol=2e-2, rtol=2e-2)


def test_gemm_layout_sanity(dev):
    # patterned asymmetric inputs catch row/col transposes exactly
    M, N, K = 128, 128, 64
    a = (torch.arange(M, device=dev).unsqueeze(1) * 0.01 +
         torch.arange(K, device=dev).unsqueeze(0) * 0.001).bfloat16()
    b = (torch.arange(N, device=dev).unsqueeze(1) * 0.02 -
         torch.arange(K, device=dev).unsqueeze(0) * 0.003).bfloat16()
    c = ops.gemm_bt_tiled(a, b)
    c_ref = ref.gemm_bt_ref(a, b)
    torch.testing.assert_close(c.float(), c_ref.float(), atol=5e-2, rtol=2e-2)


@pytest.mark.parametrize("M,N,K", [(256, 256, 512), (384, 1024, 4096),
                                   (1024, 14336, 4096), (2048, 6144, 4096)])
def test_gemm_random(dev, M, N, K):
    a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    b = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    c = ops.gemm_bt_tiled(a, b)
    c_ref = ref.gemm_bt_ref(a, b)
    torch.testing.assert_close(c.float(), c_ref.float(), atol=0.5, rtol=3e-2)


def test_gemm_8ph_layout_sanity(dev):
    # patterned asymmetric inputs catch row/col transposes exactly
    M, N, K = 256, 256, 128
    a = (torch.arange(M, device=dev).unsqueeze(1) * 0.01 +
         torch.arange(K, device=dev).unsqueeze(0) * 0.001).bfloat16()
    b = (torch.arange(N, device=dev).unsqueeze(1) * 0.02 -
         torch.arange(K, device=dev).unsqueeze(0) * 0.003).bfloat16()
    c = ops.hip_ext().gemm_bt_8ph(a, b)
    c_ref = ref.gemm_bt_ref(a, b)
    torch.testing.assert_close(c.float(), c_ref.float(), atol=5e-2, rtol=2e-2)


@pytest.mark.parametrize("M,N,K", [(256, 256, 128), (512, 512, 256),
                                   (2048, 6144, 4096), (4096, 4096, 4096),
                                   (2048, 4096, 14336)])
def test_gemm_8ph_random(dev, M, N, K):
    # race screen: two independent random rounds per shape (deep-pipelined
    # schedule; any slot-overwrite race shows as data-dependent corruption)
    for seed in (0, 1):
        g = torch.Generator(device="cpu").manual_seed(seed)
        a = torch.randn(M, K, generator=g).bfloat16().to(dev)
        b = torch.randn(N, K, generator=g).bfloat16().to(dev)
        c = ops.hip_ext().gemm_bt_8ph(a, b)
        c_ref = ref.gemm_bt_ref(a, b)
        torch.testing.assert_close(c.float(), c_ref.float(), atol=0.5, rtol=3e-2)


@pytest.mark.parametrize("M,N,K", [(256, 256, 128), (512, 512, 512),
                                   (2048, 6144, 4096), (4096, 4096, 4096),
                                   (2048, 4096, 14336)])
def test_gemm_pipe16w_random(dev, M, N, K):
    """The shipped 16-wave pipeline kernel (v13/v14 = the gemm_bt dispatch
    for big shapes): numerics + determinism race screen."""
    ext = ops.hip_ext()
    for seed in (0, 1):
        g = torch.Generator(device="cpu").manual_seed(seed)
        a = torch.randn(M, K, generator=g).bfloat16().to(dev)
        b = torch.randn(N, K, generator=g).bfloat16().to(dev)
        c = ext.gemm_bt_8ph_v(a, b, 14)
        c2 = ext.gemm_bt_8ph_v(a, b, 14)
        assert torch.equal(c, c2), "nondeterministic output (pipeline race)"
        c_ref = ref.gemm_bt_ref(a, b)
        torch.testing.assert_close(c.float(), c_ref.float(), atol=0.5, rtol=3e-2)


@pytest.mark.parametrize("var", [18, 21])
def test_gemm_asm_random(dev, var):
    """Hand-scheduled asm kernels (v18/v21 = the M>=4096 dispatch):
    numerics + determinism at two shapes."""
    ext = ops.hip_ext()
    for (M, N, K) in [(512, 512, 512), (4096, 4096, 4096)]:
        g = torch.Generator(device="cpu").manual_seed(3)
        a = torch.randn(M, K, generator=g).bfloat16().to(dev)
        b = torch.randn(N, K, generator=g).bfloat16().to(dev)
        c = ext.gemm_bt_8ph_v(a, b, var)
        assert torch.equal(c, ext.gemm_bt_8ph_v(a, b, var))
        torch.testing.assert_close(c.float(), ref.gemm_bt_ref(a, b).float(),
                                   atol=0.5, rtol=3e-2)


def test_gemm_pad_m(dev):
    # M not a multiple of 128 goes through the host-side pad
    a = torch.randn(300, 512, dtype=torch.bfloat16, device=dev)
    b = torch.randn(256, 512, dtype=torch.bfloat16, device=dev)
    c = ops.gemm_bt_tiled(a, b)
    assert c.shape == (300, 256)
    torch.testing.assert_close(c.float(), ref.gemm_bt_ref(a, b).float(), atol=0.5, rtol=3e-2)


def test_gemv_skinny(dev):
    a = torch.randn(4, 4096, dtype=torch.bfloat16, device=dev)
    b = torch.randn(1024, 4096, dtype=torch.bfloat16, device=dev)
    c = ops.gemm_bt(a, b)
    torch.testing.assert_close(c.float(), ref.gemm_bt_ref(a, b).float(), atol=0.5, rtol=3e-2)


@pytest.mark.parametrize("B,H,Hk,S", [(2, 8, 2, 256), (1, 32, 8, 1024), (2, 4, 4, 512)])
def test_attn_fwd(dev, B, H, Hk, S):
    D = 128
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(D)
    o = ops.attn_fwd(q, k, v, scale)
    o_ref = ref.attn_fwd_ref(q, k, v, scale, causal=True)
    torch.testing.assert_close(o.float(), o_ref.float(), atol=8e-2, rtol=8e-2)


def test_attn_fwd_spiked_softmax(dev):
    # force large rescales: one huge K row against each Q block
    B, H, Hk, S, D = 1, 2, 2, 512, 128
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    k[:, :, 100] *= 8  # spike inside an early tile
    v = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(D)
    o = ops.attn_fwd(q, k, v, scale)
    o_ref = ref.attn_fwd_ref(q, k, v, scale, causal=True)
    torch.testing.assert_close(o.float(), o_ref.float(), atol=8e-2, rtol=8e-2)


def test_paged_decode_attn(dev):
    B, H, Hk, D = 3, 8, 2, 128
    P, page = 64, 16
    kcache = torch.randn(P, page, Hk, D, dtype=torch.bfloat16, device=dev)
    vcache = torch.randn(P, page, Hk, D, dtype=torch.bfloat16, device=dev)
    ctx = torch.tensor([37, 200, 128], dtype=torch.int32, device=dev)
    max_pages = 16
    bt = torch.full((B, max_pages), -1, dtype=torch.int32, device=dev)
    used = 0
    for b in range(B):
        n = (int(ctx[b]) + page - 1) // page
        bt[b, :n] = torch.arange(used, used + n, dtype=torch.int32, device=dev)
        used += n
    q = torch.randn(B, H, D, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(D)
    o = ops.paged_decode_attn(q, kcache, vcache, bt, ctx, scale)
    o_ref = ref.paged_decode_attn_ref(q, kcache, vcache, bt.cpu(), ctx.cpu(), scale)
    torch.testing.assert_close(o.float(), o_ref.float(), atol=5e-2, rtol=5e-2)


def test_argmax_rows(dev):
    logits = torch.randn(64, 128256, dtype=torch.bfloat16, device=dev)
    idx = ops.argmax_rows(logits)
    idx_ref = ref.argmax_rows_ref(logits)
    torch.testing.assert_close(idx.long(), idx_ref.long())


def test_target_logprob(dev):
    logits = torch.randn(64, 128256, dtype=torch.bfloat16, device=dev) * 4
    targets = torch.randint(0, 128256, (64,), dtype=torch.int32, device=dev)
    lp = ops.target_logprob(logits, targets)
    lp_ref = ref.target_logprob_ref(logits, targets)
    torch.testing.assert_close(lp, lp_ref, atol=1e-3, rtol=1e-3)


def test_grouped_gemm_moe(dev):
    torch.manual_seed(3)
    E, K, N = 8, 4096, 1024
    # uneven segments incl. an empty expert (router imbalance)
    sizes = [300, 0, 77, 1024, 13, 512, 250, 128]
    seg = [0]
    for s in sizes:
        seg.append(seg[-1] + s)
    T = seg[-1]
    a = torch.randn(T + 128, K, dtype=torch.bfloat16, device=dev)
    w = torch.randn(E, N, K, dtype=torch.bfloat16, device=dev)
    c = ops.grouped_gemm_bt(a, w, seg)
    for e in range(E):
        s, t = seg[e], seg[e + 1]
        if t > s:
            r = ref.gemm_bt_ref(a[s:t], w[e])
            torch.testing.assert_close(c[s:t].float(), r.float(), atol=0.5, rtol=3e-2)


def test_mixtral_tiny_forward_gpu(dev):
    from senweaver_amd.models import tiny_moe
    from senweaver_amd.models.llama import LlamaModel
    model = LlamaModel(tiny_moe(), device=dev)
    tokens = torch.randint(0, 512, (2, 64), device=dev)
    h = model.prefill(tokens)
    assert torch.isfinite(h.float()).all()


def test_quant_fp8_gpu(dev):
    x = torch.randn(64, 512, dtype=torch.bfloat16, device=dev) * 2
    q, s = ops.quant_fp8(x)
    qr, sr = ref.quant_fp8_ref(x.cpu())
    torch.testing.assert_close(s.cpu(), sr, atol=1e-3, rtol=1e-3)
    back = q.cpu().view(torch.float8_e4m3fn).float() * s.cpu().unsqueeze(1)
    back_ref = qr.view(torch.float8_e4m3fn).float() * sr.unsqueeze(1)
    # the kernel multiplies by 1/scale while the reference divides: exact
    # e4m3 ties can land one quantum apart.  Require (a) few mismatches and
    # (b) the GPU roundtrip approximates x no worse than the CPU one
    d = (back - back_ref).abs()
    assert (d > 0).float().mean() < 0.01
    err_gpu = (back - x.cpu().float()).abs().max()
    err_cpu = (back_ref - x.cpu().float()).abs().max()
    assert err_gpu <= err_cpu * 1.5 + 1e-3


def test_gemm_fp8_gpu(dev):
    torch.manual_seed(5)
    M, N, K = 256, 1024, 4096
    a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    b = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    aq, asc = ops.quant_fp8(a)
    bq, bsc = ops.quant_fp8(b)
    c = ops.gemm_bt_fp8(aq, asc, bq, bsc)
    c_ref = ref.gemm_bt_fp8_ref(aq.cpu(), asc.cpu(), bq.cpu(), bsc.cpu())
    torch.testing.assert_close(c.float().cpu(), c_ref.float(), atol=1.0, rtol=5e-2)


def test_fp8_model_forward_gpu(dev):
    from senweaver_amd.models import tiny_debug
    from senweaver_amd.models.llama import LlamaModel
    m = LlamaModel(tiny_debug(), device=dev, quant="fp8")
    tokens = torch.randint(0, 512, (1, 128), device=dev)
    h = m.prefill(tokens)
    assert torch.isfinite(h.float()).all()


def test_quant_mxfp8_gpu(dev):
    torch.manual_seed(11)
    # per-block dynamic range exercises the e8m0 block scales
    x = (torch.randn(64, 512) * torch.exp2(torch.randint(-8, 8, (64, 512)).float()))
    x = x.to(torch.bfloat16).to(dev)
    q, s = ops.quant_mxfp8(x)
    qr, sr = ref.quant_mxfp8_ref(x.cpu())
    assert (s.cpu() == sr).float().mean().item() > 0.999
    back = (q.cpu().view(torch.float8_e4m3fn).float().view(64, 16, 32)
            * torch.exp2(s.cpu().float() - 127).unsqueeze(-1)).reshape(64, 512)
    xf = x.cpu().float()
    denom = xf.abs().view(64, 16, 32).amax(-1, keepdim=True).expand(64, 16, 32).reshape(64, 512)
    assert ((back - xf).abs() / (denom + 1e-30)).max().item() < 0.08


def test_gemm_mxfp8_gpu(dev):
    torch.manual_seed(6)
    M, N, K = 256, 1024, 4096
    a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    b = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    aq, asc = ops.quant_mxfp8(a)
    bq, bsc = ops.quant_mxfp8(b)
    c = ops.gemm_bt_mxfp8(aq, asc, bq, bsc)
    # vs exact dequantized product — the scaled MFMA applies scales in HW
    c_ref = ref.gemm_bt_mxfp8_ref(aq.cpu(), asc.cpu(), bq.cpu(), bsc.cpu())
    torch.testing.assert_close(c.float().cpu(), c_ref.float(), atol=1.0, rtol=5e-2)
    # and closer to the bf16-exact product than the row-wise fp8 path
    exact = (a.float() @ b.float().t()).cpu()
    rel = ((c.float().cpu() - exact).norm() / exact.norm()).item()
    assert rel < 0.05


def test_gemm_mxfp8_256_tile_gpu(dev):
    # big-tile dispatch kicks in at >=160 256x256 tiles; check it against a
    # GPU-side exact dequantized product
    torch.manual_seed(8)
    M, N, K = 4096, 4096, 1024  # 16x16=256 tiles
    a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    b = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    aq, asc = ops.quant_mxfp8(a)
    bq, bsc = ops.quant_mxfp8(b)
    c = ops.gemm_bt_mxfp8(aq, asc, bq, bsc)

    def deq(q, s):
        f = q.view(torch.float8_e4m3fn).float().view(q.shape[0], -1, 32)
        return (f * torch.exp2(s.float() - 127).unsqueeze(-1)).reshape(q.shape[0], -1)

    exact = deq(aq, asc) @ deq(bq, bsc).t()
    torch.testing.assert_close(c.float(), exact, atol=1.0, rtol=5e-2)


def test_mxfp8_model_forward_gpu(dev):
    from senweaver_amd.models import tiny_debug
    from senweaver_amd.models.llama import LlamaModel
    m = LlamaModel(tiny_debug(), device=dev, quant="mxfp8")
    m16 = LlamaModel(tiny_debug(), device=dev, seed=0)
    tokens = torch.randint(0, 512, (1, 128), device=dev)
    h = m.prefill(tokens)
    h16 = m16.prefill(tokens)
    assert torch.isfinite(h.float()).all()
    rel = ((h.float() - h16.float()).norm() / h16.float().norm()).item()
    assert rel < 0.15


def test_rope_scatter_gpu(dev):
    B, S, Hq, Hk, D = 2, 128, 4, 2, 128
    T = B * S
    qkv = torch.randn(T, (Hq + 2 * Hk) * D, dtype=torch.bfloat16, device=dev)
    q = qkv[:, : Hq * D].reshape(T, Hq, D)
    k = qkv[:, Hq * D: (Hq + Hk) * D].reshape(T, Hk, D)
    cos_sin = ops.rope_tables(4096, D).to(dev)
    pos = torch.arange(S, dtype=torch.int32, device=dev).repeat(B)
    qo, ko = ops.rope_scatter_qkv(qkv, cos_sin, pos, Hq, Hk, D, B, S)
    q_ref = ref.rope_ref(q, cos_sin, pos).reshape(B, S, Hq, D).permute(0, 2, 1, 3)
    k_ref = ref.rope_ref(k, cos_sin, pos).reshape(B, S, Hk, D).permute(0, 2, 1, 3)
    torch.testing.assert_close(qo.float(), q_ref.float(), atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(ko.float(), k_ref.float(), atol=2e-2, rtol=2e-2)


def test_decode_graph_matches_eager(dev):
    """hipGraph-captured decode must produce the same tokens as the eager path."""
    from senweaver_amd.engine.scorer import LlamaBackend
    from senweaver_amd.models import tiny_debug
    b_graph = LlamaBackend(tiny_debug(), device=dev, max_seq=256)
    out_graph = b_graph.generate("check the graph decode path", max_new_tokens=10)
    # eager: CPU backend with the same seed/model
    b_eager = LlamaBackend(tiny_debug(), device=dev, max_seq=256)
    b_eager._decode_state()  # build state
    b_eager._decode_graph = None  # force eager decode_step
    out_eager = b_eager.generate("check the graph decode path", max_new_tokens=10)
    assert out_graph == out_eager
    # repeated generate reuses the captured graph and stays deterministic
    out2 = b_graph.generate("check the graph decode path", max_new_tokens=10)
    assert out2 == out_graph


@pytest.mark.parametrize("B,H,Hk,S", [(2, 8, 2, 256), (1, 32, 8, 1024)])
def test_attn_fwd_v2(dev, B, H, Hk, S):
    D = 128
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    vt = v.transpose(-1, -2).contiguous()
    scale = 1.0 / math.sqrt(D)
    ot = ops.attn_fwd_t(q, k, vt, scale)
    o_ref = ref.attn_fwd_ref(q, k, v, scale, causal=True)
    torch.testing.assert_close(ot.transpose(-1, -2).float(), o_ref.float(),
                               atol=8e-2, rtol=8e-2)


def test_attn_fwd_v2_spiked(dev):
    B, H, Hk, S, D = 1, 2, 2, 512, 128
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    k[:, :, 100] *= 8
    v = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(D)
    ot = ops.attn_fwd_t(q, k, v.transpose(-1, -2).contiguous(), scale)
    o_ref = ref.attn_fwd_ref(q, k, v, scale, causal=True)
    torch.testing.assert_close(ot.transpose(-1, -2).float(), o_ref.float(),
                               atol=8e-2, rtol=8e-2)


def test_vt_from_qkv_gpu(dev):
    B, S, Hq, Hk, D = 2, 192, 4, 2, 128  # S not a multiple of 32-tile grid edge
    ld = (Hq + 2 * Hk) * D
    qkv = torch.randn(B * S, ld, dtype=torch.bfloat16, device=dev)
    vt = ops.vt_from_qkv(qkv, Hq, Hk, D, B, S)
    v = qkv[:, (Hq + Hk) * D:].reshape(B, S, Hk, D)
    ref_vt = v.permute(0, 2, 3, 1).contiguous()
    assert torch.equal(vt, ref_vt)


def test_attn_v3_matches_v2(dev):
    import math
    torch.manual_seed(4)
    ext = ops.hip_ext()
    for (B, H, Hk, S) in [(2, 8, 2, 512), (1, 8, 2, 384)]:  # S%256==128 case too
        D = 128
        q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
        k = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
        vt = torch.randn(B, Hk, D, S, dtype=torch.bfloat16, device=dev)
        s = 1.0 / math.sqrt(D)
        o2 = ext.attn_fwd_v2(q, k, vt, s)
        o3 = ext.attn_fwd_v3(q, k, vt, s)
        torch.testing.assert_close(o2.float(), o3.float(), atol=2e-2, rtol=1e-2)


def test_fp8_moe_forward_gpu(dev):
    from senweaver_amd.models import tiny_moe
    from senweaver_amd.models.llama import LlamaModel
    m16 = LlamaModel(tiny_moe(), device=dev, seed=4)
    m8 = LlamaModel(tiny_moe(), device=dev, seed=4, quant="fp8")
    toks = torch.randint(0, 512, (2, 64), device=dev)
    h16 = m16.prefill(toks)
    h8 = m8.prefill(toks)
    assert torch.isfinite(h8.float()).all()
    rel = ((h16.float() - h8.float()).norm() / h16.float().norm()).item()
    assert rel < 0.25


def test_moe_combine_gpu(dev):
    torch.manual_seed(12)
    T, k, H = 256, 2, 512
    Tk = T * k
    down = torch.randn(Tk, H, dtype=torch.bfloat16, device=dev)
    # a routing permutation: positions of each token's k entries
    perm = torch.randperm(Tk, device=dev)
    inv = torch.argsort(perm).to(torch.int32).reshape(T, k)
    w = torch.rand(Tk, device=dev)
    out = ops.hip_ext().moe_combine(down, inv, w)
    ref_out = torch.zeros(T, H, dtype=torch.float32, device=dev)
    tok_of_pos = perm // k  # position p holds token perm[p]//k? no — build from inv
    for j in range(k):
        pos = inv[:, j].long()
        ref_out += w[pos].unsqueeze(1) * down[pos].float()
    torch.testing.assert_close(out.float(), ref_out, atol=5e-2, rtol=2e-2)


def test_mxfp8_moe_forward_gpu(dev):
    from senweaver_amd.models import tiny_moe
    from senweaver_amd.models.llama import LlamaModel
    m16 = LlamaModel(tiny_moe(), device=dev, seed=4)
    m8 = LlamaModel(tiny_moe(), device=dev, seed=4, quant="mxfp8")
    toks = torch.randint(0, 512, (2, 64), device=dev)
    h16 = m16.prefill(toks)
    h8 = m8.prefill(toks)
    assert torch.isfinite(h8.float()).all()
    rel = ((h16.float() - h8.float()).norm() / h16.float().norm()).item()
    assert rel < 0.2


def test_gemv_norm_gpu(dev):
    torch.manual_seed(13)
    for M, with_res in [(1, True), (4, True), (3, False), (16, True)]:
        K, N = 4096, 1024
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        res = torch.randn(M, K, dtype=torch.bfloat16, device=dev) if with_res else None
        w = torch.randn(K, dtype=torch.bfloat16, device=dev)
        b = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        c, res_new = ops.gemv_norm_bt(x, res, w, b, 1e-5)
        t = (x + res) if with_res else x
        y = ref.rmsnorm_ref(t.cpu(), w.cpu(), 1e-5)
        c_ref = ref.gemm_bt_ref(y, b.cpu())
        torch.testing.assert_close(c.float().cpu(), c_ref.float(), atol=0.5, rtol=3e-2)
        torch.testing.assert_close(res_new.float().cpu(), t.float().cpu(), atol=2e-2, rtol=1e-2)


def test_rope_kv_append_gpu(dev):
    torch.manual_seed(14)
    B, Hq, Hk, D, P = 4, 4, 2, 128, 8
    ld = (Hq + 2 * Hk) * D
    qkv = torch.randn(B, ld, dtype=torch.bfloat16, device=dev)
    cs = ops.rope_tables(256, D, 500000.0).to(dev)
    pos = torch.tensor([3, 17, 42, 99], dtype=torch.int32, device=dev)
    slot = torch.tensor([5, 21, 60, 100], dtype=torch.int32, device=dev)
    kc = torch.zeros(P * 16, Hk, D, dtype=torch.bfloat16, device=dev)
    vc = torch.zeros(P * 16, Hk, D, dtype=torch.bfloat16, device=dev)
    q = ops.rope_kv_append(qkv, cs, pos, slot, kc, vc, Hq, Hk, D)
    # reference: manual slice + rope + index_copy
    qs, kvs = Hq * D, Hk * D
    q_ref = ref.rope_ref(qkv[:, :qs].reshape(B, Hq, D).cpu(), cs.cpu(), pos.cpu())
    k_ref = ref.rope_ref(qkv[:, qs:qs + kvs].reshape(B, Hk, D).cpu(), cs.cpu(), pos.cpu())
    v_ref = qkv[:, qs + kvs:].reshape(B, Hk, D).cpu()
    torch.testing.assert_close(q.cpu().float(), q_ref.float(), atol=2e-2, rtol=1e-2)
    for i in range(B):
        torch.testing.assert_close(kc[slot[i].long()].cpu().float(), k_ref[i].float(),
                                   atol=2e-2, rtol=1e-2)
        torch.testing.assert_close(vc[slot[i].long()].cpu().float(), v_ref[i].float(),
                                   atol=0, rtol=0)

@pytest.mark.parametrize("B,H,Hk,S", [(2, 8, 2, 256), (1, 32, 8, 1024),
                                      (1, 8, 8, 2048)])
def test_attn_fwd_v5(dev, B, H, Hk, S):
    """Production prefill kernel (v4 lever-matrix winner): register-staged
    async prefetch + defer-max + softmax VALU diet + sm-split."""
    D = 128
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    vt = v.transpose(-1, -2).contiguous()
    scale = 1.0 / math.sqrt(D)
    ot = ops.hip_ext().attn_fwd_v5(q, k, vt, scale)
    o_ref = ref.attn_fwd_ref(q, k, v, scale, causal=True)
    torch.testing.assert_close(ot.transpose(-1, -2).float(), o_ref.float(),
                               atol=8e-2, rtol=8e-2)


def test_attn_fwd_v5_defer_max_paths(dev):
    """defer-max correctness: a spike (forces the rescale branch) and a
    slow upward drift (keeps tiles inside the defer threshold, P bounded
    by e^8) must both match the fp32 reference."""
    B, H, Hk, S, D = 1, 4, 4, 1024, 128
    scale = 1.0 / math.sqrt(D)
    for mode in ("spike", "drift"):
        q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
        k = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
        if mode == "spike":
            k[:, :, 100] *= 8
        else:
            drift = torch.linspace(1.0, 1.5, S, device=dev)
            k = (k.float() * drift.view(1, 1, S, 1)).to(torch.bfloat16)
        v = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
        vt = v.transpose(-1, -2).contiguous()
        ot = ops.hip_ext().attn_fwd_v5(q, k.contiguous(), vt, scale)
        o_ref = ref.attn_fwd_ref(q, k, v, scale, causal=True)
        torch.testing.assert_close(ot.transpose(-1, -2).float(), o_ref.float(),
                                   atol=1e-1, rtol=1e-1)


def test_gemv_v3_engine(dev):
    """Loader/consumer LDS-DMA streaming GEMV (M=1, wide K) vs fp32 ref."""
    for (N, K) in [(4096, 14336), (1024, 8192)]:
        x = torch.randn(1, K, dtype=torch.bfloat16, device=dev)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        got = ops.hip_ext().gemv_bt_v3(x, w)
        ref_out = x.float() @ w.float().t()
        torch.testing.assert_close(got.float(), ref_out, atol=2.0, rtol=2e-2)
    # the gemm_bt M=1 dispatch routes wide-K GEMV here; parity with v2
    x = torch.randn(1, 14336, dtype=torch.bfloat16, device=dev)
    w = torch.randn(4096, 14336, dtype=torch.bfloat16, device=dev)
    via_dispatch = ops.hip_ext().gemm_bt(x, w)
    torch.testing.assert_close(via_dispatch.float(), x.float() @ w.float().t(),
                               atol=2.0, rtol=2e-2)


def test_swiglu_gemv_fused(dev):
    """Fused silu(g)*u + down-proj GEMV vs the two-op fp32 reference."""
    N, K = 4096, 14336
    gu = torch.randn(1, 2 * K, dtype=torch.bfloat16, device=dev)
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    got = ops.swiglu_gemv(gu, w)
    g = gu[:, :K].float()
    u = gu[:, K:].float()
    act = g * torch.sigmoid(g) * u
    ref_out = act @ w.float().t()
    torch.testing.assert_close(got.float(), ref_out, atol=3.0, rtol=3e-2)


def test_decode_long_context_multipage(dev):
    """Long prompt (hundreds of tokens -> tens of KV pages) decoded via the
    captured graph must match eager, exercising the multi-page block-table
    path end to end (prefill pad, paged append, split-8 decode attention)."""
    from senweaver_amd.engine.scorer import LlamaBackend
    from senweaver_amd.models import tiny_debug
    prompt = " ".join(f"segment {i} of the long context" for i in range(120))
    b_graph = LlamaBackend(tiny_debug(), device=dev, max_seq=1024)
    out_graph = b_graph.generate(prompt, max_new_tokens=24)
    b_eager = LlamaBackend(tiny_debug(), device=dev, max_seq=1024)
    b_eager._decode_state()
    b_eager._decode_graph = None
    out_eager = b_eager.generate(prompt, max_new_tokens=24)
    assert out_graph == out_eager
    assert len(out_graph) > 0


def test_decode_step_batch_lockstep(dev):
    """Batched graph decode (B=2 lockstep) must equal two independent B=1
    decodes of the same prompts."""
    from senweaver_amd.engine.graph import DecodeGraph
    from senweaver_amd.engine.kvcache import PAGE_SIZE, PagedKVCache
    from senweaver_amd.models import LlamaModel, tiny_debug

    cfg = tiny_debug()
    model = LlamaModel(cfg, device=dev, seed=5)
    prompts = [torch.randint(0, 256, (1, 64), generator=torch.Generator().manual_seed(s0))
               for s0 in (1, 2)]

    def run_single(ptoks, steps=6):
        cache = PagedKVCache(cfg, 16, torch.device(dev),
                             num_kv_heads=model.local_kv_heads)
        sq = cache.new_seq()
        hidden = model.prefill(ptoks.to(dev), cache=cache, seqs=[sq],
                               real_lens=[64])
        graph = DecodeGraph(model, cache, 16, batch=1)
        last = hidden[:, 63]
        outs = []
        nxt = int(ops.argmax_rows(model.logits(last))[0])
        for _ in range(steps):
            outs.append(nxt)
            last = graph.step(nxt, sq)
            nxt = int(ops.argmax_rows(model.logits(last))[0])
        return outs

    singles = [run_single(p) for p in prompts]

    cache = PagedKVCache(cfg, 32, torch.device(dev),
                         num_kv_heads=model.local_kv_heads)
    seqs = [cache.new_seq() for _ in range(2)]
    both = torch.cat(prompts, dim=0).to(dev)
    hidden = model.prefill(both, cache=cache, seqs=seqs, real_lens=[64, 64])
    graph = DecodeGraph(model, cache, 16, batch=2)
    last = hidden[:, 63]
    outs = [[], []]
    nxt = [int(x) for x in ops.argmax_rows(model.logits(last)).cpu()]
    for _ in range(6):
        for i in range(2):
            outs[i].append(nxt[i])
        last = graph.step_batch(nxt, seqs)
        nxt = [int(x) for x in ops.argmax_rows(model.logits(last)).cpu()]
    assert outs[0] == singles[0]
    assert outs[1] == singles[1]


def test_gemv_fp8_weights(dev):
    """fp8-weight x bf16-activation decode GEMV vs the dequant reference."""
    for (N, K, M) in [(4096, 4096, 1), (1024, 14336, 1), (2048, 4096, 3),
                      (1024, 4096, 8)]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        wq, ws = ops.quant_fp8(w)
        got = ops.gemv_fp8w(x, wq, ws)
        wf = wq.view(torch.float8_e4m3fn).float() * ws.unsqueeze(1)
        ref_out = x.float() @ wf.t()
        torch.testing.assert_close(got.float(), ref_out, atol=1.5, rtol=2e-2)


def test_gemv_mxfp8_weights(dev):
    """mxfp8-weight decode GEMV vs the block-dequant reference."""
    for (N, K, M) in [(2048, 4096, 1), (1024, 4096, 4)]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        wq, ws = ops.quant_mxfp8(w)
        got = ops.gemv_mxfp8w(x, wq, ws)
        f = wq.view(torch.float8_e4m3fn).float().view(N, K // 32, 32)
        wf = (f * torch.exp2(ws.float() - 127).unsqueeze(-1)).reshape(N, K)
        ref_out = x.float() @ wf.t()
        torch.testing.assert_close(got.float(), ref_out, atol=1.5, rtol=2e-2)


@pytest.mark.parametrize("quant", ["fp8", "mxfp8"])
def test_decode_graph_quantized(dev, quant):
    """Quantized-weight decode (fp8w/mxfp8w GEMV) through the captured
    graph must match the eager path token-for-token."""
    from senweaver_amd.engine.scorer import LlamaBackend
    from senweaver_amd.models import tiny_debug
    b_graph = LlamaBackend(tiny_debug(), device=dev, max_seq=256, quant=quant)
    out_graph = b_graph.generate("quantized decode check", max_new_tokens=8)
    b_eager = LlamaBackend(tiny_debug(), device=dev, max_seq=256, quant=quant)
    b_eager._decode_state()
    b_eager._decode_graph = None
    out_eager = b_eager.generate("quantized decode check", max_new_tokens=8)
    assert out_graph == out_eager


def test_checkpoint_roundtrip_gpu(dev, tmp_path):
    """save/load on the device: restored weights reproduce logits."""
    from senweaver_amd.models import LlamaModel, load_weights, save_weights, tiny_debug
    a = LlamaModel(tiny_debug(), device=dev, seed=1)
    b = LlamaModel(tiny_debug(), device=dev, seed=2)
    toks = torch.randint(0, 256, (1, 64),
                         generator=torch.Generator().manual_seed(3)).to(dev)
    p = str(tmp_path / "w.safetensors")
    save_weights(a, p)
    load_weights(b, p)
    torch.testing.assert_close(a.prefill(toks).float(), b.prefill(toks).float())


def test_add_bf16(dev):
    """Residual-add utility op (paths that skip the fused add+norm)."""
    a = torch.randn(333, 1024, dtype=torch.bfloat16, device=dev)
    b = torch.randn(333, 1024, dtype=torch.bfloat16, device=dev)
    got = ops.hip_ext().add_bf16(a, b)
    torch.testing.assert_close(got.float(), a.float() + b.float(),
                               atol=2e-2, rtol=2e-2)
