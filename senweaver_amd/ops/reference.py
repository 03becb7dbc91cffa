"""Pure-PyTorch fp32 reference implementations of every HIP op.

These are (a) the CPU execution path for tests and the CPU-only engine mode,
and (b) the numerics oracle the GPU tests compare the HIP kernels against
(fp32 math, same op semantics).
"""

from __future__ import annotations

import math

import torch


def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    ms = xf.pow(2).mean(-1, keepdim=True)
    y = xf * torch.rsqrt(ms + eps) * w.float()
    return y.to(x.dtype)


def fused_add_rmsnorm_ref(x: torch.Tensor, residual: torch.Tensor, w: torch.Tensor, eps: float):
    """Returns (y, new_residual). Matches the kernel: r = bf16(x + residual),
    y = rmsnorm(r)."""
    r = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm_ref(r, w, eps), r


def rope_tables(max_pos: int, dim: int, theta: float = 500000.0) -> torch.Tensor:
    """[max_pos, dim] f32 rows laid out [cos(0..d/2) | sin(0..d/2)]."""
    inv_freq = 1.0 / (theta ** (torch.arange(0, dim, 2, dtype=torch.float32) / dim))
    t = torch.arange(max_pos, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)  # [max_pos, dim/2]
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).contiguous()


def rope_ref(x: torch.Tensor, cos_sin: torch.Tensor, positions: torch.Tensor) -> torch.Tensor:
    """x: [T, H, D]; rotate-half (Llama) style."""
    T, H, D = x.shape
    half = D // 2
    cs = cos_sin[positions.long()]  # [T, D]
    cos = cs[:, :half].unsqueeze(1)
    sin = cs[:, half:].unsqueeze(1)
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    out = torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)
    return out.to(x.dtype)


def swiglu_ref(gateup: torch.Tensor) -> torch.Tensor:
    inter = gateup.shape[-1] // 2
    g = gateup[..., :inter].float()
    u = gateup[..., inter:].float()
    return (torch.nn.functional.silu(g) * u).to(gateup.dtype)


def gemm_bt_ref(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """C = A @ B^T in fp32, cast back."""
    return (a.float() @ b.float().t()).to(a.dtype)


def attn_fwd_ref(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, scale: float,
                 causal: bool = True, s_real: int | None = None) -> torch.Tensor:
    """q: [B,H,S,D], k/v: [B,Hk,S,D] (GQA broadcast), fp32 math."""
    B, H, S, D = q.shape
    Hk = k.shape[1]
    rep = H // Hk
    kf = k.float().repeat_interleave(rep, dim=1)
    vf = v.float().repeat_interleave(rep, dim=1)
    scores = torch.einsum("bhsd,bhtd->bhst", q.float(), kf) * scale
    if causal:
        mask = torch.ones(S, S, dtype=torch.bool, device=q.device).tril()
        scores = scores.masked_fill(~mask, float("-inf"))
    if s_real is not None and s_real < S:
        scores[..., s_real:] = float("-inf")
        scores[..., s_real:, :] = 0  # rows past the end: uniform (never read)
    p = torch.softmax(scores, dim=-1)
    return torch.einsum("bhst,bhtd->bhsd", p, vf).to(q.dtype)


def paged_decode_attn_ref(q, kcache, vcache, block_table, ctx_lens, scale):
    """q: [B,H,D]; caches [P, 16, Hk, D]; fp32 math."""
    B, H, D = q.shape
    Hk = kcache.shape[2]
    rep = H // Hk
    out = torch.empty_like(q)
    P = kcache.shape[0]
    kflat = kcache.float().reshape(P * 16, Hk, D)
    vflat = vcache.float().reshape(P * 16, Hk, D)
    for b in range(B):
        ctx = int(ctx_lens[b])
        pages = block_table[b]
        slots = torch.stack([pages[p // 16].long() * 16 + (p % 16) for p in range(ctx)])
        kk = kflat[slots]  # [ctx, Hk, D]
        vv = vflat[slots]
        for h in range(H):
            kvh = h // rep
            s = (kk[:, kvh] @ q[b, h].float()) * scale
            p = torch.softmax(s, dim=0)
            out[b, h] = (p @ vv[:, kvh]).to(q.dtype)
    return out


def argmax_rows_ref(logits: torch.Tensor) -> torch.Tensor:
    return logits.float().argmax(dim=-1).to(torch.int32)


def target_logprob_ref(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    lp = torch.log_softmax(logits.float(), dim=-1)
    return lp.gather(1, targets.long().unsqueeze(1)).squeeze(1)


FP8_MAX = 448.0  # OCP e4m3fn


def quant_fp8_ref(x: torch.Tensor):
    """CPU row-wise e4m3 quantization via torch.float8_e4m3fn."""
    x2 = x.reshape(-1, x.shape[-1]).float()
    amax = x2.abs().amax(dim=-1)
    scale = torch.where(amax > 0, amax / FP8_MAX, torch.ones_like(amax))
    q = (x2 / scale.unsqueeze(1)).to(torch.float8_e4m3fn)
    return q.view(torch.uint8), scale


def gemm_bt_fp8_ref(a_q, a_s, b_q, b_s):
    af = a_q.view(torch.float8_e4m3fn).float()
    bf = b_q.view(torch.float8_e4m3fn).float()
    c = (af @ bf.t()) * a_s.unsqueeze(1) * b_s.unsqueeze(0)
    return c.to(torch.bfloat16)


def quant_mxfp8_ref(x: torch.Tensor):
    """OCP MX quantization: per-32-element block, e8m0 scale = 2^(floor(log2(amax))-8).

    Returns (q_uint8[rows,K], scales_uint8[rows,K//32]) with scale byte =
    exponent + 127 (e8m0 biased).
    """
    x2 = x.reshape(-1, x.shape[-1]).float()
    rows, K = x2.shape
    blk = x2.view(rows, K // 32, 32)
    amax = blk.abs().amax(dim=-1)
    e = torch.where(amax > 0, torch.floor(torch.log2(amax)) - 8,
                    torch.full_like(amax, -127.0))
    # non-saturating variant: OCP's floor-8 scale leaves amax/X in [256,512),
    # clipping up to 12.5% off the block max; step one finer when it would
    e = torch.where(amax * torch.exp2(-e) > FP8_MAX, e + 1, e)
    e = e.clamp(-127, 127)
    scales = (e + 127).to(torch.uint8)
    q = (blk * torch.exp2(-e).unsqueeze(-1)).clamp(-FP8_MAX, FP8_MAX)
    q = q.to(torch.float8_e4m3fn).view(torch.uint8).reshape(rows, K)
    return q, scales


def gemm_bt_mxfp8_ref(a_q, a_s, b_q, b_s):
    def deq(q, s):
        rows, K = q.shape
        f = q.view(torch.float8_e4m3fn).float().view(rows, K // 32, 32)
        return (f * torch.exp2(s.float() - 127).unsqueeze(-1)).reshape(rows, K)
    return (deq(a_q, a_s) @ deq(b_q, b_s).t()).to(torch.bfloat16)
