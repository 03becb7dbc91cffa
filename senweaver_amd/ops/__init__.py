"""Op dispatch for the gfx950 compute path.

Fused and shape-special ops (attention, paged decode, RoPE/RMSNorm/SwiGLU,
V^T extraction, MX/rowwise fp8 quant, decode GEMV) run the in-repo HIP
kernels; PLAIN GEMMs go to whichever implementation measured faster on
MI355X — hipBLASLt for prefill shapes, our kernels elsewhere — per
profiles/r01_gemm_dispatch.txt (SENWEAVER_GEMM=hip forces in-repo
everywhere).

On a GPU box the HIP extension is REQUIRED — ops raise if a CUDA(=HIP)
tensor arrives and the extension is missing, so nothing silently falls back
to eager PyTorch on the hardware the kernels target.  CPU tensors use the
fp32 reference implementations (tests / CPU-only engine mode).
"""

from __future__ import annotations

import math
import os
from typing import Optional

import torch

from . import reference as ref

_ext = None
_ext_err: Optional[str] = None


def _load_ext():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    from . import build as _build

    try:
        _ext = _build.load_prebuilt()
        if _ext is None:
            _ext = _build.build()
    except Exception as e:  # pragma: no cover
        _ext_err = str(e)
        _ext = None
    return _ext


def hip_ext():
    """The loaded HIP extension module, or raise with the build error."""
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "senweaver_amd HIP extension is not available on this GPU host "
            "(refusing silent eager fallback). Build error: " + str(_ext_err)
        )
    return ext


def extension_loaded() -> bool:
    return _load_ext() is not None


def _on_gpu(*tensors) -> bool:
    return any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))


# ---------------- public ops ----------------

def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if _on_gpu(x):
        return hip_ext().rmsnorm(x.contiguous(), w.contiguous(), eps)
    return ref.rmsnorm_ref(x, w, eps)


def fused_add_rmsnorm(x: torch.Tensor, residual: torch.Tensor, w: torch.Tensor, eps: float = 1e-5):
    """Returns y; updates `residual` in place to x + residual."""
    if _on_gpu(x):
        return hip_ext().fused_add_rmsnorm(x.contiguous(), residual, w.contiguous(), eps)
    y, r = ref.fused_add_rmsnorm_ref(x, residual, w, eps)
    residual.copy_(r)
    return y


def rope_inplace(q: torch.Tensor, k: torch.Tensor, cos_sin: torch.Tensor, positions: torch.Tensor):
    """q: [T,Hq,D], k: [T,Hk,D] modified in place."""
    if _on_gpu(q):
        hip_ext().rope_inplace(q, k, cos_sin, positions.to(torch.int32))
        return
    q.copy_(ref.rope_ref(q, cos_sin, positions))
    k.copy_(ref.rope_ref(k, cos_sin, positions))


def rope_scatter_qkv(qkv: torch.Tensor, cos_sin, positions, Hq: int, Hk: int,
                     D: int, B: int, S: int):
    """Fused qkv [T, (Hq+2Hk)*D] -> rope(q,k) -> ([B,Hq,S,D], [B,Hk,S,D]).

    Reads the q/k head slices straight from the fused projection output (no
    contiguous slice copies).  The v slice is untouched (consumed separately
    as V^T)."""
    if _on_gpu(qkv):
        qo, ko = hip_ext().rope_scatter_qkv(qkv.contiguous(), cos_sin,
                                            positions.to(torch.int32),
                                            Hq, Hk, D, B, S)
        return qo, ko
    T = qkv.shape[0]
    q = qkv[:, : Hq * D].reshape(T, Hq, D)
    k = qkv[:, Hq * D: (Hq + Hk) * D].reshape(T, Hk, D)
    qr = ref.rope_ref(q, cos_sin, positions)
    kr = ref.rope_ref(k, cos_sin, positions)
    return (qr.reshape(B, S, Hq, D).permute(0, 2, 1, 3).contiguous(),
            kr.reshape(B, S, Hk, D).permute(0, 2, 1, 3).contiguous())


def rope_kv_append(qkv: torch.Tensor, cos_sin, positions, slot,
                   kcache_flat, vcache_flat, Hq: int, Hk: int, D: int
                   ) -> torch.Tensor:
    """Decode fused head dispatch: qkv row -> rope(q,k); q returned
    [B,Hq,D] contiguous; k/v written straight into the paged cache rows at
    ``slot`` (caches viewed [P*16, Hk, D]).  Replaces six ~5 us decode
    kernels per layer (3 slice copies + rope + 2 index_copy)."""
    if _on_gpu(qkv):
        return hip_ext().rope_kv_append(qkv.contiguous(), cos_sin,
                                        positions.to(torch.int32),
                                        slot.to(torch.int32),
                                        kcache_flat, vcache_flat, Hq, Hk, D)
    B = qkv.shape[0]
    qs, kvs = Hq * D, Hk * D
    q = qkv[:, :qs].reshape(B, Hq, D).clone()
    k = qkv[:, qs:qs + kvs].reshape(B, Hk, D).clone()
    v = qkv[:, qs + kvs:qs + 2 * kvs].reshape(B, Hk, D)
    qr = ref.rope_ref(q, cos_sin, positions)
    kr = ref.rope_ref(k, cos_sin, positions)
    kcache_flat.index_copy_(0, slot.long(), kr)
    vcache_flat.index_copy_(0, slot.long(), v.contiguous())
    return qr


def gemv_norm_bt(x, res_in, normw, b, eps: float = 1e-5):
    """Decode fused residual-add + RMSNorm + GEMV.

    Returns (C [M,N], res_new [M,K] = x (+ res_in)).  RMSNorm is a linear
    row scale, so the raw dot and sum-of-squares accumulate in one k-pass
    and the scale applies after the wave reduction — one kernel instead of
    fused_add_rmsnorm + gemv.  M <= 16, K %% 512 == 0, N %% 4 == 0."""
    if _on_gpu(x):
        c, res_new = hip_ext().gemv_norm_bt(x.contiguous(),
                                            res_in.contiguous() if res_in is not None else None,
                                            normw.contiguous(), b.contiguous(), eps)
        return c, res_new
    t = x + res_in if res_in is not None else x.clone()
    y = ref.rmsnorm_ref(t, normw, eps)
    return ref.gemm_bt_ref(y, b), t


def vt_from_qkv(qkv: torch.Tensor, Hq: int, Hk: int, D: int, B: int, S: int
                ) -> torch.Tensor:
    """V^T [B,Hk,D,S] straight from the fused qkv projection output.

    LDS-tiled 32x32 transpose (coalesced both ways) instead of a strided
    permute().contiguous() copy."""
    if _on_gpu(qkv):
        return hip_ext().vt_from_qkv(qkv.contiguous(), Hq, Hk, D, B, S)
    T = qkv.shape[0]
    v = qkv[:, (Hq + Hk) * D:(Hq + 2 * Hk) * D].reshape(B, S, Hk, D)
    return v.permute(0, 2, 3, 1).contiguous()


def swiglu(gateup: torch.Tensor) -> torch.Tensor:
    if _on_gpu(gateup):
        return hip_ext().swiglu(gateup.contiguous())
    return ref.swiglu_ref(gateup)


_GEMM_IMPL = os.environ.get("SENWEAVER_GEMM", "auto")  # auto | hip
_ATTN_IMPL = os.environ.get("SENWEAVER_ATTN", "auto")  # auto | v2


def swiglu_gemv(gateup: torch.Tensor, w_down: torch.Tensor) -> torch.Tensor:
    """Fused silu(g)*u + down-projection GEMV (decode, M=1, wide K).

    gateup [1, 2K]; w_down [N, K].  GPU: the loader/consumer LDS-DMA
    streaming engine computes the activation into LDS on the way in —
    one kernel replaces (swiglu + gemv).  Reference parity: reference
    decoder MLP behavior; numerics vs the fp32 torch reference in
    tests/test_kernels_gpu.py.
    """
    K = gateup.shape[-1] // 2
    if (_on_gpu(gateup) and gateup.shape[0] == 1 and K % 1024 == 0
            and 4096 < K <= 14336 and w_down.shape[0] % 8 == 0):
        return hip_ext().swiglu_gemv_bt(gateup.contiguous(), w_down.contiguous())
    act = swiglu(gateup)
    return gemm_bt(act, w_down)


def gemv_fp8w(x: torch.Tensor, w_q: torch.Tensor, w_s: torch.Tensor) -> torch.Tensor:
    """Decode GEMV with fp8 weights and UNQUANTIZED bf16 activations.

    x [M<=8,K] bf16; w_q [N,K] e4m3 + per-row scale w_s [N].  Halves the
    weight stream (decode is weight-BW-bound) and skips the per-step
    activation-quant kernel; the M-padded 128-tile fp8 GEMM it replaces
    measured 101 tok/s vs bf16's 267.  Falls back to the dequant
    reference off-GPU / off-shape.
    """
    K = x.shape[-1]
    if _on_gpu(x) and x.shape[0] <= 8 and K % 1024 == 0 and w_q.shape[0] % 4 == 0:
        return hip_ext().gemv_bt_fp8w(x.contiguous(), w_q.contiguous(),
                                      w_s.contiguous())
    wf = w_q.view(torch.float8_e4m3fn).float() * w_s.unsqueeze(1)
    return (x.float() @ wf.t()).to(x.dtype)


def gemv_mxfp8w(x: torch.Tensor, w_q: torch.Tensor, w_s: torch.Tensor) -> torch.Tensor:
    """Decode GEMV with MX block-scaled fp8 weights, bf16 activations
    (the mxfp8 analog of gemv_fp8w; e8m0 per-32-block scales)."""
    K = x.shape[-1]
    if _on_gpu(x) and x.shape[0] <= 8 and K % 1024 == 0 and w_q.shape[0] % 4 == 0:
        return hip_ext().gemv_bt_mxfp8w(x.contiguous(), w_q.contiguous(),
                                        w_s.contiguous())
    rows = w_q.shape[0]
    f = w_q.view(torch.float8_e4m3fn).float().view(rows, K // 32, 32)
    wf = (f * torch.exp2(w_s.float() - 127).unsqueeze(-1)).reshape(rows, K)
    return (x.float() @ wf.t()).to(x.dtype)


def gemm_bt(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """C[M,N] = A[M,K] @ B[N,K]^T.

    Dispatch (measured on MI355X, profiles/r01_gemm_dispatch.txt): decode-
    shaped M<=16 goes to our wave-per-row GEMV kernel (5.2 TB/s in situ, ~2x blas on
    skinny); plain prefill GEMMs go to hipBLASLt (1.28-1.58 PF/s at bench
    shapes vs our tiled kernel's 0.88-1.15 — the guide's rule: hand-write the
    fused/special ops, use the vendor library for plain GEMMs it wins).
    ``SENWEAVER_GEMM=hip`` forces the in-repo tiled kernel everywhere.
    """
    if not _on_gpu(a):
        return ref.gemm_bt_ref(a, b)
    M = a.shape[0]
    if M <= 16 and b.shape[1] % 512 == 0 and b.shape[0] % 4 == 0:
        return hip_ext().gemm_bt(a.contiguous(), b.contiguous())
    if _GEMM_IMPL != "hip":
        return a @ b.t()
    return gemm_bt_tiled(a, b)


def gemm_bt_tiled(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """The in-repo MFMA tiled GEMM (128^2 4-wave / 256^2 8-wave LDS tiles).

    Pads M to the tile.  Kept first-class for fused-epilogue growth and A/B
    benchmarking against hipBLASLt (benchmarks/kernel_bench.py gemm).
    """
    if not _on_gpu(a):
        return ref.gemm_bt_ref(a, b)
    ext = hip_ext()
    M, N = a.shape[0], b.shape[0]
    # pad M to the 256-tile when N allows it and the grid fills the chip
    tile = 256 if (N % 256 == 0 and ((M + 255) // 256) * (N // 256) >= 160) else 128
    pad = (-M) % tile
    if pad:
        a = torch.nn.functional.pad(a, (0, 0, 0, pad))
    c = ext.gemm_bt(a.contiguous(), b.contiguous())
    return c[:M] if pad else c


def quant_fp8(x: torch.Tensor):
    """Row-wise e4m3 quantization: returns (q_uint8, scale_f32[rows])."""
    if _on_gpu(x):
        q, s = hip_ext().quant_fp8(x.contiguous().reshape(-1, x.shape[-1]))
        return q, s
    return ref.quant_fp8_ref(x)


def gemm_bt_fp8(a_q, a_s, b_q, b_s):
    """C = (A_q @ B_q^T) * a_s[m] * b_s[n] on fp8 MFMA, bf16 out.

    Dispatch mirrors gemm_bt: plain shapes go to hipBLASLt's fp8 path
    (torch._scaled_mm, 2.42 PF/s @8192^3 vs our kernel's 1.37 — bitwise-
    identical result, profiles/r01_gemm_dispatch.txt); our kernel covers
    shapes the library rejects and SENWEAVER_GEMM=hip."""
    if _on_gpu(a_q):
        M = a_q.shape[0]
        if _GEMM_IMPL != "hip" and M % 16 == 0 and a_q.shape[1] % 16 == 0:
            try:
                return torch._scaled_mm(
                    a_q.view(torch.float8_e4m3fn),
                    b_q.view(torch.float8_e4m3fn).t(),
                    scale_a=a_s.unsqueeze(1), scale_b=b_s.unsqueeze(0),
                    out_dtype=torch.bfloat16)
            except RuntimeError:
                pass
        pad = (-M) % 128
        if pad:
            a_q = torch.nn.functional.pad(a_q, (0, 0, 0, pad))
            a_s = torch.nn.functional.pad(a_s, (0, pad), value=1.0)
        c = hip_ext().gemm_bt_fp8(a_q.contiguous(), a_s.contiguous(),
                                  b_q.contiguous(), b_s.contiguous())
        return c[:M] if pad else c
    return ref.gemm_bt_fp8_ref(a_q, a_s, b_q, b_s)


def quant_mxfp8(x: torch.Tensor):
    """OCP MX quant: (q_uint8[rows,K], e8m0 scales uint8[rows,K//32])."""
    if _on_gpu(x):
        q, s = hip_ext().quant_mxfp8(x.contiguous().reshape(-1, x.shape[-1]))
        return q, s
    return ref.quant_mxfp8_ref(x)


def gemm_bt_mxfp8(a_q, a_s, b_q, b_s):
    """C = dequant(A)@dequant(B)^T on the MX block-scaled fp8 MFMA (2x fp8 rate).

    Scales are e8m0 bytes per 32-element K block; the 32x32x64 scaled MFMA
    applies them in hardware, so no epilogue rescale.
    """
    if _on_gpu(a_q):
        M = a_q.shape[0]
        pad = (-M) % 128
        if pad:
            a_q = torch.nn.functional.pad(a_q, (0, 0, 0, pad))
            a_s = torch.nn.functional.pad(a_s, (0, 0, 0, pad), value=127)
        c = hip_ext().gemm_bt_mxfp8(a_q.contiguous(), a_s.contiguous(),
                                    b_q.contiguous(), b_s.contiguous())
        return c[:M] if pad else c
    return ref.gemm_bt_mxfp8_ref(a_q, a_s, b_q, b_s)


def grouped_gemm_bt(a_sorted: torch.Tensor, w: torch.Tensor, seg_starts_cpu,
                    ) -> torch.Tensor:
    """Segment-grouped C = A_seg @ W[e]^T for MoE.

    ``a_sorted``: tokens sorted by expert; ``w``: [E, N, K].
    ``seg_starts_cpu`` is either a host list/1D tensor of E+1 cumulative row
    offsets, or a DEVICE int32 tensor of E cumulative segment ENDS — the
    device form keeps the whole routed FFN sync-free on the hipBLASLt
    grouped path (torch._grouped_mm takes device offsets directly; the
    per-expert ``int(counts[e])`` host reads were 8 tiny D2H syncs per MoE
    layer).  With host offsets, ``a_sorted`` may carry >=128 pad rows for
    the in-repo kernel's tile overread; with device offsets it must be
    exactly the routed rows.  Returns at least the routed rows; rows past
    the last segment (if any) are undefined.
    """
    E = w.shape[0]
    dev_offs = (isinstance(seg_starts_cpu, torch.Tensor)
                and seg_starts_cpu.is_cuda)
    if not dev_offs:
        starts = [int(x) for x in seg_starts_cpu]
    if not _on_gpu(a_sorted):
        out = torch.zeros(a_sorted.shape[0], w.shape[1], dtype=a_sorted.dtype)
        for e in range(E):
            s, t = starts[e], starts[e + 1]
            if t > s:
                out[s:t] = ref.gemm_bt_ref(a_sorted[s:t], w[e])
        return out
    dev = a_sorted.device
    if _GEMM_IMPL != "hip" and hasattr(torch, "_grouped_mm"):
        # hipBLASLt grouped path (1.33 PF/s vs our kernel's 0.95 at Mixtral
        # shapes, profiles/r01_gemm_dispatch.txt); our kernel remains the
        # fallback for offsets/shapes the library rejects
        try:
            if dev_offs:
                return torch._grouped_mm(a_sorted, w.transpose(1, 2),
                                         offs=seg_starts_cpu)
            return torch._grouped_mm(
                a_sorted[:starts[-1]], w.transpose(1, 2),
                offs=torch.tensor(starts[1:], dtype=torch.int32, device=dev))
        except RuntimeError:
            pass
    if dev_offs:
        # fallback needs host offsets (tile map is host-built) + pad rows
        starts = [0] + seg_starts_cpu.cpu().tolist()
        a_sorted = torch.nn.functional.pad(a_sorted, (0, 0, 0, 128))
    tile_expert, tile_m0 = [], []
    for e in range(E):
        s, t = starts[e], starts[e + 1]
        for m0 in range(s, t, 128):
            tile_expert.append(e)
            tile_m0.append(m0)
    te = torch.tensor(tile_expert, dtype=torch.int32, device=dev)
    tm = torch.tensor(tile_m0, dtype=torch.int32, device=dev)
    ends = torch.tensor(starts[1:], dtype=torch.int32, device=dev)
    return hip_ext().grouped_gemm_bt(a_sorted.contiguous(), w.contiguous(), te, tm, ends)


def grouped_gemm_bt_fp8(a_q, a_s, w_q, w_s, seg_starts_cpu) -> torch.Tensor:
    """Per-expert fp8 GEMM over expert-sorted segments (fp8 MoE).

    ``a_q``/[rows] ``a_s``: rowwise-quantized sorted activations;
    ``w_q`` [E,N,K] u8 + ``w_s`` [E,N] f32; ``seg_starts_cpu``: E+1 host
    offsets, 16-ALIGNED (the MoE layer pads each expert's segment so every
    start satisfies hipBLASLt's fp8 alignment).  GPU: one torch._scaled_mm
    per non-empty expert; our fp8 kernel is the per-segment fallback.
    """
    E, N = w_q.shape[0], w_q.shape[1]
    starts = [int(x) for x in seg_starts_cpu]
    Tk = starts[-1]
    if not _on_gpu(a_q):
        out = torch.zeros(Tk, N, dtype=torch.bfloat16)
        for e in range(E):
            s, t = starts[e], starts[e + 1]
            if t > s:
                out[s:t] = ref.gemm_bt_fp8_ref(a_q[s:t], a_s[s:t], w_q[e], w_s[e])
        return out
    out = torch.empty(Tk, N, dtype=torch.bfloat16, device=a_q.device)
    for e in range(E):
        s, t = starts[e], starts[e + 1]
        if t == s:
            continue
        try:
            out[s:t] = torch._scaled_mm(
                a_q[s:t].view(torch.float8_e4m3fn),
                w_q[e].view(torch.float8_e4m3fn).t(),
                scale_a=a_s[s:t].unsqueeze(1), scale_b=w_s[e].unsqueeze(0),
                out_dtype=torch.bfloat16)
        except RuntimeError:
            out[s:t] = gemm_bt_fp8(a_q[s:t], a_s[s:t], w_q[e], w_s[e])
    return out


def grouped_gemm_bt_mxfp8(a_q, a_s, w_q, w_s, seg_starts_cpu) -> torch.Tensor:
    """Per-expert MX block-scaled fp8 GEMM over expert-sorted segments.

    ``w_q`` [E,N,K] u8 + ``w_s`` [E,N,K//32] e8m0 bytes; activations
    rowwise-MX-quantized.  One gemm_bt_mxfp8 per non-empty expert (the
    32x32x64 scaled MFMA applies the scales in hardware).
    """
    E, N = w_q.shape[0], w_q.shape[1]
    starts = [int(x) for x in seg_starts_cpu]
    Tk = starts[-1]
    if not _on_gpu(a_q):
        out = torch.zeros(Tk, N, dtype=torch.bfloat16)
    else:
        out = torch.empty(Tk, N, dtype=torch.bfloat16, device=a_q.device)
    for e in range(E):
        s, t = starts[e], starts[e + 1]
        if t > s:
            out[s:t] = gemm_bt_mxfp8(a_q[s:t], a_s[s:t], w_q[e], w_s[e])
    return out


def attn_fwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, scale: Optional[float] = None,
             vt: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Causal GQA attention. q [B,H,S,D], k/v [B,Hk,S,D] (S % 64 == 0 on GPU).

    ``vt`` may be passed pre-transposed ([B,Hk,D,S]) to skip the transpose.
    Dispatches to the swapped-QK^T v2 kernel at S % 128 == 0 (faster), the
    v1 kernel otherwise.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _on_gpu(q):
        if vt is None:
            vt = v.transpose(-1, -2).contiguous()
        if q.shape[2] % 128 == 0:
            ot = attn_fwd_t(q, k, vt, scale)
            return ot.transpose(-1, -2).contiguous()
        return hip_ext().attn_fwd(q.contiguous(), k.contiguous(), vt, scale)
    if v is None:
        v = vt.transpose(-1, -2).contiguous()
    return ref.attn_fwd_ref(q, k, v, scale, causal=True)


def attn_fwd_t(q: torch.Tensor, k: torch.Tensor, vt: torch.Tensor,
               scale: Optional[float] = None) -> torch.Tensor:
    """Causal GQA attention returning O^T [B,H,D,S] (the v2 kernel's native
    layout — saves one transpose copy when the caller wants token-major
    activations next).  S % 128 == 0 on GPU."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _on_gpu(q):
        ext = hip_ext()
        # v5 (v3 structure + register-staged async prefetch + defer-max +
        # softmax VALU diet + sm-split) when the grid still fills the chip;
        # v2's smaller blocks for tiny grids.  v5 beats v2 at EVERY measured
        # S including 8192 (580 vs ~499 TF — profiles/r02_attn_ladder.txt),
        # so the old v2-at-8192 special case is gone.
        if _ATTN_IMPL != "v2" and q.shape[2] >= 256 and \
                ((q.shape[2] + 255) // 256) * q.shape[0] * q.shape[1] >= 512:
            return ext.attn_fwd_v5(q.contiguous(), k.contiguous(),
                                   vt.contiguous(), scale)
        return ext.attn_fwd_v2(q.contiguous(), k.contiguous(),
                               vt.contiguous(), scale)
    v = vt.transpose(-1, -2).contiguous()
    o = ref.attn_fwd_ref(q, k, v, scale, causal=True)
    return o.transpose(-1, -2).contiguous()


def paged_decode_attn(q, kcache, vcache, block_table, ctx_lens, scale: Optional[float] = None):
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _on_gpu(q):
        return hip_ext().paged_decode_attn(
            q.contiguous(), kcache, vcache,
            block_table.to(torch.int32), ctx_lens.to(torch.int32), scale)
    return ref.paged_decode_attn_ref(q, kcache, vcache, block_table, ctx_lens, scale)


def argmax_rows(logits: torch.Tensor) -> torch.Tensor:
    if _on_gpu(logits):
        return hip_ext().argmax_rows(logits.contiguous())
    return ref.argmax_rows_ref(logits)


def target_logprob(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    if _on_gpu(logits):
        return hip_ext().target_logprob(logits.contiguous(), targets.to(torch.int32))
    return ref.target_logprob_ref(logits, targets)


rope_tables = ref.rope_tables
