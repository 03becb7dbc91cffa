#!/usr/bin/env python3
"""Generate ops/csrc/gemm_asm.hip — the hand-scheduled inline-asm GEMM.

The plain-HIP pipeline ladder (profiles/r02_gemm_pipeline.txt) ceilings at
~1.25-1.38 PF/s: the compiler drains the ds_read queue with lgkmcnt(0)
before each MFMA cluster and the per-tile rendezvous parks 26-36% of wave
cycles.  This kernel puts each K-tile's 24 fragment reads and 64 MFMAs in
ONE asm statement per k-half with a COUNTED lgkmcnt ladder (reads stream
while MFMAs retire; never a full drain mid-tile), accumulators pinned in
AGPRs a[0:127] (clobber-allocated), and v9's glds staging/boundary-vmcnt
structure around it.

Generated once and checked in (scripts/gen_gemm_asm.py is the source of
truth); regen with `python scripts/gen_gemm_asm.py`.

Hazard discipline (cdna_hip_programming.md §5.7, cdna_asm_programming §4):
  - ds_read -> MFMA operand: ordered by the counted s_waitcnt lgkmcnt(N).
  - v_accvgpr_write -> first MFMA C operand: s_nop 2 after the init block.
  - last MFMA -> v_accvgpr_read in the epilogue: s_nop 11 once.
  - MFMA -> MFMA accumulate chains on the same a-range: 0 wait states.
  - asm loads are invisible to hipcc: every wait is inside the strings.
"""

import os

OUT = os.path.join(os.path.dirname(__file__), "..", "senweaver_amd", "ops",
                   "csrc", "gemm_asm.hip")

# wave tile 128x64 = 8(mi) x 4(ni) fragments of 16x16; acc = a[(mi*4+ni)*4 ..]
def acc_range(mi, ni):
    base = (mi * 4 + ni) * 4
    return f"a[{base}:{base + 3}]"


def gen_khalf(kx: int, tail: str) -> str:
    """One k-half (K=32) statement: 12 ds_read_b128 + 32 MFMA, counted lgkm.

    Operand plan (outputs are early-clobber v-ranges):
      %0..%7   af0..af7  (A fragments, mi 0..7)
      %8..%11  bf0..bf3  (B fragments, ni 0..3)
      %12      aaddr (A slot byte address for this k-half, per-lane)
      %13      baddr
    """
    L = []
    # A mi0..3, B ni0..3 first (the first MFMA block's operands), then mi4..7
    for mi in range(4):
        L.append(f"ds_read_b128 %{mi}, %12 offset:{mi * 2048}")
    for ni in range(4):
        L.append(f"ds_read_b128 %{8 + ni}, %13 offset:{ni * 2048}")
    for mi in range(4, 8):
        L.append(f"ds_read_b128 %{mi}, %12 offset:{mi * 2048}")
    # first 8 reads landed -> mi0..3 x ni0..3
    L.append("s_waitcnt lgkmcnt(4)")
    for mi in range(4):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{mi}, %{8 + ni}, {acc_range(mi, ni)}")
    # the mi4..7 reads landed
    L.append("s_waitcnt lgkmcnt(0)")
    for mi in range(4, 8):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{mi}, %{8 + ni}, {acc_range(mi, ni)}")
    if tail:
        L.extend(tail.split(";"))
    body = "\\n\\t".join(L)
    outs = ", ".join(f'"=&v"(af[{i}])' for i in range(8))
    outs += ", " + ", ".join(f'"=&v"(bf[{i}])' for i in range(4))
    clob = ", ".join(f'"a{i}"' for i in range(128)) + ', "memory"'
    return (f'    asm volatile(\n        "{body}"\n'
            f'        : {outs}\n'
            f'        : "v"(aaddr{kx}), "v"(baddr{kx})\n'
            f'        : {clob});\n')


def gen_khalf_glds(kx: int, tail: str) -> str:
    """gen_khalf + 4 raw `global_load_lds_dwordx4` issues for the NEXT
    tile's operand, interleaved one per mi-row of the FIRST MFMA block
    (after that row's 4 MFMAs) — inside the hand-scheduled stream, AFTER
    the ds_read burst (the stage-first anti-pattern is glds ahead of the
    reads; v19/v20/v23 priced it at 5-15%).  m0 carries the LDS byte
    target, bumped 8192 per issue (4 chunks x 512 threads x 16 B).

    Extra operands: %14 = 64-bit sgpr base of the staged operand's K-row
    block, %15..%18 = per-lane byte offsets (st_off), %19 = m0 base
    (wave-uniform, readfirstlane'd by the caller).
    """
    L = []
    for mi in range(4):
        L.append(f"ds_read_b128 %{mi}, %12 offset:{mi * 2048}")
    for ni in range(4):
        L.append(f"ds_read_b128 %{8 + ni}, %13 offset:{ni * 2048}")
    for mi in range(4, 8):
        L.append(f"ds_read_b128 %{mi}, %12 offset:{mi * 2048}")
    L.append("s_waitcnt lgkmcnt(4)")
    for mi in range(4):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{mi}, %{8 + ni}, {acc_range(mi, ni)}")
        if mi == 0:
            L.append("s_mov_b32 m0, %19")
        else:
            L.append("s_add_u32 m0, m0, 8192")
        L.append(f"global_load_lds_dwordx4 %{15 + mi}, %14")
    L.append("s_waitcnt lgkmcnt(0)")
    for mi in range(4, 8):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{mi}, %{8 + ni}, {acc_range(mi, ni)}")
    if tail:
        L.extend(tail.split(";"))
    body = "\\n\\t".join(L)
    outs = ", ".join(f'"=&v"(af[{i}])' for i in range(8))
    outs += ", " + ", ".join(f'"=&v"(bf[{i}])' for i in range(4))
    op = "An" if kx == 0 else "Bn"
    m0 = "m0a" if kx == 0 else "m0b"
    ins = (f'"v"(aaddr{kx}), "v"(baddr{kx}), "s"({op}), '
           f'"v"(st_off[0]), "v"(st_off[1]), "v"(st_off[2]), "v"(st_off[3]), '
           f'"s"({m0})')
    clob = ", ".join(f'"a{i}"' for i in range(128)) + ', "memory"'
    return (f'    asm volatile(\n        "{body}"\n'
            f'        : {outs}\n'
            f'        : {ins}\n'
            f'        : {clob});\n')


def gen_tile_merged(tail: str) -> str:
    """Both k-halves in ONE statement: all 24 ds_read_b128 issued up-front,
    MFMAs retire behind a 4-step counted lgkmcnt ladder (16/12/4/0).

    Operands: %0..%11 kh0 frags (af0..7, bf0..3), %12..%23 kh1 frags,
    %24 aaddr0, %25 baddr0, %26 aaddr1, %27 baddr1.
    """
    L = []
    for mi in range(4):
        L.append(f"ds_read_b128 %{mi}, %24 offset:{mi * 2048}")
    for ni in range(4):
        L.append(f"ds_read_b128 %{8 + ni}, %25 offset:{ni * 2048}")
    for mi in range(4, 8):
        L.append(f"ds_read_b128 %{mi}, %24 offset:{mi * 2048}")
    # first MFMA block starts once kh0's first 8 reads land (lgkmcnt is a
    # 4-bit field: max 15 — kh1's reads are issued AFTER this block so no
    # wait ever exceeds the field width, and they overlap these MFMAs)
    L.append("s_waitcnt lgkmcnt(4)")
    for mi in range(4):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{mi}, %{8 + ni}, {acc_range(mi, ni)}")
    for mi in range(4):
        L.append(f"ds_read_b128 %{12 + mi}, %26 offset:{mi * 2048}")
    for ni in range(4):
        L.append(f"ds_read_b128 %{20 + ni}, %27 offset:{ni * 2048}")
    for mi in range(4, 8):
        L.append(f"ds_read_b128 %{12 + mi}, %26 offset:{mi * 2048}")
    L.append("s_waitcnt lgkmcnt(12)")
    for mi in range(4, 8):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{mi}, %{8 + ni}, {acc_range(mi, ni)}")
    L.append("s_waitcnt lgkmcnt(4)")
    for mi in range(4):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{12 + mi}, %{20 + ni}, {acc_range(mi, ni)}")
    L.append("s_waitcnt lgkmcnt(0)")
    for mi in range(4, 8):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{12 + mi}, %{20 + ni}, {acc_range(mi, ni)}")
    if tail:
        L.extend(tail.split(";"))
    body = "\\n\\t".join(L)
    outs = ", ".join(f'"=&v"(af[{i}])' for i in range(8))
    outs += ", " + ", ".join(f'"=&v"(bf[{i}])' for i in range(4))
    outs += ", " + ", ".join(f'"=&v"(af2[{i}])' for i in range(8))
    outs += ", " + ", ".join(f'"=&v"(bf2[{i}])' for i in range(4))
    clob = ", ".join(f'"a{i}"' for i in range(128)) + ', "memory"'
    return (f'    asm volatile(\n        "{body}"\n'
            f'        : {outs}\n'
            f'        : "v"(aaddr0), "v"(baddr0), "v"(aaddr1), "v"(baddr1)\n'
            f'        : {clob});\n')




def gen_v22_stmt1() -> str:
    """kh0 reads + kk0 MFMAs with kh1's reads issued under MFMA cover.
    Outputs: %0..%11 kh0 frags, %12..%23 kh1 frags; inputs %24..%27 addrs."""
    L = []
    for mi in range(4):
        L.append(f"ds_read_b128 %{mi}, %24 offset:{mi * 2048}")
    for ni in range(4):
        L.append(f"ds_read_b128 %{8 + ni}, %25 offset:{ni * 2048}")
    for mi in range(4, 8):
        L.append(f"ds_read_b128 %{mi}, %24 offset:{mi * 2048}")
    L.append("s_waitcnt lgkmcnt(4)")
    for mi in range(4):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{mi}, %{8 + ni}, {acc_range(mi, ni)}")
    for mi in range(4):
        L.append(f"ds_read_b128 %{12 + mi}, %26 offset:{mi * 2048}")
    for ni in range(4):
        L.append(f"ds_read_b128 %{20 + ni}, %27 offset:{ni * 2048}")
    for mi in range(4, 8):
        L.append(f"ds_read_b128 %{12 + mi}, %26 offset:{mi * 2048}")
    L.append("s_waitcnt lgkmcnt(12)")
    for mi in range(4, 8):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{mi}, %{8 + ni}, {acc_range(mi, ni)}")
    body = "\\n\\t".join(L)
    outs = ", ".join(f'"=&v"(af[{i}])' for i in range(8))
    outs += ", " + ", ".join(f'"=&v"(bf[{i}])' for i in range(4))
    outs += ", " + ", ".join(f'"=&v"(af2[{i}])' for i in range(8))
    outs += ", " + ", ".join(f'"=&v"(bf2[{i}])' for i in range(4))
    clob = ", ".join(f'"a{i}"' for i in range(128)) + ', "memory"'
    return (f'    asm volatile(\n        "{body}"\n'
            f'        : {outs}\n'
            f'        : "v"(aaddr0), "v"(baddr0), "v"(aaddr1), "v"(baddr1)\n'
            f'        : {clob});\n')


def gen_v22_stmt2(tail: str) -> str:
    """kk1 MFMAs consuming stmt1's kh1 fragments (inputs %0..%11)."""
    L = ["s_waitcnt lgkmcnt(4)"]
    for mi in range(4):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{mi}, %{8 + ni}, {acc_range(mi, ni)}")
    L.append("s_waitcnt lgkmcnt(0)")
    for mi in range(4, 8):
        for ni in range(4):
            L.append(f"v_mfma_f32_16x16x32_bf16 {acc_range(mi, ni)}, "
                     f"%{mi}, %{8 + ni}, {acc_range(mi, ni)}")
    L.extend(tail.split(";"))
    body = "\\n\\t".join(L)
    ins = ", ".join(f'"v"(af2[{i}])' for i in range(8))
    ins += ", " + ", ".join(f'"v"(bf2[{i}])' for i in range(4))
    clob = ", ".join(f'"a{i}"' for i in range(128)) + ', "memory"'
    return (f'    asm volatile(\n        "{body}"\n'
            f'        :\n        : {ins}\n'
            f'        : {clob});\n')


def main() -> None:
    init_writes = "\\n\\t".join(
        [f"v_accvgpr_write_b32 a{i}, 0" for i in range(128)] + ["s_nop 2"])
    acc_clob = ", ".join(f'"a{i}"' for i in range(128))

    epi_reads = []
    for r in range(32):
        base = r * 4
        stmt = "\\n\\t".join(
            f"v_accvgpr_read_b32 %{j}, a{base + j}" for j in range(4))
        nop = 's_nop 11\\n\\t' if r == 0 else ''
        epi_reads.append(
            f'  {{\n    float o0, o1, o2, o3;\n'
            f'    asm volatile("{nop}{stmt}"\n'
            f'        : "=v"(o0), "=v"(o1), "=v"(o2), "=v"(o3));\n'
            f'    const int mi = {r // 4}, ni = {r % 4};\n'
            f'    const long long row = c_row0 + mi * 16;\n'
            f'    ushort* crow;\n'
            f'    crow = C + (row + 0) * N; crow[c_col0 + ni * 16] = f2bf(o0);\n'
            f'    crow = C + (row + 1) * N; crow[c_col0 + ni * 16] = f2bf(o1);\n'
            f'    crow = C + (row + 2) * N; crow[c_col0 + ni * 16] = f2bf(o2);\n'
            f'    crow = C + (row + 3) * N; crow[c_col0 + ni * 16] = f2bf(o3);\n'
            f'  }}\n')
    epilogue = "".join(epi_reads)

    kh0 = gen_khalf(0, tail="")
    # steady-state tile end: boundary vmcnt(4) BEFORE the barrier (v9 contract)
    kh1_steady = gen_khalf(1, tail="s_waitcnt vmcnt(4);s_barrier")
    kh1_drain = gen_khalf(1, tail="s_waitcnt vmcnt(0);s_barrier")
    mg_steady = gen_tile_merged(tail="s_waitcnt vmcnt(4);s_barrier")
    mg_drain = gen_tile_merged(tail="s_waitcnt vmcnt(0);s_barrier")

    src = f"""// GENERATED by scripts/gen_gemm_asm.py — do not edit by hand.
//
// Hand-scheduled inline-asm bf16 GEMM (C = A @ B^T), 256x256 tile, 8 waves
// (2M x 4N, wave tile 128x64), BK=64 full-128B-row stage units, A dbuf +
// B ring-3 across 160 KiB LDS (the v9 pipeline geometry of gemm_pipe.hip),
// accumulators pinned in a[0:127].  Per K-tile: two k-half asm statements,
// each issuing its 12 ds_read_b128 then running 32 MFMAs behind a COUNTED
// lgkmcnt ladder (lgkmcnt(4) after 8 reads, lgkmcnt(0) for the rest) —
// reads stream under MFMAs instead of hipcc's full drain; the boundary
// s_waitcnt vmcnt(4) sits INSIDE the second statement before s_barrier.
// See scripts/gen_gemm_asm.py for the hazard discipline notes.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_asm_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                        ushort* __restrict__ C, int M, int N, int K) {{
  const int nwg = (M / 256) * (N / 256);
  int wgid = blockIdx.x;
  {{
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }}
  const int tiles_n = N / 256;
  const int GM = 8;
  const int tiles_m = M / 256;
  const int grp = wgid / (GM * tiles_n);
  const int rem = wgid % (GM * tiles_n);
  const int g0 = grp * GM;
  const int gh = (tiles_m - g0 < GM) ? (tiles_m - g0) : GM;
  const int tile_m = g0 + rem % gh;
  const int tile_n = rem / gh;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2;
  const int wn = wid & 3;
  const int l15 = lane & 15;
  const int kgrp = lane >> 4;

  __shared__ __attribute__((aligned(16))) ushort lds[5][256 * 64];

  const ushort* Atile = A + (long long)tile_m * 256 * K;
  const ushort* Btile = B + (long long)tile_n * 256 * K;

  const int swz0 = (kgrp + 2 * ((l15 >> 1) & 3)) & 7;
  const int frag0 = l15 * 128 + swz0 * 16;
  const int a_off = wm * 128 * 128 + frag0;
  const int b_off = wn * 64 * 128 + frag0;

  unsigned st_off[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {{
    const int s = i * 512 + tid;
    const int r = s >> 3;
    st_off[i] = ((unsigned)r * (unsigned)K
                 + ((((s & 7) - 2 * ((r >> 1) & 3)) & 7) * 8)) * 2u;  // bytes
  }}
  const int wave_chunk = tid & ~63;

  const int ntiles = K / 64;

#define AISSUE(TGT, OP, SLOT)                                                \\
  do {{                                                                      \\
    if ((TGT) < ntiles) {{                                                   \\
      const ushort* opk_ = (OP) + (TGT) * 64;                                \\
      ushort* dst_ = &lds[(SLOT)][0];                                        \\
      _Pragma("unroll") for (int i = 0; i < 4; ++i) {{                       \\
        const ushort* g = (const ushort*)((const char*)opk_ + st_off[i]);    \\
        __builtin_amdgcn_global_load_lds(                                    \\
            (const __attribute__((address_space(1))) unsigned int*)g,        \\
            (__attribute__((address_space(3))) unsigned int*)(dst_ +         \\
                (long long)(i * 512 + wave_chunk) * 8),                      \\
            16, 0, 0);                                                       \\
      }}                                                                     \\
    }}                                                                      \\
  }} while (0)

  // prologue: B(0)->2, A(0)->0, B(1)->3; then zero the AGPR accumulators
  AISSUE(0, Btile, 2);
  AISSUE(0, Atile, 0);
  AISSUE(1, Btile, 3);
  asm volatile("{init_writes}" ::: {acc_clob});
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < ntiles; ++t) {{
    const int aslot = t & 1;
    const int bslot = 2 + t % 3;
    const int bslot2 = 2 + (t + 2) % 3;
    // AS(3) pointer value == 32-bit LDS byte address (generic-pointer
    // truncation would be wrong)
    auto* ab3 = (__attribute__((address_space(3))) char*)&lds[aslot][0];
    auto* bb3 = (__attribute__((address_space(3))) char*)&lds[bslot][0];
    unsigned aaddr0 = (unsigned)(unsigned long long)ab3 + (unsigned)a_off;
    unsigned baddr0 = (unsigned)(unsigned long long)bb3 + (unsigned)b_off;
    unsigned aaddr1 = aaddr0 ^ 64u;
    unsigned baddr1 = baddr0 ^ 64u;
    short8 af[8], bf[4];
    // k-half 0: 12 reads + 32 MFMAs, counted lgkm ladder
{kh0}
    // prefetch next units while k-half 1 computes
    AISSUE(t + 1, Atile, aslot ^ 1);
    AISSUE(t + 2, Btile, bslot2);
    // k-half 1 + tile boundary (vmcnt BEFORE the rendezvous)
    if (t < ntiles - 2) {{
{kh1_steady}
    }} else {{
{kh1_drain}
    }}
  }}
#undef AISSUE

  const long long c_row0 = (long long)tile_m * 256 + wm * 128 + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * 256 + wn * 64 + l15;
{epilogue}}}

// v19: both k-halves merged into one statement per tile — 24 reads issued
// up-front, MFMAs retire behind a 16/12/4/0 counted lgkmcnt ladder; the
// next tile's staging is issued BEFORE the compute statement so the DMA
// has the whole tile in flight.
extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_asm2_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                         ushort* __restrict__ C, int M, int N, int K) {{
  const int nwg = (M / 256) * (N / 256);
  int wgid = blockIdx.x;
  {{
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }}
  const int tiles_n = N / 256;
  const int GM = 8;
  const int tiles_m = M / 256;
  const int grp = wgid / (GM * tiles_n);
  const int rem = wgid % (GM * tiles_n);
  const int g0 = grp * GM;
  const int gh = (tiles_m - g0 < GM) ? (tiles_m - g0) : GM;
  const int tile_m = g0 + rem % gh;
  const int tile_n = rem / gh;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2;
  const int wn = wid & 3;
  const int l15 = lane & 15;
  const int kgrp = lane >> 4;

  __shared__ __attribute__((aligned(16))) ushort lds[5][256 * 64];

  const ushort* Atile = A + (long long)tile_m * 256 * K;
  const ushort* Btile = B + (long long)tile_n * 256 * K;

  const int swz0 = (kgrp + 2 * ((l15 >> 1) & 3)) & 7;
  const int frag0 = l15 * 128 + swz0 * 16;
  const int a_off = wm * 128 * 128 + frag0;
  const int b_off = wn * 64 * 128 + frag0;

  unsigned st_off[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {{
    const int s = i * 512 + tid;
    const int r = s >> 3;
    st_off[i] = ((unsigned)r * (unsigned)K
                 + ((((s & 7) - 2 * ((r >> 1) & 3)) & 7) * 8)) * 2u;  // bytes
  }}
  const int wave_chunk = tid & ~63;

  const int ntiles = K / 64;

#define AISSUE2(TGT, OP, SLOT)                                               \\
  do {{                                                                      \\
    if ((TGT) < ntiles) {{                                                   \\
      const ushort* opk_ = (OP) + (TGT) * 64;                                \\
      ushort* dst_ = &lds[(SLOT)][0];                                        \\
      _Pragma("unroll") for (int i = 0; i < 4; ++i) {{                       \\
        const ushort* g = (const ushort*)((const char*)opk_ + st_off[i]);    \\
        __builtin_amdgcn_global_load_lds(                                    \\
            (const __attribute__((address_space(1))) unsigned int*)g,        \\
            (__attribute__((address_space(3))) unsigned int*)(dst_ +         \\
                (long long)(i * 512 + wave_chunk) * 8),                      \\
            16, 0, 0);                                                       \\
      }}                                                                     \\
    }}                                                                      \\
  }} while (0)

  AISSUE2(0, Btile, 2);
  AISSUE2(0, Atile, 0);
  AISSUE2(1, Btile, 3);
  asm volatile("{init_writes}" ::: {acc_clob});
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < ntiles; ++t) {{
    const int aslot = t & 1;
    const int bslot = 2 + t % 3;
    const int bslot2 = 2 + (t + 2) % 3;
    auto* ab3 = (__attribute__((address_space(3))) char*)&lds[aslot][0];
    auto* bb3 = (__attribute__((address_space(3))) char*)&lds[bslot][0];
    unsigned aaddr0 = (unsigned)(unsigned long long)ab3 + (unsigned)a_off;
    unsigned baddr0 = (unsigned)(unsigned long long)bb3 + (unsigned)b_off;
    unsigned aaddr1 = aaddr0 ^ 64u;
    unsigned baddr1 = baddr0 ^ 64u;
    short8 af[8], bf[4], af2[8], bf2[4];
    // stage FIRST: the DMA gets the whole tile's compute to hide under
    AISSUE2(t + 1, Atile, aslot ^ 1);
    AISSUE2(t + 2, Btile, bslot2);
    if (t < ntiles - 2) {{
{mg_steady}
    }} else {{
{mg_drain}
    }}
  }}
#undef AISSUE2

  const long long c_row0 = (long long)tile_m * 256 + wm * 128 + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * 256 + wn * 64 + l15;
{{epilogue2}}}}
"""
    src = src.replace("{epilogue2}", epilogue)  # same epilogue body

    # v20/v21: v18-template variants.  v20 stages BEFORE kh0; v21 adds the
    # static young-half s_setprio(1) (T5 static form).
    base = src[src.index("extern \"C\" __global__ void __launch_bounds__(512, 1)\ngemm_bt_bf16_asm_kernel"):
               src.index("// v19:")]
    v20 = base.replace("gemm_bt_bf16_asm_kernel", "gemm_bt_bf16_asm3_kernel")
    v20 = v20.replace("""    short8 af[8], bf[4];
    // k-half 0: 12 reads + 32 MFMAs, counted lgkm ladder""",
"""    short8 af[8], bf[4];
    AISSUE(t + 1, Atile, aslot ^ 1);
    AISSUE(t + 2, Btile, bslot2);
    // k-half 0: 12 reads + 32 MFMAs, counted lgkm ladder""")
    v20 = v20.replace("""    // prefetch next units while k-half 1 computes
    AISSUE(t + 1, Atile, aslot ^ 1);
    AISSUE(t + 2, Btile, bslot2);
""", "")
    st1 = gen_v22_stmt1()
    st2_steady = gen_v22_stmt2("s_waitcnt vmcnt(4);s_barrier")
    st2_drain = gen_v22_stmt2("s_waitcnt vmcnt(0);s_barrier")
    v22 = base.replace("gemm_bt_bf16_asm_kernel", "gemm_bt_bf16_asm5_kernel")
    ix0 = v22.index("    short8 af[8], bf[4];")
    ix1 = v22.index("  }\n#undef AISSUE")
    v22 = (v22[:ix0]
           + "    short8 af[8], bf[4], af2[8], bf2[4];\n"
           + st1
           + "    AISSUE(t + 1, Atile, aslot ^ 1);\n"
           + "    AISSUE(t + 2, Btile, bslot2);\n"
           + "    if (t < ntiles - 2) {\n" + st2_steady
           + "    } else {\n" + st2_drain + "    }\n"
           + v22[ix1:])
    v21 = base.replace("gemm_bt_bf16_asm_kernel", "gemm_bt_bf16_asm4_kernel")
    v21 = v21.replace("""  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < ntiles; ++t) {""",
"""  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // static young-half priority (T5)

  for (int t = 0; t < ntiles; ++t) {""")
    # v23 (asm6): A ring-of-3 + B double-buffer (3*32 + 2*32 = 160 KiB):
    # A(t+2) staged with TWO tiles of flight, B(t+1) staged at tile START
    # with a full tile of flight — the boundary vmcnt(4) then only ever
    # waits for loads that have had >= a full tile to land.
    v23 = base.replace("gemm_bt_bf16_asm_kernel", "gemm_bt_bf16_asm6_kernel")
    v23 = v23.replace("""  AISSUE(0, Btile, 2);
  AISSUE(0, Atile, 0);
  AISSUE(1, Btile, 3);""",
"""  AISSUE(0, Atile, 0);   // A ring slots 0,1,2; B dbuf slots 3,4
  AISSUE(0, Btile, 3);
  AISSUE(1, Atile, 1);""")
    v23 = v23.replace("""    const int aslot = t & 1;
    const int bslot = 2 + t % 3;
    const int bslot2 = 2 + (t + 2) % 3;""",
"""    const int aslot = t % 3;
    const int aslot2 = (t + 2) % 3;
    const int bslot = 3 + (t & 1);""")
    v23 = v23.replace("""    short8 af[8], bf[4];
    // k-half 0: 12 reads + 32 MFMAs, counted lgkm ladder""",
"""    short8 af[8], bf[4];
    AISSUE(t + 1, Btile, 3 + ((t + 1) & 1));  // the other B buffer
    // k-half 0: 12 reads + 32 MFMAs, counted lgkm ladder""")
    v23 = v23.replace("""    // prefetch next units while k-half 1 computes
    AISSUE(t + 1, Atile, aslot ^ 1);
    AISSUE(t + 2, Btile, bslot2);
""",
"""    // A(t+2) into its ring slot: two tiles of flight
    AISSUE(t + 2, Atile, aslot2);
""")
    # v24 (asm7): v21 + the next tile's staging issued as RAW
    # global_load_lds inside the asm statements (A(t+1) under kh0's first
    # MFMA block, B(t+2) under kh1's) — A gains ~a full k-half of extra
    # flight over the between-statements burst without ever preceding the
    # ds_read burst.
    kh0g = gen_khalf_glds(0, tail="")
    kh1g = gen_khalf_glds(1, tail="s_waitcnt vmcnt(4);s_barrier")
    v24 = base.replace("gemm_bt_bf16_asm_kernel", "gemm_bt_bf16_asm7_kernel")
    v24 = v24.replace("""  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < ntiles; ++t) {""",
"""  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // static young-half priority (T5)

  for (int t = 0; t < ntiles; ++t) {""")
    ia = v24.index("    short8 af[8], bf[4];")
    ib = v24.index("  }\n#undef AISSUE")
    v24 = (v24[:ia]
           + """    const unsigned wcb = (unsigned)((tid & ~63) * 16);
    const ushort* An = Atile + (long long)(t + 1) * 64;
    const ushort* Bn = Btile + (long long)(t + 2) * 64;
    const unsigned m0a = __builtin_amdgcn_readfirstlane(
        (unsigned)(unsigned long long)(__attribute__((address_space(3))) char*)&lds[aslot ^ 1][0] + wcb);
    const unsigned m0b = __builtin_amdgcn_readfirstlane(
        (unsigned)(unsigned long long)(__attribute__((address_space(3))) char*)&lds[bslot2][0] + wcb);
    short8 af[8], bf[4];
    if (t < ntiles - 2) {
""" + kh0g + kh1g + """    } else if (t == ntiles - 2) {
""" + kh0g + kh1_drain + """    } else {
""" + kh0 + kh1_drain + """    }
"""
           + v24[ib:])
    src = src + v20 + v21 + v22 + v23 + v24
    with open(os.path.abspath(OUT), "w") as f:
        f.write(src)
    print("wrote", OUT, len(src), "chars")


if __name__ == "__main__":
    main()
