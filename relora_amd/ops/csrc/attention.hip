#include "hip/hip_runtime.h"
// Causal flash attention (K3) for gfx950 — MFMA 16x16x32 bf16, LDS-tiled.
//
// Replaces the reference's SDPA call (reference modeling_llama.py:222-224,
// modeling_pythia.py:264-288): causal-only, no padding mask, dropout_p=0 —
// exactly the training configuration the reference uses.
//
// Structure (v2): 512-thread / 8-wave workgroups, 128 query (or kv) rows
// per block (16 per wave), 64-row K/V (or Q/dO) tiles, double-buffered LDS
// staging with register prefetch (issue the next tile's global loads before
// computing on the current one — T14 split), vectorized row staging plus
// vector-load/scalar-LDS-write transposes, and a per-wave private LDS
// region for the P/dS C→A-fragment relayout so the softmax step needs no
// block barrier.  Two __syncthreads per tile (the v1 kernel paid six per
// 64 kv rows and ran at ~70 TF).
//
// Row-major tiles are padded by 8 bf16 (16 B) by default: staging writes
// are conflict-free and fragment reads 2-way (bank model,
// tools/lds_bank_model.py).  Transposed tiles (V^T, K^T, Q^T, dO^T) use
// the rotated layout of t_rot() below: transpose writes conflict-free at
// hd64, everything else capped at 2-way.  Building with
// -DRELORA_AMD_ROT_V2=1 switches to the bank-model-solved variants
// (Q_V2 rotation + zero-pad XOR-swizzled row-major tiles) that are
// conflict-free on every pattern the model covers.
//
// Online softmax runs fully in registers on the MFMA C-layout (row r of a
// 16x16 tile lives in the 16 consecutive lanes with l>>4 == r>>2 at
// register r&3; row reductions are 4 shfl_xor steps over the low 4 lane
// bits).  head_dim <= 128; dims that are not multiples of 32 are
// zero-padded in the K-dim.
//
// Backward: FlashAttention-2 split — delta preprocess (rowsum(dO*O)), a dQ
// kernel (blocks over 128 q rows) and a dK/dV kernel (blocks over 128 kv
// rows), each recomputing P from the saved LSE.

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// Row-major LDS tiles: default = 8-element row padding (halves the classic
// same-slot conflict; K/P fragment reads stay 2-way per the bank model).
// RELORA_AMD_ROT_V2 = zero padding + per-row XOR swizzle on the element
// index — bank-model-verified conflict-free on every K/P write and read at
// hd64 and hd128, and saves the padding LDS.  The swizzle depth follows the
// row stride: 256-byte rows need (row&15), 128-byte rows (row&7).
#ifdef RELORA_AMD_ROT_V2
#define LPAD 0
#else
#define LPAD 8     // bf16 elements of row padding in LDS tiles (16B)
#endif
#define TILE 64    // staged tile rows (kv rows fwd/dq, q rows dkdv)

DEV_INLINE int row_swz(int row, int ldst) {
#ifdef RELORA_AMD_ROT_V2
  return ((ldst & 127) == 0 ? (row & 15) : (row & 7)) << 3;
#else
  (void)row; (void)ldst;
  return 0;
#endif
}

// swizzled element index into a row-major [R][ldst] LDS tile — EVERY
// producer and consumer of these tiles must address through this (or
// lds_frag/tile_write_rows, which do)
DEV_INLINE int lds_rm_idx(int row, int col, int ldst) {
  return (row * ldst + col) ^ row_swz(row, ldst);
}

DEV_INLINE float bf_to_f(__bf16 x) { return (float)x; }

// reduce over the 16 lanes of a C-fragment row group (low 4 lane bits)
DEV_INLINE float rowgroup_max(float x) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) x = fmaxf(x, __shfl_xor(x, m));
  return x;
}
DEV_INLINE float rowgroup_sum(float x) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) x += __shfl_xor(x, m);
  return x;
}

// load an A/B fragment from an LDS tile: lane reads row `row`, 8 elements
// at column k0. Caller guarantees 16B alignment (ldst multiple of 8).
DEV_INLINE bf16x8 lds_frag(const __bf16* tile, int row, int k0, int ldst) {
  return *reinterpret_cast<const bf16x8*>(tile + lds_rm_idx(row, k0, ldst));
}

// load an A fragment directly from global [S, hd]: lane row `grow`,
// 8 k-elements at k0; zero-pad outside.
DEV_INLINE bf16x8 global_frag(const __hip_bfloat16* src, int grow, int S, int hd,
                              int ld, int k0) {
  bf16x8 v;
  if (grow < S && k0 + 8 <= hd) {
    v = *reinterpret_cast<const bf16x8*>(src + (long)grow * ld + k0);
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      v[j] = (grow < S && k0 + j < hd) ? (__bf16)src[(long)grow * ld + k0 + j] : (__bf16)0.f;
  }
  return v;
}

// ---------------------------------------------------------------------------
// register-prefetch staging: a TILE x HD tile as NV bf16x8 slices per thread
// (512-thread mapping: slot = tid + i*512, row = slot/(HD/8), c8 = slot%(HD/8))
// ---------------------------------------------------------------------------

template <int HD, int NV, int NT = 512>
DEV_INLINE void tile_load_regs(bf16x8 (&r)[NV], const __hip_bfloat16* src,
                               int row0, int S, int hd, int ld) {
  constexpr int C8 = HD / 8;
#pragma unroll
  for (int i = 0; i < NV; ++i) {
    const int slot = threadIdx.x + i * NT;
    if (slot >= TILE * C8) break;
    const int row = slot / C8;
    const int c = (slot % C8) * 8;
    r[i] = global_frag(src, row0 + row, S, hd, ld, c);
  }
}

// write the registered tile row-major into LDS [TILE][ld]
template <int HD, int NV, int NT = 512>
DEV_INLINE void tile_write_rows(__bf16* dst, const bf16x8 (&r)[NV], int ld) {
  constexpr int C8 = HD / 8;
#pragma unroll
  for (int i = 0; i < NV; ++i) {
    const int slot = threadIdx.x + i * NT;
    if (slot >= TILE * C8) break;
    const int row = slot / C8;
    const int c = (slot % C8) * 8;
    *reinterpret_cast<bf16x8*>(dst + lds_rm_idx(row, c, ld)) = r[i];
  }
}

// Transposed tiles use a rotated layout: element (c, kv) lives at
// c*64 + t_rot(kv>>3, c)*8 + (kv&7).  The rotation keeps the 8-scalar
// transpose writes conflict-free (hd64) and caps reads at 2-way — verified
// by the in-tree bank model (tools/lds_bank_model.py); the naive dst[c][kv]
// layout put every column write of one instruction on one bank (8-16-way,
// 16% of dkdv wave cycles).  No row padding needed: stride is exactly 64.
//
// RELORA_AMD_ROT_V2: table-driven rotation (Q_V2 from the bank-model
// search) — fully conflict-free reads AND writes at hd64, conflict-free
// reads at hd128.  Numerics-neutral (same bijection used by writer and
// reader); compile with -DRELORA_AMD_ROT_V2=1 to A/B.
#ifdef RELORA_AMD_ROT_V2
__device__ constexpr unsigned char Q_V2[16][8] = {
    {4, 2, 0, 3, 6, 1, 7, 5}, {6, 0, 3, 4, 1, 5, 2, 7},
    {0, 6, 5, 1, 7, 3, 4, 2}, {0, 6, 1, 7, 5, 2, 3, 4},
    {1, 5, 2, 7, 6, 4, 0, 3}, {2, 5, 3, 0, 1, 4, 6, 7},
    {4, 6, 5, 3, 0, 1, 7, 2}, {1, 4, 2, 6, 7, 3, 5, 0},
    {3, 1, 7, 4, 5, 2, 6, 0}, {7, 3, 0, 5, 6, 2, 1, 4},
    {5, 7, 4, 6, 2, 0, 3, 1}, {3, 1, 6, 4, 0, 7, 2, 5},
    {2, 0, 1, 6, 3, 7, 5, 4}, {7, 4, 2, 5, 6, 3, 1, 0},
    {3, 7, 2, 0, 5, 4, 6, 1}, {2, 5, 7, 1, 0, 4, 6, 3}};
DEV_INLINE int t_rot(int kv_grp, int c) {
  return (Q_V2[c & 15][kv_grp] + 2 * (c >> 4)) & 7;
}
#else
DEV_INLINE int t_rot(int kv_grp, int c) {
  return ((kv_grp + (c >> 3) + (c & 7)) & 7);
}
#endif

// read a transposed-tile B fragment: channel row c, 8 kv at kv0 (mult of 8)
DEV_INLINE bf16x8 ldsT_frag(const __bf16* tile, int c, int kv0) {
  return *reinterpret_cast<const bf16x8*>(tile + c * TILE + t_rot(kv0 >> 3, c) * 8);
}

// ---------------------------------------------------------------------------
// TR layout: gfx950 ds_read_b64_tr_b16 hardware transpose-read (guide T10).
// The [TILE k-rows][HD cols] operand is stored as [k/4][n/16] subtiles of
// [4][16] row-major (64 elements, 128 B, contiguous) — the producing stage
// writes plain b128 rows (no scalar transpose pass, no rotation math), and
// the MFMA B-fragment materializes via two tr-reads per k-run of 8.
// Probe-verified semantics (tools/csrc/probe_tr.hip): each 16-lane group's
// 4-bf16 chunks FLAT[64] redistribute as OUT[lane][j] = FLAT[16j + lane].
// ---------------------------------------------------------------------------

typedef __attribute__((address_space(3))) const char as3c;

// write a register-staged tile into the TR image: thread's bf16x8 covers
// (row, c..c+8) -> one b128 store at subtile (row>>2, c>>4) offset
template <int HD, int NV>
DEV_INLINE void tile_write_tr(__bf16* dst, const bf16x8 (&r)[NV]) {
  constexpr int C8 = HD / 8;
  constexpr int N16 = HD / 16;
#pragma unroll
  for (int i = 0; i < NV; ++i) {
    const int slot = threadIdx.x + i * 512;
    if (slot >= TILE * C8) break;
    const int row = slot / C8;
    const int c = (slot % C8) * 8;
    const int off = ((row >> 2) * N16 + (c >> 4)) * 64 + (row & 3) * 16 + (c & 15);
    *reinterpret_cast<bf16x8*>(dst + off) = r[i];
  }
}

// B-fragment for the 32x32x16 MFMA from a TR image: lane l holds
// B[k0 + (l>>5)*8 + j][colbase + (l&31)], j = 0..7
template <int HD>
DEV_INLINE bf16x8 tr_frag(const __bf16* tile, int colbase, int k0, int lane) {
  constexpr int N16 = HD / 16;
  const int col = colbase + (lane & 31);
  const int kA = k0 + (lane >> 5) * 8;
  const int sub1 = (kA >> 2) * N16 + (col >> 4);
  const int sub2 = sub1 + N16;  // kA+4 row group
  const int lo = (lane & 15) * 4;
  as3c* a1 = (as3c*)(tile) + (sub1 * 64 + lo) * 2;
  as3c* a2 = (as3c*)(tile) + (sub2 * 64 + lo) * 2;
  unsigned long long r0, r1;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(r0), "=&v"(r1)
      : "v"(a1), "v"(a2));
  union { unsigned long long u[2]; bf16x8 v; } f;
  f.u[0] = r0;
  f.u[1] = r1;
  return f.v;
}

// write the registered tile transposed into LDS [HD][TILE] (rotated layout)
template <int HD, int NV, int NT = 512>
DEV_INLINE void tile_write_t(__bf16* dst, const bf16x8 (&r)[NV]) {
  constexpr int C8 = HD / 8;
#pragma unroll
  for (int i = 0; i < NV; ++i) {
    const int slot = threadIdx.x + i * NT;
    if (slot >= TILE * C8) break;
    const int row = slot / C8;
    const int cb = (slot % C8) * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = cb + j;
      dst[c * TILE + t_rot(row >> 3, c) * 8 + (row & 7)] = r[i][j];
    }
  }
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

template <int HD>  // padded head dim (multiple of 32), actual hd passed in
__global__ __launch_bounds__(512) void attn_fwd_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ out,
    float* __restrict__ lse, int S, int hd, int nh, long bst, long hst, int ld,
    float scale) {
  constexpr int KFRAGS = HD / 32;      // QK^T k-steps
  constexpr int NT_HD = HD / 16;       // PV hd tiles
  constexpr int LDK = HD + LPAD;
  constexpr int LDP = TILE + LPAD;
  constexpr int NV = (HD + 63) / 64;   // bf16x8 staging slices per thread
  // q row-fragments per wave: RF=2 (256-row blocks) halves K/V traffic and
  // staging per MFMA, but its register footprint costs hd64 its second
  // resident block (2 -> 1 block/CU, measured -20%); at hd>=96 residency is
  // 1 block/CU either way and RF=2 measured +25%
  constexpr int RF = (HD <= 64) ? 1 : 2;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds_k = (__bf16*)smem;              // [2][TILE][LDK]
  __bf16* lds_vt = lds_k + 2 * TILE * LDK;    // [2][HD][TILE] rotated
  __bf16* lds_p = lds_vt + 2 * HD * TILE;     // [RF][8][16][LDP]

  const int bh = blockIdx.y;
  // heavy blocks (high q_start: up to 8x the kv tiles of block 0) first,
  // so the causal work imbalance doesn't leave a long tail
  const int q_start = (gridDim.x - 1 - blockIdx.x) * (128 * RF);
  const long base = (long)(bh / nh) * bst + (long)(bh % nh) * hst;
  const __hip_bfloat16* qp = q + base;
  const __hip_bfloat16* kp = k + base;
  const __hip_bfloat16* vp = v + base;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;        // C-frag column / B-frag n / A-frag row
  const int kgrp = lane >> 4;       // k-element group (x8)

  // Q fragments: RF row groups of 16 per wave (A-layout); group rf covers
  // rows q_start + rf*128 + wave*16 + [0,16)
  bf16x8 qfrag[RF][KFRAGS];
#pragma unroll
  for (int rf = 0; rf < RF; ++rf)
#pragma unroll
    for (int kf = 0; kf < KFRAGS; ++kf)
      qfrag[rf][kf] = global_frag(qp, q_start + rf * 128 + wave * 16 + col,
                                  S, hd, ld, kf * 32 + kgrp * 8);

  float m_run[RF][4], l_run[RF][4];
  f32x4 o_acc[RF][NT_HD];
#pragma unroll
  for (int rf = 0; rf < RF; ++rf)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      m_run[rf][reg] = -INFINITY;
      l_run[rf][reg] = 0.f;
    }
#pragma unroll
  for (int rf = 0; rf < RF; ++rf)
#pragma unroll
    for (int t = 0; t < NT_HD; ++t) o_acc[rf][t] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int q_max_abs = min(q_start + 128 * RF - 1, S - 1);
  const int n_tiles = (q_max_abs / TILE) + 1;  // causal bound

  bf16x8 rk[NV], rv[NV];
  tile_load_regs<HD, NV>(rk, kp, 0, S, hd, ld);
  tile_load_regs<HD, NV>(rv, vp, 0, S, hd, ld);
  tile_write_rows<HD, NV>(lds_k, rk, LDK);
  tile_write_t<HD, NV>(lds_vt, rv);
  if (n_tiles > 1) {
    tile_load_regs<HD, NV>(rk, kp, TILE, S, hd, ld);
    tile_load_regs<HD, NV>(rv, vp, TILE, S, hd, ld);
  }
  __syncthreads();

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int cur = kt & 1;
    const int kv0 = kt * TILE;
    const __bf16* kb = lds_k + cur * TILE * LDK;
    const __bf16* vb = lds_vt + cur * HD * TILE;
    // per-element causal/valid checks only where the tile crosses the
    // diagonal or the sequence end
    const bool edge = (kv0 + TILE - 1 > q_start) || (kv0 + TILE > S);

    // stage tile kt+1 DURING compute (see the one-barrier-per-tile note in
    // the dq kernel): its last readers finished at the barrier ending tile
    // kt-1; placed between the two row-fragment passes' MFMA work below
    bool staged = false;

#pragma unroll
    for (int rf = 0; rf < RF; ++rf) {
      float p_val[4][4];
      __builtin_amdgcn_s_setprio(1);  // T5: favor the MFMA cluster
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kf = 0; kf < KFRAGS; ++kf)
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qfrag[rf][kf], lds_frag(kb, n * 16 + col, kf * 32 + kgrp * 8, LDK),
              acc, 0, 0, 0);
        if (edge) {
          const int kv_abs = kv0 + n * 16 + col;
#pragma unroll
          for (int reg = 0; reg < 4; ++reg) {
            const int row_abs = q_start + rf * 128 + wave * 16 + kgrp * 4 + reg;
            float s = acc[reg] * scale;
            if (kv_abs > row_abs || kv_abs >= S) s = -INFINITY;
            p_val[n][reg] = s;
          }
        } else {
#pragma unroll
          for (int reg = 0; reg < 4; ++reg) p_val[n][reg] = acc[reg] * scale;
        }
      }
      __builtin_amdgcn_s_setprio(0);

      if (!staged && kt + 1 < n_tiles) {
        staged = true;
        tile_write_rows<HD, NV>(lds_k + (cur ^ 1) * TILE * LDK, rk, LDK);
        tile_write_t<HD, NV>(lds_vt + (cur ^ 1) * HD * TILE, rv);
        if (kt + 2 < n_tiles) {
          tile_load_regs<HD, NV>(rk, kp, (kt + 2) * TILE, S, hd, ld);
          tile_load_regs<HD, NV>(rv, vp, (kt + 2) * TILE, S, hd, ld);
        }
      }

      // online softmax per row (4 regs per lane)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float rmax = fmaxf(fmaxf(p_val[0][reg], p_val[1][reg]),
                           fmaxf(p_val[2][reg], p_val[3][reg]));
        rmax = rowgroup_max(rmax);
        const float m_new = fmaxf(m_run[rf][reg], rmax);
        const float alpha = (m_new == -INFINITY) ? 1.f : __expf(m_run[rf][reg] - m_new);
        float rsum = 0.f;
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          const float p = (m_new == -INFINITY) ? 0.f : __expf(p_val[n][reg] - m_new);
          p_val[n][reg] = p;
          rsum += p;
        }
        rsum = rowgroup_sum(rsum);
        l_run[rf][reg] = l_run[rf][reg] * alpha + rsum;
        m_run[rf][reg] = m_new;
#pragma unroll
        for (int t = 0; t < NT_HD; ++t) o_acc[rf][t][reg] *= alpha;
      }

      // P relayout via this (wave, rf)'s private LDS region (no barrier)
      __bf16* pw = lds_p + (rf * 8 + wave) * 16 * LDP;
#pragma unroll
      for (int n = 0; n < 4; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg)
          pw[lds_rm_idx(kgrp * 4 + reg, n * 16 + col, LDP)] = (__bf16)p_val[n][reg];

      // PV: two K=32 steps over the 64-row kv tile
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const bf16x8 a = lds_frag(pw, col, ks * 32 + kgrp * 8, LDP);
#pragma unroll
        for (int t = 0; t < NT_HD; ++t)
          o_acc[rf][t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, ldsT_frag(vb, t * 16 + col, ks * 32 + kgrp * 8), o_acc[rf][t], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  // epilogue: O = o_acc / l, LSE = m + log(l)
#pragma unroll
  for (int rf = 0; rf < RF; ++rf)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int row_abs = q_start + rf * 128 + wave * 16 + kgrp * 4 + reg;
      if (row_abs >= S) continue;
      const float inv_l = (l_run[rf][reg] > 0.f) ? 1.f / l_run[rf][reg] : 0.f;
#pragma unroll
      for (int t = 0; t < NT_HD; ++t) {
        const int c = t * 16 + col;
        if (c < hd)
          out[base + (long)row_abs * ld + c] = __float2bfloat16(o_acc[rf][t][reg] * inv_l);
      }
      if (col == 0)
        lse[(long)bh * S + row_abs] = m_run[rf][reg] + __logf(fmaxf(l_run[rf][reg], 1e-30f));
    }
}

// ---------------------------------------------------------------------------
// forward v3 — swapped-operand 32x32x16 structure, softmax fully lane-local
//
// Per the CDNA4 guide's fused-attention pattern: compute S^T = K·Q^T
// (mfma(K, Q)) so the C-fragment's *column* index — which IS the lane index
// — is the query row.  Each lane then owns one q-row's scores (its 32 of
// the 64-kv tile; the l^32 partner lane owns the other 32), so the online
// softmax (max, exp, sum, rescale decision) is per-lane scalar code with a
// single v_permlane32_swap to merge partner halves — no cross-lane shfl
// chains and no P relayout through LDS.  P is packed to bf16 in-register
// (pair packs + two permlane32_swaps per k-step) directly into the MFMA
// B-fragment layout, and PV runs swapped too (O^T = V^T·P^T), which keeps
// the q index lane-local in the O accumulator: the online rescale and the
// final 1/l are per-lane scalar multiplies.  8 waves x 32 q-rows = 256
// q-rows per block; 64-row K/V tiles double-buffered with the same
// register-prefetch staging as v2.
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(16))) float f32x16;

// exchange a value with the lane^32 partner and combine
DEV_INLINE float swap_combine_max(float x) {
  auto r = __builtin_amdgcn_permlane32_swap(__float_as_int(x), __float_as_int(x), false, false);
  return fmaxf(__int_as_float(r[0]), __int_as_float(r[1]));
}
DEV_INLINE float swap_combine_sum(float x) {
  auto r = __builtin_amdgcn_permlane32_swap(__float_as_int(x), __float_as_int(x), false, false);
  return __int_as_float(r[0]) + __int_as_float(r[1]);
}

DEV_INLINE unsigned pack_bf16(float lo, float hi) {
  // scalar casts; the compiler fuses the pair (guide: hand-written
  // v_cvt_pk_bf16_f32 asm measured slower than letting it)
  union { __bf16 h[2]; unsigned u; } r;
  r.h[0] = (__bf16)lo;
  r.h[1] = (__bf16)hi;
  return r.u;
}

// Build the P^T B-fragment for MFMA k-step s (k = kv in [16s, 16s+16)) from
// the 32 per-lane S^T C-layout values of one kv subtile (p[0..15] = regs of
// subtile t = s>>1).  C-layout: reg r holds kv = 32t + (r&3) + 8*(r>>2) +
// 4*hi.  The fragment needs kv = 16s + hi*8 + [0,8): own even-quads supply
// half, the l^32 partner the other half — one permlane32_swap per dword
// pair completes it (guide T12).
DEV_INLINE bf16x8 pack_p_frag(const float* p16, int g) {
  unsigned a01 = pack_bf16(p16[8 * g + 0], p16[8 * g + 1]);
  unsigned a23 = pack_bf16(p16[8 * g + 2], p16[8 * g + 3]);
  unsigned b01 = pack_bf16(p16[8 * g + 4], p16[8 * g + 5]);
  unsigned b23 = pack_bf16(p16[8 * g + 6], p16[8 * g + 7]);
  auto r0 = __builtin_amdgcn_permlane32_swap((int)a01, (int)b01, false, false);
  auto r1 = __builtin_amdgcn_permlane32_swap((int)a23, (int)b23, false, false);
  union { unsigned u[4]; bf16x8 v; } f;
  f.u[0] = (unsigned)r0[0];
  f.u[1] = (unsigned)r1[0];
  f.u[2] = (unsigned)r0[1];
  f.u[3] = (unsigned)r1[1];
  return f.v;
}

#define LOG2E 1.44269504088896340736f
#define DEFER_MAX_THR 11.5f  // log2 domain ~ e^8 (guide T13; bf16 accum headroom)

template <int HD, int MINW = 2, bool TRR = false>
__global__ __launch_bounds__(512, MINW) void attn_fwd_v3_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ out,
    float* __restrict__ lse, int S, int hd, int nh, long bst, long hst, int ld,
    float scale) {
  constexpr int KSTEPS = HD / 16;  // QK^T k-steps per 32-kv subtile
  constexpr int NT32 = HD / 32;    // O^T 32-row hd tiles
  constexpr int LDK = HD + LPAD;
  constexpr int NV = (HD + 63) / 64;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds_k = (__bf16*)smem;            // [2][TILE][LDK] K rows
  __bf16* lds_vt = lds_k + 2 * TILE * LDK;  // [2][HD][TILE]  V^T rotated

  const int bh = blockIdx.y;
  const int q_start = (gridDim.x - 1 - blockIdx.x) * 256;  // heavy blocks first
  const long base = (long)(bh / nh) * bst + (long)(bh % nh) * hst;
  const __hip_bfloat16* qp = q + base;
  const __hip_bfloat16* kp = k + base;
  const __hip_bfloat16* vp = v + base;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int qcol = lane & 31;  // this lane's q row (within the wave's 32)
  const int hi = lane >> 5;    // k-run half selector
  const int q_abs = q_start + wave * 32 + qcol;

  // Q as B-fragments: lane holds Q[q_abs][ks*16 + hi*8 + 0..8)
  bf16x8 qf[KSTEPS];
#pragma unroll
  for (int ks = 0; ks < KSTEPS; ++ks)
    qf[ks] = global_frag(qp, q_abs, S, hd, ld, ks * 16 + hi * 8);

  float m2 = -INFINITY;  // running max of scores * scale * log2e
  float l_run = 0.f;
  f32x16 o_acc[NT32];
#pragma unroll
  for (int t = 0; t < NT32; ++t) o_acc[t] = (f32x16)(0.f);

  const int q_max_abs = min(q_start + 255, S - 1);
  const int n_tiles = (q_max_abs / TILE) + 1;
  const float sc2 = scale * LOG2E;

  bf16x8 rk[NV], rv[NV];
  tile_load_regs<HD, NV>(rk, kp, 0, S, hd, ld);
  tile_load_regs<HD, NV>(rv, vp, 0, S, hd, ld);
  if (TRR) tile_write_tr<HD, NV>(lds_vt, rv);
  else tile_write_t<HD, NV>(lds_vt, rv);
  tile_write_rows<HD, NV>(lds_k, rk, LDK);
  if (n_tiles > 1) {
    tile_load_regs<HD, NV>(rk, kp, TILE, S, hd, ld);
    tile_load_regs<HD, NV>(rv, vp, TILE, S, hd, ld);
  }
  __syncthreads();

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int cur = kt & 1;
    const int kv0 = kt * TILE;
    const __bf16* kb = lds_k + cur * TILE * LDK;
    const __bf16* vtb = lds_vt + cur * HD * TILE;

    // two independent online-softmax updates per staged tile, one per 32-kv
    // subtile: halves the live score registers (16 f32, not 32) — keeps the
    // hd64 kernel under the 128-VGPR / 2-blocks-per-CU occupancy cliff
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      const int kv0s = kv0 + 32 * t;
      // wave-uniform edge: the wave's lowest q row decides the masked path
      const bool edge = (kv0s + 31 > q_start + wave * 32) || (kv0s + 32 > S);
      if (kv0s > q_max_abs) break;  // fully-masked subtile (block-uniform)

      float p_val[16];
      __builtin_amdgcn_s_setprio(1);
      {
        f32x16 acc = (f32x16)(0.f);
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks)
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              lds_frag(kb, 32 * t + qcol, ks * 16 + hi * 8, LDK), qf[ks], acc, 0, 0, 0);
        if (edge) {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kv_abs = kv0s + (r & 3) + 8 * (r >> 2) + 4 * hi;
            p_val[r] = (kv_abs > q_abs || kv_abs >= S) ? -INFINITY : acc[r] * sc2;
          }
        } else {
#pragma unroll
          for (int r = 0; r < 16; ++r) p_val[r] = acc[r] * sc2;
        }
      }
      __builtin_amdgcn_s_setprio(0);

      // stage tile kt+1 between the two MFMA clusters (T14 split)
      if (t == 0 && kt + 1 < n_tiles) {
        tile_write_rows<HD, NV>(lds_k + (cur ^ 1) * TILE * LDK, rk, LDK);
        if (TRR) tile_write_tr<HD, NV>(lds_vt + (cur ^ 1) * HD * TILE, rv);
        else tile_write_t<HD, NV>(lds_vt + (cur ^ 1) * HD * TILE, rv);
        if (kt + 2 < n_tiles) {
          tile_load_regs<HD, NV>(rk, kp, (kt + 2) * TILE, S, hd, ld);
          tile_load_regs<HD, NV>(rv, vp, (kt + 2) * TILE, S, hd, ld);
        }
      }

      // online softmax, fully per-lane (q = lane&31)
      float m_tile = p_val[0];
#pragma unroll
      for (int i = 1; i < 16; ++i) m_tile = fmaxf(m_tile, p_val[i]);
      m_tile = swap_combine_max(m_tile);

      // defer-max: only rescale when the subtile max exceeds the running max
      // by more than THR (textbook-safe order — decision precedes this
      // subtile's exponentiation, l-update follows at the same scale)
      if (__any(m_tile > m2 + DEFER_MAX_THR) || m2 == -INFINITY) {
        const float m_new = fmaxf(m2, m_tile);
        const float alpha = (m_new == -INFINITY) ? 0.f : exp2f(m2 - m_new);
        l_run *= alpha;
#pragma unroll
        for (int ht = 0; ht < NT32; ++ht)
#pragma unroll
          for (int r = 0; r < 16; ++r) o_acc[ht][r] *= alpha;
        m2 = m_new;
      }

      float rsum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const float p = (m2 == -INFINITY) ? 0.f : exp2f(p_val[r] - m2);
        p_val[r] = p;
        rsum += p;
      }
      l_run += swap_combine_sum(rsum);

      // PV swapped: O^T += V^T · P^T, 2 k-steps over the 32-kv subtile
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        const bf16x8 pfrag = pack_p_frag(p_val, g);
        const int s = 2 * t + g;
#pragma unroll
        for (int ht = 0; ht < NT32; ++ht) {
          const bf16x8 vf = TRR ? tr_frag<HD>(vtb, ht * 32, s * 16, lane)
                                : ldsT_frag(vtb, ht * 32 + qcol, s * 16 + hi * 8);
          o_acc[ht] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag, o_acc[ht], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  // epilogue: per-lane 1/l, packed b32 stores (hd pairs are reg pairs)
  if (q_abs < S) {
    const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
    for (int ht = 0; ht < NT32; ++ht)
#pragma unroll
      for (int d = 0; d < 8; ++d) {
        const int hd_c = ((2 * d) & 3) + 8 * ((2 * d) >> 2) + 4 * hi + 32 * ht;
        if (hd_c + 1 < hd) {
          const unsigned w =
              pack_bf16(o_acc[ht][2 * d] * inv_l, o_acc[ht][2 * d + 1] * inv_l);
          *reinterpret_cast<unsigned*>(out + base + (long)q_abs * ld + hd_c) = w;
        } else if (hd_c < hd) {
          out[base + (long)q_abs * ld + hd_c] = __float2bfloat16(o_acc[ht][2 * d] * inv_l);
        }
      }
    if (hi == 0)
      lse[(long)bh * S + q_abs] =
          (m2 + log2f(fmaxf(l_run, 1e-30f))) * 0.6931471805599453f;
  }
}

// ---------------------------------------------------------------------------
// backward: delta preprocess
// ---------------------------------------------------------------------------

// delta = rowsum(dO * O): each wave handles 8 rows (hd <= 128), bf16x8
// vector loads, intra-8-lane-group reduction — 256-thread blocks cover 32
// rows each (the one-row-per-block version was 5% of backward time)
__global__ __launch_bounds__(256) void attn_delta_kernel(
    const __hip_bfloat16* __restrict__ dout, const __hip_bfloat16* __restrict__ o,
    float* __restrict__ delta, long n_rows, int hd, int nh, int S,
    long bst, long hst, int ld) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int seg = lane >> 3;       // 8 lanes per row
  const int sub = lane & 7;
  const long row = (long)blockIdx.x * 32 + wave * 8 + seg;  // (b*nh + h)*S + s
  if (row >= n_rows) return;
  const long s = row % S;
  const long h = (row / S) % nh;
  const long b = row / ((long)S * nh);
  const long off = b * bst + h * hst + s * ld;
  float acc = 0.f;
  for (int c = sub * 8; c < hd; c += 64) {
    if (c + 8 <= hd) {
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(dout + off + c);
      const bf16x8 bb = *reinterpret_cast<const bf16x8*>(o + off + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc += (float)a[j] * (float)bb[j];
    } else {
      for (int j = c; j < hd; ++j)
        acc += to_f32(dout[off + j]) * to_f32(o[off + j]);
    }
  }
#pragma unroll
  for (int m = 1; m < 8; m <<= 1) acc += __shfl_xor(acc, m);
  if (sub == 0) delta[row] = acc;
}

// ---------------------------------------------------------------------------
// backward v3 — same swapped-operand 32x32x16 structure as the v3 forward.
//
// dQ kernel: S^T = K·Q^T and dP^T = V·dO^T keep the q index lane-local
// (one lse/delta load per lane); dS^T is packed in-register into B-frags
// and dQ^T = K^T·dS^T accumulates with q lane-local — per-lane epilogue.
//
// dK/dV kernel: S = Q·K^T and dP = dO·V^T keep the KV index lane-local
// (dk/dv accumulators stay in this block); lse/delta are read as guarded
// f32x4 quads (the C-layout row quads are q-contiguous); P^T and dS^T pack
// in-register into A-frags for dV += P^T·dO and dK += dS^T·Q.
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(4))) float f32x4v;

DEV_INLINE f32x4 load_f32x4_guard(const float* p, long idx, long n) {
  f32x4 r;
  if (idx + 3 < n) {
    r = *reinterpret_cast<const f32x4*>(p + idx);
  } else {
#pragma unroll
    for (int i = 0; i < 4; ++i) r[i] = (idx + i < n) ? p[idx + i] : 0.f;
  }
  return r;
}

template <int HD, int NWAVE = 8, int MINW = 2>  // NWAVE*32 q rows per block;
                                  // MINW=4 caps VGPRs at 128 (occ experiment)
__global__ __launch_bounds__(NWAVE * 64, MINW) void attn_bwd_dq_v3_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    __hip_bfloat16* __restrict__ dq, int S, int hd, int nh, long bst, long hst,
    int ld, float scale) {
  constexpr int KSTEPS = HD / 16;
  constexpr int NT32 = HD / 32;
  constexpr int LDK = HD + LPAD;
  constexpr int NT = NWAVE * 64;
  constexpr int NV = (TILE * (HD / 8) + NT - 1) / NT;
  constexpr int QROWS = NWAVE * 32;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds_k = (__bf16*)smem;             // [2][TILE][LDK] K rows
  __bf16* lds_v = lds_k + 2 * TILE * LDK;    // [2][TILE][LDK] V rows
  __bf16* lds_kt = lds_v + 2 * TILE * LDK;   // [2][HD][TILE]  K^T rotated

  const int bh = blockIdx.y;
  const int q_start = (gridDim.x - 1 - blockIdx.x) * QROWS;  // heavy blocks first
  const long base = (long)(bh / nh) * bst + (long)(bh % nh) * hst;
  const __hip_bfloat16* qp = q + base;
  const __hip_bfloat16* kp = k + base;
  const __hip_bfloat16* vp = v + base;
  const __hip_bfloat16* dop = dout + base;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int qcol = lane & 31;
  const int hi = lane >> 5;
  const int q_abs = q_start + wave * 32 + qcol;

  bf16x8 qf[KSTEPS], dof[KSTEPS];
#pragma unroll
  for (int ks = 0; ks < KSTEPS; ++ks) {
    qf[ks] = global_frag(qp, q_abs, S, hd, ld, ks * 16 + hi * 8);
    dof[ks] = global_frag(dop, q_abs, S, hd, ld, ks * 16 + hi * 8);
  }
  const float sc2 = scale * LOG2E;
  const float lse2 = (q_abs < S) ? lse[(long)bh * S + q_abs] * LOG2E : 0.f;
  const float delt = (q_abs < S) ? delta[(long)bh * S + q_abs] : 0.f;

  f32x16 dq_acc[NT32];
#pragma unroll
  for (int t = 0; t < NT32; ++t) dq_acc[t] = (f32x16)(0.f);

  const int q_max_abs = min(q_start + QROWS - 1, S - 1);
  const int n_tiles = (q_max_abs / TILE) + 1;

  bf16x8 rk[NV], rv[NV];
  tile_load_regs<HD, NV, NT>(rk, kp, 0, S, hd, ld);
  tile_load_regs<HD, NV, NT>(rv, vp, 0, S, hd, ld);
  tile_write_rows<HD, NV, NT>(lds_k, rk, LDK);
  tile_write_rows<HD, NV, NT>(lds_v, rv, LDK);
  tile_write_t<HD, NV, NT>(lds_kt, rk);
  if (n_tiles > 1) {
    tile_load_regs<HD, NV, NT>(rk, kp, TILE, S, hd, ld);
    tile_load_regs<HD, NV, NT>(rv, vp, TILE, S, hd, ld);
  }
  __syncthreads();

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int cur = kt & 1;
    const int kv0 = kt * TILE;
    const __bf16* kb = lds_k + cur * TILE * LDK;
    const __bf16* vb = lds_v + cur * TILE * LDK;
    const __bf16* ktb = lds_kt + cur * HD * TILE;

    // per-32-kv subtile (halves live score registers, like the v3 forward)
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      const int kv0s = kv0 + 32 * t;
      const bool edge_t = (kv0s + 31 > q_start + wave * 32) || (kv0s + 32 > S);
      if (kv0s > q_max_abs) break;

      float ds_val[16];
      __builtin_amdgcn_s_setprio(1);
      {
        f32x16 sa = (f32x16)(0.f), dpa = (f32x16)(0.f);
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks) {
          const int c0 = ks * 16 + hi * 8;
          sa = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              lds_frag(kb, 32 * t + qcol, c0, LDK), qf[ks], sa, 0, 0, 0);
          dpa = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              lds_frag(vb, 32 * t + qcol, c0, LDK), dof[ks], dpa, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float p;
          if (edge_t) {
            const int kv_abs = kv0s + (r & 3) + 8 * (r >> 2) + 4 * hi;
            p = (kv_abs <= q_abs && kv_abs < S && q_abs < S)
                    ? exp2f(sa[r] * sc2 - lse2) : 0.f;
          } else {
            p = exp2f(sa[r] * sc2 - lse2);
          }
          ds_val[r] = p * (dpa[r] - delt) * scale;
        }
      }
      __builtin_amdgcn_s_setprio(0);

      if (t == 0 && kt + 1 < n_tiles) {
        tile_write_rows<HD, NV, NT>(lds_k + (cur ^ 1) * TILE * LDK, rk, LDK);
        tile_write_rows<HD, NV, NT>(lds_v + (cur ^ 1) * TILE * LDK, rv, LDK);
        tile_write_t<HD, NV, NT>(lds_kt + (cur ^ 1) * HD * TILE, rk);
        if (kt + 2 < n_tiles) {
          tile_load_regs<HD, NV, NT>(rk, kp, (kt + 2) * TILE, S, hd, ld);
          tile_load_regs<HD, NV, NT>(rv, vp, (kt + 2) * TILE, S, hd, ld);
        }
      }

      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        const bf16x8 dsf = pack_p_frag(ds_val, g);
        const int s = 2 * t + g;
#pragma unroll
        for (int ht = 0; ht < NT32; ++ht)
          dq_acc[ht] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              ldsT_frag(ktb, ht * 32 + qcol, s * 16 + hi * 8), dsf, dq_acc[ht], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  if (q_abs < S) {
#pragma unroll
    for (int ht = 0; ht < NT32; ++ht)
#pragma unroll
      for (int d = 0; d < 8; ++d) {
        const int hd_c = ((2 * d) & 3) + 8 * ((2 * d) >> 2) + 4 * hi + 32 * ht;
        if (hd_c + 1 < hd) {
          const unsigned w = pack_bf16(dq_acc[ht][2 * d], dq_acc[ht][2 * d + 1]);
          *reinterpret_cast<unsigned*>(dq + base + (long)q_abs * ld + hd_c) = w;
        } else if (hd_c < hd) {
          dq[base + (long)q_abs * ld + hd_c] = __float2bfloat16(dq_acc[ht][2 * d]);
        }
      }
  }
}

template <int HD, bool TRR = false>  // HD <= 64; TRR: tr-read Q^T/dO^T images
__global__ __launch_bounds__(512) void attn_bwd_dkdv_v3_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    __hip_bfloat16* __restrict__ dk, __hip_bfloat16* __restrict__ dv,
    int S, int hd, int nh, long bst, long hst, int ld, float scale) {
  constexpr int KSTEPS = HD / 16;
  constexpr int NT32 = HD / 32;
  constexpr int LDK = HD + LPAD;
  constexpr int NV = (HD + 63) / 64;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds_q = (__bf16*)smem;              // [2][TILE][LDK] Q rows
  __bf16* lds_do = lds_q + 2 * TILE * LDK;    // [2][TILE][LDK] dO rows
  __bf16* lds_qt = lds_do + 2 * TILE * LDK;   // [2][HD][TILE]  Q^T rotated
  __bf16* lds_dot = lds_qt + 2 * HD * TILE;   // [2][HD][TILE]  dO^T rotated

  const int bh = blockIdx.y;
  const int kv_start = blockIdx.x * 256;
  const long base = (long)(bh / nh) * bst + (long)(bh % nh) * hst;
  const __hip_bfloat16* qp = q + base;
  const __hip_bfloat16* kp = k + base;
  const __hip_bfloat16* vp = v + base;
  const __hip_bfloat16* dop = dout + base;
  const float* lsep = lse + (long)bh * S;
  const float* deltap = delta + (long)bh * S;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int kvcol = lane & 31;
  const int hi = lane >> 5;
  const int kv_abs = kv_start + wave * 32 + kvcol;  // this lane's kv row

  bf16x8 kf[KSTEPS], vf[KSTEPS];
#pragma unroll
  for (int ks = 0; ks < KSTEPS; ++ks) {
    kf[ks] = global_frag(kp, kv_abs, S, hd, ld, ks * 16 + hi * 8);
    vf[ks] = global_frag(vp, kv_abs, S, hd, ld, ks * 16 + hi * 8);
  }
  const float sc2 = scale * LOG2E;

  f32x16 dk_acc[NT32], dv_acc[NT32];
#pragma unroll
  for (int t = 0; t < NT32; ++t) {
    dk_acc[t] = (f32x16)(0.f);
    dv_acc[t] = (f32x16)(0.f);
  }

  const int first_qt = kv_start / TILE;
  const int n_q_tiles = (S + TILE - 1) / TILE;

  bf16x8 rq[NV], rdo[NV];
  tile_load_regs<HD, NV>(rq, qp, first_qt * TILE, S, hd, ld);
  tile_load_regs<HD, NV>(rdo, dop, first_qt * TILE, S, hd, ld);
  tile_write_rows<HD, NV>(lds_q, rq, LDK);
  tile_write_rows<HD, NV>(lds_do, rdo, LDK);
  if (TRR) {
    tile_write_tr<HD, NV>(lds_qt, rq);
    tile_write_tr<HD, NV>(lds_dot, rdo);
  } else {
    tile_write_t<HD, NV>(lds_qt, rq);
    tile_write_t<HD, NV>(lds_dot, rdo);
  }
  if (first_qt + 1 < n_q_tiles) {
    tile_load_regs<HD, NV>(rq, qp, (first_qt + 1) * TILE, S, hd, ld);
    tile_load_regs<HD, NV>(rdo, dop, (first_qt + 1) * TILE, S, hd, ld);
  }
  __syncthreads();

  for (int qt = first_qt; qt < n_q_tiles; ++qt) {
    const int cur = qt & 1;
    const int q0 = qt * TILE;
    const __bf16* qb = lds_q + cur * TILE * LDK;
    const __bf16* dob = lds_do + cur * TILE * LDK;
    const __bf16* qtb = lds_qt + cur * HD * TILE;
    const __bf16* dotb = lds_dot + cur * HD * TILE;

    // per-32-q subtile: halves the live P/dS registers and loads lse/delta
    // one quad at a time — this is what keeps the hd64 kernel from spilling
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      const int q0s = q0 + 32 * t;
      const bool edge_t = (q0s < kv_start + 255) || (q0s + 32 > S);

      float pt_val[16], dst_val[16];
      __builtin_amdgcn_s_setprio(1);
      {
        f32x16 sa = (f32x16)(0.f), dpa = (f32x16)(0.f);
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks) {
          const int c0 = ks * 16 + hi * 8;
          sa = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              lds_frag(qb, 32 * t + (lane & 31), c0, LDK), kf[ks], sa, 0, 0, 0);
          dpa = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              lds_frag(dob, 32 * t + (lane & 31), c0, LDK), vf[ks], dpa, 0, 0, 0);
        }
#pragma unroll
        for (int qd = 0; qd < 4; ++qd) {
          // lse/delta for this C-layout row quad (q-contiguous)
          const long qrow = q0s + 8 * qd + 4 * hi;
          const f32x4 lse4 = load_f32x4_guard(lsep, qrow, S);
          const f32x4 dl4 = load_f32x4_guard(deltap, qrow, S);
#pragma unroll
          for (int i = 0; i < 4; ++i) {
            const int r = 4 * qd + i;
            const float lse2 = lse4[i] * LOG2E;
            float p;
            if (edge_t) {
              const int q_abs_r = q0s + (r & 3) + 8 * (r >> 2) + 4 * hi;
              p = (q_abs_r >= kv_abs && q_abs_r < S && kv_abs < S)
                      ? exp2f(sa[r] * sc2 - lse2) : 0.f;
            } else {
              p = exp2f(sa[r] * sc2 - lse2);
            }
            pt_val[r] = p;
            dst_val[r] = p * (dpa[r] - dl4[i]) * scale;
          }
        }
      }
      __builtin_amdgcn_s_setprio(0);

      if (t == 0 && qt + 1 < n_q_tiles) {
        tile_write_rows<HD, NV>(lds_q + (cur ^ 1) * TILE * LDK, rq, LDK);
        tile_write_rows<HD, NV>(lds_do + (cur ^ 1) * TILE * LDK, rdo, LDK);
        if (TRR) {
          tile_write_tr<HD, NV>(lds_qt + (cur ^ 1) * HD * TILE, rq);
          tile_write_tr<HD, NV>(lds_dot + (cur ^ 1) * HD * TILE, rdo);
        } else {
          tile_write_t<HD, NV>(lds_qt + (cur ^ 1) * HD * TILE, rq);
          tile_write_t<HD, NV>(lds_dot + (cur ^ 1) * HD * TILE, rdo);
        }
        if (qt + 2 < n_q_tiles) {
          tile_load_regs<HD, NV>(rq, qp, (qt + 2) * TILE, S, hd, ld);
          tile_load_regs<HD, NV>(rdo, dop, (qt + 2) * TILE, S, hd, ld);
        }
      }

      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        const bf16x8 pa = pack_p_frag(pt_val, g);
        const bf16x8 dsa = pack_p_frag(dst_val, g);
        const int s = 2 * t + g;
#pragma unroll
        for (int ht = 0; ht < NT32; ++ht) {
          const bf16x8 dof = TRR ? tr_frag<HD>(dotb, ht * 32, s * 16, lane)
                                 : ldsT_frag(dotb, ht * 32 + (lane & 31), s * 16 + hi * 8);
          const bf16x8 qf2 = TRR ? tr_frag<HD>(qtb, ht * 32, s * 16, lane)
                                 : ldsT_frag(qtb, ht * 32 + (lane & 31), s * 16 + hi * 8);
          dv_acc[ht] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, dof, dv_acc[ht], 0, 0, 0);
          dk_acc[ht] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dsa, qf2, dk_acc[ht], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  // epilogue: lane holds one hd column x 16 kv rows per 32-tile
#pragma unroll
  for (int ht = 0; ht < NT32; ++ht) {
    const int hd_c = 32 * ht + (lane & 31);
    if (hd_c >= hd) continue;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kv_row = kv_start + wave * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (kv_row < S) {
        dk[base + (long)kv_row * ld + hd_c] = __float2bfloat16(dk_acc[ht][r]);
        dv[base + (long)kv_row * ld + hd_c] = __float2bfloat16(dv_acc[ht][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// backward: dQ kernel — blocks over 128 q rows, kv tiles of 64
// ---------------------------------------------------------------------------

template <int HD>
__global__ __launch_bounds__(512) void attn_bwd_dq_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    __hip_bfloat16* __restrict__ dq, int S, int hd, int nh, long bst, long hst,
    int ld, float scale) {
  constexpr int KFRAGS = HD / 32;
  constexpr int NT_HD = HD / 16;
  constexpr int LDK = HD + LPAD;
  constexpr int LDT = TILE + LPAD;
  constexpr int NV = (HD + 63) / 64;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds_k = (__bf16*)smem;              // [2][TILE][LDK] K rows
  __bf16* lds_v = lds_k + 2 * TILE * LDK;     // [2][TILE][LDK] V rows
  __bf16* lds_kt = lds_v + 2 * TILE * LDK;    // [2][HD][TILE]  K transposed (rotated)
  __bf16* lds_p = lds_kt + 2 * HD * TILE;     // [8][16][LDT]

  const int bh = blockIdx.y;
  const int q_start = (gridDim.x - 1 - blockIdx.x) * 128;  // heavy blocks first
  const long base = (long)(bh / nh) * bst + (long)(bh % nh) * hst;
  const __hip_bfloat16* qp = q + base;
  const __hip_bfloat16* kp = k + base;
  const __hip_bfloat16* vp = v + base;
  const __hip_bfloat16* dop = dout + base;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;
  const int kgrp = lane >> 4;

  const int qrow_abs = q_start + wave * 16 + col;
  bf16x8 qfrag[KFRAGS], dofrag[KFRAGS];
#pragma unroll
  for (int kf = 0; kf < KFRAGS; ++kf) {
    qfrag[kf] = global_frag(qp, qrow_abs, S, hd, ld, kf * 32 + kgrp * 8);
    dofrag[kf] = global_frag(dop, qrow_abs, S, hd, ld, kf * 32 + kgrp * 8);
  }
  // per-reg row stats
  float lse_r[4], delta_r[4];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int r = q_start + wave * 16 + kgrp * 4 + reg;
    lse_r[reg] = (r < S) ? lse[(long)bh * S + r] : 0.f;
    delta_r[reg] = (r < S) ? delta[(long)bh * S + r] : 0.f;
  }

  f32x4 dq_acc[NT_HD];
#pragma unroll
  for (int t = 0; t < NT_HD; ++t) dq_acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int q_max_abs = min(q_start + 127, S - 1);
  const int n_tiles = (q_max_abs / TILE) + 1;

  bf16x8 rk[NV], rv[NV];
  tile_load_regs<HD, NV>(rk, kp, 0, S, hd, ld);
  tile_load_regs<HD, NV>(rv, vp, 0, S, hd, ld);
  tile_write_rows<HD, NV>(lds_k, rk, LDK);
  tile_write_rows<HD, NV>(lds_v, rv, LDK);
  tile_write_t<HD, NV>(lds_kt, rk);
  if (n_tiles > 1) {
    tile_load_regs<HD, NV>(rk, kp, TILE, S, hd, ld);
    tile_load_regs<HD, NV>(rv, vp, TILE, S, hd, ld);
  }
  __syncthreads();

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int cur = kt & 1;
    const int kv0 = kt * TILE;
    const __bf16* kb = lds_k + cur * TILE * LDK;
    const __bf16* vb = lds_v + cur * TILE * LDK;
    const __bf16* ktb = lds_kt + cur * HD * TILE;
    const bool edge = (kv0 + TILE - 1 > q_start) || (kv0 + TILE > S);

    float ds_val[4][4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      f32x4 s_acc = {0.f, 0.f, 0.f, 0.f};
      f32x4 dp_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kf = 0; kf < KFRAGS; ++kf) {
        s_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qfrag[kf], lds_frag(kb, n * 16 + col, kf * 32 + kgrp * 8, LDK), s_acc, 0, 0, 0);
        dp_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dofrag[kf], lds_frag(vb, n * 16 + col, kf * 32 + kgrp * 8, LDK), dp_acc, 0, 0, 0);
      }
      const int kv_abs = kv0 + n * 16 + col;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row_abs = q_start + wave * 16 + kgrp * 4 + reg;
        float p;
        if (edge) {
          p = (kv_abs <= row_abs && kv_abs < S && row_abs < S)
                  ? __expf(s_acc[reg] * scale - lse_r[reg]) : 0.f;
        } else {
          p = __expf(s_acc[reg] * scale - lse_r[reg]);
        }
        ds_val[n][reg] = p * (dp_acc[reg] - delta_r[reg]) * scale;
      }
    }
    __builtin_amdgcn_s_setprio(0);

    if (kt + 1 < n_tiles) {  // overlapped staging, after the MFMA cluster
      tile_write_rows<HD, NV>(lds_k + (cur ^ 1) * TILE * LDK, rk, LDK);
      tile_write_rows<HD, NV>(lds_v + (cur ^ 1) * TILE * LDK, rv, LDK);
      tile_write_t<HD, NV>(lds_kt + (cur ^ 1) * HD * TILE, rk);
      if (kt + 2 < n_tiles) {
        tile_load_regs<HD, NV>(rk, kp, (kt + 2) * TILE, S, hd, ld);
        tile_load_regs<HD, NV>(rv, vp, (kt + 2) * TILE, S, hd, ld);
      }
    }

    // redistribute dS -> A layout (intra-wave)
    __bf16* pw = lds_p + wave * 16 * LDT;
#pragma unroll
    for (int n = 0; n < 4; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        pw[lds_rm_idx(kgrp * 4 + reg, n * 16 + col, LDT)] = (__bf16)ds_val[n][reg];

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const bf16x8 a = lds_frag(pw, col, ks * 32 + kgrp * 8, LDT);
#pragma unroll
      for (int t = 0; t < NT_HD; ++t)
        dq_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, ldsT_frag(ktb, t * 16 + col, ks * 32 + kgrp * 8), dq_acc[t], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row_abs = q_start + wave * 16 + kgrp * 4 + reg;
    if (row_abs >= S) continue;
#pragma unroll
    for (int t = 0; t < NT_HD; ++t) {
      const int c = t * 16 + col;
      if (c < hd)
        dq[base + (long)row_abs * ld + c] = __float2bfloat16(dq_acc[t][reg]);
    }
  }
}

// ---------------------------------------------------------------------------
// backward: dK/dV kernel — blocks over 128 kv rows, q tiles of 64
// ---------------------------------------------------------------------------

template <int HD>
__global__ __launch_bounds__(512) void attn_bwd_dkdv_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    __hip_bfloat16* __restrict__ dk, __hip_bfloat16* __restrict__ dv,
    int S, int hd, int nh, long bst, long hst, int ld, float scale) {
  constexpr int KFRAGS = HD / 32;
  constexpr int NT_HD = HD / 16;
  constexpr int LDK = HD + LPAD;
  constexpr int LDT = TILE + LPAD;
  constexpr int NV = (HD + 63) / 64;
  constexpr int NBUF = (HD <= 64) ? 2 : 1;  // hd128: 4 double-buffered tiles
                                            // exceed 160 KiB LDS

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds_q = (__bf16*)smem;                  // [NBUF][TILE][LDK]
  __bf16* lds_do = lds_q + NBUF * TILE * LDK;     // [NBUF][TILE][LDK]
  __bf16* lds_qt = lds_do + NBUF * TILE * LDK;    // [NBUF][HD][TILE] rotated
  __bf16* lds_dot = lds_qt + NBUF * HD * TILE;    // [NBUF][HD][TILE] rotated
  __bf16* lds_p = lds_dot + NBUF * HD * TILE;     // [8][16][LDT]  P^T
  __bf16* lds_p2 = lds_p + 8 * 16 * (TILE + LPAD);  // [8][16][LDT]  dS^T

  const int bh = blockIdx.y;
  const int kv_start_blk = blockIdx.x * 128;
  const long base = (long)(bh / nh) * bst + (long)(bh % nh) * hst;
  const __hip_bfloat16* qp = q + base;
  const __hip_bfloat16* kp = k + base;
  const __hip_bfloat16* vp = v + base;
  const __hip_bfloat16* dop = dout + base;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;
  const int kgrp = lane >> 4;

  const int kvrow_abs = kv_start_blk + wave * 16 + col;
  bf16x8 kfrag[KFRAGS], vfrag[KFRAGS];
#pragma unroll
  for (int kf = 0; kf < KFRAGS; ++kf) {
    kfrag[kf] = global_frag(kp, kvrow_abs, S, hd, ld, kf * 32 + kgrp * 8);
    vfrag[kf] = global_frag(vp, kvrow_abs, S, hd, ld, kf * 32 + kgrp * 8);
  }

  f32x4 dk_acc[NT_HD], dv_acc[NT_HD];
#pragma unroll
  for (int t = 0; t < NT_HD; ++t) {
    dk_acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};
    dv_acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  const int first_qt = kv_start_blk / TILE;  // causal: q >= kv
  const int n_q_tiles = (S + TILE - 1) / TILE;

  bf16x8 rq[NV], rdo[NV];
  tile_load_regs<HD, NV>(rq, qp, first_qt * TILE, S, hd, ld);
  tile_load_regs<HD, NV>(rdo, dop, first_qt * TILE, S, hd, ld);
  tile_write_rows<HD, NV>(lds_q, rq, LDK);
  tile_write_rows<HD, NV>(lds_do, rdo, LDK);
  tile_write_t<HD, NV>(lds_qt, rq);
  tile_write_t<HD, NV>(lds_dot, rdo);
  if (first_qt + 1 < n_q_tiles) {
    tile_load_regs<HD, NV>(rq, qp, (first_qt + 1) * TILE, S, hd, ld);
    tile_load_regs<HD, NV>(rdo, dop, (first_qt + 1) * TILE, S, hd, ld);
  }
  __syncthreads();

  for (int qt = first_qt; qt < n_q_tiles; ++qt) {
    const int cur = (NBUF == 2) ? (qt & 1) : 0;
    const int q_start = qt * TILE;
    const __bf16* qb = lds_q + cur * TILE * LDK;
    const __bf16* dob = lds_do + cur * TILE * LDK;
    const __bf16* qtb = lds_qt + cur * HD * TILE;
    const __bf16* dotb = lds_dot + cur * HD * TILE;
    const bool edge = (q_start < kv_start_blk + 127) || (q_start + TILE > S);

    // T = K Q^T (scores transposed), dPT = V dO^T
    float pt_val[4][4], dst_val[4][4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      f32x4 t_acc = {0.f, 0.f, 0.f, 0.f};
      f32x4 dpt_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kf = 0; kf < KFRAGS; ++kf) {
        t_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            kfrag[kf], lds_frag(qb, n * 16 + col, kf * 32 + kgrp * 8, LDK), t_acc, 0, 0, 0);
        dpt_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            vfrag[kf], lds_frag(dob, n * 16 + col, kf * 32 + kgrp * 8, LDK), dpt_acc, 0, 0, 0);
      }
      const int q_abs = q_start + n * 16 + col;
      const float lse_c = (q_abs < S) ? lse[(long)bh * S + q_abs] : 0.f;
      const float delta_c = (q_abs < S) ? delta[(long)bh * S + q_abs] : 0.f;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float p;
        if (edge) {
          const int kv_abs = kv_start_blk + wave * 16 + kgrp * 4 + reg;
          p = (q_abs >= kv_abs && q_abs < S && kv_abs < S)
                  ? __expf(t_acc[reg] * scale - lse_c) : 0.f;
        } else {
          p = __expf(t_acc[reg] * scale - lse_c);
        }
        pt_val[n][reg] = p;
        dst_val[n][reg] = p * (dpt_acc[reg] - delta_c) * scale;
      }
    }
    __builtin_amdgcn_s_setprio(0);

    if (NBUF == 2 && qt + 1 < n_q_tiles) {  // overlapped staging, mid-compute
      tile_write_rows<HD, NV>(lds_q + (cur ^ 1) * TILE * LDK, rq, LDK);
      tile_write_rows<HD, NV>(lds_do + (cur ^ 1) * TILE * LDK, rdo, LDK);
      tile_write_t<HD, NV>(lds_qt + (cur ^ 1) * HD * TILE, rq);
      tile_write_t<HD, NV>(lds_dot + (cur ^ 1) * HD * TILE, rdo);
      if (qt + 2 < n_q_tiles) {
        tile_load_regs<HD, NV>(rq, qp, (qt + 2) * TILE, S, hd, ld);
        tile_load_regs<HD, NV>(rdo, dop, (qt + 2) * TILE, S, hd, ld);
      }
    }

    // write BOTH relayouts up front into separate per-wave regions, then run
    // both MFMA groups back-to-back: one lgkm boundary and no
    // write-after-read stall between the dV and dK passes
    __bf16* pw = lds_p + wave * 16 * LDT;
    __bf16* pw2 = lds_p2 + wave * 16 * LDT;
#pragma unroll
    for (int n = 0; n < 4; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        pw[lds_rm_idx(kgrp * 4 + reg, n * 16 + col, LDT)] = (__bf16)pt_val[n][reg];
        pw2[lds_rm_idx(kgrp * 4 + reg, n * 16 + col, LDT)] = (__bf16)dst_val[n][reg];
      }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const bf16x8 a = lds_frag(pw, col, ks * 32 + kgrp * 8, LDT);
      const bf16x8 a2 = lds_frag(pw2, col, ks * 32 + kgrp * 8, LDT);
#pragma unroll
      for (int t = 0; t < NT_HD; ++t) {
        dv_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, ldsT_frag(dotb, t * 16 + col, ks * 32 + kgrp * 8), dv_acc[t], 0, 0, 0);
        dk_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a2, ldsT_frag(qtb, t * 16 + col, ks * 32 + kgrp * 8), dk_acc[t], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();

    if (NBUF == 1 && qt + 1 < n_q_tiles) {
      tile_write_rows<HD, NV>(lds_q, rq, LDK);
      tile_write_rows<HD, NV>(lds_do, rdo, LDK);
      tile_write_t<HD, NV>(lds_qt, rq);
      tile_write_t<HD, NV>(lds_dot, rdo);
      if (qt + 2 < n_q_tiles) {
        tile_load_regs<HD, NV>(rq, qp, (qt + 2) * TILE, S, hd, ld);
        tile_load_regs<HD, NV>(rdo, dop, (qt + 2) * TILE, S, hd, ld);
      }
      __syncthreads();
    }
  }

#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row_abs = kv_start_blk + wave * 16 + kgrp * 4 + reg;
    if (row_abs >= S) continue;
#pragma unroll
    for (int t = 0; t < NT_HD; ++t) {
      const int c = t * 16 + col;
      if (c < hd) {
        dk[base + (long)row_abs * ld + c] = __float2bfloat16(dk_acc[t][reg]);
        dv[base + (long)row_abs * ld + c] = __float2bfloat16(dv_acc[t][reg]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

static int pad32(int hd) { return (hd + 31) / 32 * 32; }

#define DISPATCH_HD(HDP, ...)                                     \
  do {                                                            \
    switch (HDP) {                                                \
      case 32: { constexpr int HD = 32; __VA_ARGS__; break; }     \
      case 64: { constexpr int HD = 64; __VA_ARGS__; break; }     \
      case 96: { constexpr int HD = 96; __VA_ARGS__; break; }     \
      case 128: { constexpr int HD = 128; __VA_ARGS__; break; }   \
      default: TORCH_CHECK(false, "head_dim > 128 not supported"); \
    }                                                             \
  } while (0)

// RELORA_AMD_ATTN_FWD=2 falls back to the v2 (16x16 C-layout) forward
static int attn_fwd_version() {
  static int v = [] {
    const char* e = getenv("RELORA_AMD_ATTN_FWD");
    return (e && e[0] == '2') ? 2 : 3;
  }();
  return v;
}


// Accept [B,nh,S,hd] tensors whose memory is either BHSD-contiguous or a
// transposed view of a BSHD-contiguous buffer (what
// `x.view(B,S,nh,hd).transpose(1,2)` yields — lets the model skip every
// permute+contiguous copy around attention).  Returns {bst, hst, ld}.
static std::array<long, 3> attn_strides(const torch::Tensor& q) {
  const long B = q.size(0), nh = q.size(1), S = q.size(2), hd = q.size(3);
  TORCH_CHECK(q.stride(3) == 1, "attention: head_dim must be innermost");
  const bool bhsd = q.is_contiguous();
  const bool bshd = (q.stride(1) == hd && q.stride(2) == nh * hd &&
                     q.stride(0) == S * nh * hd);
  TORCH_CHECK(bhsd || bshd, "attention: unsupported layout");
  return {q.stride(0), q.stride(1), q.stride(2)};
}

static bool same_strides(const torch::Tensor& a, const torch::Tensor& b) {
  return a.strides() == b.strides();
}

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    double scale) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4);
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16, "attn kernels are bf16-only");
  const int B = q.size(0), nh = q.size(1), S = q.size(2), hd = q.size(3);
  const int HDP = pad32(hd);
  const auto st = attn_strides(q);
  TORCH_CHECK(same_strides(q, k) && same_strides(q, v),
              "q/k/v must share a layout");
  const long bst = st[0], hst = st[1];
  const int ld = (int)st[2];
  auto out = torch::empty_strided(q.sizes(), q.strides(), q.options());
  auto lse = torch::empty({B, nh, S}, q.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  dim3 block(512);
  if (attn_fwd_version() == 3) {
    DISPATCH_HD(HDP, {
      dim3 grid((S + 255) / 256, B * nh);
      const int LDK = HD + LPAD;
      size_t smem = (2 * TILE * LDK + 2 * HD * TILE) * sizeof(__bf16);
      static const bool occ4 = [] {
        const char* e = getenv("RELORA_AMD_ATTN_OCC4");
        return !(e && e[0] == '0');  // default ON (measured +15% at hd64)
      }();
      static const bool trrf = [] {
        const char* e = getenv("RELORA_AMD_ATTN_TR_FWD");
        return !(e && e[0] == '0');  // default ON: fwd PV tr-read measured +12%
      }();
      if (occ4 && HD <= 64 && trrf) {
        hipLaunchKernelGGL((attn_fwd_v3_kernel<HD, 4, true>), grid, block, smem, stream,
                           (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                           (const __hip_bfloat16*)v.data_ptr(), (__hip_bfloat16*)out.data_ptr(),
                           lse.data_ptr<float>(), S, hd, nh, bst, hst, ld, (float)scale);
      } else if (occ4 && HD <= 64) {
        hipLaunchKernelGGL((attn_fwd_v3_kernel<HD, 4>), grid, block, smem, stream,
                           (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                           (const __hip_bfloat16*)v.data_ptr(), (__hip_bfloat16*)out.data_ptr(),
                           lse.data_ptr<float>(), S, hd, nh, bst, hst, ld, (float)scale);
      } else {
        hipLaunchKernelGGL((attn_fwd_v3_kernel<HD>), grid, block, smem, stream,
                           (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                           (const __hip_bfloat16*)v.data_ptr(), (__hip_bfloat16*)out.data_ptr(),
                           lse.data_ptr<float>(), S, hd, nh, bst, hst, ld, (float)scale);
      }
    });
    HIP_CHECK_LAST();
    return {out, lse};
  }
  DISPATCH_HD(HDP, {
    constexpr int RF = (HD <= 64) ? 1 : 2;
    dim3 grid((S + 128 * RF - 1) / (128 * RF), B * nh);
    const int LDK = HD + LPAD, LDP = TILE + LPAD;
    size_t smem = (2 * TILE * LDK + 2 * HD * TILE + RF * 8 * 16 * LDP) * sizeof(__bf16);
    hipLaunchKernelGGL((attn_fwd_kernel<HD>), grid, block, smem, stream,
                       (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                       (const __hip_bfloat16*)v.data_ptr(), (__hip_bfloat16*)out.data_ptr(),
                       lse.data_ptr<float>(), S, hd, nh, bst, hst, ld, (float)scale);
  });
  HIP_CHECK_LAST();
  return {out, lse};
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse, torch::Tensor dout,
                                    double scale) {
  const int B = q.size(0), nh = q.size(1), S = q.size(2), hd = q.size(3);
  const int HDP = pad32(hd);
  const auto st = attn_strides(q);
  const long bst = st[0], hst = st[1];
  const int ld = (int)st[2];
  // every tensor the kernels touch must share q's layout; copy dout into
  // it if the incoming grad doesn't (o always does — attn_fwd allocated it)
  TORCH_CHECK(same_strides(q, k) && same_strides(q, v) && same_strides(q, o));
  if (!(dout.stride(3) == 1 && same_strides(q, dout))) {
    auto d2 = torch::empty_strided(q.sizes(), q.strides(), q.options());
    d2.copy_(dout);
    dout = d2;
  }
  auto dq = torch::empty_strided(q.sizes(), q.strides(), q.options());
  auto dk = torch::empty_strided(q.sizes(), q.strides(), q.options());
  auto dv = torch::empty_strided(q.sizes(), q.strides(), q.options());
  auto delta = torch::empty({B, nh, S}, q.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();

  const long n_rows = (long)B * nh * S;
  hipLaunchKernelGGL(attn_delta_kernel, dim3((n_rows + 31) / 32), dim3(256), 0, stream,
                     (const __hip_bfloat16*)dout.data_ptr(), (const __hip_bfloat16*)o.data_ptr(),
                     delta.data_ptr<float>(), n_rows, hd, nh, S, bst, hst, ld);
  HIP_CHECK_LAST();

  // RELORA_AMD_ATTN_BWD=2 falls back to the v2 (16x16 C-layout) backward
  static int bwd_ver = [] {
    const char* e = getenv("RELORA_AMD_ATTN_BWD");
    return (e && e[0] == '2') ? 2 : 3;
  }();

  dim3 block(512);
  DISPATCH_HD(HDP, {
    const int LDK = HD + LPAD, LDT = TILE + LPAD;
    if (bwd_ver == 3 && HD <= 96) {  // dq v3 at hd128 hits the VGPR cap
      size_t smem_dq3 = (2 * TILE * LDK * 2 + 2 * HD * TILE) * sizeof(__bf16);
      static const bool dq4w = [] {
        // 4-wave blocks: 158-VGPR kernel fits 3 blocks/CU (vs 2 waves/SIMD)
        const char* e = getenv("RELORA_AMD_DQ_4WAVE");
        return e && e[0] == '1';
      }();
      static const bool dqocc4 = [] {
        const char* e = getenv("RELORA_AMD_DQ_OCC4");
        return e && e[0] == '1';
      }();
      if (dqocc4 && HD <= 64)
        hipLaunchKernelGGL((attn_bwd_dq_v3_kernel<HD, 8, 4>), dim3((S + 255) / 256, B * nh),
                           block, smem_dq3, stream,
                           (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                           (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                           lse.data_ptr<float>(), delta.data_ptr<float>(),
                           (__hip_bfloat16*)dq.data_ptr(), S, hd, nh, bst, hst, ld,
                           (float)scale);
      else if (dq4w && HD <= 64)
        hipLaunchKernelGGL((attn_bwd_dq_v3_kernel<HD, 4>), dim3((S + 127) / 128, B * nh),
                           dim3(256), smem_dq3, stream,
                           (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                           (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                           lse.data_ptr<float>(), delta.data_ptr<float>(),
                           (__hip_bfloat16*)dq.data_ptr(), S, hd, nh, bst, hst, ld,
                           (float)scale);
      else
        hipLaunchKernelGGL((attn_bwd_dq_v3_kernel<HD>), dim3((S + 255) / 256, B * nh), block,
                           smem_dq3, stream,
                           (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                           (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                           lse.data_ptr<float>(), delta.data_ptr<float>(),
                           (__hip_bfloat16*)dq.data_ptr(), S, hd, nh, bst, hst, ld,
                           (float)scale);
      HIP_CHECK_LAST();
    } else {
      size_t smem_dq = (2 * TILE * LDK * 2 + 2 * HD * TILE + 8 * 16 * LDT) * sizeof(__bf16);
      hipLaunchKernelGGL((attn_bwd_dq_kernel<HD>), dim3((S + 127) / 128, B * nh), block,
                         smem_dq, stream,
                         (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                         (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                         lse.data_ptr<float>(), delta.data_ptr<float>(),
                         (__hip_bfloat16*)dq.data_ptr(), S, hd, nh, bst, hst, ld,
                         (float)scale);
      HIP_CHECK_LAST();
    }
    if (bwd_ver == 3 && HD <= 64) {
      static const bool trr = [] {
        // default OFF: dkdv tr-read measured a wash (155.0 vs 157.1 TF/s) —
        // the per-fragment lgkmcnt(0) drains offset the staging savings at
        // 2 waves/SIMD; fwd at 4 waves/SIMD hides them (+12%)
        const char* e = getenv("RELORA_AMD_ATTN_TR_DKDV");
        return e && e[0] == '1';
      }();
      size_t smem_dkdv3 = (2 * TILE * LDK * 2 + 2 * HD * TILE * 2) * sizeof(__bf16);
      if (trr)
        hipLaunchKernelGGL((attn_bwd_dkdv_v3_kernel<HD, true>), dim3((S + 255) / 256, B * nh),
                           block, smem_dkdv3, stream,
                           (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                           (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                           lse.data_ptr<float>(), delta.data_ptr<float>(),
                           (__hip_bfloat16*)dk.data_ptr(), (__hip_bfloat16*)dv.data_ptr(),
                           S, hd, nh, bst, hst, ld, (float)scale);
      else
        hipLaunchKernelGGL((attn_bwd_dkdv_v3_kernel<HD>), dim3((S + 255) / 256, B * nh), block,
                           smem_dkdv3, stream,
                           (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                           (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                           lse.data_ptr<float>(), delta.data_ptr<float>(),
                           (__hip_bfloat16*)dk.data_ptr(), (__hip_bfloat16*)dv.data_ptr(),
                           S, hd, nh, bst, hst, ld, (float)scale);
      HIP_CHECK_LAST();
    } else {
      constexpr int NBUF = (HD <= 64) ? 2 : 1;
      size_t smem_dkdv =
          (NBUF * TILE * LDK * 2 + NBUF * HD * TILE * 2 + 2 * 8 * 16 * LDT) * sizeof(__bf16);
      hipLaunchKernelGGL((attn_bwd_dkdv_kernel<HD>), dim3((S + 127) / 128, B * nh), block,
                         smem_dkdv, stream,
                         (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                         (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                         lse.data_ptr<float>(), delta.data_ptr<float>(),
                         (__hip_bfloat16*)dk.data_ptr(), (__hip_bfloat16*)dv.data_ptr(),
                         S, hd, nh, bst, hst, ld, (float)scale);
      HIP_CHECK_LAST();
    }
  });
  return {dq, dk, dv};
}
