#include "hip/hip_runtime.h"
// Causal flash attention (K3) for gfx950 — MFMA 16x16x32 bf16, LDS-tiled.
//
// Replaces the reference's SDPA call (reference modeling_llama.py:222-224,
// modeling_pythia.py:264-288): causal-only, no padding mask, dropout_p=0 —
// exactly the training configuration the reference uses.
//
// Structure (forward): block = 4 waves = 64 q rows (16 per wave), KV tiles
// of 32. K is staged row-major in LDS ([kv][hd], read as contiguous-k B
// fragments), V is staged transposed ([hd][kv]). Online softmax runs fully
// in registers on the MFMA C-layout (row r of a 16x16 tile lives in the 16
// lanes with l>>4 == r>>2 at register r&3; row reductions are 4 shfl_xor
// steps over the low 4 lane bits). P is redistributed to A-fragment layout
// through a small per-wave LDS buffer. head_dim <= 128, any S; head dims
// that are not multiples of 32 (e.g. 48) are zero-padded in the K-dim.
//
// Backward: standard FlashAttention-2 split — a delta preprocess
// (rowsum(dO*O)), a dQ kernel (blocks over q tiles) and a dK/dV kernel
// (blocks over kv tiles), each recomputing P from the saved LSE.

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define LPAD 8  // bf16 elements of row padding in LDS tiles (16B)

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

// stage a [rows x cols_pad] tile row-major into LDS from global [S, hd],
// zero-padding rows >= S and cols >= hd. cols_pad is a multiple of 8.
DEV_INLINE void stage_rows(__bf16* dst, const __hip_bfloat16* src, int row0,
                           int S, int hd, int rows, int cols_pad, int ldst) {
  const int total = rows * cols_pad / 8;
  for (int t = threadIdx.x; t < total; t += blockDim.x) {
    const int r = t / (cols_pad / 8);
    const int c = (t % (cols_pad / 8)) * 8;
    __bf16* d = dst + r * ldst + c;
    const int gr = row0 + r;
    if (gr < S && c + 8 <= hd) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(src + (long)gr * hd + c);
      *reinterpret_cast<bf16x8*>(d) = v;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        d[j] = (gr < S && c + j < hd) ? (__bf16)src[(long)gr * hd + c + j] : (__bf16)0.f;
    }
  }
}

// stage a transposed tile: dst[hd][rows] from global [S, hd]
DEV_INLINE void stage_rows_t(__bf16* dst, const __hip_bfloat16* src, int row0,
                             int S, int hd, int rows, int ldst) {
  const int total = rows * hd;
  for (int t = threadIdx.x; t < total; t += blockDim.x) {
    const int r = t / hd;   // kv/q row
    const int c = t % hd;   // feature
    const int gr = row0 + r;
    dst[c * ldst + r] = (gr < S) ? (__bf16)src[(long)gr * hd + c] : (__bf16)0.f;
  }
}

// load an A/B fragment from an LDS tile: lane reads row `row`, 8 elements
// at column k0. Caller guarantees 16B alignment (ldst multiple of 8).
DEV_INLINE bf16x8 lds_frag(const __bf16* tile, int row, int k0, int ldst) {
  return *reinterpret_cast<const bf16x8*>(tile + row * ldst + k0);
}

// load an A fragment directly from global [S, hd]: lane row `grow`,
// 8 k-elements at k0; zero-pad outside.
DEV_INLINE bf16x8 global_frag(const __hip_bfloat16* src, int grow, int S, int hd, int k0) {
  bf16x8 v;
  if (grow < S && k0 + 8 <= hd) {
    v = *reinterpret_cast<const bf16x8*>(src + (long)grow * hd + k0);
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      v[j] = (grow < S && k0 + j < hd) ? (__bf16)src[(long)grow * hd + k0 + j] : (__bf16)0.f;
  }
  return v;
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

template <int HD>  // padded head dim (multiple of 32), actual hd passed in
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ out,
    float* __restrict__ lse, int S, int hd, float scale) {
  constexpr int KFRAGS = HD / 32;   // QK^T k-steps
  constexpr int NT_HD = HD / 16;    // PV hd tiles
  constexpr int LDK = HD + LPAD;    // lds_k row stride (elements)
  constexpr int LDV = 32 + LPAD;    // lds_vt row stride
  constexpr int LDP = 32 + LPAD;    // lds_p row stride

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds_k = (__bf16*)smem;                       // [32][LDK]
  __bf16* lds_vt = lds_k + 32 * LDK;                   // [HD][LDV]
  __bf16* lds_p = lds_vt + HD * LDV;                   // [4][16][LDP]

  const int bh = blockIdx.y;
  const int q_start = blockIdx.x * 64;
  const long base = (long)bh * S * hd;
  const __hip_bfloat16* qp = q + base;
  const __hip_bfloat16* kp = k + base;
  const __hip_bfloat16* vp = v + base;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;        // C-frag column / B-frag n / A-frag row
  const int kgrp = lane >> 4;       // k-element group (x8)

  // Q fragments for this wave's 16 rows (A-layout)
  const int qrow_local = wave * 16 + col;
  const int qrow_abs = q_start + qrow_local;
  bf16x8 qfrag[KFRAGS];
#pragma unroll
  for (int kf = 0; kf < KFRAGS; ++kf)
    qfrag[kf] = global_frag(qp, qrow_abs, S, hd, kf * 32 + kgrp * 8);

  // online softmax state: 4 rows per lane (rows kgrp*4 + reg of the wave tile)
  float m_run[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
  float l_run[4] = {0.f, 0.f, 0.f, 0.f};
  f32x4 o_acc[NT_HD];
#pragma unroll
  for (int t = 0; t < NT_HD; ++t) o_acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int q_max_abs = min(q_start + 63, S - 1);
  const int n_kv_tiles = (q_max_abs / 32) + 1;  // causal bound

  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int kv_start = kt * 32;
    stage_rows(lds_k, kp, kv_start, S, hd, 32, HD, LDK);
    stage_rows_t(lds_vt, vp, kv_start, S, hd, 32, LDV);
    __syncthreads();

    // scores: 2 N-subtiles of 16 kv cols
    float p_val[2][4];
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kf = 0; kf < KFRAGS; ++kf) {
        bf16x8 b = lds_frag(lds_k, n * 16 + col, kf * 32 + kgrp * 8, LDK);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kf], b, acc, 0, 0, 0);
      }
      const int kv_abs = kv_start + n * 16 + col;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row_abs = q_start + wave * 16 + kgrp * 4 + reg;
        float s = acc[reg] * scale;
        if (kv_abs > row_abs || kv_abs >= S) s = -INFINITY;
        p_val[n][reg] = s;
      }
    }

    // online softmax per row (4 regs per lane)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      float rmax = fmaxf(p_val[0][reg], p_val[1][reg]);
      rmax = rowgroup_max(rmax);
      float m_new = fmaxf(m_run[reg], rmax);
      float alpha = (m_new == -INFINITY) ? 1.f : __expf(m_run[reg] - m_new);
      float p0 = (m_new == -INFINITY) ? 0.f : __expf(p_val[0][reg] - m_new);
      float p1 = (m_new == -INFINITY) ? 0.f : __expf(p_val[1][reg] - m_new);
      p_val[0][reg] = p0;
      p_val[1][reg] = p1;
      float rsum = rowgroup_sum(p0 + p1);
      l_run[reg] = l_run[reg] * alpha + rsum;
      m_run[reg] = m_new;
#pragma unroll
      for (int t = 0; t < NT_HD; ++t) o_acc[t][reg] *= alpha;
    }

    // redistribute P (C-layout) -> A-layout via per-wave LDS
    __bf16* pw = lds_p + wave * 16 * LDP;
#pragma unroll
    for (int n = 0; n < 2; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        pw[(kgrp * 4 + reg) * LDP + n * 16 + col] = (__bf16)p_val[n][reg];
    __syncthreads();  // also protects lds_k/vt before restage

    // PV: one K=32 step over the kv tile
    bf16x8 a = lds_frag(pw, col, kgrp * 8, LDP);
#pragma unroll
    for (int t = 0; t < NT_HD; ++t) {
      bf16x8 b = lds_frag(lds_vt, t * 16 + col, kgrp * 8, LDV);
      o_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, o_acc[t], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: O = o_acc / l, LSE = m + log(l)
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row_abs = q_start + wave * 16 + kgrp * 4 + reg;
    if (row_abs >= S) continue;
    const float inv_l = (l_run[reg] > 0.f) ? 1.f / l_run[reg] : 0.f;
#pragma unroll
    for (int t = 0; t < NT_HD; ++t) {
      const int c = t * 16 + col;
      if (c < hd)
        out[base + (long)row_abs * hd + c] = __float2bfloat16(o_acc[t][reg] * inv_l);
    }
    if (col == 0)
      lse[(long)bh * S + row_abs] = m_run[reg] + __logf(fmaxf(l_run[reg], 1e-30f));
  }
}

// ---------------------------------------------------------------------------
// backward: delta preprocess
// ---------------------------------------------------------------------------

__global__ void attn_delta_kernel(const __hip_bfloat16* __restrict__ dout,
                                  const __hip_bfloat16* __restrict__ o,
                                  float* __restrict__ delta, int hd) {
  __shared__ float scratch[16];
  const long row = blockIdx.x;
  float acc = 0.f;
  for (int i = threadIdx.x; i < hd; i += blockDim.x)
    acc += to_f32(dout[row * hd + i]) * to_f32(o[row * hd + i]);
  acc = block_reduce_sum(acc, scratch);
  if (threadIdx.x == 0) delta[row] = acc;
}

// ---------------------------------------------------------------------------
// backward: dQ kernel — blocks over q tiles of 64 rows
// ---------------------------------------------------------------------------

template <int HD>
__global__ __launch_bounds__(256) void attn_bwd_dq_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    __hip_bfloat16* __restrict__ dq, int S, int hd, float scale) {
  constexpr int KFRAGS = HD / 32;
  constexpr int NT_HD = HD / 16;
  constexpr int LDK = HD + LPAD;
  constexpr int LDT = 32 + LPAD;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds_k = (__bf16*)smem;            // [32][LDK]  K rows
  __bf16* lds_v = lds_k + 32 * LDK;         // [32][LDK]  V rows
  __bf16* lds_kt = lds_v + 32 * LDK;        // [HD][LDT]  K transposed
  __bf16* lds_p = lds_kt + HD * LDT;        // [4][16][LDT]

  const int bh = blockIdx.y;
  const int q_start = blockIdx.x * 64;
  const long base = (long)bh * S * hd;
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
    qfrag[kf] = global_frag(qp, qrow_abs, S, hd, kf * 32 + kgrp * 8);
    dofrag[kf] = global_frag(dop, qrow_abs, S, hd, kf * 32 + kgrp * 8);
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

  const int q_max_abs = min(q_start + 63, S - 1);
  const int n_kv_tiles = (q_max_abs / 32) + 1;

  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int kv_start = kt * 32;
    stage_rows(lds_k, kp, kv_start, S, hd, 32, HD, LDK);
    stage_rows(lds_v, vp, kv_start, S, hd, 32, HD, LDK);
    stage_rows_t(lds_kt, kp, kv_start, S, hd, 32, LDT);
    __syncthreads();

    float ds_val[2][4];
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      f32x4 s_acc = {0.f, 0.f, 0.f, 0.f};
      f32x4 dp_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kf = 0; kf < KFRAGS; ++kf) {
        bf16x8 bk = lds_frag(lds_k, n * 16 + col, kf * 32 + kgrp * 8, LDK);
        bf16x8 bv = lds_frag(lds_v, n * 16 + col, kf * 32 + kgrp * 8, LDK);
        s_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kf], bk, s_acc, 0, 0, 0);
        dp_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dofrag[kf], bv, dp_acc, 0, 0, 0);
      }
      const int kv_abs = kv_start + n * 16 + col;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row_abs = q_start + wave * 16 + kgrp * 4 + reg;
        float p = 0.f;
        if (kv_abs <= row_abs && kv_abs < S && row_abs < S)
          p = __expf(s_acc[reg] * scale - lse_r[reg]);
        ds_val[n][reg] = p * (dp_acc[reg] - delta_r[reg]) * scale;
      }
    }

    // redistribute dS -> A layout
    __bf16* pw = lds_p + wave * 16 * LDT;
#pragma unroll
    for (int n = 0; n < 2; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        pw[(kgrp * 4 + reg) * LDT + n * 16 + col] = (__bf16)ds_val[n][reg];
    __syncthreads();

    bf16x8 a = lds_frag(pw, col, kgrp * 8, LDT);
#pragma unroll
    for (int t = 0; t < NT_HD; ++t) {
      bf16x8 b = lds_frag(lds_kt, t * 16 + col, kgrp * 8, LDT);
      dq_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, dq_acc[t], 0, 0, 0);
    }
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
        dq[base + (long)row_abs * hd + c] = __float2bfloat16(dq_acc[t][reg]);
    }
  }
}

// ---------------------------------------------------------------------------
// backward: dK/dV kernel — blocks over kv tiles of 64 rows
// ---------------------------------------------------------------------------

template <int HD>
__global__ __launch_bounds__(256) void attn_bwd_dkdv_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    __hip_bfloat16* __restrict__ dk, __hip_bfloat16* __restrict__ dv,
    int S, int hd, float scale) {
  constexpr int KFRAGS = HD / 32;
  constexpr int NT_HD = HD / 16;
  constexpr int LDK = HD + LPAD;
  constexpr int LDT = 32 + LPAD;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* lds_q = (__bf16*)smem;            // [32][LDK]  Q rows
  __bf16* lds_do = lds_q + 32 * LDK;        // [32][LDK]  dO rows
  __bf16* lds_qt = lds_do + 32 * LDK;       // [HD][LDT]  Q transposed
  __bf16* lds_dot = lds_qt + HD * LDT;      // [HD][LDT]  dO transposed
  __bf16* lds_p = lds_dot + HD * LDT;       // [4][16][LDT]

  const int bh = blockIdx.y;
  const int kv_start_blk = blockIdx.x * 64;
  const long base = (long)bh * S * hd;
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
    kfrag[kf] = global_frag(kp, kvrow_abs, S, hd, kf * 32 + kgrp * 8);
    vfrag[kf] = global_frag(vp, kvrow_abs, S, hd, kf * 32 + kgrp * 8);
  }

  f32x4 dk_acc[NT_HD], dv_acc[NT_HD];
#pragma unroll
  for (int t = 0; t < NT_HD; ++t) {
    dk_acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};
    dv_acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  const int first_qt = kv_start_blk / 32;  // causal: q >= kv
  const int n_q_tiles = (S + 31) / 32;

  for (int qt = first_qt; qt < n_q_tiles; ++qt) {
    const int q_start = qt * 32;
    stage_rows(lds_q, qp, q_start, S, hd, 32, HD, LDK);
    stage_rows(lds_do, dop, q_start, S, hd, 32, HD, LDK);
    stage_rows_t(lds_qt, qp, q_start, S, hd, 32, LDT);
    stage_rows_t(lds_dot, dop, q_start, S, hd, 32, LDT);
    __syncthreads();

    // T = K Q^T (scores transposed), dPT = V dO^T
    float pt_val[2][4], dst_val[2][4];
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      f32x4 t_acc = {0.f, 0.f, 0.f, 0.f};
      f32x4 dpt_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kf = 0; kf < KFRAGS; ++kf) {
        bf16x8 bq = lds_frag(lds_q, n * 16 + col, kf * 32 + kgrp * 8, LDK);
        bf16x8 bdo = lds_frag(lds_do, n * 16 + col, kf * 32 + kgrp * 8, LDK);
        t_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kfrag[kf], bq, t_acc, 0, 0, 0);
        dpt_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag[kf], bdo, dpt_acc, 0, 0, 0);
      }
      const int q_abs = q_start + n * 16 + col;
      const float lse_c = (q_abs < S) ? lse[(long)bh * S + q_abs] : 0.f;
      const float delta_c = (q_abs < S) ? delta[(long)bh * S + q_abs] : 0.f;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int kv_abs = kv_start_blk + wave * 16 + kgrp * 4 + reg;
        float p = 0.f;
        if (q_abs >= kv_abs && q_abs < S && kv_abs < S)
          p = __expf(t_acc[reg] * scale - lse_c);
        pt_val[n][reg] = p;
        dst_val[n][reg] = p * (dpt_acc[reg] - delta_c) * scale;
      }
    }

    __bf16* pw = lds_p + wave * 16 * LDT;
    // pass 1: dV += P^T @ dO
#pragma unroll
    for (int n = 0; n < 2; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        pw[(kgrp * 4 + reg) * LDT + n * 16 + col] = (__bf16)pt_val[n][reg];
    __syncthreads();
    {
      bf16x8 a = lds_frag(pw, col, kgrp * 8, LDT);
#pragma unroll
      for (int t = 0; t < NT_HD; ++t) {
        bf16x8 b = lds_frag(lds_dot, t * 16 + col, kgrp * 8, LDT);
        dv_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, dv_acc[t], 0, 0, 0);
      }
    }
    __syncthreads();
    // pass 2: dK += dS^T @ Q
#pragma unroll
    for (int n = 0; n < 2; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        pw[(kgrp * 4 + reg) * LDT + n * 16 + col] = (__bf16)dst_val[n][reg];
    __syncthreads();
    {
      bf16x8 a = lds_frag(pw, col, kgrp * 8, LDT);
#pragma unroll
      for (int t = 0; t < NT_HD; ++t) {
        bf16x8 b = lds_frag(lds_qt, t * 16 + col, kgrp * 8, LDT);
        dk_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, dk_acc[t], 0, 0, 0);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row_abs = kv_start_blk + wave * 16 + kgrp * 4 + reg;
    if (row_abs >= S) continue;
#pragma unroll
    for (int t = 0; t < NT_HD; ++t) {
      const int c = t * 16 + col;
      if (c < hd) {
        dk[base + (long)row_abs * hd + c] = __float2bfloat16(dk_acc[t][reg]);
        dv[base + (long)row_abs * hd + c] = __float2bfloat16(dv_acc[t][reg]);
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

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    double scale) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16, "attn kernels are bf16-only");
  const int B = q.size(0), nh = q.size(1), S = q.size(2), hd = q.size(3);
  const int HDP = pad32(hd);
  auto out = torch::empty_like(q);
  auto lse = torch::empty({B, nh, S}, q.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  dim3 grid((S + 63) / 64, B * nh), block(256);
  DISPATCH_HD(HDP, {
    const int LDK = HD + LPAD, LDV = 32 + LPAD, LDP = 32 + LPAD;
    size_t smem = (32 * LDK + HD * LDV + 4 * 16 * LDP) * sizeof(__bf16);
    hipLaunchKernelGGL((attn_fwd_kernel<HD>), grid, block, smem, stream,
                       (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                       (const __hip_bfloat16*)v.data_ptr(), (__hip_bfloat16*)out.data_ptr(),
                       lse.data_ptr<float>(), S, hd, (float)scale);
  });
  HIP_CHECK_LAST();
  return {out, lse};
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse, torch::Tensor dout,
                                    double scale) {
  const int B = q.size(0), nh = q.size(1), S = q.size(2), hd = q.size(3);
  const int HDP = pad32(hd);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto delta = torch::empty({B, nh, S}, q.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  dout = dout.contiguous();

  hipLaunchKernelGGL(attn_delta_kernel, dim3((long)B * nh * S), dim3(64), 0, stream,
                     (const __hip_bfloat16*)dout.data_ptr(), (const __hip_bfloat16*)o.data_ptr(),
                     delta.data_ptr<float>(), hd);
  HIP_CHECK_LAST();

  dim3 block(256);
  DISPATCH_HD(HDP, {
    const int LDK = HD + LPAD, LDT = 32 + LPAD;
    size_t smem_dq = (2 * 32 * LDK + HD * LDT + 4 * 16 * LDT) * sizeof(__bf16);
    hipLaunchKernelGGL((attn_bwd_dq_kernel<HD>), dim3((S + 63) / 64, B * nh), block,
                       smem_dq, stream,
                       (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                       (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (__hip_bfloat16*)dq.data_ptr(), S, hd, (float)scale);
    HIP_CHECK_LAST();
    size_t smem_dkdv = (2 * 32 * LDK + 2 * HD * LDT + 4 * 16 * LDT) * sizeof(__bf16);
    hipLaunchKernelGGL((attn_bwd_dkdv_kernel<HD>), dim3((S + 63) / 64, B * nh), block,
                       smem_dkdv, stream,
                       (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                       (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (__hip_bfloat16*)dk.data_ptr(), (__hip_bfloat16*)dv.data_ptr(),
                       S, hd, (float)scale);
    HIP_CHECK_LAST();
  });
  return {dq, dk, dv};
}
