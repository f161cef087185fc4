#include "hip/hip_runtime.h"
// Fused LoRA rank-r update kernels (K1+K2) for gfx950.
//
// The reference computes the LoRA path as three separate torch ops plus an
// add (reference relora.py:319-323): dropout(x), lora_A GEMM, lora_B GEMM,
// scale-mul, add.  Here the rank-r (<=256) reduction runs as one MFMA
// kernel that accumulates STRAIGHT INTO the main GEMM's output, with the
// dropout mask generated once (philox4x32-10, packed bits) and re-applied
// in backward — removing three [M,N]-sized memory round trips per wrapped
// Linear per direction:
//
//   fwd:  y[M,N]  += t[M,r] @ Bs[N,r]^T          (lora_add, TRANSQ=false)
//   bwd:  dx[M,K] += mask/(1-p) * (u[M,r] @ A[r,K])   (TRANSQ=true, MASK)
//
// Tile: 128x128 out per 256-thread block (4 waves, 64x64 per wave),
// mfma_f32_16x16x32_bf16, whole r staged in LDS once (no K loop over
// tiles: r <= 256).  Operand tiles are staged row-major with a 16-byte row
// pad (the attention kernels' proven layout) and read as contiguous-k
// bf16x8 fragments.

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// Default: 8-element row padding (staging writes conflict-free, fragment
// reads 2-way per the bank model).  RELORA_AMD_ROT_V2: zero padding + a
// per-row XOR swizzle on row-major tiles — conflict-free on every modeled
// pattern for r % 64 == 0 (the flagship r=128 included); see
// tools/lds_bank_model.py and tests/test_lds_bank_model.py.
#ifdef RELORA_AMD_ROT_V2
#define LPAD 0
#else
#define LPAD 8  // bf16 elements of LDS row padding (one 16B slot)
#endif

DEV_INLINE int rm_swz(int row, int ldst) {
#ifdef RELORA_AMD_ROT_V2
  return ((ldst & 127) == 0 ? (row & 15) : (row & 7)) << 3;
#else
  (void)row; (void)ldst;
  return 0;
#endif
}
// swizzled element index into a row-major [R][ldst] LDS tile — every
// producer and consumer must address through this
DEV_INLINE int rm_idx(int row, int col, int ldst) {
  return (row * ldst + col) ^ rm_swz(row, ldst);
}

// Rotated layout for transposed LDS tiles with 64-element rows: element
// (row, c) of an [R][64] image lives at row*64 + rot8(row, c), with the
// rotation verified by the in-tree bank model (tools/lds_bank_model.py):
// writes and reads cap at 2-way with zero padding (the naive layout was
// 8-way — 44% of skinny_grad wave cycles).  RELORA_AMD_ROT_V2 switches to
// the Q_V2 table (conflict-free reads, writes stay at the 2-way floor);
// numerics-neutral — writer and reader share the one bijection.
#ifdef RELORA_AMD_ROT_V2
__device__ constexpr unsigned char Q_V2_L[16][8] = {
    {4, 2, 0, 3, 6, 1, 7, 5}, {6, 0, 3, 4, 1, 5, 2, 7},
    {0, 6, 5, 1, 7, 3, 4, 2}, {0, 6, 1, 7, 5, 2, 3, 4},
    {1, 5, 2, 7, 6, 4, 0, 3}, {2, 5, 3, 0, 1, 4, 6, 7},
    {4, 6, 5, 3, 0, 1, 7, 2}, {1, 4, 2, 6, 7, 3, 5, 0},
    {3, 1, 7, 4, 5, 2, 6, 0}, {7, 3, 0, 5, 6, 2, 1, 4},
    {5, 7, 4, 6, 2, 0, 3, 1}, {3, 1, 6, 4, 0, 7, 2, 5},
    {2, 0, 1, 6, 3, 7, 5, 4}, {7, 4, 2, 5, 6, 3, 1, 0},
    {3, 7, 2, 0, 5, 4, 6, 1}, {2, 5, 7, 1, 0, 4, 6, 3}};
DEV_INLINE int rot8(int row, int c64) {
  return (((Q_V2_L[row & 15][c64 >> 3] + 2 * (row >> 4)) & 7) << 3) + (c64 & 7);
}
#else
DEV_INLINE int rot8(int row, int c64) {
  return ((((c64 >> 3) + (row >> 3) + (row & 7)) & 7) << 3) + (c64 & 7);
}
#endif
DEV_INLINE int tr64(int row, int c) { return row * 64 + rot8(row, c); }

// In-row offset for a TRANSQ (transposed) tile element at column k of a row
// with r k-elements.  Default: 64-element blocks with the 8-deep rot8
// rotation.  Under RELORA_AMD_ROT_V2 with r a multiple of 128 (256-byte
// rows), an 8-deep rotation is pigeonhole-bound to 2-way fragment reads,
// so use the closed-form 16-deep whole-row permutation instead (verified
// conflict-free reads + floor writes by the in-tree bank model).  The
// writer and reader share this one mapping, so the choice is
// numerics-neutral.
#ifdef RELORA_AMD_ROT_V2
__device__ constexpr unsigned char X16_TRQ[16] = {
    0, 2, 4, 6, 1, 3, 5, 7, 9, 11, 13, 15, 8, 10, 12, 14};
DEV_INLINE int trq_off(int row, int k, int r) {
  if ((r & 127) == 0) {
    const int grp = (k & 127) >> 3;
    const int p = (X16_TRQ[row & 15] + ((grp & 1) << 3) + (grp >> 1) +
                   2 * (row >> 4)) & 15;
    return (k & ~127) + p * 8 + (k & 7);
  }
  return (k & ~63) + rot8(row, k & 63);
}
#else
DEV_INLINE int trq_off(int row, int k, int r) {
  (void)r;
  return (k & ~63) + rot8(row, k & 63);
}
#endif


// ---------------------------------------------------------------------------
// philox4x32-10 — counter-based RNG for the dropout mask (regenerable, but we
// persist packed bits: exact replay in backward with zero recompute).
// ---------------------------------------------------------------------------

DEV_INLINE void philox_round(uint32_t& c0, uint32_t& c1, uint32_t& c2, uint32_t& c3,
                             uint32_t k0, uint32_t k1) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  uint32_t h0 = __umulhi(M0, c0), l0 = M0 * c0;
  uint32_t h1 = __umulhi(M1, c2), l1 = M1 * c2;
  uint32_t n0 = h1 ^ c1 ^ k0, n1 = l1, n2 = h0 ^ c3 ^ k1, n3 = l0;
  c0 = n0; c1 = n1; c2 = n2; c3 = n3;
}

DEV_INLINE void philox4(uint64_t seed, uint64_t idx, uint32_t out[4]) {
  uint32_t c0 = (uint32_t)idx, c1 = (uint32_t)(idx >> 32), c2 = 0x9E3779B9u, c3 = 0xBB67AE85u;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    philox_round(c0, c1, c2, c3, k0, k1);
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
}

// xd = keep ? x/(1-p) : 0; mask bit j of byte [m][k8] = keep(k8*8+j).
// One thread per 8 consecutive elements (one mask byte, one bf16x8 store).
template <typename T>
__global__ void dropout_mask_kernel(const T* __restrict__ x, T* __restrict__ xd,
                                    uint8_t* __restrict__ mask, long n8, long n,
                                    uint64_t seed, float p, float inv_keep) {
  const long i8 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i8 >= n8) return;
  uint32_t r[4];
  philox4(seed, (uint64_t)i8, r);
  const uint32_t thr = (uint32_t)(p * 65536.0f);
  const uint32_t u16s[8] = {r[0] & 0xFFFFu, r[0] >> 16, r[1] & 0xFFFFu, r[1] >> 16,
                            r[2] & 0xFFFFu, r[2] >> 16, r[3] & 0xFFFFu, r[3] >> 16};
  if (i8 * 8 + 8 <= n) {
    Vec8<T> v = load8(x + i8 * 8);
    Vec8<T> o;
    uint8_t m = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const bool keep = u16s[j] >= thr;
      m |= (uint8_t)keep << j;
      o.v[j] = keep ? from_f32<T>(to_f32(v.v[j]) * inv_keep) : from_f32<T>(0.f);
    }
    store8(xd + i8 * 8, o);
    mask[i8] = m;
  } else {  // tail: < 8 elements
    uint8_t m = 0;
    for (int j = 0; j < 8 && i8 * 8 + j < n; ++j) {
      const bool keep = u16s[j] >= thr;
      m |= (uint8_t)keep << j;
      xd[i8 * 8 + j] = keep ? from_f32<T>(to_f32(x[i8 * 8 + j]) * inv_keep)
                            : from_f32<T>(0.f);
    }
    mask[i8] = m;
  }
}

// backward of the packed-mask dropout: dx = dy * mask * inv_keep
template <typename T>
__global__ void mask_apply_kernel(const T* __restrict__ dy, T* __restrict__ dx,
                                  const uint8_t* __restrict__ mask, long n8,
                                  long n, float inv_keep) {
  const long i8 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i8 >= n8) return;
  const uint8_t m = mask[i8];
  if (i8 * 8 + 8 <= n) {
    Vec8<T> v = load8(dy + i8 * 8);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o.v[j] = ((m >> j) & 1) ? from_f32<T>(to_f32(v.v[j]) * inv_keep)
                              : from_f32<T>(0.f);
    store8(dx + i8 * 8, o);
  } else {
    for (int j = 0; j < 8 && i8 * 8 + j < n; ++j)
      dx[i8 * 8 + j] = ((m >> j) & 1)
                           ? from_f32<T>(to_f32(dy[i8 * 8 + j]) * inv_keep)
                           : from_f32<T>(0.f);
  }
}

// ---------------------------------------------------------------------------
// rank-r accumulate kernel
// out[M,N] (+)= P[M,r] @ Q^T          (TRANSQ=false: Q is [N,r] row-major)
// out[M,N] (+)= maskscale * (P[M,r] @ Q)   (TRANSQ=true: Q is [r,N] row-major)
// ---------------------------------------------------------------------------

template <bool TRANSQ, bool MASK>
__global__ __launch_bounds__(256) void lora_skinny_kernel(
    const __hip_bfloat16* __restrict__ P, const __hip_bfloat16* __restrict__ Q,
    __hip_bfloat16* __restrict__ out, const uint8_t* __restrict__ mask,
    float inv_keep, long M, int N, int r, int ldq) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int ldt = r + LPAD;
  __bf16* p_im = (__bf16*)smem;            // [128][ldt]
  __bf16* q_im = p_im + 128 * ldt;         // [128][ldt]

  const long m0 = (long)blockIdx.y * 128;
  const int n0 = blockIdx.x * 128;

  // stage P rows [128][r]
  for (int t = threadIdx.x; t < 128 * (r / 8); t += blockDim.x) {
    const int row = t / (r / 8);
    const int c = (t % (r / 8)) * 8;
    bf16x8 v;
    if (m0 + row < M) {
      v = *reinterpret_cast<const bf16x8*>(P + (m0 + row) * (long)r + c);
    } else {
      v = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
    *reinterpret_cast<bf16x8*>(p_im + rm_idx(row, c, ldt)) = v;
  }
  // stage Q tile as q_im[n][k]
  if (!TRANSQ) {
    for (int t = threadIdx.x; t < 128 * (r / 8); t += blockDim.x) {
      const int n = t / (r / 8);
      const int c = (t % (r / 8)) * 8;
      bf16x8 v;
      if (n0 + n < N) {
        v = *reinterpret_cast<const bf16x8*>(Q + (n0 + n) * (long)ldq + c);
      } else {
        v = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
      *reinterpret_cast<bf16x8*>(q_im + rm_idx(n, c, ldt)) = v;
    }
  } else {
    // Q is [r][N]: q_im[n][k] = Q[k][n0+n].  bf16x8 loads along n (contiguous
    // in global), 8 rotated scalar LDS writes each (conflict-free, see tr64)
    for (int t = threadIdx.x; t < r * 16; t += blockDim.x) {
      const int k = t / 16;
      const int nb = (t % 16) * 8;
      bf16x8 v;
      if (n0 + nb + 8 <= N) {
        v = *reinterpret_cast<const bf16x8*>(Q + (long)k * ldq + n0 + nb);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[j] = (n0 + nb + j < N) ? (__bf16)Q[(long)k * ldq + n0 + nb + j] : (__bf16)0.f;
      }
#pragma unroll
      for (int j = 0; j < 8; ++j)
        q_im[(nb + j) * ldt + trq_off(nb + j, k, r)] = v[j];
    }
  }
  __syncthreads();

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = (wave >> 1) * 64;  // wave row offset in tile
  const int wc = (wave & 1) * 64;   // wave col offset
  const int fr = lane & 15;         // fragment row/col
  const int kg = (lane >> 4) * 8;   // k-group offset

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int kk = 0; kk < r; kk += 32) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const bf16x8 a =
          *reinterpret_cast<const bf16x8*>(p_im + rm_idx(wr + mi * 16 + fr, kk + kg, ldt));
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int qrow = wc + ni * 16 + fr;
        const int koff = kk + kg;
        const bf16x8 b = *reinterpret_cast<const bf16x8*>(
            TRANSQ ? q_im + qrow * ldt + trq_off(qrow, koff, r)
                   : q_im + rm_idx(qrow, koff, ldt));
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[mi][ni], 0, 0, 0);
      }
    }
  }

  // epilogue in two phases: dump the C-layout accumulators into an LDS
  // [128][136] tile (cheap strided b16 writes), then vectorized bf16x8
  // global read-modify-writes — the per-element C-layout RMW was 64 scalar
  // global round trips per thread
  __syncthreads();  // done with p_im/q_im; reuse the LDS as the out tile
  __bf16* o_im = (__bf16*)smem;  // [128][OLD]
  constexpr int OLD = 128 + LPAD;
  const int crow = (lane >> 4) * 4;  // C-frag rows crow..crow+3, col = fr
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        o_im[rm_idx(wr + mi * 16 + crow + j, wc + ni * 16 + fr, OLD)] =
            (__bf16)acc[mi][ni][j];
  __syncthreads();

  for (int t = threadIdx.x; t < 128 * 16; t += blockDim.x) {
    const int row = t / 16;
    const int c8 = (t % 16) * 8;
    const long m = m0 + row;
    const int n = n0 + c8;
    if (m >= M || n >= N) continue;
    const __bf16* src_v = o_im + rm_idx(row, c8, OLD);
    const long flat = m * (long)N + n;  // mask bits are FLAT-packed over [M*N]
    __hip_bfloat16* o = out + flat;
    if (n + 8 <= N && (flat & 7) == 0) {
      Vec8<__hip_bfloat16> ov = load8(o);
      uint8_t mb = 0xFF;
      if (MASK) mb = mask[flat >> 3];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = (float)src_v[j];
        if (MASK) v = (mb >> j) & 1 ? v * inv_keep : 0.f;
        ov.v[j] = from_f32<__hip_bfloat16>(to_f32(ov.v[j]) + v);
      }
      store8(o, ov);
    } else {  // unaligned rows (odd N) or the row tail
      for (int j = 0; j < 8 && n + j < N; ++j) {
        float v = (float)src_v[j];
        if (MASK) {
          const uint8_t mb = mask[(flat + j) >> 3];
          v = (mb >> ((flat + j) & 7)) & 1 ? v * inv_keep : 0.f;
        }
        o[j] = from_f32<__hip_bfloat16>(to_f32(o[j]) + v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> dropout_mask_fwd(torch::Tensor x, double p, int64_t seed) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "dropout_mask_fwd: bf16 only");
  auto xd = torch::empty_like(x);
  const long n = x.numel();
  const long n8 = (n + 7) / 8;  // FLAT packing: mask bit j of byte b = elem 8b+j
  auto mask = torch::empty({n8}, x.options().dtype(torch::kUInt8));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const float inv_keep = 1.f / (1.f - (float)p);
  hipLaunchKernelGGL(dropout_mask_kernel<__hip_bfloat16>,
                     dim3((n8 + 255) / 256), dim3(256), 0, stream,
                     (const __hip_bfloat16*)x.data_ptr(), (__hip_bfloat16*)xd.data_ptr(),
                     mask.data_ptr<uint8_t>(), n8, n, (uint64_t)seed, (float)p, inv_keep);
  HIP_CHECK_LAST();
  return {xd, mask};
}

// out[M,N] += P[M,r] @ Q[N,r]^T  (forward epilogue; Q = scale*lora_B.weight)
void lora_add_nt_(torch::Tensor out, torch::Tensor P, torch::Tensor Q) {
  TORCH_CHECK(out.is_cuda() && out.is_contiguous() && P.is_contiguous() && Q.is_contiguous());
  TORCH_CHECK(out.scalar_type() == torch::kBFloat16, "lora_add: bf16 only");
  const long M = out.size(0);
  const int N = out.size(1);
  const int r = P.size(1);
  TORCH_CHECK(P.size(0) == M && Q.size(0) == N && Q.size(1) == r);
  TORCH_CHECK(r % 32 == 0 && r <= 256, "lora_add: r must be a multiple of 32, <=256");
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const size_t lds = std::max<size_t>(2u * 128 * (r + LPAD), 128u * (128 + LPAD))
                     * sizeof(__bf16);
  dim3 grid((N + 127) / 128, (M + 127) / 128), block(256);
  hipLaunchKernelGGL((lora_skinny_kernel<false, false>), grid, block, lds, stream,
                     (const __hip_bfloat16*)P.data_ptr(), (const __hip_bfloat16*)Q.data_ptr(),
                     (__hip_bfloat16*)out.data_ptr(), nullptr, 1.f, M, N, r, r);
  HIP_CHECK_LAST();
}

// out[M,K] += maskscale * (P[M,r] @ Q[r,K])  (backward dx epilogue; Q = lora_A.weight)
void lora_add_nn_(torch::Tensor out, torch::Tensor P, torch::Tensor Q,
                  torch::Tensor mask, double inv_keep) {
  TORCH_CHECK(out.is_cuda() && out.is_contiguous() && P.is_contiguous() && Q.is_contiguous());
  TORCH_CHECK(out.scalar_type() == torch::kBFloat16, "lora_add: bf16 only");
  const long M = out.size(0);
  const int N = out.size(1);
  const int r = P.size(1);
  TORCH_CHECK(P.size(0) == M && Q.size(0) == r && Q.size(1) == N);
  TORCH_CHECK(r % 32 == 0 && r <= 256, "lora_add: r must be a multiple of 32, <=256");
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const size_t lds = std::max<size_t>(2u * 128 * (r + LPAD), 128u * (128 + LPAD))
                     * sizeof(__bf16);
  dim3 grid((N + 127) / 128, (M + 127) / 128), block(256);
  const bool has_mask = mask.defined() && mask.numel() > 0;
  if (has_mask) {
    TORCH_CHECK(mask.numel() == (M * (long)N + 7) / 8, "mask size mismatch");
    hipLaunchKernelGGL((lora_skinny_kernel<true, true>), grid, block, lds, stream,
                       (const __hip_bfloat16*)P.data_ptr(),
                       (const __hip_bfloat16*)Q.data_ptr(),
                       (__hip_bfloat16*)out.data_ptr(), mask.data_ptr<uint8_t>(),
                       (float)inv_keep, M, N, r, N);
  } else {
    hipLaunchKernelGGL((lora_skinny_kernel<true, false>), grid, block, lds, stream,
                       (const __hip_bfloat16*)P.data_ptr(),
                       (const __hip_bfloat16*)Q.data_ptr(),
                       (__hip_bfloat16*)out.data_ptr(), nullptr, 1.f, M, N, r, N);
  }
  HIP_CHECK_LAST();
}

// ---------------------------------------------------------------------------
// LoRA weight gradients: out[r, C] = P[M, r]^T @ X[M, C], fp32 partials per
// M-chunk (deterministic tree sum in the caller).  hipBLASLt runs these
// skinny-output reductions as 32-workgroup launches (12% of the chip);
// chunking M restores full occupancy.
//   dA          = skinny_grad(P=u_s,  X=xd) -> [r, K]
//   dB^T        = skinny_grad(P=t_u,  X=dy) -> [r, N]
// ---------------------------------------------------------------------------

// grid: (C/128, MCHUNKS, r/128); block 256 (4 waves); out tile [128][128].
// MASK: X is consumed as dropout(X) = maskbit * X * inv_keep, applied while
// staging (so the forward never has to persist the dropped-out activations).
template <bool MASK>
__global__ __launch_bounds__(256) void skinny_grad_kernel(
    const __hip_bfloat16* __restrict__ P, const __hip_bfloat16* __restrict__ X,
    const uint8_t* __restrict__ xmask, float inv_keep,
    float* __restrict__ part, long M, int C, int r, int rows_per_chunk) {
  const int r0 = blockIdx.z * 128;      // r-tile (rank 256 spans two)
  const int rtile = min(r - r0, 128);
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // double-buffered rotated tiles; staging for m-tile t+1 overlaps the MFMA
  // work of tile t (one barrier per m-tile — same scheme as attention.hip)
  __bf16* pt = (__bf16*)smem;        // [2][128][64] rotated: P^T tiles
  __bf16* xt = pt + 2 * 128 * 64;    // [2][128][64] rotated: X^T tiles

  const int c0 = blockIdx.x * 128;
  const long m_begin = (long)blockIdx.y * rows_per_chunk;
  const long m_end = min(M, m_begin + rows_per_chunk);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;
  const int kgrp = lane >> 4;

  // per-wave output rows: wave*32 .. +32 (2 row frags), cols c0..c0+128
  f32x4 acc[2][8];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  auto stage = [&](long m0, int buf) {
    __bf16* ptb = pt + buf * 128 * 64;
    __bf16* xtb = xt + buf * 128 * 64;
    // P^T slice: pt[j][mm] = P[m0+mm][r0+j]
    for (int t = threadIdx.x; t < 64 * (rtile / 8); t += blockDim.x) {
      const int mm = t / (rtile / 8);
      const int j8 = (t % (rtile / 8)) * 8;
      bf16x8 v;
      if (m0 + mm < m_end) {
        v = *reinterpret_cast<const bf16x8*>(P + (m0 + mm) * (long)r + r0 + j8);
      } else {
        v = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) ptb[tr64(j8 + j, mm)] = v[j];
    }
    // X^T: xt[cc][mm] = X[m0+mm][c0+cc]; vec8 over the row direction
    for (int t = threadIdx.x; t < 64 * 16; t += blockDim.x) {
      const int mm = t / 16;
      const int c8 = (t % 16) * 8;
      bf16x8 v;
      const long xoff = (m0 + mm) * (long)C + c0 + c8;
      if (m0 + mm < m_end && c0 + c8 + 8 <= C && (xoff & 7) == 0) {
        v = *reinterpret_cast<const bf16x8*>(X + xoff);
        if (MASK) {
          // flat-packed bits; an aligned vec8 covers bits j..j+7 of bytes
          // (xoff>>3) and possibly (xoff>>3)+1 when xoff&7 != 0 — here
          // (xoff&7)==0 so exactly one byte
          const uint8_t mb = xmask[xoff >> 3];
#pragma unroll
          for (int j = 0; j < 8; ++j)
            v[j] = (mb >> j) & 1 ? (__bf16)((float)v[j] * inv_keep) : (__bf16)0.f;
        }
      } else if (m0 + mm < m_end) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = (c0 + c8 + j < C) ? (float)(__bf16)X[xoff + j] : 0.f;
          if (MASK && c0 + c8 + j < C) {
            const uint8_t mb = xmask[(xoff + j) >> 3];
            f = (mb >> ((xoff + j) & 7)) & 1 ? f * inv_keep : 0.f;
          }
          v[j] = (__bf16)f;
        }
      } else {
        v = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) xtb[tr64(c8 + j, mm)] = v[j];
    }
  };

  stage(m_begin, 0);
  __syncthreads();
  for (long m0 = m_begin; m0 < m_end; m0 += 64) {
    const int buf = (int)(((m0 - m_begin) >> 6) & 1);
    if (m0 + 64 < m_end) stage(m0 + 64, buf ^ 1);
    const __bf16* ptb = pt + buf * 128 * 64;
    const __bf16* xtb = xt + buf * 128 * 64;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int jrow = wave * 32 + i * 16 + col;
        const bf16x8 a = (jrow < rtile)
            ? *reinterpret_cast<const bf16x8*>(ptb + tr64(jrow, ks * 32 + kgrp * 8))
            : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const bf16x8 b = *reinterpret_cast<const bf16x8*>(
              xtb + tr64(j * 16 + col, ks * 32 + kgrp * 8));
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[i][j], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // epilogue: fp32 partial plane for this chunk
  float* plane = part + (long)blockIdx.y * r * C;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int rbase = wave * 32 + i * 16 + (kgrp << 2);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = c0 + j * 16 + col;
      if (c >= C) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int rr = rbase + reg;
        if (rr < rtile) plane[(long)(r0 + rr) * C + c] = acc[i][j][reg];
      }
    }
  }
}

// combine the MCHUNK fp32 partial planes: out = scale * sum_chunks(part),
// emitted in the requested dtype, optionally transposed ([C, r] instead of
// [r, C]) — folds the .sum(0).t().contiguous().mul_(scale).to(bf16) chain
// (5 launches per wrapped Linear) into one kernel.
template <typename T, bool TRANSP>
__global__ void skinny_combine_kernel(const float* __restrict__ part,
                                      T* __restrict__ out, int r, int C,
                                      int nchunks, float scale) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= (long)r * C) return;
  float acc = 0.f;
  for (int c = 0; c < nchunks; ++c) acc += part[(long)c * r * C + i];
  acc *= scale;
  if (TRANSP) {
    const int rr = i / C, cc = i % C;
    out[(long)cc * r + rr] = from_f32<T>(acc);
  } else {
    out[i] = from_f32<T>(acc);
  }
}

// out = scale * (P[M,r]^T @ dropout_mask(X)[M,C]) in `dtype`; [C, r] when
// transpose_out.  `xmask` empty -> X used as-is.
torch::Tensor skinny_grad(torch::Tensor P, torch::Tensor X, torch::Tensor xmask,
                          double inv_keep, double scale,
                          bool transpose_out, torch::ScalarType dtype) {
  TORCH_CHECK(P.is_cuda() && P.is_contiguous() && X.is_contiguous());
  TORCH_CHECK(P.scalar_type() == torch::kBFloat16 && X.scalar_type() == torch::kBFloat16);
  const long M = P.size(0);
  const int r = P.size(1);
  const int C = X.size(1);
  TORCH_CHECK(X.size(0) == M && r % 8 == 0 && r <= 256);
  // chunk M so the grid fills the chip: (C/128)*chunks >= ~512
  int chunks = 1;
  const int ctiles = (C + 127) / 128;
  while (chunks < 32 && ctiles * chunks < 512 && (M + chunks - 1) / chunks > 256)
    chunks <<= 1;
  int rows = (int)((M + chunks - 1) / chunks);
  rows = (rows + 63) / 64 * 64;  // multiple of the m-tile
  chunks = (int)((M + rows - 1) / rows);
  auto part = torch::empty({chunks, r, C}, P.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const size_t lds = (4u * 128 * 64) * sizeof(__bf16);
  dim3 grid(ctiles, chunks, (r + 127) / 128), block(256);
  const bool has_mask = xmask.defined() && xmask.numel() > 0;
  if (has_mask) {
    TORCH_CHECK(xmask.numel() == (M * (long)C + 7) / 8, "skinny_grad mask size");
    hipLaunchKernelGGL((skinny_grad_kernel<true>), grid, block, lds, stream,
                       (const __hip_bfloat16*)P.data_ptr(),
                       (const __hip_bfloat16*)X.data_ptr(),
                       xmask.data_ptr<uint8_t>(), (float)inv_keep,
                       part.data_ptr<float>(), M, C, r, rows);
  } else {
    hipLaunchKernelGGL((skinny_grad_kernel<false>), grid, block, lds, stream,
                       (const __hip_bfloat16*)P.data_ptr(),
                       (const __hip_bfloat16*)X.data_ptr(), nullptr, 1.f,
                       part.data_ptr<float>(), M, C, r, rows);
  }
  HIP_CHECK_LAST();
  auto out = transpose_out
      ? torch::empty({C, r}, P.options().dtype(dtype))
      : torch::empty({r, C}, P.options().dtype(dtype));
  dim3 cgrid(((long)r * C + 255) / 256), cblock(256);
  const float s = (float)scale;
  if (dtype == torch::kBFloat16) {
    if (transpose_out)
      hipLaunchKernelGGL((skinny_combine_kernel<__hip_bfloat16, true>), cgrid, cblock, 0,
                         stream, part.data_ptr<float>(),
                         (__hip_bfloat16*)out.data_ptr(), r, C, chunks, s);
    else
      hipLaunchKernelGGL((skinny_combine_kernel<__hip_bfloat16, false>), cgrid, cblock, 0,
                         stream, part.data_ptr<float>(),
                         (__hip_bfloat16*)out.data_ptr(), r, C, chunks, s);
  } else {
    TORCH_CHECK(dtype == torch::kFloat32, "skinny_grad: bf16 or fp32 output");
    if (transpose_out)
      hipLaunchKernelGGL((skinny_combine_kernel<float, true>), cgrid, cblock, 0, stream,
                         part.data_ptr<float>(), out.data_ptr<float>(), r, C, chunks, s);
    else
      hipLaunchKernelGGL((skinny_combine_kernel<float, false>), cgrid, cblock, 0, stream,
                         part.data_ptr<float>(), out.data_ptr<float>(), r, C, chunks, s);
  }
  HIP_CHECK_LAST();
  return out;
}

torch::Tensor dropout_mask_bwd(torch::Tensor dy, torch::Tensor mask, double p) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16);
  auto dx = torch::empty_like(dy);
  const long n = dy.numel();
  const long n8 = (n + 7) / 8;
  TORCH_CHECK(mask.numel() >= n8, "mask too small");
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const float inv_keep = (float)(1.0 / (1.0 - p));
  hipLaunchKernelGGL(mask_apply_kernel<__hip_bfloat16>,
                     dim3((n8 + 255) / 256), dim3(256), 0, stream,
                     (const __hip_bfloat16*)dy.data_ptr(),
                     (__hip_bfloat16*)dx.data_ptr(),
                     mask.data_ptr<uint8_t>(), n8, n, inv_keep);
  HIP_CHECK_LAST();
  return dx;
}
