#include "hip/hip_runtime.h"
// RMSNorm (K5) and LayerNorm (K6) for gfx950.
// Reference numerics: fp32 variance, product rounded at bf16
// (reference modeling_llama.py:74-91).
//
// fwd: one block per row, vectorized 8-wide loads, single pass.
// bwd: dx one block per row; dw/db one thread per column striding rows
// (coalesced across threads; the re-read of x/dy is HBM-bound and overlaps).

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

// ---------------------------------------------------------------------------
// RMSNorm forward: y = w * (x * rsqrt(mean(x^2)+eps)); saves invrms per row.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void rmsnorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                   T* __restrict__ y, float* __restrict__ invrms,
                                   int H, float eps) {
  __shared__ float scratch[16];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)H;
  T* yr = y + row * (long)H;

  float ss = 0.f;
  const int vec_end = (H / 8) * 8;
  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> v = load8(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32(v.v[j]);
      ss += f * f;
    }
  }
  for (int i = vec_end + threadIdx.x; i < H; i += blockDim.x) {
    float f = to_f32(xr[i]);
    ss += f * f;
  }
  ss = block_reduce_sum(ss, scratch);
  const float ir = rsqrtf(ss / H + eps);
  if (threadIdx.x == 0) invrms[row] = ir;

  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> v = load8(xr + i);
    Vec8<T> wv = load8(w + i);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      // match torch: (x * invrms) rounded to T, then * w rounded to T
      T h = from_f32<T>(to_f32(v.v[j]) * ir);
      o.v[j] = from_f32<T>(to_f32(h) * to_f32(wv.v[j]));
    }
    store8(yr + i, o);
  }
  for (int i = vec_end + threadIdx.x; i < H; i += blockDim.x) {
    T h = from_f32<T>(to_f32(xr[i]) * ir);
    yr[i] = from_f32<T>(to_f32(h) * to_f32(w[i]));
  }
}


// ---------------------------------------------------------------------------
// Fused residual-add + RMSNorm (K16): s = x + res (bf16, same rounding as
// the unfused residual add), y = rmsnorm(s); backward folds the +dsum of
// the residual fork into the dx epilogue — removes one [M,H] add kernel in
// each direction per call.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void rmsnorm_fwd_add_kernel(const T* __restrict__ x, const T* __restrict__ res,
                                       const T* __restrict__ w, T* __restrict__ sum_o,
                                       T* __restrict__ y, float* __restrict__ invrms,
                                       int H, float eps) {
  __shared__ float scratch[16];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)H;
  const T* rr = res + row * (long)H;
  T* sr = sum_o + row * (long)H;
  T* yr = y + row * (long)H;

  float ss = 0.f;
  const int vec_end = (H / 8) * 8;
  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> v = load8(xr + i);
    Vec8<T> rv = load8(rr + i);
    Vec8<T> sv;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sv.v[j] = from_f32<T>(to_f32(v.v[j]) + to_f32(rv.v[j]));
      float f = to_f32(sv.v[j]);
      ss += f * f;
    }
    store8(sr + i, sv);
  }
  for (int i = vec_end + threadIdx.x; i < H; i += blockDim.x) {
    T s = from_f32<T>(to_f32(xr[i]) + to_f32(rr[i]));
    sr[i] = s;
    float f = to_f32(s);
    ss += f * f;
  }
  ss = block_reduce_sum(ss, scratch);
  const float ir = rsqrtf(ss / H + eps);
  if (threadIdx.x == 0) invrms[row] = ir;

  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> v = load8(sr + i);
    Vec8<T> wv = load8(w + i);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      T h = from_f32<T>(to_f32(v.v[j]) * ir);
      o.v[j] = from_f32<T>(to_f32(h) * to_f32(wv.v[j]));
    }
    store8(yr + i, o);
  }
  for (int i = vec_end + threadIdx.x; i < H; i += blockDim.x) {
    T h = from_f32<T>(to_f32(sr[i]) * ir);
    yr[i] = from_f32<T>(to_f32(h) * to_f32(w[i]));
  }
}

// dx = invrms * (g - xhat * mean(g*xhat)) + dsum   (x here = the saved sum)
template <typename T>
__global__ void rmsnorm_bwd_dx_add_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                          const float* __restrict__ invrms,
                                          const T* __restrict__ dy,
                                          const T* __restrict__ dsum,
                                          T* __restrict__ dx, int H) {
  __shared__ float scratch[16];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)H;
  const T* dyr = dy + row * (long)H;
  const T* dsr = dsum + row * (long)H;
  T* dxr = dx + row * (long)H;
  const float ir = invrms[row];

  float dot = 0.f;
  const int vec_end = (H / 8) * 8;
  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> xv = load8(xr + i);
    Vec8<T> dv = load8(dyr + i);
    Vec8<T> wv = load8(w + i);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dot += to_f32(dv.v[j]) * to_f32(wv.v[j]) * to_f32(xv.v[j]) * ir;
  }
  for (int i = vec_end + threadIdx.x; i < H; i += blockDim.x)
    dot += to_f32(dyr[i]) * to_f32(w[i]) * to_f32(xr[i]) * ir;
  dot = block_reduce_sum(dot, scratch) / H;

  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> xv = load8(xr + i);
    Vec8<T> dv = load8(dyr + i);
    Vec8<T> wv = load8(w + i);
    Vec8<T> ds = load8(dsr + i);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = to_f32(dv.v[j]) * to_f32(wv.v[j]);
      float xh = to_f32(xv.v[j]) * ir;
      o.v[j] = from_f32<T>(ir * (g - xh * dot) + to_f32(ds.v[j]));
    }
    store8(dxr + i, o);
  }
  for (int i = vec_end + threadIdx.x; i < H; i += blockDim.x) {
    float g = to_f32(dyr[i]) * to_f32(w[i]);
    float xh = to_f32(xr[i]) * ir;
    dxr[i] = from_f32<T>(ir * (g - xh * dot) + to_f32(dsr[i]));
  }
}

// dx = invrms * (g - xhat * mean(g * xhat)), g = dy*w, xhat = x*invrms
template <typename T>
__global__ void rmsnorm_bwd_dx_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                      const float* __restrict__ invrms,
                                      const T* __restrict__ dy, T* __restrict__ dx,
                                      int H) {
  __shared__ float scratch[16];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)H;
  const T* dyr = dy + row * (long)H;
  T* dxr = dx + row * (long)H;
  const float ir = invrms[row];

  float dot = 0.f;
  const int vec_end = (H / 8) * 8;
  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> xv = load8(xr + i);
    Vec8<T> dv = load8(dyr + i);
    Vec8<T> wv = load8(w + i);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dot += to_f32(dv.v[j]) * to_f32(wv.v[j]) * to_f32(xv.v[j]) * ir;
  }
  for (int i = vec_end + threadIdx.x; i < H; i += blockDim.x)
    dot += to_f32(dyr[i]) * to_f32(w[i]) * to_f32(xr[i]) * ir;
  dot = block_reduce_sum(dot, scratch) / H;  // mean(g * xhat)

  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> xv = load8(xr + i);
    Vec8<T> dv = load8(dyr + i);
    Vec8<T> wv = load8(w + i);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = to_f32(dv.v[j]) * to_f32(wv.v[j]);
      float xh = to_f32(xv.v[j]) * ir;
      o.v[j] = from_f32<T>(ir * (g - xh * dot));
    }
    store8(dxr + i, o);
  }
  for (int i = vec_end + threadIdx.x; i < H; i += blockDim.x) {
    float g = to_f32(dyr[i]) * to_f32(w[i]);
    float xh = to_f32(xr[i]) * ir;
    dxr[i] = from_f32<T>(ir * (g - xh * dot));
  }
}

// dw[col] = sum_rows dy[r][col] * x[r][col] * invrms[r]  (fp32 out)
// Two stages: blockIdx.y row-chunks write fp32 partials (fills the chip —
// the single-stage column loop left 255/256 CUs idle and was 37% of a
// training step), then a small combine kernel sums the chunk axis.
template <typename T>
__global__ void rmsnorm_bwd_dw_partial_kernel(const T* __restrict__ x,
                                              const float* __restrict__ invrms,
                                              const T* __restrict__ dy,
                                              float* __restrict__ dw_part,
                                              long M, int H, int rows_per_chunk) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H) return;
  const long r0 = (long)blockIdx.y * rows_per_chunk;
  const long r1 = min(M, r0 + rows_per_chunk);
  float acc = 0.f;
  for (long r = r0; r < r1; ++r)
    acc += to_f32(dy[r * H + col]) * to_f32(x[r * H + col]) * invrms[r];
  dw_part[(long)blockIdx.y * H + col] = acc;
}

__global__ void col_combine_kernel(const float* __restrict__ part,
                                   float* __restrict__ out, int H, int nchunks,
                                   int nout) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H * nout) return;
  float acc = 0.f;
  for (int c = 0; c < nchunks; ++c) acc += part[(long)c * H * nout + col];
  out[col] = acc;
}

static int reduce_chunks(long M) {
  // enough chunks to fill 256 CUs even when H/256 is small, few enough that
  // the combine stays trivial
  long c = (M + 63) / 64;
  return (int)(c < 1 ? 1 : (c > 256 ? 256 : c));
}

// ---------------------------------------------------------------------------
// LayerNorm
// ---------------------------------------------------------------------------

template <typename T>
__global__ void layernorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                     const T* __restrict__ b, T* __restrict__ y,
                                     float* __restrict__ mean_out,
                                     float* __restrict__ invstd_out, int H, float eps) {
  __shared__ float scratch[16];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)H;
  T* yr = y + row * (long)H;

  float s = 0.f, ss = 0.f;
  const int vec_end = (H / 8) * 8;
  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> v = load8(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32(v.v[j]);
      s += f;
      ss += f * f;
    }
  }
  for (int i = vec_end + threadIdx.x; i < H; i += blockDim.x) {
    float f = to_f32(xr[i]);
    s += f;
    ss += f * f;
  }
  s = block_reduce_sum(s, scratch);
  ss = block_reduce_sum(ss, scratch);
  const float mu = s / H;
  const float var = fmaxf(ss / H - mu * mu, 0.f);
  const float istd = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mu;
    invstd_out[row] = istd;
  }

  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> v = load8(xr + i);
    Vec8<T> wv = load8(w + i);
    Vec8<T> bv = load8(b + i);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xh = (to_f32(v.v[j]) - mu) * istd;
      o.v[j] = from_f32<T>(xh * to_f32(wv.v[j]) + to_f32(bv.v[j]));
    }
    store8(yr + i, o);
  }
  for (int i = vec_end + threadIdx.x; i < H; i += blockDim.x) {
    float xh = (to_f32(xr[i]) - mu) * istd;
    yr[i] = from_f32<T>(xh * to_f32(w[i]) + to_f32(b[i]));
  }
}

// dx = istd * (g - mean(g) - xhat * mean(g * xhat)), g = dy*w
template <typename T>
__global__ void layernorm_bwd_dx_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                        const float* __restrict__ mean,
                                        const float* __restrict__ invstd,
                                        const T* __restrict__ dy, T* __restrict__ dx,
                                        int H) {
  __shared__ float scratch[16];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)H;
  const T* dyr = dy + row * (long)H;
  T* dxr = dx + row * (long)H;
  const float mu = mean[row];
  const float istd = invstd[row];

  float sg = 0.f, sgx = 0.f;
  for (int i = threadIdx.x; i < H; i += blockDim.x) {
    float g = to_f32(dyr[i]) * to_f32(w[i]);
    float xh = (to_f32(xr[i]) - mu) * istd;
    sg += g;
    sgx += g * xh;
  }
  sg = block_reduce_sum(sg, scratch) / H;
  sgx = block_reduce_sum(sgx, scratch) / H;

  for (int i = threadIdx.x; i < H; i += blockDim.x) {
    float g = to_f32(dyr[i]) * to_f32(w[i]);
    float xh = (to_f32(xr[i]) - mu) * istd;
    dxr[i] = from_f32<T>(istd * (g - sg - xh * sgx));
  }
}

// Same two-stage structure as rmsnorm's dw; partials laid out [chunk][2][H]
// (dw plane then db plane) so one col_combine_kernel call with nout=2 sums both.
template <typename T>
__global__ void layernorm_bwd_dwdb_partial_kernel(const T* __restrict__ x,
                                                  const float* __restrict__ mean,
                                                  const float* __restrict__ invstd,
                                                  const T* __restrict__ dy,
                                                  float* __restrict__ part,
                                                  long M, int H, int rows_per_chunk) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H) return;
  const long r0 = (long)blockIdx.y * rows_per_chunk;
  const long r1 = min(M, r0 + rows_per_chunk);
  float accw = 0.f, accb = 0.f;
  for (long r = r0; r < r1; ++r) {
    float d = to_f32(dy[r * H + col]);
    float xh = (to_f32(x[r * H + col]) - mean[r]) * invstd[r];
    accw += d * xh;
    accb += d;
  }
  part[(long)blockIdx.y * 2 * H + col] = accw;
  part[(long)blockIdx.y * 2 * H + H + col] = accb;
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

static int norm_block(int H) {
  if (H <= 512) return 64;
  if (H <= 2048) return 256;
  return 512;
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  const long M = x.size(0);
  const int H = x.size(1);
  auto y = torch::empty_like(x);
  auto invrms = torch::empty({M}, x.options().dtype(torch::kFloat32));
  dim3 grid(M), block(norm_block(H));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(rmsnorm_fwd_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)w.data_ptr(),
                       (__hip_bfloat16*)y.data_ptr(), invrms.data_ptr<float>(), H, (float)eps);
  } else {
    hipLaunchKernelGGL(rmsnorm_fwd_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(), y.data_ptr<float>(),
                       invrms.data_ptr<float>(), H, (float)eps);
  }
  HIP_CHECK_LAST();
  return {y, invrms};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor x, torch::Tensor w,
                                       torch::Tensor invrms, torch::Tensor dy) {
  const long M = x.size(0);
  const int H = x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::empty({H}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  dim3 grid(M), block(norm_block(H));
  const int nchunks = reduce_chunks(M);
  const int rows_per_chunk = (int)((M + nchunks - 1) / nchunks);
  auto part = torch::empty({nchunks, H}, x.options().dtype(torch::kFloat32));
  dim3 gridc((H + 255) / 256, nchunks), blockc(256);
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(rmsnorm_bwd_dx_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)w.data_ptr(),
                       invrms.data_ptr<float>(), (const __hip_bfloat16*)dy.data_ptr(),
                       (__hip_bfloat16*)dx.data_ptr(), H);
    hipLaunchKernelGGL(rmsnorm_bwd_dw_partial_kernel<__hip_bfloat16>, gridc, blockc, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), invrms.data_ptr<float>(),
                       (const __hip_bfloat16*)dy.data_ptr(), part.data_ptr<float>(),
                       M, H, rows_per_chunk);
  } else {
    hipLaunchKernelGGL(rmsnorm_bwd_dx_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(), invrms.data_ptr<float>(),
                       dy.data_ptr<float>(), dx.data_ptr<float>(), H);
    hipLaunchKernelGGL(rmsnorm_bwd_dw_partial_kernel<float>, gridc, blockc, 0, stream,
                       x.data_ptr<float>(), invrms.data_ptr<float>(), dy.data_ptr<float>(),
                       part.data_ptr<float>(), M, H, rows_per_chunk);
  }
  hipLaunchKernelGGL(col_combine_kernel, dim3((H + 255) / 256), blockc, 0, stream,
                     part.data_ptr<float>(), dw.data_ptr<float>(), H, nchunks, 1);
  HIP_CHECK_LAST();
  return {dx, dw};
}

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  const long M = x.size(0);
  const int H = x.size(1);
  auto y = torch::empty_like(x);
  auto mean = torch::empty({M}, x.options().dtype(torch::kFloat32));
  auto invstd = torch::empty({M}, x.options().dtype(torch::kFloat32));
  dim3 grid(M), block(norm_block(H));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(layernorm_fwd_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)w.data_ptr(),
                       (const __hip_bfloat16*)b.data_ptr(), (__hip_bfloat16*)y.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(), H, (float)eps);
  } else {
    hipLaunchKernelGGL(layernorm_fwd_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(), b.data_ptr<float>(),
                       y.data_ptr<float>(), mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       H, (float)eps);
  }
  HIP_CHECK_LAST();
  return {y, mean, invstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor mean, torch::Tensor invstd,
                                         torch::Tensor dy) {
  const long M = x.size(0);
  const int H = x.size(1);
  auto dx = torch::empty_like(x);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  dim3 grid(M), block(norm_block(H));
  const int nchunks = reduce_chunks(M);
  const int rows_per_chunk = (int)((M + nchunks - 1) / nchunks);
  auto part = torch::empty({nchunks, 2, H}, x.options().dtype(torch::kFloat32));
  auto dwdb = torch::empty({2, H}, x.options().dtype(torch::kFloat32));
  dim3 gridc((H + 255) / 256, nchunks), blockc(256);
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(layernorm_bwd_dx_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)w.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       (const __hip_bfloat16*)dy.data_ptr(), (__hip_bfloat16*)dx.data_ptr(), H);
    hipLaunchKernelGGL(layernorm_bwd_dwdb_partial_kernel<__hip_bfloat16>, gridc, blockc, 0,
                       stream, (const __hip_bfloat16*)x.data_ptr(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), (const __hip_bfloat16*)dy.data_ptr(),
                       part.data_ptr<float>(), M, H, rows_per_chunk);
  } else {
    hipLaunchKernelGGL(layernorm_bwd_dx_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), dy.data_ptr<float>(), dx.data_ptr<float>(), H);
    hipLaunchKernelGGL(layernorm_bwd_dwdb_partial_kernel<float>, gridc, blockc, 0, stream,
                       x.data_ptr<float>(), mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       dy.data_ptr<float>(), part.data_ptr<float>(), M, H, rows_per_chunk);
  }
  hipLaunchKernelGGL(col_combine_kernel, dim3((2 * H + 255) / 256), blockc, 0, stream,
                     part.data_ptr<float>(), dwdb.data_ptr<float>(), H, nchunks, 2);
  HIP_CHECK_LAST();
  return {dx, dwdb[0], dwdb[1]};
}

std::vector<torch::Tensor> rmsnorm_fwd_add(torch::Tensor x, torch::Tensor res,
                                           torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous() && res.is_contiguous());
  TORCH_CHECK(x.sizes() == res.sizes());
  const long M = x.size(0);
  const int H = x.size(1);
  auto sum_o = torch::empty_like(x);
  auto y = torch::empty_like(x);
  auto invrms = torch::empty({M}, x.options().dtype(torch::kFloat32));
  dim3 grid(M), block(norm_block(H));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(rmsnorm_fwd_add_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)res.data_ptr(),
                       (const __hip_bfloat16*)w.data_ptr(), (__hip_bfloat16*)sum_o.data_ptr(),
                       (__hip_bfloat16*)y.data_ptr(), invrms.data_ptr<float>(), H, (float)eps);
  } else {
    hipLaunchKernelGGL(rmsnorm_fwd_add_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), res.data_ptr<float>(), w.data_ptr<float>(),
                       sum_o.data_ptr<float>(), y.data_ptr<float>(),
                       invrms.data_ptr<float>(), H, (float)eps);
  }
  HIP_CHECK_LAST();
  return {y, sum_o, invrms};
}

std::vector<torch::Tensor> rmsnorm_bwd_add(torch::Tensor x, torch::Tensor w,
                                           torch::Tensor invrms, torch::Tensor dy,
                                           torch::Tensor dsum) {
  // identical to rmsnorm_bwd but dx += dsum fused; dw reuses the standard path
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(dy.is_contiguous() && dsum.is_contiguous());
  const long M = x.size(0);
  const int H = x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::empty({H}, x.options().dtype(torch::kFloat32));
  dim3 grid(M), block(norm_block(H));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int nchunks = reduce_chunks(M);
  const int rows_per_chunk = (int)((M + nchunks - 1) / nchunks);
  auto part = torch::empty({nchunks, H}, x.options().dtype(torch::kFloat32));
  dim3 gridc((H + 255) / 256, nchunks), blockc(256);
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(rmsnorm_bwd_dx_add_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)w.data_ptr(),
                       invrms.data_ptr<float>(), (const __hip_bfloat16*)dy.data_ptr(),
                       (const __hip_bfloat16*)dsum.data_ptr(),
                       (__hip_bfloat16*)dx.data_ptr(), H);
    hipLaunchKernelGGL(rmsnorm_bwd_dw_partial_kernel<__hip_bfloat16>, gridc, blockc, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), invrms.data_ptr<float>(),
                       (const __hip_bfloat16*)dy.data_ptr(), part.data_ptr<float>(),
                       M, H, rows_per_chunk);
  } else {
    hipLaunchKernelGGL(rmsnorm_bwd_dx_add_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(), invrms.data_ptr<float>(),
                       dy.data_ptr<float>(), dsum.data_ptr<float>(), dx.data_ptr<float>(), H);
    hipLaunchKernelGGL(rmsnorm_bwd_dw_partial_kernel<float>, gridc, blockc, 0, stream,
                       x.data_ptr<float>(), invrms.data_ptr<float>(), dy.data_ptr<float>(),
                       part.data_ptr<float>(), M, H, rows_per_chunk);
  }
  hipLaunchKernelGGL(col_combine_kernel, dim3((H + 255) / 256), blockc, 0, stream,
                     part.data_ptr<float>(), dw.data_ptr<float>(), H, nchunks, 1);
  HIP_CHECK_LAST();
  return {dx, dw};
}
