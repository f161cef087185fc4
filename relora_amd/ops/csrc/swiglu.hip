#include "hip/hip_runtime.h"
// SwiGLU (K7): y = silu(gate) * up, fused elementwise fwd/bwd
// (reference modeling_llama.py:157-158). Vectorized 8-wide with scalar tail
// (intermediate sizes like 5461 are odd).

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

DEV_INLINE float silu_f(float x) { return x / (1.f + __expf(-x)); }

template <typename T>
__global__ void swiglu_fwd_kernel(const T* __restrict__ g, const T* __restrict__ u,
                                  T* __restrict__ y, long n) {
  const long vec_n = n / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < vec_n;
       i += (long)gridDim.x * blockDim.x) {
    Vec8<T> gv = load8(g + i * 8);
    Vec8<T> uv = load8(u + i * 8);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o.v[j] = from_f32<T>(silu_f(to_f32(gv.v[j])) * to_f32(uv.v[j]));
    store8(y + i * 8, o);
  }
  // tail
  long start = vec_n * 8;
  for (long i = start + (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    y[i] = from_f32<T>(silu_f(to_f32(g[i])) * to_f32(u[i]));
}

template <typename T>
__global__ void swiglu_bwd_kernel(const T* __restrict__ g, const T* __restrict__ u,
                                  const T* __restrict__ dy, T* __restrict__ dg,
                                  T* __restrict__ du, long n) {
  const long vec_n = n / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < vec_n;
       i += (long)gridDim.x * blockDim.x) {
    Vec8<T> gv = load8(g + i * 8);
    Vec8<T> uv = load8(u + i * 8);
    Vec8<T> dv = load8(dy + i * 8);
    Vec8<T> og, ou;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = to_f32(gv.v[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      float si = gf * sig;
      float d = to_f32(dv.v[j]);
      og.v[j] = from_f32<T>(d * to_f32(uv.v[j]) * sig * (1.f + gf * (1.f - sig)));
      ou.v[j] = from_f32<T>(d * si);
    }
    store8(dg + i * 8, og);
    store8(du + i * 8, ou);
  }
  long start = vec_n * 8;
  for (long i = start + (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float gf = to_f32(g[i]);
    float sig = 1.f / (1.f + __expf(-gf));
    float d = to_f32(dy[i]);
    dg[i] = from_f32<T>(d * to_f32(u[i]) * sig * (1.f + gf * (1.f - sig)));
    du[i] = from_f32<T>(d * gf * sig);
  }
}

// GELU (K8, pythia MLP — reference modeling_pythia.py:395-406):
// exact erf form matching F.gelu's default
DEV_INLINE float gelu_f(float x) {
  return 0.5f * x * (1.f + erff(x * 0.70710678118654752f));
}
DEV_INLINE float gelu_grad_f(float x) {
  const float cdf = 0.5f * (1.f + erff(x * 0.70710678118654752f));
  const float pdf = 0.3989422804014327f * __expf(-0.5f * x * x);
  return cdf + x * pdf;
}

template <typename T>
__global__ void gelu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y, long n) {
  const long vec_n = n / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < vec_n;
       i += (long)gridDim.x * blockDim.x) {
    Vec8<T> xv = load8(x + i * 8);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o.v[j] = from_f32<T>(gelu_f(to_f32(xv.v[j])));
    store8(y + i * 8, o);
  }
  long start = vec_n * 8;
  for (long i = start + (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    y[i] = from_f32<T>(gelu_f(to_f32(x[i])));
}

template <typename T>
__global__ void gelu_bwd_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                                T* __restrict__ dx, long n) {
  const long vec_n = n / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < vec_n;
       i += (long)gridDim.x * blockDim.x) {
    Vec8<T> xv = load8(x + i * 8);
    Vec8<T> dv = load8(dy + i * 8);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o.v[j] = from_f32<T>(to_f32(dv.v[j]) * gelu_grad_f(to_f32(xv.v[j])));
    store8(dx + i * 8, o);
  }
  long start = vec_n * 8;
  for (long i = start + (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    dx[i] = from_f32<T>(to_f32(dy[i]) * gelu_grad_f(to_f32(x[i])));
}

static dim3 ew_grid(long n) {
  long blocks = (n / 8 + 255) / 256;
  if (blocks < 1) blocks = 1;
  if (blocks > 2048) blocks = 2048;  // grid-stride the rest (guide G11)
  return dim3(blocks);
}

torch::Tensor swiglu_fwd(torch::Tensor g, torch::Tensor u) {
  TORCH_CHECK(g.is_cuda() && g.is_contiguous() && u.is_contiguous());
  auto y = torch::empty_like(g);
  long n = g.numel();
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (g.scalar_type() == torch::kBFloat16)
    hipLaunchKernelGGL(swiglu_fwd_kernel<__hip_bfloat16>, ew_grid(n), dim3(256), 0, stream,
                       (const __hip_bfloat16*)g.data_ptr(), (const __hip_bfloat16*)u.data_ptr(),
                       (__hip_bfloat16*)y.data_ptr(), n);
  else
    hipLaunchKernelGGL(swiglu_fwd_kernel<float>, ew_grid(n), dim3(256), 0, stream,
                       g.data_ptr<float>(), u.data_ptr<float>(), y.data_ptr<float>(), n);
  HIP_CHECK_LAST();
  return y;
}

std::vector<torch::Tensor> swiglu_bwd(torch::Tensor g, torch::Tensor u, torch::Tensor dy) {
  auto dg = torch::empty_like(g);
  auto du = torch::empty_like(u);
  long n = g.numel();
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (g.scalar_type() == torch::kBFloat16)
    hipLaunchKernelGGL(swiglu_bwd_kernel<__hip_bfloat16>, ew_grid(n), dim3(256), 0, stream,
                       (const __hip_bfloat16*)g.data_ptr(), (const __hip_bfloat16*)u.data_ptr(),
                       (const __hip_bfloat16*)dy.data_ptr(), (__hip_bfloat16*)dg.data_ptr(),
                       (__hip_bfloat16*)du.data_ptr(), n);
  else
    hipLaunchKernelGGL(swiglu_bwd_kernel<float>, ew_grid(n), dim3(256), 0, stream,
                       g.data_ptr<float>(), u.data_ptr<float>(), dy.data_ptr<float>(),
                       dg.data_ptr<float>(), du.data_ptr<float>(), n);
  HIP_CHECK_LAST();
  return {dg, du};
}

torch::Tensor gelu_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  auto y = torch::empty_like(x);
  const long n = x.numel();
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (x.scalar_type() == torch::kBFloat16)
    hipLaunchKernelGGL(gelu_fwd_kernel<__hip_bfloat16>, ew_grid(n), dim3(256), 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), (__hip_bfloat16*)y.data_ptr(), n);
  else
    hipLaunchKernelGGL(gelu_fwd_kernel<float>, ew_grid(n), dim3(256), 0, stream,
                       (const float*)x.data_ptr(), (float*)y.data_ptr(), n);
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor gelu_bwd(torch::Tensor x, torch::Tensor dy) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
  auto dx = torch::empty_like(x);
  const long n = x.numel();
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (x.scalar_type() == torch::kBFloat16)
    hipLaunchKernelGGL(gelu_bwd_kernel<__hip_bfloat16>, ew_grid(n), dim3(256), 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)dy.data_ptr(),
                       (__hip_bfloat16*)dx.data_ptr(), n);
  else
    hipLaunchKernelGGL(gelu_bwd_kernel<float>, ew_grid(n), dim3(256), 0, stream,
                       (const float*)x.data_ptr(), (const float*)dy.data_ptr(),
                       (float*)dx.data_ptr(), n);
  HIP_CHECK_LAST();
  return dx;
}
