#include "hip/hip_runtime.h"
// RoPE apply (K4), rotate-half convention with fp32 cos/sin tables
// (reference modeling_llama.py:126-141; partial rotary for pythia
// modeling_pythia.py:184-197). One kernel handles q and k; INVERSE=true
// computes the backward rotation (dq = dy*cos + rot_inv(dy*sin)).
//
// Layout: x [B, nh, S, hd] — BHSD-contiguous or a transposed view of a
// BSHD buffer (`x.view(B,S,nh,hd).transpose(1,2)`); the strided form lets
// the whole q/k path from the QKV projection through attention run with
// ZERO permute/contiguous copies.  cos/sin [S_cache, R] fp32 with
// duplicated halves (cos[i] == cos[i + R/2]); R <= hd, pass-through tail.

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

template <typename T, bool INVERSE>
__global__ void rope_kernel(const T* __restrict__ x, T* __restrict__ y,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t,
                            int S, int hd, int R, long total_rows,
                            int nh, long bst, long hst, long ld) {
  // one thread per (row, i) pair with i < R/2; rows = B*nh*S
  const int half = R / 2;
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long npairs = total_rows * half;
  if (idx >= npairs) return;
  const long row = idx / half;
  const int i = idx % half;
  const long s = row % S;  // position within sequence
  const long h = (row / S) % nh;
  const long b = row / ((long)S * nh);
  const long off = b * bst + h * hst + s * ld;

  const float c = cos_t[s * R + i];
  const float sn = sin_t[s * R + i];
  const T* xr = x + off;
  T* yr = y + off;
  const float x1 = to_f32(xr[i]);
  const float x2 = to_f32(xr[i + half]);
  if (INVERSE) {
    yr[i] = from_f32<T>(x1 * c + x2 * sn);
    yr[i + half] = from_f32<T>(x2 * c - x1 * sn);
  } else {
    yr[i] = from_f32<T>(x1 * c - x2 * sn);
    yr[i + half] = from_f32<T>(x2 * c + x1 * sn);
  }
}

template <typename T>
__global__ void copy_tail_kernel(const T* __restrict__ x, T* __restrict__ y,
                                 int hd, int R, long total_rows,
                                 int S, int nh, long bst, long hst, long ld) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const int tail = hd - R;
  const long n = total_rows * tail;
  if (idx >= n) return;
  const long row = idx / tail;
  const int i = R + idx % tail;
  const long s = row % S;
  const long h = (row / S) % nh;
  const long b = row / ((long)S * nh);
  const long off = b * bst + h * hst + s * ld;
  y[off + i] = x[off + i];
}

template <typename T>
static void rope_launch(const torch::Tensor& x, torch::Tensor& y,
                        const torch::Tensor& cos_t, const torch::Tensor& sin_t,
                        bool inverse, hipStream_t stream) {
  const int nh = x.size(1);
  const int S = x.size(2);
  const int hd = x.size(3);
  const int R = cos_t.size(1);
  const long bst = x.stride(0), hst = x.stride(1), ld = x.stride(2);
  const long rows = (long)x.size(0) * nh * S;
  const long npairs = rows * (R / 2);
  dim3 block(256);
  dim3 grid((npairs + 255) / 256);
  if (inverse)
    hipLaunchKernelGGL((rope_kernel<T, true>), grid, block, 0, stream,
                       (const T*)x.data_ptr(), (T*)y.data_ptr(),
                       cos_t.data_ptr<float>(), sin_t.data_ptr<float>(), S, hd, R,
                       rows, nh, bst, hst, ld);
  else
    hipLaunchKernelGGL((rope_kernel<T, false>), grid, block, 0, stream,
                       (const T*)x.data_ptr(), (T*)y.data_ptr(),
                       cos_t.data_ptr<float>(), sin_t.data_ptr<float>(), S, hd, R,
                       rows, nh, bst, hst, ld);
  if (R < hd) {
    const long n = rows * (hd - R);
    hipLaunchKernelGGL(copy_tail_kernel<T>, dim3((n + 255) / 256), block, 0, stream,
                       (const T*)x.data_ptr(), (T*)y.data_ptr(), hd, R, rows,
                       S, nh, bst, hst, ld);
  }
}

static bool rope_layout_ok(const torch::Tensor& q) {
  if (q.stride(3) != 1) return false;
  const long nh = q.size(1), S = q.size(2), hd = q.size(3);
  return q.is_contiguous() ||
         (q.stride(1) == hd && q.stride(2) == nh * hd && q.stride(0) == S * nh * hd);
}

std::vector<torch::Tensor> rope_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor cos_t, torch::Tensor sin_t,
                                    bool inverse) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4);
  TORCH_CHECK(rope_layout_ok(q), "rope: unsupported q layout");
  TORCH_CHECK(k.sizes() == q.sizes() && k.strides() == q.strides(),
              "rope: k must share q's layout");
  TORCH_CHECK(cos_t.scalar_type() == torch::kFloat32 && cos_t.is_contiguous());
  TORCH_CHECK(cos_t.size(0) >= q.size(2), "rope cache shorter than sequence");
  auto qo = torch::empty_strided(q.sizes(), q.strides(), q.options());
  auto ko = torch::empty_strided(k.sizes(), k.strides(), k.options());
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (q.scalar_type() == torch::kBFloat16) {
    rope_launch<__hip_bfloat16>(q, qo, cos_t, sin_t, inverse, stream);
    rope_launch<__hip_bfloat16>(k, ko, cos_t, sin_t, inverse, stream);
  } else {
    rope_launch<float>(q, qo, cos_t, sin_t, inverse, stream);
    rope_launch<float>(k, ko, cos_t, sin_t, inverse, stream);
  }
  HIP_CHECK_LAST();
  return {qo, ko};
}
