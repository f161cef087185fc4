#include "hip/hip_runtime.h"
// Fused cross-entropy row kernels (K10): one-pass online max+sumexp row
// stats and the in-place softmax-minus-onehot gradient. The LM-head GEMMs
// run in hipBLASLt (library GEMMs); these kernels remove the 3 extra fp32
// passes over the [chunk, V] logits that torch's logsumexp/softmax would
// cost, and the full [M, V] logits are never materialized (the memory hot
// spot the reference flags at modeling_llama.py:696-697).

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

// per-row online max/sumexp + target-logit gather.
template <typename T>
__global__ void ce_row_stats_kernel(const T* __restrict__ logits,
                                    const long* __restrict__ labels,
                                    float* __restrict__ lse_out,
                                    float* __restrict__ tgt_out,
                                    int V, long ignore_index) {
  __shared__ float scratch[16];
  const long row = blockIdx.x;
  const T* lr = logits + row * (long)V;

  float m = -INFINITY, s = 0.f;
  const int vec_end = (V / 8) * 8;
  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> v = load8(lr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32(v.v[j]);
      if (f > m) {
        s *= __expf(m - f);
        m = f;
      }
      s += __expf(f - m);
    }
  }
  for (int i = vec_end + threadIdx.x; i < V; i += blockDim.x) {
    float f = to_f32(lr[i]);
    if (f > m) {
      s *= __expf(m - f);
      m = f;
    }
    s += __expf(f - m);
  }
  // combine (m, s) across the block: global max then rescaled sums
  float gm = block_reduce_max(m, scratch);
  float gs = block_reduce_sum(s * __expf(m - gm), scratch);

  if (threadIdx.x == 0) {
    lse_out[row] = gm + __logf(gs);
    long lab = labels[row];
    tgt_out[row] = (lab == ignore_index) ? 0.f : to_f32(lr[lab]);
  }
}

// in place: logits[row] <- (softmax(logits[row]) - onehot(label)) * gscale
template <typename T>
__global__ void ce_grad_kernel(T* __restrict__ logits, const long* __restrict__ labels,
                               const float* __restrict__ lse, float gscale,
                               long M, int V, long ignore_index) {
  const long row = blockIdx.x;
  T* lr = logits + row * (long)V;
  const long lab = labels[row];
  const bool valid = lab != ignore_index;
  const float l = lse[row];

  const int vec_end = (V / 8) * 8;
  for (int i = threadIdx.x * 8; i < vec_end; i += blockDim.x * 8) {
    Vec8<T> v = load8(lr + i);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float p = valid ? __expf(to_f32(v.v[j]) - l) : 0.f;
      if (valid && (long)(i + j) == lab) p -= 1.f;
      o.v[j] = from_f32<T>(p * gscale);
    }
    store8(lr + i, o);
  }
  for (int i = vec_end + threadIdx.x; i < V; i += blockDim.x) {
    float p = valid ? __expf(to_f32(lr[i]) - l) : 0.f;
    if (valid && (long)i == lab) p -= 1.f;
    lr[i] = from_f32<T>(p * gscale);
  }
}

std::vector<torch::Tensor> ce_row_stats(torch::Tensor logits, torch::Tensor labels,
                                        long ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(labels.scalar_type() == torch::kInt64);
  const long M = logits.size(0);
  const int V = logits.size(1);
  auto lse = torch::empty({M}, logits.options().dtype(torch::kFloat32));
  auto tgt = torch::empty({M}, logits.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  dim3 grid(M), block(512);
  if (logits.scalar_type() == torch::kBFloat16)
    hipLaunchKernelGGL(ce_row_stats_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       (const __hip_bfloat16*)logits.data_ptr(), labels.data_ptr<long>(),
                       lse.data_ptr<float>(), tgt.data_ptr<float>(), V, ignore_index);
  else
    hipLaunchKernelGGL(ce_row_stats_kernel<float>, grid, block, 0, stream,
                       logits.data_ptr<float>(), labels.data_ptr<long>(),
                       lse.data_ptr<float>(), tgt.data_ptr<float>(), V, ignore_index);
  HIP_CHECK_LAST();
  return {lse, tgt};
}

void ce_grad_(torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
              double gscale, long ignore_index) {
  const long M = logits.size(0);
  const int V = logits.size(1);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  dim3 grid(M), block(512);
  if (logits.scalar_type() == torch::kBFloat16)
    hipLaunchKernelGGL(ce_grad_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       (__hip_bfloat16*)logits.data_ptr(), labels.data_ptr<long>(),
                       lse.data_ptr<float>(), (float)gscale, M, V, ignore_index);
  else
    hipLaunchKernelGGL(ce_grad_kernel<float>, grid, block, 0, stream,
                       logits.data_ptr<float>(), labels.data_ptr<long>(),
                       lse.data_ptr<float>(), (float)gscale, M, V, ignore_index);
  HIP_CHECK_LAST();
}
