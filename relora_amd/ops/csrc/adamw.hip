#include "hip/hip_runtime.h"
// Fused multi-tensor AdamW (K11) and L2-norm / scale (K12) for gfx950.
// Chunk-table multi-tensor apply: the host packs up to MT_MAX tensor
// pointers into kernel arguments; each block processes one MT_CHUNK-element
// chunk. Matches torch.optim.AdamW math (decoupled wd, bias correction) with
// fp32 internal math regardless of state dtype.

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

#define MT_MAX 48
#define MT_CHUNK 65536

struct AdamWArgs {
  void* p[MT_MAX];
  void* g[MT_MAX];
  void* m[MT_MAX];
  void* v[MT_MAX];
  long numel[MT_MAX];
  unsigned char tensor_of_block[320];  // which tensor a block belongs to
  int chunk_of_block[320];             // chunk INDEX within that tensor
};

template <typename T>
__global__ void fused_adamw_kernel(AdamWArgs args, float lr, float beta1, float beta2,
                                   float eps, float wd, float bc1, float bc2) {
  const int t = args.tensor_of_block[blockIdx.x];
  const long start = (long)args.chunk_of_block[blockIdx.x] * MT_CHUNK;
  const long n = args.numel[t];
  T* p = (T*)args.p[t] + start;
  const T* g = (const T*)args.g[t] + start;
  T* m = (T*)args.m[t] + start;
  T* v = (T*)args.v[t] + start;
  const long n_rem = n - start;
  const long len = n_rem < MT_CHUNK ? n_rem : MT_CHUNK;

  const float decay = 1.f - lr * wd;
  const float step_size = lr / bc1;
  const float inv_bc2 = 1.f / bc2;

  for (long i = threadIdx.x; i < len; i += blockDim.x) {
    float pf = to_f32(p[i]) * decay;
    float gf = to_f32(g[i]);
    float mf = beta1 * to_f32(m[i]) + (1.f - beta1) * gf;
    float vf = beta2 * to_f32(v[i]) + (1.f - beta2) * gf * gf;
    m[i] = from_f32<T>(mf);
    v[i] = from_f32<T>(vf);
    const float denom = sqrtf(vf * inv_bc2) + eps;
    p[i] = from_f32<T>(pf - step_size * mf / denom);
  }
}

void fused_adamw(std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs, std::vector<torch::Tensor> exp_avg_sqs,
                 double lr, double beta1, double beta2, double eps, double wd, long step) {
  TORCH_CHECK(params.size() == grads.size() && params.size() == exp_avgs.size());
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);

  size_t i = 0;
  while (i < params.size()) {
    AdamWArgs args;
    int nt = 0, nb = 0;
    while (i < params.size() && nt < MT_MAX) {
      long numel = params[i].numel();
      int chunks = (int)((numel + MT_CHUNK - 1) / MT_CHUNK);
      if (nb + chunks > 320) break;
      args.p[nt] = params[i].data_ptr();
      args.g[nt] = grads[i].data_ptr();
      args.m[nt] = exp_avgs[i].data_ptr();
      args.v[nt] = exp_avg_sqs[i].data_ptr();
      args.numel[nt] = numel;
      for (int c = 0; c < chunks; ++c) {
        args.tensor_of_block[nb] = (unsigned char)nt;
        args.chunk_of_block[nb] = c;
        ++nb;
      }
      ++nt;
      ++i;
    }
    TORCH_CHECK(nb > 0, "tensor too large for one multi-tensor batch");
    if (params[0].scalar_type() == torch::kBFloat16)
      hipLaunchKernelGGL(fused_adamw_kernel<__hip_bfloat16>, dim3(nb), dim3(256), 0, stream,
                         args, (float)lr, (float)beta1, (float)beta2, (float)eps,
                         (float)wd, bc1, bc2);
    else
      hipLaunchKernelGGL(fused_adamw_kernel<float>, dim3(nb), dim3(256), 0, stream,
                         args, (float)lr, (float)beta1, (float)beta2, (float)eps,
                         (float)wd, bc1, bc2);
    HIP_CHECK_LAST();
  }
}

// ---------------------------------------------------------------------------
// multi-tensor L2 norm (sum of squares -> atomic add) and scale
// ---------------------------------------------------------------------------

struct NormArgs {
  void* g[MT_MAX];
  long numel[MT_MAX];
  unsigned char tensor_of_block[320];
  int chunk_of_block[320];
};

template <typename T>
__global__ void l2norm_kernel(NormArgs args, float* __restrict__ out) {
  __shared__ float scratch[16];
  const int t = args.tensor_of_block[blockIdx.x];
  const long start = (long)args.chunk_of_block[blockIdx.x] * MT_CHUNK;
  const T* g = (const T*)args.g[t] + start;
  const long n_rem = args.numel[t] - start;
  const long len = n_rem < MT_CHUNK ? n_rem : MT_CHUNK;
  float ss = 0.f;
  for (long i = threadIdx.x; i < len; i += blockDim.x) {
    float f = to_f32(g[i]);
    ss += f * f;
  }
  ss = block_reduce_sum(ss, scratch);
  if (threadIdx.x == 0) atomicAdd(out, ss);
}

template <typename T>
__global__ void scale_kernel(NormArgs args, float scale) {
  const int t = args.tensor_of_block[blockIdx.x];
  const long start = (long)args.chunk_of_block[blockIdx.x] * MT_CHUNK;
  T* g = (T*)args.g[t] + start;
  const long n_rem = args.numel[t] - start;
  const long len = n_rem < MT_CHUNK ? n_rem : MT_CHUNK;
  for (long i = threadIdx.x; i < len; i += blockDim.x)
    g[i] = from_f32<T>(to_f32(g[i]) * scale);
}

template <typename Fn>
static void norm_batches(std::vector<torch::Tensor>& grads, Fn&& launch) {
  size_t i = 0;
  while (i < grads.size()) {
    NormArgs args;
    int nt = 0, nb = 0;
    while (i < grads.size() && nt < MT_MAX) {
      long numel = grads[i].numel();
      int chunks = (int)((numel + MT_CHUNK - 1) / MT_CHUNK);
      if (nb + chunks > 320) break;
      args.g[nt] = grads[i].data_ptr();
      args.numel[nt] = numel;
      for (int c = 0; c < chunks; ++c) {
        args.tensor_of_block[nb] = (unsigned char)nt;
        args.chunk_of_block[nb] = c;
        ++nb;
      }
      ++nt;
      ++i;
    }
    TORCH_CHECK(nb > 0, "tensor too large for one multi-tensor batch");
    launch(args, nb);
  }
}

torch::Tensor multi_tensor_l2norm(std::vector<torch::Tensor> grads) {
  TORCH_CHECK(!grads.empty());
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  auto out = torch::zeros({1}, grads[0].options().dtype(torch::kFloat32));
  const bool bf16 = grads[0].scalar_type() == torch::kBFloat16;
  norm_batches(grads, [&](NormArgs& args, int nb) {
    if (bf16)
      hipLaunchKernelGGL(l2norm_kernel<__hip_bfloat16>, dim3(nb), dim3(256), 0, stream,
                         args, out.data_ptr<float>());
    else
      hipLaunchKernelGGL(l2norm_kernel<float>, dim3(nb), dim3(256), 0, stream,
                         args, out.data_ptr<float>());
    HIP_CHECK_LAST();
  });
  return out.sqrt_().squeeze(0);
}

void multi_tensor_scale_(std::vector<torch::Tensor> grads, double scale) {
  TORCH_CHECK(!grads.empty());
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const bool bf16 = grads[0].scalar_type() == torch::kBFloat16;
  norm_batches(grads, [&](NormArgs& args, int nb) {
    if (bf16)
      hipLaunchKernelGGL(scale_kernel<__hip_bfloat16>, dim3(nb), dim3(256), 0, stream,
                         args, (float)scale);
    else
      hipLaunchKernelGGL(scale_kernel<float>, dim3(nb), dim3(256), 0, stream,
                         args, (float)scale);
    HIP_CHECK_LAST();
  });
}
