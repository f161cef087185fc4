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

// Grid-stride variant for tensors too big for the chunk table (embeddings,
// lm_head: ~65M elements on llama_1b). A dedicated launch also fills the
// whole chip (256 CUs) instead of the table's 320-block ceiling.
template <typename T>
__global__ void adamw_single_kernel(T* p, const T* __restrict__ g, T* m, T* v, long n,
                                    float lr, float beta1, float beta2, float eps,
                                    float wd, float bc1, float bc2) {
  const float decay = 1.f - lr * wd;
  const float step_size = lr / bc1;
  const float inv_bc2 = 1.f / bc2;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
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

static int single_grid(long numel) {
  long blocks = (numel + 256 * 4 - 1) / (256 * 4);
  return (int)(blocks < 16384 ? blocks : 16384);
}

void fused_adamw(std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs, std::vector<torch::Tensor> exp_avg_sqs,
                 double lr, double beta1, double beta2, double eps, double wd, long step) {
  TORCH_CHECK(params.size() == grads.size() && params.size() == exp_avgs.size());
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  const bool bf16 = params[0].scalar_type() == torch::kBFloat16;
  for (auto& t : params)
    TORCH_CHECK(t.scalar_type() == params[0].scalar_type(),
                "fused_adamw: all tensors in one call must share a dtype");

  size_t i = 0;
  while (i < params.size()) {
    {  // oversized tensors get their own full-chip launch
      long numel = params[i].numel();
      if ((numel + MT_CHUNK - 1) / MT_CHUNK > 320) {
        dim3 grid(single_grid(numel));
        if (bf16)
          hipLaunchKernelGGL(adamw_single_kernel<__hip_bfloat16>, grid, dim3(256), 0, stream,
                             (__hip_bfloat16*)params[i].data_ptr(),
                             (const __hip_bfloat16*)grads[i].data_ptr(),
                             (__hip_bfloat16*)exp_avgs[i].data_ptr(),
                             (__hip_bfloat16*)exp_avg_sqs[i].data_ptr(), numel,
                             (float)lr, (float)beta1, (float)beta2, (float)eps,
                             (float)wd, bc1, bc2);
        else
          hipLaunchKernelGGL(adamw_single_kernel<float>, grid, dim3(256), 0, stream,
                             (float*)params[i].data_ptr(), (const float*)grads[i].data_ptr(),
                             (float*)exp_avgs[i].data_ptr(), (float*)exp_avg_sqs[i].data_ptr(),
                             numel, (float)lr, (float)beta1, (float)beta2, (float)eps,
                             (float)wd, bc1, bc2);
        HIP_CHECK_LAST();
        ++i;
        continue;
      }
    }
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

template <typename T>
__global__ void l2norm_single_kernel(const T* __restrict__ g, long n, float* __restrict__ out) {
  __shared__ float scratch[16];
  const long stride = (long)gridDim.x * blockDim.x;
  float ss = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    float f = to_f32(g[i]);
    ss += f * f;
  }
  ss = block_reduce_sum(ss, scratch);
  if (threadIdx.x == 0) atomicAdd(out, ss);
}

template <typename T>
__global__ void scale_single_kernel(T* g, long n, float scale) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    g[i] = from_f32<T>(to_f32(g[i]) * scale);
}

// `launch` runs a packed chunk-table batch; `launch_single` one oversized tensor.
template <typename Fn, typename FnSingle>
static void norm_batches(std::vector<torch::Tensor>& grads, Fn&& launch,
                         FnSingle&& launch_single) {
  for (auto& t : grads)
    TORCH_CHECK(t.scalar_type() == grads[0].scalar_type(),
                "multi-tensor op: all tensors in one call must share a dtype");
  size_t i = 0;
  while (i < grads.size()) {
    if ((grads[i].numel() + MT_CHUNK - 1) / MT_CHUNK > 320) {
      launch_single(grads[i], single_grid(grads[i].numel()));
      ++i;
      continue;
    }
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
  }, [&](torch::Tensor& g, int grid) {
    if (bf16)
      hipLaunchKernelGGL(l2norm_single_kernel<__hip_bfloat16>, dim3(grid), dim3(256), 0,
                         stream, (const __hip_bfloat16*)g.data_ptr(), g.numel(),
                         out.data_ptr<float>());
    else
      hipLaunchKernelGGL(l2norm_single_kernel<float>, dim3(grid), dim3(256), 0, stream,
                         (const float*)g.data_ptr(), g.numel(), out.data_ptr<float>());
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
  }, [&](torch::Tensor& g, int grid) {
    if (bf16)
      hipLaunchKernelGGL(scale_single_kernel<__hip_bfloat16>, dim3(grid), dim3(256), 0,
                         stream, (__hip_bfloat16*)g.data_ptr(), g.numel(), (float)scale);
    else
      hipLaunchKernelGGL(scale_single_kernel<float>, dim3(grid), dim3(256), 0, stream,
                         (float*)g.data_ptr(), g.numel(), (float)scale);
    HIP_CHECK_LAST();
  });
}
