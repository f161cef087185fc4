#include "hip/hip_runtime.h"
// Blockwise weight quantization (K15) for gfx950 — the frozen-W memory
// saver behind `--quantize 4bit|8bit`.
//
// The reference delegates to bitsandbytes (CUDA-only: Params4bit /
// quantize_4bit / dequantize_blockwise, reference relora.py:225-238,
// 277-299).  MI355X-native equivalents:
//   * NF4: the standard 16-level normal-float codebook, 64-element blocks,
//     fp32 absmax per block, two codes packed per byte — bit-compatible
//     with the published NF4 format;
//   * int8: symmetric linear blockwise (absmax/127) with 256-element
//     blocks.  (bitsandbytes uses a non-uniform dynamic map here; ours is
//     linear — documented framework difference, same memory footprint.)
//
// Training-path use: weights live quantized at rest; forward dequantizes
// into a transient bf16 buffer feeding the normal MFMA GEMM path, and the
// ReLoRA merge does dequant -> W += s*BA -> requant (relora.py).

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

// NF4 codebook (QLoRA paper, bitsandbytes nf4 table)
__constant__ float NF4_CODE[16] = {
    -1.0f, -0.6961928009986877f, -0.5250730514526367f, -0.39491748809814453f,
    -0.28444138169288635f, -0.18477343022823334f, -0.09105003625154495f, 0.0f,
    0.07958029955625534f, 0.16093020141124725f, 0.24611230194568634f,
    0.33791524171829224f, 0.44070982933044434f, 0.5626170039176941f,
    0.7229568362236023f, 1.0f};

#define NF4_BLOCK 64
#define I8_BLOCK 256

DEV_INLINE int nf4_nearest(float v) {
  // 15-ary comparison against midpoints; v in [-1, 1]
  int best = 0;
  float bd = fabsf(v - NF4_CODE[0]);
#pragma unroll
  for (int i = 1; i < 16; ++i) {
    const float d = fabsf(v - NF4_CODE[i]);
    if (d < bd) { bd = d; best = i; }
  }
  return best;
}

// one wave per block of 64 elements -> 32 packed bytes + 1 absmax
template <typename T>
__global__ void quantize_nf4_kernel(const T* __restrict__ x, uint8_t* __restrict__ q,
                                    float* __restrict__ absmax, long nblocks, long n) {
  const long blk = (long)blockIdx.x * (blockDim.x / WAVE) + (threadIdx.x / WAVE);
  if (blk >= nblocks) return;
  const int lane = threadIdx.x & (WAVE - 1);
  const long i = blk * NF4_BLOCK + lane;
  const float v = (lane < NF4_BLOCK && i < n) ? to_f32(x[i]) : 0.f;
  float am = fabsf(v);
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) am = fmaxf(am, __shfl_down(am, off));
  am = __shfl(am, 0);
  if (lane == 0) absmax[blk] = am;
  const float inv = (am > 0.f) ? 1.f / am : 0.f;
  if (lane < NF4_BLOCK / 2) {
    const long j = blk * NF4_BLOCK + lane * 2;
    const int hi = (j < n) ? nf4_nearest(to_f32(x[j]) * inv) : 0;
    const int lo = (j + 1 < n) ? nf4_nearest(to_f32(x[j + 1]) * inv) : 0;
    q[blk * (NF4_BLOCK / 2) + lane] = (uint8_t)((hi << 4) | lo);
  }
}

template <typename T>
__global__ void dequantize_nf4_kernel(const uint8_t* __restrict__ q,
                                      const float* __restrict__ absmax,
                                      T* __restrict__ out, long n) {
  const long b = (long)blockIdx.x * blockDim.x + threadIdx.x;  // byte index
  if (b * 2 >= n) return;
  const long blk = b / (NF4_BLOCK / 2);
  const float am = absmax[blk];
  const uint8_t code = q[b];
  out[b * 2] = from_f32<T>(NF4_CODE[code >> 4] * am);
  if (b * 2 + 1 < n) out[b * 2 + 1] = from_f32<T>(NF4_CODE[code & 0xF] * am);
}

template <typename T>
__global__ void quantize_int8_kernel(const T* __restrict__ x, int8_t* __restrict__ q,
                                     float* __restrict__ absmax, long nblocks, long n) {
  __shared__ float scratch[16];
  const long blk = blockIdx.x;
  if (blk >= nblocks) return;
  const long s = blk * I8_BLOCK;
  const long e = min(n, s + (long)I8_BLOCK);
  float am = 0.f;
  for (long i = s + threadIdx.x; i < e; i += blockDim.x) am = fmaxf(am, fabsf(to_f32(x[i])));
  am = block_reduce_max(am, scratch);
  if (threadIdx.x == 0) absmax[blk] = am;
  const float scale = (am > 0.f) ? 127.f / am : 0.f;
  for (long i = s + threadIdx.x; i < e; i += blockDim.x) {
    const float v = to_f32(x[i]) * scale;
    q[i] = (int8_t)lrintf(fminf(fmaxf(v, -127.f), 127.f));
  }
}

template <typename T>
__global__ void dequantize_int8_kernel(const int8_t* __restrict__ q,
                                       const float* __restrict__ absmax,
                                       T* __restrict__ out, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float am = absmax[i / I8_BLOCK];
  out[i] = from_f32<T>((float)q[i] * (am / 127.f));
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> quantize_nf4(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const long n = x.numel();
  TORCH_CHECK(n % 2 == 0, "nf4: even numel required");
  const long nblocks = (n + NF4_BLOCK - 1) / NF4_BLOCK;
  auto q = torch::empty({(n + 1) / 2}, x.options().dtype(torch::kUInt8));
  auto absmax = torch::empty({nblocks}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int waves_per_block = 4;
  dim3 grid((nblocks + waves_per_block - 1) / waves_per_block), block(waves_per_block * WAVE);
  if (x.scalar_type() == torch::kBFloat16)
    hipLaunchKernelGGL(quantize_nf4_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), q.data_ptr<uint8_t>(),
                       absmax.data_ptr<float>(), nblocks, n);
  else
    hipLaunchKernelGGL(quantize_nf4_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), q.data_ptr<uint8_t>(),
                       absmax.data_ptr<float>(), nblocks, n);
  HIP_CHECK_LAST();
  return {q, absmax};
}

torch::Tensor dequantize_nf4(torch::Tensor q, torch::Tensor absmax, long n,
                             torch::ScalarType dtype) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous());
  auto out = torch::empty({n}, q.options().dtype(dtype));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const long nbytes = (n + 1) / 2;
  dim3 grid((nbytes + 255) / 256), block(256);
  if (dtype == torch::kBFloat16)
    hipLaunchKernelGGL(dequantize_nf4_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       q.data_ptr<uint8_t>(), absmax.data_ptr<float>(),
                       (__hip_bfloat16*)out.data_ptr(), n);
  else
    hipLaunchKernelGGL(dequantize_nf4_kernel<float>, grid, block, 0, stream,
                       q.data_ptr<uint8_t>(), absmax.data_ptr<float>(),
                       out.data_ptr<float>(), n);
  HIP_CHECK_LAST();
  return out;
}

std::vector<torch::Tensor> quantize_int8(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const long n = x.numel();
  const long nblocks = (n + I8_BLOCK - 1) / I8_BLOCK;
  auto q = torch::empty({n}, x.options().dtype(torch::kChar));
  auto absmax = torch::empty({nblocks}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  dim3 grid(nblocks), block(256);
  if (x.scalar_type() == torch::kBFloat16)
    hipLaunchKernelGGL(quantize_int8_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), q.data_ptr<int8_t>(),
                       absmax.data_ptr<float>(), nblocks, n);
  else
    hipLaunchKernelGGL(quantize_int8_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), q.data_ptr<int8_t>(),
                       absmax.data_ptr<float>(), nblocks, n);
  HIP_CHECK_LAST();
  return {q, absmax};
}

torch::Tensor dequantize_int8(torch::Tensor q, torch::Tensor absmax, long n,
                              torch::ScalarType dtype) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous());
  auto out = torch::empty({n}, q.options().dtype(dtype));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  dim3 grid((n + 255) / 256), block(256);
  if (dtype == torch::kBFloat16)
    hipLaunchKernelGGL(dequantize_int8_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       q.data_ptr<int8_t>(), absmax.data_ptr<float>(),
                       (__hip_bfloat16*)out.data_ptr(), n);
  else
    hipLaunchKernelGGL(dequantize_int8_kernel<float>, grid, block, 0, stream,
                       q.data_ptr<int8_t>(), absmax.data_ptr<float>(),
                       out.data_ptr<float>(), n);
  HIP_CHECK_LAST();
  return out;
}
