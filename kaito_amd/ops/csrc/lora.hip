// Batched multi-adapter LoRA kernels for gfx950 (BGMV/SGMV class).
//
// Replaces vLLM's punica SGMV CUDA kernels (SURVEY.md §2.3 "LoRA
// multi-adapter serving" row). Per-token adapter indices; adapter weights
// are stacked [num_adapters, ...]; rank<=64. Both stages are memory-bound
// on adapter weights — per-adapter slabs are small (r*K) and hit L2/L3
// across the tokens sharing an adapter.
//
//   shrink: tmp[t, r] = scale * sum_k x[t, k] * A[idx[t], r, k]
//   expand: y[t, o]  += sum_r tmp[t, r] * B[idx[t], o, r]
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

__global__ void lora_shrink_kernel(
    float* __restrict__ tmp,          // [T, R] f32
    const short* __restrict__ x,      // [T, K] bf16 (row stride x_stride)
    const short* __restrict__ A,      // [L, R, K] bf16
    const int* __restrict__ idx,      // [T] adapter id; <0 → skip (zero)
    const int T, const int R, const int K,
    const int64_t x_stride, const float scale) {
  const int t = blockIdx.x;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int nw = blockDim.x >> 6;
  const int a = idx[t];
  for (int r = wave; r < R; r += nw) {
    if (a < 0) {
      if (lane == 0) tmp[(int64_t)t * R + r] = 0.f;
      continue;
    }
    const short* xr = x + (int64_t)t * x_stride;
    const short* ar = A + (((int64_t)a * R) + r) * K;
    float acc = 0.f;
    for (int k = lane * 8; k < K; k += 64 * 8) {
      short8_t xv = *reinterpret_cast<const short8_t*>(xr + k);
      short8_t av = *reinterpret_cast<const short8_t*>(ar + k);
#pragma unroll
      for (int j = 0; j < 8; j++)
        acc += bf16_to_f32(xv[j]) * bf16_to_f32(av[j]);
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) tmp[(int64_t)t * R + r] = acc * scale;
  }
}

__global__ void lora_expand_kernel(
    short* __restrict__ y,            // [T, O] bf16 (row stride y_stride)
    const float* __restrict__ tmp,    // [T, R]
    const short* __restrict__ B,      // [L, O, R] bf16
    const int* __restrict__ idx,
    const int T, const int O, const int R,
    const int64_t y_stride) {
  const int t = blockIdx.x;
  const int a = idx[t];
  if (a < 0) return;
  __shared__ float s_tmp[64];
  if (threadIdx.x < R) s_tmp[threadIdx.x] = tmp[(int64_t)t * R + threadIdx.x];
  __syncthreads();
  for (int o = blockIdx.y * blockDim.x + threadIdx.x; o < O;
       o += gridDim.y * blockDim.x) {
    const short* br = B + (((int64_t)a * O) + o) * R;
    float acc = 0.f;
    for (int r = 0; r < R; r++) acc += s_tmp[r] * bf16_to_f32(br[r]);
    short* yp = y + (int64_t)t * y_stride + o;
    *yp = f32_to_bf16(bf16_to_f32(*yp) + acc);
  }
}

void lora_shrink(at::Tensor tmp, at::Tensor x, at::Tensor A, at::Tensor idx,
                 double scale) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16);
  TORCH_CHECK(A.is_contiguous() && tmp.is_contiguous());
  TORCH_CHECK(x.stride(-1) == 1);
  TORCH_CHECK(idx.dtype() == at::kInt);
  const int T = x.size(0);
  const int K = x.size(-1);
  const int R = A.size(1);
  TORCH_CHECK(A.size(2) == K && K % 8 == 0, "K must be multiple of 8");
  TORCH_CHECK(R <= 64);
  if (T == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(lora_shrink_kernel, dim3(T), dim3(256), 0, stream,
      tmp.data_ptr<float>(), (const short*)x.data_ptr(),
      (const short*)A.data_ptr(), idx.data_ptr<int>(),
      T, R, K, x.stride(0), (float)scale);
}

void lora_expand(at::Tensor y, at::Tensor tmp, at::Tensor B, at::Tensor idx) {
  TORCH_CHECK(y.is_cuda() && y.dtype() == at::kBFloat16);
  TORCH_CHECK(B.is_contiguous() && tmp.is_contiguous());
  TORCH_CHECK(y.stride(-1) == 1);
  const int T = y.size(0);
  const int O = y.size(-1);
  const int R = B.size(2);
  TORCH_CHECK(B.size(1) == O && R <= 64);
  if (T == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  const int oy = std::min((O + 255) / 256, 16);
  hipLaunchKernelGGL(lora_expand_kernel, dim3(T, oy), dim3(256), 0, stream,
      (short*)y.data_ptr(), tmp.data_ptr<float>(),
      (const short*)B.data_ptr(), idx.data_ptr<int>(),
      T, O, R, y.stride(0));
}

}  // namespace kaito
