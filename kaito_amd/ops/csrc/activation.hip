// Fused SiLU-and-mul (SwiGLU gate) for gfx950: out = silu(x[:, :I]) * x[:, I:].
// Memory-bound; short8-vectorized grid-stride loop.
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

__global__ void silu_and_mul_kernel(
    short* __restrict__ out,        // [T, I]
    const short* __restrict__ x,    // [T, 2I]  (gate | up)
    const int64_t T, const int I) {
  const int nvec = I / 8;
  const int64_t total = T * (int64_t)nvec;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / nvec;
    const int v = (int)(idx % nvec);
    const short8_t g = *reinterpret_cast<const short8_t*>(x + row * 2 * I + v * 8);
    const short8_t u = *reinterpret_cast<const short8_t*>(x + row * 2 * I + I + v * 8);
    short8_t o;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float gf = bf16_to_f32(g[j]);
      float uf = bf16_to_f32(u[j]);
      float s = gf / (1.f + __expf(-gf));
      o[j] = f32_to_bf16(s * uf);
    }
    *reinterpret_cast<short8_t*>(out + row * I + v * 8) = o;
  }
}

KAITO_DEV float gelu_tanh_f(float v) {
  // tanh-approx GELU (HF gelu_pytorch_tanh — gemma GeGLU / phi-2 MLP)
  const float c = 0.7978845608028654f;  // sqrt(2/pi)
  const float t = tanhf(c * (v + 0.044715f * v * v * v));
  return 0.5f * v * (1.f + t);
}

__global__ void gelu_and_mul_kernel(
    short* __restrict__ out,        // [T, I]
    const short* __restrict__ x,    // [T, 2I]  (gate | up)
    const int64_t T, const int I) {
  const int nvec = I / 8;
  const int64_t total = T * (int64_t)nvec;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / nvec;
    const int v = (int)(idx % nvec);
    const short8_t g = *reinterpret_cast<const short8_t*>(
        x + row * 2 * I + v * 8);
    const short8_t u = *reinterpret_cast<const short8_t*>(
        x + row * 2 * I + I + v * 8);
    short8_t o;
#pragma unroll
    for (int j = 0; j < 8; j++)
      o[j] = f32_to_bf16(gelu_tanh_f(bf16_to_f32(g[j])) * bf16_to_f32(u[j]));
    *reinterpret_cast<short8_t*>(out + row * I + v * 8) = o;
  }
}

__global__ void gelu_kernel(short* __restrict__ out,
                            const short* __restrict__ x, const int64_t n8) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    const short8_t v = *reinterpret_cast<const short8_t*>(x + i * 8);
    short8_t o;
#pragma unroll
    for (int j = 0; j < 8; j++)
      o[j] = f32_to_bf16(gelu_tanh_f(bf16_to_f32(v[j])));
    *reinterpret_cast<short8_t*>(out + i * 8) = o;
  }
}

void gelu_and_mul(at::Tensor out, at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  const int I2 = x.size(-1);
  TORCH_CHECK(I2 % 16 == 0);
  const int I = I2 / 2;
  const int64_t T = x.numel() / I2;
  auto stream = at::hip::getCurrentHIPStream();
  const int block = 256;
  const int64_t work = T * (I / 8);
  const int grid = (int)std::min<int64_t>((work + block - 1) / block, 2048);
  hipLaunchKernelGGL(gelu_and_mul_kernel, dim3(grid), dim3(block), 0, stream,
      (short*)out.data_ptr(), (const short*)x.data_ptr(), T, I);
}

void gelu(at::Tensor out, at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(x.numel() % 8 == 0);
  const int64_t n8 = x.numel() / 8;
  auto stream = at::hip::getCurrentHIPStream();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((n8 + block - 1) / block, 2048);
  hipLaunchKernelGGL(gelu_kernel, dim3(grid), dim3(block), 0, stream,
      (short*)out.data_ptr(), (const short*)x.data_ptr(), n8);
}

void silu_and_mul(at::Tensor out, at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  const int I2 = x.size(-1);
  TORCH_CHECK(I2 % 16 == 0);
  const int I = I2 / 2;
  const int64_t T = x.numel() / I2;
  auto stream = at::hip::getCurrentHIPStream();
  const int block = 256;
  const int64_t work = T * (I / 8);
  const int grid = (int)std::min<int64_t>((work + block - 1) / block, 2048);
  hipLaunchKernelGGL(silu_and_mul_kernel, dim3(grid), dim3(block), 0, stream,
      (short*)out.data_ptr(), (const short*)x.data_ptr(), T, I);
}

}  // namespace kaito
