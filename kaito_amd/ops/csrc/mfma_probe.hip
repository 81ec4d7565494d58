// Tiny single-wave MFMA tile GEMM used by tests to verify the gfx950
// mfma_f32_16x16x32_bf16 fragment layout assumptions that
// prefill_attention.hip relies on. C[16,16] = A[16,32] @ B[32,16].
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

__global__ void mfma_tile_kernel(float* __restrict__ c,
                                 const short* __restrict__ a,   // [16,32]
                                 const short* __restrict__ b) { // [32,16]
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 4, lo = lane & 15;
  // A: row=lo, k=hi*8+[0..7] contiguous
  bf16x8 af = *reinterpret_cast<const bf16x8*>(a + lo * 32 + hi * 8);
  // B: col=lo, k=hi*8+[0..7] → strided gather (column of B)
  bf16x8 bf;
#pragma unroll
  for (int j = 0; j < 8; j++) bf[j] = b[(hi * 8 + j) * 16 + lo];
  f32x4 acc{0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
  // C/D: col=lo, row=hi*4+r
#pragma unroll
  for (int r = 0; r < 4; r++) c[(hi * 4 + r) * 16 + lo] = acc[r];
}

at::Tensor mfma_tile_gemm(at::Tensor a, at::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == at::kBFloat16);
  TORCH_CHECK(a.sizes() == at::IntArrayRef({16, 32}) &&
              b.sizes() == at::IntArrayRef({32, 16}));
  auto c = at::empty({16, 16}, a.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_tile_kernel, dim3(1), dim3(64), 0, stream,
      c.data_ptr<float>(), (const short*)a.contiguous().data_ptr(),
      (const short*)b.contiguous().data_ptr());
  return c;
}

}  // namespace kaito
