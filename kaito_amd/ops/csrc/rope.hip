// Rotary position embedding (neox / llama "rotate-half" style), in-place
// on Q and K. gfx950: trig tables are precomputed on device once (host-side
// cache tensor) — on-device sinf/cosf per element turns a memory-bound op
// VALU-bound (guide Appendix B).
//
// cos_sin_cache: [max_pos, rot_dim] f32 laid out as [cos(rot_dim/2) | sin(rot_dim/2)].
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

// q: [T, QH*D], k: [T, KH*D] (contiguous rows); positions: [T] int32/64.
// rotate-half: out[i]       = x[i]*cos[i] - x[i+R/2]*sin[i]
//              out[i+R/2]   = x[i+R/2]*cos[i] + x[i]*sin[i]      (i < R/2)
__global__ void rope_kernel(
    short* __restrict__ q, short* __restrict__ k,
    const int64_t* __restrict__ positions,
    const float* __restrict__ cos_sin,  // [max_pos, R]
    const int T, const int QH, const int KH, const int D, const int R,
    const int64_t q_stride, const int64_t k_stride) {
  const int token = blockIdx.x;
  if (token >= T) return;
  const int64_t pos = positions[token];
  const float* cs = cos_sin + pos * R;
  const int half = R / 2;
  const int total_heads = QH + KH;

  // Each thread handles one (head, i<half) pair; vectorize by 2 along i.
  for (int idx = threadIdx.x; idx < total_heads * half / 2; idx += blockDim.x) {
    const int pair = idx * 2;               // i offset within half, step 2
    const int h = pair / half;
    const int i = pair % half;
    short* base = (h < QH)
        ? q + (int64_t)token * q_stride + (int64_t)h * D
        : k + (int64_t)token * k_stride + (int64_t)(h - QH) * D;
    // load x[i..i+1] and x[i+half..i+half+1]
    short2 xa = *reinterpret_cast<short2*>(base + i);
    short2 xb = *reinterpret_cast<short2*>(base + i + half);
    float2 c = *reinterpret_cast<const float2*>(cs + i);
    float2 s = *reinterpret_cast<const float2*>(cs + half + i);
    float a0 = bf16_to_f32(xa.x), a1 = bf16_to_f32(xa.y);
    float b0 = bf16_to_f32(xb.x), b1 = bf16_to_f32(xb.y);
    short2 oa, ob;
    oa.x = f32_to_bf16(a0 * c.x - b0 * s.x);
    oa.y = f32_to_bf16(a1 * c.y - b1 * s.y);
    ob.x = f32_to_bf16(b0 * c.x + a0 * s.x);
    ob.y = f32_to_bf16(b1 * c.y + a1 * s.y);
    *reinterpret_cast<short2*>(base + i) = oa;
    *reinterpret_cast<short2*>(base + i + half) = ob;
  }
}

void rotary_embedding(at::Tensor positions, at::Tensor q, at::Tensor k,
                      int64_t head_dim, at::Tensor cos_sin_cache) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == at::kBFloat16);
  TORCH_CHECK(positions.dtype() == at::kLong);
  TORCH_CHECK(q.stride(-1) == 1 && k.stride(-1) == 1,
              "rope: innermost dim must be contiguous");
  const int T = positions.size(0);
  const int R = cos_sin_cache.size(1);
  const int D = (int)head_dim;
  const int QH = (int)(q.size(-1) / D) * (q.dim() == 3 ? (int)q.size(1) : 1);
  const int KH = (int)(k.size(-1) / D) * (k.dim() == 3 ? (int)k.size(1) : 1);
  TORCH_CHECK(R <= D && R % 4 == 0, "rot_dim must be <= head_dim, mult of 4");
  auto stream = at::hip::getCurrentHIPStream();
  const int block = 256;
  hipLaunchKernelGGL(rope_kernel, dim3(T), dim3(block), 0, stream,
      (short*)q.data_ptr(), (short*)k.data_ptr(),
      positions.data_ptr<int64_t>(), cos_sin_cache.data_ptr<float>(),
      T, QH, KH, D, R, q.stride(0), k.stride(0));
}

}  // namespace kaito
