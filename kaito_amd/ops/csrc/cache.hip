// Paged-KV cache scatter: writes freshly-computed K/V rows into the block
// pool. Cache layout (chosen for the decode kernel's coalescing):
//   k_cache / v_cache: [num_blocks, KV_HEADS, BLOCK_SIZE, HEAD_DIM] bf16
// so one (block, kv_head) slab is BLOCK_SIZE*HEAD_DIM contiguous bf16 —
// a wave reads it with dense 16B lanes.
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

// FP8 = true: cache stores OCP e4m3 bytes (halves decode attention
// bytes + doubles KV capacity; unscaled like vLLM's fp8 default).
template <bool FP8>
__global__ void reshape_and_cache_kernel(
    const short* __restrict__ k,       // [T, KH*D]
    const short* __restrict__ v,       // [T, KH*D]
    void* __restrict__ k_cache,        // [B, KH, BS, D] bf16 | fp8
    void* __restrict__ v_cache,
    const int64_t* __restrict__ slots, // [T] = block*BS + off ; <0 = skip
    const int T, const int KH, const int D, const int BS,
    const int64_t kv_stride) {
  const int token = blockIdx.x;
  if (token >= T) return;
  const int64_t slot = slots[token];
  if (slot < 0) return;
  const int64_t block = slot / BS;
  const int off = (int)(slot % BS);
  const int nvec = KH * D / 8;
  const short8_t* kv = reinterpret_cast<const short8_t*>(k + (int64_t)token * kv_stride);
  const short8_t* vv = reinterpret_cast<const short8_t*>(v + (int64_t)token * kv_stride);
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    const int h = (i * 8) / D;
    const int d = (i * 8) % D;
    const int64_t dst = (((block * KH + h) * BS + off) * D + d) / 8;
    if constexpr (FP8) {
      const short8_t kr = kv[i], vr = vv[i];
      u8x8_t ko, vo;
#pragma unroll
      for (int j = 0; j < 4; j++) {
        const uint32_t kw = f32x2_to_fp8x2(bf16_to_f32(kr[2 * j]),
                                           bf16_to_f32(kr[2 * j + 1]));
        const uint32_t vw = f32x2_to_fp8x2(bf16_to_f32(vr[2 * j]),
                                           bf16_to_f32(vr[2 * j + 1]));
        ko[2 * j] = kw & 0xFF;     ko[2 * j + 1] = (kw >> 8) & 0xFF;
        vo[2 * j] = vw & 0xFF;     vo[2 * j + 1] = (vw >> 8) & 0xFF;
      }
      reinterpret_cast<u8x8_t*>(k_cache)[dst] = ko;
      reinterpret_cast<u8x8_t*>(v_cache)[dst] = vo;
    } else {
      reinterpret_cast<short8_t*>(k_cache)[dst] = kv[i];
      reinterpret_cast<short8_t*>(v_cache)[dst] = vv[i];
    }
  }
}

void reshape_and_cache(at::Tensor k, at::Tensor v, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor slot_mapping) {
  TORCH_CHECK(k.is_cuda() && k.dtype() == at::kBFloat16);
  TORCH_CHECK(k.stride(-1) == 1 && v.stride(-1) == 1);
  TORCH_CHECK(k.dim() == 3 && k.stride(1) == k.size(2),
              "k must be [T, KH, D] with contiguous heads");
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v must share row stride");
  TORCH_CHECK(slot_mapping.dtype() == at::kLong);
  const int T = slot_mapping.size(0);
  const int KH = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int D = k_cache.size(3);
  TORCH_CHECK(D % 8 == 0);
  auto stream = at::hip::getCurrentHIPStream();
  const int block = std::min(256, KH * D / 8);
  if (T == 0) return;
  if (k_cache.dtype() == at::kByte) {
    hipLaunchKernelGGL((reshape_and_cache_kernel<true>), dim3(T),
        dim3(block), 0, stream,
        (const short*)k.data_ptr(), (const short*)v.data_ptr(),
        k_cache.data_ptr(), v_cache.data_ptr(),
        slot_mapping.data_ptr<int64_t>(), T, KH, D, BS, k.stride(0));
  } else {
    hipLaunchKernelGGL((reshape_and_cache_kernel<false>), dim3(T),
        dim3(block), 0, stream,
        (const short*)k.data_ptr(), (const short*)v.data_ptr(),
        k_cache.data_ptr(), v_cache.data_ptr(),
        slot_mapping.data_ptr<int64_t>(), T, KH, D, BS, k.stride(0));
  }
}

}  // namespace kaito
