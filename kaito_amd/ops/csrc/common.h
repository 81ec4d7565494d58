// Common device helpers for kaito_amd CDNA4 (gfx950 / MI355X) kernels.
//
// Design notes (MI355X-first):
//  - wavefront = 64 lanes; all cross-lane reductions use 64-wide shuffles.
//  - bf16 global loads are ALWAYS vectorized (short4/short8 reinterpret):
//    hipcc does not auto-vectorize scalar bf16 loads (2-2.5x penalty).
//  - LDS = 32 banks x 4B; row-major tiles with 128B-multiple strides are
//    32-way conflicts on ds_read_b128 -> XOR-swizzle byte offsets.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define KAITO_WAVE 64
#define KAITO_DEV __device__ __forceinline__

// ---- vector types for wide loads --------------------------------------
typedef short  short4_t  __attribute__((ext_vector_type(4)));   // 8B  = 4 bf16
typedef short  short8_t  __attribute__((ext_vector_type(8)));   // 16B = 8 bf16
typedef float  float4_t  __attribute__((ext_vector_type(4)));
typedef float  float2_t  __attribute__((ext_vector_type(2)));
typedef short  bf16x8    __attribute__((ext_vector_type(8)));   // MFMA A/B frag (4 VGPRs)
typedef float  f32x4     __attribute__((ext_vector_type(4)));   // MFMA C/D frag 16x16
typedef __hip_bfloat16 bf16_t;

KAITO_DEV float bf16_to_f32(short u) {
  union { float f; uint32_t i; } v;
  v.i = (uint32_t)(uint16_t)u << 16;
  return v.f;
}

// ---- fp8 (OCP e4m3fn — gfx950 native converts; NOT the MI300 fnuz) ----
typedef uint8_t u8x8_t __attribute__((ext_vector_type(8)));   // 8 fp8 = 8B
typedef float   f32x2_t __attribute__((ext_vector_type(2)));

// pack two f32 into two e4m3 bytes (low half of the returned word)
KAITO_DEV uint32_t f32x2_to_fp8x2(float a, float b) {
  return (uint32_t)__builtin_amdgcn_cvt_pk_fp8_f32(a, b, 0, false);
}

// unpack bytes [0:1] (SEL=false) or [2:3] (SEL=true) of a word to 2 f32
// (the builtin's word-select must be an immediate)
template <bool SEL>
KAITO_DEV f32x2_t fp8x2_to_f32x2(uint32_t w) {
  return __builtin_amdgcn_cvt_pk_f32_fp8(w, SEL);
}

KAITO_DEV void fp8x8_to_f32(const u8x8_t v, float* __restrict__ out) {
  const uint32_t* w = reinterpret_cast<const uint32_t*>(&v);
#pragma unroll
  for (int s = 0; s < 2; s++) {
    const f32x2_t lo = fp8x2_to_f32x2<false>(w[s]);
    const f32x2_t hi = fp8x2_to_f32x2<true>(w[s]);
    out[4 * s + 0] = lo[0]; out[4 * s + 1] = lo[1];
    out[4 * s + 2] = hi[0]; out[4 * s + 3] = hi[1];
  }
}

KAITO_DEV short f32_to_bf16(float f) {
  union { float f; uint32_t i; } v;
  v.f = f;
  // round-to-nearest-even
  uint32_t rounding = 0x7FFF + ((v.i >> 16) & 1);
  return (short)((v.i + rounding) >> 16);
}

// ---- DPP cross-lane adds (VALU-speed; __shfl_xor lowers to
// ds_bpermute_b32 which costs a ~30-60 cyc LDS-crossbar trip each).
// xor1 = quad_perm(1,0,3,2)=0xB1; xor2 = quad_perm(2,3,0,1)=0x4E;
// xor4 = ROW_HALF_MIRROR=0x141; xor8 = ROW_ROR:8=0x128 (i^8 == i+8 mod 16
// within a row).
template <int CTRL>
KAITO_DEV float dpp_xor_add(float x) {
  union { float f; int i; } u, v;
  u.f = x;
  v.i = __builtin_amdgcn_update_dpp(0, u.i, CTRL, 0xf, 0xf, true);
  return x + v.f;
}

// ---- wave-level reductions (64 lanes) ---------------------------------
KAITO_DEV float wave_reduce_sum(float x) {
  x = dpp_xor_add<0xB1>(x);    // xor1 (quad_perm)
  x = dpp_xor_add<0x4E>(x);    // xor2 (quad_perm)
  x = dpp_xor_add<0x141>(x);   // xor4-equivalent (row_half_mirror)
  x = dpp_xor_add<0x128>(x);   // xor8-equivalent (row_ror:8)
#pragma unroll
  for (int off = 16; off < 64; off <<= 1) x += __shfl_xor(x, off, 64);
  return x;
}

KAITO_DEV float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, 64));
  return x;
}

// Reduce within a group of `W` consecutive lanes (W power of 2, <= 64).
// W<=16 uses pure-DPP butterflies (every lane ends with the group sum).
template <int W>
KAITO_DEV float group_reduce_sum(float x) {
  if constexpr (W >= 2) x = dpp_xor_add<0xB1>(x);
  if constexpr (W >= 4) x = dpp_xor_add<0x4E>(x);
  if constexpr (W >= 8) x = dpp_xor_add<0x141>(x);
  if constexpr (W >= 16) x = dpp_xor_add<0x128>(x);
#pragma unroll
  for (int off = 16; off < W; off <<= 1) x += __shfl_xor(x, off, 64);
  return x;
}

template <int W>
KAITO_DEV float group_reduce_max(float x) {
#pragma unroll
  for (int off = W / 2; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, 64));
  return x;
}

// Block-level reduce over all waves via LDS scratch (caller provides
// __shared__ float scratch[nwaves]); valid for blockDim.x % 64 == 0.
KAITO_DEV float block_reduce_sum(float x, float* scratch) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
  x = wave_reduce_sum(x);
  if (lane == 0) scratch[wave] = x;
  __syncthreads();
  x = (lane < nwaves) ? scratch[lane] : 0.f;
  x = wave_reduce_sum(x);  // cheap: only first nwaves lanes carry data
  return x;
}

#define HIP_CHECK_KAITO(expr)                                              \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));            \
    }                                                                      \
  } while (0)
