// W4A16 group-quantized linear kernels (AWQ/GPTQ-class presets —
// reference inventory row "quantized GEMM"; the catalog
// carries qwen3-8b-awq etc.).
//
// Native layout (loaders repack AWQ checkpoints into this):
//   qweight: uint32 [N, K/8]   8 consecutive K nibbles per word, row-major
//   scales:  float  [N, K/G]   group size G (typ. 128), w = s*q - z
//   zeros:   float  [N, K/G]   z folded as s*zq at repack time
//
// Two paths:
//   w4a16_gemv    : M <= 32 decode shapes. Memory-bound on the 4-bit
//                   weights (N*K/2 bytes vs 2*N*K for bf16 -> ~4x less
//                   traffic than a bf16 GEMM at small M). One wave per
//                   output row; lane takes 8-k packs round-robin so the
//                   qweight reads coalesce per wave; x reads are short8.
//   w4a16_dequant : tile dequant to bf16 scratch; large-M callers then run
//                   the scratch through hipBLASLt MFMA (dequant is
//                   O(N*K) against the GEMM's O(M*N*K) -> noise at M>=128).
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

// ---------------------------------------------------------------- gemv
// grid.x = N / ROWS, block = ROWS waves; each wave owns one output row.
template <int M_TILE>
__global__ __launch_bounds__(256, 4)
void w4a16_gemv_kernel(short* __restrict__ out,          // [M, N] bf16
                       const short* __restrict__ x,      // [M, K] bf16
                       const uint32_t* __restrict__ qw,  // [N, K/8]
                       const float* __restrict__ scales, // [N, K/G]
                       const float* __restrict__ zeros,  // [N, K/G]
                       int M, int N, int K, int group) {
  // Each wave owns one output row n. A lane takes a CONTIGUOUS 8-word
  // (64-k) slice per tile so its slice sits inside ONE quant group:
  // scale/zero load once per slice (vs per word), q loads are uint4
  // (16B), x loads are short8 (16B). Weight traffic is then the pure
  // N*K/2 bytes and the kernel runs at HBM rate for small M.
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n = blockIdx.x * 4 + wave;
  if (n >= N) return;
  const int kw = K >> 3;                      // u32 words per row
  const uint32_t* qrow = qw + (int64_t)n * kw;
  const int gstride = K / group;
  const float* srow = scales + (int64_t)n * gstride;
  const float* zrow = zeros + (int64_t)n * gstride;

  float acc[M_TILE];
#pragma unroll
  for (int m = 0; m < M_TILE; ++m) acc[m] = 0.f;

  typedef uint32_t uint4_t __attribute__((ext_vector_type(4)));
  for (int t = lane * 8; t < kw; t += 64 * 8) {  // 8 contiguous words/lane
    const int k0 = t << 3;                       // 64 k per slice
    const float s = srow[k0 / group];
    const float z = zrow[k0 / group];
    const uint4_t qa = *reinterpret_cast<const uint4_t*>(qrow + t);
    const uint4_t qb = *reinterpret_cast<const uint4_t*>(qrow + t + 4);
    float dotm[M_TILE], xsm[M_TILE];
#pragma unroll
    for (int m = 0; m < M_TILE; ++m) { dotm[m] = 0.f; xsm[m] = 0.f; }
#pragma unroll
    for (int w = 0; w < 8; ++w) {
      const uint32_t q = (w < 4) ? qa[w] : qb[w - 4];
      float qv[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) qv[j] = (float)((q >> (4 * j)) & 0xF);
#pragma unroll
      for (int m = 0; m < M_TILE; ++m) {
        if (m >= M) break;
        const short8_t xv = *reinterpret_cast<const short8_t*>(
            x + (int64_t)m * K + k0 + w * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float xf = bf16_to_f32(xv[j]);
          dotm[m] += xf * qv[j];
          xsm[m] += xf;
        }
      }
    }
#pragma unroll
    for (int m = 0; m < M_TILE; ++m) acc[m] += s * dotm[m] - z * xsm[m];
  }
#pragma unroll
  for (int m = 0; m < M_TILE; ++m) {
    if (m >= M) break;
    const float r = wave_reduce_sum(acc[m]);
    if (lane == 0) out[(int64_t)m * N + n] = f32_to_bf16(r);
  }
}

// ---------------------------------------------------------------- dequant
// out[N, K] bf16 <- dequantized weights. Elementwise, one u32 pack per
// thread; writes are short8 (16B) so the store path is fully vectorized.
__global__ __launch_bounds__(256)
void w4a16_dequant_kernel(short* __restrict__ out,
                          const uint32_t* __restrict__ qw,
                          const float* __restrict__ scales,
                          const float* __restrict__ zeros,
                          int N, int K, int group) {
  const int kw = K >> 3;
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (int64_t)N * kw) return;
  const int n = idx / kw;
  const int w = idx % kw;
  const int k0 = w << 3;
  const int g = k0 / group;
  const int gstride = K / group;
  const float s = scales[(int64_t)n * gstride + g];
  const float z = zeros[(int64_t)n * gstride + g];
  const uint32_t q = qw[idx];
  short8_t o;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    o[j] = f32_to_bf16(s * (float)((q >> (4 * j)) & 0xF) - z);
  *reinterpret_cast<short8_t*>(out + (int64_t)n * K + k0) = o;
}

// ------------------------------------------------------- fused MFMA GEMM
// Marlin-class mid/large-M path: C[M,N] = x[M,K]bf16 @ dequant(W)^T with
// the dequant INLINE in the K-loop — the 4-bit weights go HBM → register
// → (dequant) → swizzled LDS → MFMA, never touching a full-precision
// scratch buffer (the round-1 path dequantized the whole weight to
// global memory first: 5x the weight traffic, and it lost to bf16 at
// saturation — docs/ROADMAP.md #2).
//
// Structure: 2-barrier 64x64x64 tile, 4 waves as 2x2, MFMA 16x16x32
// (same skeleton as moe.hip). A staged with global_load_lds
// (pre-swizzled source); B: each thread loads u32 packed words (one
// word == one 16-B bf16 chunk after dequant), dequants with its group's
// (s, z), and ds_writes the chunk at the XOR-swizzled slot.
__global__ __launch_bounds__(256, 2)
void w4a16_gemm_kernel(short* __restrict__ out,          // [M, N] bf16
                       const short* __restrict__ x,      // [M, K] bf16
                       const uint32_t* __restrict__ qw,  // [N, K/8]
                       const float* __restrict__ scales, // [N, K/G]
                       const float* __restrict__ zeros,
                       int M, int N, int K, int group) {
  constexpr int BM = 64, BN = 64, BK = 64;
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wm = wave >> 1, wn = wave & 1;
  const int lo = lane & 15, hi = lane >> 4;
  const int kw = K >> 3;
  const int gstride = K / group;

  __shared__ short lds_a[BM * BK];
  __shared__ short lds_b[BN * BK];

  f32x4 acc[2][2];
#pragma unroll
  for (int a = 0; a < 2; a++)
#pragma unroll
    for (int b = 0; b < 2; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += BK) {
    // ---- A tile via global_load_lds (rows clamped at the M tail) ----
#pragma unroll
    for (int i = 0; i < 2; i++) {
      const int seg = i * 4 + wave;
      const int idx = seg * 64 + lane;
      const int row = idx >> 3;
      const int c = (idx & 7) ^ (row & 7);
      const int srow = min(m0 + row, M - 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)
              (x + (int64_t)srow * K + k0 + c * 8),
          (__attribute__((address_space(3))) unsigned int*)
              (lds_a + seg * 512),
          16, 0, 0);
    }
    // ---- B tile: packed u32 → inline dequant → swizzled ds_write ----
#pragma unroll
    for (int rep = 0; rep < 2; rep++) {
      const int id = rep * 256 + threadIdx.x;     // 512 words per tile
      const int row = id >> 3;                    // n row in tile
      const int wslot = id & 7;                   // 8 words (64 k) / row
      const int n = n0 + row;
      const uint32_t q = qw[(int64_t)n * kw + (k0 >> 3) + wslot];
      const int g = (k0 + wslot * 8) / group;
      const float s = scales[(int64_t)n * gstride + g];
      const float z = zeros[(int64_t)n * gstride + g];
      short8_t dq;
#pragma unroll
      for (int j = 0; j < 8; j++)
        dq[j] = f32_to_bf16(s * (float)((q >> (4 * j)) & 0xF) - z);
      const int c = wslot ^ (row & 7);
      *reinterpret_cast<short8_t*>(lds_b + row * BK + c * 8) = dq;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK / 32; kk++) {
      bf16x8 af[2], bf[2];
#pragma unroll
      for (int r = 0; r < 2; r++) {
        {
          const int row = wm * 32 + r * 16 + lo;
          const int c = (kk * 4 + hi) ^ (row & 7);
          af[r] = *reinterpret_cast<const bf16x8*>(lds_a + row * BK + c * 8);
        }
        {
          const int row = wn * 32 + r * 16 + lo;
          const int c = (kk * 4 + hi) ^ (row & 7);
          bf[r] = *reinterpret_cast<const bf16x8*>(lds_b + row * BK + c * 8);
        }
      }
#pragma unroll
      for (int a = 0; a < 2; a++)
#pragma unroll
        for (int b = 0; b < 2; b++)
          acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[a], bf[b], acc[a][b], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int a = 0; a < 2; a++) {
    const int row_base = wm * 32 + a * 16 + hi * 4;
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int row = m0 + row_base + r;
      if (row >= M) continue;
#pragma unroll
      for (int b = 0; b < 2; b++) {
        const int col = n0 + wn * 32 + b * 16 + lo;
        out[(int64_t)row * N + col] = f32_to_bf16(acc[a][b][r]);
      }
    }
  }
}

void w4a16_gemm(at::Tensor out, at::Tensor x, at::Tensor qweight,
                at::Tensor scales, at::Tensor zeros, int64_t group) {
  const int M = x.size(0), K = x.size(1), N = qweight.size(0);
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(K % 64 == 0 && N % 64 == 0 && group % 64 == 0 &&
              K % group == 0, "w4a16_gemm: 64|K, 64|N, 64|group required");
  dim3 grid((M + 63) / 64, N / 64), block(256);
  auto stream = at::hip::getCurrentHIPStream();
  w4a16_gemm_kernel<<<grid, block, 0, stream>>>(
      (short*)out.data_ptr(), (const short*)x.data_ptr(),
      (const uint32_t*)qweight.data_ptr(), scales.data_ptr<float>(),
      zeros.data_ptr<float>(), M, N, K, (int)group);
}

void w4a16_gemv(at::Tensor out, at::Tensor x, at::Tensor qweight,
                at::Tensor scales, at::Tensor zeros, int64_t group) {
  const int M = x.size(0), K = x.size(1), N = qweight.size(0);
  TORCH_CHECK(M <= 32, "w4a16_gemv: M must be <= 32 (use dequant+GEMM)");
  TORCH_CHECK(K % 64 == 0 && group % 64 == 0 && K % group == 0,
              "w4a16_gemv: 64|K and 64|group required");
  dim3 grid((N + 3) / 4), block(256);
  auto stream = at::hip::getCurrentHIPStream();
  auto launch = [&](auto mt) {
    w4a16_gemv_kernel<decltype(mt)::value><<<grid, block, 0, stream>>>(
        (short*)out.data_ptr(), (const short*)x.data_ptr(),
        (const uint32_t*)qweight.data_ptr(), scales.data_ptr<float>(),
        zeros.data_ptr<float>(), M, N, K, (int)group);
  };
  if (M <= 1) launch(std::integral_constant<int, 1>{});
  else if (M <= 4) launch(std::integral_constant<int, 4>{});
  else if (M <= 8) launch(std::integral_constant<int, 8>{});
  else if (M <= 16) launch(std::integral_constant<int, 16>{});
  else launch(std::integral_constant<int, 32>{});
}

void w4a16_dequant(at::Tensor out, at::Tensor qweight, at::Tensor scales,
                   at::Tensor zeros, int64_t group) {
  const int N = qweight.size(0), K = out.size(1);
  TORCH_CHECK(K % 8 == 0 && K % group == 0);
  const int64_t total = (int64_t)N * (K >> 3);
  dim3 grid((total + 255) / 256), block(256);
  auto stream = at::hip::getCurrentHIPStream();
  w4a16_dequant_kernel<<<grid, block, 0, stream>>>(
      (short*)out.data_ptr(), (const uint32_t*)qweight.data_ptr(),
      scales.data_ptr<float>(), zeros.data_ptr<float>(), N, K, (int)group);
}

}  // namespace kaito
