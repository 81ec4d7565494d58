// Diagnostic kernels: isolate achievable HBM bandwidth for the paged
// decode-attention access pattern (same grid/lane/address walk, no
// softmax/compute) so kernel-side vs memory-side limits can be separated.
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

// MODE 0: pure read; MODE 1: read + q·k dot + p·v accumulate (no
// softmax/cross-lane reduces) — isolates the shfl/softmax cost.
template <int D, int BS, int MODE>
__global__ __launch_bounds__(256, 2) void paged_read_bw_kernel(
    float* __restrict__ out,             // [T*KH] checksums
    const short* __restrict__ k_cache,
    const short* __restrict__ v_cache,
    const int* __restrict__ block_tables,
    const int* __restrict__ seq_lens,
    const int KH, const int max_blocks) {
  constexpr int NV = D / 64;
  const int seq = blockIdx.x;
  const int kvh = blockIdx.y;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int tg = lane >> 3;
  const int dc = lane & 7;
  const int seq_len = seq_lens[seq];
  const int* bt = block_tables + (int64_t)seq * max_blocks;
  const int nchunks = (seq_len + 7) / 8;
  constexpr int G = 4;
  float qreg[G][NV * 8];
  float vacc[G][NV * 8];
#pragma unroll
  for (int g = 0; g < G; g++)
#pragma unroll
    for (int j = 0; j < NV * 8; j++) { qreg[g][j] = 0.01f * j; vacc[g][j] = 0.f; }
  float acc = 0.f;
  for (int c = wave; c < nchunks; c += 4) {
    const int tok = c * 8 + tg;
    const int tok_c = (tok < seq_len) ? tok : (seq_len - 1);
    const int blk = bt[tok_c / BS];
    const int64_t b = (((int64_t)blk * KH + kvh) * BS + (tok_c % BS)) * D + dc * 8;
    if (MODE == 0) {
#pragma unroll
      for (int vv = 0; vv < NV; vv++) {
        short8_t kx = *reinterpret_cast<const short8_t*>(k_cache + b + vv * 64);
        short8_t vx = *reinterpret_cast<const short8_t*>(v_cache + b + vv * 64);
        acc += (float)kx[0] + (float)vx[7];
      }
    } else {
      float kreg[NV * 8], vreg[NV * 8];
#pragma unroll
      for (int vv = 0; vv < NV; vv++) {
        short8_t kx = *reinterpret_cast<const short8_t*>(k_cache + b + vv * 64);
        short8_t vx = *reinterpret_cast<const short8_t*>(v_cache + b + vv * 64);
#pragma unroll
        for (int j = 0; j < 8; j++) {
          kreg[vv * 8 + j] = bf16_to_f32(kx[j]);
          vreg[vv * 8 + j] = bf16_to_f32(vx[j]);
        }
      }
#pragma unroll
      for (int g = 0; g < G; g++) {
        float p = 0.f;
#pragma unroll
        for (int j = 0; j < NV * 8; j++) p += qreg[g][j] * kreg[j];
#pragma unroll
        for (int j = 0; j < NV * 8; j++) vacc[g][j] += p * vreg[j];
      }
    }
  }
  if (MODE == 1)
#pragma unroll
    for (int g = 0; g < G; g++)
#pragma unroll
      for (int j = 0; j < NV * 8; j++) acc += vacc[g][j];
  acc = wave_reduce_sum(acc);
  if (threadIdx.x == 0) out[(int64_t)seq * KH + kvh] = acc;
}

void paged_read_bw(at::Tensor out, at::Tensor k_cache, at::Tensor v_cache,
                   at::Tensor block_tables, at::Tensor seq_lens,
                   int64_t mode) {
  const int T = seq_lens.size(0);
  const int KH = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int D = k_cache.size(3);
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(BS == 16 && (D == 128 || D == 64));
  auto stream = at::hip::getCurrentHIPStream();
#define RB_LAUNCH(D_, M_)                                                     \
  hipLaunchKernelGGL((paged_read_bw_kernel<D_, 16, M_>), dim3(T, KH),          \
      dim3(256), 0, stream, out.data_ptr<float>(),                            \
      (const short*)k_cache.data_ptr(), (const short*)v_cache.data_ptr(),     \
      block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(), KH, max_blocks)
  if (D == 128) { if (mode == 0) RB_LAUNCH(128, 0); else RB_LAUNCH(128, 1); }
  else { if (mode == 0) RB_LAUNCH(64, 0); else RB_LAUNCH(64, 1); }
#undef RB_LAUNCH
}

}  // namespace kaito
