// Paged-attention DECODE kernel for gfx950 (MI355X).
//
// Replaces the vLLM decode attention the reference delegates to
// (SURVEY.md §2.3 row "Paged-attention prefill/decode"). MI355X-first
// design, NOT a CUDA port:
//  - memory-bound regime: one (seq, kv_head) workgroup streams the whole
//    K/V history once and shares it across the GQA group's Q heads.
//  - wave = 64 lanes: lane (tg,dc) = (lane>>3, lane&7) → 8 tokens/wave-step,
//    16 dims/lane → 32 B/lane vectorized bf16 loads (2KB contiguous per
//    wave-step from one [block, kv_head] slab).
//  - flash-decoding online softmax kept PER-LANE inside the K/V stream
//    (single cross-lane merge at the end; in-loop cross-lane work is just
//    the 8-lane dot reduce, done as DPP adds), LDS cross-wave combine.
//
// Cache layout: [num_blocks, KV_HEADS, BLOCK_SIZE, HEAD_DIM] bf16.
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

template <int D, int G, int BS>
__global__ __launch_bounds__(256, G <= 4 ? 3 : 2) void paged_attention_kernel(
    short* __restrict__ out,             // [T, QH, D] bf16
    const short* __restrict__ q,         // [T, QH, D] bf16
    const short* __restrict__ k_cache,   // [B, KH, BS, D]
    const short* __restrict__ v_cache,
    const int* __restrict__ block_tables,// [T, max_blocks]
    const int* __restrict__ seq_lens,    // [T]
    const float scale, const int KH, const int max_blocks,
    const int64_t q_stride) {
  constexpr int DL = D / 8;       // dims per lane (16 for D=128)
  constexpr int NW = 4;           // waves per workgroup
  const int seq = blockIdx.x;
  const int kvh = blockIdx.y;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int tg = lane >> 3;       // token slot in chunk (0..7)
  const int dc = lane & 7;        // dim chunk (0..7)
  const int seq_len = seq_lens[seq];
  const int QH = KH * G;

  const int* bt = block_tables + (int64_t)seq * max_blocks;

  // ---- preload Q for this kv-head's group (f32) ----
  // lane-dim mapping: vector vv holds dims vv*64 + dc*8 + [0..7] so each
  // 16B lane-load is CONTIGUOUS across the 8 dc-lanes (dense 128B/token
  // transactions; the naive dc*DL mapping strides 32B between lanes and
  // halves effective HBM bandwidth).
  // Q stays PACKED bf16: the QK dot runs on v_dot2c_f32_bf16 (2-wide
  // bf16 dot-accumulate, one VALU op per dim pair — vs convert+FMA at 2
  // ops/dim), and packed Q halves its VGPR footprint. bf16xbf16 products
  // are exact in the f32 accumulator, so numerics match the f32 path up
  // to summation order.
  short8_t qreg[G][DL / 8];
#pragma unroll
  for (int g = 0; g < G; g++) {
#pragma unroll
    for (int vv = 0; vv < DL / 8; vv++) {
      qreg[g][vv] = *reinterpret_cast<const short8_t*>(
          q + (int64_t)seq * q_stride + (kvh * G + g) * D + vv * 64 + dc * 8);
    }
  }

  float m[G], l[G], acc[G][DL];
#pragma unroll
  for (int g = 0; g < G; g++) {
    m[g] = -1e30f; l[g] = 0.f;
#pragma unroll
    for (int j = 0; j < DL; j++) acc[g][j] = 0.f;
  }

  const int nchunks = (seq_len + 7) / 8;  // 8 tokens per wave-step

  // software-pipelined K/V streaming: issue chunk c+NW's loads (raw bf16
  // registers) before computing chunk c, so HBM latency hides under the
  // dot/softmax VALU work (guide T14 async-split).
  auto chunk_base = [&](int c) -> int64_t {
    const int tok = c * 8 + tg;
    const int tok_c = (tok < seq_len) ? tok : (seq_len - 1);
    const int blk = bt[tok_c / BS];
    return (((int64_t)blk * KH + kvh) * BS + (tok_c % BS)) * D + dc * 8;
  };
  constexpr int NV = DL / 8;              // short8 vectors per lane (2 for D=128)

  // minimal-register variant: no explicit prefetch regs — loads are issued
  // at the top of each iteration and occupancy (3 waves/SIMD) hides the
  // latency across waves.
  auto load_chunk = [&](int c, short8_t (&kd)[NV], short8_t (&vd)[NV]) {
    const int64_t b = chunk_base(c);
#pragma unroll
    for (int vv = 0; vv < NV; vv++) {
      kd[vv] = *reinterpret_cast<const short8_t*>(k_cache + b + vv * 64);
      vd[vv] = *reinterpret_cast<const short8_t*>(v_cache + b + vv * 64);
    }
  };

  // one chunk's update: dot → PER-LANE online softmax → accumulate.
  // The dot's 8-lane reduce is the ONLY cross-lane op in the loop; each
  // lane tracks its own running (m, l, acc) for the tokens it owns and the
  // lane-local partials are merged ONCE after the loop (measured: with
  // per-chunk max/sum butterflies the kernel ran at 3.4 TB/s vs the
  // 6.0 TB/s dot-only ceiling — shuffles were the entire gap).
  auto process = [&](int c, short8_t (&kd)[NV], short8_t (&vd)[NV]) {
    const bool valid_c = c < nchunks;
    if (!valid_c) return;
    const bool valid = (c * 8 + tg) < seq_len;
    typedef __bf16 bf16x2_t __attribute__((ext_vector_type(2)));
    float s[G];
#pragma unroll
    for (int g = 0; g < G; g++) {
      float p = 0.f;
#pragma unroll
      for (int vv = 0; vv < NV; vv++) {
        const bf16x2_t* kp = reinterpret_cast<const bf16x2_t*>(&kd[vv]);
        const bf16x2_t* qp = reinterpret_cast<const bf16x2_t*>(&qreg[g][vv]);
#pragma unroll
        for (int jj = 0; jj < 4; jj++)
          p = __builtin_amdgcn_fdot2_f32_bf16(qp[jj], kp[jj], p, false);
      }
      p = group_reduce_sum<8>(p) * scale;   // dot over 8 dc-lanes
      s[g] = valid ? p : -1e30f;
    }
    float kreg[DL];
#pragma unroll
    for (int vv = 0; vv < NV; vv++)
#pragma unroll
      for (int j = 0; j < 8; j++) kreg[vv * 8 + j] = bf16_to_f32(vd[vv][j]);
    if (!valid) return;           // masked lanes accumulate nothing
#pragma unroll
    for (int g = 0; g < G; g++) {
      float p;
      // float2 views: the rescale and accumulate lower to packed
      // v_pk_mul_f32 / v_pk_fma_f32 (2 lanes per VALU op)
      float2_t* a2 = reinterpret_cast<float2_t*>(acc[g]);
      if (s[g] > m[g]) {          // lane-local rescale (no cross-lane max)
        const float corr = __expf(m[g] - s[g]);
        l[g] *= corr;
#pragma unroll
        for (int j = 0; j < DL / 2; j++) a2[j] *= corr;
        m[g] = s[g];
        p = 1.0f;
      } else {
        p = __expf(s[g] - m[g]);
      }
      l[g] += p;
      const float2_t* k2 = reinterpret_cast<const float2_t*>(kreg);
#pragma unroll
      for (int j = 0; j < DL / 2; j++) a2[j] += p * k2[j];
    }
  };

  // (re-tried a 2-deep load pipeline post-dot2: the double buffer pushes
  // the kernel to 168 VGPR with 156 B/lane of spill inside the 3-wave
  // bound — worse than just letting occupancy hide the latency)
  for (int c = wave; c < nchunks; c += NW) {
    short8_t kd[NV], vd[NV];
    load_chunk(c, kd, vd);
    process(c, kd, vd);
  }

  // ---- merge lane-local softmax state (once, not per chunk) ----
  // m is uniform across the 8 dc-lanes of a token, so reducing over the
  // stride-8 (token-group) lanes yields the wave max/sum.
#pragma unroll
  for (int g = 0; g < G; g++) {
    float gm = m[g];
#pragma unroll
    for (int off = 8; off < 64; off <<= 1)
      gm = fmaxf(gm, __shfl_xor(gm, off, 64));
    const float factor = __expf(m[g] - gm);   // 0 for empty lanes
    m[g] = gm;
    l[g] *= factor;
#pragma unroll
    for (int off = 8; off < 64; off <<= 1)
      l[g] += __shfl_xor(l[g], off, 64);
#pragma unroll
    for (int j = 0; j < DL; j++) acc[g][j] *= factor;
  }

  // ---- reduce acc across the 8 token-group lanes (same dc) ----
#pragma unroll
  for (int g = 0; g < G; g++)
#pragma unroll
    for (int j = 0; j < DL; j++)
#pragma unroll
      for (int off = 8; off < 64; off <<= 1)
        acc[g][j] += __shfl_xor(acc[g][j], off, 64);

  // ---- cross-wave combine via LDS ----
  __shared__ float s_acc[NW][G][D];
  __shared__ float s_ml[NW][G][2];
  if (tg == 0) {
#pragma unroll
    for (int g = 0; g < G; g++) {
#pragma unroll
      for (int vv = 0; vv < NV; vv++)
#pragma unroll
        for (int j = 0; j < 8; j++)
          s_acc[wave][g][vv * 64 + dc * 8 + j] = acc[g][vv * 8 + j];
      if (dc == 0) { s_ml[wave][g][0] = m[g]; s_ml[wave][g][1] = l[g]; }
    }
  }
  __syncthreads();

  // 256 threads: thread handles (g, d) pairs strided.
  for (int idx = threadIdx.x; idx < G * D; idx += 256) {
    const int g = idx / D;
    const int d = idx % D;
    float gm = -1e30f;
#pragma unroll
    for (int w = 0; w < NW; w++) gm = fmaxf(gm, s_ml[w][g][0]);
    float num = 0.f, den = 0.f;
#pragma unroll
    for (int w = 0; w < NW; w++) {
      const float e = __expf(s_ml[w][g][0] - gm);
      num += e * s_acc[w][g][d];
      den += e * s_ml[w][g][1];
    }
    out[((int64_t)seq * QH + kvh * G + g) * D + d] =
        f32_to_bf16(num / fmaxf(den, 1e-20f));
  }
}

#define PA_LAUNCH(D_, G_)                                                      \
  hipLaunchKernelGGL((paged_attention_kernel<D_, G_, 16>), dim3(T, KH),        \
      dim3(256), 0, stream, (short*)out.data_ptr(),                           \
      (const short*)query.data_ptr(), (const short*)k_cache.data_ptr(),       \
      (const short*)v_cache.data_ptr(), block_tables.data_ptr<int>(),         \
      seq_lens.data_ptr<int>(), (float)scale, KH, max_blocks,                 \
      query.stride(0))

void paged_attention(at::Tensor out, at::Tensor query, at::Tensor k_cache,
                     at::Tensor v_cache, at::Tensor block_tables,
                     at::Tensor seq_lens, double scale) {
  TORCH_CHECK(query.is_cuda() && query.dtype() == at::kBFloat16);
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(query.stride(-1) == 1 && query.stride(1) == query.size(2),
              "query must be [T, QH, D] with contiguous heads");
  TORCH_CHECK(block_tables.dtype() == at::kInt && seq_lens.dtype() == at::kInt);
  const int T = query.size(0);
  const int QH = query.size(1);
  const int D = query.size(2);
  const int KH = k_cache.size(1);
  const int BS = k_cache.size(2);
  TORCH_CHECK(BS == 16, "block_size must be 16");
  TORCH_CHECK(QH % KH == 0);
  const int G = QH / KH;
  const int max_blocks = block_tables.size(1);
  auto stream = at::hip::getCurrentHIPStream();
  if (T == 0) return;
  if (D == 128) {
    switch (G) {
      case 1: PA_LAUNCH(128, 1); break;
      case 2: PA_LAUNCH(128, 2); break;
      case 3: PA_LAUNCH(128, 3); break;
      case 4: PA_LAUNCH(128, 4); break;
      case 5: PA_LAUNCH(128, 5); break;
      case 6: PA_LAUNCH(128, 6); break;
      case 7: PA_LAUNCH(128, 7); break;
      case 8: PA_LAUNCH(128, 8); break;
      default: TORCH_CHECK(false, "unsupported GQA group ", G);
    }
  } else if (D == 64) {
    switch (G) {
      case 1: PA_LAUNCH(64, 1); break;
      case 2: PA_LAUNCH(64, 2); break;
      case 4: PA_LAUNCH(64, 4); break;
      case 8: PA_LAUNCH(64, 8); break;
      default: TORCH_CHECK(false, "unsupported GQA group ", G);
    }
  } else {
    TORCH_CHECK(false, "unsupported head_dim ", D);
  }
}

}  // namespace kaito
