// Paged CONTEXT prefill attention for gfx950: varlen causal attention
// where Q covers only the SUFFIX tokens of each sequence and K/V are read
// from the paged cache (which already holds the full context including the
// suffix). Enables chunked prefill and partial prefix-cache restore.
//
// Same MFMA 16x16x32 structure as prefill_attention.hip (XOR-swizzled K
// tile, transposed V tile, per-wave online softmax); the cooperative
// staging gathers rows through the block table instead of a contiguous
// K/V tensor. Causal rule: q row qi (0-based within the suffix) has
// absolute position kv_len - q_len + qi and attends kv positions <= it.
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

constexpr int CTX_QTILE = 64;
constexpr int CTX_KVT = 32;
constexpr int CTX_PAD = 40;

template <int D, int BS, bool FP8 = false>
__global__ __launch_bounds__(256, 2) void context_attn_kernel(
    short* __restrict__ out,            // [Tq, QH, D]
    const short* __restrict__ q,        // [Tq, QH, D] (suffix tokens)
    const void* __restrict__ k_cache,   // [B, KH, BS, D] bf16 | e4m3
    const void* __restrict__ v_cache,
    const int* __restrict__ tile_seq,
    const int* __restrict__ tile_qbase,   // within-suffix q row base
    const int* __restrict__ cu_seqlens_q, // [batch+1] suffix lens prefix-sum
    const int* __restrict__ kv_lens,      // [batch] total context length
    const int* __restrict__ block_tables, // [batch, max_blocks]
    const float scale, const int QH, const int KH, const int max_blocks,
    const int64_t q_stride,
    const int window,                   // 0 = full causal
    const float* __restrict__ sinks) {  // [QH] or nullptr
  constexpr int KK = D / 32;
  constexpr int DT = D / 16;
  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (QH / KH);
  const int seq = tile_seq[tile];
  const int q0 = tile_qbase[tile];
  const int tok0 = cu_seqlens_q[seq];
  const int q_len = cu_seqlens_q[seq + 1] - tok0;
  const int kv_len = kv_lens[seq];
  const int q_abs0 = kv_len - q_len;     // absolute pos of suffix row 0
  const int* bt = block_tables + (int64_t)seq * max_blocks;

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 4;
  const int lo = lane & 15;

  __shared__ short lds_k[CTX_KVT * D];
  __shared__ short lds_vt[D * CTX_PAD];
  __shared__ short lds_p[4][16 * CTX_PAD];

  const int qrow = q0 + wave * 16 + lo;             // within-suffix
  const bool qvalid = qrow < q_len;
  const int qtok = tok0 + (qvalid ? qrow : q_len - 1);
  bf16x8 qfrag[KK];
#pragma unroll
  for (int kk = 0; kk < KK; kk++)
    qfrag[kk] = *reinterpret_cast<const bf16x8*>(
        q + (int64_t)qtok * q_stride + qh * D + kk * 32 + hi * 8);

  float mrow[4], lrow[4];
  f32x4 ofrag[DT];
#pragma unroll
  for (int r = 0; r < 4; r++) { mrow[r] = -1e30f; lrow[r] = 0.f; }
#pragma unroll
  for (int dt = 0; dt < DT; dt++) ofrag[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  // causal bound: highest absolute q position in this tile
  const int kv_end = min(kv_len, q_abs0 + q0 + CTX_QTILE);
  // sliding window: earliest key visible to this tile's lowest q row
  const int kv_begin = (window > 0 && q_abs0 + q0 - window + 1 > 0)
      ? ((q_abs0 + q0 - window + 1) / CTX_KVT) * CTX_KVT : 0;
  for (int kv0 = kv_begin; kv0 < kv_end; kv0 += CTX_KVT) {
    {
      const int nvec = CTX_KVT * D / 8;
      for (int i = threadIdx.x; i < nvec; i += 256) {
        const int row = (i * 8) / D;
        const int col = (i * 8) % D;
        const int kvp = min(kv0 + row, kv_len - 1);
        const int blk = bt[kvp / BS];
        const int64_t src =
            (((int64_t)blk * KH + kvh) * BS + (kvp % BS)) * D + col;
        short8_t kd, vd;
        if constexpr (FP8) {   // e4m3 cache → bf16 LDS; body unchanged
          float kf[8], vf[8];
          fp8x8_to_f32(reinterpret_cast<const u8x8_t*>(k_cache)[src / 8],
                       kf);
          fp8x8_to_f32(reinterpret_cast<const u8x8_t*>(v_cache)[src / 8],
                       vf);
#pragma unroll
          for (int j = 0; j < 8; j++) {
            kd[j] = f32_to_bf16(kf[j]);
            vd[j] = f32_to_bf16(vf[j]);
          }
        } else {
          kd = *reinterpret_cast<const short8_t*>(
              (const short*)k_cache + src);
          vd = *reinterpret_cast<const short8_t*>(
              (const short*)v_cache + src);
        }
        const int bo = col * 2;
        const int swz = bo ^ ((row & 7) << 4);
        *reinterpret_cast<short8_t*>(&lds_k[row * D + swz / 2]) = kd;
#pragma unroll
        for (int j = 0; j < 8; j++) lds_vt[(col + j) * CTX_PAD + row] = vd[j];
      }
    }
    __syncthreads();

    f32x4 sfrag[2];
#pragma unroll
    for (int kt = 0; kt < 2; kt++) {
      sfrag[kt] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < KK; kk++) {
        const int krow = kt * 16 + lo;
        const int bo = (kk * 32 + hi * 8) * 2;
        const int swz = bo ^ ((krow & 7) << 4);
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(&lds_k[krow * D + swz / 2]);
        sfrag[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], kf,
                                                            sfrag[kt], 0, 0, 0);
      }
    }

    float p[2][4];
#pragma unroll
    for (int kt = 0; kt < 2; kt++)
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const int qr = q0 + wave * 16 + hi * 4 + r;      // within-suffix
        const int q_abs = q_abs0 + qr;
        const int kvp = kv0 + kt * 16 + lo;
        float sv = sfrag[kt][r] * scale;
        if (kvp > q_abs || kvp >= kv_len || qr >= q_len ||
            (window > 0 && kvp <= q_abs - window)) sv = -1e30f;
        p[kt][r] = sv;
      }
    float mnew[4];
#pragma unroll
    for (int r = 0; r < 4; r++) {
      float mx = fmaxf(p[0][r], p[1][r]);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) mx = fmaxf(mx, __shfl_xor(mx, off, 64));
      mnew[r] = fmaxf(mrow[r], mx);
      const float corr = __expf(mrow[r] - mnew[r]);
      lrow[r] *= corr;
#pragma unroll
      for (int dt = 0; dt < DT; dt++) ofrag[dt][r] *= corr;
      mrow[r] = mnew[r];
      float ps = 0.f;
#pragma unroll
      for (int kt = 0; kt < 2; kt++) {
        p[kt][r] = __expf(p[kt][r] - mnew[r]);
        ps += p[kt][r];
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) ps += __shfl_xor(ps, off, 64);
      lrow[r] += ps;
    }

    {
      short* pl = lds_p[wave];
#pragma unroll
      for (int kt = 0; kt < 2; kt++)
#pragma unroll
        for (int r = 0; r < 4; r++)
          pl[(hi * 4 + r) * CTX_PAD + kt * 16 + lo] = f32_to_bf16(p[kt][r]);
    }
    __builtin_amdgcn_s_waitcnt(0);
    bf16x8 pfrag = *reinterpret_cast<const bf16x8*>(
        &lds_p[wave][lo * CTX_PAD + hi * 8]);
#pragma unroll
    for (int dt = 0; dt < DT; dt++) {
      bf16x8 vf = *reinterpret_cast<const bf16x8*>(
          &lds_vt[(dt * 16 + lo) * CTX_PAD + hi * 8]);
      ofrag[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vf, ofrag[dt],
                                                          0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 4; r++) {
    const int qr = q0 + wave * 16 + hi * 4 + r;
    if (qr >= q_len) continue;
    float den = lrow[r];
    if (sinks != nullptr) den += __expf(sinks[qh] - mrow[r]);
    const float inv = 1.f / fmaxf(den, 1e-20f);
    const int64_t obase = ((int64_t)(tok0 + qr) * QH + qh) * D;
#pragma unroll
    for (int dt = 0; dt < DT; dt++)
      out[obase + dt * 16 + lo] = f32_to_bf16(ofrag[dt][r] * inv);
  }
}

void context_attention(at::Tensor out, at::Tensor q, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor tile_seq,
                       at::Tensor tile_qbase, at::Tensor cu_seqlens_q,
                       at::Tensor kv_lens, at::Tensor block_tables,
                       double scale, int64_t window, at::Tensor sinks) {
  const float* sink_ptr = nullptr;
  if (sinks.numel() > 0) {
    TORCH_CHECK(sinks.dtype() == at::kFloat && sinks.is_cuda() &&
                sinks.numel() == q.size(1));
    sink_ptr = sinks.data_ptr<float>();
  }
  TORCH_CHECK(q.is_cuda() && q.dtype() == at::kBFloat16);
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(q.stride(-1) == 1 && q.stride(1) == q.size(2));
  const int QH = q.size(1);
  const int D = q.size(2);
  const int KH = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int ntiles = tile_seq.size(0);
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(BS == 16 && (D == 128 || D == 64 || D == 256),
              "context attn: D 64/128/256");
  auto stream = at::hip::getCurrentHIPStream();
  if (ntiles == 0) return;
#define CTX_LAUNCH(D_)                                                        \
  do {                                                                       \
    if (k_cache.dtype() == at::kByte)                                        \
      hipLaunchKernelGGL((context_attn_kernel<D_, 16, true>),                 \
          dim3(ntiles, QH), dim3(256), 0, stream, (short*)out.data_ptr(),    \
          (const short*)q.data_ptr(), k_cache.data_ptr(),                    \
          v_cache.data_ptr(), tile_seq.data_ptr<int>(),                      \
          tile_qbase.data_ptr<int>(), cu_seqlens_q.data_ptr<int>(),          \
          kv_lens.data_ptr<int>(), block_tables.data_ptr<int>(),             \
          (float)scale, QH, KH, max_blocks, q.stride(0), (int)window,        \
          sink_ptr);                                                         \
    else                                                                     \
      hipLaunchKernelGGL((context_attn_kernel<D_, 16, false>),                \
          dim3(ntiles, QH), dim3(256), 0, stream, (short*)out.data_ptr(),    \
          (const short*)q.data_ptr(), k_cache.data_ptr(),                    \
          v_cache.data_ptr(), tile_seq.data_ptr<int>(),                      \
          tile_qbase.data_ptr<int>(), cu_seqlens_q.data_ptr<int>(),          \
          kv_lens.data_ptr<int>(), block_tables.data_ptr<int>(),             \
          (float)scale, QH, KH, max_blocks, q.stride(0), (int)window,        \
          sink_ptr);                                                         \
  } while (0)
  if (D == 128) CTX_LAUNCH(128);
  else if (D == 256) CTX_LAUNCH(256);
  else CTX_LAUNCH(64);
#undef CTX_LAUNCH
}

}  // namespace kaito
