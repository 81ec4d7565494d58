// Split-phase paged-attention DECODE kernel for gfx950 (MI355X).
//
// Round-2 restructure of paged_attention.hip. The r01 kernel reached
// 4.33 TB/s against a 6.0 TB/s read+dot diagnostic ceiling; the measured
// gap was the per-chunk online-softmax serial chain (compare → exp →
// rescale of the accumulator) riding INSIDE the K/V streaming loop at
// 3 waves/SIMD (profiles/r01_decode_profile.md). This kernel removes
// that chain from the memory loops entirely:
//
//   phase A  stream K only: dot + DC-lane DPP reduce + running MAX (one
//            fmaxf — no exp, no rescale), raw scores parked in LDS.
//   phase B  tiny throughput pass: one exp per (token, head) over the
//            LDS scores with a wave-uniform max, probabilities written
//            back in place.
//   phase C  stream V only: vector LDS broadcast of the G
//            probabilities + pure packed v_pk_fma_f32 accumulate.
//
// Lane layout: DC = D/8 dim-lanes × TPC = 64/DC tokens per wave-step,
// 8 dims (one short8, 16 B) per lane. For D=128 that is 16 dim-lanes ×
// 4 tokens — each token's 16 lanes read 256 B CONTIGUOUS, and the
// per-lane accumulator is acc[G][8] (32 VGPR at G=4), small enough that
// the whole kernel fits the 128-VGPR 4-waves/SIMD occupancy bound with
// no spill (the r01 kernel ran at 3 waves with acc[G][16]).
//
// Memory-level parallelism: each wave owns a CONTIGUOUS chunk span and
// the streaming loops process one KV BLOCK (4 chunks = 16 tokens =
// 4 KiB) per iteration — ONE block-table read per block (vs per-token),
// no per-lane address clamp (a clamped block index keeps every
// intra-block offset valid), and 4 independent 16 B loads in flight per
// lane per iteration.
//
// Long histories are processed in SUPER-CHUNKS of SC chunks per wave;
// carried (M, l, acc) state is rescaled once per super-chunk. All LDS
// score regions are wave-private, so phases need no __syncthreads
// (same-wave LDS RAW is ordered by the compiler's lgkmcnt).
//
// Cache layout: [num_blocks, KV_HEADS, BLOCK_SIZE, HEAD_DIM] bf16
// (same as paged_attention.hip; replaces the vLLM decode attention the
// reference delegates to — SURVEY.md §2.3).
#include "common.h"
#include <cstdlib>
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

// OCC = target workgroups/CU (4 ⇒ 4 waves/SIMD at ≤128 VGPRs).
// Measured (tools/bench_decode_attn.py --sweep-occ): OCC=3 == OCC=4
// exactly and OCC=5 spills — the kernel is NOT wave-count-limited, so
// at OCC=3 there are ~170 VGPRs of free headroom. UB (blocks per
// streaming iteration) spends that headroom on memory-level
// parallelism: UB=2 keeps 8 independent 16 B loads in flight per lane
// (two KV blocks) instead of 4. Variants selectable at runtime via
// KAITO_PA_SP_OCC / KAITO_PA_SP_UB for A/B on real hardware.
template <int D, int G, int BS, bool FP8 = false, int OCC = (G <= 4 ? 4 : 3),
          int UB = 1>
__global__ __launch_bounds__(256, OCC)
void paged_attention_sp_kernel(
    short* __restrict__ out,             // [T, QH, D] bf16
    const short* __restrict__ q,         // [T, QH, D] bf16
    const void* __restrict__ k_cache,    // [B, KH, BS, D] bf16 | e4m3
    const void* __restrict__ v_cache,
    const int* __restrict__ block_tables,// [T, max_blocks]
    const int* __restrict__ seq_lens,    // [T]
    const float scale, const int KH, const int max_blocks,
    const int64_t q_stride,
    const int window,                    // 0 = full attention
    const float* __restrict__ sinks) {   // [QH] or nullptr (gpt-oss)
  constexpr int DC = D / 8;       // dim-chunk lanes (16 for D=128)
  constexpr int TPC = 64 / DC;    // tokens per wave-step (4 for D=128)
  constexpr int CPB = BS / TPC;   // chunks per KV block (4 for D=128)
  constexpr int NW = 4;           // waves per workgroup
  constexpr int G4 = G <= 2 ? 2 : (G <= 4 ? 4 : 8);  // padded score row
  constexpr int SC = (G <= 4 ? 256 : 128) / TPC;  // chunks per super-chunk
  static_assert(SC % CPB == 0, "super-chunk must align to KV blocks");
  const int seq = blockIdx.x;
  const int kvh = blockIdx.y;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int tg = lane / DC;       // token slot in chunk (0..TPC-1)
  const int dc = lane % DC;       // dim chunk (8 dims each)
  const int seq_len = seq_lens[seq];
  const int QH = KH * G;

  const int* bt = block_tables + (int64_t)seq * max_blocks;

  // scores/probabilities: wave-private [SC][TPC tokens][G4] f32
  __shared__ float s_p[NW][SC][TPC][G4];

  float M[G], lpart[G], acc[G][8];
#pragma unroll
  for (int g = 0; g < G; g++) {
    M[g] = -1e30f; lpart[g] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; j++) acc[g][j] = 0.f;
  }

  // sliding window: only tokens in [start, seq_len) are attended; the
  // chunk range is aligned DOWN to the containing KV block (tokens
  // before `start` in that block are masked by `valid`)
  const int start = (window > 0 && seq_len > window) ? seq_len - window : 0;
  const int c_lo = (start / BS) * CPB;
  const int nchunks = (seq_len + TPC - 1) / TPC;
  const int last_blk = (seq_len - 1) / BS;
  // contiguous block-aligned chunk span per wave
  const int total_c = nchunks - c_lo;
  const int n_per = ((total_c + NW * CPB - 1) / (NW * CPB)) * CPB;
  const int span0 = c_lo + wave * n_per;
  const int n_i = max(0, min(nchunks, span0 + n_per) - span0);

  typedef __bf16 bf16x2_t __attribute__((ext_vector_type(2)));

  for (int s0 = 0; s0 < n_i; s0 += SC) {
    const int n_ii = min(SC, n_i - s0);

    // ---------------- phase A: stream K, raw scores to LDS -------------
    // Q is (re)loaded per super-chunk so its registers die before
    // phase C (Q and the accumulator never coexist). Lane dc holds dims
    // dc*8..dc*8+8 — one contiguous 16B load per (head, lane). The fp8
    // cache path pre-converts Q to f32 (the dot runs on fmaf over the
    // hardware fp8→f32 pair converts instead of v_dot2 bf16).
    short8_t qreg[G];
    float qf[FP8 ? G : 1][FP8 ? 8 : 1];
#pragma unroll
    for (int g = 0; g < G; g++) {
      qreg[g] = *reinterpret_cast<const short8_t*>(
          q + (int64_t)seq * q_stride + (kvh * G + g) * D + dc * 8);
      if constexpr (FP8) {
#pragma unroll
        for (int j = 0; j < 8; j++) qf[g][j] = bf16_to_f32(qreg[g][j]);
      }
    }
    float mloc[G];
#pragma unroll
    for (int g = 0; g < G; g++) mloc[g] = -1e30f;
    for (int ii0 = 0; ii0 < n_ii; ii0 += UB * CPB) {
      const int c0 = span0 + s0 + ii0;           // multiple of CPB
      // UB block bases; clamped indices keep every offset valid
      int64_t gb[UB];
#pragma unroll
      for (int b = 0; b < UB; b++) {
        const int blk = bt[min(c0 / CPB + b, last_blk)];
        gb[b] = (((int64_t)blk * KH + kvh) * BS + tg) * D + dc * 8;
      }
      short8_t kd[FP8 ? 1 : UB * CPB];
      u8x8_t kd8[FP8 ? UB * CPB : 1];
#pragma unroll
      for (int u = 0; u < UB * CPB; u++) {
        if constexpr (FP8)
          kd8[u] = reinterpret_cast<const u8x8_t*>(k_cache)[
              (gb[u / CPB] + (u % CPB) * (TPC * D)) / 8];
        else
          kd[u] = *reinterpret_cast<const short8_t*>(
              (const short*)k_cache + gb[u / CPB] + (u % CPB) * (TPC * D));
      }
      const int nu = min(UB * CPB, n_ii - ii0);
      // compile-time trip count: a runtime-bounded loop would make
      // kd[u] a runtime-indexed register array → scratch (rule #20)
#pragma unroll
      for (int u = 0; u < UB * CPB; u++) {
        if (u >= nu) break;
        const int tok = (c0 + u) * TPC + tg;
        const bool valid = tok >= start && tok < seq_len;
        float s[G];
        float kf[FP8 ? 8 : 1];
        if constexpr (FP8) fp8x8_to_f32(kd8[u], kf);
#pragma unroll
        for (int g = 0; g < G; g++) {
          float p = 0.f;
          if constexpr (FP8) {
#pragma unroll
            for (int j = 0; j < 8; j++) p = fmaf(qf[g][j], kf[j], p);
          } else {
            const bf16x2_t* kp = reinterpret_cast<const bf16x2_t*>(&kd[u]);
            const bf16x2_t* qp = reinterpret_cast<const bf16x2_t*>(&qreg[g]);
#pragma unroll
            for (int jj = 0; jj < 4; jj++)
              p = __builtin_amdgcn_fdot2_f32_bf16(qp[jj], kp[jj], p, false);
          }
          p = group_reduce_sum<DC>(p) * scale;   // dot over DC dim-lanes
          s[g] = valid ? p : -1e30f;
          mloc[g] = fmaxf(mloc[g], s[g]);
        }
        // park raw scores: lane with dc<G writes s[dc] (static unroll —
        // a runtime-indexed register array would spill to scratch)
        if (dc < G) {
          float sval = s[0];
#pragma unroll
          for (int g = 1; g < G; g++) if (dc == g) sval = s[g];
          s_p[wave][ii0 + u][tg][dc] = sval;
        }
      }
    }

    // ---- wave-uniform max; rescale carried state once per super ----
#pragma unroll
    for (int g = 0; g < G; g++) {
      float gm = mloc[g];               // uniform over dc; varies over tg
#pragma unroll
      for (int off = DC; off < 64; off <<= 1)
        gm = fmaxf(gm, __shfl_xor(gm, off, 64));
      const float Mnew = fmaxf(M[g], gm);
      if (Mnew > M[g]) {
        const float corr = __expf(M[g] - Mnew);   // 0 on first super
        lpart[g] *= corr;
        float2_t* a2 = reinterpret_cast<float2_t*>(acc[g]);
#pragma unroll
        for (int j = 0; j < 4; j++) a2[j] *= corr;
        M[g] = Mnew;
      }
    }

    // ---------------- phase B: exp over LDS scores (throughput) --------
    const int entries = n_ii * TPC;     // (ii, tg) pairs
#pragma unroll
    for (int g = 0; g < G; g++) {
      for (int idx = lane; idx < entries; idx += 64) {
        float* slot = &s_p[wave][0][0][0] + idx * G4 + g;
        const float sv = *slot;
        const float p = sv > -1e29f ? __expf(sv - M[g]) : 0.f;
        lpart[g] += p;
        *slot = p;
      }
    }

    // ---------------- phase C: stream V, pure FMA accumulate -----------
    for (int ii0 = 0; ii0 < n_ii; ii0 += UB * CPB) {
      const int c0 = span0 + s0 + ii0;
      int64_t gb[UB];
#pragma unroll
      for (int b = 0; b < UB; b++) {
        const int blk = bt[min(c0 / CPB + b, last_blk)];
        gb[b] = (((int64_t)blk * KH + kvh) * BS + tg) * D + dc * 8;
      }
      short8_t vd[FP8 ? 1 : UB * CPB];
      u8x8_t vd8[FP8 ? UB * CPB : 1];
#pragma unroll
      for (int u = 0; u < UB * CPB; u++) {
        if constexpr (FP8)
          vd8[u] = reinterpret_cast<const u8x8_t*>(v_cache)[
              (gb[u / CPB] + (u % CPB) * (TPC * D)) / 8];
        else
          vd[u] = *reinterpret_cast<const short8_t*>(
              (const short*)v_cache + gb[u / CPB] + (u % CPB) * (TPC * D));
      }
      const int nu = min(UB * CPB, n_ii - ii0);
#pragma unroll
      for (int u = 0; u < UB * CPB; u++) {
        if (u >= nu) break;
        // G probabilities for this token: one vector LDS broadcast
        float pg[G];
        if constexpr (G4 == 2) {
          float2_t pv = *reinterpret_cast<float2_t*>(
              &s_p[wave][ii0 + u][tg][0]);
#pragma unroll
          for (int g = 0; g < G; g++) pg[g] = pv[g];
        } else if constexpr (G4 == 4) {
          float4_t pv = *reinterpret_cast<float4_t*>(
              &s_p[wave][ii0 + u][tg][0]);
#pragma unroll
          for (int g = 0; g < G; g++) pg[g] = pv[g];
        } else {
          float4_t pv0 = *reinterpret_cast<float4_t*>(
              &s_p[wave][ii0 + u][tg][0]);
          float4_t pv1 = *reinterpret_cast<float4_t*>(
              &s_p[wave][ii0 + u][tg][4]);
#pragma unroll
          for (int g = 0; g < G; g++) pg[g] = g < 4 ? pv0[g] : pv1[g - 4];
        }
        float vreg[8];
        if constexpr (FP8) {
          fp8x8_to_f32(vd8[u], vreg);
        } else {
#pragma unroll
          for (int j = 0; j < 8; j++) vreg[j] = bf16_to_f32(vd[u][j]);
        }
        const float2_t* v2 = reinterpret_cast<const float2_t*>(vreg);
#pragma unroll
        for (int g = 0; g < G; g++) {
          float2_t* a2 = reinterpret_cast<float2_t*>(acc[g]);
#pragma unroll
          for (int j = 0; j < 4; j++) a2[j] += pg[g] * v2[j];
        }
      }
    }
  }

  // ---- merge over the TPC token-group lanes (M already wave-uniform) --
#pragma unroll
  for (int g = 0; g < G; g++) {
    lpart[g] = wave_reduce_sum(lpart[g]);
#pragma unroll
    for (int j = 0; j < 8; j++)
#pragma unroll
      for (int off = DC; off < 64; off <<= 1)
        acc[g][j] += __shfl_xor(acc[g][j], off, 64);
  }

  // ---- cross-wave combine via LDS (same structure as the r01 kernel) --
  __shared__ float s_acc[NW][G][D];
  __shared__ float s_ml[NW][G][2];
  if (tg == 0) {
#pragma unroll
    for (int g = 0; g < G; g++) {
#pragma unroll
      for (int j = 0; j < 8; j++)
        s_acc[wave][g][dc * 8 + j] = acc[g][j];
      if (dc == 0) { s_ml[wave][g][0] = M[g]; s_ml[wave][g][1] = lpart[g]; }
    }
  }
  __syncthreads();

  for (int idx = threadIdx.x; idx < G * D; idx += 256) {
    const int g = idx / D;
    const int d = idx % D;
    float gm = -1e30f;
#pragma unroll
    for (int w = 0; w < NW; w++) gm = fmaxf(gm, s_ml[w][g][0]);
    float num = 0.f, den = 0.f;
    float sink = sinks != nullptr ? sinks[kvh * G + g] : -1e30f;
    gm = fmaxf(gm, sink);       // keep exp() bounded when sink dominates
#pragma unroll
    for (int w = 0; w < NW; w++) {
      const float e = __expf(s_ml[w][g][0] - gm);
      num += e * s_acc[w][g][d];
      den += e * s_ml[w][g][1];
    }
    if (sinks != nullptr)       // learned sink absorbs softmax mass
      den += __expf(sink - gm);
    out[((int64_t)seq * QH + kvh * G + g) * D + d] =
        f32_to_bf16(num / fmaxf(den, 1e-20f));
  }
}

// KAITO_PA_SP_OCC=3|4|5 selects the occupancy variant (default: the
// template default, i.e. 4 for G<=4). Only instantiated for bf16
// caches; the fp8 path always uses the default.
static int pa_sp_occ_env() {
  static int occ = []() {
    const char* e = getenv("KAITO_PA_SP_OCC");
    return e ? atoi(e) : 0;
  }();
  return occ;
}

static int pa_sp_ub_env() {
  static int ub = []() {
    const char* e = getenv("KAITO_PA_SP_UB");
    return e ? atoi(e) : 0;
  }();
  return ub;
}

#define PA_SP_ARGS                                                            \
  dim3(T, KH), dim3(256), 0, stream, (short*)out.data_ptr(),                  \
      (const short*)query.data_ptr(), k_cache.data_ptr(),                     \
      v_cache.data_ptr(), block_tables.data_ptr<int>(),                       \
      seq_lens.data_ptr<int>(), (float)scale, KH, max_blocks,                 \
      query.stride(0), (int)window, sink_ptr

#define PA_SP_LAUNCH(D_, G_)                                                   \
  do {                                                                        \
    if (fp8)                                                                  \
      hipLaunchKernelGGL((paged_attention_sp_kernel<D_, G_, 16, true>),        \
          PA_SP_ARGS);                                                        \
    else if (G_ <= 4 && pa_sp_ub_env() == 2)                                  \
      hipLaunchKernelGGL((paged_attention_sp_kernel<D_, G_, 16, false, 3, 2>), \
          PA_SP_ARGS);                                                        \
    else if (G_ <= 4 && pa_sp_occ_env() == 5)                                 \
      hipLaunchKernelGGL((paged_attention_sp_kernel<D_, G_, 16, false, 5>),    \
          PA_SP_ARGS);                                                        \
    else if (G_ <= 4 && pa_sp_occ_env() == 3)                                 \
      hipLaunchKernelGGL((paged_attention_sp_kernel<D_, G_, 16, false, 3>),    \
          PA_SP_ARGS);                                                        \
    else                                                                      \
      hipLaunchKernelGGL((paged_attention_sp_kernel<D_, G_, 16, false>),       \
          PA_SP_ARGS);                                                        \
  } while (0)

void paged_attention_sp(at::Tensor out, at::Tensor query, at::Tensor k_cache,
                        at::Tensor v_cache, at::Tensor block_tables,
                        at::Tensor seq_lens, double scale, int64_t window,
                        at::Tensor sinks) {
  const float* sink_ptr = nullptr;
  if (sinks.numel() > 0) {
    TORCH_CHECK(sinks.dtype() == at::kFloat && sinks.is_cuda() &&
                sinks.numel() == query.size(1));
    sink_ptr = sinks.data_ptr<float>();
  }
  TORCH_CHECK(query.is_cuda() && query.dtype() == at::kBFloat16);
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(query.stride(-1) == 1 && query.stride(1) == query.size(2),
              "query must be [T, QH, D] with contiguous heads");
  TORCH_CHECK(block_tables.dtype() == at::kInt && seq_lens.dtype() == at::kInt);
  const bool fp8 = k_cache.dtype() == at::kByte;   // OCP e4m3 KV cache
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
      case 1: PA_SP_LAUNCH(128, 1); break;
      case 2: PA_SP_LAUNCH(128, 2); break;
      case 3: PA_SP_LAUNCH(128, 3); break;
      case 4: PA_SP_LAUNCH(128, 4); break;
      case 5: PA_SP_LAUNCH(128, 5); break;
      case 6: PA_SP_LAUNCH(128, 6); break;
      case 7: PA_SP_LAUNCH(128, 7); break;
      case 8: PA_SP_LAUNCH(128, 8); break;
      default: TORCH_CHECK(false, "unsupported GQA group ", G);
    }
  } else if (D == 64) {
    switch (G) {
      case 1: PA_SP_LAUNCH(64, 1); break;
      case 2: PA_SP_LAUNCH(64, 2); break;
      case 4: PA_SP_LAUNCH(64, 4); break;
      case 8: PA_SP_LAUNCH(64, 8); break;
      default: TORCH_CHECK(false, "unsupported GQA group ", G);
    }
  } else if (D == 256) {   // gemma-3-4b class
    switch (G) {
      case 1: PA_SP_LAUNCH(256, 1); break;
      case 2: PA_SP_LAUNCH(256, 2); break;
      case 4: PA_SP_LAUNCH(256, 4); break;
      default: TORCH_CHECK(false, "unsupported GQA group ", G);
    }
  } else {
    TORCH_CHECK(false, "unsupported head_dim ", D);
  }
}

}  // namespace kaito
