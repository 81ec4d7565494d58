// MLA (multi-head latent attention) DECODE kernel for gfx950 (MI355X).
//
// DeepSeek V2/V3/R1 absorbed decode (models/mla.py): one compressed
// latent row per cached token — c_kv[R=512] ‖ k_rope[P=64], 576 bf16 —
// shared by EVERY head. Replaces the vLLM MLA backend the reference
// delegates to for the deepseek presets (SURVEY.md §2.3,
// supported_models.yaml deepseek-v3-0324 / deepseek-r1-0528).
//
// Shape: scores[h][t] = q_latent[h]·cache[t] (the full 576-dot — rope
// term included by construction), out[h] = Σ_t softmax·c_kv[t][0:512].
// Compared with GQA decode (paged_attention_sp.hip) the arithmetic
// intensity is ~16 heads × 2 dots per byte, so the kernel is organized
// around HBM-byte REUSE, not pure streaming:
//
//   * grid (seq, head-tile of 16); each 256-thread workgroup stages one
//     16-token KV block (18 KB) into LDS cooperatively, then all 16
//     heads consume it — every HBM byte is read ONCE per tile. At
//     deepseek-v3 TP=8 each rank holds exactly 16 heads = one tile;
//     V2-Lite (16 heads) is one tile at TP=1.
//   * score phase: lane = (head&3)<<4 | slice, 4 heads per wave; each
//     lane dots a 36-dim slice (9 × b64 LDS reads, bf16 v_dot2) and the
//     16 slice-lanes shfl-reduce. The 4 same-slice lanes read the same
//     LDS address → broadcast, conflict-free.
//   * online softmax per block: per-head running (M, l) and the
//     accumulator rescale, amortized over 16 heads × 512 dims (the
//     per-block rescale is ~6% of the accumulate FLOPs — cheap here,
//     unlike the GQA kernel where it forced the split-phase design).
//   * accumulate phase: thread (head = tid>>4, dim-slice = tid&15)
//     carries 32 f32 accumulator registers; p broadcasts from LDS.
//   * all LDS handoffs except the staging barrier are intra-wave
//     (score→exp→accumulate stay inside the head's own wave), so each
//     block iteration needs only TWO __syncthreads.
//
// Cache layout: [num_blocks, BLOCK_SIZE=16, R+P] bf16 (the aliased
// (c, c) pair from model_runner.profile_and_allocate_kv).
//
// Known limit: grid = T × ceil(H/16) — below ~512 concurrent sequences
// the launch underfills 256 CUs; a flash-decoding token-split tier is
// the designated fix (docs/ROADMAP.md).
#include "common.h"
#include <cstdlib>
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

typedef short short4_t __attribute__((ext_vector_type(4)));

// OCC: workgroups/CU. LDS (19.3 KB/wg) allows 8; VGPRs decide the
// real bound — selectable via KAITO_MLA_OCC for A/B on hardware.
// DB: software double-buffer THROUGH REGISTERS — the next block's
// global loads (5 x b128 per thread) issue before the current block's
// compute, hiding the HBM latency the single-buffer stage exposes at
// every __syncthreads; LDS footprint unchanged (KAITO_MLA_DB).
template <int R, int P, int BS, int OCC = 4, bool DB = false>
__global__ __launch_bounds__(256, OCC)
void mla_decode_kernel(
    short* __restrict__ out,            // [T, H, R] bf16
    const short* __restrict__ q,        // [T, H, R+P] bf16
    const short* __restrict__ cache,    // [NB, BS, R+P] bf16
    const int* __restrict__ block_tables,  // [T, max_blocks]
    const int* __restrict__ seq_lens,   // [T]
    const float scale, const int H, const int max_blocks) {
  constexpr int DT = R + P;             // 576
  constexpr int SL = DT / 16;           // 36 dims per score slice
  constexpr int AD = R / 16;            // 32 dims per accum slice
  constexpr int HT = 16;                // heads per workgroup
  static_assert(DT % 16 == 0 && R % 16 == 0 && SL % 12 == 0
                && AD % 8 == 0);

  const int seq = blockIdx.x;
  const int h0 = blockIdx.y * HT;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int seq_len = seq_lens[seq];
  const int nblocks = (seq_len + BS - 1) / BS;
  const int* bt = block_tables + (int64_t)seq * max_blocks;

  // score-phase role: 4 heads per wave, 16 dim-slices per head
  const int sc_head = wave * 4 + (lane >> 4);     // 0..15 == tid>>4
  const int slice = lane & 15;
  // accum/exp-phase role (same head): dim/token slice
  const int ac_h = tid >> 4;                      // == sc_head
  const int ac_s = tid & 15;

  __shared__ short s_kv[BS * DT];                 // 18 KB staged block
  __shared__ float s_p[BS][HT];                   // scores→probs

  // per-head online-softmax state, held REDUNDANTLY in registers by all
  // 16 lanes of the head's thread group (every lane derives identical
  // values from the same shfl reductions) — no cross-lane LDS handoff
  float M = -1e30f, l = 0.f;

  // q slice (36 bf16 = 9 x b64) for the score phase; zero for padded
  // heads so their scores stay finite (their output is never written)
  short4_t qv[SL / 4];
  const bool head_ok = h0 + sc_head < H;
  {
    // clamp the ROW (not just the value) so a potentially if-converted
    // load never dereferences past the q tensor for padded head tiles
    const int hq = min(h0 + sc_head, H - 1);
    const short* qp = q + ((int64_t)seq * H + hq) * DT + slice * SL;
#pragma unroll
    for (int j = 0; j < SL / 4; j++) {
      qv[j] = *reinterpret_cast<const short4_t*>(qp + j * 4);
      if (!head_ok) qv[j] = short4_t{0, 0, 0, 0};
    }
  }

  float acc[AD];
#pragma unroll
  for (int j = 0; j < AD; j++) acc[j] = 0.f;

  typedef __bf16 bf16x2_t __attribute__((ext_vector_type(2)));

  constexpr int NCH = (BS * DT / 8 + 255) / 256;   // b128 chunks/thread
  short8_t pf[DB ? NCH : 1];
  if constexpr (DB) {   // prologue: block 0 loads in flight
    const short8_t* src = reinterpret_cast<const short8_t*>(
        cache + (int64_t)bt[0] * (BS * DT));
#pragma unroll
    for (int c = 0; c < NCH; c++) {
      const int idx = c * 256 + tid;
      if (idx < BS * DT / 8) pf[c] = src[idx];
    }
  }

  for (int b = 0; b < nblocks; b++) {
    // ---- stage one KV block: 1152 x b128, 256 threads ----
    if constexpr (DB) {
      short8_t* dst = reinterpret_cast<short8_t*>(s_kv);
#pragma unroll
      for (int c = 0; c < NCH; c++) {
        const int idx = c * 256 + tid;
        if (idx < BS * DT / 8) dst[idx] = pf[c];
      }
    } else {
      const short8_t* src = reinterpret_cast<const short8_t*>(
          cache + (int64_t)bt[b] * (BS * DT));
      short8_t* dst = reinterpret_cast<short8_t*>(s_kv);
#pragma unroll
      for (int c = 0; c < NCH; c++) {
        const int idx = c * 256 + tid;
        if (idx < BS * DT / 8) dst[idx] = src[idx];
      }
    }
    __syncthreads();
    if constexpr (DB) {   // issue NEXT block's loads before computing
      if (b + 1 < nblocks) {
        const short8_t* src = reinterpret_cast<const short8_t*>(
            cache + (int64_t)bt[b + 1] * (BS * DT));
#pragma unroll
        for (int c = 0; c < NCH; c++) {
          const int idx = c * 256 + tid;
          if (idx < BS * DT / 8) pf[c] = src[idx];
        }
      }
    }

    // ---- scores for the 16 tokens (intra-wave handoff to exp) ----
    // 3 independent dot accumulators per token + 2-token unroll: the
    // naive single-accumulator form is an 18-deep serial v_dot2 chain
    // (~170 dependent cycles/token) that stalls the SIMD
#pragma unroll 2
    for (int t = 0; t < BS; t++) {
      const short* kp = s_kv + t * DT + slice * SL;
      float p0 = 0.f, p1 = 0.f, p2 = 0.f;
#pragma unroll
      for (int j = 0; j < SL / 12; j++) {
        const short4_t a = *reinterpret_cast<const short4_t*>(kp + j * 12);
        const short4_t bq = *reinterpret_cast<const short4_t*>(
            kp + j * 12 + 4);
        const short4_t c = *reinterpret_cast<const short4_t*>(
            kp + j * 12 + 8);
        const bf16x2_t* ka = reinterpret_cast<const bf16x2_t*>(&a);
        const bf16x2_t* kb = reinterpret_cast<const bf16x2_t*>(&bq);
        const bf16x2_t* kc = reinterpret_cast<const bf16x2_t*>(&c);
        const bf16x2_t* qa = reinterpret_cast<const bf16x2_t*>(
            &qv[j * 3]);
        const bf16x2_t* qb = reinterpret_cast<const bf16x2_t*>(
            &qv[j * 3 + 1]);
        const bf16x2_t* qc = reinterpret_cast<const bf16x2_t*>(
            &qv[j * 3 + 2]);
        p0 = __builtin_amdgcn_fdot2_f32_bf16(qa[0], ka[0], p0, false);
        p1 = __builtin_amdgcn_fdot2_f32_bf16(qb[0], kb[0], p1, false);
        p2 = __builtin_amdgcn_fdot2_f32_bf16(qc[0], kc[0], p2, false);
        p0 = __builtin_amdgcn_fdot2_f32_bf16(qa[1], ka[1], p0, false);
        p1 = __builtin_amdgcn_fdot2_f32_bf16(qb[1], kb[1], p1, false);
        p2 = __builtin_amdgcn_fdot2_f32_bf16(qc[1], kc[1], p2, false);
      }
      float part = (p0 + p1) + p2;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        part += __shfl_xor(part, off, 64);
      if (slice == 0) {
        const bool valid = b * BS + t < seq_len;
        s_p[t][sc_head] = valid ? part * scale : -1e30f;
      }
    }

    // ---- online softmax update (thread (h, t); state in registers) ----
    float corr;
    {
      const int t = ac_s;
      float s = s_p[t][ac_h];
      float bm = s;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        bm = fmaxf(bm, __shfl_xor(bm, off, 64));
      const float Mnew = fmaxf(M, bm);
      const float p = s > -1e29f ? __expf(s - Mnew) : 0.f;
      float bsum = p;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        bsum += __shfl_xor(bsum, off, 64);
      s_p[t][ac_h] = p;
      corr = Mnew > M ? __expf(M - Mnew) : 1.f;
      l = l * corr + bsum;
      M = Mnew;
    }

    // ---- accumulate c_kv into the latent-space output ----
    // packed v_pk_fma_f32 (2 fma/instr) like paged_attention_sp phase C
    {
      float2_t* a2 = reinterpret_cast<float2_t*>(acc);
      const float2_t corr2 = {corr, corr};
#pragma unroll
      for (int j = 0; j < AD / 2; j++) a2[j] *= corr2;
      for (int t = 0; t < BS; t++) {
        const float p = s_p[t][ac_h];
        const float2_t p2 = {p, p};
        const short* vp = s_kv + t * DT + ac_s * AD;
#pragma unroll
        for (int c = 0; c < AD / 8; c++) {
          const short8_t v8 = *reinterpret_cast<const short8_t*>(vp + c * 8);
          float vreg[8];
#pragma unroll
          for (int j = 0; j < 8; j++) vreg[j] = bf16_to_f32(v8[j]);
          const float2_t* v2 = reinterpret_cast<const float2_t*>(vreg);
#pragma unroll
          for (int k = 0; k < 4; k++) a2[c * 4 + k] += p2 * v2[k];
        }
      }
    }
    __syncthreads();   // s_kv free for the next block's staging
  }

  // ---- epilogue: normalize + write [T, H, R] ----
  if (h0 + ac_h < H) {
    const float inv = 1.f / fmaxf(l, 1e-20f);
    short* op = out + ((int64_t)seq * H + (h0 + ac_h)) * R + ac_s * AD;
#pragma unroll
    for (int j = 0; j < AD; j++) op[j] = f32_to_bf16(acc[j] * inv);
  }
}

void mla_decode(at::Tensor out, at::Tensor q, at::Tensor cache,
                at::Tensor block_tables, at::Tensor seq_lens, double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == at::kBFloat16 && q.is_contiguous());
  TORCH_CHECK(cache.dtype() == at::kBFloat16 && cache.is_contiguous());
  TORCH_CHECK(out.is_contiguous() && out.dtype() == at::kBFloat16);
  TORCH_CHECK(block_tables.dtype() == at::kInt
              && seq_lens.dtype() == at::kInt);
  const int T = q.size(0);
  const int H = q.size(1);
  const int DT = q.size(2);
  const int BS = cache.size(1);
  const int R = out.size(2);
  TORCH_CHECK(cache.size(2) == DT && out.size(0) == T && out.size(1) == H);
  TORCH_CHECK(BS == 16, "block_size must be 16");
  const int max_blocks = block_tables.size(1);
  auto stream = at::hip::getCurrentHIPStream();
  if (T == 0) return;
  const dim3 grid(T, (H + 15) / 16);
  static const int occ_env = []() {
    const char* e = getenv("KAITO_MLA_OCC");
    return e ? atoi(e) : 0;
  }();
  // register double-buffer measured faster at every shape (247.9 vs
  // 258.8 us bs=768, 183.5 vs 209.7 bs=256) — default ON
  static const bool db_env = []() {
    const char* e = getenv("KAITO_MLA_DB");
    return e == nullptr || atoi(e) != 0;
  }();
  if (R == 512 && DT == 576) {
#define MLA_LAUNCH(OCC_, DB_)                                                 \
    hipLaunchKernelGGL((mla_decode_kernel<512, 64, 16, OCC_, DB_>), grid,      \
                       dim3(256), 0, stream, (short*)out.data_ptr(),          \
                       (const short*)q.data_ptr(),                            \
                       (const short*)cache.data_ptr(),                        \
                       block_tables.data_ptr<int>(),                          \
                       seq_lens.data_ptr<int>(), (float)scale, H, max_blocks)
    if (db_env) {
      switch (occ_env) {
        case 5: MLA_LAUNCH(5, true); break;
        case 6: MLA_LAUNCH(6, true); break;
        default: MLA_LAUNCH(4, true); break;
      }
    } else {
      switch (occ_env) {
        case 5: MLA_LAUNCH(5, false); break;
        case 6: MLA_LAUNCH(6, false); break;
        case 8: MLA_LAUNCH(8, false); break;
        default: MLA_LAUNCH(4, false); break;
      }
    }
#undef MLA_LAUNCH
  } else {
    TORCH_CHECK(false, "unsupported MLA dims r=", R, " r+rope=", DT,
                " (deepseek family is 512/576)");
  }
}

// Fused MLA cache write: latent row (c_kv ‖ k_rope) scattered to the
// paged cache in ONE kernel. Replaces a torch where + cat + index_copy_
// chain that showed up at ~7% of a v2-lite serving run
// (profiles/r02_mla_kernels.md). One wave per token: 64 lanes x 16 B
// covers the 512-dim c_kv; lanes 0-7 append the 64 rope dims.
template <int R, int P>
__global__ __launch_bounds__(64, 8)
void mla_cache_write_kernel(
    short* __restrict__ cache,          // [NB, BS, R+P] bf16
    const short* __restrict__ c_kv,     // [T, R]
    const short* __restrict__ k_pe,     // [T, P] (row stride pe_stride)
    const int64_t* __restrict__ slots,  // [T]; <0 = skip (padding)
    const int64_t pe_stride) {
  constexpr int DT = R + P;
  const int t = blockIdx.x;
  const int lane = threadIdx.x;
  const int64_t slot = slots[t];
  if (slot < 0) return;
  short* dst = cache + slot * DT;
  static_assert(R == 64 * 8 && P == 8 * 8);
  *reinterpret_cast<short8_t*>(dst + lane * 8) =
      *reinterpret_cast<const short8_t*>(c_kv + (int64_t)t * R + lane * 8);
  if (lane < P / 8)
    *reinterpret_cast<short8_t*>(dst + R + lane * 8) =
        *reinterpret_cast<const short8_t*>(k_pe + t * pe_stride + lane * 8);
}

void mla_cache_write(at::Tensor cache, at::Tensor c_kv, at::Tensor k_pe,
                     at::Tensor slots) {
  TORCH_CHECK(cache.is_cuda() && cache.dtype() == at::kBFloat16
              && cache.is_contiguous());
  TORCH_CHECK(c_kv.is_contiguous() && c_kv.dtype() == at::kBFloat16);
  TORCH_CHECK(k_pe.stride(-1) == 1 && k_pe.dtype() == at::kBFloat16);
  TORCH_CHECK(slots.dtype() == at::kLong);
  const int T = c_kv.size(0);
  const int R = c_kv.size(1);
  const int P = k_pe.size(1);
  TORCH_CHECK(cache.size(2) == R + P && k_pe.size(0) == T
              && slots.size(0) == T);
  auto stream = at::hip::getCurrentHIPStream();
  if (T == 0) return;
  if (R == 512 && P == 64) {
    hipLaunchKernelGGL((mla_cache_write_kernel<512, 64>), dim3(T), dim3(64),
                       0, stream, (short*)cache.data_ptr(),
                       (const short*)c_kv.data_ptr(),
                       (const short*)k_pe.data_ptr(),
                       slots.data_ptr<int64_t>(), k_pe.stride(0));
  } else {
    TORCH_CHECK(false, "unsupported MLA cache dims ", R, "+", P);
  }
}

}  // namespace kaito
