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
#include <algorithm>
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
// PH: diagnostic phase mask (1 = stage only, 2 = +scores, 3 = full) —
// timing ablation via KAITO_MLA_PH; output is garbage for PH<3.
// MF: MFMA score phase — C[16h x 16t] via mfma_f32_16x16x32_bf16 with
// the 18 K-steps split 4/5/4/5 across the 4 waves (partials reduced
// through LDS), Q K-range PRELOADED per wave (Q is block-invariant),
// s_kv rows padded +16 B so B-fragment b128 reads hit the 2-cycle
// LDS minimum instead of a 16-way bank conflict (KAITO_MLA_MF).
// MV: MFMA V-accumulate — out[16h x 512d] += P[16h x 16t(+16 zero-pad
// k)] @ C_kv[16t x 512d] via 8 d-tiles per wave; V staged TRANSPOSED
// (s_vt[dim][token]) during the cooperative copy so B-fragments are
// contiguous b128 reads (same trick as prefill_attention); requires MF.
// SP: flash-decoding token-split — blockIdx.z picks a contiguous
// partition of the KV blocks; each split writes UN-normalized partials
// (acc, M, l) to scratch and mla_merge_kernel combines them. Fills the
// 256-CU chip when T x head-tiles alone can't (docs/ROADMAP.md).
template <int R, int P, int BS, int OCC = 4, bool DB = false, int PH = 3,
          bool MF = false, bool MV = false, bool SP = false>
__global__ __launch_bounds__(256, OCC)
void mla_decode_kernel(
    short* __restrict__ out,            // [T, H, R] bf16
    const short* __restrict__ q,        // [T, H, R+P] bf16
    const short* __restrict__ cache,    // [NB, BS, R+P] bf16
    const int* __restrict__ block_tables,  // [T, max_blocks]
    const int* __restrict__ seq_lens,   // [T]
    const float scale, const int H, const int max_blocks,
    float* __restrict__ part_acc = nullptr,   // [T, S, 16, R]
    float* __restrict__ part_ml = nullptr,    // [T, S, 16, 2]
    const int S = 1) {
  constexpr int DT = R + P;             // 576
  constexpr int SL = DT / 16;           // 36 dims per score slice
  constexpr int AD = R / 16;            // 32 dims per accum slice
  constexpr int HT = 16;                // heads per workgroup
  static_assert(DT % 16 == 0 && R % 16 == 0 && SL % 12 == 0
                && AD % 8 == 0);

  const int seq = blockIdx.x;
  const int h0 = blockIdx.y * HT;
  const int split = SP ? blockIdx.z : 0;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int seq_len = seq_lens[seq];
  const int nb_all = (seq_len + BS - 1) / BS;
  const int b_lo = SP ? (nb_all * split) / S : 0;
  const int nblocks = SP ? (nb_all * (split + 1)) / S : nb_all;
  const int* bt = block_tables + (int64_t)seq * max_blocks;

  // score-phase role: 4 heads per wave, 16 dim-slices per head
  const int sc_head = wave * 4 + (lane >> 4);     // 0..15 == tid>>4
  const int slice = lane & 15;
  // accum/exp-phase role (same head): dim/token slice
  const int ac_h = tid >> 4;                      // == sc_head
  const int ac_s = tid & 15;

  constexpr int RS = MF ? DT + 8 : DT;   // padded row (bank spread)
  constexpr int TP = 20;                 // s_vt token stride (16+4 pad)
  __shared__ short s_kv[BS * RS];                 // staged block
  __shared__ float s_p[BS][HT];                   // scores→probs
  __shared__ float s_part[MF ? 4 : 1][HT][HT];    // per-wave C partials
  __shared__ short s_vt[MV ? R * TP : 1];         // V transposed (20.5KB)
  __shared__ short s_pb[MV ? HT : 1][TP];         // P row-major bf16
  __shared__ float s_corr[HT], s_l[HT];
  __shared__ float s_M2[SP ? HT : 1];             // running max (splits)

  // per-head online-softmax state, held REDUNDANTLY in registers by all
  // 16 lanes of the head's thread group (every lane derives identical
  // values from the same shfl reductions) — no cross-lane LDS handoff
  float M = -1e30f, l = 0.f;

  constexpr int KSTEPS = DT / 32;                 // MFMA K-chunks (18)
  constexpr int KW = (KSTEPS + 3) / 4 + 1;        // per-wave upper bound
  const int kb0 = (KSTEPS * wave) / 4;            // ragged 4/5/4/5 split
  const int kb1 = (KSTEPS * (wave + 1)) / 4;
  const int hi = lane >> 4;

  // q for the score phase; zero for padded heads so their scores stay
  // finite (their output is never written). Rows are CLAMPED (not just
  // the value) so an if-converted load can't read past the q tensor.
  short4_t qv[MF ? 1 : SL / 4];
  bf16x8 af[MF ? KW : 1];
  if constexpr (MF) {
    // A-fragment rows: head = lane&15; k = kc*32 + (lane>>4)*8 + [0..7]
    const bool ok = h0 + (lane & 15) < H;
    const int hq = min(h0 + (lane & 15), H - 1);
    const short* qp = q + ((int64_t)seq * H + hq) * DT;
#pragma unroll
    for (int j = 0; j < KW; j++) {
      if (kb0 + j >= kb1) break;
      af[j] = *reinterpret_cast<const bf16x8*>(
          qp + (kb0 + j) * 32 + hi * 8);
      if (!ok) af[j] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  } else {
    const bool head_ok = h0 + sc_head < H;
    const int hq = min(h0 + sc_head, H - 1);
    const short* qp = q + ((int64_t)seq * H + hq) * DT + slice * SL;
#pragma unroll
    for (int j = 0; j < SL / 4; j++) {
      qv[j] = *reinterpret_cast<const short4_t*>(qp + j * 4);
      if (!head_ok) qv[j] = short4_t{0, 0, 0, 0};
    }
  }

  float acc[MV ? 1 : AD];
  f32x4 accv[MV ? 8 : 1];
  if constexpr (MV) {
#pragma unroll
    for (int j = 0; j < 8; j++) accv[j] = f32x4{0.f, 0.f, 0.f, 0.f};
  } else {
#pragma unroll
    for (int j = 0; j < AD; j++) acc[j] = 0.f;
  }

  typedef __bf16 bf16x2_t __attribute__((ext_vector_type(2)));

  constexpr int NCH = (BS * DT / 8 + 255) / 256;   // b128 chunks/thread
  short8_t pf[DB ? NCH : 1];
  if constexpr (DB) {   // prologue: first block's loads in flight
    const short8_t* src = reinterpret_cast<const short8_t*>(
        cache + (int64_t)bt[b_lo] * (BS * DT));
#pragma unroll
    for (int c = 0; c < NCH; c++) {
      const int idx = c * 256 + tid;
      if (idx < BS * DT / 8) pf[c] = src[idx];
    }
  }

  for (int b = b_lo; b < nblocks; b++) {
    // ---- stage one KV block: 1152 x b128, 256 threads ----
    if constexpr (DB) {
#pragma unroll
      for (int c = 0; c < NCH; c++) {
        const int idx = c * 256 + tid;
        if (idx < BS * DT / 8) {
          *reinterpret_cast<short8_t*>(
              s_kv + (idx / (DT / 8)) * RS + (idx % (DT / 8)) * 8) = pf[c];
          if constexpr (MV) {
            const int tok = idx / (DT / 8);
            const int d0 = (idx % (DT / 8)) * 8;
            if (d0 < R) {
#pragma unroll
              for (int j = 0; j < 8; j++)
                s_vt[(d0 + j) * TP + tok] = pf[c][j];
            }
          }
        }
      }
    } else {
      const short8_t* src = reinterpret_cast<const short8_t*>(
          cache + (int64_t)bt[b] * (BS * DT));
#pragma unroll
      for (int c = 0; c < NCH; c++) {
        const int idx = c * 256 + tid;
        if (idx < BS * DT / 8)
          *reinterpret_cast<short8_t*>(
              s_kv + (idx / (DT / 8)) * RS + (idx % (DT / 8)) * 8) =
              src[idx];
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
    if constexpr (MF) {
      f32x4 cp = {0.f, 0.f, 0.f, 0.f};
      const int tcol = lane & 15;
#pragma unroll
      for (int j = 0; j < KW; j++) {
        if (kb0 + j >= kb1) break;
        const bf16x8 bfr = *reinterpret_cast<const bf16x8*>(
            s_kv + tcol * RS + (kb0 + j) * 32 + hi * 8);
        cp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[j], bfr, cp,
                                                     0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; r++)
        s_part[wave][hi * 4 + r][tcol] = cp[r];
      __syncthreads();                 // cross-wave partial handoff
    } else if constexpr (PH >= 2) {
#pragma unroll 2
    for (int t = 0; t < BS; t++) {
      const short* kp = s_kv + t * RS + slice * SL;
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
    }

    // ---- online softmax update (thread (h, t); state in registers) ----
    float corr = 1.f;
    if constexpr (PH >= 3) {
      const int t = ac_s;
      float s;
      if constexpr (MF) {
        s = ((s_part[0][ac_h][t] + s_part[1][ac_h][t])
             + (s_part[2][ac_h][t] + s_part[3][ac_h][t])) * scale;
        if (b * BS + t >= seq_len) s = -1e30f;
      } else {
        s = s_p[t][ac_h];
      }
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
      if constexpr (MV) {
        s_pb[ac_h][t] = f32_to_bf16(p);
        if (t == 0) {
          s_corr[ac_h] = corr;
          s_l[ac_h] = l;
          if constexpr (SP) s_M2[ac_h] = M;
        }
      }
    }
    if constexpr (MV) __syncthreads();   // s_pb/s_corr to all waves

    // ---- accumulate c_kv into the latent-space output ----
    if constexpr (MV) {
      // rescale C fragments: row hi*4+r is head hi*4+r
      float cr[4];
#pragma unroll
      for (int r = 0; r < 4; r++) cr[r] = s_corr[hi * 4 + r];
#pragma unroll
      for (int j = 0; j < 8; j++)
#pragma unroll
        for (int r = 0; r < 4; r++) accv[j][r] *= cr[r];
      const int tok0 = (hi & 1) * 8;    // hi>=2: A is zero, B ignored
      const bf16x8 pa = hi < 2
          ? *reinterpret_cast<const bf16x8*>(&s_pb[lane & 15][hi * 8])
          : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 8; j++) {     // 8 dim-tiles per wave
        const int dim = (wave * 8 + j) * 16 + (lane & 15);
        const bf16x8 vb = *reinterpret_cast<const bf16x8*>(
            &s_vt[dim * TP + tok0]);
        accv[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vb, accv[j],
                                                          0, 0, 0);
      }
    } else if constexpr (PH >= 3) {
      float2_t* a2 = reinterpret_cast<float2_t*>(acc);
      const float2_t corr2 = {corr, corr};
#pragma unroll
      for (int j = 0; j < AD / 2; j++) a2[j] *= corr2;
      for (int t = 0; t < BS; t++) {
        const float p = s_p[t][ac_h];
        const float2_t p2 = {p, p};
        const short* vp = s_kv + t * RS + ac_s * AD;
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

  if constexpr (PH < 3) {
    // ablation variants: fake consumer keeps the staging/score stores
    // alive (never taken at runtime — seq_len >= 1)
    if (seq_len < 0)
      out[tid] = s_kv[tid] + (short)s_p[tid & 15][tid >> 4];
    return;
  }

  if constexpr (SP) {
    // split partials: UN-normalized latent acc + (M, l) per head; the
    // merge kernel rescales across splits. Empty split (b_lo >=
    // nblocks, short sequences) contributes exp(-1e30)=0.
    const bool empty = b_lo >= nblocks;
    const int HTOT = gridDim.y * HT;     // scratch covers ALL head tiles
#pragma unroll
    for (int j = 0; j < 8; j++) {
      const int dim = (wave * 8 + j) * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const int h = hi * 4 + r;
        part_acc[(((int64_t)seq * S + split) * HTOT + h0 + h) * R + dim] =
            accv[j][r];
      }
    }
    if (wave == 0 && (lane & 15) == 0) {
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const int h = hi * 4 + r;
        float* ml = part_ml
            + (((int64_t)seq * S + split) * HTOT + h0 + h) * 2;
        ml[0] = empty ? -1e30f : s_M2[h];
        ml[1] = empty ? 0.f : s_l[h];
      }
    }
    return;
  }

  // ---- epilogue: normalize + write [T, H, R] ----
  if constexpr (MV) {
    // C-fragment layout: row hi*4+r = head, col = dim-within-tile
#pragma unroll
    for (int j = 0; j < 8; j++) {
      const int dim = (wave * 8 + j) * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const int h = hi * 4 + r;
        if (h0 + h < H)
          out[((int64_t)seq * H + (h0 + h)) * R + dim] =
              f32_to_bf16(accv[j][r] / fmaxf(s_l[h], 1e-20f));
      }
    }
    return;
  }
  if (h0 + ac_h < H) {
    const float inv = 1.f / fmaxf(l, 1e-20f);
    short* op = out + ((int64_t)seq * H + (h0 + ac_h)) * R + ac_s * AD;
#pragma unroll
    for (int j = 0; j < AD; j++) op[j] = f32_to_bf16(acc[j] * inv);
  }
}

// Combine flash-decoding split partials: out[h] = sum_s w_s*acc_s /
// sum_s w_s*l_s with w_s = exp(M_s - max M). Thread (head = tid>>4,
// dim-slice = tid&15) mirrors the main kernel's accum layout.
template <int R>
__global__ __launch_bounds__(256, 8)
void mla_merge_kernel(short* __restrict__ out,
                      const float* __restrict__ part_acc,
                      const float* __restrict__ part_ml,
                      const int H, const int S) {
  constexpr int HT = 16;
  constexpr int AD = R / 16;
  const int seq = blockIdx.x;
  const int h = (int)(threadIdx.x >> 4) + blockIdx.y * HT;
  const int ds = threadIdx.x & 15;
  if (h >= H) return;
  const int HTOT = gridDim.y * HT;
  const int64_t base = (int64_t)seq * S;
  float gm = -1e30f;
  for (int sp = 0; sp < S; sp++)
    gm = fmaxf(gm, part_ml[((base + sp) * HTOT + h) * 2]);
  float den = 0.f;
  float num[AD];
#pragma unroll
  for (int j = 0; j < AD; j++) num[j] = 0.f;
  for (int sp = 0; sp < S; sp++) {
    const float* ml = part_ml + ((base + sp) * HTOT + h) * 2;
    const float w = __expf(ml[0] - gm);
    den += w * ml[1];
    const float* pa = part_acc + ((base + sp) * HTOT + h) * R + ds * AD;
#pragma unroll
    for (int j = 0; j < AD; j++) num[j] += w * pa[j];
  }
  const float inv = 1.f / fmaxf(den, 1e-20f);
  short* op = out + ((int64_t)seq * H + h) * R + ds * AD;
#pragma unroll
  for (int j = 0; j < AD; j++) op[j] = f32_to_bf16(num[j] * inv);
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
    static const int ph_env = []() {
      const char* e = getenv("KAITO_MLA_PH");
      return e ? atoi(e) : 3;
    }();
    // MFMA scores measured 159 vs 245 us (bs=768), 86 vs 176 (bs=256),
    // 184 vs 295 (bs=1024) — default ON (KAITO_MLA_MF=0 reverts)
    static const bool mf_env = []() {
      const char* e = getenv("KAITO_MLA_MF");
      return e == nullptr || atoi(e) != 0;
    }();
    // MFMA V-accumulate: 139.8 vs 158.7 us (bs=768), equal at bs=1024
    // — default ON (KAITO_MLA_MV=0 reverts)
    static const bool mv_env = []() {
      const char* e = getenv("KAITO_MLA_MV");
      return e == nullptr || atoi(e) != 0;
    }();
    if (ph_env == 1) {
      hipLaunchKernelGGL((mla_decode_kernel<512, 64, 16, 4, true, 1>), grid,
                         dim3(256), 0, stream, (short*)out.data_ptr(),
                         (const short*)q.data_ptr(),
                         (const short*)cache.data_ptr(),
                         block_tables.data_ptr<int>(),
                         seq_lens.data_ptr<int>(), (float)scale, H,
                         max_blocks);
    } else if (ph_env == 2) {
      hipLaunchKernelGGL((mla_decode_kernel<512, 64, 16, 4, true, 2>), grid,
                         dim3(256), 0, stream, (short*)out.data_ptr(),
                         (const short*)q.data_ptr(),
                         (const short*)cache.data_ptr(),
                         block_tables.data_ptr<int>(),
                         seq_lens.data_ptr<int>(), (float)scale, H,
                         max_blocks);
    } else if (mf_env && mv_env) {
      // flash-decoding token-split when the plain grid underfills the
      // 256-CU chip (KAITO_MLA_SPLIT overrides; 0 disables)
      const int tiles = (H + 15) / 16;
      int S = 1;
      const char* se = getenv("KAITO_MLA_SPLIT");
      if (se != nullptr) {
        S = atoi(se);
      } else if (T * tiles < 320) {
        S = std::min<int>(8, std::max<int>(2, 512 / std::max(T * tiles, 1)));
      }
      if (S > 1) {
        auto opts = at::TensorOptions()
            .dtype(at::kFloat).device(q.device());
        at::Tensor pacc = at::empty(
            {(int64_t)T * S * tiles * 16 * 512}, opts);
        at::Tensor pml = at::empty({(int64_t)T * S * tiles * 16 * 2}, opts);
        hipLaunchKernelGGL(
            (mla_decode_kernel<512, 64, 16, 4, true, 3, true, true, true>),
            dim3(T, tiles, S), dim3(256), 0, stream,
            (short*)out.data_ptr(), (const short*)q.data_ptr(),
            (const short*)cache.data_ptr(), block_tables.data_ptr<int>(),
            seq_lens.data_ptr<int>(), (float)scale, H, max_blocks,
            pacc.data_ptr<float>(), pml.data_ptr<float>(), S);
        hipLaunchKernelGGL((mla_merge_kernel<512>), dim3(T, tiles),
                           dim3(256), 0, stream, (short*)out.data_ptr(),
                           pacc.data_ptr<float>(), pml.data_ptr<float>(),
                           H, S);
      } else {
        hipLaunchKernelGGL(
            (mla_decode_kernel<512, 64, 16, 4, true, 3, true, true>), grid,
            dim3(256), 0, stream, (short*)out.data_ptr(),
            (const short*)q.data_ptr(), (const short*)cache.data_ptr(),
            block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(),
            (float)scale, H, max_blocks);
      }
    } else if (mf_env) {
      hipLaunchKernelGGL(
          (mla_decode_kernel<512, 64, 16, 4, true, 3, true>), grid,
          dim3(256), 0, stream, (short*)out.data_ptr(),
          (const short*)q.data_ptr(), (const short*)cache.data_ptr(),
          block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(),
          (float)scale, H, max_blocks);
    } else if (db_env) {
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
