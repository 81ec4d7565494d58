// Fused MoE grouped GEMMs for gfx950 (MI355X).
//
// Replaces the round-1 per-expert torch GEMM loop whose host-side read
// of expert counts (`counts.tolist()`) forced eager MoE decode
// (docs/ROADMAP.md #3). Everything here is device-side and
// static-shaped, so MoE decode steps capture into hipGraphs:
//
//   * tokens are sorted by expert on the GPU (torch argsort) and the
//     kernels get `sorted_ids` (token index per sorted row) +
//     `offsets` ([E+1] exclusive prefix sums);
//   * the grid covers the WORST-CASE tile count (ceil(TK/BM) m-tiles
//     per expert); tiles beyond an expert's real row range exit after
//     reading two ints — no moe-align padding, no token dropping, no
//     host sync;
//   * kernel 1: act[s, :] = silu(x[tok] @ Wg[e]^T) * (x[tok] @ Wu[e]^T)
//     with the A rows GATHERED through sorted_ids;
//   * kernel 2: out[tok, :] += gate_s * (act[s] @ Wd[e]^T), scattered
//     with f32 atomics.
//
// GEMM structure (guide §5, 2-barrier, BM=BN=BK=64, 4 waves as 2x2,
// MFMA 16x16x32 bf16): A and B tiles staged to LDS with
// global_load_lds (16 B lanes, wave-uniform destination) using the
// pre-swizzled-SOURCE XOR pattern so the ds_read_b128 fragment reads
// are bank-conflict-free (guide §6 G4: row-major [64][64] bf16 would be
// a 32-way conflict; `chunk ^= row & 7` spreads the column across 8
// banks, applied identically on store-source and read).
//
// Expert parallelism: callers pass `e_base` and size the grid's
// expert dimension to the LOCAL expert count; TP shards the expert
// intermediate dim (IE_local) — both reuse these kernels unchanged.
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

namespace {

constexpr int BM = 64, BK = 64;
constexpr int NW = 4;  // waves (2x2 over the BM x BN tile)

template <int ROWS = 64>
KAITO_DEV void stage_tile_gathered(
    const short* __restrict__ src_base,  // element base (row stride given)
    int64_t row_stride,                  // elements per source row
    const int* __restrict__ row_ids,     // ROWS source rows
    short* __restrict__ lds,             // [ROWS][BK] linear
    int wave, int lane) {
  // ROWSx64 bf16 tile = ROWS/8 wave-segments of 64 lanes x 16 B.
#pragma unroll
  for (int i = 0; i < ROWS / 32; i++) {
    const int seg = i * NW + wave;
    const int idx = seg * 64 + lane;
    const int row = idx >> 3;
    const int c = (idx & 7) ^ (row & 7);   // pre-swizzled source chunk
    const short* src = src_base + (int64_t)row_ids[row] * row_stride + c * 8;
    // segment = 64 lanes x 16 B = 512 shorts; HW writes lane i at
    // base + i*16 (wave-uniform destination)
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(lds + seg * 512),
        16, 0, 0);
  }
}

template <int ROWS = 64>
KAITO_DEV void stage_tile_rows(
    const short* __restrict__ src,       // first row, element base
    int64_t row_stride,
    short* __restrict__ lds, int wave, int lane) {
#pragma unroll
  for (int i = 0; i < ROWS / 32; i++) {
    const int seg = i * NW + wave;
    const int idx = seg * 64 + lane;
    const int row = idx >> 3;
    const int c = (idx & 7) ^ (row & 7);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)
            (src + (int64_t)row * row_stride + c * 8),
        (__attribute__((address_space(3))) unsigned int*)(lds + seg * 512),
        16, 0, 0);
  }
}

// read one MFMA operand fragment from a swizzled [64][BK] LDS tile:
// lane (lo, hi) wants row `row0+lo`, k-chunk hi of the kk-th 32-k slab.
KAITO_DEV bf16x8 frag_read(const short* __restrict__ lds, int row, int kk,
                           int hi) {
  const int c = (kk * 4 + hi) ^ (row & 7);
  return *reinterpret_cast<const bf16x8*>(lds + row * BK + c * 8);
}

KAITO_DEV float silu_f(float v) { return v / (1.f + __expf(-v)); }

// act_mode 1: gpt-oss clamped swiglu — gate=min(g,7), up=clamp(u,±7),
// (up+1) * gate*sigmoid(1.702*gate)
KAITO_DEV float moe_act_f(int mode, float g, float u) {
  if (mode == 1) {
    g = fminf(g, 7.f);
    u = fminf(fmaxf(u, -7.f), 7.f);
    const float glu = g / (1.f + __expf(-1.702f * g));
    return (u + 1.f) * glu;
  }
  return silu_f(g) * u;
}

}  // namespace

// act[s, n] = silu(x[tok_s] @ Wg[e]^T) * (x[tok_s] @ Wu[e]^T)
// grid: (max_m_tiles, n_local_experts, IE/BN); block 256.
template <int BN>
__global__ __launch_bounds__(256, 3)
void moe_gate_silu_kernel(
    short* __restrict__ act,             // [TK, IE] bf16
    const short* __restrict__ x,         // [T, H] bf16
    const short* __restrict__ w,         // [E, 2*IE, H] bf16
    const int* __restrict__ sorted_ids,  // [TK]
    const int* __restrict__ offsets,     // [E+1] (global expert ids)
    const short* __restrict__ bias,      // [E, 2*IE] bf16 or nullptr
    const int act_mode,
    const int e_base, const int H, const int IE) {
  const int e = e_base + blockIdx.y;
  const int m0 = offsets[e] + blockIdx.x * BM;
  const int m_end = offsets[e + 1];
  if (m0 >= m_end) return;
  const int m_rem = min(BM, m_end - m0);
  const int n0 = blockIdx.z * BN;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wm = wave >> 1, wn = wave & 1;   // 2x2 wave grid
  const int lo = lane & 15, hi = lane >> 4;
  constexpr int NR = BN / 32;   // 16-col frags per wave along N

  // single-buffered 2-barrier staging: an explicit double-buffer
  // measured +0% here (occupancy already hides the staging latency —
  // these kernels run at ~5 TB/s weight streaming, tools/
  // bench_moe_kernels.py) and the saved LDS buys 2x the resident blocks
  __shared__ short lds_a[BM * BK];
  __shared__ short lds_bg[BN * BK];
  __shared__ short lds_bu[BN * BK];
  __shared__ int s_rows[BM];
  for (int i = threadIdx.x; i < BM; i += 256)
    s_rows[i] = sorted_ids[m0 + min(i, m_rem - 1)];
  __syncthreads();

  const short* wg = w + (int64_t)e * (2 * IE) * H + (int64_t)n0 * H;
  const short* wu = wg + (int64_t)IE * H;

  f32x4 accg[2][NR], accu[2][NR];
#pragma unroll
  for (int a = 0; a < 2; a++)
#pragma unroll
    for (int b = 0; b < NR; b++) {
      accg[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};
      accu[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};
    }

  for (int k0 = 0; k0 < H; k0 += BK) {
    stage_tile_gathered<BM>(x + k0, H, s_rows, lds_a, wave, lane);
    stage_tile_rows<BN>(wg + k0, H, lds_bg, wave, lane);
    stage_tile_rows<BN>(wu + k0, H, lds_bu, wave, lane);
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK / 32; kk++) {
      bf16x8 af[2], bg[NR], bu[NR];
#pragma unroll
      for (int r = 0; r < 2; r++)
        af[r] = frag_read(lds_a, wm * 32 + r * 16 + lo, kk, hi);
#pragma unroll
      for (int b = 0; b < NR; b++) {
        bg[b] = frag_read(lds_bg, wn * (BN / 2) + b * 16 + lo, kk, hi);
        bu[b] = frag_read(lds_bu, wn * (BN / 2) + b * 16 + lo, kk, hi);
      }
#pragma unroll
      for (int a = 0; a < 2; a++)
#pragma unroll
        for (int b = 0; b < NR; b++) {
          accg[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[a], bg[b], accg[a][b], 0, 0, 0);
          accu[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[a], bu[b], accu[a][b], 0, 0, 0);
        }
    }
    __syncthreads();
  }

  // epilogue: silu(g)*u → act[m0+row][n0+col] (rows are SORTED order,
  // contiguous — the scatter happens in the down kernel)
#pragma unroll
  for (int a = 0; a < 2; a++) {
    const int row_base = wm * 32 + a * 16 + hi * 4;
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int row = row_base + r;
      if (row >= m_rem) continue;
#pragma unroll
      for (int b = 0; b < NR; b++) {
        const int col = n0 + wn * (BN / 2) + b * 16 + lo;
        float g = accg[a][b][r], u = accu[a][b][r];
        if (bias != nullptr) {
          g += bf16_to_f32(bias[(int64_t)e * 2 * IE + col]);
          u += bf16_to_f32(bias[(int64_t)e * 2 * IE + IE + col]);
        }
        act[(int64_t)(m0 + row) * IE + col] =
            f32_to_bf16(moe_act_f(act_mode, g, u));
      }
    }
  }
}

// out[tok_s, n] += gate_s * (act[s] @ Wd[e]^T); out is f32, atomics.
// grid: (max_m_tiles, n_local_experts, H/BN); block 256.
template <int BN>
__global__ __launch_bounds__(256, 3)
void moe_down_scatter_kernel(
    float* __restrict__ out,             // [T, H] f32 (pre-zeroed)
    const short* __restrict__ act,       // [TK, IE] bf16
    const short* __restrict__ w,         // [E, H, IE] bf16
    const int* __restrict__ sorted_ids,  // [TK]
    const float* __restrict__ gates,     // [TK] sorted gate weights
    const int* __restrict__ offsets,     // [E+1]
    const short* __restrict__ bias,      // [E, H] bf16 or nullptr
    const int e_base, const int H, const int IE) {
  const int e = e_base + blockIdx.y;
  const int m0 = offsets[e] + blockIdx.x * BM;
  const int m_end = offsets[e + 1];
  if (m0 >= m_end) return;
  const int m_rem = min(BM, m_end - m0);
  const int n0 = blockIdx.z * BN;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wm = wave >> 1, wn = wave & 1;
  const int lo = lane & 15, hi = lane >> 4;
  constexpr int NR = BN / 32;

  __shared__ short lds_a[BM * BK];
  __shared__ short lds_b[BN * BK];

  const short* wd = w + (int64_t)e * H * IE + (int64_t)n0 * IE;

  f32x4 acc[2][NR];
#pragma unroll
  for (int a = 0; a < 2; a++)
#pragma unroll
    for (int b = 0; b < NR; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < IE; k0 += BK) {
    stage_tile_rows<BM>(act + (int64_t)m0 * IE + k0, IE, lds_a, wave,
                        lane);
    stage_tile_rows<BN>(wd + k0, IE, lds_b, wave, lane);
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK / 32; kk++) {
      bf16x8 af[2], bf[NR];
#pragma unroll
      for (int r = 0; r < 2; r++)
        af[r] = frag_read(lds_a, wm * 32 + r * 16 + lo, kk, hi);
#pragma unroll
      for (int b = 0; b < NR; b++)
        bf[b] = frag_read(lds_b, wn * (BN / 2) + b * 16 + lo, kk, hi);
#pragma unroll
      for (int a = 0; a < 2; a++)
#pragma unroll
        for (int b = 0; b < NR; b++)
          acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[a], bf[b], acc[a][b], 0, 0, 0);
    }
    __syncthreads();
  }

  // NOTE: rows past m_end within this tile read garbage A rows — only
  // rows < m_rem are scattered. Rows of the NEXT expert that fall in
  // this tile's [m0, m0+BM) range are handled by THAT expert's tiles.
#pragma unroll
  for (int a = 0; a < 2; a++) {
    const int row_base = wm * 32 + a * 16 + hi * 4;
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int row = row_base + r;
      if (row >= m_rem) continue;
      const int tok = sorted_ids[m0 + row];
      const float g = gates[m0 + row];
#pragma unroll
      for (int b = 0; b < NR; b++) {
        const int col = n0 + wn * (BN / 2) + b * 16 + lo;
        float v = acc[a][b][r];
        if (bias != nullptr) v += bf16_to_f32(bias[(int64_t)e * H + col]);
        atomicAdd(&out[(int64_t)tok * H + col], g * v);
      }
    }
  }
}

void moe_gate_silu(at::Tensor act, at::Tensor x, at::Tensor w_gate_up,
                   at::Tensor sorted_ids, at::Tensor offsets,
                   at::Tensor bias, int64_t act_mode,
                   int64_t e_base, int64_t n_local_experts) {
  const short* bias_ptr = nullptr;
  if (bias.numel() > 0) {
    TORCH_CHECK(bias.dtype() == at::kBFloat16 && bias.is_contiguous());
    bias_ptr = (const short*)bias.data_ptr();
  }
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(act.is_contiguous() && w_gate_up.is_contiguous());
  TORCH_CHECK(sorted_ids.dtype() == at::kInt && offsets.dtype() == at::kInt);
  const int TK = act.size(0);
  const int IE = act.size(1);
  const int H = x.size(1);
  TORCH_CHECK(H % BK == 0 && IE % 64 == 0,
              "H/IE must be multiples of 64 for the fused MoE path");
  TORCH_CHECK(w_gate_up.size(1) == 2 * IE && w_gate_up.size(2) == H);
  const int mtiles = (TK + BM - 1) / BM;
  if (mtiles == 0 || n_local_experts == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  // BN=128 when it divides IE: halves the A-tile re-reads across
  // n-tiles (the prefill-regime cost) at the same weight traffic
  const int BN = IE % 128 == 0 ? 128 : 64;
  dim3 grid(mtiles, n_local_experts, IE / BN), block(256);
  if (BN == 128)
    moe_gate_silu_kernel<128><<<grid, block, 0, stream>>>(
        (short*)act.data_ptr(), (const short*)x.data_ptr(),
        (const short*)w_gate_up.data_ptr(), sorted_ids.data_ptr<int>(),
        offsets.data_ptr<int>(), bias_ptr, (int)act_mode, (int)e_base, H,
        IE);
  else
    moe_gate_silu_kernel<64><<<grid, block, 0, stream>>>(
        (short*)act.data_ptr(), (const short*)x.data_ptr(),
        (const short*)w_gate_up.data_ptr(), sorted_ids.data_ptr<int>(),
        offsets.data_ptr<int>(), bias_ptr, (int)act_mode, (int)e_base, H,
        IE);
}

void moe_down_scatter(at::Tensor out, at::Tensor act, at::Tensor w_down,
                      at::Tensor sorted_ids, at::Tensor gates,
                      at::Tensor offsets, at::Tensor bias, int64_t e_base,
                      int64_t n_local_experts) {
  const short* bias_ptr = nullptr;
  if (bias.numel() > 0) {
    TORCH_CHECK(bias.dtype() == at::kBFloat16 && bias.is_contiguous());
    bias_ptr = (const short*)bias.data_ptr();
  }
  TORCH_CHECK(out.is_cuda() && out.dtype() == at::kFloat &&
              out.is_contiguous());
  TORCH_CHECK(act.is_contiguous() && w_down.is_contiguous());
  TORCH_CHECK(gates.dtype() == at::kFloat);
  const int TK = act.size(0);
  const int IE = act.size(1);
  const int H = out.size(1);
  TORCH_CHECK(H % 64 == 0 && IE % BK == 0);
  TORCH_CHECK(w_down.size(1) == H && w_down.size(2) == IE);
  const int mtiles = (TK + BM - 1) / BM;
  if (mtiles == 0 || n_local_experts == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  const int BN = H % 128 == 0 ? 128 : 64;
  dim3 grid(mtiles, n_local_experts, H / BN), block(256);
  if (BN == 128)
    moe_down_scatter_kernel<128><<<grid, block, 0, stream>>>(
        out.data_ptr<float>(), (const short*)act.data_ptr(),
        (const short*)w_down.data_ptr(), sorted_ids.data_ptr<int>(),
        gates.data_ptr<float>(), offsets.data_ptr<int>(), bias_ptr,
        (int)e_base, H, IE);
  else
    moe_down_scatter_kernel<64><<<grid, block, 0, stream>>>(
        out.data_ptr<float>(), (const short*)act.data_ptr(),
        (const short*)w_down.data_ptr(), sorted_ids.data_ptr<int>(),
        gates.data_ptr<float>(), offsets.data_ptr<int>(), bias_ptr,
        (int)e_base, H, IE);
}

}  // namespace kaito
