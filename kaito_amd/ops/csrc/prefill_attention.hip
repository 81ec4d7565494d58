// Varlen causal flash-attention PREFILL kernel for gfx950 — MFMA
// (16x16x32 bf16) with LDS-tiled K/V, online softmax.
//
// Replaces vLLM's prefill attention. MI355X-first:
//  - MFMA 16x16x32 bf16 per-wave tiles (no warp-group MMA on CDNA).
//  - K tile LDS layout XOR-swizzled (bank-conflict fix, guide §6 G4).
//  - V staged transposed (Vt[d][kv], padded rows) so the PV B-fragment is a
//    contiguous ds_read_b128.
//  - workgroup = 4 waves × 16 q-rows = 64-row Q tile; K/V tiles of 32
//    staged cooperatively and shared by all 4 waves.
//
// Fragment layouts (gfx950 mfma_f32_16x16x32_bf16), verified by
// tests/test_gpu_kernels.py::test_mfma_tile_gemm against torch:
//   A[16,32]: lane holds row=lane&15, k=(lane>>4)*8+[0..7]
//   B[32,16]: lane holds col=lane&15, k=(lane>>4)*8+[0..7]
//   C/D[16,16]: lane holds col=lane&15, row=(lane>>4)*4+reg
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

constexpr int QTILE = 64;   // q rows per workgroup (16 per wave)
constexpr int KVT = 32;     // kv tile
constexpr int VT_PAD = 40;  // Vt row stride in bf16 (32 + 8 pad)

KAITO_DEV bf16x8 mfma_bf16_frag_zero() { return bf16x8{0,0,0,0,0,0,0,0}; }

template <int D>
__global__ __launch_bounds__(256, 2) void prefill_attn_kernel(
    short* __restrict__ out,          // [T, QH, D]
    const short* __restrict__ q,      // [T, QH, D]
    const short* __restrict__ k,      // [T, KH, D]
    const short* __restrict__ v,      // [T, KH, D]
    const int* __restrict__ tile_seq,   // [ntiles] seq index
    const int* __restrict__ tile_qbase, // [ntiles] q row base within seq
    const int* __restrict__ cu_seqlens, // [batch+1]
    const float scale, const int QH, const int KH,
    const int64_t q_stride, const int64_t kv_stride,
    const int window,                   // 0 = full causal
    const float* __restrict__ sinks) {  // [QH] or nullptr
  constexpr int KK = D / 32;          // MFMA k-steps over head_dim (4 for 128)
  constexpr int DT = D / 16;          // output d-tiles (8 for 128)
  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (QH / KH);
  const int seq = tile_seq[tile];
  const int q0 = tile_qbase[tile];    // within-seq q row of tile start
  const int tok0 = cu_seqlens[seq];
  const int slen = cu_seqlens[seq + 1] - tok0;

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 4;           // 0..3
  const int lo = lane & 15;

  // LDS: K tile (swizzled linear [KVT][D] bf16), Vt [D][VT_PAD] bf16,
  // P scratch per wave [16][40] bf16.
  __shared__ short lds_k[KVT * D];
  __shared__ short lds_vt[D * VT_PAD];
  __shared__ short lds_p[4][16 * VT_PAD];

  // ---- preload Q fragments (A-layout): row=lo, k=hi*8+kk*32 ----
  const int qrow_local = wave * 16 + lo;           // 0..63 in tile
  const int qrow = q0 + qrow_local;                // within-seq
  const bool qvalid = qrow < slen;
  const int qtok = tok0 + (qvalid ? qrow : slen - 1);
  bf16x8 qfrag[KK];
#pragma unroll
  for (int kk = 0; kk < KK; kk++) {
    const short8_t* qp = reinterpret_cast<const short8_t*>(
        q + (int64_t)qtok * q_stride + qh * D + kk * 32 + hi * 8);
    qfrag[kk] = *reinterpret_cast<const bf16x8*>(qp);
  }

  // per-lane softmax state: 4 q rows (row = hi*4 + r within the wave's 16)
  float mrow[4], lrow[4];
  f32x4 ofrag[DT];
#pragma unroll
  for (int r = 0; r < 4; r++) { mrow[r] = -1e30f; lrow[r] = 0.f; }
#pragma unroll
  for (int dt = 0; dt < DT; dt++) ofrag[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kv_end = min(slen, q0 + QTILE);        // causal bound for tile
  // sliding window: the earliest key any q row of this tile can see
  const int kv_begin = (window > 0 && q0 - window + 1 > 0)
      ? ((q0 - window + 1) / KVT) * KVT : 0;
  for (int kv0 = kv_begin; kv0 < kv_end; kv0 += KVT) {
    // ---- cooperative stage: K tile swizzled + V transposed ----
    // 256 threads × short8: K tile = KVT*D/8 vectors (512 for D=128).
    {
      const int nvec = KVT * D / 8;
      for (int i = threadIdx.x; i < nvec; i += 256) {
        const int row = (i * 8) / D;               // kv row in tile
        const int col = (i * 8) % D;               // dim
        const int kvp = kv0 + row;
        const int tok = tok0 + min(kvp, slen - 1);
        short8_t kd = *reinterpret_cast<const short8_t*>(
            k + (int64_t)tok * kv_stride + kvh * D + col);
        // swizzle byte offset within row: 16B-granular XOR of row bits
        const int bo = col * 2;
        const int swz = bo ^ ((row & 7) << 4);
        *reinterpret_cast<short8_t*>(&lds_k[row * D + swz / 2]) = kd;
        // V: transpose into Vt[d][kv]
        short8_t vd = *reinterpret_cast<const short8_t*>(
            v + (int64_t)tok * kv_stride + kvh * D + col);
#pragma unroll
        for (int j = 0; j < 8; j++) lds_vt[(col + j) * VT_PAD + row] = vd[j];
      }
    }
    __syncthreads();

    // ---- S = Q·K^T for this wave's 16 q rows × 32 kv cols ----
    f32x4 sfrag[2];
#pragma unroll
    for (int kt = 0; kt < 2; kt++) {
      sfrag[kt] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < KK; kk++) {
        // B-frag from K row (col=lo → kv row kv0+kt*16+lo), k=hi*8
        const int krow = kt * 16 + lo;
        const int bo = (kk * 32 + hi * 8) * 2;
        const int swz = bo ^ ((krow & 7) << 4);
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(&lds_k[krow * D + swz / 2]);
        sfrag[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], kf, sfrag[kt], 0, 0, 0);
      }
    }

    // ---- mask + online softmax ----
    // lane holds (col=lo → kv = kv0+kt*16+lo, rows hi*4+r)
    float p[2][4];
#pragma unroll
    for (int kt = 0; kt < 2; kt++) {
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const int qr = q0 + wave * 16 + hi * 4 + r;
        const int kvp = kv0 + kt * 16 + lo;
        float sv = sfrag[kt][r] * scale;
        if (kvp > qr || kvp >= slen || qr >= slen ||
            (window > 0 && kvp <= qr - window)) sv = -1e30f;
        p[kt][r] = sv;
      }
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

    // ---- P → LDS (C/D layout → A layout via padded LDS tile) ----
    // lds_p[wave] is [16 rows][VT_PAD] bf16; lane writes rows hi*4+r, col:
    // kt*16+lo.
    {
      short* pl = lds_p[wave];
#pragma unroll
      for (int kt = 0; kt < 2; kt++)
#pragma unroll
        for (int r = 0; r < 4; r++)
          pl[(hi * 4 + r) * VT_PAD + kt * 16 + lo] = f32_to_bf16(p[kt][r]);
    }
    __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt(0): LDS writes visible to own wave
    // A-frag of P: row=lo, k(kv)=hi*8+[0..7] → contiguous in lds_p row.
    bf16x8 pfrag = *reinterpret_cast<const bf16x8*>(
        &lds_p[wave][lo * VT_PAD + hi * 8]);

    // ---- O += P·V : 8 d-tiles, B-frag from Vt ----
#pragma unroll
    for (int dt = 0; dt < DT; dt++) {
      // B[32kv,16d]: lane col=lo → d = dt*16+lo; k rows kv=hi*8+[0..7]
      bf16x8 vf = *reinterpret_cast<const bf16x8*>(
          &lds_vt[(dt * 16 + lo) * VT_PAD + hi * 8]);
      ofrag[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vf, ofrag[dt], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: out[q, qh, d] = O / l (+ sink mass in the denom) ----
#pragma unroll
  for (int r = 0; r < 4; r++) {
    const int qr = q0 + wave * 16 + hi * 4 + r;
    if (qr >= slen) continue;
    float den = lrow[r];
    if (sinks != nullptr) den += __expf(sinks[qh] - mrow[r]);
    const float inv = 1.f / fmaxf(den, 1e-20f);
    const int64_t obase = ((int64_t)(tok0 + qr) * QH + qh) * D;
#pragma unroll
    for (int dt = 0; dt < DT; dt++)
      out[obase + dt * 16 + lo] = f32_to_bf16(ofrag[dt][r] * inv);
  }
}

void prefill_attention(at::Tensor out, at::Tensor q, at::Tensor k, at::Tensor v,
                       at::Tensor tile_seq, at::Tensor tile_qbase,
                       at::Tensor cu_seqlens, double scale, int64_t window,
                       at::Tensor sinks) {
  const float* sink_ptr = nullptr;
  if (sinks.numel() > 0) {
    TORCH_CHECK(sinks.dtype() == at::kFloat && sinks.is_cuda() &&
                sinks.numel() == q.size(1));
    sink_ptr = sinks.data_ptr<float>();
  }
  TORCH_CHECK(q.is_cuda() && q.dtype() == at::kBFloat16);
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(q.stride(-1) == 1 && q.stride(1) == q.size(2));
  TORCH_CHECK(k.stride(-1) == 1 && k.stride(1) == k.size(2));
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v must share row stride");
  TORCH_CHECK(tile_seq.dtype() == at::kInt && cu_seqlens.dtype() == at::kInt);
  const int QH = q.size(1);
  const int D = q.size(2);
  const int KH = k.size(1);
  const int ntiles = tile_seq.size(0);
  TORCH_CHECK(QH % KH == 0);
  auto stream = at::hip::getCurrentHIPStream();
  if (ntiles == 0) return;
#define PF_LAUNCH(D_)                                                         \
  hipLaunchKernelGGL((prefill_attn_kernel<D_>), dim3(ntiles, QH), dim3(256),  \
      0, stream, (short*)out.data_ptr(), (const short*)q.data_ptr(),         \
      (const short*)k.data_ptr(), (const short*)v.data_ptr(),                \
      tile_seq.data_ptr<int>(), tile_qbase.data_ptr<int>(),                  \
      cu_seqlens.data_ptr<int>(), (float)scale, QH, KH,                       \
      q.stride(0), k.stride(0), (int)window, sink_ptr)
  switch (D) {
    case 256: PF_LAUNCH(256); break;
    case 128: PF_LAUNCH(128); break;
    case 96:  PF_LAUNCH(96);  break;
    case 64:  PF_LAUNCH(64);  break;
    default: TORCH_CHECK(false, "prefill: unsupported head_dim ", D);
  }
#undef PF_LAUNCH
}

}  // namespace kaito
