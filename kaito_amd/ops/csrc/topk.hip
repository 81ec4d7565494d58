// Top-K selection kernel for gfx950 — the selection half of flat k-NN
// search (distances come from an MFMA GEMM; this kernel selects the K best
// per query row). Replaces faiss-cpu's CPU selection in the reference's
// RAG engine (SURVEY.md §2.3 "Embedding forward + FAISS" row).
//
// One workgroup per query row; each thread keeps its K best over a strided
// scan in registers (insertion-sorted, K<=32), then K rounds of block-wide
// argmax over the 256*K candidates in LDS.
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

template <int K>
__global__ void topk_kernel(
    float* __restrict__ out_vals,    // [Q, K]
    int* __restrict__ out_idx,       // [Q, K]
    const float* __restrict__ scores,// [Q, N] (higher = better)
    const int N, const int k) {      // k <= K: rounds actually emitted
  const int q = blockIdx.x;
  const float* row = scores + (int64_t)q * N;
  // thread-local top-K (descending)
  float v[K];
  int ix[K];
#pragma unroll
  for (int i = 0; i < K; i++) { v[i] = -1e30f; ix[i] = -1; }
  for (int j = threadIdx.x; j < N; j += blockDim.x) {
    const float s = row[j];
    if (s > v[K - 1]) {
      int p = K - 1;
      while (p > 0 && v[p - 1] < s) { v[p] = v[p - 1]; ix[p] = ix[p - 1]; --p; }
      v[p] = s; ix[p] = j;
    }
  }
  __shared__ float lv[256 * K > 16384 ? 1 : 256 * K];
  __shared__ int li[256 * K > 16384 ? 1 : 256 * K];
  static_assert(256 * K <= 16384, "K too large for LDS merge");
#pragma unroll
  for (int i = 0; i < K; i++) {
    lv[threadIdx.x * K + i] = v[i];
    li[threadIdx.x * K + i] = ix[i];
  }
  __syncthreads();
  // K rounds of block argmax (each round: 256 threads scan their own slot
  // head; tree-reduce in LDS).
  __shared__ float rv[256];
  __shared__ int ri[256];
  __shared__ int head[256];
  head[threadIdx.x] = 0;
  __syncthreads();
  for (int round = 0; round < k; round++) {
    const int h = head[threadIdx.x];
    rv[threadIdx.x] = (h < K) ? lv[threadIdx.x * K + h] : -1e30f;
    ri[threadIdx.x] = threadIdx.x;
    __syncthreads();
    for (int off = 128; off > 0; off >>= 1) {
      if (threadIdx.x < off) {
        if (rv[threadIdx.x + off] > rv[threadIdx.x]) {
          rv[threadIdx.x] = rv[threadIdx.x + off];
          ri[threadIdx.x] = ri[threadIdx.x + off];
        }
      }
      __syncthreads();
    }
    if (threadIdx.x == 0) {
      const int winner = ri[0];
      out_vals[(int64_t)q * k + round] = rv[0];
      out_idx[(int64_t)q * k + round] = li[winner * K + head[winner]];
      head[winner] += 1;
    }
    __syncthreads();
  }
}

void topk(at::Tensor out_vals, at::Tensor out_idx, at::Tensor scores,
          int64_t k) {
  TORCH_CHECK(scores.is_cuda() && scores.dtype() == at::kFloat);
  TORCH_CHECK(scores.is_contiguous());
  const int Q = scores.size(0);
  const int N = scores.size(1);
  TORCH_CHECK(k <= 32 && k >= 1, "topk supports k in [1,32]");
  TORCH_CHECK(k <= N);
  auto stream = at::hip::getCurrentHIPStream();
#define TK_LAUNCH(K_)                                                          \
  hipLaunchKernelGGL((topk_kernel<K_>), dim3(Q), dim3(256), 0, stream,        \
      out_vals.data_ptr<float>(), out_idx.data_ptr<int>(),                    \
      scores.data_ptr<float>(), N, (int)k)
  if (k <= 1) TK_LAUNCH(1);
  else if (k <= 2) TK_LAUNCH(2);
  else if (k <= 4) TK_LAUNCH(4);
  else if (k <= 8) TK_LAUNCH(8);
  else if (k <= 16) TK_LAUNCH(16);
  else TK_LAUNCH(32);
#undef TK_LAUNCH
}

}  // namespace kaito
