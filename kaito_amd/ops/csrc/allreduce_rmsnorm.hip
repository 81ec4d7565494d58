// Fused one-shot all-reduce + RMSNorm for TP decode over xGMI.
//
// SURVEY.md hard part #2 / §8: on the 8-GPU MI355X mesh every GPU has a
// direct link to every peer, so for the SMALL tensors of a decode step
// ([T<=1024, H] bf16) a one-shot all-reduce — each rank reads all N peer
// buffers directly over xGMI and reduces locally — beats a ring (which
// serializes 2(N-1) per-link hops). Fusing the RMSNorm epilogue removes
// one full read+write of the hidden states (the optimization the
// reference explicitly disables on NVIDIA, interface.go:439-446).
//
// The kernel takes a device array of N peer base pointers. On a real TP
// group those are hipIpc-mapped peer buffers (parallel/one_shot.py wires
// the handle exchange); the single-GPU numerics test passes N local
// buffers — the kernel is identical either way.
#include "common.h"
#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

namespace kaito {

// one workgroup per row; 256 threads stride H.
__global__ __launch_bounds__(256)
void allreduce_rmsnorm_kernel(short* __restrict__ out,       // [T, H]
                              const uint64_t* __restrict__ ptrs,  // [N]
                              const short* __restrict__ weight,   // [H]
                              float eps, int N, int H) {
  const int row = blockIdx.x;
  extern __shared__ float s_red[];
  // pass 1: sum across ranks, accumulate sum of squares
  float ss = 0.f;
  for (int h = threadIdx.x; h < H; h += blockDim.x) {
    float v = 0.f;
    for (int r = 0; r < N; r++) {
      const short* src = reinterpret_cast<const short*>(ptrs[r]);
      v += bf16_to_f32(src[(int64_t)row * H + h]);
    }
    // stash the reduced value in out (bf16) for pass 2; keep f32 in ss
    out[(int64_t)row * H + h] = f32_to_bf16(v);
    ss += v * v;
  }
  ss = block_reduce_sum(ss, s_red);
  __shared__ float s_scale;
  if (threadIdx.x == 0) s_scale = rsqrtf(ss / H + eps);
  __syncthreads();
  const float scale = s_scale;
  for (int h = threadIdx.x; h < H; h += blockDim.x) {
    const float v = bf16_to_f32(out[(int64_t)row * H + h]);
    out[(int64_t)row * H + h] =
        f32_to_bf16(v * scale * bf16_to_f32(weight[h]));
  }
}

void allreduce_rmsnorm(at::Tensor out, at::Tensor ptrs, at::Tensor weight,
                       double eps) {
  const int T = out.size(0), H = out.size(1), N = ptrs.size(0);
  TORCH_CHECK(out.dtype() == at::kBFloat16 && out.is_contiguous());
  TORCH_CHECK(ptrs.dtype() == at::kLong && ptrs.is_cuda());
  dim3 grid(T), block(256);
  auto stream = at::hip::getCurrentHIPStream();
  const int smem = 32 * sizeof(float);
  allreduce_rmsnorm_kernel<<<grid, block, smem, stream>>>(
      (short*)out.data_ptr(), (const uint64_t*)ptrs.data_ptr(),
      (const short*)weight.data_ptr(), (float)eps, N, H);
}

// ======================================================================
// Graph-capturable one-shot all-reduce (+residual) + RMSNorm.
//
// The v1 kernel above needs host-side torch.distributed barriers around
// it, which (a) cost ~100us of host latency per call and (b) make it
// impossible to capture the decode step into a hipGraph. This variant
// fuses the synchronization INTO the kernel (vLLM custom-allreduce
// style): each rank owns an IPC-mapped signal page; block `row` of the
// kernel bumps a device-memory epoch counter, release-stores the epoch
// into every peer's signal slot [row][self] over xGMI, and acquire-spins
// on its own slots until all peers arrive. Combined with DOUBLE-BUFFERED
// staging on the Python side (a peer entering call k+1 has necessarily
// finished reading call k's buffer — kernels on one stream serialize),
// one in-kernel barrier per call suffices and no host sync is needed:
// the whole decode step, collectives included, captures into a hipGraph.
//
// The epoch lives in device memory and is incremented BY the kernel, so
// graph replays keep advancing it. Epoch comparison uses signed distance
// so the uint32 wrap is harmless.
__device__ __forceinline__ void one_shot_barrier(
    const uint64_t* __restrict__ sig_ptrs, int rank, int N, int row,
    uint32_t* __restrict__ counter) {
  __shared__ uint32_t s_flag;
  if (threadIdx.x == 0) s_flag = ++counter[row];  // single writer per row
  __syncthreads();
  const uint32_t flag = s_flag;
  if ((int)threadIdx.x < N) {
    uint32_t* peer = reinterpret_cast<uint32_t*>(sig_ptrs[threadIdx.x]);
    __hip_atomic_store(&peer[row * 8 + rank], flag, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_SYSTEM);
    uint32_t* own = reinterpret_cast<uint32_t*>(sig_ptrs[rank]);
    while ((int32_t)(__hip_atomic_load(&own[row * 8 + threadIdx.x],
                                       __ATOMIC_ACQUIRE,
                                       __HIP_MEMORY_SCOPE_SYSTEM) - flag) < 0) {
      __builtin_amdgcn_s_sleep(1);
    }
  }
  __syncthreads();
}

// one workgroup per row; vectorized short8 (8 bf16 / 16B per lane-step).
__global__ __launch_bounds__(256)
void one_shot_ar_rmsnorm_kernel(
    short* __restrict__ out,             // [T, H]
    short* __restrict__ residual,        // [T, H] in-place or nullptr
    const uint64_t* __restrict__ ptrs,   // [N] staging base ptrs
    const uint64_t* __restrict__ sig_ptrs,  // [N] signal base ptrs
    uint32_t* __restrict__ counter,      // [max_rows] local epochs
    const short* __restrict__ weight,    // [H]
    float eps, int N, int H, int rank) {
  const int row = blockIdx.x;
  one_shot_barrier(sig_ptrs, rank, N, row, counter);

  extern __shared__ float s_red[];
  float ss = 0.f;
  for (int h8 = threadIdx.x * 8; h8 < H; h8 += blockDim.x * 8) {
    float v[8];
#pragma unroll
    for (int j = 0; j < 8; j++) v[j] = 0.f;
    for (int r = 0; r < N; r++) {
      const short8_t sv = *reinterpret_cast<const short8_t*>(
          reinterpret_cast<const short*>(ptrs[r]) + (int64_t)row * H + h8);
#pragma unroll
      for (int j = 0; j < 8; j++) v[j] += bf16_to_f32(sv[j]);
    }
    short8_t stash;
    if (residual != nullptr) {
      short8_t* res = reinterpret_cast<short8_t*>(
          residual + (int64_t)row * H + h8);
      const short8_t rv = *res;
#pragma unroll
      for (int j = 0; j < 8; j++) {
        // residual stream stays bf16 (matches fused_add_rms_norm)
        stash[j] = f32_to_bf16(v[j] + bf16_to_f32(rv[j]));
        const float rf = bf16_to_f32(stash[j]);
        ss += rf * rf;
      }
      *res = stash;
    } else {
#pragma unroll
      for (int j = 0; j < 8; j++) {
        stash[j] = f32_to_bf16(v[j]);
        const float rf = bf16_to_f32(stash[j]);
        ss += rf * rf;
      }
      *reinterpret_cast<short8_t*>(out + (int64_t)row * H + h8) = stash;
    }
  }
  ss = block_reduce_sum(ss, s_red);
  __shared__ float s_scale;
  if (threadIdx.x == 0) s_scale = rsqrtf(ss / H + eps);
  __syncthreads();
  const float scale = s_scale;
  const short* src = residual != nullptr ? residual : out;
  for (int h8 = threadIdx.x * 8; h8 < H; h8 += blockDim.x * 8) {
    const short8_t rv = *reinterpret_cast<const short8_t*>(
        src + (int64_t)row * H + h8);
    const short8_t wv = *reinterpret_cast<const short8_t*>(weight + h8);
    short8_t o;
#pragma unroll
    for (int j = 0; j < 8; j++)
      o[j] = f32_to_bf16(bf16_to_f32(rv[j]) * scale * bf16_to_f32(wv[j]));
    *reinterpret_cast<short8_t*>(out + (int64_t)row * H + h8) = o;
  }
}

void one_shot_ar_rmsnorm(at::Tensor out, at::Tensor residual,
                         at::Tensor ptrs, at::Tensor sig_ptrs,
                         at::Tensor counter, at::Tensor weight, double eps,
                         int64_t rank) {
  const int T = out.size(0), H = out.size(1), N = ptrs.size(0);
  TORCH_CHECK(out.dtype() == at::kBFloat16 && out.is_contiguous());
  TORCH_CHECK(H % (256 * 8) == 0 || H % 8 == 0, "H must be 8-aligned");
  TORCH_CHECK(ptrs.dtype() == at::kLong && ptrs.is_cuda());
  TORCH_CHECK(sig_ptrs.dtype() == at::kLong && sig_ptrs.is_cuda());
  TORCH_CHECK(counter.dtype() == at::kInt && counter.size(0) >= T,
              "epoch counter smaller than row count");
  short* res = nullptr;
  if (residual.numel() > 0) {
    TORCH_CHECK(residual.is_contiguous() && residual.sizes() == out.sizes());
    res = (short*)residual.data_ptr();
  }
  dim3 grid(T), block(256);
  auto stream = at::hip::getCurrentHIPStream();
  const int smem = 32 * sizeof(float);
  one_shot_ar_rmsnorm_kernel<<<grid, block, smem, stream>>>(
      (short*)out.data_ptr(), res, (const uint64_t*)ptrs.data_ptr(),
      (const uint64_t*)sig_ptrs.data_ptr(), (uint32_t*)counter.data_ptr(),
      (const short*)weight.data_ptr(), (float)eps, N, H, (int)rank);
}

// ---- hipIpc plumbing for cross-process peer mapping --------------------
// (multi-GPU TP: each rank shares its buffer handle; peers open it and
// pass the mapped pointer into allreduce_rmsnorm's ptrs array. Exchange
// happens over torch.distributed all_gather_object in parallel/one_shot.py.)
at::Tensor ipc_handle(at::Tensor t) {
  TORCH_CHECK(t.is_cuda());
  hipIpcMemHandle_t h;
  HIP_CHECK_KAITO(hipIpcGetMemHandle(&h, t.data_ptr()));
  auto out = at::empty({(int64_t)sizeof(h)}, at::kByte);
  memcpy(out.data_ptr(), &h, sizeof(h));
  return out;
}

int64_t ipc_open(at::Tensor handle_bytes) {
  TORCH_CHECK(handle_bytes.numel() == (int64_t)sizeof(hipIpcMemHandle_t));
  hipIpcMemHandle_t h;
  memcpy(&h, handle_bytes.data_ptr(), sizeof(h));
  void* ptr = nullptr;
  HIP_CHECK_KAITO(hipIpcOpenMemHandle(&ptr, h,
                                      hipIpcMemLazyEnablePeerAccess));
  return (int64_t)ptr;
}

void ipc_close(int64_t ptr) {
  HIP_CHECK_KAITO(hipIpcCloseMemHandle((void*)ptr));
}

}  // namespace kaito
