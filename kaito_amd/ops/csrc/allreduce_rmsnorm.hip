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
