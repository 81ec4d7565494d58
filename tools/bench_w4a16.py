import sys, os; sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch, time
from kaito_amd.models.quant import QuantLinear
torch.manual_seed(0)
K, N = 4096, 6144   # llama-8b qkv-ish shape
lin = torch.nn.Linear(K, N, bias=False).to("cuda", torch.bfloat16)
ql = QuantLinear.from_float(lin.weight.data.float(), 128).to("cuda")
def t(f, n=200):
    for _ in range(20): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e6
for M in (1, 4, 8, 16, 32):
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    us_q = t(lambda: ql(x))
    us_b = t(lambda: lin(x))
    gb = (N*K/2 + M*K*2 + M*N*2)/1e9
    print(f"M={M:3d}  w4a16={us_q:7.1f}us ({gb/us_q*1e6:.2f} TB/s)  bf16={us_b:7.1f}us  speedup={us_b/us_q:.2f}x")
from kaito_amd import ops
for M in (64, 128, 256, 512, 1024):
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    us_f = t(lambda: ops.w4a16_gemm(x, ql.qweight, ql.scales, ql.zeros, 128))
    def dq():
        w = ops.w4a16_dequant(ql.qweight, ql.scales, ql.zeros, 128)
        return torch.nn.functional.linear(x, w)
    us_d = t(dq)
    us_b = t(lambda: lin(x))
    fl = 2 * M * N * K / 1e12
    print(f"M={M:4d}  fused={us_f:7.1f}us ({fl/us_f*1e6:.0f} TF)  "
          f"dequant+blaslt={us_d:7.1f}us  bf16={us_b:7.1f}us  "
          f"fused-vs-bf16={us_b/us_f:.2f}x")
