import sys, os, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from kaito_amd import ops
ops.load_extension()
dev = "cuda"
import sys as _s
# mixtral-8x7b shapes: decode T=256 (TK=512) or prefill T=8192
T = int(_s.argv[1]) if len(_s.argv) > 1 else 256
K_TOP, E, H, IE = 2, 8, 4096, 14336
TK = T * K_TOP
torch.manual_seed(0)
x = (torch.randn(T, H, device=dev) * 0.1).to(torch.bfloat16)
wgu = (torch.randn(E, 2 * IE, H, device=dev) * 0.02).to(torch.bfloat16)
wd = (torch.randn(E, H, IE, device=dev) * 0.02).to(torch.bfloat16)
# balanced routing
flat_e = torch.arange(TK, device=dev) % E
order = torch.argsort(flat_e)
sorted_tok = (torch.arange(TK, device=dev) // K_TOP).int()[order]
gates = torch.full((TK,), 0.5, device=dev)
counts = torch.bincount(flat_e, minlength=E).int()
offsets = torch.zeros(E + 1, dtype=torch.int32, device=dev)
torch.cumsum(counts, 0, out=offsets[1:].long()) if False else None
offsets[1:] = torch.cumsum(counts, 0)
pad = (TK + 63) // 64 * 64 + 64
act = torch.empty(pad, IE, dtype=torch.bfloat16, device=dev)
out32 = torch.zeros(T, H, dtype=torch.float32, device=dev)

def t(f, n=50):
    for _ in range(10): f()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e6

us_a = t(lambda: ops.moe_gate_silu(act, x, wgu, sorted_tok, offsets, 0, E))
us_b = t(lambda: ops.moe_down_scatter(out32, act, wd, sorted_tok, gates, offsets, 0, E))
gb_a = E * 2 * IE * H * 2 / 1e9
gb_b = E * H * IE * 2 / 1e9
print(f"gate_silu: {us_a:.1f} us  weightBW={gb_a/us_a*1e6:.2f} TB/s (floor ~{gb_a/6.0*1e3:.0f} us)")
print(f"down_scat: {us_b:.1f} us  weightBW={gb_b/us_b*1e6:.2f} TB/s (floor ~{gb_b/6.0*1e3:.0f} us)")
# reference: hipBLASLt per-expert GEMMs on the same shapes
xs = x[sorted_tok.long()]
def blaslt():
    o = torch.empty(TK, 2 * IE, dtype=torch.bfloat16, device=dev)
    for e in range(E):
        s, t2 = int(offsets[e]), int(offsets[e + 1])
        torch.matmul(xs[s:t2], wgu[e].T, out=o[s:t2])
    return o
print(f"blaslt gate_up loop: {t(blaslt):.1f} us")
