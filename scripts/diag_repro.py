import sys, torch, math
sys.path.insert(0, ".")
import kaito_amd.ops as ops
ops.load_extension()
dev = "cuda"
T, KH, G, D, BS = 8, 2, 4, 128, 16
L = 64
mb = (L + BS - 1) // BS
NB = T * mb + 1
kc = torch.randn(NB, KH, BS, D, device=dev).to(torch.bfloat16)
vc = torch.randn_like(kc)
bt = torch.arange(1, T * mb + 1, dtype=torch.int32, device=dev).reshape(T, mb)
sl = torch.full((T,), L, dtype=torch.int32, device=dev)
q = torch.randn(T, KH * G, D, device=dev).to(torch.bfloat16)
print("attn...", flush=True)
o = ops.paged_attention(q, kc, vc, bt, sl, 0.088)
torch.cuda.synchronize(); print("attn ok", flush=True)
outd = torch.empty(T * KH, dtype=torch.float32, device=dev)
print("diag...", flush=True)
torch.ops.kaito.paged_read_bw(outd, kc, vc, bt, sl)
torch.cuda.synchronize(); print("diag ok", outd.sum().item(), flush=True)
