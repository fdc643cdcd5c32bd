import torch, math, sys
sys.path.insert(0, "/root/repo")
from megatron_amd.ops import ext
mod = ext.load(required=True)
B, S, H, D = 4, 4096, 32, 128
q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
out, lse = mod.flash_attn_fwd(q, k, v, True, 1.0/math.sqrt(D), -1)
dout = torch.randn_like(out)
for _ in range(5):
    mod.flash_attn_bwd(dout, q, k, v, out, lse, True, 1.0/math.sqrt(D), -1)
torch.cuda.synchronize()
