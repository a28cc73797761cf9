import torch
from lpp_amd import ops
ext = ops.extension()
B, S, H, D = 1, 4096, 64, 128
q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q); do = torch.randn_like(q)
o, lse2 = ext.attention_fwd(q, k, v)
for _ in range(20):
    ext.attention_fwd(q, k, v)
    ext.attention_bwd(do, q, k, v, o, lse2)
torch.cuda.synchronize()
