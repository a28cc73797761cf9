#!/usr/bin/env python
"""Attention backward microbench at the production shape (for rocprofv3
PMC collection runs): B=4, S=4096, H=64, D=128, bf16."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch

from lpp_amd import ops

ext = ops.extension()
B, S, H, D = 4, 4096, 64, 128
q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
k, v, do_ = torch.randn_like(q), torch.randn_like(q), torch.randn_like(q)
o, lse = ext.attention_fwd(q, k, v)
for _ in range(5):
    ext.attention_bwd(do_, q, k, v, o, lse)
torch.cuda.synchronize()
print("attn bwd probe done")
