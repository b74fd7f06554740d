"""Attention bug localization, round 3: which slope does head h>0 use?"""
import math
import sys

sys.path.insert(0, "/root/repo")
import torch
from photon_amd.ops import hip_ext
from photon_amd.ops.attention import reference_attention_fp32

ext = hip_ext()
dev = "cuda:0"
torch.manual_seed(0)

B, H, S, D = 1, 2, 64, 64
q = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
k = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
v = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)

slopes = torch.tensor([0.25, 0.0625])
o, lse = ext.attn_fwd(q, k, v, slopes, False)

for h in range(H):
    for trial, s in (("own", slopes[h]), ("s0", slopes[0]), ("s1", slopes[1]),
                     ("zero", torch.tensor(0.0))):
        ref = reference_attention_fp32(
            q[:, h : h + 1], k[:, h : h + 1], v[:, h : h + 1],
            s.reshape(1), causal=False,
        )
        e = (o[:, h : h + 1].float() - ref).abs().max().item()
        print(f"head {h} vs slope[{trial}]: err {e:.4f}")

# equal slopes control
se = torch.tensor([0.25, 0.25])
o2, _ = ext.attn_fwd(q, k, v, se, False)
ref2 = reference_attention_fp32(q, k, v, se, causal=False)
print("equal slopes err:", (o2.float() - ref2).abs().max().item())

# B=2 H=1: batch indexing control
q3 = torch.randn(2, 1, S, D, device=dev, dtype=torch.bfloat16)
k3 = torch.randn(2, 1, S, D, device=dev, dtype=torch.bfloat16)
v3 = torch.randn(2, 1, S, D, device=dev, dtype=torch.bfloat16)
o3, _ = ext.attn_fwd(q3, k3, v3, torch.tensor([0.25]), False)
ref3 = reference_attention_fp32(q3, k3, v3, torch.tensor([0.25]), causal=False)
print("B=2 H=1 err:", (o3.float() - ref3).abs().max().item())
