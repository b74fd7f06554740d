import sys
sys.path.insert(0, "/root/repo")
import torch
from photon_amd.ops.attention import alibi_slopes, flash_attention
q = torch.randn(32,12,2048,64, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
slopes = alibi_slopes(12).to("cuda")
for _ in range(5):
    with torch.no_grad():
        flash_attention(q,k,v,slopes,causal=True,impl="flash")
torch.cuda.synchronize()
