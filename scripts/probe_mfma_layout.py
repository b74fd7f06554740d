"""Determine the real gfx950 mfma_f32_32x32x16_bf16 operand layouts.

Runs mfma_probe (fragments loaded as lane l elem j -> mem col 8*(l>>5)+j)
and checks which candidate k-map reproduces the observed D.
"""
import itertools
import torch
from photon_amd.ops import hip_ext

ext = hip_ext()
dev = "cuda:0"
torch.manual_seed(0)
A = torch.randint(-3, 4, (32, 16), device=dev).to(torch.bfloat16)   # A[i][k]
BT = torch.randint(-3, 4, (32, 16), device=dev).to(torch.bfloat16)  # BT[j][k]
D = ext.mfma_probe(A.contiguous(), BT.contiguous())

Af = A.float().cpu()
BTf = BT.float().cpu()
Dc = D.cpu()

def kmap_contig(hi, j):     # k = 8*hi + j
    return 8 * hi + j
def kmap_stacked(hi, j):    # k = 4*hi + (j&3) + 8*(j>>2)
    return 4 * hi + (j & 3) + 8 * (j >> 2)
def kmap_interleave(hi, j): # k = hi + 2*j  (unlikely)
    return hi + 2 * j

cands = {"contig": kmap_contig, "stacked": kmap_stacked, "inter": kmap_interleave}

for an, afn in cands.items():
    for bn, bfn in cands.items():
        # reconstruct effective A_hw, B_hw given our load pattern
        A_hw = torch.zeros(32, 16)
        B_hw = torch.zeros(16, 32)
        for l in range(64):
            hi, lq = l >> 5, l & 31
            for j in range(8):
                A_hw[lq, afn(hi, j)] = Af[lq, 8 * hi + j]
                B_hw[bfn(hi, j), lq] = BTf[lq, 8 * hi + j]
        D_pred = A_hw @ B_hw
        err = (D_pred - Dc).abs().max().item()
        if err < 1e-3:
            print(f"MATCH: A={an} B={bn} err={err}")
        else:
            print(f"  no: A={an} B={bn} err={err:.1f}")

# pack probe: which (row, col) value lands in each frag slot
P = ext.pack_probe(torch.zeros(1, device=dev)).cpu()
print("pack_probe lane 0..3, 32..35 (value = row*64 + col):")
for l in (0, 1, 32, 33):
    vals = [int(v) for v in P[l]]
    decoded = [(v // 64, v % 64) for v in vals]
    print(f" lane {l}: {decoded}")
