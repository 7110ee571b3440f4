"""Empirically determine the gfx950 v_mfma_f32_16x16x32_bf16 fragment maps.

Runs one MFMA on prepacked per-lane fragments and checks which
(A-pack, B-pack, C-unpack) candidate triple reproduces A @ B. Run on a GPU
box:  python tools/mfma_probe.py
"""

import itertools
import sys

import torch

sys.path.insert(0, ".")
from agentainer_amd import ops  # noqa: E402

assert torch.cuda.is_available()
mod = ops._load_hip()

torch.manual_seed(0)
A = torch.randn(16, 32, dtype=torch.float32)
B = torch.randn(32, 16, dtype=torch.float32)
Ab = A.to(torch.bfloat16)
Bb = B.to(torch.bfloat16)
want = Ab.float() @ Bb.float()

# candidate packings: lane l, element j -> (row, k)
A_PACKS = {
    "row=l%16,k=(l/16)*8+j": lambda l, j: (l % 16, (l // 16) * 8 + j),
    "row=l%16,k=(l/16)+4*j": lambda l, j: (l % 16, (l // 16) + 4 * j),
    "row=l/4, k=(l%4)*8+j": lambda l, j: (l // 4, (l % 4) * 8 + j),
    "row=l%16,k=(l/16)*4+j%4+16*(j/4)": lambda l, j: (l % 16, (l // 16) * 4 + j % 4 + 16 * (j // 4)),
}
B_PACKS = {
    "k=(l/16)*8+j,col=l%16": lambda l, j: ((l // 16) * 8 + j, l % 16),
    "k=(l/16)+4*j,col=l%16": lambda l, j: ((l // 16) + 4 * j, l % 16),
    "k=(l%4)*8+j,col=l/4": lambda l, j: ((l % 4) * 8 + j, l // 4),
    "k=(l/16)*4+j%4+16*(j/4),col=l%16": lambda l, j: ((l // 16) * 4 + j % 4 + 16 * (j // 4), l % 16),
}
C_UNPACKS = {
    "row=(l/16)*4+r,col=l%16": lambda l, r: ((l // 16) * 4 + r, l % 16),
    "row=l%16,col=(l/16)*4+r": lambda l, r: (l % 16, (l // 16) * 4 + r),
}


def pack(mat, fn):
    out = torch.zeros(64, 8, dtype=torch.bfloat16)
    for l in range(64):
        for j in range(8):
            r, c = fn(l, j)
            out[l, j] = mat[r, c] if mat.shape[0] > r else 0
    return out


results = []
for an, af in A_PACKS.items():
    for bn, bf in B_PACKS.items():
        a = pack(Ab, af).cuda()
        b = pack(Bb, bf).cuda()  # bf returns (k, col) indices into B [32,16]
        c = torch.zeros(64, 4, dtype=torch.float32, device="cuda")
        mod.mfma_probe(c, a.view(-1).contiguous(), b.view(-1).contiguous())
        c = c.cpu()
        for cn, cf in C_UNPACKS.items():
            D = torch.zeros(16, 16)
            for l in range(64):
                for r in range(4):
                    rr, cc = cf(l, r)
                    D[rr, cc] = c[l, r]
            diff = (D - want).abs().max().item()
            results.append((diff, an, bn, cn))

results.sort()
for diff, an, bn, cn in results[:6]:
    print(f"diff={diff:10.5f}  A[{an}]  B[{bn}]  C[{cn}]")
