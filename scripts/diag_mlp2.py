"""Precise stage-level check of mlp3_fwd/bwd outputs (GPU box)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from openembedding_amd.ops import require_hip

ext = require_hip()
DEV = "cuda:0"
torch.manual_seed(0)
M, K0, K0p, H = 2048, 247, 256, 400
Hp = 416

x0 = torch.zeros(M, K0p, device=DEV, dtype=torch.bfloat16)
x0[:, :K0] = (torch.randn(M, K0, device=DEV) * 0.5).to(torch.bfloat16)
w1p = torch.zeros(H, K0p, device=DEV, dtype=torch.bfloat16)
w1p[:, :K0] = (torch.randn(H, K0, device=DEV) * 0.05).to(torch.bfloat16)
w2 = (torch.randn(H, H, device=DEV) * 0.05).to(torch.bfloat16)
w3 = (torch.randn(H, H, device=DEV) * 0.05).to(torch.bfloat16)
w4 = (torch.randn(H, device=DEV) * 0.05).to(torch.bfloat16)
bs = [(torch.randn(H, device=DEV) * 0.1).to(torch.bfloat16) for _ in range(3)]
b4 = (torch.randn(1, device=DEV) * 0.1).to(torch.bfloat16)
w2p = torch.zeros(H, Hp, device=DEV, dtype=torch.bfloat16); w2p[:, :H] = w2
w3p = torch.zeros(H, Hp, device=DEV, dtype=torch.bfloat16); w3p[:, :H] = w3

out, a1, a2, a3 = ext.mlp3_fwd(x0, w1p, bs[0], w2p, bs[1], w3p, bs[2], w4, b4)


def check(name, got, ref, atol=2e-2):
    d = (got.float() - ref.float()).abs()
    bad = (d > atol + 2e-2 * ref.float().abs()).sum().item()
    zr_g = (got.float().abs().sum(1) == 0).sum().item()
    zr_r = (ref.float().abs().sum(1) == 0).sum().item() if ref.dim() == 2 else 0
    print(f"{name}: maxdiff={d.max().item():.4f} bad={bad}/{got.numel()} "
          f"zero_rows got={zr_g} ref={zr_r}")


check("a1", a1, torch.relu(x0 @ w1p.t() + bs[0]))
check("a2", a2, torch.relu(a1 @ w2.t() + bs[1]))
check("a3", a3, torch.relu(a2 @ w3.t() + bs[2]))
check("out", out.unsqueeze(1), (a3 @ w4 + b4).float().unsqueeze(1), atol=3e-2)

dout = torch.randn(M, device=DEV)
w3tp = torch.zeros(H, Hp, device=DEV, dtype=torch.bfloat16); w3tp[:, :H] = w3.t()
w2tp = torch.zeros(H, Hp, device=DEV, dtype=torch.bfloat16); w2tp[:, :H] = w2.t()
w1tp = torch.zeros(K0p, Hp, device=DEV, dtype=torch.bfloat16)
w1tp[:, :H] = w1p.t()
dx0, dz1, dz2, dz3 = ext.mlp3_bwd(dout, a1, a2, a3, w4, w3tp, w2tp, w1tp)

rz3 = (dout.unsqueeze(1).to(torch.bfloat16) * w4) * (a3 > 0)
rz2 = ((rz3 @ w3) * (a2 > 0)).to(torch.bfloat16)
rz1 = ((rz2 @ w2) * (a1 > 0)).to(torch.bfloat16)
rx0 = (rz1 @ w1p).to(torch.bfloat16)
check("dz3", dz3, rz3)
check("dz2", dz2, rz2)
check("dz1", dz1, rz1)
check("dx0", dx0, rx0)
print("done")
