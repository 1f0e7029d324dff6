#!/usr/bin/env python3
"""CIN kernel diagnostics (run on a GPU box).

Checks each kernel against an EXACT-operand fp32 reference (same bf16
roundings the kernel performs, fp32 accumulation), so only accumulation
order separates them — tolerances are tight. One-hot weight tests pin the
index maps exactly."""

import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import torch  # noqa: E402

from openembedding_amd.ops import require_hip  # noqa: E402

DEV = "cuda:0"
ext = require_hip()
bf = torch.bfloat16


def pad_w(W, Kp, Op=None):
    O, K = W.shape
    wp = torch.zeros(O, Kp, dtype=bf, device=DEV)
    wp[:, :K] = W
    if Op is None:
        return wp
    wt = torch.zeros(Kp, Op, dtype=bf, device=DEV)
    wt[:K, :O] = W.t()
    return wp, wt


def vref(x0p, xkp, F, H):
    # kernel rounding: inputs staged bf16, fp32 product, bf16 store
    # (identical to the chunked-torch path's roundings)
    v = (x0p.to(bf).float().unsqueeze(2)
         * xkp.to(bf).float().unsqueeze(1))          # [N, F, H]
    return v.reshape(x0p.shape[0], F * H).to(bf).float()


def check(name, got, ref, atol=2e-3, rtol=1e-3):
    d = (got - ref).abs()
    rel = d / ref.abs().clamp(min=1e-6)
    ok = bool((d <= atol + rtol * ref.abs()).all())
    print(f"{name}: max_abs={d.max():.3e} max_rel={rel.max():.3e} "
          f"{'OK' if ok else 'FAIL'}")
    return ok


def main():
    torch.manual_seed(0)
    allok = True
    for (B, F, H, O, d) in [(32, 4, 8, 16, 1), (512, 26, 128, 128, 9),
                            (512, 26, 26, 128, 9), (104, 13, 64, 64, 4)]:
        N = B * d
        K = F * H
        Kp = (K + 31) // 32 * 32
        Op = (O + 31) // 32 * 32
        x0p = torch.randn(N, F, device=DEV)
        xkp = torch.randn(N, H, device=DEV)
        W = (torch.randn(O, K, device=DEV) * 0.05)
        wp, wt = pad_w(W, Kp, Op)
        wbf = wp[:, :K].float()

        # one-hot W: out[n][o] = V[n][o*step]
        step = max(1, K // O)
        W1 = torch.zeros(O, K, device=DEV)
        for o in range(O):
            W1[o, (o * step) % K] = 1.0
        wp1 = pad_w(W1, Kp)
        v = vref(x0p, xkp, F, H)
        out1 = ext.cin_fwd(x0p, xkp, wp1)
        ref1 = v[:, [(o * step) % K for o in range(O)]]
        allok &= check(f"fwd-onehot {B}x{F}x{H}x{O}x{d}", out1, ref1,
                       atol=1e-5, rtol=1e-5)

        # random W vs exact-operand fp32 GEMM
        out = ext.cin_fwd(x0p, xkp, wp)
        ref = v @ wbf.t()
        allok &= check(f"fwd-rand   {B}x{F}x{H}x{O}x{d}", out, ref)

        # dx kernel: P = W^T dZ then contractions, exact-operand fp32 ref
        doutp = torch.randn(N, O, device=DEV)
        dx0, dxk = ext.cin_dx(doutp, wt, x0p, xkp)
        dz = doutp.to(bf).float()
        p = (dz @ wbf).view(N, F, H)       # fp32 accum of bf16 operands
        rx0 = (p * xkp.unsqueeze(1)).sum(2)
        rxk = (p * x0p.unsqueeze(2)).sum(1)
        allok &= check(f"dx0        {B}x{F}x{H}x{O}x{d}", dx0, rx0,
                       atol=5e-2, rtol=2e-2)
        allok &= check(f"dxk        {B}x{F}x{H}x{O}x{d}", dxk, rxk,
                       atol=5e-2, rtol=2e-2)

        # dW kernel vs exact-operand fp32
        dzt = doutp.t().contiguous().to(bf)
        x0t = x0p.t().contiguous().to(bf)
        xkt = xkp.t().contiguous().to(bf)
        dw = ext.cin_dw(dzt, x0t, xkt, O, 8)
        bref = (x0t.float().unsqueeze(1) * xkt.float().unsqueeze(0)
                ).to(bf).float()           # [F, H, N] rounded like kernel
        rdw = torch.einsum("on,fhn->ofh", dzt.float(),
                           bref).reshape(O, K)
        allok &= check(f"dW         {B}x{F}x{H}x{O}x{d}", dw, rdw,
                       atol=1e-2, rtol=2e-2)
    print("ALL OK" if allok else "FAILURES PRESENT")
    sys.exit(0 if allok else 1)


if __name__ == "__main__":
    main()
