"""CTR model zoo: LR, WDL, DeepFM, xDeepFM.

The reference trains these via DeepCTR/Keras (examples/criteo_deepctr_*.py,
test/benchmark/criteo_deepctr.py); here they are native torch modules whose
sparse features go through the PS-backed CombinedEmbedding (one fused pull
per step for all 26 fields — see openembedding_amd/torch/__init__.py).

All models take (dense [B,13] float32, sparse [B,26] int64 per-field ids)
and return logits [B].
"""

from __future__ import annotations

import math
from typing import List, Optional, Sequence

import torch
import torch.nn as nn

from ..ops import hip_available
from ..torch import CombinedEmbedding
from .criteo import CRITEO_FIELD_VOCABS, N_DENSE


class _FusedCTRHeadFn(torch.autograd.Function):
    """Fused interaction head (ops/csrc/ctrhead.hip): one kernel assembles
    deep_in (cast fused) and computes first-order + FM + dense-linear
    partial logits; one kernel does the whole backward. Replaces ~20 small
    elementwise/reduce launches per step (profiles/)."""

    @staticmethod
    def forward(ctx, e_all, dense, w, b, use_fm, out_bf16, pad32=False):
        from ..ops import require_hip
        ext = require_hip()
        wf = w.reshape(-1).contiguous()
        e_all = e_all.contiguous()
        dense = dense.contiguous()
        deep_in, partial, s = ext.ctr_head_fwd(
            e_all, dense, wf, b.reshape(-1).contiguous(), use_fm, out_bf16,
            pad32)
        ctx.save_for_backward(e_all, dense, wf, s)
        ctx.use_fm = use_fm
        return deep_in, partial

    @staticmethod
    def backward(ctx, d_deep_in, d_partial):
        from ..ops import require_hip
        ext = require_hip()
        e_all, dense, wf, s = ctx.saved_tensors
        de_all, d_dense, dw, db = ext.ctr_head_bwd(
            e_all, dense, wf, d_deep_in.contiguous(),
            d_partial.contiguous(), s, ctx.use_fm)
        return de_all, d_dense, dw.view(1, -1), db, None, None, None


class _FusedMLP3Fn(torch.autograd.Function):
    """Whole 3-hidden-layer MLP forward in ONE bf16 MFMA kernel
    (ops/csrc/mlp.hip): activations flow layer-to-layer through LDS,
    bias+ReLU fused, hidden activations saved for backward. Backward:
    dgrad chain + wgrads via bf16 torch matmuls (wgrad shapes K=M are the
    library-friendly ones)."""

    @staticmethod
    def forward(ctx, x0, partial, w1, b1, w2, b2, w3, b3, w4, b4, bufs):
        from ..ops import require_hip
        ext = require_hip()
        # refresh padded weight copies (zero tails allocated once in bufs)
        # in ONE launch — three aten copy_ were ~3.8 us of launch/ramp
        # each for <1 MB moved (profiles/step_attrib_r2.txt)
        ext.mlp3_pack(w1, bufs["w1p"], False, w2, bufs["w2p"], False,
                      w3, bufs["w3p"], False)
        w4f = w4.reshape(-1).contiguous()
        # partial (the head's first-order+FM+dense logits) folds into the
        # final-dot epilogue, replacing the `partial + dnn` add kernel
        out, a1, a2, a3 = ext.mlp3_fwd(x0, bufs["w1p"], b1, bufs["w2p"], b2,
                                       bufs["w3p"], b3, w4f, b4,
                                       None if partial is None
                                       else partial.contiguous())
        ctx.save_for_backward(x0, w1, w2, w3, w4f, a1, a2, a3)
        ctx.bufs = bufs
        ctx._params = (w1, b1, w2, b2, w3, b3, w4, b4)
        ctx._has_partial = partial is not None
        return out

    @staticmethod
    def backward(ctx, dout):
        from ..ops import require_hip
        ext = require_hip()
        x0, w1, w2, w3, w4f, a1, a2, a3 = ctx.saved_tensors
        bufs = ctx.bufs
        K0 = w1.shape[1]
        # padded transposed weights for the fused dgrad chain (one launch;
        # w1tp takes w1p as source so its K0p..K0 tail rows stay zero)
        ext.mlp3_pack(w3, bufs["w3tp"], True, w2, bufs["w2tp"], True,
                      bufs["w1p"], bufs["w1tp"], True)
        # prebound flat-optimizer grads? decided BEFORE the dgrad launch:
        # the dz3-assembly loop there carries the head wgrad/bias sums
        # (it reads dout and a3 anyway), killing the separate head pass
        params = ctx._params
        prebound = all(p.grad is not None for p in params)
        bias_fused = (prebound
                      and params[1].grad.dtype == torch.bfloat16
                      and all(params[i].grad.is_contiguous()
                              for i in (1, 3, 5, 6)))
        bscratch = None
        if bias_fused:
            H0 = a1.shape[1]
            bscratch = bufs.get("bscratch")
            if bscratch is None or bscratch.numel() != 4 * H0 + 1:
                bscratch = torch.zeros(4 * H0 + 1, dtype=torch.float32,
                                       device=a1.device)
                bufs["bscratch"] = bscratch
        dx0, dz1, dz2, dz3 = ext.mlp3_bwd(
            dout.contiguous(), a1, a2, a3, w4f,
            bufs["w3tp"], bufs["w2tp"], bufs["w1tp"], bscratch)
        # bias grads as GEMVs (fallback paths): torch's column-sum of
        # row-major bf16 ran ~16 us each (reduce_kernel); ones@dz is a
        # hipBLASLt GEMV. The default path below does them in the fused
        # bias kernel instead and never materializes d/ones.
        def _d():
            return dout.unsqueeze(0).to(a3.dtype)      # [1, M]

        def _ones():
            ones = bufs.get("ones")
            if ones is None or ones.shape[1] != a1.shape[0]:
                ones = torch.ones(1, a1.shape[0], dtype=a1.dtype,
                                  device=a1.device)
                bufs["ones"] = ones
            return ones
        # With pre-bound .grad views (the flat-optimizer bench path), the
        # wgrads ACCUMULATE in place via beta=1 addmm_ — one fused GEMM per
        # param instead of GEMM + autograd's separate accumulate-add (8 add
        # kernels/step in the profile) — and autograd gets None.
        if prebound:
            w1g, b1g, w2g, b2g, w3g, b3g, w4g, b4g = (p.grad for p in params)
            M, H = dz1.shape
            K0p = x0.shape[1]
            wgrad_fused = (w1g.dtype == torch.bfloat16
                           and w1g.is_contiguous() and w2g.is_contiguous()
                           and w3g.is_contiguous()
                           and M % 32 == 0 and K0p <= 512)
            # when BOTH fused paths run, the wgrad kernel carries the three
            # dz column sums (bias grads) from its LDS-staged tiles; the
            # head sums already rode the dgrad kernel above — the separate
            # bias pass disappears entirely (only its finisher runs)
            bias_in_wgrad = wgrad_fused and bias_fused
            if wgrad_fused:
                # K0p <= 512: at the dim9 shapes the fused kernel beats the
                # hipBLASLt trio (12.29 -> 12.62M); at dim64 (K0p 1696)
                # hipBLASLt's wide-N tiles win (4.90 vs 4.75M) — measured,
                # profiles r2m-r2o
                # all three wgrads in ONE MFMA launch (+= via fp32 scratch,
                # self-cleaning finisher) — the hipBLASLt trio ran ~59 us
                # per step on 64x16 macro tiles (profiles/, round 2)
                scratch = bufs.get("wscratch")
                want = H * K0p + 2 * H * H
                if scratch is None or scratch.numel() != want:
                    scratch = torch.zeros(want, dtype=torch.float32,
                                          device=dz1.device)
                    bufs["wscratch"] = scratch
                ext.mlp3_wgrad(dz1, dz2, dz3, x0, a1, a2, scratch,
                               w1g, w2g, w3g,
                               bscratch if bias_in_wgrad else None)
            else:
                w1g.addmm_(dz1.t(), x0[:, :K0])
                w2g.addmm_(dz2.t(), a1)
                w3g.addmm_(dz3.t(), a2)
            dpart = dout if ctx._has_partial else None
            if bias_fused:
                # finisher folds the fp32 scratch (dz sums from the wgrad
                # kernel or here; head sums from the dgrad kernel) into the
                # bf16 grads and re-zeros it. with_dz=False when the fused
                # wgrad already summed the dz columns; with_head always
                # False (the dgrad kernel carried dw4/db4); both False
                # launches only the finisher.
                ext.mlp3_bias_bwd(dout.contiguous(), dz1, dz2, dz3, a3,
                                  bscratch, b1g, b2g, b3g,
                                  w4g.reshape(-1), b4g,
                                  not bias_in_wgrad, False)
            else:
                d, ones = _d(), _ones()
                w4g.addmm_(d, a3)
                b1g.unsqueeze(0).addmm_(ones, dz1)
                b2g.unsqueeze(0).addmm_(ones, dz2)
                b3g.unsqueeze(0).addmm_(ones, dz3)
                b4g.add_(d.sum())
            return (dx0, dpart, None, None, None, None, None, None, None,
                    None, None)
        # wgrads stay library GEMMs — their K = batch shapes run well
        d, ones = _d(), _ones()
        dpart = dout if ctx._has_partial else None
        dw1 = dz1.t() @ x0[:, :K0]
        dw2 = dz2.t() @ a1
        dw3 = dz3.t() @ a2
        dw4 = d @ a3                                   # [1, H]
        db1 = (ones @ dz1).reshape(-1)
        db2 = (ones @ dz2).reshape(-1)
        db3 = (ones @ dz3).reshape(-1)
        db4 = d.sum(1)
        return (dx0, dpart, dw1, db1, dw2, db2, dw3, db3, dw4, db4, None)


def _mlp(in_dim: int, hidden: Sequence[int], out_dim: int = 1) -> nn.Sequential:
    layers: List[nn.Module] = []
    d = in_dim
    for h in hidden:
        layers += [nn.Linear(d, h), nn.ReLU()]
        d = h
    layers.append(nn.Linear(d, out_dim))
    return nn.Sequential(*layers)


class _CTRBase(nn.Module):
    """Shared CTR skeleton.

    MI355X-first layout: the per-key first-order ("linear"/wide) weight is
    stored as an EXTRA COLUMN of the same PS variable (rows are dim+1 wide)
    instead of the reference's separate dim-1 Embedding — one fused
    unique/all_to_all/gather per step instead of two, with identical
    per-element optimizer math (all sparse optimizers are element-wise; the
    per-row scalar states advance on the same touch pattern either way)."""

    def __init__(self, field_vocabs: Optional[List[int]], dim: int,
                 sparse_as_dense_size: int = 0, hash_mode: bool = False):
        super().__init__()
        self.field_vocabs = list(field_vocabs or CRITEO_FIELD_VOCABS)
        self.dim = dim
        self.n_fields = len(self.field_vocabs)
        self.embedding = CombinedEmbedding(self.field_vocabs, dim + 1,
                                           hash_mode=hash_mode)
        self.dense_linear = nn.Linear(N_DENSE, 1)
        # fused interaction head on GPU (ctrhead.hip); deep_in dtype follows
        # head_bf16 (fp32 measured faster than bf16 autocast on MI355X for
        # these skinny GEMMs — gpurun_out/bench_fp32graph.log)
        self.head_bf16 = False
        # whole-MLP fused kernel (mlp.hip): enabled by convert_mlp_bf16 when
        # the dnn matches Linear/ReLU x3 + Linear(H,1), H <= 512 (% 16)
        self.fused_mlp = False
        self._w1pad = None

    def _embed(self, sparse: torch.Tensor):
        """-> (e [B,F,dim], linear_w [B,F])."""
        e_all = self.embedding(sparse)
        return e_all[..., :self.dim], e_all[..., self.dim]

    def _first_order(self, dense: torch.Tensor, linear_w: torch.Tensor
                     ) -> torch.Tensor:
        return linear_w.sum(dim=1) + self.dense_linear(dense).squeeze(-1)

    def _use_fused_head(self, t: torch.Tensor) -> bool:
        # kernel limit: each wave lane covers <=2 row columns -> dim+1 <= 128
        return (t.is_cuda and 1 <= self.dim + 1 <= 128 and hip_available())

    def _fused_head(self, dense: torch.Tensor, sparse: torch.Tensor,
                    use_fm: bool):
        """-> (e_all [B,F,dim+1], deep_in [B,F*dim+ND(+pad)], partial [B])."""
        e_all = self.embedding(sparse)
        deep_in, partial = _FusedCTRHeadFn.apply(
            e_all, dense, self.dense_linear.weight, self.dense_linear.bias,
            use_fm, self.head_bf16, self.fused_mlp)
        return e_all, deep_in, partial

    def _dnn_out(self, deep_in: torch.Tensor,
                 partial: torch.Tensor = None) -> torch.Tensor:
        """dnn logits [B] (+ `partial` carried in, when given — the fused
        kernel folds the add into its final-dot epilogue)."""
        if (self.fused_mlp and deep_in.is_cuda
                and deep_in.dtype == torch.bfloat16):
            l1, l2, l3, l4 = self.dnn[0], self.dnn[2], self.dnn[4], self.dnn[6]
            K0p = deep_in.shape[1]
            if self._w1pad is None or self._w1pad["w1p"].shape[1] != K0p:
                H = l1.weight.shape[0]
                Hp = (H + 31) // 32 * 32
                z = lambda *s: torch.zeros(*s, dtype=deep_in.dtype,  # noqa: E731
                                           device=deep_in.device)
                self._w1pad = {"w1p": z(H, K0p), "w2p": z(H, Hp),
                               "w3p": z(H, Hp), "w3tp": z(H, Hp),
                               "w2tp": z(H, Hp), "w1tp": z(K0p, Hp)}
            return _FusedMLP3Fn.apply(
                deep_in, partial, l1.weight, l1.bias, l2.weight, l2.bias,
                l3.weight, l3.bias, l4.weight, l4.bias, self._w1pad)
        w_dtype = next(self.dnn.parameters()).dtype
        out = self.dnn(deep_in.to(w_dtype)).squeeze(-1).float()
        return out if partial is None else partial + out


class LR(_CTRBase):
    """Logistic regression (reference examples/criteo_lr_subclass.py:
    subclassed model, hash-mode embedding dim 1)."""

    def __init__(self, field_vocabs: Optional[List[int]] = None):
        super().__init__(field_vocabs, dim=0)

    def forward(self, dense: torch.Tensor, sparse: torch.Tensor) -> torch.Tensor:
        _, lin = self._embed(sparse)
        return self._first_order(dense, lin)


class WDL(_CTRBase):
    """Wide & Deep (reference benchmark model 'WDL')."""

    def __init__(self, field_vocabs: Optional[List[int]] = None, dim: int = 9,
                 hidden: Sequence[int] = (400, 400, 400), **kw):
        super().__init__(field_vocabs, dim, **kw)
        self.dnn = _mlp(self.n_fields * dim + N_DENSE, hidden)

    def forward(self, dense: torch.Tensor, sparse: torch.Tensor) -> torch.Tensor:
        if self._use_fused_head(dense):
            _, deep_in, partial = self._fused_head(dense, sparse, use_fm=False)
            return self._dnn_out(deep_in, partial)
        e, lin = self._embed(sparse)                     # [B, F, d], [B, F]
        deep_in = torch.cat([e.flatten(1), dense], dim=1)
        return self._first_order(dense, lin) + self._dnn_out(deep_in)


class DeepFM(_CTRBase):
    """DeepFM (reference primary benchmark model, BASELINE.md DeepFM dim9)."""

    def __init__(self, field_vocabs: Optional[List[int]] = None, dim: int = 9,
                 hidden: Sequence[int] = (400, 400, 400), **kw):
        super().__init__(field_vocabs, dim, **kw)
        self.dnn = _mlp(self.n_fields * dim + N_DENSE, hidden)

    def forward(self, dense: torch.Tensor, sparse: torch.Tensor) -> torch.Tensor:
        if self._use_fused_head(dense):
            _, deep_in, partial = self._fused_head(dense, sparse, use_fm=True)
            return self._dnn_out(deep_in, partial)
        e, lin = self._embed(sparse)                     # [B, F, d], [B, F]
        # FM second order: 0.5*((sum_f e)^2 - sum_f e^2) summed over dim
        s = e.sum(dim=1)
        fm2 = 0.5 * (s * s - (e * e).sum(dim=1)).sum(dim=1)
        deep_in = torch.cat([e.flatten(1), dense], dim=1)
        return (self._first_order(dense, lin) + fm2
                + self._dnn_out(deep_in))


class _CINLayerFn(torch.autograd.Function):
    """One CIN layer as an implicit-outer-product GEMM.

    out[b,o,dd] = sum_{f,h} W[o, f*H+h] * x0[b,f,dd] * xk[b,h,dd]

    The reference path (einsum -> conv1d) materializes the outer-product
    tensor z [B, F*H, d] — ~0.5 GB/step at the benchmark shape, and the
    dominant cost of round-1 xDeepFM. Here the operand V[(b,dd), f*H+h] =
    x0*xk is built per d-slice (27 MB transient) right before its GEMM and
    REBUILT in backward instead of saved, so peak memory is O(B*F*H) per
    slice and every op is hipGraph-capturable. ``cdt`` is the compute dtype
    (bf16 on the native-MFMA path, fp32 for matched-precision runs)."""

    @staticmethod
    def _vmat(x0p, xkp, B, F, H, d):
        # V [d, B, F*H] = per-slice outer products, built from the
        # d-leading contiguous layouts (one broadcast mul, coalesced)
        return (x0p.unsqueeze(3) * xkp.unsqueeze(2)).reshape(d, B, F * H)

    @staticmethod
    def _hip_ok(x0, W, cdt):
        if not (x0.is_cuda and cdt == torch.bfloat16 and hip_available()):
            return False
        O, K = W.shape
        F = x0.shape[1]
        H = K // F
        n = x0.shape[0] * x0.shape[2]    # columns; dw operands pad to 32
        return (F <= 32 and H <= 128 and 16 <= O <= 128 and O % 16 == 0
                and n % 32 == 0)

    @staticmethod
    def _bufs(W, F, H):
        # padded bf16 weight mirrors, cached on the parameter and
        # refreshed (copy_) every forward — same pattern as the fused MLP
        O, K = W.shape
        Kp = (K + 31) // 32 * 32
        Op = (O + 31) // 32 * 32
        bufs = getattr(W, "_cin_bufs", None)
        if bufs is None or bufs["wp"].shape != (O, Kp):
            bufs = {
                "wp": torch.zeros(O, Kp, dtype=torch.bfloat16,
                                  device=W.device),
                "wt": torch.zeros(Kp, Op, dtype=torch.bfloat16,
                                  device=W.device),
            }
            W._cin_bufs = bufs
        bufs["wp"][:, :K].copy_(W)
        bufs["wt"][:K, :O].copy_(W.t())
        return bufs

    @staticmethod
    def forward(ctx, x0, xk, W, cdt):
        B, F, d = x0.shape
        H = xk.shape[1]
        hip = _CINLayerFn._hip_ok(x0, W, cdt)
        if hip:
            # implicit-GEMM HIP kernels (ops/csrc/cin.hip): V is built in
            # LDS per block and never touches HBM (the torch path below
            # materializes ~245 MB per build; rocprof showed those builds
            # + re-reads as ~70% of the xDeepFM step)
            from ..ops import require_hip
            ext = require_hip()
            x0p = x0.permute(2, 0, 1).reshape(d * B, F).contiguous()
            xkp = xk.permute(2, 0, 1).reshape(d * B, H).contiguous()
            bufs = _CINLayerFn._bufs(W, F, H)
            out = ext.cin_fwd(x0p, xkp, bufs["wp"])       # [d*B, O]
            ctx.save_for_backward(x0p, xkp, W)
            ctx.meta = (B, F, H, d, True)
            return out.view(d, B, -1).permute(1, 2, 0).float()
        x0p = x0.permute(2, 0, 1).to(cdt).contiguous()   # [d, B, F]
        xkp = xk.permute(2, 0, 1).to(cdt).contiguous()   # [d, B, H]
        Wt = W.to(cdt).t().contiguous()                  # [F*H, O]
        v = _CINLayerFn._vmat(x0p, xkp, B, F, H, d)      # [d, B, F*H]
        out = torch.matmul(v, Wt)                        # [d, B, O]
        ctx.save_for_backward(x0p, xkp, W)
        ctx.meta = (B, F, H, d, False)
        ctx.cdt = cdt
        return out.permute(1, 2, 0).float()              # [B, O, d]

    @staticmethod
    def backward(ctx, dout):
        x0p, xkp, W = ctx.saved_tensors
        B, F, H, d, hip = ctx.meta
        if hip:
            from ..ops import require_hip
            ext = require_hip()
            N = d * B
            bufs = W._cin_bufs
            doutp = dout.permute(2, 0, 1).reshape(N, -1).contiguous()
            dx0p, dxkp = ext.cin_dx(doutp, bufs["wt"], x0p, xkp)
            # transposed bf16 operands for the weight-grad GEMM
            dzt = doutp.t().contiguous().to(torch.bfloat16)
            x0t = x0p.t().contiguous().to(torch.bfloat16)
            xkt = xkp.t().contiguous().to(torch.bfloat16)
            dW = ext.cin_dw(dzt, x0t, xkt, W.shape[0], 8)
            return (dx0p.view(d, B, F).permute(1, 2, 0),
                    dxkp.view(d, B, H).permute(1, 2, 0),
                    dW.to(W.dtype), None)
        cdt = ctx.cdt
        Wc = W.to(cdt)                                    # [O, F*H]
        g = dout.permute(2, 0, 1).to(cdt).contiguous()    # [d, B, O]
        v = _CINLayerFn._vmat(x0p, xkp, B, F, H, d)       # rebuilt, not saved
        # dW = sum_d g_d^T @ v_d  (d-batched GEMM, fp32 accumulate)
        dW = torch.matmul(g.transpose(1, 2), v).float().sum(0)
        p = torch.matmul(g, Wc).view(d, B, F, H)          # [d, B, F, H]
        dx0 = (p * xkp.unsqueeze(2)).sum(3)               # [d, B, F]
        dxk = (p * x0p.unsqueeze(3)).sum(2)               # [d, B, H]
        return (dx0.permute(1, 2, 0).float(),
                dxk.permute(1, 2, 0).float(), dW.to(W.dtype), None)


class CIN(nn.Module):
    """Compressed Interaction Network (xDeepFM component).

    MI355X-first: each layer is the implicit-GEMM _CINLayerFn above instead
    of the reference's einsum+Conv1d (DeepCTR CIN); weights are plain
    [O, F*H] matrices (identical math to a kernel-size-1 Conv1d)."""

    def __init__(self, n_fields: int, dim: int,
                 layer_sizes: Sequence[int] = (128, 128)):
        super().__init__()
        self.layer_sizes = list(layer_sizes)
        self.weights = nn.ParameterList()
        h_prev = n_fields
        for h in self.layer_sizes:
            w = nn.Parameter(torch.empty(h, n_fields * h_prev))
            nn.init.kaiming_uniform_(w, a=math.sqrt(5))
            self.weights.append(w)
            h_prev = h
        self.fc = nn.Linear(sum(self.layer_sizes), 1)
        # compute dtype of the layer GEMMs on GPU (set bf16 by
        # convert_mlp_bf16; fp32 keeps matched-precision runs honest)
        self.compute_dtype = torch.float32

    def forward(self, e: torch.Tensor) -> torch.Tensor:  # e: [B, F, d]
        x0 = e
        xk = e
        outs = []
        for w in self.weights:
            z = _CINLayerFn.apply(x0, xk, w, self.compute_dtype)
            xk = torch.relu(z)
            outs.append(xk.sum(dim=2))                   # [B, Hk]
        return self.fc(torch.cat(outs, dim=1)).squeeze(-1)


class xDeepFM(_CTRBase):
    """xDeepFM (reference benchmark model 'xDeepFM', compute-bound case)."""

    def __init__(self, field_vocabs: Optional[List[int]] = None, dim: int = 9,
                 hidden: Sequence[int] = (400, 400, 400),
                 cin_layers: Sequence[int] = (128, 128), **kw):
        super().__init__(field_vocabs, dim, **kw)
        self.dnn = _mlp(self.n_fields * dim + N_DENSE, hidden)
        self.cin = CIN(self.n_fields, dim, cin_layers)

    def forward(self, dense: torch.Tensor, sparse: torch.Tensor) -> torch.Tensor:
        if self._use_fused_head(dense):
            e_all, deep_in, partial = self._fused_head(dense, sparse,
                                                       use_fm=False)
            e = e_all[..., :self.dim]
            return self._dnn_out(deep_in, partial + self.cin(e))
        e, lin = self._embed(sparse)
        deep_in = torch.cat([e.flatten(1), dense], dim=1)
        return (self._first_order(dense, lin) + self.cin(e)
                + self._dnn_out(deep_in))


def convert_mlp_bf16(model: _CTRBase) -> _CTRBase:
    """Native-bf16 MLP: the dnn's weights LIVE in bf16 (straight into MFMA
    GEMMs, no per-step autocast casts; the flat optimizer keeps an fp32
    master) and the fused head emits bf16 deep_in. dense_linear stays fp32
    (it is consumed by the fused head kernel); embeddings, FM math and the
    loss stay fp32."""
    if hasattr(model, "dnn"):
        model.dnn.to(torch.bfloat16)
        seq = model.dnn
        ok = (len(seq) == 7
              and all(isinstance(seq[i], nn.Linear) for i in (0, 2, 4, 6))
              and seq[0].out_features == seq[2].out_features
              == seq[4].out_features == seq[6].in_features
              and seq[6].out_features == 1
              and seq[0].out_features % 16 == 0
              and seq[0].out_features <= 416)
        model.fused_mlp = ok
    model.head_bf16 = True
    if hasattr(model, "cin"):
        # CIN layer GEMMs in bf16 (fp32 accumulation inside the library
        # GEMM); weights and outputs stay fp32
        model.cin.compute_dtype = torch.bfloat16
    return model


MODELS = {"lr": LR, "wdl": WDL, "deepfm": DeepFM, "xdeepfm": xDeepFM}
