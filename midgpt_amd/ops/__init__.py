"""Op dispatch: hand-written HIP/CDNA4 kernels on GPU, fp32 PyTorch
reference implementations on CPU.

The HIP extension is built IN-TREE as ``midgpt_amd/ops/_C*.so`` (see
``setup.py`` / ``__graft_entry__.build``). On a GPU box the
extension is REQUIRED: ops raise if it is missing so that a silent eager
fallback can never masquerade as the native path. Set ``MIDGPT_FORCE_REF=1``
to explicitly run the reference path on GPU (debugging only).
"""
from __future__ import annotations

import math
import os

import torch

from midgpt_amd.ops import reference as ref

_C = None
_C_ERR: str | None = None


def _try_load_ext():
    global _C, _C_ERR
    if _C is not None:
        return _C
    try:
        import importlib
        _C = importlib.import_module("midgpt_amd.ops._C")
    except Exception as e:  # pragma: no cover
        _C_ERR = repr(e)
        _C = None
    return _C


_try_load_ext()


def have_ext() -> bool:
    return _C is not None


def _use_hip(*tensors) -> bool:
    """True iff tensors live on a GPU. Raises loudly if the extension is
    missing on GPU (unless MIDGPT_FORCE_REF=1)."""
    on_gpu = any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if not on_gpu:
        return False
    if os.environ.get("MIDGPT_FORCE_REF") == "1":
        return False
    if _C is None:
        raise RuntimeError(
            "midgpt_amd HIP extension (midgpt_amd/ops/_C) is not built but a GPU "
            f"tensor reached a hot op. Build it with __graft_entry__.build(). "
            f"Import error: {_C_ERR}")
    return True


# ----------------------------------------------------------------------------
# RMSNorm (reference src/layers.py:60-75; plan K5)
# ----------------------------------------------------------------------------
class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        shp = x.shape
        x2d = x.reshape(-1, shp[-1])
        if _use_hip(x):
            y, invrms = _C.rmsnorm_fwd(x2d.contiguous(), weight, eps)
        else:
            xf = x2d.float()
            invrms = torch.rsqrt(xf.pow(2).mean(dim=-1) + eps)
            y = xf * invrms[:, None]
            if weight is not None:
                y = y * weight.float()
            y = y.to(x.dtype)
        ctx.save_for_backward(x2d, invrms, *( [weight] if weight is not None else [] ))
        ctx.eps = eps
        ctx.has_w = weight is not None
        return y.reshape(shp)

    @staticmethod
    def backward(ctx, dy):
        saved = ctx.saved_tensors
        x2d, invrms = saved[0], saved[1]
        weight = saved[2] if ctx.has_w else None
        dy2d = dy.reshape(x2d.shape)
        if _use_hip(x2d):
            dx, dw = _C.rmsnorm_bwd(dy2d.contiguous(), x2d.contiguous(), weight,
                                    invrms, ctx.eps)
        else:
            D = x2d.shape[-1]
            xf = x2d.float()
            dyf = dy2d.float()
            g = dyf * weight.float() if weight is not None else dyf
            # y = x * r, r = (mean(x^2)+eps)^-1/2
            # dx = r*g - x * r^3 * mean(x*g)
            r = invrms[:, None]
            mean_xg = (xf * g).mean(dim=-1, keepdim=True)
            dx = (r * g - xf * r.pow(3) * mean_xg).to(x2d.dtype)
            dw = (dyf * xf * r).sum(dim=0).to(weight.dtype) if weight is not None else None
        return dx.reshape(dy.shape), dw, None


def rmsnorm(x, weight=None, eps: float = 1e-6):
    return _RMSNorm.apply(x, weight, eps)


# ----------------------------------------------------------------------------
# Fused QK-LayerNorm + RoPE over the packed QKV projection
# (reference src/model.py:59-69; plan K3+K4). Input qkv: (B, T, 3, H, C)
# produced by the c_attn GEMM; outputs q,k,v: (B, H, T, C), with LayerNorm
# (weight, no bias, eps 1e-6) and GPT-J interleaved RoPE applied to q,k.
# ----------------------------------------------------------------------------
class _QKVPrep(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, qw, kw, sin, cos, eps):
        B, T, three, H, C = qkv.shape
        assert three == 3
        if _use_hip(qkv):
            q, k, v, qstats, kstats = _C.qkv_prep_fwd(qkv.contiguous(), qw, kw,
                                                      sin, cos, eps)
        else:
            qr = qkv[:, :, 0].permute(0, 2, 1, 3)  # (B,H,T,C)
            kr = qkv[:, :, 1].permute(0, 2, 1, 3)
            v = qkv[:, :, 2].permute(0, 2, 1, 3).contiguous()
            qn, qstats = _ln_fwd_stats(qr, qw, eps)
            kn, kstats = _ln_fwd_stats(kr, kw, eps)
            q = ref.apply_rope(qn, sin, cos).contiguous()
            k = ref.apply_rope(kn, sin, cos).contiguous()
        ctx.save_for_backward(qkv, qw, kw, sin, cos, qstats, kstats)
        ctx.eps = eps
        return q, k, v

    @staticmethod
    def backward(ctx, dq, dk, dv):
        qkv, qw, kw, sin, cos, qstats, kstats = ctx.saved_tensors
        if _use_hip(qkv):
            dqkv, dqw, dkw = _C.qkv_prep_bwd(dq.contiguous(), dk.contiguous(),
                                             dv.contiguous(), qkv.contiguous(),
                                             qw, kw, sin, cos, qstats, kstats,
                                             ctx.eps)
        else:
            # inverse RoPE rotation on dq/dk: d(rope(x)) -> cos*dy - rot(sin*dy)...
            # rope(x) = x*cos + rot(x)*sin with rot = rotate_every_two.
            # d/dx: dx = dy*cos + rot^T(dy*sin); rot^T = inverse rotation
            # rot([a,b]) = [-b, a]  =>  rot^T([a,b]) = [b, -a] = -rot
            sin2 = torch.repeat_interleave(sin, 2, dim=-1)
            cos2 = torch.repeat_interleave(cos, 2, dim=-1)
            dqn = (dq.float() * cos2 - ref.rotate_every_two(dq.float() * sin2))
            dkn = (dk.float() * cos2 - ref.rotate_every_two(dk.float() * sin2))
            qr = qkv[:, :, 0].permute(0, 2, 1, 3)
            kr = qkv[:, :, 1].permute(0, 2, 1, 3)
            dqr, dqw = _ln_bwd(dqn, qr, qw, qstats)
            dkr, dkw = _ln_bwd(dkn, kr, kw, kstats)
            dqkv = torch.empty_like(qkv)
            dqkv[:, :, 0] = dqr.permute(0, 2, 1, 3).to(qkv.dtype)
            dqkv[:, :, 1] = dkr.permute(0, 2, 1, 3).to(qkv.dtype)
            dqkv[:, :, 2] = dv.permute(0, 2, 1, 3).to(qkv.dtype)
        return dqkv, dqw, dkw, None, None, None


def _ln_fwd_stats(x, w, eps):
    """LayerNorm fwd returning (y, stats) with stats = (mean, invstd) fp32
    stacked on last dim -> shape (*x.shape[:-1], 2)."""
    xf = x.float()
    mu = xf.mean(dim=-1, keepdim=True)
    var = xf.var(dim=-1, unbiased=False, keepdim=True)
    invstd = torch.rsqrt(var + eps)
    y = ((xf - mu) * invstd * w.float()).to(x.dtype)
    stats = torch.cat([mu, invstd], dim=-1)  # (..., 2)
    return y, stats


def _ln_bwd(dyf, x, w, stats):
    """LayerNorm bwd (no bias). dyf fp32, x input tensor, stats (...,2)."""
    xf = x.float()
    mu = stats[..., 0:1]
    invstd = stats[..., 1:2]
    xhat = (xf - mu) * invstd
    g = dyf * w.float()
    D = xf.shape[-1]
    mean_g = g.mean(dim=-1, keepdim=True)
    mean_gx = (g * xhat).mean(dim=-1, keepdim=True)
    dx = invstd * (g - mean_g - xhat * mean_gx)
    dw = (dyf * xhat).sum(dim=tuple(range(dyf.dim() - 1))).to(w.dtype)
    return dx, dw


def qkv_prep(qkv, q_ln_weight, k_ln_weight, sin, cos, eps: float = 1e-6):
    return _QKVPrep.apply(qkv, q_ln_weight, k_ln_weight, sin, cos, eps)


# ----------------------------------------------------------------------------
# Flash-style causal attention (reference src/model.py:71-79; plan K1/K2)
# ----------------------------------------------------------------------------
class _Attention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v):
        if _use_hip(q):
            q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
            o, lse = _C.attn_fwd(q, k, v)
        else:
            B, H, T, C = q.shape
            s = torch.matmul(q.float(), k.float().transpose(-1, -2)) / math.sqrt(C)
            mask = torch.ones(T, T, dtype=torch.bool, device=q.device).tril()
            s = s.masked_fill(~mask, float("-inf"))
            lse = torch.logsumexp(s, dim=-1)  # (B,H,T) fp32
            a = torch.exp(s - lse[..., None])
            o = torch.matmul(a.to(v.dtype), v)
        ctx.save_for_backward(q, k, v, o, lse)
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        if _use_hip(q):
            dq, dk, dv = _C.attn_bwd(do.contiguous(), q, k, v, o, lse)
        else:
            B, H, T, C = q.shape
            scale = 1.0 / math.sqrt(C)
            s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
            mask = torch.ones(T, T, dtype=torch.bool, device=q.device).tril()
            s = s.masked_fill(~mask, float("-inf"))
            p = torch.exp(s - lse[..., None])          # (B,H,T,T) fp32
            dof = do.float()
            dv = torch.matmul(p.transpose(-1, -2), dof)
            dp = torch.matmul(dof, v.float().transpose(-1, -2))
            delta = (dof * o.float()).sum(dim=-1, keepdim=True)  # rowsum(dO*O)
            ds = p * (dp - delta)
            dq = torch.matmul(ds, k.float()) * scale
            dk = torch.matmul(ds.transpose(-1, -2), q.float()) * scale
            dq, dk, dv = dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)
        return dq, dk, dv


def flash_attention(q, k, v):
    """Causal attention, q/k/v (B,H,T,C). Softmax scale 1/sqrt(C) in fp32."""
    return _Attention.apply(q, k, v)


# ----------------------------------------------------------------------------
# GELU (tanh approx) fwd/bwd (reference src/model.py:30; plan K7): a
# hand-written u16x8 elementwise pair replacing the torch library kernels
# (same numerics as F.gelu(approximate="tanh"); fp32 internal).
# ----------------------------------------------------------------------------
class _Gelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        if _use_hip(x) and x.dtype == torch.bfloat16 and x.numel() % 8 == 0:
            return _C.gelu_fwd(x.contiguous())
        return torch.nn.functional.gelu(x, approximate="tanh")

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        if _use_hip(x) and x.dtype == torch.bfloat16 and x.numel() % 8 == 0:
            return _C.gelu_bwd(dy.contiguous(), x.contiguous())
        xf = x.float()
        c0, c1 = 0.7978845608028654, 0.044715
        inner = c0 * (xf + c1 * xf ** 3)
        t = torch.tanh(inner)
        dg = 0.5 * (1 + t) + 0.5 * xf * (1 - t * t) * c0 * (1 + 3 * c1 * xf ** 2)
        return (dy.float() * dg).to(x.dtype)


def gelu(x):
    return _Gelu.apply(x)


# ----------------------------------------------------------------------------
# Embedding: gather fwd, scatter-add bwd (reference src/layers.py:13-34;
# plan K8). The forward gather is a plain coalesced index_select (torch);
# the backward is the hand-written HIP fp32-accurate scatter-add.
# ----------------------------------------------------------------------------
class _Embedding(torch.autograd.Function):
    @staticmethod
    def forward(ctx, idx, weight):
        ctx.save_for_backward(idx)
        ctx.V = weight.shape[0]
        ctx.wdtype = weight.dtype
        return weight[idx]

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        D = dy.shape[-1]
        dy2d = dy.reshape(-1, D)
        if _use_hip(dy) and dy.dtype == torch.bfloat16:
            dw = _C.embedding_bwd(dy2d.contiguous(), idx.reshape(-1), ctx.V)
        else:
            dw32 = torch.zeros(ctx.V, D, dtype=torch.float32, device=dy.device)
            dw32.index_add_(0, idx.reshape(-1), dy2d.float())
            dw = dw32.to(ctx.wdtype)
        return None, dw


def embedding(idx, weight):
    """idx (...,) int64, weight (V, D) -> (..., D)."""
    return _Embedding.apply(idx, weight)


# ----------------------------------------------------------------------------
# Fused softmax cross-entropy over the vocab (reference src/train.py:76-77;
# plan K9). Never materializes the fp32 softmax over V on the HIP path.
# ----------------------------------------------------------------------------
class _CrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets):
        N, V = logits.shape
        if _use_hip(logits):
            loss_sum, lse = _C.ce_fwd(logits, targets)
            loss = loss_sum / N
        else:
            lf = logits.float()
            lse = torch.logsumexp(lf, dim=-1)
            picked = lf.gather(1, targets[:, None]).squeeze(1)
            loss = (lse - picked).mean()
        ctx.save_for_backward(logits, targets, lse)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, lse = ctx.saved_tensors
        N, V = logits.shape
        if _use_hip(logits):
            dlogits = _C.ce_bwd(logits, targets, lse, dloss.float().reshape(1))
        else:
            lf = logits.float()
            p = torch.exp(lf - lse[:, None])
            p.scatter_add_(1, targets[:, None],
                           -torch.ones(N, 1, device=p.device, dtype=p.dtype))
            dlogits = (p * (dloss.float() / N)).to(logits.dtype)
        return dlogits, None


def cross_entropy(logits, targets):
    """logits (N,V), targets (N,) int64 -> scalar mean loss (fp32)."""
    return _CrossEntropy.apply(logits, targets)


# ----------------------------------------------------------------------------
# Fused AdamW step on flat buffers (plan K10/K13). Semantics match the optax
# chain clip_by_global_norm(1.0) -> scale_by_adam -> add_decayed_weights(
# wd/lr_peak) -> scale_by_schedule -> scale(-1)  (reference src/train.py:153-159).
# ----------------------------------------------------------------------------
# ----------------------------------------------------------------------------
# hipBLASLt DGELU-fused MLP backward: the gelu backward lives in the dgrad
# GEMM epilogue — no separate elementwise gelu-backward kernel (plan K7).
# (GELU_AUX forward fusion has no gfx950 hipblaslt algorithms on ROCm 7.2 —
# probed via _C.lt_probe — so the forward keeps torch's GEMM + gelu;
# numerics: hipBLASLt GELU is the tanh approximation, matching the
# reference jax.nn.gelu default.)
# ----------------------------------------------------------------------------
class _FusedMLP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d, w1, w2):
        h = torch.mm(x2d, w1.t())
        g = torch.nn.functional.gelu(h, approximate="tanh")
        y = torch.mm(g, w2.t())
        ctx.save_for_backward(x2d, w1, w2, h, g)
        return y

    @staticmethod
    def backward(ctx, dy):
        x2d, w1, w2, h, g = ctx.saved_tensors
        dy = dy.contiguous()
        dh = _C.matmul_dgelu(dy, w2, h)      # dgrad GEMM + dgelu epilogue
        dw2 = torch.mm(dy.t(), g)
        dx = torch.mm(dh, w1)
        dw1 = torch.mm(dh.t(), x2d)
        return dx, dw1, dw2


def fused_mlp(x, w1, w2):
    """x (..., D) -> gelu(x @ w1.T) @ w2.T with GEMM-epilogue GELU."""
    shp = x.shape
    y = _FusedMLP.apply(x.reshape(-1, shp[-1]).contiguous(), w1, w2)
    return y.reshape(*shp[:-1], w2.shape[0])


def adamw_step(master: torch.Tensor, grad: torch.Tensor, m: torch.Tensor,
               v: torch.Tensor, out_bf16: torch.Tensor | None,
               *, lr: float, beta1: float, beta2: float, eps: float,
               wd_over_peak_lr: float, grad_scale: float, clip_norm: float,
               sq_sum: torch.Tensor, step: int):
    """In-place AdamW on flat fp32 buffers; optionally writes the bf16
    working copy. ``step`` is 1-indexed (bias correction t).

    The effective gradient is ``grad * grad_scale * clip`` where
    ``clip = min(1, clip_norm / (grad_scale * sqrt(sq_sum)))`` — computed
    ON DEVICE from the (already all-reduced) squared-norm scalar ``sq_sum``
    so the step path never syncs to host.
    """
    if master.is_cuda and _use_hip(master):
        _C.adamw_step(master, grad, m, v,
                      out_bf16 if out_bf16 is not None
                      else master.new_empty(0, dtype=torch.bfloat16),
                      out_bf16 is not None, sq_sum,
                      lr, beta1, beta2, eps, wd_over_peak_lr,
                      grad_scale, clip_norm, step)
        return
    gnorm = float(sq_sum.float().sqrt()) * grad_scale
    clip_coef = grad_scale * min(1.0, clip_norm / (gnorm + 1e-12))
    g = grad * clip_coef
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    mhat = m / (1 - beta1 ** step)
    vhat = v / (1 - beta2 ** step)
    update = mhat / (vhat.sqrt() + eps) + wd_over_peak_lr * master
    master.add_(update, alpha=-lr)
    if out_bf16 is not None:
        out_bf16.copy_(master.to(torch.bfloat16))
