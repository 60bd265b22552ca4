"""Pure-PyTorch reference implementations of the hot ops.

These define the numerics contract (fp32 internal math) that the HIP
kernels in ``csrc/`` are tested against, and serve as the CPU execution
path. Semantics follow the reference implementation:

- RMSNorm: ``x * rsqrt(mean(x^2) + eps)``, optional weight
  (reference src/layers.py:60-75; block norms weightless).
- QK-LayerNorm: LayerNorm over head dim, weight, no bias, eps 1e-6
  (reference src/model.py:52-53).
- RoPE: GPT-J interleaved pairing — ``rotate_every_two`` =
  ``[a b c d] -> [-b a -d c]`` (reference src/layers.py:85-99).
- Attention: causal, fp32 softmax with 1/sqrt(C) scale folded in
  (reference src/model.py:71-77).
- Cross-entropy: fp32 logits, mean over tokens (reference src/train.py:76-77).
"""
from __future__ import annotations

import math

import torch
import torch.nn.functional as F


def rmsnorm(x: torch.Tensor, weight=None, eps: float = 1e-6) -> torch.Tensor:
    """Row-wise RMS normalization over the last dim. fp32 internal math."""
    xf = x.float()
    out = xf * torch.rsqrt(xf.pow(2).mean(dim=-1, keepdim=True) + eps)
    if weight is not None:
        out = out * weight.float()
    return out.to(x.dtype)


def qk_layernorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    """LayerNorm over the last (head) dim with weight, no bias."""
    xf = x.float()
    mu = xf.mean(dim=-1, keepdim=True)
    var = xf.var(dim=-1, unbiased=False, keepdim=True)
    out = (xf - mu) * torch.rsqrt(var + eps) * weight.float()
    return out.to(x.dtype)


def rope_tables(head_dim: int, seq_len: int, base: float = 10000.0,
                device=None, dtype=torch.float32):
    """sin/cos tables of shape (T, C/2). inv_freq = base^(-2i/C)
    (reference src/layers.py:79-82, host-precomputed)."""
    inv_freq = 1.0 / (base ** (torch.arange(0, head_dim, 2, device=device,
                                            dtype=torch.float64) / head_dim))
    t = torch.arange(seq_len, device=device, dtype=torch.float64)
    freqs = torch.outer(t, inv_freq)  # (T, C/2)
    return freqs.sin().to(dtype), freqs.cos().to(dtype)


def rotate_every_two(x: torch.Tensor) -> torch.Tensor:
    """[a b c d] -> [-b a -d c] (GPT-J interleaved; reference src/layers.py:85-89)."""
    x1 = x[..., ::2]
    x2 = x[..., 1::2]
    out = torch.stack((-x2, x1), dim=-1)
    return out.flatten(-2)


def apply_rope(x: torch.Tensor, sin: torch.Tensor, cos: torch.Tensor) -> torch.Tensor:
    """x: (..., T, C); sin/cos: (T, C/2). Duplicates sin/cos to C interleaved
    and returns x*cos + rotate_every_two(x)*sin (reference src/layers.py:92-99)."""
    sin2 = torch.repeat_interleave(sin, 2, dim=-1).to(torch.float32)
    cos2 = torch.repeat_interleave(cos, 2, dim=-1).to(torch.float32)
    xf = x.float()
    out = xf * cos2 + rotate_every_two(xf) * sin2
    return out.to(x.dtype)


def causal_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     dropout_p: float = 0.0, training: bool = False) -> torch.Tensor:
    """q,k,v: (B, H, T, C). Materialized reference (O(T^2) memory):
    scores = QK^T masked, softmax((scores)/sqrt(C)) in fp32, then @V."""
    B, H, T, C = q.shape
    s = torch.matmul(q.float(), k.float().transpose(-1, -2))
    mask = torch.ones(T, T, dtype=torch.bool, device=q.device).tril()
    s = s.masked_fill(~mask, float("-inf"))
    a = torch.softmax(s / math.sqrt(C), dim=-1)
    if dropout_p > 0.0 and training:
        a = F.dropout(a, p=dropout_p, training=True)
    return torch.matmul(a.to(v.dtype), v)


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    """logits (N, V) any float dtype, targets (N,) int64. fp32 math, mean."""
    return F.cross_entropy(logits.float(), targets.reshape(-1))
