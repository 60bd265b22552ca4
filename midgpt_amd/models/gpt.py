"""GPT model — batched PyTorch modules over the midgpt_amd hot-op layer.

Numerics contract (matching the reference):
- Linear: bias-free, weight (out,in), truncated-normal +-2sigma scaled by
  1/sqrt(in)                                  (reference src/layers.py:37-57)
- Embedding wte: normal * 1/sqrt(D); lm_head initialized from the SAME
  array (tied at init, untied afterwards)     (reference src/model.py:134-138)
- RMSNorm: weightless for block norms (eps 1e-6) and final norm (eps 1e-5)
                                              (reference src/model.py:94-95,133)
- QK-LayerNorm over head dim, weight, no bias, eps 1e-6
                                              (reference src/model.py:52-53)
- RoPE: GPT-J interleaved pairing, host-precomputed sin/cos tables
                                              (reference src/layers.py:79-99)
- Attention: causal, fp32 softmax with 1/sqrt(C) scale
                                              (reference src/model.py:71-77)
- MLP: gelu(tanh approx) 4x expansion, no bias (reference src/model.py:17-31)

Unlike the reference (per-token modules lifted by vmap), everything here is
natively batched (B, T, ...) so each op is ONE kernel launch on the GPU.
"""
from __future__ import annotations

import contextlib
import math
import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from midgpt_amd import ops
from midgpt_amd.config import GPTConfig
from midgpt_amd.ops.reference import rope_tables


_SCOPES = os.environ.get("MIDGPT_SCOPES") == "1"


def _scope(name):
    """Named trace scopes (reference jax.named_scope parity); enabled with
    MIDGPT_SCOPES=1 so the hot path stays annotation-free by default."""
    return torch.profiler.record_function(name) if _SCOPES \
        else contextlib.nullcontext()


def _init_linear_(w: torch.Tensor, in_features: int, generator=None):
    """Truncated normal on [-2, 2] scaled by 1/sqrt(in_features)."""
    nn.init.trunc_normal_(w, mean=0.0, std=1.0, a=-2.0, b=2.0, generator=generator)
    with torch.no_grad():
        w.mul_(1.0 / math.sqrt(in_features))


class Linear(nn.Module):
    """Bias-free linear, weight (out, in)."""

    def __init__(self, in_features: int, out_features: int, generator=None):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        _init_linear_(self.weight, in_features, generator)

    def forward(self, x):
        return F.linear(x, self.weight.to(x.dtype))


class CausalSelfAttention(nn.Module):
    def __init__(self, config: GPTConfig, generator=None):
        super().__init__()
        D, H = config.n_embd, config.n_head
        assert D % H == 0
        self.n_head, self.n_embd = H, D
        self.head_dim = D // H
        self.c_attn = Linear(D, 3 * D, generator)
        self.c_proj = Linear(D, D, generator)
        # QK-LayerNorm weights (eps 1e-6, no bias)
        self.q_ln_weight = nn.Parameter(torch.ones(self.head_dim))
        self.k_ln_weight = nn.Parameter(torch.ones(self.head_dim))
        self.dropout = config.dropout
        self.attn_dropout = nn.Dropout(config.dropout)
        self.resid_dropout = nn.Dropout(config.dropout)

    def forward(self, x, sin, cos):
        with _scope("causal_sa"):
            return self._forward(x, sin, cos)

    def _forward(self, x, sin, cos):
        B, T, D = x.shape
        qkv = self.c_attn(x).view(B, T, 3, self.n_head, self.head_dim)
        q, k, v = ops.qkv_prep(qkv, self.q_ln_weight, self.k_ln_weight,
                               sin[:T], cos[:T])
        if self.dropout > 0.0 and self.training:
            # Attention-matrix dropout requires materialized probabilities;
            # only the shakespeare_char config uses it (reference
            # src/model.py:78). Not a hot path.
            from midgpt_amd.ops import reference as refops
            o = refops.causal_attention(q, k, v, self.dropout, True)
        else:
            o = ops.flash_attention(q, k, v)
        o = o.transpose(1, 2).reshape(B, T, D)
        return self.resid_dropout(self.c_proj(o))


class MLP(nn.Module):
    def __init__(self, config: GPTConfig, generator=None):
        super().__init__()
        D = config.n_embd
        self.c_fc = Linear(D, 4 * D, generator)
        self.c_proj = Linear(4 * D, D, generator)
        self.dropout = nn.Dropout(config.dropout)

    def forward(self, x):
        with _scope("mlp"):
            # ops.fused_mlp (hipBLASLt DGELU-epilogue backward) is available
            # but measured SLOWER end-to-end on gfx950/ROCm 7.2 (the few
            # epilogue-capable algos lose more GEMM throughput than the
            # saved elementwise pass) -> opt-in via MIDGPT_FUSED_MLP=1.
            if (x.is_cuda and x.dtype == torch.bfloat16 and ops.have_ext()
                    and os.environ.get("MIDGPT_FUSED_MLP") == "1"):
                return self.dropout(ops.fused_mlp(x, self.c_fc.weight,
                                                  self.c_proj.weight))
            # torch's gelu kernels win IN-STEP (the hand-written pair ties
            # them in isolation, but its nontemporal stores defeat the
            # L2 handoff to the next GEMM — rocprof: 136 vs 109 ms/4
            # steps at xl). MIDGPT_GELU_HIP=1 opts into the native pair.
            if os.environ.get("MIDGPT_GELU_HIP") == "1":
                return self.dropout(self.c_proj(ops.gelu(self.c_fc(x))))
            return self.dropout(
                self.c_proj(F.gelu(self.c_fc(x), approximate="tanh")))


class Block(nn.Module):
    """Pre-norm residual block; ln1/ln2 weightless RMSNorm (eps 1e-6)."""

    def __init__(self, config: GPTConfig, generator=None):
        super().__init__()
        self.attn = CausalSelfAttention(config, generator)
        self.mlp = MLP(config, generator)

    def forward(self, x, sin, cos):
        with _scope("block"):
            x = x + self.attn(ops.rmsnorm(x, None, 1e-6), sin, cos)
            x = x + self.mlp(ops.rmsnorm(x, None, 1e-6))
            return x


class GPT(nn.Module):
    def __init__(self, config: GPTConfig, generator=None):
        super().__init__()
        self.config = config
        D, V = config.n_embd, config.vocab_size
        self.wte = nn.Parameter(torch.randn(V, D, generator=generator) / math.sqrt(D))
        self.drop = nn.Dropout(config.dropout)
        self.blocks = nn.ModuleList(Block(config, generator)
                                    for _ in range(config.n_layer))
        # lm_head tied AT INIT to wte (separate parameter afterwards —
        # reference src/model.py:137-138)
        self.lm_head = Linear(D, V)
        with torch.no_grad():
            self.lm_head.weight.copy_(self.wte)
        sin, cos = rope_tables(config.head_dim, config.block_size)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.remat = False  # set by the trainer (jax.checkpoint parity)

    def _block_fn(self, blk, x):
        return blk(x, self.rope_sin, self.rope_cos)

    def forward(self, idx: torch.Tensor) -> torch.Tensor:
        """idx (B, T) int64 -> logits (B, T, V) in compute dtype."""
        x = self.drop(ops.embedding(idx, self.wte))
        for blk in self.blocks:
            if self.remat and torch.is_grad_enabled():
                x = torch.utils.checkpoint.checkpoint(
                    self._block_fn, blk, x, use_reentrant=False)
            else:
                x = self._block_fn(blk, x)
        x = ops.rmsnorm(x, None, 1e-5)
        return self.lm_head(x)

    def loss(self, idx: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
        logits = self.forward(idx)
        return ops.cross_entropy(logits.reshape(-1, logits.shape[-1]),
                                 targets.reshape(-1))


def count_params(model: GPT) -> int:
    """Parameter count excluding lm_head (parity with reference
    src/model.py:161-164 which excludes the (un)tied head)."""
    n = sum(p.numel() for p in model.parameters())
    return n - model.lm_head.weight.numel()
