"""Autoregressive generation with incremental KV-cache decode.

Functional parity with the reference generate() (sample.py:68-95):
temperature-scaled categorical sampling from the last position's logits,
cropping to block_size. Improvement: an incremental KV cache while the
sequence fits in block_size (the reference re-runs the full forward per
token). Once the window slides past block_size we fall back to the
reference's recompute-from-cropped-window behavior so RoPE positions
match exactly.
"""
from __future__ import annotations

import math

import torch
import torch.nn.functional as F

from midgpt_amd.ops import reference as ref


def _attn_cached(q, k, v):
    """q: (B,H,Tq,C) attends to k,v: (B,H,Tk,C) with causal alignment at the
    END (the Tq new positions are the last Tq of Tk). fp32 softmax."""
    B, H, Tq, C = q.shape
    Tk = k.shape[2]
    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) / math.sqrt(C)
    if Tq > 1:
        qpos = torch.arange(Tk - Tq, Tk, device=q.device)[:, None]
        kpos = torch.arange(Tk, device=q.device)[None, :]
        s = s.masked_fill(kpos > qpos, float("-inf"))
    a = torch.softmax(s, dim=-1).to(v.dtype)
    return torch.matmul(a, v)


def _forward_cached(model, idx_new: torch.Tensor, cache: list, pos: int):
    """Run idx_new (B, Tnew) through the model appending to cache; returns
    last-position logits (B, V)."""
    cfg = model.config
    H, C = cfg.n_head, cfg.head_dim
    B, Tn = idx_new.shape
    sin = model.rope_sin[pos:pos + Tn]
    cos = model.rope_cos[pos:pos + Tn]
    x = F.embedding(idx_new, model.wte)
    for li, blk in enumerate(model.blocks):
        h = ref.rmsnorm(x, None, 1e-6)
        qkv = blk.attn.c_attn(h).view(B, Tn, 3, H, C)
        q = qkv[:, :, 0].permute(0, 2, 1, 3)
        k = qkv[:, :, 1].permute(0, 2, 1, 3)
        v = qkv[:, :, 2].permute(0, 2, 1, 3)
        q = ref.qk_layernorm(q, blk.attn.q_ln_weight)
        k = ref.qk_layernorm(k, blk.attn.k_ln_weight)
        q = ref.apply_rope(q, sin, cos)
        k = ref.apply_rope(k, sin, cos)
        if cache[li] is None:
            cache[li] = (k, v)
        else:
            pk, pv = cache[li]
            cache[li] = (torch.cat([pk, k], dim=2), torch.cat([pv, v], dim=2))
        k, v = cache[li]
        o = _attn_cached(q, k, v)
        o = o.transpose(1, 2).reshape(B, Tn, cfg.n_embd)
        x = x + blk.attn.c_proj(o)
        x = x + blk.mlp(ref.rmsnorm(x, None, 1e-6))
    x = ref.rmsnorm(x[:, -1:], None, 1e-5)
    return model.lm_head(x)[:, -1]


@torch.no_grad()
def generate(model, idx: torch.Tensor, max_new_tokens: int,
             temperature: float = 1.0, top_k: int | None = None,
             generator: torch.Generator | None = None) -> torch.Tensor:
    """idx (B, T0) int64 -> (B, T0 + max_new_tokens)."""
    model.eval()
    block = model.config.block_size
    cache = [None] * model.config.n_layer
    # prompts longer than block_size condition on the last block_size tokens
    # (reference sample.py:74-81 crops to the trailing window)
    logits = _forward_cached(model, idx[:, -block:], cache, 0)
    for _ in range(max_new_tokens):
        nxt = _sample(logits, temperature, top_k, generator)
        idx = torch.cat([idx, nxt[:, None]], dim=1)
        if idx.shape[1] <= block:
            logits = _forward_cached(model, nxt[:, None], cache,
                                     idx.shape[1] - 1)
        else:
            # window slid: reference behavior — recompute cropped window
            win = idx[:, -block:]
            cache = [None] * model.config.n_layer
            logits = _forward_cached(model, win, cache, 0)
            cache = [None] * model.config.n_layer  # cache positions invalid
    return idx


def _sample(logits, temperature, top_k, generator):
    if temperature <= 0:
        return logits.argmax(dim=-1)
    lf = logits.float() / temperature
    if top_k is not None:
        kth = torch.topk(lf, top_k, dim=-1).values[:, -1:]
        lf = lf.masked_fill(lf < kth, float("-inf"))
    p = torch.softmax(lf, dim=-1)
    return torch.multinomial(p.cpu() if generator is not None and
                             generator.device.type == "cpu" and p.is_cuda else p,
                             1, generator=generator).squeeze(-1).to(logits.device)
