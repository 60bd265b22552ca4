"""Property-based tests for the op layer's mathematical invariants (CPU,
using the same dispatch wrappers the model uses)."""

import torch
from hypothesis import given, settings, strategies as st

from midgpt_amd import ops
from midgpt_amd.ops import reference as ref


@given(st.integers(0, 2 ** 31 - 1), st.floats(0.5, 8.0))
@settings(max_examples=20, deadline=None)
def test_rmsnorm_scale_invariance(seed, scl):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(4, 64, generator=g) * 10  # large vs eps
    y1 = ops.rmsnorm(x, None, 1e-6)
    y2 = ops.rmsnorm(x * scl, None, 1e-6)
    assert torch.allclose(y1, y2, atol=1e-4)


@given(st.integers(0, 2 ** 31 - 1))
@settings(max_examples=20, deadline=None)
def test_rope_is_a_rotation(seed):
    """RoPE preserves the norm of every interleaved pair exactly."""
    g = torch.Generator().manual_seed(seed)
    T, C = 16, 32
    x = torch.randn(2, T, C, generator=g)
    sin, cos = ref.rope_tables(C, T)
    y = ref.apply_rope(x, sin, cos)
    nx = x.view(2, T, C // 2, 2).norm(dim=-1)
    ny = y.view(2, T, C // 2, 2).norm(dim=-1)
    assert torch.allclose(nx, ny, atol=1e-5)


@given(st.integers(0, 2 ** 31 - 1), st.floats(-5.0, 5.0))
@settings(max_examples=20, deadline=None)
def test_cross_entropy_shift_invariance(seed, shift):
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(16, 33, generator=g)
    targets = torch.randint(0, 33, (16,), generator=g)
    l1 = ops.cross_entropy(logits, targets)
    l2 = ops.cross_entropy(logits + shift, targets)
    assert torch.allclose(l1, l2, atol=1e-5)


@given(st.integers(0, 2 ** 31 - 1))
@settings(max_examples=10, deadline=None)
def test_attention_uniform_value_rows(seed):
    """If every V row equals v0, attention output is v0 for every query
    (softmax weights sum to 1)."""
    g = torch.Generator().manual_seed(seed)
    B, H, T, C = 1, 2, 8, 16
    q = torch.randn(B, H, T, C, generator=g)
    k = torch.randn(B, H, T, C, generator=g)
    v0 = torch.randn(B, H, 1, C, generator=g)
    v = v0.expand(B, H, T, C).contiguous()
    o = ops.flash_attention(q, k, v)
    assert torch.allclose(o, v.clone(), atol=1e-5)


def test_adamw_zero_grad_is_pure_decay():
    """With zero gradients (and zero moments), the update is exactly
    -lr * (wd/lr_peak) * theta — the reference's independent weight decay."""
    master = torch.randn(100)
    expect = master * (1 - 1e-3 * 0.1)
    ops.adamw_step(master, torch.zeros(100), torch.zeros(100),
                   torch.zeros(100), None, lr=1e-3, beta1=0.9, beta2=0.95,
                   eps=1e-8, wd_over_peak_lr=0.1, grad_scale=1.0,
                   clip_norm=1.0, sq_sum=torch.zeros(()), step=1)
    assert torch.allclose(master, expect, atol=1e-7)


@given(st.integers(0, 2 ** 31 - 1))
@settings(max_examples=10, deadline=None)
def test_grad_clip_bounds_update_norm(seed):
    """After clipping, the effective gradient norm never exceeds clip_norm."""
    g = torch.Generator().manual_seed(seed)
    grad = torch.randn(256, generator=g) * 100  # way over the clip
    m = torch.zeros(256)
    v = torch.zeros(256)
    master = torch.zeros(256)
    sq = (grad * grad).sum()
    ops.adamw_step(master, grad, m, v, None, lr=1.0, beta1=0.0, beta2=0.0,
                   eps=1e-8, wd_over_peak_lr=0.0, grad_scale=1.0,
                   clip_norm=1.0, sq_sum=sq, step=1)
    # b1=b2=0 -> m = g_clipped; mhat = m; vhat = g^2; update = sign-ish...
    # check the clipped gradient's norm directly via m
    assert float(m.norm()) <= 1.0 + 1e-5


@given(total=st.integers(1, 5000), world=st.integers(1, 8),
       nb=st.integers(1, 6))
@settings(max_examples=60, deadline=None)
def test_bucket_piece_map_partitions_buffer(total, world, nb):
    """The bucketed ZeRO piece maps must tile the padded buffer exactly:
    every rank's pieces are disjoint, bucket-aligned, and the union over
    ranks covers [0, padded) (checkpoint reassembly correctness)."""
    shard = (total + world - 1) // world
    padded = shard * world
    nb_eff = max(1, min(nb, shard))
    cuts = [shard * i // nb_eff for i in range(nb_eff + 1)]
    buckets = [(cuts[i] * world, (cuts[i + 1] - cuts[i]) * world)
               for i in range(nb_eff)]
    assert sum(n for _, n in buckets) == padded
    covered = []
    for rank in range(world):
        po = 0
        for (o_b, n_b) in buckets:
            p_b = n_b // world
            covered.append((o_b + rank * p_b, p_b))
            po += p_b
        assert po == shard
    covered.sort()
    pos = 0
    for off, n in covered:
        if n == 0:
            continue
        assert off == pos, (off, pos)
        pos += n
    assert pos == padded
