"""CPU numerics tests for the reference op layer (the oracle the HIP
kernels are tested against) and the autograd dispatch wrappers."""

import torch

from midgpt_amd import ops
from midgpt_amd.ops import reference as ref


def test_rmsnorm_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(7, 64, dtype=torch.float64)
    y = ref.rmsnorm(x.float(), eps=1e-6)
    expect = x / torch.sqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6)
    assert torch.allclose(y.double(), expect, atol=1e-5)


def test_rmsnorm_autograd_matches_torch_autograd():
    torch.manual_seed(1)
    x = torch.randn(5, 32, requires_grad=True)
    y = ops.rmsnorm(x, None, 1e-6)
    g = torch.randn_like(y)
    (y * g).sum().backward()
    x2 = x.detach().clone().requires_grad_(True)
    y2 = x2 * torch.rsqrt(x2.pow(2).mean(-1, keepdim=True) + 1e-6)
    (y2 * g).sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)


def test_qk_layernorm_matches_torch():
    torch.manual_seed(2)
    x = torch.randn(3, 4, 5, 64)
    w = torch.randn(64)
    y = ref.qk_layernorm(x, w, eps=1e-6)
    expect = torch.nn.functional.layer_norm(x, (64,), weight=w, eps=1e-6)
    assert torch.allclose(y, expect, atol=1e-5)


def test_rope_shift_equivariance():
    """Ported property test (reference scripts/test_rotary.py): attention
    scores of RoPE'd Q,K are shift-equivariant along T."""
    torch.manual_seed(3)
    H, T, C, shift = 2, 32, 16, 5
    q = torch.randn(H, T, C)
    k = torch.randn(H, T, C)
    sin, cos = ref.rope_tables(C, T)
    qr = ref.apply_rope(q, sin, cos)
    kr = ref.apply_rope(k, sin, cos)
    scores = torch.matmul(qr, kr.transpose(-1, -2))
    q2 = torch.roll(q, shift, dims=1)
    k2 = torch.roll(k, shift, dims=1)
    qr2 = ref.apply_rope(q2, sin, cos)
    kr2 = ref.apply_rope(k2, sin, cos)
    scores2 = torch.matmul(qr2, kr2.transpose(-1, -2))
    valid = scores[:, : T - shift, : T - shift]
    valid2 = scores2[:, shift:, shift:]
    assert torch.allclose(valid, valid2, atol=1e-4)


def test_rope_interleaved_pairing():
    """GPT-J interleaved: rotate_every_two([a,b,c,d]) == [-b,a,-d,c]."""
    x = torch.tensor([1.0, 2.0, 3.0, 4.0])
    out = ref.rotate_every_two(x)
    assert torch.equal(out, torch.tensor([-2.0, 1.0, -4.0, 3.0]))


def test_attention_reference_vs_flash_dispatch_cpu():
    torch.manual_seed(4)
    B, H, T, C = 2, 3, 16, 8
    q = torch.randn(B, H, T, C)
    k = torch.randn(B, H, T, C)
    v = torch.randn(B, H, T, C)
    o1 = ref.causal_attention(q, k, v)
    o2 = ops.flash_attention(q, k, v)
    assert torch.allclose(o1, o2, atol=1e-5)


def test_attention_backward_matches_autograd():
    torch.manual_seed(5)
    B, H, T, C = 1, 2, 8, 4
    q = torch.randn(B, H, T, C, requires_grad=True)
    k = torch.randn(B, H, T, C, requires_grad=True)
    v = torch.randn(B, H, T, C, requires_grad=True)
    o = ops.flash_attention(q, k, v)
    g = torch.randn_like(o)
    (o * g).sum().backward()
    grads = [q.grad.clone(), k.grad.clone(), v.grad.clone()]
    q2, k2, v2 = (t.detach().clone().requires_grad_(True) for t in (q, k, v))
    o2 = ref.causal_attention(q2, k2, v2)
    (o2 * g).sum().backward()
    for a, b in zip(grads, [q2.grad, k2.grad, v2.grad]):
        assert torch.allclose(a, b, atol=1e-4), (a - b).abs().max()


def test_cross_entropy_matches_torch():
    torch.manual_seed(6)
    logits = torch.randn(64, 101, requires_grad=True)
    targets = torch.randint(0, 101, (64,))
    loss = ops.cross_entropy(logits, targets)
    expect = torch.nn.functional.cross_entropy(logits, targets)
    assert torch.allclose(loss, expect, atol=1e-6)
    loss.backward()
    l2 = logits.detach().clone().requires_grad_(True)
    torch.nn.functional.cross_entropy(l2, targets).backward()
    assert torch.allclose(logits.grad, l2.grad, atol=1e-6)


def test_qkv_prep_matches_composed_ops():
    torch.manual_seed(7)
    B, T, H, C = 2, 8, 3, 16
    qkv = torch.randn(B, T, 3, H, C, requires_grad=True)
    qw = torch.randn(C, requires_grad=True)
    kw = torch.randn(C, requires_grad=True)
    sin, cos = ref.rope_tables(C, T)
    q, k, v = ops.qkv_prep(qkv, qw, kw, sin, cos)
    # composed reference
    qr = ref.qk_layernorm(qkv[:, :, 0].permute(0, 2, 1, 3), qw)
    kr = ref.qk_layernorm(qkv[:, :, 1].permute(0, 2, 1, 3), kw)
    qe = ref.apply_rope(qr, sin, cos)
    ke = ref.apply_rope(kr, sin, cos)
    ve = qkv[:, :, 2].permute(0, 2, 1, 3)
    assert torch.allclose(q, qe, atol=1e-5)
    assert torch.allclose(k, ke, atol=1e-5)
    assert torch.allclose(v, ve, atol=1e-6)
    # backward vs autograd through the composed reference
    g = torch.randn_like(q)
    (q * g).sum().backward()
    got = qkv.grad.clone(), qw.grad.clone()
    qkv2 = qkv.detach().clone().requires_grad_(True)
    qw2 = qw.detach().clone().requires_grad_(True)
    qr2 = ref.qk_layernorm(qkv2[:, :, 0].permute(0, 2, 1, 3), qw2)
    qe2 = ref.apply_rope(qr2, sin, cos)
    (qe2 * g).sum().backward()
    assert torch.allclose(got[0], qkv2.grad, atol=1e-4)
    assert torch.allclose(got[1], qw2.grad, atol=1e-4)


def test_adamw_step_matches_optax_chain_semantics():
    """Simulate the optax chain clip->adam->decay->schedule->descend by hand
    and compare with ops.adamw_step."""
    torch.manual_seed(8)
    n = 257
    master = torch.randn(n)
    grad = torch.randn(n) * 3
    m = torch.zeros(n)
    v = torch.zeros(n)
    lr, b1, b2, eps = 3e-3, 0.9, 0.95, 1e-8
    wd_over_peak = 0.1
    # two steps to exercise bias correction
    master2, m2, v2 = master.clone(), m.clone(), v.clone()
    for step in (1, 2):
        g = grad * (0.5 ** step)
        sq = (g * g).sum()
        ops.adamw_step(master, g, m, v, None, lr=lr, beta1=b1, beta2=b2,
                       eps=eps, wd_over_peak_lr=wd_over_peak, grad_scale=1.0,
                       clip_norm=1.0, sq_sum=sq, step=step)
        # manual optax chain
        gn = float(sq.sqrt())
        gc = g * min(1.0, 1.0 / (gn + 1e-12))
        m2 = b1 * m2 + (1 - b1) * gc
        v2 = b2 * v2 + (1 - b2) * gc * gc
        mhat = m2 / (1 - b1 ** step)
        vhat = v2 / (1 - b2 ** step)
        upd = mhat / (vhat.sqrt() + eps) + wd_over_peak * master2
        master2 = master2 - lr * upd
        assert torch.allclose(master, master2, atol=1e-6)


def test_gelu_reference_path_matches_torch():
    """ops.gelu CPU fallback (fwd + hand-derived bwd) vs torch autograd."""
    import torch

    from midgpt_amd import ops
    torch.manual_seed(0)
    x = (torch.randn(64, 33) * 3).requires_grad_(True)
    y = ops.gelu(x)
    x2 = x.detach().clone().requires_grad_(True)
    y2 = torch.nn.functional.gelu(x2, approximate="tanh")
    assert torch.allclose(y, y2, atol=1e-6)
    g = torch.randn_like(y)
    (y * g).sum().backward()
    (y2 * g).sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-5), \
        (x.grad - x2.grad).abs().max()
