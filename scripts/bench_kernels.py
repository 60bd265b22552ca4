"""Microbenchmarks for the HIP hot kernels (run on the GPU box).
Prints per-kernel time and effective TF/s or TB/s."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import argparse
import time

import torch

from midgpt_amd import ops
from midgpt_amd.ops import reference as ref

assert torch.cuda.is_available()
DEV = "cuda:0"


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_attn(B=32, H=16, T=1024, C=128):
    q = torch.randn(B, H, T, C, device=DEV, dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    o, lse = ops._C.attn_fwd(q, k, v)
    do = torch.randn_like(o)
    t_fwd = timeit(lambda: ops._C.attn_fwd(q, k, v))
    t_bwd = timeit(lambda: ops._C.attn_bwd(do, q, k, v, o, lse))
    flops_fwd = 4 * B * H * T * T * C / 2  # causal
    flops_bwd = flops_fwd * 2.5
    print(f"attn_fwd  B{B} H{H} T{T} C{C}: {t_fwd*1e3:8.3f} ms  "
          f"{flops_fwd/t_fwd/1e12:7.1f} TF/s (causal-effective)")
    print(f"attn_bwd  B{B} H{H} T{T} C{C}: {t_bwd*1e3:8.3f} ms  "
          f"{flops_bwd/t_bwd/1e12:7.1f} TF/s (causal-effective)")


def bench_gelu(N=131072, D=8192):
    x = torch.randn(N, D, device=DEV).to(torch.bfloat16)
    dy = torch.randn_like(x)
    t = timeit(lambda: ops._C.gelu_fwd(x))
    print(f"gelu_fwd N{N} D{D}: {t*1e3:8.3f} ms  {2*N*D*2/t/1e12:6.2f} TB/s")
    t = timeit(lambda: ops._C.gelu_bwd(dy, x))
    print(f"gelu_bwd N{N} D{D}: {t*1e3:8.3f} ms  {3*N*D*2/t/1e12:6.2f} TB/s")


def bench_embedding(N=131072, V=50304, D=2048):
    idx = torch.randint(0, V, (N,), device=DEV)
    dy = torch.randn(N, D, device=DEV).to(torch.bfloat16)
    t = timeit(lambda: ops._C.embedding_bwd(dy, idx, V), iters=10)
    print(f"embedding_bwd N{N} V{V} D{D}: {t*1e3:8.3f} ms")


def bench_rmsnorm(N=131072, D=2048):
    x = torch.randn(N, D, device=DEV, dtype=torch.bfloat16)
    y, invr = ops._C.rmsnorm_fwd(x, None, 1e-6)
    t = timeit(lambda: ops._C.rmsnorm_fwd(x, None, 1e-6))
    gb = 2 * N * D * 2 / 1e9
    print(f"rmsnorm_fwd N{N} D{D}: {t*1e3:8.3f} ms  {gb/t/1e3:6.2f} TB/s")
    dy = torch.randn_like(x)
    t = timeit(lambda: ops._C.rmsnorm_bwd(dy, x, None, invr, 1e-6))
    gb = 5 * N * D * 2 / 1e9
    print(f"rmsnorm_bwd N{N} D{D}: {t*1e3:8.3f} ms  {gb/t/1e3:6.2f} TB/s")


def bench_qkv_prep(B=32, T=1024, H=16, C=128):
    qkv = torch.randn(B, T, 3, H, C, device=DEV, dtype=torch.bfloat16)
    qw = torch.ones(C, device=DEV)
    kw = torch.ones(C, device=DEV)
    sin, cos = ref.rope_tables(C, T, device=DEV)
    sin, cos = sin.contiguous(), cos.contiguous()
    t = timeit(lambda: ops._C.qkv_prep_fwd(qkv, qw, kw, sin, cos, 1e-6))
    gb = 2 * B * T * 3 * H * C * 2 / 1e9
    print(f"qkv_prep_fwd: {t*1e3:8.3f} ms  {gb/t/1e3:6.2f} TB/s")


def bench_ce(N=131072, V=50304):
    logits = torch.randn(N, V, device=DEV, dtype=torch.bfloat16)
    targets = torch.randint(0, V, (N,), device=DEV)
    loss, lse = ops._C.ce_fwd(logits, targets)
    t = timeit(lambda: ops._C.ce_fwd(logits, targets), iters=5)
    gb = N * V * 2 / 1e9
    print(f"ce_fwd N{N} V{V}: {t*1e3:8.3f} ms  {gb/t/1e3:6.2f} TB/s")
    gs = torch.ones(1, device=DEV)
    t = timeit(lambda: ops._C.ce_bwd(logits, targets, lse, gs), iters=5)
    print(f"ce_bwd N{N} V{V}: {t*1e3:8.3f} ms  {2*gb/t/1e3:6.2f} TB/s")


def bench_adamw(n=1_600_000_000 // 8):
    master = torch.randn(n, device=DEV)
    grad = torch.randn(n, device=DEV)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    out = torch.empty(n, device=DEV, dtype=torch.bfloat16)
    sq = (grad * grad).sum()
    t = timeit(lambda: ops._C.adamw_step(master, grad, m, v, out, True, sq,
                                         1e-3, 0.9, 0.95, 1e-8, 0.1, 1.0, 1.0, 5),
               iters=5)
    gb = n * (4 * 4 + 3 * 4 + 2) / 1e9  # r: m,v,g,master; w: m,v,master; bf16
    print(f"adamw n{n}: {t*1e3:8.3f} ms  {gb/t/1e3:6.2f} TB/s")


def bench_gemm():
    for (M, K, N, tag) in [(32768, 2048, 6144, "c_attn"),
                           (32768, 2048, 8192, "c_fc"),
                           (32768, 8192, 2048, "mlp_proj"),
                           (32768, 2048, 50304, "lm_head")]:
        a = torch.randn(M, K, device=DEV, dtype=torch.bfloat16)
        b = torch.randn(N, K, device=DEV, dtype=torch.bfloat16)
        t = timeit(lambda: a @ b.t(), iters=10)
        fl = 2 * M * K * N
        print(f"gemm {tag} {M}x{K}x{N}: {t*1e3:8.3f} ms  {fl/t/1e12:7.1f} TF/s")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--which", default="all")
    args = p.parse_args()
    w = args.which
    if w in ("all", "attn"):
        bench_attn()
        bench_attn(B=32, H=12, T=1024, C=64)
        bench_attn(B=4, H=32, T=4096, C=128)  # 7B shape
    if w in ("all", "gelu"):
        bench_gelu()
    if w in ("all", "embed"):
        bench_embedding()
    if w in ("all", "norm"):
        bench_rmsnorm()
        bench_qkv_prep()
    if w in ("all", "ce"):
        bench_ce()
    if w in ("all", "adamw"):
        bench_adamw()
    if w in ("all", "gemm"):
        bench_gemm()
