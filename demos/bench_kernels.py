"""Kernel micro-benchmark harness: HIP kernels vs their eager references.

Run on an MI355X:
    python demos/bench_kernels.py [--iters 50]
Prints per-op timings (ms) for the fused kernels and the eager PyTorch
equivalent at representative shapes.  CPU fallback times the eager path
only (useful as a smoke test of the harness itself).
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters, sync):
    fn()  # warmup
    if sync:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    if sync:
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=50)
    args = p.parse_args()
    cuda = torch.cuda.is_available()
    dev = "cuda" if cuda else "cpu"
    dt = torch.bfloat16 if cuda else torch.float32
    rows = []

    from agilerl_amd import ops

    # GAE scan (T, N)
    T, N = 128, 65536 if cuda else 512
    r = torch.randn(T, N, device=dev)
    v = torch.randn(T, N, device=dev)
    d = (torch.rand(T, N, device=dev) < 0.01).float()
    lv = torch.randn(N, device=dev)
    rows.append(("gae_scan", f"({T},{N})",
                 timeit(lambda: ops.gae_scan(r, v, d, lv, 0.99, 0.95), args.iters, cuda)))

    # RMSNorm (B*T, H)
    x = torch.randn(8192, 4096, device=dev, dtype=dt)
    w = torch.ones(4096, device=dev, dtype=dt)
    rows.append(("rms_norm", "(8192,4096)",
                 timeit(lambda: ops.rms_norm(x, w), args.iters, cuda)))
    ref = torch.nn.RMSNorm(4096, device=dev, dtype=dt)
    rows.append(("rms_norm/eager", "(8192,4096)",
                 timeit(lambda: ref(x), args.iters, cuda)))

    # SwiGLU
    g = torch.randn(8192, 4096, device=dev, dtype=dt)
    u = torch.randn(8192, 4096, device=dev, dtype=dt)
    rows.append(("swiglu", "(8192,4096)",
                 timeit(lambda: ops.swiglu(g, u), args.iters, cuda)))
    rows.append(("swiglu/eager", "(8192,4096)",
                 timeit(lambda: torch.nn.functional.silu(g) * u, args.iters, cuda)))

    # fused lm_head logprobs
    if cuda:
        from agilerl_amd.ops.fused_logprobs import fused_linear_logprobs

        h = torch.randn(4096, 4096, device=dev, dtype=dt)
        W = torch.randn(32000, 4096, device=dev, dtype=dt)
        tgt = torch.randint(0, 32000, (4096,), device=dev)
        rows.append(("fused_logprobs", "(4096,4096)x(32000,4096)",
                     timeit(lambda: fused_linear_logprobs(h, W, tgt), args.iters, True)))

    # paged attention decode
    from agilerl_amd.ops.paged_attn import paged_attention_decode

    B, Hq, Hkv, D, S, P = (64, 32, 8, 128, 16, 4096) if cuda else (4, 4, 2, 16, 4, 16)
    q = torch.randn(B, Hq, D, device=dev, dtype=dt)
    kp = torch.randn(P, S, Hkv, D, device=dev, dtype=dt)
    vp = torch.randn(P, S, Hkv, D, device=dev, dtype=dt)
    maxp = P // B
    table = torch.arange(B * maxp, device=dev, dtype=torch.int32).reshape(B, maxp)
    lengths = torch.full((B,), S * maxp - 3, device=dev, dtype=torch.int32)
    rows.append(("paged_attn_decode", f"B{B} Hq{Hq} len{int(lengths[0])}",
                 timeit(lambda: paged_attention_decode(q, kp, vp, table, lengths),
                        args.iters, cuda)))
    rows.append(("paged_attn(hint)", f"B{B} Hq{Hq} len{int(lengths[0])}",
                 timeit(lambda: paged_attention_decode(
                     q, kp, vp, table, lengths, max_len_hint=int(lengths[0])),
                        args.iters, cuda)))

    # skinny decode GEMM vs hipBLASLt at M=8 (Llama-3-8B projection shapes)
    if cuda:
        from agilerl_amd.ops.backend import extension as _ext

        ext = _ext()
        for (M, N, K) in [(8, 4096, 4096), (8, 1024, 4096), (8, 14336, 4096),
                          (8, 4096, 14336), (8, 128256, 4096)]:
            x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
            w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
            gb = N * K * 2 / 1e9
            ms_sg = timeit(lambda: ext.skinny_gemm(x, w), args.iters, True)
            ms_bl = timeit(lambda: x @ w.t(), args.iters, True)
            rows.append((f"skinny_gemm", f"({M},{K})x({N},{K})T {gb:.2f}GB",
                         ms_sg))
            rows.append((f"skinny/blaslt", f"-> {gb/ms_sg*1000:.0f} vs {gb/ms_bl*1000:.0f} GB/s",
                         ms_bl))

    print(f"{'op':<22} {'shape':<28} {'ms':>8}")
    for name, shape, ms in rows:
        print(f"{name:<22} {shape:<28} {ms:8.3f}")


if __name__ == "__main__":
    main()
