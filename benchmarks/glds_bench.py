"""A/B the glds (direct-to-LDS) GEMM path against the register-staging path
and hipBLASLt (torch bf16 matmul) at the trace-scale model shapes.

    python benchmarks/glds_bench.py [--iters 50]
"""
import argparse
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


def bench(fn, iters, *args):
    for _ in range(5):
        fn(*args)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn(*args)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    import pertgnn._C as C

    dev = torch.device("cuda:0")
    # (name, M, N(out), K): qkvs fwd + dgrad at the realistic bench scale
    shapes = [
        ("qkvs fwd", 180736, 1024, 256),
        ("qkvs odd-m", 180186, 1024, 256),  # 26-row wgrad tail + 90-row dgrad tail
        ("qkvs fwd b256", 45056, 1024, 256),
        ("512H qkvs", 90112, 2048, 512),
    ]
    print(f"{'name':<16} {'M':>7} {'N':>5} {'K':>4}  {'a16o16':>14} {'blaslt16':>14}")
    for name, m, n, k in shapes:
        x = (torch.randn(m, k, device=dev) * 0.5).to(torch.bfloat16)
        w = torch.randn(n, k, device=dev) * 0.1
        b = torch.randn(n, device=dev)
        # parity vs fp32 matmul on bf16-rounded operands
        y = C.linear_fwd_a16o16(x, w, b, False)
        ref = (x.float() @ w.to(torch.bfloat16).float().t() + b)
        err = (y.float() - ref).abs().max().item()
        rel = err / ref.abs().max().item()
        t = bench(C.linear_fwd_a16o16, args.iters, x, w, b, False)
        wt16 = w.to(torch.bfloat16)
        trb = bench(lambda: x @ wt16.t(), args.iters)
        fl = 2.0 * m * n * k
        print(f"{name:<16} {m:>7} {n:>5} {k:>4}  "
              f"{t * 1e6:7.1f}us {fl / t / 1e12:5.0f}TF  "
              f"{trb * 1e6:7.1f}us {fl / trb / 1e12:5.0f}TF  relerr={rel:.2e}")

        # dgrad: dx[M,K'] = g[M,N'] . w'[N',K'] with N'=n? model: g[M,4H] w4[4H,H]
        g = (torch.randn(m, n, device=dev) * 0.5).to(torch.bfloat16)
        w2 = torch.randn(n, k, device=dev) * 0.1  # [n rows, k cols]
        dx = C.linear_dgrad16_o16(g, w2, False)
        ref2 = g.float() @ w2.to(torch.bfloat16).float()
        err2 = (dx.float() - ref2).abs().max().item()
        rel2 = err2 / ref2.abs().max().item()
        td = bench(C.linear_dgrad16_o16, args.iters, g, w2, False)
        w216 = w2.to(torch.bfloat16)
        trd = bench(lambda: g @ w216, args.iters)
        print(f"{'  dgrad':<16} {m:>7} {n:>5} {k:>4}  "
              f"{td * 1e6:7.1f}us {fl / td / 1e12:5.0f}TF  "
              f"{trd * 1e6:7.1f}us {fl / trd / 1e12:5.0f}TF  relerr={rel2:.2e}")

        # wgrad: dw[n,k] = g[m,n]^T x[m,k] (+ fused bias grad)
        dw, db = C.linear_wgrad16_b16(g, x, True, False)
        ref3 = g.float().t() @ x.float()
        rel3 = (dw - ref3).abs().max().item() / ref3.abs().max().item()
        reldb = (db - g.float().sum(0)).abs().max().item() / max(
            g.float().sum(0).abs().max().item(), 1e-9)
        tw = bench(C.linear_wgrad16_b16, args.iters, g, x, True, False)
        trw = bench(lambda: g.t().float() @ x.float(), 5)
        print(f"{'  wgrad':<16} {m:>7} {n:>5} {k:>4}  "
              f"{tw * 1e6:7.1f}us {fl / tw / 1e12:5.0f}TF  "
              f"{trw * 1e6:7.1f}us {fl / trw / 1e12:5.0f}TF  relerr={rel3:.2e}"
              f" db={reldb:.2e}")


if __name__ == "__main__":
    main()
