"""Microbench of the BN channel-reduction kernels at the flagship shape
(n=180780 rows x h=256, bf16 streams) across PERTGNN_BN_BLOCKS settings.

Run (GPU box): for B in 512 1024 2048 2825; do
  PERTGNN_BN_BLOCKS=$B PYTHONPATH=. python benchmarks/bn_micro.py; done
"""
import os
import time

import torch


def timeit(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    import pertgnn._C as C

    dev = torch.device("cuda:0")
    n, h = 180780, 256
    x = torch.randn(n, h, device=dev).to(torch.bfloat16)
    g = torch.randn(n, h, device=dev).to(torch.bfloat16)
    y = torch.relu(torch.randn(n, h, device=dev)).to(torch.bfloat16)
    mean = torch.zeros(h, device=dev)
    invstd = torch.ones(h, device=dev)

    t_stats = timeit(lambda: C.bn_stats(x))
    t_bwd = timeit(lambda: C.bn_bwd_partials16(g, x, y, mean, invstd, True, 1.0))
    gamma = torch.ones(h, device=dev)
    part = C.bn_bwd_partials16(g, x, y, mean, invstd, True, 1.0)
    t_apply = timeit(lambda: C.bn_bwd_apply16(g, x, y, mean, invstd, gamma,
                                              part, part, True, 1.0))
    gb = n * h * 2 / 1e9
    blocks = os.environ.get("PERTGNN_BN_BLOCKS", "512(default)")
    print(f"blocks={blocks:>14}  bn_stats {t_stats:7.1f} us ({gb / t_stats * 1e6 / 1e3:4.1f} TB/s)"
          f"  bn_bwd_partials {t_bwd:7.1f} us ({3 * gb / t_bwd * 1e6 / 1e3:4.1f} TB/s)"
          f"  bn_bwd_apply {t_apply:7.1f} us ({4 * gb / t_apply * 1e6 / 1e3:4.1f} TB/s)")

    # parity across block counts (vs plain torch fp32 on rounded operands)
    p = C.bn_stats(x)
    xs = x.float()
    ref_s = xs.sum(0)
    ref_q = (xs * xs).sum(0)
    rel = max((p[:h] - ref_s).abs().max().item() / ref_s.abs().max().item(),
              (p[h:2 * h] - ref_q).abs().max().item() / ref_q.abs().max().item())
    assert p[2 * h].item() == n
    print(f"  stats relerr vs fp32: {rel:.2e}")


if __name__ == "__main__":
    main()
