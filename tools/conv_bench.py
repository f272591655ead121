#!/usr/bin/env python3
"""Per-shape conv microbenchmark: MFMA kernels vs torch/MIOpen.

Times fwd / bwd-data / bwd-weight for the WRN conv shapes at batch 128 and
prints a table + dispatch recommendation. Run on the GPU box:
  python tools/conv_bench.py
"""
import sys
import time

import torch

sys.path.insert(0, ".")
from fast_autoaugment_amd.ops import ext

C = ext()

SHAPES = [
    # Cin, H, Cout, k, stride        (WRN-40-2 & WRN-28-10 shapes)
    (3, 32, 16, 3, 1),
    (16, 32, 32, 3, 1), (32, 32, 32, 3, 1),
    (32, 32, 64, 3, 2), (64, 16, 64, 3, 1),
    (64, 16, 128, 3, 2), (128, 8, 128, 3, 1),
    (16, 32, 32, 1, 1), (32, 16, 64, 1, 2), (64, 8, 128, 1, 2),
    (16, 32, 160, 3, 1), (160, 32, 160, 3, 1),
    (160, 32, 320, 3, 2), (320, 16, 320, 3, 1),
    (320, 16, 640, 3, 2), (640, 8, 640, 3, 1),
    (3, 224, 64, 7, 2),    # ResNet ImageNet 7x7 stem
]


def timeit(fn, iters=50, warm=10):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    B = 128
    dev = torch.device("cuda")
    print(f"{'shape':>26} | {'dir':>4} | {'faa us':>8} | {'torch us':>8} | win")
    for (Cin, H, Cout, k, s) in SHAPES:
        x = torch.randn(B, Cin, H, H, device=dev).bfloat16().contiguous(
            memory_format=torch.channels_last)
        w = torch.randn(Cout, Cin, k, k, device=dev).bfloat16().contiguous(
            memory_format=torch.channels_last)
        b = torch.randn(Cout, device=dev).bfloat16()
        Ho = (H + 2 * (k // 2) - k) // s + 1
        dy = torch.randn(B, Cout, Ho, Ho, device=dev).bfloat16().contiguous(
            memory_format=torch.channels_last)

        tag = f"{Cin}x{H}x{H}->{Cout} k{k}s{s}"
        t_faa = timeit(lambda: C.conv2d_fwd(x, w, b, s, k // 2))
        t_ref = timeit(lambda: torch.nn.functional.conv2d(x, w, b, stride=s, padding=k // 2))
        print(f"{tag:>26} | {'fwd':>4} | {t_faa:8.1f} | {t_ref:8.1f} | {'faa' if t_faa < t_ref else 'torch'}")

        if s == 1:
            t_faa = timeit(lambda: C.conv2d_bwd_data(dy, w, 1, k // 2, H, H))
            t_ref = timeit(lambda: torch.nn.grad.conv2d_input(list(x.shape), w, dy,
                                                              stride=1, padding=k // 2))
            print(f"{tag:>26} | {'bwdD':>4} | {t_faa:8.1f} | {t_ref:8.1f} | {'faa' if t_faa < t_ref else 'torch'}")

        t_faa = timeit(lambda: C.conv2d_bwd_weight(dy, x, s, k // 2, k, k, True))
        t_ref = timeit(lambda: torch.nn.grad.conv2d_weight(x, list(w.shape), dy,
                                                           stride=s, padding=k // 2))
        print(f"{tag:>26} | {'bwdW':>4} | {t_faa:8.1f} | {t_ref:8.1f} | {'faa' if t_faa < t_ref else 'torch'}")


if __name__ == "__main__":
    main()
