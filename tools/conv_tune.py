#!/usr/bin/env python3
"""Round-2 conv tuning sweep: per-shape fwd time across tile forcings and
split-K, vs MIOpen. Run on a GPU box:

  python tools/conv_tune.py            # flagship + deep-stage shapes
  python tools/conv_tune.py --big      # WRN-28-10 / shake C>=256 shapes
"""
import argparse
import os
import sys
import time

sys.path.insert(0, ".")
import torch


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--big", action="store_true")
    ap.add_argument("--d8", action="store_true",
                    help="sweep FAA_CONV_D8 variants on the deep 8x8 shapes")
    ap.add_argument("--wrw", action="store_true",
                    help="sweep wrw v3 slice counts vs v2 vs MIOpen")
    ap.add_argument("--dw", action="store_true",
                    help="depthwise fwd: v3 LDS-tile vs register-sliding vs torch")
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    if args.dw:
        import math
        from fast_autoaugment_amd.ops import ext as _ext
        C = _ext()
        shapes_d = [(64, 32, 112, 3, 1), (64, 96, 112, 3, 2), (64, 144, 56, 5, 2),
                    (64, 240, 28, 5, 1), (64, 576, 14, 5, 1), (64, 1152, 7, 5, 1)]
        print(f"{'shape':<24} {'torch':>8} {'v3':>8} {'old':>8}")
        for B, Ch, H, k, s in shapes_d:
            x = (torch.randn(B, Ch, H, H, device="cuda") * 0.5).bfloat16() \
                .contiguous(memory_format=torch.channels_last)
            w = (torch.randn(Ch, 1, k, k, device="cuda") * 0.2).bfloat16() \
                .contiguous(memory_format=torch.channels_last)
            ph = max((math.ceil(H / s) - 1) * s + k - H, 0)
            pl, pt = ph // 2, ph // 2
            pr, pb = ph - pl, ph - pt
            xp = torch.nn.functional.pad(x.float(), (pl, pr, pt, pb)).bfloat16() \
                .contiguous(memory_format=torch.channels_last)
            row = [f"{B}x{Ch}x{H}^2 k{k}s{s}"]
            row.append(f"{bench(lambda: torch.nn.functional.conv2d(xp, w, stride=s, groups=Ch), args.iters):8.1f}")
            os.environ.pop("FAA_DW_V3", None)
            row.append(f"{bench(lambda: C.dwconv_fwd(x, w, torch.Tensor(), s, pt, pb, pl, pr), args.iters):8.1f}")
            os.environ["FAA_DW_V3"] = "0"
            row.append(f"{bench(lambda: C.dwconv_fwd(x, w, torch.Tensor(), s, pt, pb, pl, pr), args.iters):8.1f}")
            os.environ.pop("FAA_DW_V3", None)
            print(" ".join(row))
        return
    if args.wrw:
        from fast_autoaugment_amd.ops import ext as _ext
        C = _ext()
        shapes_w = [(128, 16, 32, 32, 3, 1), (128, 32, 32, 32, 3, 1),
                    (128, 64, 16, 64, 3, 1), (128, 128, 8, 128, 3, 1),
                    (128, 160, 32, 160, 3, 1), (128, 320, 16, 320, 3, 1),
                    (128, 640, 8, 640, 3, 1)]
        slc = ["8", "16", "32", "64", "128", "256"]
        print(f"{'shape':<24} {'miopen':>8} {'v2':>8} {'v3auto':>8} {'v3ws':>8} "
              + " ".join(f"s{v:>7}" for v in slc))
        for B, Cin, H, Cout, k, s in shapes_w:
            x = (torch.randn(B, Cin, H, H, device="cuda") * 0.5).bfloat16() \
                .contiguous(memory_format=torch.channels_last)
            dy = (torch.randn(B, Cout, H, H, device="cuda") * 0.1).bfloat16() \
                .contiguous(memory_format=torch.channels_last)
            w_shape = [Cout, Cin, k, k]
            row = [f"{B}x{Cin}x{H}^2->{Cout}"]
            row.append(f"{bench(lambda: torch.nn.grad.conv2d_weight(x, w_shape, dy, stride=s, padding=k//2), args.iters):8.1f}")
            os.environ.pop("FAA_WRW_V3", None)
            row.append(f"{bench(lambda: C.conv2d_bwd_weight(dy, x, s, k//2, k, k, False), args.iters):8.1f}")
            os.environ["FAA_WRW_V3"] = "1"
            os.environ.pop("FAA_WRW3_SLICES", None)
            row.append(f"{bench(lambda: C.conv2d_bwd_weight(dy, x, s, k//2, k, k, False), args.iters):8.1f}")
            os.environ["FAA_WRW3_WSPLIT"] = "1"
            row.append(f"{bench(lambda: C.conv2d_bwd_weight(dy, x, s, k//2, k, k, False), args.iters):8.1f}")
            os.environ.pop("FAA_WRW3_WSPLIT", None)
            for v in slc:
                os.environ["FAA_WRW3_SLICES"] = v
                row.append(f"{bench(lambda: C.conv2d_bwd_weight(dy, x, s, k//2, k, k, False), args.iters):8.1f}")
            os.environ.pop("FAA_WRW3_SLICES", None)
            os.environ.pop("FAA_WRW_V3", None)
            print(" ".join(row))
        return
    if args.d8:
        from fast_autoaugment_amd.ops import ext as _ext
        C = _ext()
        variants = ["ib2", "ib4", "ib2sk2", "ib4sk2", "ib4sk4"]
        shapes8 = [(128, 128, 8, 128, 3, 1), (128, 256, 8, 256, 3, 1),
                   (128, 384, 8, 384, 3, 1), (128, 640, 8, 640, 3, 1)]
        print(f"{'shape':<26} {'miopen':>8} {'auto':>8} "
              + " ".join(f"{v:>8}" for v in variants))
        for B, Cin, H, Cout, k, s in shapes8:
            x = (torch.randn(B, Cin, H, H, device="cuda") * 0.5).bfloat16() \
                .contiguous(memory_format=torch.channels_last)
            w = (torch.randn(Cout, Cin, k, k, device="cuda") * 0.05).bfloat16() \
                .contiguous(memory_format=torch.channels_last)
            row = [f"{B}x{Cin}x{H}^2->{Cout}"]
            row.append(f"{bench(lambda: torch.nn.functional.conv2d(x, w, stride=s, padding=k//2), args.iters):8.1f}")
            os.environ.pop("FAA_CONV_D8", None)
            os.environ["FAA_CONV_DIRECT"] = "big"
            row.append(f"{bench(lambda: C.conv2d_fwd(x, w, torch.Tensor(), s, k//2), args.iters):8.1f}")
            for v in variants:
                os.environ["FAA_CONV_D8"] = v
                row.append(f"{bench(lambda: C.conv2d_fwd(x, w, torch.Tensor(), s, k//2), args.iters):8.1f}")
            os.environ.pop("FAA_CONV_D8", None)
            os.environ.pop("FAA_CONV_DIRECT", None)
            print(" ".join(row))
        return
    from fast_autoaugment_amd.ops import ext
    C = ext()
    shapes = ([(128, 256, 16, 256, 3, 1), (128, 384, 8, 384, 3, 1),
               (128, 640, 8, 640, 3, 1), (128, 320, 16, 320, 3, 1)]
              if args.big else
              [(128, 16, 32, 32, 3, 1), (128, 32, 32, 32, 3, 1),
               (128, 64, 16, 64, 3, 1), (128, 128, 8, 128, 3, 1),
               (128, 64, 16, 128, 3, 2)])
    print(f"{'shape':<28} {'miopen':>8} {'direct':>8} {'im2col':>8} "
          + " ".join(f"{t:>8}" for t in ["32x32", "32x64", "64x32", "64x64"])
          + f" {'i2c+sk':>8}")
    for B, Cin, H, Cout, k, s in shapes:
        x = (torch.randn(B, Cin, H, H, device="cuda") * 0.5).bfloat16() \
            .contiguous(memory_format=torch.channels_last)
        w = (torch.randn(Cout, Cin, k, k, device="cuda") * 0.05).bfloat16() \
            .contiguous(memory_format=torch.channels_last)
        row = [f"{B}x{Cin}x{H}^2->{Cout} k{k}s{s}"]
        row.append(f"{bench(lambda: torch.nn.functional.conv2d(x, w, stride=s, padding=k//2), args.iters):8.1f}")
        os.environ.pop("FAA_CONV_TILE", None)
        os.environ.pop("FAA_CONV_SPLITK", None)
        os.environ.pop("FAA_CONV_DIRECT", None)     # default: direct when eligible
        row.append(f"{bench(lambda: C.conv2d_fwd(x, w, torch.Tensor(), s, k//2), args.iters):8.1f}")
        os.environ["FAA_CONV_DIRECT"] = "0"
        row.append(f"{bench(lambda: C.conv2d_fwd(x, w, torch.Tensor(), s, k//2), args.iters):8.1f}")
        for tile in ["32x32", "32x64", "64x32", "64x64"]:
            os.environ["FAA_CONV_TILE"] = tile
            row.append(f"{bench(lambda: C.conv2d_fwd(x, w, torch.Tensor(), s, k//2), args.iters):8.1f}")
        os.environ.pop("FAA_CONV_TILE", None)
        os.environ["FAA_CONV_SPLITK"] = "1"
        row.append(f"{bench(lambda: C.conv2d_fwd(x, w, torch.Tensor(), s, k//2), args.iters):8.1f}")
        os.environ.pop("FAA_CONV_SPLITK", None)
        os.environ.pop("FAA_CONV_DIRECT", None)
        print(" ".join(row))


if __name__ == "__main__":
    main()
