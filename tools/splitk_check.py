#!/usr/bin/env python3
"""Split-K conv fwd numerics check (FAA_CONV_SPLITK=1 vs torch fp32)."""
import os
import sys
sys.path.insert(0, ".")
os.environ["FAA_CONV_SPLITK"] = "1"
import torch
from fast_autoaugment_amd.ops import ext
C = ext()
torch.manual_seed(0)
for B, Cin, H, Cout, k, s in [(128, 64, 8, 64, 3, 1), (8, 32, 8, 64, 3, 2),
                              (128, 128, 8, 128, 3, 1), (4, 48, 9, 40, 1, 1)]:
    x = torch.randn(B, Cin, H, H, device="cuda") * 0.5
    w = torch.randn(Cout, Cin, k, k, device="cuda") * 0.05
    b = torch.randn(Cout, device="cuda") * 0.1
    ref = torch.nn.functional.conv2d(x, w, b, stride=s, padding=k // 2)
    got = C.conv2d_fwd(x.bfloat16().contiguous(memory_format=torch.channels_last),
                       w.bfloat16().contiguous(memory_format=torch.channels_last),
                       b.bfloat16(), s, k // 2).float()
    err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    print(f"B{B} C{Cin}->{Cout} H{H} k{k} s{s}: rel err {err:.4f}")
    assert err < 2e-2, "SPLITK MISMATCH"
print("SPLITK_OK")
