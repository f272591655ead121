"""Minimal repro: capture ONLY conv2d_fwd in a hipGraph, replay, compare."""
import sys
import torch
sys.path.insert(0, ".")
from fast_autoaugment_amd.ops import ext
C = ext()

def trial(seed):
    torch.manual_seed(seed)
    x = (torch.randn(128, 16, 32, 32, device="cuda") * 0.5).bfloat16().contiguous(
        memory_format=torch.channels_last)
    w = (torch.randn(32, 16, 3, 3, device="cuda") * 0.1).bfloat16().contiguous(
        memory_format=torch.channels_last)
    b = torch.randn(32, device="cuda").bfloat16()
    ref = C.conv2d_fwd(x, w, b, 1, 1).float().clone()
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(3):
            C.conv2d_fwd(x, w, b, 1, 1)
    torch.cuda.current_stream().wait_stream(side)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        y = C.conv2d_fwd(x, w, b, 1, 1)
    bad = 0
    for i in range(10):
        g.replay()
        torch.cuda.synchronize()
        d = (y.float() - ref).abs().max().item()
        if d > 0.1 or torch.isnan(y.float()).any():
            bad += 1
    return bad

tot = 0
for s in range(6):
    bad = trial(s)
    tot += bad
    print(f"trial {s}: {bad}/10 replays wrong")
print("TOTAL bad replays:", tot)


def trial_colsum(seed):
    torch.manual_seed(seed)
    dy = (torch.randn(128, 32, 32, 32, device="cuda") * 0.3).bfloat16().contiguous(
        memory_format=torch.channels_last)
    ref = C.colsum_bf16(dy).float().clone()
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(3):
            C.colsum_bf16(dy)
    torch.cuda.current_stream().wait_stream(side)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        out = C.colsum_bf16(dy)
    bad = 0
    for i in range(10):
        g.replay()
        torch.cuda.synchronize()
        if (out.float() - ref).abs().max().item() > 0.5 or torch.isnan(out.float()).any():
            bad += 1
    return bad


print("=== colsum graph replay ===")
tot = 0
for s in range(6):
    b = trial_colsum(s)
    tot += b
    print(f"colsum trial {s}: {b}/10 wrong")
print("COLSUM total bad:", tot)
