import sys, torch
sys.path.insert(0, ".")
from fast_autoaugment_amd.ops import ext
C = ext()
# hot WRN convs through the round-2 DIRECT tiled kernel: stage-3 (128ch
# 8x8) and stage-1 (32ch 32x32), b128 k3 s1, 50 dispatches each
for (Cin, H, Cout) in [(128, 8, 128), (32, 32, 32)]:
    x = torch.randn(128, Cin, H, H, device="cuda").bfloat16().contiguous(
        memory_format=torch.channels_last)
    w = torch.randn(Cout, Cin, 3, 3, device="cuda").bfloat16().contiguous(
        memory_format=torch.channels_last)
    b = torch.randn(Cout, device="cuda").bfloat16()
    for _ in range(50):
        y = C.conv2d_fwd(x, w, b, 1, 1)
    torch.cuda.synchronize()
print("conv pmc probe done")
