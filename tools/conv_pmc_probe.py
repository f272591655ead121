import sys, torch
sys.path.insert(0, ".")
from fast_autoaugment_amd.ops import ext
C = ext()
# hot WRN stage-3 conv: 128ch 8x8 b128 k3
x = torch.randn(128, 128, 8, 8, device="cuda").bfloat16().contiguous(memory_format=torch.channels_last)
w = torch.randn(128, 128, 3, 3, device="cuda").bfloat16().contiguous(memory_format=torch.channels_last)
b = torch.randn(128, device="cuda").bfloat16()
for _ in range(50):
    y = C.conv2d_fwd(x, w, b, 1, 1)
torch.cuda.synchronize()
print("conv pmc probe done")
