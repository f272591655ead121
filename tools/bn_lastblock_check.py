#!/usr/bin/env python3
"""Round-2 validation for FAA_BN_LASTBLOCK=1 (run BEFORE adding a GPU
test for it — the kernel tail was written without GPU budget left in
round 1)."""
import sys
sys.path.insert(0, ".")
import torch


def main():
    """FAA_BN_LASTBLOCK=1 (last reduce block performs the finalize inline,
    staged for round 2) must match the default two-kernel path."""
    import os
    torch.manual_seed(3)
    for Ch in (32, 61):
        x = torch.randn(8, Ch, 16, 16, device=torch.device("cuda:0"), dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        bn = torch.nn.BatchNorm2d(Ch, momentum=0.3).to(torch.device("cuda:0"))
        bn2 = torch.nn.BatchNorm2d(Ch, momentum=0.3).to(torch.device("cuda:0"))
        bn2.load_state_dict(bn.state_dict())
        from fast_autoaugment_amd.ops.bnrelu import fused_bn_relu
        bn.train(); bn2.train()
        out_ref = fused_bn_relu(x, bn)
        os.environ["FAA_BN_LASTBLOCK"] = "1"
        try:
            out = fused_bn_relu(x, bn2)
        finally:
            os.environ.pop("FAA_BN_LASTBLOCK", None)
        assert (out.float() - out_ref.float()).abs().max().item() < 1e-5
        assert (bn2.running_mean - bn.running_mean).abs().max().item() < 1e-5
        assert (bn2.running_var - bn.running_var).abs().max().item() < 1e-5


def main_bwd():
    import os
    torch.manual_seed(5)
    dev = torch.device("cuda:0")
    for Ch in (32, 61):
        x = torch.randn(8, Ch, 16, 16, device=dev, dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last).requires_grad_(True)
        x2 = x.detach().clone().requires_grad_(True)
        bn = torch.nn.BatchNorm2d(Ch).to(dev)
        bn2 = torch.nn.BatchNorm2d(Ch).to(dev)
        bn2.load_state_dict(bn.state_dict())
        from fast_autoaugment_amd.ops.bnrelu import fused_bn_relu
        g = torch.randn(8, Ch, 16, 16, device=dev, dtype=torch.bfloat16)
        fused_bn_relu(x, bn).backward(g)
        os.environ["FAA_BN_LASTBLOCK"] = "1"
        try:
            fused_bn_relu(x2, bn2).backward(g)
        finally:
            os.environ.pop("FAA_BN_LASTBLOCK", None)
        assert (x.grad.float() - x2.grad.float()).abs().max().item() < 1e-5
        assert (bn.weight.grad.float() - bn2.weight.grad.float()).abs().max().item() < 1e-4
        assert (bn.bias.grad.float() - bn2.bias.grad.float()).abs().max().item() < 1e-4


if __name__ == "__main__":
    main()
    main_bwd()
    print("BN_LASTBLOCK_OK")
