"""MFMA conv kernels vs plain torch (fp32 reference, asymmetric random data).

Guide rule G9: transpose bugs hide under symmetric inputs — all tests use
independent random tensors and check fwd, bwd-data, bwd-weight, bias grads.
"""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from fast_autoaugment_amd.ops import ext
    C = ext()


def dev():
    return torch.device("cuda:0")


SHAPES = [
    # B, Cin, H, Cout, k, stride   (WRN-40-2 + stress shapes)
    (8, 3, 32, 16, 3, 1),        # stem (Cin=3 tail masking)
    (8, 16, 32, 32, 3, 1),
    (8, 32, 32, 32, 3, 1),
    (8, 32, 32, 64, 3, 2),       # stride 2
    (8, 64, 16, 64, 3, 1),
    (8, 64, 16, 128, 3, 2),
    (8, 128, 8, 128, 3, 1),
    (8, 16, 32, 32, 1, 1),       # 1x1 shortcut
    (8, 64, 16, 128, 1, 2),      # 1x1 stride 2
    (4, 160, 16, 160, 3, 1),     # WRN-28-10 non-pow2 channels
    (3, 48, 9, 40, 3, 1),        # odd sizes (M not multiple of 64)
    # big-config shapes (direct tiled kernel, cin-slab loop, C >= 256)
    (4, 256, 16, 256, 3, 1),
    (4, 320, 8, 320, 3, 1),
    (2, 640, 8, 640, 3, 1),
    # ResNet-50 ImageNet spatials (masked direct tiles: 56/28/14/7)
    (2, 64, 56, 64, 3, 1),
    (2, 128, 28, 128, 3, 1),
    (2, 256, 14, 256, 3, 1),
    (2, 512, 7, 512, 3, 1),
]


def _mk(B, Cin, H, Cout, k, stride, seed=0):
    torch.manual_seed(seed)
    x = torch.randn(B, Cin, H, H, device=dev()) * 0.5
    w = torch.randn(Cout, Cin, k, k, device=dev()) * (1.0 / np.sqrt(Cin * k * k))
    b = torch.randn(Cout, device=dev()) * 0.1
    return x, w, b


@pytest.mark.parametrize("B,Cin,H,Cout,k,stride", SHAPES)
def test_conv_fwd(B, Cin, H, Cout, k, stride):
    x, w, b = _mk(B, Cin, H, Cout, k, stride)
    ref = torch.nn.functional.conv2d(x, w, b, stride=stride, padding=k // 2)
    xb = x.bfloat16().contiguous(memory_format=torch.channels_last)
    wb = w.bfloat16().contiguous(memory_format=torch.channels_last)
    got = C.conv2d_fwd(xb, wb, b.bfloat16(), stride, k // 2).float()
    scale = ref.abs().max().item() + 1e-3
    err = (got - ref).abs().max().item() / scale
    assert err < 2e-2, f"fwd rel err {err}"


@pytest.mark.parametrize("B,Cin,H,Cout,k,stride", [s for s in SHAPES if s[5] == 1])
def test_conv_bwd_data(B, Cin, H, Cout, k, stride):
    x, w, _ = _mk(B, Cin, H, Cout, k, stride, seed=1)
    dy = torch.randn(B, Cout, H, H, device=dev()) * 0.5
    ref = torch.nn.grad.conv2d_input(list(x.shape), w, dy, stride=1, padding=k // 2)
    dyb = dy.bfloat16().contiguous(memory_format=torch.channels_last)
    wb = w.bfloat16().contiguous(memory_format=torch.channels_last)
    got = C.conv2d_bwd_data(dyb, wb, 1, k // 2, H, H).float()
    scale = ref.abs().max().item() + 1e-3
    err = (got - ref).abs().max().item() / scale
    assert err < 2e-2, f"bwd-data rel err {err}"


@pytest.mark.parametrize("B,Cin,H,Cout,k,stride", SHAPES)
def test_conv_bwd_weight(B, Cin, H, Cout, k, stride):
    x, w, _ = _mk(B, Cin, H, Cout, k, stride, seed=2)
    Ho = (H + 2 * (k // 2) - k) // stride + 1
    dy = torch.randn(B, Cout, Ho, Ho, device=dev()) * 0.1
    ref_w = torch.nn.grad.conv2d_weight(x, list(w.shape), dy, stride=stride, padding=k // 2)
    ref_b = dy.sum(dim=(0, 2, 3))
    dyb = dy.bfloat16().contiguous(memory_format=torch.channels_last)
    xb = x.bfloat16().contiguous(memory_format=torch.channels_last)
    dw, db = C.conv2d_bwd_weight(dyb, xb, stride, k // 2, k, k, True)
    scale = ref_w.abs().max().item() + 1e-3
    err = (dw.float() - ref_w).abs().max().item() / scale
    assert err < 2e-2, f"bwd-weight rel err {err}"
    berr = (db.float() - ref_b).abs().max().item() / (ref_b.abs().max().item() + 1e-3)
    assert berr < 2e-2, f"bwd-bias rel err {berr}"


def test_patched_conv_module_end_to_end():
    """WRN block convs through patch_convs: autograd grads vs fp32 torch."""
    from fast_autoaugment_amd.ops.conv import patch_convs
    torch.manual_seed(0)
    m = torch.nn.Conv2d(32, 64, 3, stride=1, padding=1).to(dev())
    mref = torch.nn.Conv2d(32, 64, 3, stride=1, padding=1).to(dev())
    mref.load_state_dict(m.state_dict())
    m = m.to(torch.bfloat16).to(memory_format=torch.channels_last)
    assert patch_convs(m) == 1
    x = torch.randn(8, 32, 16, 16, device=dev())
    xb = x.bfloat16().contiguous(memory_format=torch.channels_last).requires_grad_(True)
    xr = x.clone().requires_grad_(True)
    y = m(xb)
    yr = mref(xr)
    g = torch.randn_like(yr)
    y.backward(g.bfloat16())
    yr.backward(g)
    for got, ref in [(y.float(), yr), (xb.grad.float(), xr.grad),
                     (m.weight.grad.float(), mref.weight.grad),
                     (m.bias.grad.float(), mref.bias.grad)]:
        scale = ref.abs().max().item() + 1e-3
        assert (got - ref).abs().max().item() / scale < 3e-2


@pytest.mark.parametrize("C,H,k,stride", [
    (32, 32, 3, 1), (96, 56, 3, 2), (144, 28, 5, 1), (240, 28, 5, 2),
    (576, 14, 5, 1), (1152, 7, 5, 1),
])
def test_depthwise_fwd_bwd(C, H, k, stride):
    """Depthwise kernels vs torch grouped conv (TF-SAME asymmetric pads)."""
    import math
    torch.manual_seed(0)
    x = (torch.randn(8, C, H, H, device=dev()) * 0.5)
    w = torch.randn(C, 1, k, k, device=dev()) * 0.2
    b = torch.randn(C, device=dev()) * 0.1
    ph = max((math.ceil(H / stride) - 1) * stride + k - H, 0)
    pl, pr, pt, pb = ph // 2, ph - ph // 2, ph // 2, ph - ph // 2
    xp = torch.nn.functional.pad(x, (pl, pr, pt, pb))
    ref = torch.nn.functional.conv2d(xp, w, b, stride=stride, groups=C)

    xb = x.bfloat16().contiguous(memory_format=torch.channels_last)
    wb = w.bfloat16().contiguous(memory_format=torch.channels_last)
    got = C_ext().dwconv_fwd(xb, wb, b.bfloat16(), stride, pt, pb, pl, pr).float()
    scale = ref.abs().max().item() + 1e-3
    assert (got - ref).abs().max().item() / scale < 2e-2

    # bwd-data + bwd-weight vs torch
    dy = torch.randn_like(ref) * 0.2
    dyb = dy.bfloat16().contiguous(memory_format=torch.channels_last)
    ref_dx = torch.nn.grad.conv2d_input(list(xp.shape), w, dy, stride=stride,
                                        groups=C)[:, :, pt:pt + H, pl:pl + H]
    got_dx = C_ext().dwconv_bwd_data(dyb, wb, stride, pt, pl, H, H).float()
    s2 = ref_dx.abs().max().item() + 1e-3
    assert (got_dx - ref_dx).abs().max().item() / s2 < 2e-2
    ref_dw = torch.nn.grad.conv2d_weight(xp, list(w.shape), dy, stride=stride, groups=C)
    got_dw = C_ext().dwconv_bwd_weight(dyb, xb, stride, pt, pl, k, k).float()
    s3 = ref_dw.abs().max().item() + 1e-3
    assert (got_dw - ref_dw).abs().max().item() / s3 < 2e-2


def C_ext():
    from fast_autoaugment_amd.ops import ext
    return ext()


def test_conv_fwd_splitk_matches():
    """FAA_CONV_SPLITK=1 K-partitioned fwd (fp32-atomic workspace) vs torch
    (validated in round 1, tools/splitk_check.py; default-off pending
    round-2 measurement)."""
    import os
    os.environ["FAA_CONV_SPLITK"] = "1"
    try:
        torch.manual_seed(0)
        for B, Cin, H, Cout, k, s in [(128, 64, 8, 64, 3, 1), (8, 32, 8, 64, 3, 2)]:
            x = torch.randn(B, Cin, H, H, device=dev()) * 0.5
            w = torch.randn(Cout, Cin, k, k, device=dev()) * 0.05
            b = torch.randn(Cout, device=dev()) * 0.1
            ref = torch.nn.functional.conv2d(x, w, b, stride=s, padding=k // 2)
            got = C_ext().conv2d_fwd(
                x.bfloat16().contiguous(memory_format=torch.channels_last),
                w.bfloat16().contiguous(memory_format=torch.channels_last),
                b.bfloat16(), s, k // 2).float()
            err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
            assert err < 2e-2, f"splitk rel err {err}"
    finally:
        os.environ.pop("FAA_CONV_SPLITK", None)


@pytest.mark.parametrize("M,Ch", [(128 * 32 * 32, 16), (128 * 8 * 8, 128),
                                  (128 * 16 * 16, 160), (64 * 8 * 8, 320),
                                  (128 * 8 * 8, 640), (1000, 8)])
def test_colsum_v2_matches_reference(M, Ch):
    """Replay-safe colsum v2 (dbias) vs fp32 torch sum, incl. C > blockDim
    shapes for the big-config coverage (C up to 640)."""
    torch.manual_seed(3)
    dy = (torch.randn(M, Ch, device=dev()) * 0.5).bfloat16()
    # view as NHWC channels-last image so the C++ side sees a 4D tensor
    dy4 = dy.view(M, 1, 1, Ch).permute(0, 3, 1, 2).contiguous(
        memory_format=torch.channels_last)
    ref = dy.float().sum(dim=0)
    got = C.colsum_bf16(dy4).float()
    err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    assert err < 1e-2, f"colsum rel err {err}"


@pytest.mark.parametrize("d8", ["ib2", "ib4", "ib2sk2", "ib4sk2", "ib4sk4"])
def test_conv_fwd_direct_d8_variants(d8):
    """W=8 big-C direct-kernel variants (IB4 / cin-split-K) vs fp32 torch."""
    import os
    torch.manual_seed(7)
    B, Cin, H, Cout = 8, 256, 8, 256
    x = torch.randn(B, Cin, H, H, device=dev()) * 0.5
    w = torch.randn(Cout, Cin, 3, 3, device=dev()) * 0.02
    b = torch.randn(Cout, device=dev()) * 0.1
    ref = torch.nn.functional.conv2d(x, w, b, stride=1, padding=1)
    os.environ["FAA_CONV_DIRECT"] = "big"
    os.environ["FAA_CONV_D8"] = d8
    try:
        got = C.conv2d_fwd(
            x.bfloat16().contiguous(memory_format=torch.channels_last),
            w.bfloat16().contiguous(memory_format=torch.channels_last),
            b.bfloat16(), 1, 1).float()
    finally:
        os.environ.pop("FAA_CONV_D8", None)
        os.environ.pop("FAA_CONV_DIRECT", None)
    err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    assert err < 2e-2, f"d8={d8} rel err {err}"


@pytest.mark.parametrize("B,Cin,H,Cout", [(8, 32, 32, 64), (8, 64, 16, 128),
                                          (4, 160, 32, 320)])
def test_conv_bwd_data_stride2_phase(B, Cin, H, Cout):
    """Phase-decomposition stride-2 3x3 bwd-data vs fp32 torch."""
    torch.manual_seed(11)
    x = torch.randn(B, Cin, H, H, device=dev()) * 0.5
    w = torch.randn(Cout, Cin, 3, 3, device=dev()) * 0.05
    dy = torch.randn(B, Cout, H // 2, H // 2, device=dev()) * 0.5
    ref = torch.nn.grad.conv2d_input(list(x.shape), w, dy, stride=2, padding=1)
    dyb = dy.bfloat16().contiguous(memory_format=torch.channels_last)
    wb = w.bfloat16().contiguous(memory_format=torch.channels_last)
    got = C.conv2d_bwd_data_s2(dyb, wb, H, H, False).float()
    err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    assert err < 2e-2, f"s2 bwd-data rel err {err}"


def test_batched_flip_matches_per_conv():
    """conv_flip_all (one launch) == per-conv weight_flip semantics, via the
    cached bwd-data path producing identical dx."""
    import os
    from fast_autoaugment_amd.ops.conv import (conv_flip_all, patch_convs,
                                               _flip_cache)
    torch.manual_seed(3)
    m = torch.nn.Conv2d(32, 64, 3, padding=1).to(dev()).to(torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    assert patch_convs(m) == 1
    conv_flip_all()
    dy = torch.randn(8, 64, 16, 16, device=dev()).bfloat16().contiguous(
        memory_format=torch.channels_last)
    ref = C.conv2d_bwd_data(dy, m.weight, 1, 1, 16, 16).float()
    w2 = _flip_cache[m.weight.data_ptr()]
    got = C.conv2d_fwd(dy, w2, torch.Tensor(), 1, 1).float()
    err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    assert err < 1e-3, f"batched flip mismatch {err}"


@pytest.mark.parametrize("wsplit", ["0", "1"])
def test_wrw3_small_c_variants(wsplit):
    """wrw v3 at small C: quadrant config vs wave-split + LDS cross-wave
    reduce, both vs fp32 torch."""
    import os
    torch.manual_seed(13)
    for (B, Cin, H, Cout, k) in [(8, 16, 32, 32, 3), (8, 32, 16, 32, 3),
                                 (8, 32, 32, 32, 1)]:
        x = torch.randn(B, Cin, H, H, device=dev()) * 0.5
        w_shape = [Cout, Cin, k, k]
        dy = torch.randn(B, Cout, H, H, device=dev()) * 0.1
        ref = torch.nn.grad.conv2d_weight(x, w_shape, dy, stride=1, padding=k // 2)
        os.environ["FAA_WRW_V3"] = "1"
        os.environ["FAA_WRW3_WSPLIT"] = wsplit
        try:
            dw, _ = C.conv2d_bwd_weight(
                dy.bfloat16().contiguous(memory_format=torch.channels_last),
                x.bfloat16().contiguous(memory_format=torch.channels_last),
                1, k // 2, k, k, False)
        finally:
            os.environ.pop("FAA_WRW_V3", None)
            os.environ.pop("FAA_WRW3_WSPLIT", None)
        err = (dw.float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
        assert err < 2e-2, f"wsplit={wsplit} {Cin}x{H}->{Cout} k{k} rel err {err}"


@pytest.mark.parametrize("B,Cm,H,G", [(8, 384, 16, 4), (8, 128, 32, 4),
                                      (4, 256, 8, 4)])
def test_conv_fwd_grouped(B, Cm, H, G):
    """Grouped 3x3 s1 (cardinality-4 ShakeResNeXt branches) vs fp32 torch."""
    torch.manual_seed(17)
    x = torch.randn(B, Cm, H, H, device=dev()) * 0.5
    w = torch.randn(Cm, Cm // G, 3, 3, device=dev()) * 0.05
    b = torch.randn(Cm, device=dev()) * 0.1
    ref = torch.nn.functional.conv2d(x, w, b, stride=1, padding=1, groups=G)
    got = C.conv2d_fwd_grouped(
        x.bfloat16().contiguous(memory_format=torch.channels_last),
        w.bfloat16().contiguous(memory_format=torch.channels_last),
        b.bfloat16(), 1, 1, G).float()
    err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    assert err < 2e-2, f"grouped rel err {err}"


def test_patched_grouped_conv_end_to_end():
    from fast_autoaugment_amd.ops.conv import patch_convs
    torch.manual_seed(2)
    m = torch.nn.Conv2d(128, 128, 3, padding=1, groups=4).to(dev())
    mref = torch.nn.Conv2d(128, 128, 3, padding=1, groups=4).to(dev())
    mref.load_state_dict(m.state_dict())
    m = m.to(torch.bfloat16).to(memory_format=torch.channels_last)
    assert patch_convs(m) == 1
    x = torch.randn(8, 128, 16, 16, device=dev())
    xb = x.bfloat16().contiguous(memory_format=torch.channels_last).requires_grad_(True)
    xr = x.clone().requires_grad_(True)
    y = m(xb)
    yr = mref(xr)
    g = torch.randn_like(yr)
    y.backward(g.bfloat16())
    yr.backward(g)
    for got, ref in [(y.float(), yr), (xb.grad.float(), xr.grad),
                     (m.weight.grad.float(), mref.weight.grad)]:
        scale = ref.abs().max().item() + 1e-3
        assert (got - ref).abs().max().item() / scale < 3e-2
