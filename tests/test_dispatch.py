"""CPU tests for the measured kernel-dispatch tables (ops/conv.py).

These rules were derived from per-shape microbenchmarks
(profiles/*, gpurun_out/call*.log); the tests pin the TABLE so a refactor
cannot silently change which implementation a shape runs on.
"""
import torch

from fast_autoaugment_amd.ops import conv as C


def test_wrw_dispatch_table():
    # stem / deep-8px / >=160@32px in-house; everything else fallback
    assert C._faa_wrw_wins(3, 32, 16)            # stem
    assert C._faa_wrw_wins(128, 8, 128)          # stage-3 WRN
    assert C._faa_wrw_wins(160, 32, 160)         # WRN-28-10 32px
    assert C._faa_wrw_wins(320, 32, 320)
    assert not C._faa_wrw_wins(16, 32, 32)       # MIOpen wins (measured)
    assert not C._faa_wrw_wins(32, 32, 32)
    assert not C._faa_wrw_wins(64, 16, 64)
    assert not C._faa_wrw_wins(128, 16, 128)     # 128ch only at 8px
    assert not C._faa_wrw_wins(320, 16, 320)
    assert not C._faa_wrw_wins(640, 8, 640)


def test_wrw_dispatch_env_forcing(monkeypatch):
    monkeypatch.setenv("FAA_WRW", "faa")
    assert C._faa_wrw_wins(64, 16, 64)
    monkeypatch.setenv("FAA_WRW", "torch")
    assert not C._faa_wrw_wins(3, 32, 16)


def _conv(cin, cout, k, s, groups=1):
    return torch.nn.Conv2d(cin, cout, k, stride=s, padding=k // 2,
                           groups=groups)


def test_fwd_patch_eligibility():
    # measured small/mid shapes always eligible
    assert C._eligible(_conv(16, 32, 3, 1))
    assert C._eligible(_conv(64, 128, 3, 2))
    assert C._eligible(_conv(16, 32, 1, 1))
    # big channels only through the direct 3x3 s1 path
    assert C._eligible(_conv(320, 320, 3, 1))
    assert not C._eligible(_conv(320, 320, 1, 1))    # 1x1 big: unmeasured
    assert not C._eligible(_conv(320, 640, 3, 2))    # s2 big: im2col loses
    # non-conv geometries stay out
    assert not C._eligible(torch.nn.Conv2d(32, 32, 3, padding=2))
    assert not C._eligible(torch.nn.Conv2d(32, 32, 3, padding=1, dilation=2))


def test_runtime_fwd_dispatch_rules():
    m160 = _conv(160, 160, 3, 1)
    m640 = _conv(640, 640, 3, 1)
    x32 = torch.zeros(2, 160, 32, 32)
    x8_640 = torch.zeros(2, 640, 8, 8)
    assert C._runtime_faa_ok(m160, x32)
    assert not C._runtime_faa_ok(m640, x8_640)       # MIOpen keeps 640@8px
    # non-CIFAR spatial (9x9) at big channels: no direct geometry -> torch
    assert not C._runtime_faa_ok(_conv(320, 320, 3, 1), torch.zeros(2, 320, 9, 9))


def test_grouped_eligibility():
    # ShakeResNeXt cardinality-4 branches
    assert C._grouped_eligible(_conv(384, 384, 3, 1, groups=4))
    assert C._grouped_eligible(_conv(128, 128, 3, 1, groups=4))
    # depthwise is NOT the grouped path
    assert not C._grouped_eligible(_conv(64, 64, 3, 1, groups=64))
    # per-group cout must tile by 32
    assert not C._grouped_eligible(_conv(96, 96, 3, 1, groups=4))
    # stride-2 grouped stays on torch
    assert not C._grouped_eligible(_conv(384, 384, 3, 2, groups=4))


def test_dbias_cpu_fallback():
    dy = torch.randn(4, 12, 8, 8)   # C % 8 != 0 -> always torch sum
    ref = dy.sum(dim=(0, 2, 3))
    assert torch.allclose(C._dbias(dy), ref)


def test_runtime_fwd_dispatch_imagenet_spatials():
    # measured (gpurun_out/call35.log): direct wins 56^2 and 14^2; MIOpen
    # keeps 28^2 and 7^2
    assert C._runtime_faa_ok(_conv(64, 64, 3, 1), torch.zeros(2, 64, 56, 56))
    assert C._runtime_faa_ok(_conv(256, 256, 3, 1), torch.zeros(2, 256, 14, 14))
    assert not C._runtime_faa_ok(_conv(128, 128, 3, 1), torch.zeros(2, 128, 28, 28))
    assert not C._runtime_faa_ok(_conv(512, 512, 3, 1), torch.zeros(2, 512, 7, 7))
    # 1x1s keep the small-channel blanket rule
    assert C._runtime_faa_ok(_conv(64, 64, 1, 1), torch.zeros(2, 64, 28, 28))
