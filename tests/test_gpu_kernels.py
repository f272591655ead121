"""GPU numerics tests: every HIP kernel vs its CPU/plain-torch fp32 reference."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from fast_autoaugment_amd.ops import ext
    C = ext()   # fail loudly if the extension is missing on a GPU box


def dev():
    return torch.device("cuda:0")


# ------------------------------------------------------------ aug pipeline

def _run_pipeline_both(prog_mod, seed=0, B=8, H=32, W=32, cutout=16, pad=4):
    from fast_autoaugment_amd.aug import cpu_exec, ops as aug_ops
    rng = np.random.default_rng(seed)
    imgs = rng.integers(0, 256, size=(64, H, W, 3), dtype=np.uint8)
    sel = rng.integers(0, 64, size=B)
    prog = np.zeros((B, aug_ops.PROG_SLOTS, aug_ops.PROG_WIDTH), np.float32)
    prog_mod(prog, rng)
    post = aug_ops.compile_post(B, W, H, rng, pad=pad, cutout_len=cutout, train=True)
    mean = np.array([0.4914, 0.4822, 0.4465], np.float32)
    std = np.array([0.2023, 0.1994, 0.2010], np.float32)

    ref = cpu_exec.run_pipeline_cpu(imgs[sel], prog, post, mean, std)  # [B,H,W,3] f32

    timgs = torch.from_numpy(imgs).to(dev())
    out = C.aug_pipeline(timgs, torch.from_numpy(sel).to(dev()),
                         torch.from_numpy(prog).to(dev()),
                         torch.from_numpy(post).to(dev()),
                         torch.from_numpy(mean).to(dev()),
                         torch.from_numpy(std).to(dev()), False)
    # out is [B,3,H,W] channels_last -> compare as NHWC
    got = out.permute(0, 2, 3, 1).contiguous().cpu().numpy()
    return got, ref


def test_aug_pipeline_passthrough():
    got, ref = _run_pipeline_both(lambda prog, rng: None, cutout=0, pad=0)
    np.testing.assert_allclose(got, ref, atol=0, rtol=0)


def test_aug_pipeline_post_only():
    got, ref = _run_pipeline_both(lambda prog, rng: None)
    np.testing.assert_allclose(got, ref, atol=0, rtol=0)


@pytest.mark.parametrize("code,params", [
    (2, [0] * 6),                                  # autocontrast
    (3, [0] * 6),                                  # invert
    (4, [0] * 6),                                  # equalize
    (5, [0] * 6),                                  # flip
    (6, [128.7, 0, 0, 0, 0, 0]),                   # solarize
    (7, [5, 0, 0, 0, 0, 0]),                       # posterize
    (8, [1.37, 0, 0, 0, 0, 0]),                    # contrast
    (9, [0.42, 0, 0, 0, 0, 0]),                    # color
    (10, [1.9, 0, 0, 0, 0, 0]),                    # brightness
    (11, [0.3, 0, 0, 0, 0, 0]),                    # sharpness
    (12, [5, 7, 20, 22, 0, 0]),                    # cutout fill
    (13, [0.35, 3, 0, 0, 0, 0]),                   # sample pairing (batch slot 3)
])
def test_aug_pipeline_single_op(code, params):
    def setp(prog, rng):
        prog[:, 0, 0] = code
        prog[:, 0, 1:1 + len(params)] = params
    got, ref = _run_pipeline_both(setp, cutout=0, pad=0)
    np.testing.assert_allclose(got, ref, atol=0, rtol=0)


@pytest.mark.parametrize("m", [
    (1.0, 0.2, 0.0, 0.0, 1.0, 0.0),                          # shear x
    (1.0, 0.0, 3.7, 0.0, 1.0, -2.2),                          # translate
])
def test_aug_pipeline_affine(m):
    def setp(prog, rng):
        prog[:, 0, 0] = 1
        prog[:, 0, 1:7] = m
    got, ref = _run_pipeline_both(setp, cutout=0, pad=0)
    np.testing.assert_allclose(got, ref, atol=0, rtol=0)


def test_aug_pipeline_rotate_matrix():
    from fast_autoaugment_amd.aug.ops import _rotate_matrix
    m = np.array(_rotate_matrix(32, 32, -17.3), np.float32)

    def setp(prog, rng):
        prog[:, 0, 0] = 1
        prog[:, 0, 1:7] = m
    got, ref = _run_pipeline_both(setp, cutout=0, pad=0)
    np.testing.assert_allclose(got, ref, atol=0, rtol=0)


def test_aug_pipeline_random_policy_programs():
    """Random archive-driven programs, chained ops, full post stage."""
    from fast_autoaugment_amd import policies
    from fast_autoaugment_amd.aug import ops as aug_ops
    pol = policies.get_archive("fa_reduced_cifar10")

    def setp(prog, rng):
        p2 = aug_ops.compile_program(pol, prog.shape[0], 32, 32, rng)
        prog[:] = p2
    got, ref = _run_pipeline_both(setp, B=32)
    np.testing.assert_allclose(got, ref, atol=0, rtol=0)


def test_aug_pipeline_large_image_path():
    """>64x64 images take the global-workspace path."""
    got, ref = _run_pipeline_both(lambda prog, rng: None, B=4, H=96, W=96,
                                  cutout=0, pad=0)
    np.testing.assert_allclose(got, ref, atol=0, rtol=0)


def test_aug_pipeline_bf16_output():
    from fast_autoaugment_amd.aug import ops as aug_ops
    rng = np.random.default_rng(3)
    imgs = rng.integers(0, 256, size=(16, 32, 32, 3), dtype=np.uint8)
    sel = np.arange(8)
    prog = np.zeros((8, aug_ops.PROG_SLOTS, aug_ops.PROG_WIDTH), np.float32)
    post = np.zeros((8, 6), np.float32)
    mean = np.zeros(3, np.float32)
    std = np.ones(3, np.float32)
    t = lambda a: torch.from_numpy(a).to(dev())
    out = C.aug_pipeline(t(imgs), t(sel), t(prog), t(post), t(mean), t(std), True)
    assert out.dtype == torch.bfloat16
    ref = torch.from_numpy(imgs[sel].astype(np.float32) / 255.0).to(dev())
    got = out.permute(0, 2, 3, 1).float()
    assert (got - ref).abs().max().item() < 4e-3   # bf16 resolution


# ------------------------------------------------------------- elementwise

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_scale_bcast(dtype):
    x = torch.randn(8, 16, 4, 4, device=dev(), dtype=dtype)
    s = torch.rand(8, device=dev())
    out = C.scale_bcast(x, s)
    # kernel computes in fp32 and rounds once: compare against fp32 math
    ref = (x.float() * s.view(-1, 1, 1, 1)).to(dtype)
    assert (out.float() - ref.float()).abs().max().item() == 0


def test_scale_lerp():
    x1 = torch.randn(8, 16, 4, 4, device=dev())
    x2 = torch.randn(8, 16, 4, 4, device=dev())
    a = torch.rand(8, device=dev())
    out = C.scale_lerp(x1, x2, a)
    av = a.view(-1, 1, 1, 1)
    ref = av * x1 + (1 - av) * x2
    assert (out - ref).abs().max().item() < 1e-6


def test_swish_fwd_bwd():
    x = torch.randn(4, 64, 8, 8, device=dev())
    out = C.swish_fwd(x)
    ref = x * torch.sigmoid(x)
    assert (out - ref).abs().max().item() < 1e-5
    g = torch.randn_like(x)
    gout = C.swish_bwd(g, x)
    s = torch.sigmoid(x)
    gref = g * (s * (1 + x * (1 - s)))
    assert (gout - gref).abs().max().item() < 1e-5


def test_mixup_kernel():
    x = torch.randn(8, 3, 8, 8, device=dev())
    perm = torch.randperm(8, device=dev())
    out = C.mixup_fwd(x, perm, 0.7)
    ref = 0.7 * x + 0.3 * x[perm]
    assert (out - ref).abs().max().item() < 1e-6


def test_pad_add():
    x = torch.randn(4, 20, 8, 8, device=dev()).contiguous(memory_format=torch.channels_last)
    sc = torch.randn(4, 12, 8, 8, device=dev()).contiguous(memory_format=torch.channels_last)
    out = C.pad_add(x, sc)
    ref = x + torch.nn.functional.pad(sc, (0, 0, 0, 0, 0, 8))
    assert (out - ref).abs().max().item() < 1e-6


# ------------------------------------------------------------------- loss

def test_label_smooth_ce_fwd_bwd():
    from fast_autoaugment_amd.metrics import CrossEntropyLabelSmooth
    torch.manual_seed(0)
    for eps in [0.0, 0.1]:
        logits = torch.randn(64, 100, device=dev(), requires_grad=True)
        target = torch.randint(0, 100, (64,), device=dev())
        crit = CrossEntropyLabelSmooth(100, eps)
        loss = crit(logits, target)
        # CPU reference
        lcpu = logits.detach().cpu().requires_grad_(True)
        ref = CrossEntropyLabelSmooth(100, eps)(lcpu, target.cpu())
        assert abs(loss.item() - ref.item()) < 1e-4
        loss.backward()
        ref.backward()
        assert (logits.grad.cpu() - lcpu.grad).abs().max().item() < 1e-5


# ------------------------------------------------------------------- step

def test_sgd_fused_step_matches_cpu():
    from fast_autoaugment_amd.parallel.flat import flatten_module
    from fast_autoaugment_amd.optim import FusedSGD
    torch.manual_seed(0)
    m_gpu = torch.nn.Sequential(torch.nn.Conv2d(3, 8, 3, padding=1),
                                torch.nn.BatchNorm2d(8),
                                torch.nn.Conv2d(8, 8, 1)).to(dev())
    m_cpu = torch.nn.Sequential(torch.nn.Conv2d(3, 8, 3, padding=1),
                                torch.nn.BatchNorm2d(8),
                                torch.nn.Conv2d(8, 8, 1))
    m_cpu.load_state_dict(m_gpu.state_dict())
    fg = flatten_module(m_gpu)
    fc = flatten_module(m_cpu)
    og = FusedSGD(fg, lr=0.1, momentum=0.9, nesterov=True, weight_decay=0.01, grad_clip=5.0)
    oc = FusedSGD(fc, lr=0.1, momentum=0.9, nesterov=True, weight_decay=0.01, grad_clip=5.0)
    x = torch.randn(4, 3, 8, 8)
    for i in range(3):
        for m, o, xx in [(m_gpu, og, x.to(dev())), (m_cpu, oc, x)]:
            o.zero_grad()
            m(xx).square().mean().backward()
            o.step()
    assert (fg.flat_param.cpu() - fc.flat_param).abs().max().item() < 1e-5


def test_sgd_fused_clip_active():
    """With a big gradient the clip must engage identically to torch's."""
    from fast_autoaugment_amd.parallel.flat import FlatParams
    from fast_autoaugment_amd.optim import FusedSGD
    p = torch.randn(1000, device=dev()) * 10
    g = torch.randn(1000, device=dev()) * 100
    flat = FlatParams(p.clone(), g.clone(), 1000, [])
    opt = FusedSGD(flat, lr=0.1, momentum=0.9, nesterov=True,
                   weight_decay=0.0, grad_clip=5.0)
    opt.step()
    # torch reference
    gref = g.clone()
    total = gref.norm(2)
    gref.mul_(min(1.0, 5.0 / (total.item() + 1e-6)))
    buf = gref.clone()
    upd = gref + 0.9 * buf
    pref = p - 0.1 * upd
    assert (flat.flat_param - pref).abs().max().item() < 1e-5


def test_ema_lerp():
    s = torch.randn(1000, device=dev())
    x = torch.randn(1000, device=dev())
    ref = (1 - 0.99) * x + 0.99 * s
    C.ema_lerp_(s, x, 0.99)
    assert (s - ref).abs().max().item() < 1e-6


# ---------------------------------------------------------------- bn_relu

@pytest.mark.parametrize("dtype,tol", [(torch.float32, 2e-5), (torch.bfloat16, 3e-2)])
@pytest.mark.parametrize("Ch", [
    32,    # vec path, C < blockDim
    512,   # vec path, C > blockDim (multi-pass LDS fold)
    640,   # WRN-28-10 widest stage
    61,    # any-C path, odd channels (PyramidNet rounded widths)
    340,   # any-C path, C % 8 == 4, C > blockDim
])
def test_bn_relu_fwd_bwd_vs_torch(Ch, dtype, tol):
    torch.manual_seed(0)
    N, H, W = 8, 16, 16
    x = torch.randn(N, Ch, H, W, device=dev(), dtype=dtype).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)
    bn = torch.nn.BatchNorm2d(Ch, momentum=0.9).to(dev())
    bn.weight.data.uniform_(0.5, 1.5)
    bn.bias.data.uniform_(-0.5, 0.5)

    # reference: plain fp32 BN + relu
    xref = x.detach().float().clone().requires_grad_(True)
    bnref = torch.nn.BatchNorm2d(Ch, momentum=0.9).to(dev())
    bnref.load_state_dict(bn.state_dict())
    ref = torch.relu(bnref(xref))

    from fast_autoaugment_amd.ops.bnrelu import fused_bn_relu
    bn.train()
    out = fused_bn_relu(x, bn)
    assert (out.float() - ref).abs().max().item() < tol
    # running stats must match
    assert (bn.running_mean - bnref.running_mean).abs().max().item() < tol
    assert (bn.running_var - bnref.running_var).abs().max().item() < tol

    g = torch.randn_like(ref)
    ref.backward(g)
    out.backward(g.to(dtype))
    assert (x.grad.float() - xref.grad).abs().max().item() < tol * 4
    assert (bn.weight.grad - bnref.weight.grad).abs().max().item() < tol * 10
    assert (bn.bias.grad - bnref.bias.grad).abs().max().item() < tol * 10


def test_bn_relu_eval_mode():
    N, Ch = 4, 16
    x = torch.randn(N, Ch, 8, 8, device=dev()).contiguous(memory_format=torch.channels_last)
    bn = torch.nn.BatchNorm2d(Ch).to(dev())
    bn.running_mean.uniform_(-1, 1)
    bn.running_var.uniform_(0.5, 2)
    bn.eval()
    from fast_autoaugment_amd.ops.bnrelu import fused_bn_relu
    out = fused_bn_relu(x, bn)
    ref = torch.relu(bn(x))
    assert (out - ref).abs().max().item() < 1e-5


# ----------------------------------------------------------- model smoke

def test_wrn_forward_backward_gpu():
    from fast_autoaugment_amd.models import build_model
    m = build_model({"type": "wresnet40_2"}, 10).to(dev()).to(
        memory_format=torch.channels_last)
    x = torch.randn(8, 3, 32, 32, device=dev()).contiguous(memory_format=torch.channels_last)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = m(x)
        loss = y.float().square().mean()
    loss.backward()
    assert torch.isfinite(loss)
    assert all(torch.isfinite(p.grad).all() for p in m.parameters() if p.grad is not None)


def test_sgd_fused_step_mixed_matches_cpu():
    """bf16 working weights + fp32 master: GPU kernel vs CPU reference path."""
    from fast_autoaugment_amd.parallel.flat import flatten_module
    from fast_autoaugment_amd.optim import FusedSGD
    torch.manual_seed(0)
    # conv-only model: torch CPU BatchNorm cannot mix bf16 activations with
    # fp32 buffers (the GPU path uses the fused HIP BN instead)
    mk = lambda: torch.nn.Sequential(torch.nn.Conv2d(3, 8, 3, padding=1),
                                     torch.nn.Conv2d(8, 8, 1))
    m_gpu, m_cpu = mk().to(dev()), mk()
    m_cpu.load_state_dict(m_gpu.state_dict())
    fg = flatten_module(m_gpu, work_dtype=torch.bfloat16)
    fc = flatten_module(m_cpu, work_dtype=torch.bfloat16)
    og = FusedSGD(fg, lr=0.1, momentum=0.9, nesterov=True, weight_decay=0.01, grad_clip=5.0)
    oc = FusedSGD(fc, lr=0.1, momentum=0.9, nesterov=True, weight_decay=0.01, grad_clip=5.0)
    x = torch.randn(4, 3, 8, 8)
    for i in range(3):
        for m, o, xx in [(m_gpu, og, x.to(dev()).bfloat16()), (m_cpu, oc, x.bfloat16())]:
            o.zero_grad()
            m(xx).float().square().mean().backward()
            o.step()
    # bf16 grads differ slightly between CPU/GPU conv kernels; masters track
    assert (fg.flat_master.cpu() - fc.flat_master).abs().max().item() < 5e-3
    # working copy is the bf16 quantization of the master
    assert (fg.flat_param.float().cpu() -
            fg.flat_master.cpu().bfloat16().float()).abs().max().item() == 0


def test_aug_pipeline_imagenet():
    """EffNet crop-resize + jitter + lighting pipeline vs the CPU reference.
    Small sizes keep the pure-python bicubic reference fast."""
    from fast_autoaugment_amd.aug import cpu_exec, ops as aug_ops
    from fast_autoaugment_amd.aug.imagenet import compile_post_imagenet
    rng = np.random.default_rng(11)
    H = W = 48
    OS = 32
    B = 6
    imgs = rng.integers(0, 256, size=(16, H, W, 3), dtype=np.uint8)
    sel = rng.integers(0, 16, size=B)
    prog = np.zeros((B, aug_ops.PROG_SLOTS, aug_ops.PROG_WIDTH), np.float32)
    prog[:, 0, 0] = 3  # invert as a phase-A op
    post = compile_post_imagenet(B, W, H, rng, OS, train=True)
    mean = np.array([0.485, 0.456, 0.406], np.float32)
    std = np.array([0.229, 0.224, 0.225], np.float32)

    ref = cpu_exec.run_pipeline_imagenet_cpu(imgs[sel], prog, post, mean, std, OS, OS)

    t = lambda a: torch.from_numpy(np.ascontiguousarray(a)).to(dev())
    out = C.aug_pipeline_imagenet(t(imgs), t(sel), t(prog), t(post), t(mean), t(std),
                                  OS, OS, False)
    got = out.permute(0, 2, 3, 1).contiguous().cpu().numpy()
    # resize accumulation order differs CPU vs GPU: allow 1/255 per pixel
    err = np.abs(got - ref) * std.reshape(1, 1, 1, 3) * 255.0
    assert err.max() < 1.5, f"imagenet pipeline max err {err.max()}"


def test_aug_pipeline_imagenet_eval_centercrop():
    from fast_autoaugment_amd.aug import cpu_exec, ops as aug_ops
    from fast_autoaugment_amd.aug.imagenet import compile_post_imagenet
    rng = np.random.default_rng(12)
    H = W = 64
    OS = 40
    B = 4
    imgs = rng.integers(0, 256, size=(8, H, W, 3), dtype=np.uint8)
    sel = np.arange(B)
    prog = np.zeros((B, aug_ops.PROG_SLOTS, aug_ops.PROG_WIDTH), np.float32)
    post = compile_post_imagenet(B, W, H, rng, OS, train=False)
    mean = np.zeros(3, np.float32)
    std = np.ones(3, np.float32)
    ref = cpu_exec.run_pipeline_imagenet_cpu(imgs[sel], prog, post, mean, std, OS, OS)
    t = lambda a: torch.from_numpy(np.ascontiguousarray(a)).to(dev())
    out = C.aug_pipeline_imagenet(t(imgs), t(sel), t(prog), t(post), t(mean), t(std),
                                  OS, OS, False)
    got = out.permute(0, 2, 3, 1).contiguous().cpu().numpy()
    assert np.abs(got - ref).max() * 255 < 1.5


def test_fused_rmsprop_gpu_matches_cpu():
    from fast_autoaugment_amd.optim import FusedRMSpropTF
    from fast_autoaugment_amd.parallel.flat import flatten_module
    torch.manual_seed(0)
    mk = lambda: torch.nn.Sequential(torch.nn.Conv2d(3, 8, 3, padding=1),
                                     torch.nn.Conv2d(8, 8, 1))
    mg, mc = mk().to(dev()), mk()
    mc.load_state_dict(mg.state_dict())
    fg = flatten_module(mg, work_dtype=torch.bfloat16)
    fc = flatten_module(mc, work_dtype=torch.bfloat16)
    og = FusedRMSpropTF(fg, lr=0.01, weight_decay=1e-5, grad_clip=2.0)
    oc = FusedRMSpropTF(fc, lr=0.01, weight_decay=1e-5, grad_clip=2.0)
    x = torch.randn(4, 3, 8, 8)
    for _ in range(3):
        for m, o, xx in [(mg, og, x.to(dev()).bfloat16()), (mc, oc, x.bfloat16())]:
            o.zero_grad()
            m(xx).float().square().mean().backward()
            o.step()
    assert (fg.flat_master.cpu() - fc.flat_master).abs().max().item() < 5e-3


def test_pad_add_fwd_bwd_vs_torch():
    """Fused channel-pad residual add (PyramidNet) vs F.pad reference."""
    import torch.nn.functional as F
    from fast_autoaugment_amd.ops.functional import pad_add
    torch.manual_seed(0)
    for dtype, tol in [(torch.float32, 1e-6), (torch.bfloat16, 1e-2)]:
        out = torch.randn(4, 37, 8, 8, device=dev(), dtype=dtype).contiguous(
            memory_format=torch.channels_last).requires_grad_(True)
        sc = torch.randn(4, 21, 8, 8, device=dev(), dtype=dtype).contiguous(
            memory_format=torch.channels_last).requires_grad_(True)
        oref = out.detach().clone().requires_grad_(True)
        sref = sc.detach().clone().requires_grad_(True)
        y = pad_add(out, sc)
        yr = oref + F.pad(sref, (0, 0, 0, 0, 0, 37 - 21))
        assert (y - yr).abs().max().item() < tol
        g = torch.randn_like(yr)
        y.backward(g)
        yr.backward(g)
        assert (out.grad - oref.grad).abs().max().item() < tol
        assert (sc.grad - sref.grad).abs().max().item() < tol


@pytest.mark.parametrize("Ch", [32, 61])
def test_bn_swish_fwd_bwd_vs_torch(Ch):
    """Fused BN+swish (EfficientNet pattern) vs fp32 torch BN + x*sigmoid."""
    torch.manual_seed(1)
    dtype, tol = torch.bfloat16, 3e-2
    x = torch.randn(8, Ch, 14, 14, device=dev(), dtype=dtype).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)
    bn = torch.nn.BatchNorm2d(Ch, momentum=0.1, eps=1e-3).to(dev())
    bn.weight.data.uniform_(0.5, 1.5)
    bn.bias.data.uniform_(-0.5, 0.5)
    xref = x.detach().float().clone().requires_grad_(True)
    bnref = torch.nn.BatchNorm2d(Ch, momentum=0.1, eps=1e-3).to(dev())
    bnref.load_state_dict(bn.state_dict())
    z = bnref(xref)
    ref = z * torch.sigmoid(z)

    from fast_autoaugment_amd.ops.bnrelu import fused_bn_swish
    bn.train()
    out = fused_bn_swish(x, bn)
    assert (out.float() - ref).abs().max().item() < tol
    g = torch.randn_like(ref)
    ref.backward(g)
    out.backward(g.to(dtype))
    assert (x.grad.float() - xref.grad).abs().max().item() < tol * 4
    assert (bn.weight.grad.float() - bnref.weight.grad).abs().max().item() < tol * 10
    assert (bn.bias.grad.float() - bnref.bias.grad).abs().max().item() < tol * 10
    # eval mode
    bn.eval(); bnref.eval()
    xe = torch.randn(4, Ch, 7, 7, device=dev(), dtype=dtype).contiguous(
        memory_format=torch.channels_last)
    ze = bnref(xe.float())
    oute = fused_bn_swish(xe, bn)
    assert (oute.float() - ze * torch.sigmoid(ze)).abs().max().item() < tol



def test_residual_add_bn_stats_fusion():
    """Fused residual-add + BN fwd-reduce handoff == separate add + BN."""
    import os
    torch.manual_seed(21)
    dev_ = dev()
    a = (torch.randn(16, 64, 16, 16, device=dev_) * 0.5).bfloat16().contiguous(
        memory_format=torch.channels_last)
    b = (torch.randn(16, 64, 16, 16, device=dev_) * 0.5).bfloat16().contiguous(
        memory_format=torch.channels_last)
    out, scratch = C.residual_add_bn_stats(a, b)
    ref_out = a + b
    assert torch.equal(out, ref_out), "fused add output differs from a+b"

    from fast_autoaugment_amd.ops.bnrelu import fused_bn_relu, residual_add
    bn1 = torch.nn.BatchNorm2d(64, momentum=0.3).to(dev_).train()
    bn2 = torch.nn.BatchNorm2d(64, momentum=0.3).to(dev_).train()
    bn2.load_state_dict(bn1.state_dict())
    os.environ["FAA_ADD_BN_FUSE"] = "1"
    try:
        y1 = fused_bn_relu(residual_add(a, b), bn1)
    finally:
        os.environ.pop("FAA_ADD_BN_FUSE", None)
    y2 = fused_bn_relu(ref_out, bn2)
    s = y2.float().abs().max().item() + 1e-3
    assert (y1.float() - y2.float()).abs().max().item() / s < 1e-2
    assert (bn1.running_mean - bn2.running_mean).abs().max().item() < 1e-4
    assert (bn1.running_var - bn2.running_var).abs().max().item() < 1e-4
