"""CPU unit tests: policies codec, schedules, optimizers, EMA, metrics, flat params."""
import math

import torch

from fast_autoaugment_amd import policies
from fast_autoaugment_amd.common import EMA
from fast_autoaugment_amd.lr_scheduler import build_scheduler
from fast_autoaugment_amd.metrics import CrossEntropyLabelSmooth, accuracy
from fast_autoaugment_amd.optim import RMSpropTF
from fast_autoaugment_amd.parallel.flat import flatten_module


def test_policy_archives_load():
    for name in ["fa_reduced_cifar10", "fa_reduced_svhn", "fa_resnet50_rimagenet"]:
        pol = policies.get_archive(name)
        assert len(pol) > 400
        for sub in pol:
            for (op, pr, lvl) in sub:
                assert op in policies.ALL_OPS
                assert 0.0 <= pr <= 1.0


def test_policy_codec_roundtrip():
    pol = policies.get_archive("fa_reduced_cifar10")[:5]
    cfg = policies.policy_encoder(pol)
    back = policies.policy_decoder(cfg, 5, 2)
    assert [[tuple(op) for op in sub] for sub in back] == \
           [[tuple(op) for op in sub] for sub in pol]


def test_remove_duplicates():
    p = [[("Rotate", 0.5, 0.5), ("Color", 0.1, 0.2)],
         [("Rotate", 0.9, 0.1), ("Color", 0.3, 0.4)],
         [("Color", 0.1, 0.2), ("Rotate", 0.5, 0.5)]]
    out = policies.remove_duplicates(p)
    assert len(out) == 2


def _sched_conf(stype, epochs, warm=5):
    return {"epoch": epochs,
            "lr_schedule": {"type": stype, "warmup": {"multiplier": 1, "epoch": warm}}}


def test_cosine_warmup_schedule():
    opt = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=0.1)
    s = build_scheduler(_sched_conf("cosine", 200), opt, 0.1)
    s.step(0.0)
    assert opt.param_groups[0]["lr"] == 0.0          # warmup start
    s.step(5.0)
    assert abs(opt.param_groups[0]["lr"] - 0.1) < 1e-9
    s.step(205.0)
    assert opt.param_groups[0]["lr"] < 1e-4          # cosine tail ~0


def test_resnet_schedule():
    opt = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=0.1)
    s = build_scheduler(_sched_conf("resnet", 270), opt, 0.4)
    s.step(50.0)
    assert abs(opt.param_groups[0]["lr"] - 0.4) < 1e-9
    s.step(100.0)
    assert abs(opt.param_groups[0]["lr"] - 0.04) < 1e-9
    s.step(250.0)
    assert abs(opt.param_groups[0]["lr"] - 0.0004) < 1e-9


def test_rmsprop_tf_semantics():
    """TF semantics: ms init ones, eps inside sqrt (reference rmsprop.py:80-97)."""
    p = torch.nn.Parameter(torch.tensor([1.0]))
    opt = RMSpropTF([p], lr=0.1, alpha=0.9, momentum=0.9, eps=0.001)
    p.grad = torch.tensor([0.5])
    opt.step()
    # ms = 1 + (0.25-1)*0.1 = 0.925 ; mom = 0.1*0.5/sqrt(0.925+0.001)
    expect_mom = 0.1 * 0.5 / math.sqrt(0.925 + 0.001)
    assert abs(p.item() - (1.0 - expect_mom)) < 1e-6


def test_ema_warmup_and_lerp():
    m = torch.nn.Linear(4, 4)
    ema = EMA(0.999)
    ema(m, step=0)       # mu = min(.999, 1/10) -> initial copy
    w0 = ema.shadow["weight"].clone()
    with torch.no_grad():
        m.weight.add_(1.0)
    ema(m, step=1)       # mu = 2/11
    mu = 2.0 / 11
    expect = (1 - mu) * m.weight + mu * w0
    assert torch.allclose(ema.shadow["weight"], expect, atol=1e-6)


def test_ema_fp32_shadow_moves_on_bf16_weights():
    """At mu=0.9999 the per-step increment (1-mu)*delta is below bf16 ULP; the
    shadow must be fp32 so the EMA keeps integrating (ADVICE r1 high,
    reference common.py:44-51 never hit this because its params are fp32)."""
    m = torch.nn.Linear(16, 16).bfloat16()
    ema = EMA(0.9999)
    ema(m)  # no step arg -> mu stays 0.9999 (initial copy)
    assert ema.shadow["weight"].dtype == torch.float32
    w0 = ema.shadow["weight"].clone()
    # 200 tiny updates: each (1-mu)*delta ~ 1e-6, invisible at bf16 ULP (~0.008
    # at magnitude 1) but must accumulate in fp32
    with torch.no_grad():
        m.weight.fill_(1.0)
    for _ in range(200):
        ema(m)
    moved = (ema.shadow["weight"] - w0).abs().max().item()
    # after 200 steps shadow should have moved ~ (1 - mu^200)*(1 - w0) ~ 2% of gap
    assert moved > 1e-3, f"EMA shadow frozen (moved {moved})"
    # integer buffers still copied verbatim
    sd = {"num_batches_tracked": torch.tensor(3)}
    class _M(torch.nn.Module):
        def state_dict(self, *a, **k):
            return sd
    ema2 = EMA(0.9999)
    ema2(_M())
    assert ema2.shadow["num_batches_tracked"].dtype == torch.int64


def test_label_smooth_ce_matches_manual():
    torch.manual_seed(0)
    logits = torch.randn(8, 10)
    target = torch.randint(0, 10, (8,))
    crit = CrossEntropyLabelSmooth(10, 0.1)
    loss = crit(logits, target)
    # manual
    logp = torch.log_softmax(logits, 1)
    t = torch.full_like(logp, 0.1 / 10)
    t.scatter_(1, target[:, None], 1 - 0.1 + 0.1 / 10)
    manual = -(t * logp).sum(1).mean()
    assert torch.allclose(loss, manual, atol=1e-6)
    # epsilon=0 equals plain CE
    crit0 = CrossEntropyLabelSmooth(10, 0.0)
    assert torch.allclose(crit0(logits, target),
                          torch.nn.functional.cross_entropy(logits, target), atol=1e-6)


def test_accuracy_topk():
    logits = torch.tensor([[0.9, 0.1, 0.0], [0.1, 0.8, 0.1], [0.5, 0.4, 0.1]])
    target = torch.tensor([0, 1, 2])
    top1, top2 = accuracy(logits, target, (1, 2))
    assert abs(top1.item() - 2 / 3) < 1e-6
    assert abs(top2.item() - 2 / 3) < 1e-6


def test_flatten_module_views_and_decay_order():
    m = torch.nn.Sequential(torch.nn.Conv2d(3, 8, 3), torch.nn.BatchNorm2d(8),
                            torch.nn.Conv2d(8, 4, 1))
    flat = flatten_module(m)
    # all params are views of the flat buffer
    for p in m.parameters():
        assert p.data_ptr() >= flat.flat_param.data_ptr()
        assert p.grad is not None
    # BN params must sit in the no-decay tail
    bn_w = dict(m.named_parameters())["1.weight"]
    off = (bn_w.data_ptr() - flat.flat_param.data_ptr()) // 4
    assert off >= flat.n_decay
    # training still works through the views
    x = torch.randn(2, 3, 8, 8)
    y = m(x).sum()
    y.backward()
    assert flat.flat_grad.abs().sum() > 0


def test_flat_optimizer_equivalence():
    """A step through flat views must equal a step on a regular clone."""
    torch.manual_seed(0)
    m1 = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.Linear(8, 2))
    m2 = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.Linear(8, 2))
    m2.load_state_dict(m1.state_dict())
    flatten_module(m1)
    o1 = torch.optim.SGD(m1.parameters(), lr=0.1, momentum=0.9, nesterov=True)
    o2 = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=0.9, nesterov=True)
    x = torch.randn(4, 8)
    for _ in range(3):
        for m, o in [(m1, o1), (m2, o2)]:
            o.zero_grad(set_to_none=False)
            m(x).sum().backward()
            o.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6)


def test_fused_rmsprop_tf_cpu_matches_reference():
    """FusedRMSpropTF flat step vs the per-tensor RMSpropTF (same math)."""
    import torch as T
    from fast_autoaugment_amd.optim import FusedRMSpropTF, RMSpropTF
    from fast_autoaugment_amd.parallel.flat import flatten_module
    T.manual_seed(0)
    m1 = T.nn.Sequential(T.nn.Conv2d(3, 8, 3, padding=1), T.nn.Conv2d(8, 4, 1))
    m2 = T.nn.Sequential(T.nn.Conv2d(3, 8, 3, padding=1), T.nn.Conv2d(8, 4, 1))
    m2.load_state_dict(m1.state_dict())
    flat = flatten_module(m1, work_dtype=T.bfloat16)
    o1 = FusedRMSpropTF(flat, lr=0.01, weight_decay=0.0, grad_clip=0.0)
    o2 = RMSpropTF(m2.parameters(), lr=0.01, alpha=0.9, momentum=0.9, eps=1e-3)
    x = T.randn(2, 3, 8, 8)
    for _ in range(3):
        o1.zero_grad()
        m1(x.bfloat16()).float().square().mean().backward()
        o1.step()
        o2.zero_grad()
        m2(x).square().mean().backward()
        o2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        # bf16 grads vs fp32 grads: loose tolerance
        assert (p1.float() - p2).abs().max().item() < 5e-2


def test_sample_pairing_compiles_to_pairing_slots():
    """SamplePairing policies compile to OP_PAIRING slots with an in-range
    batch-slot partner in BOTH compilers (scalar + vectorized)."""
    import numpy as np
    from fast_autoaugment_amd.aug import ops as aug_ops
    policy = [[("SamplePairing", 1.0, 0.5)]]
    for compiler in (aug_ops.compile_program, aug_ops.compile_program_fast):
        rng = np.random.default_rng(5)
        prog = compiler(policy, 16, 32, 32, rng)
        codes = prog[:, 0, 0]
        assert (codes == float(aug_ops.OpCode.PAIRING)).all(), compiler.__name__
        alphas = prog[:, 0, 1]
        # level 0.5 in range [0, 0.4] -> alpha 0.2
        assert np.allclose(alphas, 0.2), compiler.__name__
        partners = prog[:, 0, 2]
        assert ((partners >= 0) & (partners < 16)).all(), compiler.__name__
