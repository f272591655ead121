#!/usr/bin/env python3
"""Does a hipGraph-captured train step actually TRAIN? (call11 freeze bisect)

Three variants on one fixed batch, 60 steps each, losses every 10:
  eager      plain loop (reference)
  graph_fb   capture fwd+bwd only; opt.step eager between replays
  graph_all  capture fwd+bwd+opt.step (the trainer _GraphedTrainStep shape)
A frozen loss in graph_* that decreases in eager localizes the breakage.
"""
import sys

sys.path.insert(0, ".")
import torch


def build(lr=0.05):
    from fast_autoaugment_amd.metrics import CrossEntropyLabelSmooth
    from fast_autoaugment_amd.models import build_model
    from fast_autoaugment_amd.optim import FusedSGD
    from fast_autoaugment_amd.ops.conv import patch_convs
    from fast_autoaugment_amd.parallel.flat import flatten_module
    torch.manual_seed(0)
    m = build_model({"type": "wresnet40_2"}, 10).cuda().to(
        memory_format=torch.channels_last)
    flat = flatten_module(m, work_dtype=torch.bfloat16)
    patch_convs(m)
    opt = FusedSGD(flat, lr=lr, momentum=0.9, nesterov=True,
                   weight_decay=2e-4, grad_clip=5.0)
    crit = CrossEntropyLabelSmooth(10, 0.0)
    m.train()
    return m, flat, opt, crit


def data_batch():
    torch.manual_seed(7)
    d = (torch.randn(128, 3, 32, 32, device="cuda") * 0.5).bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (128,), device="cuda")
    return d, y


def run(mode, steps=60):
    m, flat, opt, crit = build()
    d, y = data_batch()

    def body():
        opt.zero_grad(set_to_none=False)
        loss = crit(m(d), y)
        loss.backward()
        return loss

    losses = []
    if mode == "eager":
        for i in range(steps):
            loss = body()
            opt.step()
            if i % 10 == 0 or i == steps - 1:
                losses.append(round(loss.item(), 4))
    else:
        for _ in range(3):
            loss = body()
            opt.step()
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        if mode == "graph_all":
            with torch.cuda.graph(g):
                loss_s = body()
                opt.step()
        else:
            with torch.cuda.graph(g):
                loss_s = body()
        for i in range(steps - 3):
            g.replay()
            if mode == "graph_fb":
                opt.step()
            if i % 10 == 0 or i == steps - 4:
                losses.append(round(loss_s.item(), 4))
    print(f"{mode:>10}: {losses}", flush=True)




def run_gather(mode, steps=60):
    """bench.py-shaped capture variants.

    gather_bwd: side-stream warmup, p.grad=None, capture fwd+BACKWARD
                (round-1/2 bench gather mode — AccumulateGrad inside capture)
    gather_ag : capture fwd + torch.autograd.grad (NO AccumulateGrad)
    all_ag    : gather_ag + gather_grads + opt.step all inside the graph
    """
    from fast_autoaugment_amd.ops import ext
    CX = ext()
    m, flat, opt, crit = build()
    d, y = data_batch()

    def fwd_loss():
        return crit(m(d), y)

    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(3):
            opt.zero_grad(set_to_none=False)
            loss = fwd_loss()
            loss.backward()
            opt.step()
    torch.cuda.current_stream().wait_stream(side)
    del loss
    torch.cuda.synchronize()

    g = torch.cuda.CUDAGraph()
    base = flat.flat_param.data_ptr()
    if mode == "gather_bwd":
        for p in flat.params:
            p.grad = None
        with torch.cuda.graph(g):
            loss_s = fwd_loss()
            loss_s.backward()
        gs = [p.grad for p in flat.params]
    else:
        with torch.cuda.graph(g):
            loss_s = fwd_loss()
            gs = torch.autograd.grad(loss_s, flat.params)
            if mode == "all_ag":
                rows, tb = [], None
                for p, gr in zip(flat.params, gs):
                    rows.append([gr.data_ptr(), (p.data.data_ptr() - base) // 2,
                                 gr.numel()])
                tb = torch.tensor(rows, dtype=torch.int64, device="cuda")
                CX.gather_grads(tb, flat.flat_grad)
                opt.step()
    if mode != "all_ag":
        rows = [[gr.data_ptr(), (p.data.data_ptr() - base) // 2, gr.numel()]
                for p, gr in zip(flat.params, gs)]
        tb = torch.tensor(rows, dtype=torch.int64, device="cuda")
        flat.flat_grad.zero_()
    losses = []
    for i in range(steps - 3):
        g.replay()
        if mode != "all_ag":
            CX.gather_grads(tb, flat.flat_grad)
            opt.step()
        if i % 10 == 0 or i == steps - 4:
            losses.append(round(loss_s.item(), 4))
    print(f"{mode:>10}: {losses}", flush=True)


def run_class(steps=60, vary_data=False):
    """Drive the actual engine._GraphedTrainStep class (freeze bisect)."""
    from fast_autoaugment_amd.engine.trainer import _GraphedTrainStep
    m, flat, opt, crit = build()
    d, y = data_batch()
    gstep = _GraphedTrainStep(m, crit, opt, False)
    losses = []
    torch.manual_seed(77)
    for i in range(steps):
        if vary_data:
            di = (d + 0.01 * torch.randn_like(d.float()).bfloat16()).contiguous(
                memory_format=torch.channels_last)
        else:
            di = d
        res = gstep.step(di, y)
        if res is None:
            opt.zero_grad(set_to_none=False)
            loss = crit(m(di), y)
            loss.backward()
            opt.step()
            opt.zero_grad(set_to_none=False)
            lv = float(loss.item())
            del loss
        else:
            lv = float(res[0].item())
        if i % 10 == 0 or i == steps - 1:
            losses.append(round(lv, 4))
        if i == 10:
            gnorm = float(flat.flat_grad.float().norm().item())
            print(f"  [class vary={vary_data}] grad norm after gather @10: {gnorm:.4f}",
                  flush=True)
    print(f"class(vary={vary_data}): {losses}", flush=True)


if __name__ == "__main__":
    for mode in (sys.argv[1:] or ["eager", "graph_fb", "graph_all"]):
        if mode.startswith(("gather", "all_")):
            run_gather(mode)
        elif mode == "class_fixed":
            run_class(vary_data=False)
        elif mode == "class_vary":
            run_class(vary_data=True)
        else:
            run(mode)
