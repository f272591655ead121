#!/usr/bin/env python3
"""In-process hipGraph corruption hunt.

Captures the flagship fwd+bwd (gather grad mode, like bench.py) in a
hipGraph and replays it many times per variant, checking params/grads for
non-finite values after every replay. The FAA_DBIAS variants isolate the
mechanism behind the round-1 "colsum quarantine":

  colsum   dbias = colsum v2 in-graph (round-2 default)
  legacy   dbias = round-1 atomic colsum in-graph
  torch    dbias = at::sum in-graph (round-1 quarantine default)
  dummy    dbias = at::sum + a DISCARDED colsum launch (perturbation
           control: if this flakes, colsum itself is innocent)
  static   colsum v2 into non-pool buffers + clone

Usage (GPU box):  python tools/nan_hunt.py [--replays 60] [--repeats 3]
                  [--variants colsum,torch,dummy,...] [--no-eager-tail]
"""
import argparse
import os
import sys

sys.path.insert(0, ".")
import torch


def build(model_name="wresnet40_2"):
    from fast_autoaugment_amd.metrics import CrossEntropyLabelSmooth
    from fast_autoaugment_amd.models import build_model
    from fast_autoaugment_amd.optim import FusedSGD
    from fast_autoaugment_amd.ops.conv import patch_convs
    from fast_autoaugment_amd.parallel.flat import flatten_module
    torch.manual_seed(0)
    model = build_model({"type": model_name}, 10).cuda().to(
        memory_format=torch.channels_last)
    flat = flatten_module(model, work_dtype=torch.bfloat16)
    patch_convs(model)
    opt = FusedSGD(flat, lr=0.1, momentum=0.9, nesterov=True,
                   weight_decay=2e-4, grad_clip=5.0)
    crit = CrossEntropyLabelSmooth(10, 0.0)
    model.train()
    return model, flat, opt, crit


def run_variant(mode, replays, eager_tail, batch=128, seed=0,
                side_warmup=False):
    os.environ["FAA_DBIAS"] = mode
    model, flat, opt, crit = build()
    torch.manual_seed(1234 + seed)
    x = (torch.randn(batch, 3, 32, 32, device="cuda") * 0.5).bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (batch,), device="cuda")

    def fwd_bwd():
        loss = crit(model(x), y)
        loss.backward()
        return loss

    # eager warmup (3 full steps, also instantiates any static buffers);
    # --side-warmup mirrors bench.py's side-stream warmup (the last
    # structural difference vs the bench graph in the c128 bisect)
    def warm():
        for _ in range(3):
            opt.zero_grad()
            fwd_bwd()
            opt.step()
    if side_warmup:
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            warm()
        torch.cuda.current_stream().wait_stream(side)
    else:
        warm()
    torch.cuda.synchronize()

    # gather-mode capture: assignment-mode backward
    for p in flat.params:
        p.grad = None
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        fwd_bwd()
    base = flat.flat_param.data_ptr()
    rows = []
    for p in flat.params:
        gr = p.grad
        assert gr is not None and gr.dtype == torch.bfloat16
        rows.append([gr.data_ptr(), (p.data.data_ptr() - base) // 2, gr.numel()])
    table = torch.tensor(rows, dtype=torch.int64, device="cuda")
    flat.flat_grad.zero_()
    from fast_autoaugment_amd.ops import ext
    CX = ext()

    bad_at = -1
    bad_param = None
    for i in range(replays):
        g.replay()
        if eager_tail:
            CX.gather_grads(table, flat.flat_grad)
            opt.step()
        gbad = ~torch.isfinite(flat.flat_grad.float()).all()
        pbad = ~torch.isfinite(flat.flat_param.float()).all()
        if bool(gbad) or bool(pbad):
            bad_at = i
            # localize: which param region went non-finite first
            fg = flat.flat_grad.float()
            for (pi, p) in enumerate(flat.params):
                off = (p.data.data_ptr() - base) // 2
                seg = fg[off:off + p.numel()]
                if not torch.isfinite(seg).all():
                    bad_param = (pi, list(p.shape))
                    break
            break
    del g
    torch.cuda.synchronize()
    return bad_at, bad_param


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--replays", type=int, default=60)
    ap.add_argument("--repeats", type=int, default=3)
    ap.add_argument("--variants", type=str,
                    default="colsum,legacy,torch,dummy,static")
    ap.add_argument("--no-eager-tail", action="store_true")
    ap.add_argument("--side-warmup", action="store_true")
    args = ap.parse_args()
    for mode in args.variants.split(","):
        for rep in range(args.repeats):
            bad_at, bad_param = run_variant(mode, args.replays,
                                            not args.no_eager_tail, seed=rep,
                                            side_warmup=args.side_warmup)
            status = "CLEAN" if bad_at < 0 else f"NAN@replay{bad_at} param={bad_param}"
            print(f"{mode:>8} rep{rep} tail={not args.no_eager_tail} "
                  f"sidewarm={args.side_warmup}: {status}", flush=True)


if __name__ == "__main__":
    main()
