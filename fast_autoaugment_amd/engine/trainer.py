"""Training engine (reference train.py:35-322).

train_and_eval/run_epoch keep the reference's control flow and semantics —
manual L2 decay on non-BN params included in the clip norm (train.py:40,61),
grad-norm clip default 5.0 (train.py:63-65), per-step fractional-epoch LR
schedule (train.py:90-91), per-step EMA (train.py:69-70), eval every 5
epochs, NaN guard, `.pth` checkpoint layout {epoch, log, optimizer, model,
ema} (train.py:307-317) — on an MI355X-native substrate: bf16 autocast +
channels_last NHWC, GPU-resident augmented loaders, fused HIP loss/step
kernels, and RCCL flat-buffer data parallelism.
"""
from __future__ import annotations

import math
import os
from collections import OrderedDict
from typing import Optional

import torch
from torch import nn

from ..common import EMA, get_logger
from ..config import Config as C
from ..data import get_dataloaders
from ..lr_scheduler import build_scheduler
from ..metrics import (Accumulator, CrossEntropyLabelSmooth,
                       CrossEntropyMixUpLabelSmooth, accuracy, mixup,
                       get_summary_writer)
from ..models import get_model, num_class
from ..optim import build_optimizer

logger = get_logger("faa_amd.trainer")


def _decay_params(model: nn.Module):
    """Non-BN parameters for the manual weight-decay term (train.py:40)."""
    from ..parallel.flat import bn_param_names
    bn = bn_param_names(model)
    return [p for name, p in model.named_parameters() if name not in bn]


def _apply_manual_wd(params, wd: float):
    """grad += wd * p — the gradient of the reference's loss-side WD term."""
    with torch.no_grad():
        ps = [p for p in params if p.grad is not None]
        if not ps:
            return
        torch._foreach_add_([p.grad for p in ps], ps, alpha=wd)


class _GraphedTrainStep:
    """hipGraph-captured train step for the trainer hot loop.

    The eager loop issues ~1500 kernel launches per step at 61% GPU-busy
    (profiles/train_profile_r02.txt). The capture body computes grads with
    ``torch.autograd.grad`` — NOT ``loss.backward()`` — because backward
    routes through per-param AccumulateGrad nodes whose recorded stream
    (from the first eager backward) injects cross-stream syncs into the
    capture: segfault when that stream is the default stream, silently
    mis-captured dependency edges otherwise (tools/graph_train_check.py,
    gpurun_out/call12/13.log — the round-1 "colsum corruption" mechanism).
    Grads land at stable capture-pool addresses; one gather kernel packs
    them into the flat buffer and the fused optimizer steps eagerly after
    each replay (4 launches/step). Batches are copied into static input
    buffers; loss/preds read back after replay. Falls back (returns None)
    on shape changes or layout-incompatible grads.
    """

    WARMUP = 3

    def __init__(self, model, loss_fn, optimizer, flip_batch: bool):
        self.model = model
        self.loss_fn = loss_fn
        self.opt = optimizer
        self.flip = flip_batch
        self.graph = None
        self.shape = None
        self.warm = 0
        self.dead = False
        self.replays = 0
        # platform-bug workaround (docs/GRAPH_NAN.md): the graph pool
        # corrupts after ~100-150 replays when eager allocations churn
        # between replays; re-capturing with a FRESH graph+pool every
        # RECAP_EVERY steps stays under the onset (~1 extra eager-step
        # cost amortized over the window)
        self.recap_every = int(os.environ.get("FAA_TRAIN_GRAPH_RECAP", "64"))

    def _capture(self, data, label):
        import torch as _t
        flat = self.opt.flat
        self.data_s = data.clone()
        self.label_s = label.clone()
        _t.cuda.synchronize()
        g = _t.cuda.CUDAGraph()
        with _t.cuda.graph(g):
            if self.flip:
                from ..ops.conv import conv_flip_all
                conv_flip_all()
            preds = self.model(self.data_s)
            loss = self.loss_fn(preds, self.label_s)
            gs = _t.autograd.grad(loss, flat.params, allow_unused=True)
        base = flat.flat_param.data_ptr()
        rows = []
        for p, gr in zip(flat.params, gs):
            if gr is None:
                continue
            if gr.dtype != _t.bfloat16:
                return False
            if gr.dim() == 4:
                ok = (gr.is_contiguous(memory_format=_t.channels_last)
                      or (gr.is_contiguous()
                          and (gr.size(1) == 1 or gr.size(2) * gr.size(3) == 1)))
            else:
                ok = gr.is_contiguous()
            if not ok:
                return False
            rows.append([gr.data_ptr(), (p.data.data_ptr() - base) // 2, gr.numel()])
        self.table = _t.tensor(rows, dtype=_t.int64, device=data.device)
        flat.flat_grad.zero_()          # pad gaps / unused params stay zero
        self.graph = g
        self.loss_s, self.preds_s = loss, preds
        return True

    def step(self, data, label):
        if self.dead:
            return None
        if self.shape is None:
            self.shape = tuple(data.shape)
        if tuple(data.shape) != self.shape:
            return None
        if self.graph is None:
            self.warm += 1
            if self.warm <= self.WARMUP:
                return None                      # caller runs the eager step
            if not self._capture(data, label):
                self.dead = True
                return None
            self.replays = 0
            # capture records but does not execute: replay for this batch
        elif self.replays >= self.recap_every > 0:
            import gc
            self.graph = None
            torch.cuda.synchronize()
            gc.collect()                         # free the old graph + pool
            if not self._capture(data, label):
                self.dead = True
                return None
            self.replays = 0
        else:
            self.data_s.copy_(data, non_blocking=True)
            self.label_s.copy_(label, non_blocking=True)
        self.replays += 1
        self.graph.replay()
        from ..ops import ext
        ext().gather_grads(self.table, self.opt.flat.flat_grad)
        self.opt.step()
        if os.environ.get("FAA_TRAIN_GRAPH_DEBUG") == "1":
            self._dbg = getattr(self, "_dbg", 0) + 1
            if self._dbg in (1, 10, 40):
                import torch as _t
                _t.cuda.synchronize()
                fl = self.opt.flat
                print(f"# graphstep dbg n={self._dbg} loss={float(self.loss_s):.4f} "
                      f"|g|={float(fl.flat_grad.float().norm()):.4f} "
                      f"|p|={float(fl.flat_param.float().norm()):.4f} "
                      f"lr={self.opt.param_groups[0]['lr']:.5f} "
                      f"rows={self.table.size(0)} "
                      f"data|m|={float(self.data_s.float().abs().mean()):.4f}",
                      flush=True)
        return self.loss_s, self.preds_s


def run_epoch(model, loader, loss_fn, optimizer, desc_default="", epoch=0,
              writer=None, verbose=False, scheduler=None, is_master=True,
              ema: Optional[EMA] = None, wd: float = 0.0, device="cpu",
              autocast_dtype=None):
    conf = C.get()
    decay_params = _decay_params(model) if optimizer else []
    metrics = Accumulator()
    cnt = 0
    total_steps = max(len(loader), 1)
    steps = 0
    use_mixup = conf.get_value("mixup", 0.0) > 0.0 and optimizer is not None
    amp = (autocast_dtype is not None and device != "cpu")
    # per-step scalars accumulate on-device; ONE host sync per epoch
    acc = torch.zeros(3, device=device)

    flip_batch = (optimizer is not None and device != "cpu"
                  and os.environ.get("FAA_FLIP_BATCH") == "1")
    # graph-captured train step (pure-bf16 fused-optimizer path only)
    graphed = None
    if (optimizer is not None and device != "cpu" and not use_mixup and not amp
            and os.environ.get("FAA_TRAIN_GRAPHS", "0") == "1"):
        from ..optim import FusedRMSpropTF, FusedSGD
        if isinstance(optimizer, (FusedSGD, FusedRMSpropTF)):
            graphed = getattr(optimizer, "_faa_graph_step", None)
            if graphed is None:
                graphed = _GraphedTrainStep(model, loss_fn, optimizer, flip_batch)
                optimizer._faa_graph_step = graphed
    for data, label in loader:
        steps += 1
        data = data.to(device, non_blocking=True)
        label = label.to(device, non_blocking=True)
        if device != "cpu":
            data = data.contiguous(memory_format=torch.channels_last)

        res = graphed.step(data, label) if graphed is not None else None
        if res is not None:
            loss, preds = res
            if ema is not None:
                ema(model, (epoch - 1) * total_steps + steps)
        else:
            if flip_batch:
                from ..ops.conv import conv_flip_all
                conv_flip_all()   # one launch refreshes all bwd-data repacks
            with torch.autocast("cuda", dtype=autocast_dtype, enabled=amp):
                if use_mixup:
                    data, targets, shuffled_targets, lam = mixup(data, label, conf["mixup"])
                    preds = model(data)
                    loss = loss_fn(preds, targets, shuffled_targets, lam)
                else:
                    preds = model(data)
                    loss = loss_fn(preds, label)

            if optimizer:
                loss.backward()
                if hasattr(model, "finish_gradient_sync"):
                    model.finish_gradient_sync()
                from ..optim import FusedSGD, FusedRMSpropTF
                if not isinstance(optimizer, (FusedSGD, FusedRMSpropTF)):
                    # FusedSGD folds manual WD + global clip into its kernels
                    if wd > 0.0:
                        _apply_manual_wd(decay_params, wd)
                    grad_clip = conf["optimizer"].get("clip", 5.0)
                    if grad_clip > 0:
                        nn.utils.clip_grad_norm_(model.parameters(), grad_clip)
                optimizer.step()
                optimizer.zero_grad(set_to_none=False)
                if ema is not None:
                    ema(model, (epoch - 1) * total_steps + steps)

        with torch.no_grad():
            top1, top5 = accuracy(preds, label, (1, 5))
            n = len(data)
            acc[0] += loss.detach().float() * n
            acc[1] += top1 * n
            acc[2] += top5 * n
        cnt += n
        if scheduler is not None:
            scheduler.step(epoch - 1 + float(steps) / total_steps)
        del preds, loss, top1, top5, data, label

    if cnt == 0:
        return metrics
    vals = acc.cpu()
    metrics.add_dict({"loss": float(vals[0]), "top1": float(vals[1]), "top5": float(vals[2])})
    metrics /= cnt
    if optimizer:
        metrics.metrics["lr"] = optimizer.param_groups[0]["lr"]
    if verbose and is_master:
        logger.info("[%s %03d/%03d] %s", desc_default, epoch, conf["epoch"], metrics)
        if writer is not None:
            for key, value in metrics.items():
                writer.add_scalar(key, value, epoch)
    return metrics


def train_and_eval(tag, dataroot, test_ratio=0.0, cv_fold=0, reporter=None,
                   metric="last", save_path=None, only_eval=False,
                   local_rank=-1, evaluation_interval=5, log_path=None):
    conf = C.get()
    use_cuda = torch.cuda.is_available()
    device = "cuda" if use_cuda else "cpu"
    autocast_dtype = None
    if use_cuda:
        prec = conf.get_value("precision", "bf16")
        autocast_dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
                          "fp32": None}[prec]

    world_size, rank = 1, 0
    if local_rank >= 0:
        import torch.distributed as dist
        if not dist.is_initialized():
            dist.init_process_group(backend="nccl" if use_cuda else "gloo",
                                    init_method="env://")
        world_size, rank = dist.get_world_size(), dist.get_rank()
        if use_cuda:
            torch.cuda.set_device(local_rank)
        conf["lr"] = conf["lr"] * world_size     # linear LR scaling (train.py:117)
    is_master = local_rank < 0 or rank == 0

    if not reporter:
        reporter = lambda **kwargs: 0

    max_epoch = conf["epoch"]
    nc = num_class(conf["dataset"])
    out_dtype = torch.bfloat16 if autocast_dtype == torch.bfloat16 else torch.float32
    trainsampler, trainloader, validloader, testloader_ = get_dataloaders(
        conf["dataset"], conf["batch"], dataroot, test_ratio, split_idx=cv_fold,
        multinode=(local_rank >= 0), rank=rank, world_size=world_size,
        device=device, out_dtype=out_dtype)

    # Fast path: pure-bf16 flat weights + fused optimizer + MFMA convs on GPU.
    use_fast = (use_cuda and conf["optimizer"]["type"] in ("sgd", "rmsprop")
                and autocast_dtype == torch.bfloat16
                and conf.get_value("pure_bf16", True))
    work_dtype = torch.bfloat16 if use_fast else None

    model = get_model(conf["model"], nc, local_rank=local_rank, device=device,
                      work_dtype=work_dtype)
    model_ema = get_model(conf["model"], nc, local_rank=-1, device=device)
    model_ema.eval()
    if use_fast:
        # EMA evaluation model follows the bf16 working dtype (buffers stay fp32)
        for p in model_ema.parameters():
            p.data = p.data.to(torch.bfloat16)

    criterion_ce = criterion = CrossEntropyLabelSmooth(nc, conf.get_value("lb_smooth", 0))
    if conf.get_value("mixup", 0.0) > 0.0:
        criterion = CrossEntropyMixUpLabelSmooth(nc, conf.get_value("lb_smooth", 0))

    if use_fast:
        from ..ops.conv import patch_convs
        from ..optim import FusedSGD
        from ..parallel.flat import flatten_module
        flat = model.flat if hasattr(model, "flat") else \
            flatten_module(model, work_dtype=torch.bfloat16)
        patch_convs(model)
        if conf["optimizer"]["type"] == "sgd":
            optimizer = FusedSGD(flat, lr=conf["lr"],
                                 momentum=conf["optimizer"].get("momentum", 0.9),
                                 nesterov=conf["optimizer"].get("nesterov", True),
                                 weight_decay=conf["optimizer"].get("decay", 0.0),
                                 grad_clip=conf["optimizer"].get("clip", 5.0))
        else:
            from ..optim import FusedRMSpropTF
            optimizer = FusedRMSpropTF(flat, lr=conf["lr"], alpha=0.9,
                                       momentum=0.9, eps=0.001,
                                       weight_decay=conf["optimizer"].get("decay", 0.0),
                                       grad_clip=conf["optimizer"].get("clip", 0.0))
        autocast_dtype = None      # the model computes natively in bf16
    else:
        optimizer = build_optimizer(conf["optimizer"], model.parameters(), conf["lr"])
    scheduler = build_scheduler(conf.conf, optimizer, conf["lr"])

    writers = [get_summary_writer(f"./logs/{tag}/{x}", bool(tag) and is_master)
               for x in ["train", "valid", "test"]]

    ema = EMA(conf["optimizer"]["ema"]) if conf["optimizer"].get("ema", 0) > 0 and is_master else None

    result = OrderedDict()
    epoch_start = 1
    # checkpoint load + resume (reference train.py:191-218)
    if save_path != "test.pth" and save_path and os.path.exists(save_path):
        logger.info("%s found. loading...", save_path)
        data = torch.load(save_path, map_location=device, weights_only=False)
        key = "model" if "model" in data else "state_dict"
        if "epoch" not in data:
            model.load_state_dict(data)
        else:
            logger.info("checkpoint epoch@%d", data["epoch"])
            sd = data[key]
            raw = model.module if hasattr(model, "module") else model
            raw.load_state_dict({k.replace("module.", ""): v for k, v in sd.items()})
            if use_fast:
                flat.flat_master.copy_(flat.flat_param.float())
            try:
                optimizer.load_state_dict(data["optimizer"])
            except Exception as e:
                logger.warning("optimizer state not loaded (%s); fresh momentum", e)
            if data["epoch"] < max_epoch:
                epoch_start = data["epoch"]
            else:
                only_eval = True
            if ema is not None:
                saved = data.get("ema")
                if isinstance(saved, dict) and saved:
                    ema.shadow = {k: v.to(device) for k, v in saved.items()}
        del data
    elif save_path and not os.path.exists(save_path):
        if only_eval:
            logger.warning("checkpoint not found; only-eval off.")
        only_eval = False

    if local_rank >= 0:
        import torch.distributed as dist
        raw = model.module if hasattr(model, "module") else model
        for _, x in raw.state_dict().items():
            dist.broadcast(x, 0)
        if use_cuda:
            torch.cuda.synchronize()

    if only_eval:
        logger.info("evaluation only+")
        model.eval()
        rs = dict()
        with torch.no_grad():
            rs["train"] = run_epoch(model, trainloader, criterion, None, desc_default="train",
                                    epoch=0, writer=writers[0], is_master=is_master,
                                    device=device, autocast_dtype=autocast_dtype)
            rs["valid"] = run_epoch(model, validloader, criterion, None, desc_default="valid",
                                    epoch=0, writer=writers[1], is_master=is_master,
                                    device=device, autocast_dtype=autocast_dtype)
            rs["test"] = run_epoch(model, testloader_, criterion, None, desc_default="*test",
                                   epoch=0, writer=writers[2], is_master=is_master,
                                   device=device, autocast_dtype=autocast_dtype)
            if ema is not None and len(ema) > 0:
                model_ema.load_state_dict({k.replace("module.", ""): v
                                           for k, v in ema.state_dict().items()})
                rs["valid"] = run_epoch(model_ema, validloader, criterion_ce, None,
                                        desc_default="valid(EMA)", epoch=0, writer=writers[1],
                                        verbose=is_master, device=device,
                                        autocast_dtype=autocast_dtype)
                rs["test"] = run_epoch(model_ema, testloader_, criterion_ce, None,
                                       desc_default="*test(EMA)", epoch=0, writer=writers[2],
                                       verbose=is_master, device=device,
                                       autocast_dtype=autocast_dtype)
        for key in ["loss", "top1", "top5"]:
            for setname in ["train", "valid", "test"]:
                if setname in rs:
                    result[f"{key}_{setname}"] = rs[setname][key]
        result["epoch"] = 0
        return result

    # train loop
    best_top1 = 0.0
    for epoch in range(epoch_start, max_epoch + 1):
        if local_rank >= 0:
            trainsampler.set_epoch(epoch)
        model.train()
        rs = dict()
        rs["train"] = run_epoch(model, trainloader, criterion, optimizer,
                                desc_default="train", epoch=epoch, writer=writers[0],
                                verbose=(is_master and local_rank <= 0),
                                scheduler=scheduler, ema=ema,
                                wd=conf["optimizer"].get("decay", 0.0),
                                is_master=is_master, device=device,
                                autocast_dtype=autocast_dtype)
        model.eval()
        from ..ops.bnrelu import sync_bn_trackers
        sync_bn_trackers(model.module if hasattr(model, "module") else model)

        if math.isnan(rs["train"]["loss"]):
            raise Exception("train loss is NaN.")

        # EMA->model periodic sync (reference train.py:262-270). ema exists
        # only on the master rank, so EVERY rank must enter the broadcast
        # collective or the job deadlocks — gate on the config alone and let
        # non-masters participate receive-only.
        if (conf["optimizer"].get("ema", 0) > 0
                and conf["optimizer"].get("ema_interval", -1) > 0
                and epoch % conf["optimizer"]["ema_interval"] == 0):
            raw = model.module if hasattr(model, "module") else model
            if ema is not None:
                raw.load_state_dict(ema.state_dict())
            if local_rank >= 0:
                import torch.distributed as dist
                for _, x in raw.state_dict().items():
                    dist.broadcast(x, 0)

        if is_master and (epoch % evaluation_interval == 0 or epoch == max_epoch):
            with torch.no_grad():
                rs["valid"] = run_epoch(model, validloader, criterion_ce, None,
                                        desc_default="valid", epoch=epoch, writer=writers[1],
                                        verbose=is_master, device=device,
                                        autocast_dtype=autocast_dtype)
                rs["test"] = run_epoch(model, testloader_, criterion_ce, None,
                                       desc_default="*test", epoch=epoch, writer=writers[2],
                                       verbose=is_master, device=device,
                                       autocast_dtype=autocast_dtype)
                if ema is not None and len(ema) > 0:
                    model_ema.load_state_dict({k.replace("module.", ""): v
                                               for k, v in ema.state_dict().items()})
                    rs["valid"] = run_epoch(model_ema, validloader, criterion_ce, None,
                                            desc_default="valid(EMA)", epoch=epoch,
                                            writer=writers[1], verbose=is_master,
                                            device=device, autocast_dtype=autocast_dtype)
                    rs["test"] = run_epoch(model_ema, testloader_, criterion_ce, None,
                                           desc_default="*test(EMA)", epoch=epoch,
                                           writer=writers[2], verbose=is_master,
                                           device=device, autocast_dtype=autocast_dtype)

            logger.info("epoch=%d [train] loss=%.4f top1=%.4f [valid] top1=%.4f [test] top1=%.4f",
                        epoch, rs["train"]["loss"], rs["train"]["top1"],
                        rs["valid"]["top1"], rs["test"]["top1"])

            if metric == "last" or rs[metric]["top1"] > best_top1:
                if metric != "last":
                    best_top1 = rs[metric]["top1"]
                for key in ["loss", "top1", "top5"]:
                    for setname in ["train", "valid", "test"]:
                        result[f"{key}_{setname}"] = rs[setname][key]
                result["epoch"] = epoch

                reporter(loss_valid=rs["valid"]["loss"], top1_valid=rs["valid"]["top1"],
                         loss_test=rs["test"]["loss"], top1_test=rs["test"]["top1"])

                if is_master and save_path:
                    raw = model.module if hasattr(model, "module") else model
                    logger.info("save model@%d to %s", epoch, save_path)
                    # .pth layout compatible with the reference (train.py:307-317);
                    # bf16 working weights are stored as fp32 for interchange
                    sd_out = {k: (v.float() if v.dtype == torch.bfloat16 else v)
                              for k, v in raw.state_dict().items()}
                    torch.save({
                        "epoch": epoch,
                        "log": {
                            "train": rs["train"].get_dict(),
                            "valid": rs["valid"].get_dict(),
                            "test": rs["test"].get_dict(),
                        },
                        "optimizer": optimizer.state_dict(),
                        "model": sd_out,
                        "ema": ema.state_dict() if ema is not None else None,
                    }, save_path)

    # drop the cached step graph + its capture pool before the next fold
    # trains in this worker process (search runs folds sequentially)
    if hasattr(optimizer, "_faa_graph_step"):
        optimizer._faa_graph_step = None
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        import gc
        gc.collect()
    del model
    if metric != "last":
        # reference train.py:321 assigns best_top1 unconditionally, which
        # zeroes top1_test under metric='last'; we keep the recorded value.
        result["top1_test"] = best_top1
    return result
