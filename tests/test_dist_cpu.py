"""Multi-process data-parallel tests over gloo (CPU, world_size=2).

Validates FlatDDP's broadcast-init + bucketed all-reduce semantics against
single-process training, and the distributed loader sharding.
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _find_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _ddp_worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(100 + rank)   # different init per rank: broadcast must fix it

    from fast_autoaugment_amd.parallel.ddp import FlatDDP

    model = torch.nn.Sequential(
        torch.nn.Conv2d(3, 8, 3, padding=1), torch.nn.BatchNorm2d(8),
        torch.nn.ReLU(), torch.nn.Conv2d(8, 4, 1))
    ddp = FlatDDP(model, bucket_bytes=1 << 10)   # small buckets: several fire
    opt = torch.optim.SGD(ddp.parameters(), lr=0.1, momentum=0.9)

    # fixed per-rank shards of a common batch
    torch.manual_seed(7)
    data = torch.randn(4 * world, 3, 8, 8)
    shard = data[rank::world]
    for _ in range(3):
        opt.zero_grad(set_to_none=False)
        out = ddp(shard)
        out.square().mean().backward()
        ddp.finish_gradient_sync()
        opt.step()
    flat = ddp.flat.flat_param.detach().clone()
    results[rank] = flat
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_flatddp_matches_single_process():
    world = 2
    port = _find_port()
    ctx = mp.get_context("spawn")
    with mp.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_ddp_worker, args=(r, world, port, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(150)
            assert p.exitcode == 0
        r0, r1 = results[0], results[1]

    # ranks end identical
    assert torch.allclose(r0, r1, atol=1e-6)

    # single-process reference: full batch, grads averaged over world
    torch.manual_seed(100 + 0)
    from fast_autoaugment_amd.parallel.flat import flatten_module
    model = torch.nn.Sequential(
        torch.nn.Conv2d(3, 8, 3, padding=1), torch.nn.BatchNorm2d(8),
        torch.nn.ReLU(), torch.nn.Conv2d(8, 4, 1))
    flat = flatten_module(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    torch.manual_seed(7)
    data = torch.randn(4 * 2, 3, 8, 8)
    for _ in range(3):
        opt.zero_grad(set_to_none=False)
        # mean over per-rank shard losses == ddp's averaged gradient
        loss = (model(data[0::2]).square().mean() + model(data[1::2]).square().mean()) / 2
        loss.backward()
        opt.step()
    assert torch.allclose(r0, flat.flat_param, atol=1e-5)


def test_loader_distributed_sharding():
    from fast_autoaugment_amd.data.loader import AugLoader, TensorStore
    imgs = np.zeros((100, 32, 32, 3), np.uint8)
    labels = np.arange(100, dtype=np.int64)
    store = TensorStore(imgs, labels, device="cpu")
    mean = np.zeros(3, np.float32)
    std = np.ones(3, np.float32)
    seen = []
    for rank in range(2):
        ld = AugLoader(store, 10, None, train=False, mean=mean, std=std,
                       shuffle=True, drop_last=False, rank=rank, world_size=2,
                       seed=5, prefetch=0)
        ld.set_epoch(3)
        labels_seen = []
        for _, lab in ld:
            labels_seen.extend(lab.tolist())
        seen.append(set(labels_seen))
        assert len(labels_seen) == 50
    # shards are disjoint and cover everything
    assert seen[0] | seen[1] == set(range(100))
    assert not (seen[0] & seen[1])


def _bench_dp_worker(rank, world, port, results):
    """Mirrors bench.py's distributed step on CPU/gloo: AugLoader batches ->
    FlatDDP fwd/bwd -> finish_gradient_sync -> FusedSGD step."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(50 + rank)

    from fast_autoaugment_amd.data.loader import AugLoader, TensorStore
    from fast_autoaugment_amd.metrics import CrossEntropyLabelSmooth
    from fast_autoaugment_amd.models import build_model
    from fast_autoaugment_amd.optim import FusedSGD
    from fast_autoaugment_amd.parallel.ddp import FlatDDP
    from fast_autoaugment_amd.policies import resolve_aug

    rng = np.random.default_rng(9)
    imgs = rng.integers(0, 256, (64, 32, 32, 3), dtype=np.uint8)
    labels = (np.arange(64) % 10).astype(np.int64)
    store = TensorStore(imgs, labels, device="cpu")
    mean = np.zeros(3, np.float32)
    std = np.ones(3, np.float32)
    loader = AugLoader(store, 8, resolve_aug("fa_reduced_cifar10"), train=True,
                       mean=mean, std=std, cutout=4, rank=rank, world_size=world,
                       seed=3, prefetch=0)
    model = build_model({"type": "wresnet40_2"}, 10)
    ddp = FlatDDP(model)
    opt = FusedSGD(ddp.flat, lr=0.05, weight_decay=2e-4, grad_clip=5.0)
    crit = CrossEntropyLabelSmooth(10, 0.0)
    it = iter(loader)
    for _ in range(2):
        data, label = next(it)
        opt.zero_grad()
        loss = crit(ddp(data), label)
        loss.backward()
        ddp.finish_gradient_sync()
        opt.step()
    results[rank] = ddp.flat.flat_param.detach().clone()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_bench_style_dp_step():
    world = 2
    port = _find_port()
    ctx = mp.get_context("spawn")
    with mp.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_bench_dp_worker, args=(r, world, port, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(280)
            assert p.exitcode == 0
        assert torch.allclose(results[0], results[1], atol=1e-6)
        assert torch.isfinite(results[0]).all()


@pytest.mark.timeout(600)
def test_bench_torchrun_cpu_dry_run():
    """The driver launches bench.py under torch.distributed.run; dry-run that
    exact invocation on CPU/gloo (FAA_BENCH_CPU=1) and check the JSON line."""
    import json
    import subprocess
    import sys
    env = dict(os.environ, FAA_BENCH_CPU="1", FAA_SYNTH_TRAIN="256",
               FAA_SYNTH_TEST="64")
    root = os.path.join(os.path.dirname(__file__), "..")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_find_port()), "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch", "16"],
        env=env, cwd=root, capture_output=True, text=True, timeout=550)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 2
    assert out["metric"] == "images/sec"
    assert out["config"]["parallelism"] == "dp2"
    assert out["config"]["global_batch"] == 32


def test_train_dist_launcher_node_rank_env(tmp_path, monkeypatch):
    """Inner launcher must honor NODE_RANK/NNODES from the ssh fan-out."""
    import subprocess
    import sys as _sys
    import os as _os
    repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    # patch torch.distributed.run with an echo stub via a sitecustomize-free
    # trick: just inspect the constructed command by running with a fake
    # python -m target is hard; instead check launch_local's args directly.
    _sys.path.insert(0, repo)
    import importlib
    td = importlib.import_module("train_dist")
    captured = {}

    def fake_launch(np_per_node, master_addr, master_port, node_rank, nnodes, rest):
        captured.update(node_rank=node_rank, nnodes=nnodes, np=np_per_node)
        return 0

    monkeypatch.setattr(td, "launch_local", fake_launch)
    monkeypatch.setenv("NODE_RANK", "2")
    monkeypatch.setenv("NNODES", "4")
    monkeypatch.setattr(_sys, "argv", ["train_dist.py", "--np", "3"])
    try:
        td.main()
    except SystemExit as e:
        assert e.code == 0
    assert captured == {"node_rank": 2, "nnodes": 4, "np": 3}


def test_bench_cpu_dry_run_imagenet(tmp_path):
    """bench.py CPU dry-run must exercise the ImageNet pipeline branch
    (run_pipeline_imagenet_cpu + 18-wide post), not just CIFAR."""
    import subprocess
    import sys as _sys
    import os as _os
    repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    env = dict(_os.environ, FAA_BENCH_CPU="1", FAA_SYNTH_TRAIN="16", FAA_SYNTH_TEST="8")
    r = subprocess.run(
        [_sys.executable, "bench.py", "--model", "resnet50", "--dataset", "imagenet",
         "--batch", "2", "--steps", "1", "--warmup", "0"],
        cwd=repo, env=env, capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, r.stderr[-1500:]
    assert '"image": "224x224"' in r.stdout


def _ddp8_worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(200 + rank)
    from fast_autoaugment_amd.parallel.ddp import FlatDDP
    model = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.ReLU(),
                                torch.nn.Linear(32, 4))
    ddp = FlatDDP(model, bucket_bytes=1 << 9)
    opt = torch.optim.SGD(ddp.parameters(), lr=0.05)
    torch.manual_seed(11)
    data = torch.randn(2 * world, 16)
    for _ in range(2):
        opt.zero_grad(set_to_none=False)
        ddp(data[rank::world]).square().mean().backward()
        ddp.finish_gradient_sync()
        opt.step()
    results[rank] = ddp.flat.flat_param.detach().clone()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_flatddp_world8_matches_single_process():
    """DP=8 bucketed all-reduce equivalence on gloo (VERDICT r1 item 4:
    the scaling world size the driver runs on hardware)."""
    world = 8
    port = _find_port()
    ctx = mp.get_context("spawn")
    with mp.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_ddp8_worker, args=(r, world, port, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(250)
            assert p.exitcode == 0
        flats = [results[r] for r in range(world)]
    for f in flats[1:]:
        assert torch.allclose(flats[0], f, atol=1e-6)

    torch.manual_seed(200 + 0)
    from fast_autoaugment_amd.parallel.flat import flatten_module
    model = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.ReLU(),
                                torch.nn.Linear(32, 4))
    flat = flatten_module(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    torch.manual_seed(11)
    data = torch.randn(2 * 8, 16)
    for _ in range(2):
        opt.zero_grad(set_to_none=False)
        loss = sum(model(data[r::8]).square().mean() for r in range(8)) / 8
        loss.backward()
        opt.step()
    assert torch.allclose(flats[0], flat.flat_param, atol=1e-5)
