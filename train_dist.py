#!/usr/bin/env python3
"""Multi-GPU / multi-node launcher (reference train_dist.py:105-150).

Single node (the primary MI355X target: 8 GPUs over RCCL/xGMI):
  python train_dist.py --np 8 -c confs/wresnet28x10_cifar.yaml --save ckpt.pth
Multi node: --hosts host1:8,host2:8 launches per-host via ssh, rendezvous on
--master-addr/--master-port (env:// like the reference).

Child processes are process-group leaders; on launcher death they receive
SIGTERM (the reference vendored Horovod's safe_shell_exec for this —
start_new_session + terminate-on-exit covers the same orphan cleanup).
"""
import argparse
import os
import shlex
import signal
import subprocess
import sys


def launch_local(np_per_node, master_addr, master_port, node_rank, nnodes, rest):
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        f"--nnodes={nnodes}", f"--nproc-per-node={np_per_node}",
        f"--node-rank={node_rank}",
        f"--master-addr={master_addr}", f"--master-port={master_port}",
        os.path.join(os.path.dirname(os.path.abspath(__file__)), "train.py"),
    ] + rest
    proc = subprocess.Popen(cmd, start_new_session=True)

    def _fwd(sig, frame):
        try:
            os.killpg(proc.pid, signal.SIGTERM)
        except ProcessLookupError:
            pass

    signal.signal(signal.SIGTERM, _fwd)
    signal.signal(signal.SIGINT, _fwd)
    rc = proc.wait()
    return rc


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--np", type=int, default=8, help="processes (GPUs) per node")
    parser.add_argument("--hosts", type=str, default=None,
                        help="host1:np,host2:np for multi-node ssh fan-out")
    parser.add_argument("--master-addr", type=str, default="127.0.0.1")
    parser.add_argument("--master-port", type=int, default=29500)
    args, rest = parser.parse_known_args()

    if not args.hosts:
        # NODE_RANK/NNODES arrive via env when this is the per-host inner
        # launcher of an ssh fan-out (see below)
        node_rank = int(os.environ.get("NODE_RANK", "0"))
        nnodes = int(os.environ.get("NNODES", "1"))
        sys.exit(launch_local(args.np, args.master_addr, args.master_port,
                              node_rank, nnodes, rest))

    hosts = [h.split(":") for h in args.hosts.split(",")]
    procs = []
    for rank, (host, np_h) in enumerate(hosts):
        inner = (f"cd {shlex.quote(os.getcwd())} && "
                 f"{sys.executable} {shlex.quote(os.path.abspath(__file__))} "
                 f"--np {np_h} --master-addr {args.master_addr} "
                 f"--master-port {args.master_port} "
                 + " ".join(shlex.quote(r) for r in rest))
        env_inner = (f"MASTER_ADDR={args.master_addr} MASTER_PORT={args.master_port} "
                     f"NODE_RANK={rank} NNODES={len(hosts)} {inner}")
        if host in ("localhost", "127.0.0.1"):
            procs.append(subprocess.Popen(["bash", "-c", env_inner],
                                          start_new_session=True))
        else:
            procs.append(subprocess.Popen(["ssh", "-o", "BatchMode=yes", host,
                                           env_inner], start_new_session=True))

    def _fan(sig, frame):
        for p in procs:
            try:
                os.killpg(p.pid, signal.SIGTERM)
            except ProcessLookupError:
                pass

    signal.signal(signal.SIGTERM, _fan)
    signal.signal(signal.SIGINT, _fan)
    rc = 0
    for p in procs:
        rc |= p.wait()
    sys.exit(rc)


if __name__ == "__main__":
    main()
