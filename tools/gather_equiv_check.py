"""Compare gather vs flat grad modes: fp32 masters after ONE stepped replay
from identical state/input (flat mode's grad buffer holds post-clip values,
so grads themselves aren't comparable; masters are)."""
import os
import subprocess
import sys

import torch

sys.path.insert(0, ".")


def run(mode, tag):
    env = dict(os.environ, FAA_BENCH_IMGS="2048", FAA_BENCH_SAVE=f"/tmp/m_{tag}.pt")
    r = subprocess.run([sys.executable, "bench.py", "--steps", "1", "--warmup", "2",
                        "--grad-mode", mode], env=env, capture_output=True, text=True,
                       timeout=280)
    assert r.returncode == 0, r.stderr[-2000:]
    return torch.load(f"/tmp/m_{tag}.pt").float()


a = run("flat", "f1")
a2 = run("flat", "f2")
b = run("gather", "g1")
print("flat-vs-flat  max abs diff:", (a - a2).abs().max().item())
d = (a - b).abs().max().item()
print("flat-vs-gather max abs diff:", d)
rerun_noise = (a - a2).abs().max().item()
assert d <= max(5 * rerun_noise, 1e-4), "gather diverges beyond rerun noise"
print("EQUIVALENT (within split-K atomics rerun noise)")
