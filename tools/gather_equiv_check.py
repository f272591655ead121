"""Train 40 graph steps in flat and gather grad modes with identical host RNG
and compare the fp32 masters: must match to bf16-grad determinism (exact)."""
import os
import subprocess
import sys

import torch

sys.path.insert(0, ".")


def run(mode):
    env = dict(os.environ, FAA_BENCH_IMGS="2048", FAA_BENCH_SAVE=f"/tmp/m_{mode}.pt")
    r = subprocess.run([sys.executable, "bench.py", "--steps", "40", "--warmup", "5",
                        "--grad-mode", mode], env=env, capture_output=True, text=True,
                       timeout=280)
    assert r.returncode == 0, r.stderr[-1500:]


run("flat")
run("gather")
a = torch.load("/tmp/m_flat.pt")
b = torch.load("/tmp/m_gather.pt")
d = (a - b).abs().max().item()
print("max |master_flat - master_gather| =", d)
assert d == 0.0, "gather mode diverged from flat mode"
print("EQUIVALENT")
