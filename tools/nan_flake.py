import os, subprocess, sys
import torch
sys.path.insert(0, ".")

def run(tag, sync="0", n=4, **extra_env):
    bad = 0
    for i in range(n):
        env = dict(os.environ, FAA_BENCH_IMGS="2048", FAA_BENCH_SAVE="/tmp/f.pt",
                   FAA_BENCH_SYNC_UPLOAD=sync, **extra_env)
        r = subprocess.run([sys.executable, "bench.py", "--steps", "1", "--warmup", "2"],
                           env=env, capture_output=True, text=True, timeout=280)
        assert r.returncode == 0, r.stderr[-400:]
        t = torch.load("/tmp/f.pt").float()
        if torch.isnan(t).any():
            bad += 1
    print(f"{tag}: {bad}/{n} runs NaN")

import sys
specs = sys.argv[1:] or ["baseline:"]
for spec in specs:
    name, _, envs = spec.partition(":")
    extra = dict(kv.split("=") for kv in envs.split(",") if kv)
    run(name, n=6, **extra)
