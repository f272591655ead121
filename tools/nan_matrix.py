import os, subprocess, sys
import torch
sys.path.insert(0, ".")

def run(tag, imgs, steps, warmup, grad=False, mode="flat"):
    env = dict(os.environ, FAA_BENCH_IMGS=str(imgs), FAA_BENCH_SAVE=f"/tmp/x_{tag}.pt")
    if grad:
        env["FAA_BENCH_SAVE_GRAD"] = "1"
    r = subprocess.run([sys.executable, "bench.py", "--steps", str(steps),
                        "--warmup", str(warmup), "--grad-mode", mode],
                       env=env, capture_output=True, text=True, timeout=280)
    if r.returncode != 0:
        print(tag, "FAILED", r.stderr[-300:]); return
    t = torch.load(f"/tmp/x_{tag}.pt").float()
    print(f"{tag:24s} nan={torch.isnan(t).sum().item():8d} max={t.nan_to_num().abs().max().item():.3e}")

run("i2048_s1_w2", 2048, 1, 2)
run("i512_s1_w2", 512, 1, 2)
run("i50000_s1_w2", 50000, 1, 2)
run("i2048_s50_w2", 2048, 50, 2)
run("i2048_s1_w10", 2048, 1, 10)
run("i2048_s1_w2_grad", 2048, 1, 2, grad=True)
run("i512_s300_w10", 512, 300, 10)
