import os, subprocess, sys
import torch
sys.path.insert(0, ".")

def run(tag, *extra):
    env = dict(os.environ, FAA_BENCH_IMGS="2048", FAA_BENCH_SAVE=f"/tmp/n_{tag}.pt")
    r = subprocess.run([sys.executable, "bench.py", "--steps", "1", "--warmup", "2",
                        *extra], env=env, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-1500:]
    t = torch.load(f"/tmp/n_{tag}.pt").float()
    n_nan = torch.isnan(t).sum().item()
    n_inf = torch.isinf(t).sum().item()
    print(f"{tag}: numel={t.numel()} nan={n_nan} inf={n_inf} max={t.nan_to_num().abs().max().item():.4f}")
    if n_nan:
        idx = torch.nonzero(torch.isnan(t)).flatten()
        print("  first nan offsets:", idx[:8].tolist(), " last:", idx[-4:].tolist())
    return t

run("eager_flat", "--graphs", "0")
run("graph_flat", "--graphs", "1", "--grad-mode", "flat")
run("graph_gath", "--graphs", "1", "--grad-mode", "gather")
# more steps eager
env = dict(os.environ, FAA_BENCH_IMGS="2048", FAA_BENCH_SAVE="/tmp/n_e50.pt")
r = subprocess.run([sys.executable, "bench.py", "--steps", "50", "--warmup", "2", "--graphs", "0"],
                   env=env, capture_output=True, text=True, timeout=280)
t = torch.load("/tmp/n_e50.pt").float()
print("eager 50 steps: nan=", torch.isnan(t).sum().item(), "max=", t.nan_to_num().abs().max().item())
