"""DDP bucket-size sweep on a multi-GPU node (ROADMAP item 3): runs bench.py
under torchrun for each --bucket-mb and prints a table. Usage:
    python tools/bucket_sweep.py --gpus 8 [--model resnet50] [--sizes 16 32 64 128]
"""
import argparse
import json
import subprocess
import sys


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=8)
    p.add_argument("--model", default="resnet50")
    p.add_argument("--sizes", type=float, nargs="+",
                   default=[16.0, 32.0, 64.0, 128.0])
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=10)
    args = p.parse_args()

    rows = []
    for mb in args.sizes:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={args.gpus}", "--master-addr", "127.0.0.1",
               "--master-port", "29533", "bench.py", "--gpus", str(args.gpus),
               "--steps", str(args.steps), "--warmup", str(args.warmup),
               "--model", args.model, "--bucket-mb", str(mb)]
        r = subprocess.run(cmd, capture_output=True, text=True, timeout=1800)
        line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
        if r.returncode != 0 or not line:
            print(f"bucket {mb} MiB: FAILED rc={r.returncode}\n"
                  f"{r.stderr[-500:]}")
            continue
        j = json.loads(line[-1])
        rows.append((mb, j["value"], j["ms_per_step"]))
        print(f"bucket {mb:6.1f} MiB: {j['value']:9.1f} img/s  "
              f"{j['ms_per_step']:.2f} ms/step")
    if rows:
        best = max(rows, key=lambda r: r[1])
        print(f"best: {best[0]} MiB at {best[1]:.1f} img/s")


if __name__ == "__main__":
    main()
