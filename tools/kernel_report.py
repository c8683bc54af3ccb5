"""First-class per-kernel / per-layer time report (VERDICT r1 missing #3).

The reference publishes per-layer epoch times (report Tables 4-7,
Sequential/Main.cpp:51-54).  This framework fuses whole phases into
single kernels, so host-side per-layer timers cannot exist; the honest
equivalent is the device-side per-kernel split, which this tool produces
in one command instead of a hand-written rocprofv3 invocation:

    python tools/kernel_report.py [--model lenet5|deepcnn]
        [--batch-size N] [--steps K]

Runs bench.py under `rocprofv3 --kernel-trace --stats`, parses the
kernel stats, maps kernels to the reference's layer vocabulary, and
prints a per-step table.  Needs a GPU box with rocprofv3 (errors out
clearly otherwise).  `python -m parallel_cnn_amd.train --profile` gives
the coarser host-side phase split without rocprofv3.
"""
from __future__ import annotations

import argparse
import csv
import os
import shutil
import subprocess
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

# kernel -> (layer/phase, note) in the reference's table vocabulary
KERNEL_MAP = [
    # LeNet path (3 fused kernels)
    ("pcnn::k_fwdbwd", "conv+pool+fc fwd & bwd-data",
     "fused: the reference's fp_c1..bp_preact_c1 chain"),
    ("pcnn::k_wgrad", "all weight grads",
     "bp_weight_f/s1/c1 + biases, one kernel"),
    ("pcnn::k_update", "SGD update", "apply_grad x3 + grad zero"),
    ("pcnn::k_train_steps", "fused whole-step loop", ""),
    # DeepCNN path
    ("k_gemm_smallk", "conv fwd (stage 0, small-K GEMM)", ""),
    ("k_gemm<", "conv fwd GEMM + dgrad-as-conv", "implicit im2col"),
    ("k_wgrad_multi", "ALL conv weight grads (one launch)", ""),
    ("k_wgrad_gemm", "conv weight grad (+bias colsum)", ""),
    ("k_split_epi", "GEMM split-K combine + epilogues", ""),
    ("k_im2col", "im2col (stage 0)", ""),
    ("k_pool_wbwd", "pool wgrad+bwd (fused)", ""),
    ("k_pool_fwd", "pool fwd", ""),
    ("k_pool_bwd", "pool bwd", ""),
    ("k_colsum", "conv bias grad", ""),
    ("k_col2im", "col2im + sigmoid bwd", ""),
    ("k_fc_fwd", "fc fwd (+loss, +fused fc bwd-data)", ""),
    ("k_fc_bwd", "fc bwd-data", ""),
    ("k_fc_wgrad", "fc weight grad", ""),
    ("k_update_cast_all", "SGD update + weight cast (fused)", ""),
    ("k_cast_wt", "weight cast", ""),
    ("k_update_n", "SGD update", ""),
    ("k_pad_channels", "input channel pad", ""),
    ("k_remap_dw8", "padded-dW remap", ""),
]


def classify(name: str):
    for key, layer, note in KERNEL_MAP:
        if key in name:
            return layer, note
    if "at::native" in name or "rocclr" in name:
        return None, None  # torch housekeeping outside the step
    return name.split("(")[0][-40:], ""


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="lenet5",
                   choices=["lenet5", "deepcnn"])
    p.add_argument("--batch-size", type=int, default=None)
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--warmup", type=int, default=20)
    args = p.parse_args()
    bs = args.batch_size or (256 if args.model == "deepcnn" else 64)

    if shutil.which("rocprofv3") is None:
        sys.exit("rocprofv3 not found — run on a GPU box "
                 "(ROCm bin on PATH)")
    out = tempfile.mkdtemp(prefix="pcnn_kreport_", dir="/tmp")
    env = dict(os.environ, TMPDIR="/tmp")
    cmd = ["rocprofv3", "--kernel-trace", "--stats",
           "--output-format", "csv", "-d", out, "-o", "r", "--",
           sys.executable, os.path.join(REPO, "bench.py"),
           "--model", args.model, "--batch-size", str(bs),
           "--steps", str(args.steps), "--warmup", str(args.warmup)]
    r = subprocess.run(cmd, cwd="/tmp", env=env, capture_output=True,
                       text=True, timeout=900)
    if r.returncode != 0:
        sys.exit(f"profiling run failed:\n{r.stdout[-1500:]}"
                 f"\n{r.stderr[-1500:]}")
    stats = os.path.join(out, "r_kernel_stats.csv")
    if not os.path.exists(stats):
        sys.exit(f"no kernel stats at {stats}")

    # warmup + timed-region repeats both hit the device; the update
    # kernel runs exactly once per training step — use its call count as
    # the step normalizer (fall back to the max count)
    rows = []
    with open(stats) as f:
        for row in csv.DictReader(f):
            layer, note = classify(row["Name"])
            if layer is None:
                continue
            rows.append((layer, note, int(row["Calls"]),
                         float(row["TotalDurationNs"]) / 1e3))
    upd = [c for l, n, c, t in rows if "update" in l.lower()]
    steps = upd[0] if upd else max((c for l, n, c, t in rows), default=1)
    print(f"Per-kernel device-time report — {args.model}, bs={bs} "
          f"({steps} profiled steps)")
    print(f"{'layer / phase':44s} {'calls/step':>10s} {'us/step':>9s}")
    for layer, note, calls, tot_us in sorted(rows, key=lambda r: -r[3]):
        print(f"{layer:44s} {calls / steps:>10.2f} {tot_us / steps:>9.2f}"
              + (f"   [{note}]" if note else ""))
    print(f"{'TOTAL (device)':44s} {'':>10s} "
          f"{sum(t for *_x, t in rows) / steps:>9.2f}")
    bench_line = [l for l in r.stdout.splitlines() if l.startswith("{")]
    if bench_line:
        print("\nbench:", bench_line[-1])
    return 0


if __name__ == "__main__":
    sys.exit(main())
