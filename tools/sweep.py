"""Parameter sweeps of bench.py on one GPU (run via gpurun)."""
import json
import subprocess
import sys


def run(args):
    out = subprocess.run([sys.executable, "bench.py"] + args,
                         capture_output=True, text=True, timeout=600)
    for line in out.stdout.strip().splitlines():
        if line.startswith("{"):
            return json.loads(line)
    raise RuntimeError(f"no JSON from bench {args}: {out.stdout} {out.stderr}")


def main():
    rows = []
    for chunk in (1, 2, 4, 8, 16, 32, 64):
        r = run(["--gpus", "1", "--steps", "500", "--warmup", "50",
                 "--wgrad-chunk", str(chunk)])
        rows.append(("bf16", chunk, r["value"], r["ms_per_step"] * 1e3))
        print(f"bf16 chunk={chunk:3d}: {r['value']:10.0f} img/s  "
              f"{r['ms_per_step']*1e3:7.2f} us/step", flush=True)
    for chunk in (2, 8):
        r = run(["--gpus", "1", "--steps", "500", "--warmup", "50",
                 "--act-dtype", "fp32", "--wgrad-chunk", str(chunk)])
        print(f"fp32 chunk={chunk:3d}: {r['value']:10.0f} img/s  "
              f"{r['ms_per_step']*1e3:7.2f} us/step", flush=True)
    for bs in (256, 1024, 4096):
        r = run(["--gpus", "1", "--steps", "200", "--warmup", "20",
                 "--batch-size", str(bs), "--wgrad-chunk", "8"])
        print(f"bf16 bs={bs:5d} chunk=8: {r['value']:10.0f} img/s  "
              f"{r['ms_per_step']*1e3:7.2f} us/step", flush=True)


if __name__ == "__main__":
    main()
