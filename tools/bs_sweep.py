import json, subprocess, sys
for b in (64, 256, 1024, 4096, 16384, 65536):
    out = subprocess.run([sys.executable, "bench.py", "--gpus", "1",
                          "--steps", "50", "--warmup", "10",
                          "--batch-size", str(b)],
                         capture_output=True, text=True, timeout=600)
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    r = json.loads(line)
    print(f"lenet bs={b:6d}: {round(r['value']):>10} img/s "
          f"{r['ms_per_step']*1000:8.1f} us/step")
