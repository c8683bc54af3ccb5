"""Isolate the torch-process step overhead: drive the extern-C launchers
via ctypes on raw hipMalloc buffers, with and without torch imported.

Usage: python tools/ctypes_probe.py [--torch]
"""
import ctypes
import glob
import sys
import time

USE_TORCH = "--torch" in sys.argv
if USE_TORCH:
    import torch  # noqa: F401 - presence of the runtime is the variable
    torch.zeros(1, device="cuda")  # force full HIP init
    sys.path.insert(0, ".")
    import parallel_cnn_amd._C as _C_mod  # loads with torch's libs
    lib = ctypes.CDLL(_C_mod.__file__)
    hip_path = [l.split()[-1] for l in open("/proc/self/maps")
                if "amdhip64" in l][0]
    hip = ctypes.CDLL(hip_path)
else:
    # load torch's lib deps first so _C resolves, without torch python init
    for dep in ("libc10.so", "libtorch_cpu.so"):
        try:
            ctypes.CDLL("/usr/local/lib/python3.10/dist-packages/torch/lib/"
                        + dep, mode=ctypes.RTLD_GLOBAL)
        except OSError:
            pass
    lib = ctypes.CDLL(glob.glob("parallel_cnn_amd/_C.*.so")[0],
                      mode=ctypes.RTLD_GLOBAL)
    hip_path = [l.split()[-1] for l in open("/proc/self/maps")
                if "amdhip64" in l][0]
    hip = ctypes.CDLL(hip_path)

def hipcheck(r):
    assert r == 0, r

def dmalloc(nbytes):
    p = ctypes.c_void_p()
    hipcheck(hip.hipMalloc(ctypes.byref(p), ctypes.c_size_t(nbytes)))
    hipcheck(hip.hipMemset(p, 0, ctypes.c_size_t(nbytes)))
    return p

B = 64
N_PARAMS = 2343
x = dmalloc(B * 784 * 2)          # bf16
params = dmalloc(N_PARAMS * 4)
grads = dmalloc(N_PARAMS * 4)
a1 = dmalloc(B * 3456 * 2)
a2 = dmalloc(B * 216 * 2)
y = dmalloc(B * 10 * 4)
dz = dmalloc(B * 10 * 4)
dz2 = dmalloc(B * 216 * 4)
dz1 = dmalloc(B * 3456 * 2)
labels = dmalloc(B * 4)
loss = dmalloc(4)

fwd = lib.pcnn_launch_fwdbwd
wgr = lib.pcnn_launch_wgrad
upd = lib.pcnn_launch_update
NULLS = ctypes.c_void_p(0)

def step():
    hipcheck(fwd(x, params, a1, a2, y, dz, dz2, dz1, labels, loss, NULLS,
                 B, 1, 0, NULLS))
    hipcheck(wgr(x, a1, a2, dz, dz2, dz1, grads, B, 1, 0, NULLS))
    upd.restype = ctypes.c_int
    hipcheck(upd(params, grads, ctypes.c_float(0.001), NULLS))

for n in (200, 2000):
    t0 = time.perf_counter()
    for _ in range(n):
        step()
    hipcheck(hip.hipDeviceSynchronize())
    dt = time.perf_counter() - t0
    if n == 2000:
        tag = "with-torch" if USE_TORCH else "no-torch"
        print(f"{tag}: {dt / n * 1e6:.1f} us/step")
