import sys, os, ctypes, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import numpy as np, conflux_amd
lib = conflux_amd.lib()
rng = np.random.default_rng(5)
# warmup: first kernel launch in a fresh process pays code-object load
_P = np.ascontiguousarray(5 + rng.random((1024, 512)))
_ip = np.zeros(512, dtype=np.int32)
lib.conflux_lu_debug_getrf(1024, 512, _P.ctypes.data_as(ctypes.c_void_p),
                           _ip.ctypes.data_as(ctypes.c_void_p))
for (n, v) in [(16384, 512), (32768, 512), (1024, 512)]:
    P = np.ascontiguousarray(5 + rng.random((n, v)))
    ipiv = np.zeros(v, dtype=np.int32)
    t0 = time.perf_counter()
    rc = lib.conflux_lu_debug_getrf(n, v, P.ctypes.data_as(ctypes.c_void_p), ipiv.ctypes.data_as(ctypes.c_void_p))
    dt = time.perf_counter() - t0
    print(f"getrf n={n} v={v}: {dt*1e3:.1f} ms ({dt*1e6/v:.2f} us/col) rc={rc}")
