import sys, os, ctypes
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import conflux_amd
lib = conflux_amd.lib()
shapes = [(16384,16384,512), (8192,8192,512), (16384,16384,256), (4096,4096,512), (8192,8192,256)]
if len(sys.argv) > 1:
    shapes = [tuple(int(x) for x in sys.argv[1].split(","))]
for (M,N,K) in shapes:
    tf = ctypes.c_double()
    rc = lib.conflux_lu_debug_dgemm_bench(M, N, K, 3, ctypes.byref(tf))
    print(f"M={M} N={N} K={K}: {tf.value:.2f} TF/s (rc={rc})")
