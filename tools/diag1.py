import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import ctypes, numpy as np, conflux_amd
lib = conflux_amd.lib()
def dgemm(A,B,C):
    C1 = C.copy()
    rc = lib.conflux_lu_debug_dgemm(A.shape[0], B.shape[1], A.shape[1],
        A.ctypes.data_as(ctypes.c_void_p), B.ctypes.data_as(ctypes.c_void_p),
        C1.ctypes.data_as(ctypes.c_void_p))
    print("rc=", rc)
    return C1
M,N,K = 16,16,4
A = np.arange(M*K,dtype=np.float64).reshape(M,K)+1
B = (np.arange(K*N,dtype=np.float64).reshape(K,N)+1)*0.001
C = np.zeros((M,N))
C1 = dgemm(A,B,C)
ref = -A@B
print("max|C1-ref| =", np.abs(C1-ref).max())
print("C1[:2,:4] =", C1[:2,:4])
print("ref[:2,:4] =", ref[:2,:4])
print("C1 transposed match?", np.abs(C1-ref.T).max() if M==N else "n/a")
print("nonzero count:", np.count_nonzero(C1))
