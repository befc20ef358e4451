import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import numpy as np
from conflux_amd import Engine
from oracle import residual_check
# regenerate the fixture first (CPU, ~10 s):
#   python3 -c "from oracle import *; import numpy as np; A=gen_matrix(2048); \
#      r=lu_oracle(A, Params(2048,256,2,2,2)); \
#      np.savez_compressed('tools/oracle2048.npz', perm=r['perm'], F=r['F'], A=A)"
d = np.load(os.path.join(os.path.dirname(__file__), "oracle2048.npz"))
A, perm_ref, F_ref = d["A"], d["perm"], d["F"]
with Engine(2048, 256, 2, 2, 2, rank=-1) as e:
    e.store_factors(True)
    e.set_matrix_global(A)
    e.factor()
    perm = e.get_perm()
    F = e.get_F_global()
print("perm bit-exact:", np.array_equal(perm, perm_ref))
print("F maxdiff:", np.abs(F - F_ref).max())
print("residual:", residual_check(A, perm, F))
assert np.array_equal(perm, perm_ref)
assert np.abs(F - F_ref).max() < 1e-10
assert residual_check(A, perm, F) < 1e-13
print("BIGSIM OK")
