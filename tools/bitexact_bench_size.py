"""One-off deep verification: pivots bit-exact vs the oracle restatement at
the FULL bench size N=16384, v=512 (the oracle needs ~5 min of host CPU, so
this runs as a manual job rather than a suite test).  Also checks factor
agreement and the size-independent residual."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import numpy as np

from conflux_amd import Engine
from oracle import Params, gen_matrix, lu_oracle, residual_check

N, v = 16384, 512
t0 = time.time()
A = gen_matrix(N)
print(f"gen {time.time()-t0:.0f}s", flush=True)
t0 = time.time()
ref = lu_oracle(A, Params(N, v, 1, 1, 1))
print(f"oracle {time.time()-t0:.0f}s", flush=True)
t0 = time.time()
with Engine(N, v, 1, 1, 1, rank=-1) as e:
    e.store_factors(True)
    e.set_matrix_global(A)
    e.factor()
    perm = e.get_perm()
    F = e.get_F_global()
print(f"engine {time.time()-t0:.0f}s", flush=True)
ok = np.array_equal(perm, ref["perm"])
dmax = float(np.abs(F - ref["F"]).max())
res = float(residual_check(A, perm, F))
print(f"pivots bit-exact: {ok}; max|dF| = {dmax:.3e}; residual = {res:.3e}")
assert ok and res < 1e-13
print("BITEXACT16K OK")
