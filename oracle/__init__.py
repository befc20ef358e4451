# ============================================================================
# ORACLE — TEST INFRASTRUCTURE ONLY.
#
# This package is a CPU (numpy/scipy) restatement of the reference
# eth-cscs/conflux LU hot path (src/conflux/lu/conflux_opt.hpp LU_rep<T>),
# used exclusively as the parity checker for the MI355X-native engine.
#
# Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
# import, call, link or execute anything in here — and there only as the
# checker / reported CPU baseline, never as the thing measured or shipped.
# The product path (conflux_amd + libconflux_lu.so) never routes through
# this package and fails loudly if its HIP extension is missing.
#
# Parity pinning: the oracle is checked against
#   - the reference's own hard-coded known-answer matrices
#     (lu_params.hpp:157-363), captured as fixtures under tests/golden/,
#   - LAPACK partial pivoting (scipy.linalg.lu_factor) on the 1x1x1 grid,
#   - the reference's own LU loop compiled in-container (oracle/_ref,
#     recipe under oracle/ref_build/) on small grids under mpiexec.
# ============================================================================
from .lu_oracle import Params, lu_oracle, lu_nopivot, residual_check
from .gen_input import gen_matrix
