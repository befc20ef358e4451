"""conflux_amd — Python bindings for the MI355X-native CONFLUX LU engine.

The PRODUCT is the native library (libconflux_lu.so: C++ host + HIP/gfx950
kernels + RCCL) and the conflux_miniapp CLI.  This module is a thin ctypes
loader used by bench.py and the tests.  It FAILS LOUDLY if the HIP extension
is missing — there is no CPU fallback anywhere in the product path.
"""
import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "libconflux_lu.so")

UID_BYTES = 128


class ConfluxLuError(RuntimeError):
    pass


def _load():
    if not os.path.exists(_LIB_PATH):
        raise ConfluxLuError(
            f"HIP engine not built: {_LIB_PATH} missing. "
            "Run `make -C conflux_amd` (or __graft_entry__.build()). "
            "The product has no CPU fallback.")
    lib = ctypes.CDLL(_LIB_PATH)
    lib.conflux_lu_create.argtypes = [ctypes.c_int] * 7 + [
        ctypes.c_char_p, ctypes.POINTER(ctypes.c_void_p)]
    lib.conflux_lu_make_uid.argtypes = [ctypes.c_char_p]
    lib.conflux_lu_init_matrix.argtypes = [ctypes.c_void_p, ctypes.c_uint64]
    lib.conflux_lu_init_matrix_spd.argtypes = [ctypes.c_void_p, ctypes.c_uint64]
    lib.conflux_chol_factor.argtypes = [ctypes.c_void_p,
                                        ctypes.POINTER(ctypes.c_double)]
    lib.conflux_lu_set_matrix_local.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    lib.conflux_lu_set_matrix_sim.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                              ctypes.c_void_p]
    lib.conflux_lu_store_factors.argtypes = [ctypes.c_void_p, ctypes.c_int]
    lib.conflux_lu_set_pivoting.argtypes = [ctypes.c_void_p, ctypes.c_int]
    lib.conflux_lu_factor.argtypes = [ctypes.c_void_p,
                                      ctypes.POINTER(ctypes.c_double)]
    lib.conflux_lu_get_factors.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                           ctypes.c_void_p]
    lib.conflux_lu_get_factors_sim.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                               ctypes.c_void_p, ctypes.c_void_p]
    lib.conflux_lu_dims.argtypes = [ctypes.c_void_p] + [
        ctypes.POINTER(ctypes.c_int)] * 6
    lib.conflux_lu_kernel_stats.argtypes = [
        ctypes.c_void_p, ctypes.c_int, ctypes.POINTER(ctypes.c_double),
        ctypes.POINTER(ctypes.c_long), ctypes.POINTER(ctypes.c_double)]
    lib.conflux_lu_validate.argtypes = [ctypes.c_void_p,
                                        ctypes.POINTER(ctypes.c_double)]
    lib.conflux_chol_validate.argtypes = [ctypes.c_void_p,
                                          ctypes.POINTER(ctypes.c_double)]
    lib.conflux_lu_destroy.argtypes = [ctypes.c_void_p]
    lib.conflux_lu_build_info.restype = ctypes.c_char_p
    lib.conflux_lu_debug_dgemm.argtypes = [
        ctypes.c_int, ctypes.c_longlong, ctypes.c_int, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_void_p]
    lib.conflux_lu_debug_getrf.argtypes = [ctypes.c_int, ctypes.c_int,
                                           ctypes.c_void_p, ctypes.c_void_p]
    lib.conflux_lu_debug_dgemm_bench.argtypes = [
        ctypes.c_int, ctypes.c_longlong, ctypes.c_int, ctypes.c_int,
        ctypes.POINTER(ctypes.c_double)]
    lib.conflux_lu_debug_trsm.argtypes = [
        ctypes.c_int, ctypes.c_int, ctypes.c_longlong, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_void_p]
    return lib


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = _load()
    return _lib


def _chk(rc, what):
    if rc != 0:
        raise ConfluxLuError(f"{what} failed: rc={rc}")


class Engine:
    """One factorization context.  rank=-1, world=P -> single-process
    multi-rank SIMULATION on one GPU (full choreography, D2D transport);
    rank>=0 -> one process per GPU over RCCL."""

    def __init__(self, N, v, Px, Py, Pz, rank=-1, world=None, uid=None):
        import numpy as np  # noqa: F401
        self.N, self.v, self.Px, self.Py, self.Pz = N, v, Px, Py, Pz
        P = Px * Py * Pz
        world = P if world is None else world
        self.world = world
        self.sim = rank < 0
        self._h = ctypes.c_void_p()
        _chk(lib().conflux_lu_create(N, v, Px, Py, Pz, rank, world, uid,
                                     ctypes.byref(self._h)), "create")
        d = [ctypes.c_int() for _ in range(6)]
        _chk(lib().conflux_lu_dims(self._h, *[ctypes.byref(x) for x in d]),
             "dims")
        self.Ml, self.Nl, self.Nt, self.nlayr, self.M, self.Npad = \
            (x.value for x in d)

    @staticmethod
    def make_uid():
        buf = ctypes.create_string_buffer(UID_BYTES)
        _chk(lib().conflux_lu_make_uid(buf), "make_uid")
        return buf.raw

    def init_matrix(self, seed=42):
        _chk(lib().conflux_lu_init_matrix(self._h, seed), "init_matrix")

    def set_matrix_global(self, A):
        """Distribute a full N x N numpy matrix (sim mode only)."""
        import numpy as np
        assert self.sim
        v = self.v
        for pi in range(self.Px):
            for pj in range(self.Py):
                loc = np.zeros((self.Ml, self.Nl))
                for lti in range(self.Ml // v):
                    for ltj in range(self.Nl // v):
                        gti, gtj = lti * self.Px + pi, ltj * self.Py + pj
                        loc[lti * v:(lti + 1) * v, ltj * v:(ltj + 1) * v] = \
                            A[gti * v:(gti + 1) * v, gtj * v:(gtj + 1) * v]
                loc = np.ascontiguousarray(loc)
                for pk in range(self.Pz):
                    g = (pi * self.Py + pj) * self.Pz + pk
                    buf = loc if pk == 0 else None
                    _chk(lib().conflux_lu_set_matrix_sim(
                        self._h, g,
                        buf.ctypes.data_as(ctypes.c_void_p) if buf is not None
                        else None), "set_matrix_sim")

    def set_matrix_local(self, local):
        ptr = None
        if local is not None:
            import numpy as np
            local = np.ascontiguousarray(local, dtype=np.float64)
            ptr = local.ctypes.data_as(ctypes.c_void_p)
        _chk(lib().conflux_lu_set_matrix_local(self._h, ptr), "set_matrix")

    def store_factors(self, enable):
        _chk(lib().conflux_lu_store_factors(self._h, int(enable)), "store")

    def set_pivoting(self, mode):
        """1 = tournament (default), 0 = none (EmptyPivot fast path for
        diagonally dominant inputs; perm stays identity)."""
        _chk(lib().conflux_lu_set_pivoting(self._h, int(mode)), "pivoting")

    def factor(self):
        ms = ctypes.c_double()
        _chk(lib().conflux_lu_factor(self._h, ctypes.byref(ms)), "factor")
        return ms.value

    def init_matrix_spd(self, seed=42):
        _chk(lib().conflux_lu_init_matrix_spd(self._h, seed), "init_spd")

    def factor_cholesky(self):
        ms = ctypes.c_double()
        _chk(lib().conflux_chol_factor(self._h, ctypes.byref(ms)), "chol")
        return ms.value

    def validate(self):
        """Device-side ||PA - LU||_F / ||A||_F of the last factorization
        (conflux_lu_validate; stripe-streamed).  COLLECTIVE when world > 1:
        every rank must call; all ranks return the broadcast residual."""
        r = ctypes.c_double()
        _chk(lib().conflux_lu_validate(self._h, ctypes.byref(r)), "validate")
        return r.value

    def validate_cholesky(self):
        """Device-side ||A - L L^T||_F / ||A||_F of the last Cholesky
        factorization.  COLLECTIVE when world > 1 (like validate)."""
        r = ctypes.c_double()
        _chk(lib().conflux_chol_validate(self._h, ctypes.byref(r)),
             "chol_validate")
        return r.value

    def get_perm(self):
        import numpy as np
        perm = np.zeros(self.M, dtype=np.int32)
        _chk(lib().conflux_lu_get_factors(self._h, None,
                                          perm.ctypes.data_as(ctypes.c_void_p)),
             "get_factors")
        return perm

    def get_F_global(self):
        """Assemble the global factored matrix F (pivoted rows) — sim mode."""
        import numpy as np
        assert self.sim
        v = self.v
        F = np.zeros((self.Npad, self.Npad))
        loc = np.zeros((self.Ml, self.Nl))
        for pi in range(self.Px):
            for pj in range(self.Py):
                g = (pi * self.Py + pj) * self.Pz + 0
                _chk(lib().conflux_lu_get_factors_sim(
                    self._h, g, loc.ctypes.data_as(ctypes.c_void_p), None),
                    "get_factors_sim")
                for lti in range(self.Ml // v):
                    for ltj in range(self.Nl // v):
                        gti, gtj = lti * self.Px + pi, ltj * self.Py + pj
                        F[gti * v:(gti + 1) * v, gtj * v:(gtj + 1) * v] = \
                            loc[lti * v:(lti + 1) * v, ltj * v:(ltj + 1) * v]
        return F

    def get_F_local(self):
        import numpy as np
        F = np.zeros((self.Ml, self.Nl))
        _chk(lib().conflux_lu_get_factors(self._h,
                                          F.ctypes.data_as(ctypes.c_void_p),
                                          None), "get_factors")
        return F

    def kernel_stats(self):
        out = {}
        names = {0: "dgemm_trailing", 1: "panel_getrf", 2: "trsm", 3: "rowmove"}
        for k, name in names.items():
            s = ctypes.c_double()
            n = ctypes.c_long()
            f = ctypes.c_double()
            _chk(lib().conflux_lu_kernel_stats(self._h, k, ctypes.byref(s),
                                               ctypes.byref(n),
                                               ctypes.byref(f)), "stats")
            out[name] = dict(seconds=s.value, launches=n.value, flops=f.value)
        return out

    def close(self):
        if self._h:
            lib().conflux_lu_destroy(self._h)
            self._h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
