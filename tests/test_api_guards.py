"""C-ABI argument guards (no GPU needed: rejected before any HIP call)."""
import ctypes

import pytest


@pytest.fixture(scope="module")
def lib():
    import conflux_amd
    try:
        return conflux_amd.lib()
    except conflux_amd.ConfluxLuError:
        pytest.skip("HIP engine not built")


@pytest.mark.parametrize("args", [
    (256, 32, 2, 4, 1),   # Px != Py
    (256, 32, 3, 3, 1),   # non-power-of-two Px
    (256, 33, 1, 1, 2),   # v % Pz != 0
    (64, 64, 1, 1, 1),    # Ml < 2v (reference buffer-sizing envelope)
    (0, 32, 1, 1, 1),     # degenerate
])
def test_create_rejects_unsupported_grids(lib, args):
    N, v, Px, Py, Pz = args
    h = ctypes.c_void_p()
    rc = lib.conflux_lu_create(N, v, Px, Py, Pz, -1, Px * Py * Pz, None,
                               ctypes.byref(h))
    assert rc == -1  # CONFLUX_LU_EARG


def test_create_rounds_n_up(lib):
    """Non-multiple N is rounded up like the reference (lu_params.hpp:67-71),
    not rejected.  On a GPU-less box create proceeds past the argument
    checks and fails at the first HIP call instead (EHIP, not EARG)."""
    h = ctypes.c_void_p()
    rc = lib.conflux_lu_create(250, 32, 1, 1, 1, -1, 1, None,
                               ctypes.byref(h))
    assert rc in (0, -2)
    if rc == 0:
        d = [ctypes.c_int() for _ in range(6)]
        lib.conflux_lu_dims(h, *[ctypes.byref(x) for x in d])
        assert d[5].value == 256  # N_padded
        lib.conflux_lu_destroy(h)


def test_header_symbols_all_exported(lib):
    """Every symbol include/conflux_lu.h declares is exported by the built
    library (no compute — pure dlsym checks)."""
    import re
    import os
    hdr = open(os.path.join(os.path.dirname(__file__), "..", "include",
                            "conflux_lu.h")).read()
    decls = sorted(set(re.findall(r"\b(conflux_\w+)\s*\(", hdr)))
    assert len(decls) >= 15
    missing = [d for d in decls if not hasattr(lib, d)]
    assert not missing, f"undefined in libconflux_lu.so: {missing}"
    assert b"MI355X" in lib.conflux_lu_build_info()
