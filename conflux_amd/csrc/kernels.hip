// kernels.hip — CDNA4 (gfx950) kernels of the MI355X-native CONFLUX LU engine.
//
// Re-expressions of the reference's compute units (SURVEY.md §2 checklist):
//   k_dgemm_f64        <- trailing-update cblas_dgemm (conflux_opt.hpp:1628-1633)
//                         and the panel-update GEMMs of blocked getrf/TRSM;
//                         hand-written v_mfma_f64_16x16x4_f64, LDS-staged.
//   k_panel_factor     <- LAPACKE_dgetrf partial pivoting
//                         (conflux_opt.hpp:143-166 LUP): persistent sub-panel
//                         kernel, LDS-resident rows, grid-wide first-max
//                         argmax per column via an agent-scope
//                         release/acquire slab handshake.
//   k_trsm_*           <- cblas_dtrsm Right/Upper/NonUnit (:1347) and
//                         Left/Lower/Unit (:1539), 32-wide diagonal blocks +
//                         k_dgemm_f64 updates.
//   k_row_gather/scatter/move, k_swap_map/k_rowperm_*, k_copy2d, ...
//                      <- push_pivots_up / permute_rows / dlaswp / mcopy
//                         (conflux_opt.hpp:176-218, utils.hpp:48-160,
//                          memory_utils.hpp:8-34) as coalesced index-vector
//                         kernels.
//   k_init_matrix      <- lu_params::InitMatrix random fill (splitmix64
//                         variant, oracle/gen_input.py — bit-identical).
//
// All fp64.  No CUDA-compat shims, no library GEMMs on the hot path.
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdlib>

#include "kernels.hpp"

#define DEVFN __device__ __forceinline__

typedef double f64x4 __attribute__((ext_vector_type(4)));

namespace ck {

// ---------------------------------------------------------------------------
// input generator (matches oracle/gen_input.py bit-for-bit)
// ---------------------------------------------------------------------------
DEVFN uint64_t splitmix64(uint64_t z) {
    z += 0x9E3779B97F4A7C15ull;
    z ^= z >> 30;
    z *= 0xBF58476D1CE4E5B9ull;
    z ^= z >> 27;
    z *= 0x94D049BB133111EBull;
    z ^= z >> 31;
    return z;
}

DEVFN double gen_entry(int64_t gi, int64_t gj, uint64_t seed) {
    uint64_t key = ((uint64_t)gi << 32) ^ (uint64_t)gj;
    key += seed * 0xBF58476D1CE4E5B9ull;
    const uint64_t h = splitmix64(key);
    return 5.0 + (double)(h >> 11) * (1.0 / 9007199254740992.0);
}

// Grid-stride: a flat launch over Ml x Nl elements would need 2^32 threads
// at N=65536, one past HIP's 32-bit total-thread launch limit — the launch
// is REJECTED SILENTLY and the matrix stays zero (found the hard way; the
// launchers below cap the grid and every at-risk kernel strides).
__global__ void k_init_matrix(double *__restrict__ A, int Ml, int Nl, int v,
                              int Px, int Py, int pi, int pj, int zero_layer,
                              uint64_t seed, int spd, int64_t Nglob) {
    const int64_t total = (int64_t)Ml * Nl;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        if (zero_layer) {
            A[idx] = 0.0;
            continue;
        }
        const int r = (int)(idx / Nl), c = (int)(idx % Nl);
        // local (r, c) -> global (i, j): tile-cyclic map (layout.cpp:95-123)
        const int64_t gi = (int64_t)(r / v * Px + pi) * v + r % v;
        const int64_t gj = (int64_t)(c / v * Py + pj) * v + c % v;
        if (spd) {
            // symmetric positive definite: sym(gen)/1 + 2N on the diagonal
            double x = 0.5 * (gen_entry(gi, gj, seed) + gen_entry(gj, gi, seed));
            if (gi == gj) x += 2.0 * (double)Nglob;
            A[idx] = x;
            continue;
        }
        A[idx] = gen_entry(gi, gj, seed);
    }
}

// ---------------------------------------------------------------------------
// 2D strided copies / zero / add (mcopy & reduce-combine re-expressions)
// ---------------------------------------------------------------------------
__global__ void k_copy2d(const double *__restrict__ src, int64_t lds,
                         double *__restrict__ dst, int64_t ldd,
                         int rows, int64_t cols) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < (int64_t)rows * cols; i += stride) {
        const int64_t r = i / cols, c = i % cols;
        dst[r * ldd + c] = src[r * lds + c];
    }
}

__global__ void k_zero2d(double *__restrict__ dst, int64_t ldd, int rows,
                         int64_t cols) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < (int64_t)rows * cols; i += stride)
        dst[(i / cols) * ldd + i % cols] = 0.0;
}

__global__ void k_add2d(const double *__restrict__ src, int64_t lds,
                        double *__restrict__ dst, int64_t ldd, int rows,
                        int64_t cols) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < (int64_t)rows * cols; i += stride)
        dst[(i / cols) * ldd + i % cols] += src[(i / cols) * lds + i % cols];
}

// dst row i <- src row idx[i]   (gather; push_pivots_up phase 1/3,
// candidate winner permute — utils.hpp inverse_permute_rows)
__global__ void k_row_gather(const double *__restrict__ src, int64_t lds,
                             double *__restrict__ dst, int64_t ldd,
                             const int *__restrict__ idx, int n_rows,
                             int64_t cols) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < (int64_t)n_rows * cols; i += stride) {
        const int64_t r = i / cols, c = i % cols;
        dst[r * ldd + c] = src[(int64_t)idx[r] * lds + c];
    }
}

// dst row idx[i] <- src row i   (scatter; push_pivots_up phase 2,
// A01 pivot-order placement — conflux_opt.hpp:1251-1258)
__global__ void k_row_scatter(const double *__restrict__ src, int64_t lds,
                              double *__restrict__ dst, int64_t ldd,
                              const int *__restrict__ idx, int n_rows,
                              int64_t cols) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < (int64_t)n_rows * cols; i += stride) {
        const int64_t r = i / cols, c = i % cols;
        dst[(int64_t)idx[r] * ldd + c] = src[r * lds + c];
    }
}

// src rows idx[i] copied to dst rows dst_idx[i] within SAME buffer is unsafe;
// engine always stages through separate buffers (3-phase like the reference).



// gather/scatter with a skipped column range [skip0, skip0+skipn): the
// sub-panel columns were already swapped inside k_panel_factor.
__global__ void k_rowperm_gather_skip(const double *__restrict__ src,
                                      int64_t lds, double *__restrict__ tmp,
                                      const int *__restrict__ src_idx,
                                      int row_base, int n_rows, int64_t skip0,
                                      int64_t skipn, int64_t tot_cols) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < (int64_t)n_rows * tot_cols; i += stride) {
        const int64_t r = i / tot_cols, cc = i % tot_cols;
        const int64_t c = (cc < skip0) ? cc : cc + skipn;
        tmp[r * tot_cols + cc] = src[(int64_t)(row_base + src_idx[r]) * lds + c];
    }
}

__global__ void k_rowperm_scatter_skip(const double *__restrict__ tmp,
                                       double *__restrict__ dst, int64_t ldd,
                                       const int *__restrict__ dst_idx,
                                       int row_base, int n_rows, int64_t skip0,
                                       int64_t skipn, int64_t tot_cols) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < (int64_t)n_rows * tot_cols; i += stride) {
        const int64_t r = i / tot_cols, cc = i % tot_cols;
        const int64_t c = (cc < skip0) ? cc : cc + skipn;
        dst[(int64_t)(dst_idx[r] + row_base) * ldd + c] = tmp[r * tot_cols + cc];
    }
}

// dst row dst_idx[i] <- src row src_idx[i]; row sets must be disjoint
// (push_pivots_up phase 2: early non-pivots into vacated late-pivot slots)
__global__ void k_row_move(const double *__restrict__ src, int64_t lds,
                           double *__restrict__ dst, int64_t ldd,
                           const int *__restrict__ src_idx,
                           const int *__restrict__ dst_idx, int n_rows,
                           int64_t cols) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < (int64_t)n_rows * cols; i += stride) {
        const int64_t r = i / cols, c = i % cols;
        dst[(int64_t)dst_idx[r] * ldd + c] = src[(int64_t)src_idx[r] * lds + c];
    }
}



// ---------------------------------------------------------------------------
// panel column step: argmax + swap + scale + rank-1 on a col-major sub-panel
// ---------------------------------------------------------------------------
// One launch per column c of an nb-wide sub-panel held col-major in `cm`
// (ld = ldc, rows 0..m).  Matches LAPACK dgetrf partial pivoting: pivot =
// first row (smallest index) with max |value| in column c among rows c..m;
// swap full sub-panel rows c <-> piv; scale col c below diag by 1/pivot
// (multiply by reciprocal, as dscal does); rank-1 update cols c+1..nb.
//
// Inter-block protocol (cdna_hip_programming.md §6 G16, R1 form):
//   every block: phase-A local argmax over its rows -> sc1 payload stores ->
//   vmcnt drain -> relaxed-agent flag(epoch); block 0 polls all flags,
//   reduces with the first-max rule, performs the swap OUTSIDE the disputed
//   region, publishes both affected rows sc1 + flag; blocks then finish from
//   registers + the published slab (no plain re-reads of swapped data).
#define PANEL_TPB 256
// (rows per block is now a template parameter of k_panel_factor —
// QR*TPB, default 256x256 = 1 row/thread from the r02 shape sweep;
// panel_shape() below picks the instantiation)
#define PANEL_NB 32             // sub-panel width == register column budget

// --------------------------------------------------------------------------
// persistent sub-panel factorization (v2): ONE launch factors a whole
// nb(<=32)-wide sub-panel.  Each block owns QR*TPB rows, held in LDS for
// all nb columns; the only cross-block traffic per column is the slab
// handshake: every block publishes its candidate row (full nb values) and
// the block owning the diagonal row publishes it, so the row swap is
// performed by the OWNING blocks from slab data — no cross-block reads of
// matrix memory, no fences on the bulk (G16 R1: sc1 payload -> drain ->
// flag; consumers poll relaxed + read sc1).
// --------------------------------------------------------------------------
// All per-column slabs are DOUBLE-BUFFERED by column parity: a fast block
// may publish column c+1 while a slow one still reads column c (it cannot
// reach c+2's publish before every block published c+1, so two slots
// suffice).  Single-buffered slabs raced (old flag + new payload is
// observable: no ordering between one block's reads and another's writes).
struct PanelSync2 {
    unsigned long long key_abs[2][CONFLUX_PANEL_MAX_BLOCKS];
    unsigned long long key_flag[2][CONFLUX_PANEL_MAX_BLOCKS];  // (epoch<<32)|row
    double cand_row[2][CONFLUX_PANEL_MAX_BLOCKS][PANEL_NB];
    double diag_row[2][PANEL_NB];
    unsigned int diag_flag[2];
    unsigned int err;
    // diagnostics: total key-poll iterations (all blocks, all columns) —
    // read+reset by conflux_panel_spin_stats (CONFLUX_SPIN_STATS=1)
    unsigned long long spin_sum;
};


typedef unsigned int __attribute__((address_space(1))) gu32;
typedef unsigned long long __attribute__((address_space(1))) gu64;
#define RLX_AGENT __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT

DEVFN void st_rlx_u64(unsigned long long *p, unsigned long long v) {
    __hip_atomic_store((gu64 *)p, v, RLX_AGENT);
}
DEVFN unsigned long long ld_rlx_u64(const unsigned long long *p) {
    return __hip_atomic_load((const gu64 *)p, RLX_AGENT);
}
DEVFN void st_rlx_u32(unsigned int *p, unsigned int v) {
    __hip_atomic_store((gu32 *)p, v, RLX_AGENT);
}
DEVFN unsigned int ld_rlx_u32(const unsigned int *p) {
    return __hip_atomic_load((const gu32 *)p, RLX_AGENT);
}
DEVFN void st_rlx_f64(double *p, double v) {
    union { double d; unsigned long long u; } x;
    x.d = v;
    st_rlx_u64((unsigned long long *)p, x.u);
}
DEVFN double ld_rlx_f64(const double *p) {
    union { double d; unsigned long long u; } x;
    x.u = ld_rlx_u64((const unsigned long long *)p);
    return x.d;
}
DEVFN void drain_stores() { asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); }

typedef int i32x4 __attribute__((ext_vector_type(4)));
union F64x2Bits {
    double d[2];
    i32x4 v;
};

// ---------------------------------------------------------------------------
// TRSM diagonal-block solvers (32-wide; the v%32!=0 fallback path and the
// sub-panel solve inside factor_panel — the whole-panel solves use the
// fused k_trsm_*_mfma kernels below)
// ---------------------------------------------------------------------------
// X (nb x N, row-major ld) <- L^{-1} X with L (nb x nb, ld ldl) unit-lower:
// thread j owns column j; column values kept in registers.
// cblas_dtrsm Left/Lower/NoTrans/Unit re-expression (conflux_opt.hpp:1539).
__global__ __launch_bounds__(256) void k_trsm_left_lower_unit32(
    const double *__restrict__ L, int64_t ldl, double *__restrict__ X,
    int64_t ldx, int nb, int64_t N) {
    __shared__ double sL[32][33];
    const int tid = threadIdx.x;
    for (int i = tid; i < nb * nb; i += 256) sL[i / nb][i % nb] = L[(i / nb) * ldl + i % nb];
    __syncthreads();
    const int64_t j = (int64_t)blockIdx.x * 256 + tid;
    if (j >= N) return;
    double x[32];
#pragma unroll
    for (int r = 0; r < 32; ++r) x[r] = (r < nb) ? X[r * ldx + j] : 0.0;
#pragma unroll
    for (int r = 1; r < 32; ++r) {
        double acc = x[r];
#pragma unroll
        for (int i = 0; i < r; ++i) acc -= sL[r][i] * x[i];
        if (r < nb) x[r] = acc;
    }
#pragma unroll
    for (int r = 0; r < 32; ++r)
        if (r < nb) X[r * ldx + j] = x[r];
}

// ---------------------------------------------------------------------------
// persistent sub-panel getrf: ONE launch factors the nb(<=32)-wide sub-panel
// with LAPACK partial pivoting (first-max rule).  Blocks own QR*TPB rows
// held in LDS for all nb columns; per column the only cross-block traffic is
// the slab handshake; the row swap is performed by the OWNING blocks from
// slab-published rows (no cross-block matrix reads, no write races).
// ---------------------------------------------------------------------------
template <int QR, int TPB>  // rows/block = QR*TPB; QR=1 halves the
                            // per-column local update (1 row/thread);
                            // TPB=256 gives 67 KB LDS (2 blocks/CU)
__global__ __launch_bounds__(TPB) void k_panel_factor(
    double *__restrict__ panel, int64_t ldp, int m, int nb,
    PanelSync2 *__restrict__ sync, int *__restrict__ ipiv,
    unsigned int epoch0, int nblocks, int *__restrict__ swap_dst,
    int *__restrict__ swap_src, int backoff) {
    constexpr int RPB = QR * TPB;
    const int tid = threadIdx.x, bid = blockIdx.x;
    const int r0 = bid * RPB + tid;
    __shared__ double rows[QR][TPB][PANEL_NB + 1];
    __shared__ double piv_lds[PANEL_NB];
    __shared__ double diag_lds[PANEL_NB];
    __shared__ double red_abs[TPB];
    __shared__ int red_row[TPB];
    __shared__ unsigned int sh_info[2];
    // block 0 lane 0 composes the dlaswp row-permutation incrementally as
    // pivots are decided (replaces the separate k_swap_map launch):
    __shared__ int sm_base[PANEL_NB];
    __shared__ int sm_opos[PANEL_NB], sm_osrc[PANEL_NB];
    __shared__ int sm_novf;
    if (bid == 0 && tid == 0) {
        for (int i = 0; i < nb; ++i) sm_base[i] = i;
        sm_novf = 0;
    }
    const auto srsrc = __builtin_amdgcn_make_buffer_rsrc(
        (void *)sync, (short)0, (int)sizeof(PanelSync2), 0x00020000);

    unsigned long long spin_acc = 0;  // key-poll iterations (diagnostics)
    for (int q = 0; q < QR; ++q) {
        const int r = r0 + q * TPB;
        for (int cc = 0; cc < nb; ++cc)
            rows[q][tid][cc] = (r < m) ? panel[(int64_t)r * ldp + cc] : 0.0;
    }

    // ---- cooperative publication of column pc's candidate: lanes 0-15
    // stream the winner row, lanes 16-31 the diag row, as 16-byte sc1
    // buffer stores.  The owning (thread, q) of any row r is computable
    // from r directly (r = bid*RPB + tid + q*TPB), so the publishing lanes
    // read the owner's LDS slot; all publishing lanes are wave 0, one wave
    // drain covers them, then lane 0 posts the key granule (R2: the flag
    // carries the row) and — if this block owns row pc — the diag flag.
    auto publish_col = [&](int pc, double wa, int wrow) {
        const unsigned int pep = epoch0 + (unsigned)pc;
        const int ppar = (int)(pep & 1u);
        if (tid < 16) {
            const int lrow = (wrow < m) ? wrow - bid * RPB : 0;
            const int wq = lrow >= TPB;
            const int wtid = lrow - wq * TPB;
            F64x2Bits x;
            x.d[0] = rows[wq][wtid][2 * tid];
            x.d[1] = rows[wq][wtid][2 * tid + 1];
            if (2 * tid < nb)
                __builtin_amdgcn_raw_buffer_store_b128(
                    x.v, srsrc,
                    (int)offsetof(PanelSync2, cand_row) +
                        (ppar * CONFLUX_PANEL_MAX_BLOCKS + bid) * PANEL_NB *
                            8 +
                        16 * tid,
                    0, /*sc1*/ 16);
        } else if (tid < 32 && pc >= bid * RPB &&
                   pc < (bid + 1) * RPB) {
            const int l = tid - 16;
            const int lrow = pc - bid * RPB;
            const int dq = lrow >= TPB;
            const int dtid = lrow - dq * TPB;
            F64x2Bits x;
            x.d[0] = rows[dq][dtid][2 * l];
            x.d[1] = rows[dq][dtid][2 * l + 1];
            if (2 * l < nb)
                __builtin_amdgcn_raw_buffer_store_b128(
                    x.v, srsrc,
                    (int)offsetof(PanelSync2, diag_row) +
                        ppar * PANEL_NB * 8 + 16 * l,
                    0, /*sc1*/ 16);
        }
        // key_abs is payload like the slabs: store it BEFORE the joint
        // drain so ONE vmcnt covers slabs + abs and the flag store follows
        // immediately (one L2 round trip per column instead of two on the
        // last publisher's critical path)
        if (tid == 0) {
            union { double d; unsigned long long u; } a;
            a.d = wa;
            st_rlx_u64(&sync->key_abs[ppar][bid], a.u);
        }
        if (tid < 64) drain_stores();
        if (tid == 0) {
            st_rlx_u64(&sync->key_flag[ppar][bid],
                       ((unsigned long long)pep << 32) | (unsigned)wrow);
            if (pc >= bid * RPB && pc < (bid + 1) * RPB)
                st_rlx_u32(&sync->diag_flag[ppar], pep);
        }
    };
    // combine the four per-wave partials (redundantly on wave 0's lanes)
    auto combine_partials = [&](double &wa, int &wrow) {
        wa = red_abs[0];
        wrow = red_row[0];
        for (int wv = 1; wv < TPB / 64; ++wv) {
            const double oa = red_abs[wv];
            const int orr = red_row[wv];
            if (oa > wa || (oa == wa && orr < wrow)) { wa = oa; wrow = orr; }
        }
    };
    // per-wave shuffle-tree + LDS partial write of a (|value|, row) pair
    auto reduce_to_partials = [&](double amax, int arow) {
        for (int w = 32; w > 0; w >>= 1) {
            const double oa = __shfl_down(amax, w);
            const int orr = __shfl_down(arow, w);
            if (oa > amax || (oa == amax && orr < arow)) {
                amax = oa;
                arow = orr;
            }
        }
        if ((tid & 63) == 0) {
            red_abs[tid >> 6] = amax;
            red_row[tid >> 6] = arow;
        }
    };

    // column 0's candidate: first-max over own rows, then publish.  Every
    // later column's candidate is reduced and published at the TAIL of the
    // previous column's update (the values are final there), so a column
    // step starts directly at the key poll — one barrier and one LDS read
    // pass fewer per column.
    {
        double amax = -1.0;
        int arow = m;
        for (int q = 0; q < QR; ++q) {
            const int r = r0 + q * TPB;
            if (r >= 0 && r < m) {
                const double a = fabs(rows[q][tid][0]);
                if (a > amax || (a == amax && r < arow)) { amax = a; arow = r; }
            }
        }
        reduce_to_partials(amax, arow);
        __syncthreads();
        double wa;
        int wrow;
        combine_partials(wa, wrow);
        publish_col(0, wa, wrow);
    }

    for (int c = 0; c < nb; ++c) {
        const unsigned int epoch = epoch0 + (unsigned)c;
        const int par = (int)(epoch & 1u);  // slab parity slot
        const int diag_off =
            (int)offsetof(PanelSync2, diag_row) + par * PANEL_NB * 8;

        // ---- EVERY block reduces the global winner itself (redundant,
        // deterministic): lane b polls block b's flag granule, reads its
        // |value|, shuffle-tree with the first-max rule.  Saves the central
        // reducer's publish round trip.
        if (tid < 64) {
            double a_d = -1.0;
            int a_row = m, a_win = 0;
            for (int b = tid; b < nblocks; b += 64) {
                unsigned long long g;
                unsigned spins = 0;
                while (((g = ld_rlx_u64(&sync->key_flag[par][b])) >> 32) !=
                       epoch) {
                    if (backoff == 1) __builtin_amdgcn_s_sleep(1);
                    else if (backoff == 2) __builtin_amdgcn_s_sleep(2);
                    else if (backoff >= 4) __builtin_amdgcn_s_sleep(4);
                    if (++spins > 800000000u) {
                        st_rlx_u32(&sync->err, 1u + (unsigned)c);
                        break;
                    }
                }
                spin_acc += spins;
                union { double d; unsigned long long u; } a;
                a.u = ld_rlx_u64(&sync->key_abs[par][b]);
                const int rr = (int)(g & 0xffffffffu);
                if (a.d > a_d || (a.d == a_d && rr < a_row)) {
                    a_d = a.d;
                    a_row = rr;
                    a_win = b;
                }
            }
            for (int w = 32; w > 0; w >>= 1) {
                const double od = __shfl_down(a_d, w);
                const int orr = __shfl_down(a_row, w);
                const int ow = __shfl_down(a_win, w);
                if (od > a_d || (od == a_d && orr < a_row)) {
                    a_d = od;
                    a_row = orr;
                    a_win = ow;
                }
            }
            if (tid == 0) {
                if (bid == 0) {
                    ipiv[c] = a_row;
                    // compose the swap (c <-> a_row) into the row map
                    int *b;
                    if (a_row < nb) {
                        b = &sm_base[a_row];
                    } else {
                        int j = 0;
                        for (; j < sm_novf && sm_opos[j] != a_row; ++j) {
                        }
                        if (j == sm_novf) {
                            sm_opos[j] = a_row;
                            sm_osrc[j] = a_row;
                            ++sm_novf;
                        }
                        b = &sm_osrc[j];
                    }
                    const int t = sm_base[c];
                    sm_base[c] = *b;
                    *b = t;
                }
                sh_info[0] = (unsigned)a_row;
                sh_info[1] = (unsigned)a_win;
                // wait the diagonal row publication too
                unsigned spins = 0;
                while (ld_rlx_u32(&sync->diag_flag[par]) != epoch) {
                    __builtin_amdgcn_s_sleep(1);
                    if (++spins > 800000000u) {
                        st_rlx_u32(&sync->err, 1000000u + (unsigned)c);
                        break;
                    }
                }
            }
        }
        __syncthreads();
        const int piv = (int)sh_info[0];
        const int win = (int)sh_info[1];
        if (tid < 16 && 2 * tid < nb) {
            F64x2Bits x;
            x.v = __builtin_amdgcn_raw_buffer_load_b128(
                srsrc,
                (int)offsetof(PanelSync2, cand_row) +
                    (par * CONFLUX_PANEL_MAX_BLOCKS + win) * PANEL_NB * 8 +
                    16 * tid,
                0, /*sc1*/ 16);
            piv_lds[2 * tid] = x.d[0];
            piv_lds[2 * tid + 1] = x.d[1];
            x.v = __builtin_amdgcn_raw_buffer_load_b128(
                srsrc, diag_off + 16 * tid, 0, /*sc1*/ 16);
            diag_lds[2 * tid] = x.d[0];
            diag_lds[2 * tid + 1] = x.d[1];
        }
        __syncthreads();
        const double pivval = piv_lds[c];
        const bool do_scale = pivval != 0.0;
        const double recip = do_scale ? 1.0 / pivval : 0.0;

        // ---- update own rows; track the next column's candidate inline --
        double namax = -1.0;
        int narow = m;
        for (int q = 0; q < QR; ++q) {
            const int r = r0 + q * TPB;
            if (r >= m) continue;
            double *my = rows[q][tid];
            if (r == c) {
                for (int cc = 0; cc < nb; ++cc) my[cc] = piv_lds[cc];
            } else if (r == piv) {
                for (int cc = 0; cc < c; ++cc) my[cc] = diag_lds[cc];
                const double v0 = diag_lds[c];
                const double l = do_scale ? v0 * recip : v0;
                my[c] = l;
                for (int cc = c + 1; cc < nb; ++cc)
                    my[cc] = diag_lds[cc] - l * piv_lds[cc];
            } else if (r > c) {
                const double l = do_scale ? my[c] * recip : my[c];
                my[c] = l;
                for (int cc = c + 1; cc < nb; ++cc)
                    my[cc] -= l * piv_lds[cc];
            }
            // candidates for column c+1 are exactly the rows updated above
            if (r > c && c + 1 < nb) {
                const double a = fabs(my[c + 1]);
                if (a > namax || (a == namax && r < narow)) {
                    namax = a;
                    narow = r;
                }
            }
        }
        if (c + 1 < nb) reduce_to_partials(namax, narow);
        __syncthreads();
        if (c + 1 < nb) {
            // tail-publish column c+1's candidate (values final after the
            // barrier above; the publish overlaps other blocks' update
            // tails instead of gating the next column's start)
            double wa;
            int wrow;
            combine_partials(wa, wrow);
            publish_col(c + 1, wa, wrow);
        }
    }

    // write back (plain stores; next kernels see them at the launch boundary)
    for (int q = 0; q < QR; ++q) {
        const int r = r0 + q * TPB;
        if (r >= m) continue;
        for (int cc = 0; cc < nb; ++cc)
            panel[(int64_t)r * ldp + cc] = rows[q][tid][cc];
    }
    // spin diagnostics: wave-0 lanes aggregated, one atomicAdd per block
    if (tid < 64) {
        unsigned long long t = spin_acc;
        for (int w2 = 32; w2 > 0; w2 >>= 1) t += __shfl_down(t, w2);
        if (tid == 0 && t)
            (void)__hip_atomic_fetch_add((gu64 *)&sync->spin_sum, t,
                                         RLX_AGENT);
    }
    // block 0: emit the composed row-permutation map (identity-padded by
    // duplicating entry 0 — benign identical concurrent writes downstream).
    // sm_* visibility: lane 0's last update precedes the column loop's
    // closing __syncthreads, so no extra barrier (and none would be legal
    // in this divergent tail).
    if (bid == 0 && tid < 2 * PANEL_NB && swap_dst) {
        const int i = tid;
        int d, sv;
        if (i < nb) {
            d = i;
            sv = sm_base[i];
        } else if (i - nb < sm_novf) {
            d = sm_opos[i - nb];
            sv = sm_osrc[i - nb];
        } else {
            d = 0;
            sv = sm_base[0];
        }
        if (i < 2 * nb) {
            swap_dst[i] = d;
            swap_src[i] = sv;
        }
    }
}

// X (M x nb, row-major ld) <- X U^{-1} with U (nb x nb, ld ldu) upper
// non-unit: 256 rows per block staged through LDS (coalesced), thread r owns
// row r.  cblas_dtrsm Right/Upper/NoTrans/NonUnit (conflux_opt.hpp:1347).
__global__ __launch_bounds__(256) void k_trsm_right_upper32(
    const double *__restrict__ U, int64_t ldu, double *__restrict__ X,
    int64_t ldx, int nb, int64_t M, int trans) {
    __shared__ double sU[32][33];
    __shared__ double sX[256][33];
    const int tid = threadIdx.x;
    // trans: sU[i][c] = U[c][i] — solves X*L^T = B for lower-triangular L
    // (cblas_dtrsm Right/Lower/Trans/NonUnit, reference Cholesky.cpp:280)
    for (int i = tid; i < nb * nb; i += 256)
        sU[i / nb][i % nb] = trans ? U[(i % nb) * ldu + i / nb]
                                   : U[(i / nb) * ldu + i % nb];
    const int64_t r0 = (int64_t)blockIdx.x * 256;
    const int rows = (int)min((int64_t)256, M - r0);
    // stage rows r0..r0+rows coalesced: thread t covers elements t, t+256, ...
    for (int i = tid; i < rows * nb; i += 256) sX[i / nb][i % nb] = X[(r0 + i / nb) * ldx + i % nb];
    __syncthreads();
    if (tid < rows) {
        double x[32];
#pragma unroll
        for (int c = 0; c < 32; ++c) x[c] = (c < nb) ? sX[tid][c] : 0.0;
#pragma unroll
        for (int c = 0; c < 32; ++c) {
            double acc = x[c];
#pragma unroll
            for (int i = 0; i < c; ++i) acc -= x[i] * sU[i][c];
            if (c < nb) x[c] = acc / sU[c][c];
        }
#pragma unroll
        for (int c = 0; c < 32; ++c)
            if (c < nb) sX[tid][c] = x[c];
    }
    __syncthreads();
    for (int i = tid; i < rows * nb; i += 256) X[(r0 + i / nb) * ldx + i % nb] = sX[i / nb][i % nb];
}

// ---------------------------------------------------------------------------
// MFMA fp64 GEMM:  C (M x N, ldc) -= A (M x K, lda) * B (K x N, ldb)
// ---------------------------------------------------------------------------
// 128x128 block tile, BK=16, 4 waves (2x2), 64x64 per wave as 4x4 fragments
// of v_mfma_f64_16x16x4_f64.  Register-staged global->LDS with the
// write-after-barrier placement (guide §5.5 T14); A transposed in LDS so
// fragment reads are conflict-free; XCD-aware block swizzle (T1).
#define GEMM_BM 128
#define GEMM_BN 128
#define GEMM_BK 16
#define GEMM_TPB 256

// tile order: strips of `strip_w` tile-columns walked row-major, so the
// ~512 concurrently-resident workgroups form a 2D window (e.g. 64 wg =
// 8 rows x 8 cols at strip_w 8) instead of a 1D row slice.  Cuts the
// B-panel re-read traffic from ntm full passes to one pass per strip that
// stays L2/LLC-hot, and keeps each A row-panel hot for strip_w consecutive
// tiles.  Bijective for ragged tails; strip_w <= 1 is the flat order.
DEVFN void gemm_tile_of(int wg, int ntm, int ntn, int strip_w, int &tm,
                        int &tn) {
    if (strip_w > 1 && strip_w < ntn) {
        const int per = ntm * strip_w;
        const int full = ntn / strip_w;
        const int strip = wg / per;
        if (strip < full) {
            const int rem = wg % per;
            tm = rem / strip_w;
            tn = strip * strip_w + rem % strip_w;
        } else {
            const int rem = wg - full * per;
            const int wc = ntn - full * strip_w;
            tm = rem / wc;
            tn = full * strip_w + rem % wc;
        }
    } else {
        tm = wg / ntn;
        tn = wg % ntn;
    }
}

__global__ __launch_bounds__(GEMM_TPB) void k_dgemm_f64(
    const double *__restrict__ A, int64_t lda, const double *__restrict__ B,
    int64_t ldb, double *__restrict__ C, int64_t ldc, int M, int64_t N, int K,
    int ntm, int ntn, int strip_w) {
    // bijective XCD swizzle (guide §5: q/r form)
    int wg = blockIdx.x;
    {
        const int nwg = ntm * ntn;
        const int q = nwg >> 3, r = nwg & 7;
        const int xcd = wg & 7, idx = wg >> 3;
        // inverse of dispatch round-robin: give each XCD a contiguous chunk
        wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
        if (nwg < 8) wg = blockIdx.x;   // tiny grids: identity
    }
    int tm, tn;
    gemm_tile_of(wg, ntm, ntn, strip_w, tm, tn);
    const int row0 = tm * GEMM_BM;
    const int64_t col0 = (int64_t)tn * GEMM_BN;

    __shared__ double As[2][GEMM_BK][GEMM_BM + 1];   // transposed, padded
    __shared__ double Bs[2][GEMM_BK][GEMM_BN + 2];   // double-buffered

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wm0 = (wave >> 1) * 64;             // wave's 64x64 sub-tile
    const int wn0 = (wave & 1) * 64;
    const int frow = lane & 15;                   // fragment row/col lane part
    const int fk = lane >> 4;                     // fragment k lane part

    f64x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = f64x4{0, 0, 0, 0};

    // staging registers: 8 A elements + 8 B elements per thread per K-tile
    // (element e = tid + i*256 over the 2048-element tile; A reads coalesce
    // over the 16-wide rows, B reads over the 128-wide rows)
    double ra[8], rb[8];
    const int ktiles = (K + GEMM_BK - 1) / GEMM_BK;

    auto load_a = [&](int kt) {
        const int kk = kt * GEMM_BK;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
            const int e = tid + i * GEMM_TPB;          // 0..2047
            const int r = e >> 4, k = e & 15;          // row-major in tile
            const int gr = row0 + r;
            ra[i] = (gr < M && kk + k < K) ? A[(int64_t)gr * lda + kk + k] : 0.0;
        }
    };
    auto load_b = [&](int kt) {
        const int kk = kt * GEMM_BK;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
            const int e = tid + i * GEMM_TPB;          // 0..2047
            const int k = e >> 7, c = e & 127;
            const int64_t gc = col0 + c;
            rb[i] = (kk + k < K && gc < N) ? B[(int64_t)(kk + k) * ldb + gc] : 0.0;
        }
    };
    auto write_lds = [&](int buf) {
#pragma unroll
        for (int i = 0; i < 8; ++i) {
            const int e = tid + i * GEMM_TPB;
            As[buf][e & 15][e >> 4] = ra[i];
        }
#pragma unroll
        for (int i = 0; i < 8; ++i) {
            const int e = tid + i * GEMM_TPB;
            Bs[buf][e >> 7][e & 127] = rb[i];
        }
    };

    load_a(0);
    load_b(0);
    write_lds(0);
    __syncthreads();

    int cur = 0;
    for (int kt = 0; kt < ktiles; ++kt) {
        if (kt + 1 < ktiles) {           // issue next tile's global loads
            load_a(kt + 1);
            load_b(kt + 1);
        }
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
            const int k = kk * 4 + fk;
            double af[4], bf[4];
#pragma unroll
            for (int i = 0; i < 4; ++i) af[i] = As[cur][k][wm0 + i * 16 + frow];
#pragma unroll
            for (int j = 0; j < 4; ++j) bf[j] = Bs[cur][k][wn0 + j * 16 + frow];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
        // write t+1 into the other buffer (everyone finished reading it at
        // the barrier that ended step t-1), then one barrier per K-step
        if (kt + 1 < ktiles) write_lds(cur ^ 1);
        __syncthreads();
        cur ^= 1;
    }

    // epilogue: C -= acc   (read-modify-write, coalesced over frag columns)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
#pragma unroll
            for (int q = 0; q < 4; ++q) {
                // f64 16x16x4 C/D map (hardware-verified by mfma_probe):
                // lane l, reg q -> D[4*q + (l>>4)][l & 15]
                const int r = row0 + wm0 + i * 16 + q * 4 + fk;
                const int64_t cidx = col0 + wn0 + j * 16 + frow;
                if (r < M && cidx < N) {
                    C[(int64_t)r * ldc + cidx] -= acc[i][j][q];
                }
            }
        }
    }
}

// NOTE (r02 measured): __launch_bounds__(512, 4) is load-bearing — it
// forces a 128-register allocation (with a small scratch spill) whose
// schedule sustains 57.7 TF; relaxing to (512, 2) (166 regs, no spill,
// same 2-waves/SIMD occupancy) drops to 44.3, and software-pipelining the
// fragment loads on top of it reaches only 45.  Keep the tight bound.
__global__ __launch_bounds__(512, 4) void k_dgemm_f64_w8(
    const double *__restrict__ A, int64_t lda, const double *__restrict__ B,
    int64_t ldb, double *__restrict__ C, int64_t ldc, int M, int64_t N, int K,
    int ntm, int ntn, int strip_w, int ntc) {
    __shared__ double As[2][GEMM_BK][GEMM_BM + 1];   // transposed, padded
    __shared__ double Bs[2][GEMM_BK][GEMM_BN + 2];   // double-buffered
    const int nwg = ntm * ntn;
    // persistent grid-stride over tiles: gridDim may be capped below nwg so
    // the concurrent panel kernel (needs whole CUs: 135 KB LDS/block) can
    // co-schedule during lookahead.  gridDim == nwg -> one iteration,
    // identical to the plain launch.
    for (int vwg = blockIdx.x; vwg < nwg; vwg += gridDim.x) {
    // bijective XCD swizzle (guide §5: q/r form)
    int wg = vwg;
    {
        const int q = nwg >> 3, r = nwg & 7;
        const int xcd = vwg & 7, idx = vwg >> 3;
        // inverse of dispatch round-robin: give each XCD a contiguous chunk
        wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
        if (nwg < 8) wg = vwg;          // tiny grids: identity
    }
    int tm, tn;
    gemm_tile_of(wg, ntm, ntn, strip_w, tm, tn);
    const int row0 = tm * GEMM_BM;
    const int64_t col0 = (int64_t)tn * GEMM_BN;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wm0 = (wave >> 2) * 64;             // wave's 64x32 sub-tile
    const int wn0 = (wave & 3) * 32;
    const int frow = lane & 15;                   // fragment row/col lane part
    const int fk = lane >> 4;                     // fragment k lane part

    f64x4 acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = f64x4{0, 0, 0, 0};

    // staging registers: 8 A elements + 8 B elements per thread per K-tile
    // (element e = tid + i*256 over the 2048-element tile; A reads coalesce
    // over the 16-wide rows, B reads over the 128-wide rows)
    double ra[4], rb[4];
    const int ktiles = (K + GEMM_BK - 1) / GEMM_BK;

    auto load_a = [&](int kt) {
        const int kk = kt * GEMM_BK;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 512;               // 0..2047
            const int r = e >> 4, k = e & 15;          // row-major in tile
            const int gr = row0 + r;
            ra[i] = (gr < M && kk + k < K) ? A[(int64_t)gr * lda + kk + k] : 0.0;
        }
    };
    auto load_b = [&](int kt) {
        const int kk = kt * GEMM_BK;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 512;               // 0..2047
            const int k = e >> 7, c = e & 127;
            const int64_t gc = col0 + c;
            rb[i] = (kk + k < K && gc < N) ? B[(int64_t)(kk + k) * ldb + gc] : 0.0;
        }
    };
    auto write_lds = [&](int buf) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 512;
            As[buf][e & 15][e >> 4] = ra[i];
        }
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 512;
            Bs[buf][e >> 7][e & 127] = rb[i];
        }
    };

    load_a(0);
    load_b(0);
    write_lds(0);
    __syncthreads();

    int cur = 0;
    for (int kt = 0; kt < ktiles; ++kt) {
        if (kt + 1 < ktiles) {           // issue next tile's global loads
            load_a(kt + 1);
            load_b(kt + 1);
        }
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
            const int k = kk * 4 + fk;
            double af[4], bf[2];
#pragma unroll
            for (int i = 0; i < 4; ++i) af[i] = As[cur][k][wm0 + i * 16 + frow];
#pragma unroll
            for (int j = 0; j < 2; ++j) bf[j] = Bs[cur][k][wn0 + j * 16 + frow];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
        // write t+1 into the other buffer (everyone finished reading it at
        // the barrier that ended step t-1), then one barrier per K-step
        if (kt + 1 < ktiles) write_lds(cur ^ 1);
        __syncthreads();
        cur ^= 1;
    }

    // epilogue: C -= acc   (read-modify-write, coalesced over frag columns)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
#pragma unroll
            for (int q = 0; q < 4; ++q) {
                // f64 16x16x4 C/D map (hardware-verified by mfma_probe):
                // lane l, reg q -> D[4*q + (l>>4)][l & 15]
                const int r = row0 + wm0 + i * 16 + q * 4 + fk;
                const int64_t cidx = col0 + wn0 + j * 16 + frow;
                if (r < M && cidx < N) {
                    double *cp = &C[(int64_t)r * ldc + cidx];
                    if (ntc) {
                        // non-temporal RMW: C has zero reuse — keep its
                        // 4.3 GB/launch from evicting hot lines (the
                        // concurrent panel's sync slabs) out of the LLC
                        __builtin_nontemporal_store(
                            __builtin_nontemporal_load(cp) - acc[i][j][q],
                            cp);
                    } else {
                        *cp -= acc[i][j][q];
                    }
                }
            }
        }
    }
    // next tile re-stages buffer 0; the kt-loop's closing barrier already
    // ordered every wave's last LDS read before it
    if (vwg + gridDim.x < nwg) __syncthreads();
    }
}


// 256x128 block-tile variant (CONFLUX_GEMM_VARIANT=3): 8 waves as 4x2 of
// 64x64 fragments — double the accumulator chains per wave and half the
// barriers per flop vs the 128x128 w8 kernel, at 99 KB LDS (1 wg/CU,
// 2 waves/SIMD).  Plain (non-persistent) launch: meant for the uncapped
// full-chip launches; the capped overlap path keeps the w8 kernel.
__global__ __launch_bounds__(512) void k_dgemm_f64_bm256(
    const double *__restrict__ A, int64_t lda, const double *__restrict__ B,
    int64_t ldb, double *__restrict__ C, int64_t ldc, int M, int64_t N, int K,
    int ntm, int ntn, int strip_w) {
    int wg = blockIdx.x;
    {
        const int nwg = ntm * ntn;
        const int q = nwg >> 3, r = nwg & 7;
        const int xcd = wg & 7, idx = wg >> 3;
        wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
        if (nwg < 8) wg = blockIdx.x;
    }
    int tm, tn;
    gemm_tile_of(wg, ntm, ntn, strip_w, tm, tn);
    const int row0 = tm * 256;
    const int64_t col0 = (int64_t)tn * GEMM_BN;

    constexpr int BK8 = 8;  // half-depth K-tile: 49.5 KB LDS -> 2 wg/CU
    __shared__ double As[2][BK8][256 + 1];   // transposed, padded
    __shared__ double Bs[2][BK8][GEMM_BN + 2];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wm0 = (wave >> 1) * 64;             // wave's 64x64 sub-tile
    const int wn0 = (wave & 1) * 64;
    const int frow = lane & 15;
    const int fk = lane >> 4;

    f64x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = f64x4{0, 0, 0, 0};

    double ra[4], rb[2];
    const int ktiles = (K + BK8 - 1) / BK8;

    auto load_a = [&](int kt) {
        const int kk = kt * BK8;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 512;               // 0..2047
            const int r = e >> 3, k = e & 7;
            const int gr = row0 + r;
            ra[i] = (gr < M && kk + k < K) ? A[(int64_t)gr * lda + kk + k] : 0.0;
        }
    };
    auto load_b = [&](int kt) {
        const int kk = kt * BK8;
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            const int e = tid + i * 512;               // 0..1023
            const int k = e >> 7, c = e & 127;
            const int64_t gc = col0 + c;
            rb[i] = (kk + k < K && gc < N) ? B[(int64_t)(kk + k) * ldb + gc] : 0.0;
        }
    };
    auto write_lds = [&](int buf) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 512;
            As[buf][e & 7][e >> 3] = ra[i];
        }
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            const int e = tid + i * 512;
            Bs[buf][e >> 7][e & 127] = rb[i];
        }
    };

    load_a(0);
    load_b(0);
    write_lds(0);
    __syncthreads();

    int cur = 0;
    for (int kt = 0; kt < ktiles; ++kt) {
        if (kt + 1 < ktiles) {
            load_a(kt + 1);
            load_b(kt + 1);
        }
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            const int k = kk * 4 + fk;
            double af[4], bf[4];
#pragma unroll
            for (int i = 0; i < 4; ++i) af[i] = As[cur][k][wm0 + i * 16 + frow];
#pragma unroll
            for (int j = 0; j < 4; ++j) bf[j] = Bs[cur][k][wn0 + j * 16 + frow];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
        if (kt + 1 < ktiles) write_lds(cur ^ 1);
        __syncthreads();
        cur ^= 1;
    }

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
#pragma unroll
            for (int q = 0; q < 4; ++q) {
                const int r = row0 + wm0 + i * 16 + q * 4 + fk;
                const int64_t cidx = col0 + wn0 + j * 16 + frow;
                if (r < M && cidx < N) C[(int64_t)r * ldc + cidx] -= acc[i][j][q];
            }
        }
    }
}

// NT variant: C -= A * B^T with B stored (N x K) row-major — the
// computeA11 low-rank update (reference Cholesky.cpp:345-351).
// Tile-cyclic lower-triangle mask (the Cholesky c4 single-launch update):
// workgroups whose GLOBAL v-tile row < v-tile col exit immediately, so one
// rectangular launch covers the trapezoid of tiles with i >= j and spends
// nothing on the rest.  mask.v == 0 disables.  A 128-wide workgroup never
// straddles a v-tile boundary (v % 128 == 0 enforced by the launcher).
struct TriMask {
    int v;          // global tile size (0 = no mask)
    int r0off;      // launch-local row 0's offset within the rank-local A11
    int64_t c0off;  // same for columns
    int Px, Py, pi, pj;
};

__global__ __launch_bounds__(512, 4) void k_dgemm_f64_w8_nt(
    const double *__restrict__ A, int64_t lda, const double *__restrict__ B,
    int64_t ldb, double *__restrict__ C, int64_t ldc, int M, int64_t N, int K,
    int ntm, int ntn, int strip_w, TriMask mask) {
    // bijective XCD swizzle (guide §5: q/r form)
    int wg = blockIdx.x;
    {
        const int nwg = ntm * ntn;
        const int q = nwg >> 3, r = nwg & 7;
        const int xcd = wg & 7, idx = wg >> 3;
        // inverse of dispatch round-robin: give each XCD a contiguous chunk
        wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
        if (nwg < 8) wg = blockIdx.x;   // tiny grids: identity
    }
    int tm, tn;
    gemm_tile_of(wg, ntm, ntn, strip_w, tm, tn);
    const int row0 = tm * GEMM_BM;
    const int64_t col0 = (int64_t)tn * GEMM_BN;
    if (mask.v > 0) {
        const int gti = (int)((mask.r0off + row0) / mask.v) * mask.Px + mask.pi;
        const int gtj = (int)((mask.c0off + col0) / mask.v) * mask.Py + mask.pj;
        if (gti < gtj) return;  // above the global tile diagonal: untouched
    }

    __shared__ double As[2][GEMM_BK][GEMM_BM + 1];   // transposed, padded
    __shared__ double Bs[2][GEMM_BK][GEMM_BN + 2];   // double-buffered

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wm0 = (wave >> 2) * 64;             // wave's 64x32 sub-tile
    const int wn0 = (wave & 3) * 32;
    const int frow = lane & 15;                   // fragment row/col lane part
    const int fk = lane >> 4;                     // fragment k lane part

    f64x4 acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = f64x4{0, 0, 0, 0};

    // staging registers: 8 A elements + 8 B elements per thread per K-tile
    // (element e = tid + i*256 over the 2048-element tile; A reads coalesce
    // over the 16-wide rows, B reads over the 128-wide rows)
    double ra[4], rb[4];
    const int ktiles = (K + GEMM_BK - 1) / GEMM_BK;

    auto load_a = [&](int kt) {
        const int kk = kt * GEMM_BK;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 512;               // 0..2047
            const int r = e >> 4, k = e & 15;          // row-major in tile
            const int gr = row0 + r;
            ra[i] = (gr < M && kk + k < K) ? A[(int64_t)gr * lda + kk + k] : 0.0;
        }
    };
    auto load_b = [&](int kt) {
        // B is (N x K) row-major here: Bs[k][n] <- B[col0+n][kk+k]
        // (coalesced over k within each B row, like the A staging)
        const int kk = kt * GEMM_BK;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 512;               // 0..2047
            const int n = e >> 4, k = e & 15;
            const int64_t gc = col0 + n;
            rb[i] = (kk + k < K && gc < N) ? B[gc * ldb + kk + k] : 0.0;
        }
    };
    auto write_lds = [&](int buf) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 512;
            As[buf][e & 15][e >> 4] = ra[i];
        }
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 512;
            Bs[buf][e & 15][e >> 4] = rb[i];
        }
    };

    load_a(0);
    load_b(0);
    write_lds(0);
    __syncthreads();

    int cur = 0;
    for (int kt = 0; kt < ktiles; ++kt) {
        if (kt + 1 < ktiles) {           // issue next tile's global loads
            load_a(kt + 1);
            load_b(kt + 1);
        }
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
            const int k = kk * 4 + fk;
            double af[4], bf[2];
#pragma unroll
            for (int i = 0; i < 4; ++i) af[i] = As[cur][k][wm0 + i * 16 + frow];
#pragma unroll
            for (int j = 0; j < 2; ++j) bf[j] = Bs[cur][k][wn0 + j * 16 + frow];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
        // write t+1 into the other buffer (everyone finished reading it at
        // the barrier that ended step t-1), then one barrier per K-step
        if (kt + 1 < ktiles) write_lds(cur ^ 1);
        __syncthreads();
        cur ^= 1;
    }

    // epilogue: C -= acc   (read-modify-write, coalesced over frag columns)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
#pragma unroll
            for (int q = 0; q < 4; ++q) {
                // f64 16x16x4 C/D map (hardware-verified by mfma_probe):
                // lane l, reg q -> D[4*q + (l>>4)][l & 15]
                const int r = row0 + wm0 + i * 16 + q * 4 + fk;
                const int64_t cidx = col0 + wn0 + j * 16 + frow;
                if (r < M && cidx < N) {
                    C[(int64_t)r * ldc + cidx] -= acc[i][j][q];
                }
            }
        }
    }
}


// ---------------------------------------------------------------------------
// glds variant: async global->LDS staging (__builtin_amdgcn_global_load_lds,
// 16-byte pieces), one barrier per K-step, XOR-swizzled SOURCE addresses so
// the lane-linear LDS image reads conflict-free (guide rule 21 / T2).
// Layouts: As [BK][BM] row-of-m linear, A-chunk (row, p) stored at p^(row&7);
//          Bs [BK][BN] linear, B-chunk (k, np) stored at np^((k&3)<<1).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(512, 4) void k_dgemm_f64_glds(
    const double *__restrict__ A, int64_t lda, const double *__restrict__ B,
    int64_t ldb, double *__restrict__ C, int64_t ldc, int M, int64_t N, int K,
    int ntm, int ntn, int strip_w) {
    int wg = blockIdx.x;
    {
        const int nwg = ntm * ntn;
        const int q = nwg >> 3, r = nwg & 7;
        const int xcd = wg & 7, idx = wg >> 3;
        wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
        if (nwg < 8) wg = blockIdx.x;
    }
    int tm, tn;
    gemm_tile_of(wg, ntm, ntn, strip_w, tm, tn);
    const int row0 = tm * GEMM_BM;
    const int64_t col0 = (int64_t)tn * GEMM_BN;

    // ONE shared object (a second one makes hipcc drain vmcnt(0) before
    // every ds_read — guide §5 trap (a)).  [2 buffers][A 16*128 | B 16*128]
    __shared__ double smem[2 * (GEMM_BK * GEMM_BM + GEMM_BK * GEMM_BN)];
    double *As[2] = {smem, smem + 2 * GEMM_BK * GEMM_BM + 0};
    // carve: buf b: A at b*(BK*BM), B at 2*BK*BM + b*(BK*BN)
    As[0] = smem;
    As[1] = smem + GEMM_BK * GEMM_BM;
    double *Bs0 = smem + 2 * GEMM_BK * GEMM_BM;
    double *Bs[2] = {Bs0, Bs0 + GEMM_BK * GEMM_BN};

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wm0 = (wave >> 2) * 64;
    const int wn0 = (wave & 3) * 32;
    const int frow = lane & 15;
    const int fk = lane >> 4;

    f64x4 acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = f64x4{0, 0, 0, 0};

    const int ktiles = (K + GEMM_BK - 1) / GEMM_BK;
    // A: 1024 16B chunks (row, p), linear ci = row*8 + p_lds; 2 waves-worth
    // per wave (16 wave-instructions over 8 waves = 2 each).
    // chunk for this lane at issue w (w = 0,1): ci = wave*128 + w*64 + lane
    // B: 1024 chunks (k, np), ci = k*64 + np_lds; same split.
    auto stage = [&](int buf, int kt) {
        const int kk = kt * GEMM_BK;
#pragma unroll
        for (int w = 0; w < 2; ++w) {
            const int ci = wave * 128 + w * 64 + lane;
            // A chunk (row, p_lds) holds source k-pair p_lds ^ (row & 7);
            // out-of-range rows load the block's first row (junk rows are
            // discarded by the epilogue bounds).  K is a multiple of 16
            // everywhere in this engine (v, nlayr, NB all are), so there is
            // no K-tail.
            const int arow = ci >> 3, ap = ci & 7;
            const int asrc_p = ap ^ (arow & 7);
            const double *agp =
                A + (int64_t)(row0 + ((row0 + arow < M) ? arow : 0)) * lda +
                kk + 2 * asrc_p;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t *)agp,
                (__attribute__((address_space(3))) uint32_t
                     *)(As[buf] + (int64_t)(wave * 128 + w * 64) * 2),
                16, 0, 0);
            // B chunk (k, np_lds) holds source col-pair np_lds ^ ((k&3)<<1).
            // Clamp only when the chunk's FIRST column is out of range (a
            // half-valid chunk must still load its valid column; its second
            // 8 bytes may read past row end — within hipMalloc's page
            // granularity, and its value only lands in discarded lanes).
            const int ck = ci >> 6, cnp = ci & 63;
            const int bsrc_np = cnp ^ ((ck & 3) << 1);
            const double *bgp =
                B + (int64_t)(kk + ck) * ldb + col0 + 2 * bsrc_np;
            if (col0 + 2 * bsrc_np >= N) bgp = B + (int64_t)(kk + ck) * ldb;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t *)bgp,
                (__attribute__((address_space(3))) uint32_t
                     *)(Bs[buf] + (int64_t)(ci - lane) * 2),
                16, 0, 0);
        }
    };

    stage(0, 0);
    int cur = 0;
    for (int kt = 0; kt < ktiles; ++kt) {
        // ONE barrier per K-step: its per-wave vmcnt(0) drains the stage
        // issued LAST iteration (it flew under that iteration's compute),
        // and the barrier itself guarantees every wave finished reading the
        // buffer the next stage overwrites.
        __syncthreads();
        if (kt + 1 < ktiles) stage(cur ^ 1, kt + 1);
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
            const int k = kk * 4 + fk;
            const int ap = k >> 1, ae = k & 1;
            double af[4], bf[2];
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int row = wm0 + i * 16 + frow;
                af[i] = As[cur][row * 16 + 2 * (ap ^ (row & 7)) + ae];
            }
#pragma unroll
            for (int j = 0; j < 2; ++j) {
                const int n = wn0 + j * 16 + frow;
                const int np = (n >> 1) ^ ((k & 3) << 1);
                bf[j] = Bs[cur][k * 128 + 2 * np + (n & 1)];
            }
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
        cur ^= 1;
    }

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
#pragma unroll
            for (int q = 0; q < 4; ++q) {
                const int r = row0 + wm0 + i * 16 + q * 4 + fk;
                const int64_t cidx = col0 + wn0 + j * 16 + frow;
                if (r < M && cidx < N) C[(int64_t)r * ldc + cidx] -= acc[i][j][q];
            }
        }
    }
}


// ---------------------------------------------------------------------------
// Cholesky kernels (CONFCHOX path; reference
// src/conflux/cholesky/Cholesky.cpp): k_potrf32 <- the 32-wide micro-factor
// of LAPACKE_dpotrf('L') (Cholesky.cpp:192); the right/lower/trans TRSM is
// k_trsm_right_tri32 with trans=1 (updateA10's dtrsm, Cholesky.cpp:280);
// the NoTrans x Trans low-rank update (computeA11, Cholesky.cpp:345-351) is
// k_dgemm_f64_w8_nt below.
// ---------------------------------------------------------------------------
// single block: in-place lower Cholesky of the nb x nb (nb <= 32) diagonal
// block, row-major ld.  Sequential over columns in LDS (no pivoting).
__global__ __launch_bounds__(256) void k_potrf32(double *__restrict__ A,
                                                 int64_t lda, int nb) {
    __shared__ double sA[32][33];
    const int tid = threadIdx.x;
    for (int i = tid; i < nb * nb; i += 256)
        sA[i / nb][i % nb] = A[(i / nb) * lda + i % nb];
    __syncthreads();
    for (int c = 0; c < nb; ++c) {
        if (tid == 0) sA[c][c] = sqrt(sA[c][c]);
        __syncthreads();
        const double inv = 1.0 / sA[c][c];
        if (tid > c && tid < nb) sA[tid][c] *= inv;
        __syncthreads();
        // trailing update: element (i, j), i > c, c < j <= i
        for (int e = tid; e < nb * nb; e += 256) {
            const int i = e / nb, j = e % nb;
            if (i > c && j > c && j <= i) sA[i][j] -= sA[i][c] * sA[j][c];
        }
        __syncthreads();
    }
    for (int i = tid; i < nb * nb; i += 256)
        A[(i / nb) * lda + i % nb] = sA[i / nb][i % nb];
}

// 32x32 LU WITHOUT pivoting (the Python prototype's EmptyPivot fast path,
// SURVEY §8f4 — for diagonally dominant inputs): per column, scale below
// the diagonal by 1/pivot and rank-1 the trailing sub-block.  Single block,
// same shape as k_potrf32.
__global__ __launch_bounds__(256) void k_getrf32_nopiv(double *__restrict__ A,
                                                       int64_t lda, int nb) {
    __shared__ double sA[32][33];
    const int tid = threadIdx.x;
    for (int i = tid; i < nb * nb; i += 256)
        sA[i / nb][i % nb] = A[(i / nb) * lda + i % nb];
    __syncthreads();
    for (int c = 0; c < nb; ++c) {
        const double piv = sA[c][c];
        const double inv = (piv != 0.0) ? 1.0 / piv : 0.0;  // LAPACK-style:
        __syncthreads();                                    // no scaling on 0
        if (tid > c && tid < nb && piv != 0.0) sA[tid][c] *= inv;
        __syncthreads();
        for (int e = tid; e < nb * nb; e += 256) {
            const int i = e / nb, j = e % nb;
            if (i > c && j > c) sA[i][j] -= sA[i][c] * sA[c][j];
        }
        __syncthreads();
    }
    for (int i = tid; i < nb * nb; i += 256)
        A[(i / nb) * lda + i % nb] = sA[i / nb][i % nb];
}

// ---------------------------------------------------------------------------
// misc small kernels for the distributed path
// ---------------------------------------------------------------------------
// candidate pack: cand row i = [ gri[f+i] | A10[f+i, 0..v) ]  (prepend_column,
// utils.hpp:13-26 + step-1 copy conflux_opt.hpp:698-705)
// cand row i = [ gri[f+s] | A10[f+s, :] ] with s = idx ? idx[i] : i;
// s >= n_src stands for the reference's zero-padded candidate rows
// (step0_padding, conflux_opt.hpp:604-614).
__global__ void k_pack_candidate(const double *__restrict__ A10, int64_t lda,
                                 const int *__restrict__ gri, int f, int n_src,
                                 int n_out, int v, const int *__restrict__ idx,
                                 double *__restrict__ cand) {
    const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= (int64_t)n_out * (v + 1)) return;
    const int r = (int)(i / (v + 1)), c = (int)(i % (v + 1));
    const int s = idx ? idx[r] : r;
    if (s >= n_src) { cand[i] = 0.0; return; }
    cand[i] = (c == 0) ? (double)gri[f + s] : A10[(int64_t)(f + s) * lda + c - 1];
}

__global__ void k_extract_col0_int(const double *__restrict__ cand, int stride,
                                   int n, int *__restrict__ out) {
    const int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) out[i] = (int)cand[(int64_t)i * stride];
}

// slab pack for the A10 spread (conflux_opt.hpp:1389-1399): column slab
// [pk*nlayr, (pk+1)*nlayr) of X (n x v) -> contiguous n x nlayr at slab pk
__global__ void k_slab_pack(const double *__restrict__ X, int64_t ldx, int n,
                            int nlayr, int Pz, double *__restrict__ out) {
    const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= (int64_t)n * nlayr * Pz) return;
    const int pk = (int)(i / ((int64_t)n * nlayr));
    const int64_t rest = i % ((int64_t)n * nlayr);
    const int r = (int)(rest / nlayr), c = (int)(rest % nlayr);
    out[i] = X[(int64_t)r * ldx + pk * nlayr + c];
}

}  // namespace ck

// ---------------------------------------------------------------------------
// launch wrappers (host side)
// ---------------------------------------------------------------------------
using namespace ck;

static inline int64_t cdiv64(int64_t a, int64_t b) { return (a + b - 1) / b; }

static inline unsigned cap_grid(int64_t total) {
    int64_t g = cdiv64(total, 256);
    if (g > (1 << 22)) g = 1 << 22;  // grid-stride kernels cover the rest
    return (unsigned)g;
}

void launch_init_matrix(double *A, int Ml, int Nl, int v, int Px, int Py,
                        int pi, int pj, int zero_layer, uint64_t seed,
                        hipStream_t s) {
    const int64_t total = (int64_t)Ml * Nl;
    hipLaunchKernelGGL(k_init_matrix, dim3(cap_grid(total)), dim3(256), 0, s,
                       A, Ml, Nl, v, Px, Py, pi, pj, zero_layer, seed, 0,
                       (int64_t)0);
}

void launch_init_matrix_spd(double *A, int Ml, int Nl, int v, int Px, int Py,
                            int pi, int pj, int zero_layer, uint64_t seed,
                            int64_t Nglob, hipStream_t s) {
    const int64_t total = (int64_t)Ml * Nl;
    hipLaunchKernelGGL(k_init_matrix, dim3(cap_grid(total)), dim3(256), 0, s,
                       A, Ml, Nl, v, Px, Py, pi, pj, zero_layer, seed, 1,
                       Nglob);
}

void launch_copy2d(const double *src, int64_t lds, double *dst, int64_t ldd,
                   int rows, int64_t cols, hipStream_t s) {
    if (rows <= 0 || cols <= 0) return;
    hipLaunchKernelGGL(k_copy2d, dim3(cap_grid((int64_t)rows * cols)),
                       dim3(256), 0, s, src, lds, dst, ldd, rows, cols);
}

void launch_zero2d(double *dst, int64_t ldd, int rows, int64_t cols,
                   hipStream_t s) {
    if (rows <= 0 || cols <= 0) return;
    hipLaunchKernelGGL(k_zero2d, dim3(cap_grid((int64_t)rows * cols)),
                       dim3(256), 0, s, dst, ldd, rows, cols);
}

void launch_add2d(const double *src, int64_t lds, double *dst, int64_t ldd,
                  int rows, int64_t cols, hipStream_t s) {
    if (rows <= 0 || cols <= 0) return;
    hipLaunchKernelGGL(k_add2d, dim3(cap_grid((int64_t)rows * cols)),
                       dim3(256), 0, s, src, lds, dst, ldd, rows, cols);
}

void launch_row_gather(const double *src, int64_t lds, double *dst,
                       int64_t ldd, const int *idx, int n_rows, int64_t cols,
                       hipStream_t s) {
    if (n_rows <= 0 || cols <= 0) return;
    hipLaunchKernelGGL(k_row_gather, dim3(cap_grid((int64_t)n_rows * cols)),
                       dim3(256), 0, s, src, lds, dst, ldd, idx, n_rows, cols);
}

void launch_row_scatter(const double *src, int64_t lds, double *dst,
                        int64_t ldd, const int *idx, int n_rows, int64_t cols,
                        hipStream_t s) {
    if (n_rows <= 0 || cols <= 0) return;
    hipLaunchKernelGGL(k_row_scatter, dim3(cap_grid((int64_t)n_rows * cols)),
                       dim3(256), 0, s, src, lds, dst, ldd, idx, n_rows, cols);
}





void launch_rowperm_skip(double *mat, int64_t ld, const int *dst_idx,
                         const int *src_idx, int row_base, int n_rows,
                         int64_t skip0, int64_t skipn, int64_t tot_cols,
                         double *tmp, hipStream_t s) {
    if (n_rows <= 0 || tot_cols <= 0) return;
    const int64_t n = (int64_t)n_rows * tot_cols;
    hipLaunchKernelGGL(k_rowperm_gather_skip, dim3(cap_grid(n)), dim3(256),
                       0, s, mat, ld, tmp, src_idx, row_base, n_rows, skip0,
                       skipn, tot_cols);
    hipLaunchKernelGGL(k_rowperm_scatter_skip, dim3(cap_grid(n)), dim3(256),
                       0, s, tmp, mat, ld, dst_idx, row_base, n_rows, skip0,
                       skipn, tot_cols);
}

// panel block shape: CONFLUX_PANEL_RPB (rows/block) x CONFLUX_PANEL_TPB
// (threads).  r02 A/B at N=16384: 256 rows/block 1-row/thread (67 KB LDS,
// 2 blocks/CU) beats the original 512x256 2-rows/thread: 205 vs 223
// ms/step in context, panel 122 vs 137 ms sequential.  Default 256x256.
static void panel_shape(int *qr, int *tpb) {
    static int g_qr = -1, g_tpb = 0;
    if (g_qr < 0) {
        const char *r = getenv("CONFLUX_PANEL_RPB");
        const char *t = getenv("CONFLUX_PANEL_TPB");
        g_tpb = t ? atoi(t) : 256;
        if (g_tpb != 512 && g_tpb != 128) g_tpb = 256;
        const int rpb = r ? atoi(r) : 256;
        g_qr = (rpb / g_tpb >= 2) ? 2 : 1;
    }
    *qr = g_qr;
    *tpb = g_tpb;
}

static int panel_qr() {
    int qr, tpb;
    panel_shape(&qr, &tpb);
    return qr;
}

int conflux_panel_blocks_per_cu() {  // LDS/block ~= QR*TPB*264 B + tails
    int qr, tpb;
    panel_shape(&qr, &tpb);
    const int rows = qr * tpb;
    return rows <= 128 ? 4 : rows <= 256 ? 2 : 1;  // 34/67/135 KB of 160
}

int launch_panel_factor(double *panel, int64_t ldp, int m, int nb, void *sync,
                        int *ipiv, unsigned int epoch0, int *swap_dst,
                        int *swap_src, hipStream_t s) {
    static int backoff = -1;
    if (backoff < 0) {
        const char *e = getenv("CONFLUX_PANEL_SLEEP");
        backoff = e ? atoi(e) : 2;  // r02 sweep at the 256x256 shape:
                                    // 0/1/2/4 -> 203.5/202.2/201.5/201.5
                                    // ms/step in context
    }
    int qr, tpb;
    panel_shape(&qr, &tpb);
    int rpb = qr * tpb;
    int nblocks = (int)cdiv64(m, rpb);
    if (nblocks < 1) nblocks = 1;
    if (nblocks > CONFLUX_PANEL_MAX_BLOCKS && rpb < 512) {
        // very tall panel (m > 64k rows): auto-promote to 512-row blocks
        qr = 2; tpb = 256; rpb = 512;
        nblocks = (int)cdiv64(m, rpb);
    }
    if (nblocks > CONFLUX_PANEL_MAX_BLOCKS) return -1;  // not resident: refuse
    if (qr == 1 && tpb == 512)
        hipLaunchKernelGGL((k_panel_factor<1, 512>), dim3(nblocks), dim3(512),
                           0, s, panel, ldp, m, nb, (PanelSync2 *)sync, ipiv,
                           epoch0, nblocks, swap_dst, swap_src, backoff);
    else if (qr == 1 && tpb == 128)
        hipLaunchKernelGGL((k_panel_factor<1, 128>), dim3(nblocks), dim3(128),
                           0, s, panel, ldp, m, nb, (PanelSync2 *)sync, ipiv,
                           epoch0, nblocks, swap_dst, swap_src, backoff);
    else if (qr == 1)
        hipLaunchKernelGGL((k_panel_factor<1, 256>), dim3(nblocks), dim3(256),
                           0, s, panel, ldp, m, nb, (PanelSync2 *)sync, ipiv,
                           epoch0, nblocks, swap_dst, swap_src, backoff);
    else
        hipLaunchKernelGGL((k_panel_factor<2, 256>), dim3(nblocks), dim3(256),
                           0, s, panel, ldp, m, nb, (PanelSync2 *)sync, ipiv,
                           epoch0, nblocks, swap_dst, swap_src, backoff);
    return 0;
}

void conflux_panel_spin_read(void *sync, unsigned long long *out,
                             hipStream_t s) {
    (void)hipMemcpyAsync(out, (char *)sync + offsetof(PanelSync2, spin_sum),
                         8, hipMemcpyDeviceToHost, s);
    (void)hipStreamSynchronize(s);
    (void)hipMemsetAsync((char *)sync + offsetof(PanelSync2, spin_sum), 0, 8,
                         s);
}

int conflux_panel_sync_bytes() { return (int)sizeof(PanelSync2); }
int conflux_panel_nb() { return PANEL_NB; }
int conflux_panel_rpb() {
    int qr, tpb;
    panel_shape(&qr, &tpb);
    return qr * tpb;
}

void launch_trsm_left_lower_unit32(const double *L, int64_t ldl, double *X,
                                   int64_t ldx, int nb, int64_t N,
                                   hipStream_t s) {
    if (N <= 0 || nb <= 0) return;
    hipLaunchKernelGGL(k_trsm_left_lower_unit32, dim3(cdiv64(N, 256)),
                       dim3(256), 0, s, L, ldl, X, ldx, nb, N);
}

void launch_trsm_right_upper32(const double *U, int64_t ldu, double *X,
                               int64_t ldx, int nb, int64_t M, int trans,
                               hipStream_t s) {
    if (M <= 0 || nb <= 0) return;
    hipLaunchKernelGGL(k_trsm_right_upper32, dim3(cdiv64(M, 256)), dim3(256),
                       0, s, U, ldu, X, ldx, nb, M, trans);
}

// ---------------------------------------------------------------------------
// Fused whole-panel TRSMs, MFMA edition: ONE launch applies the full v x v
// triangle (v % 32 == 0).  Left-looking over 32-wide panels; the
// rank-32 updates run on the matrix cores (v_mfma_f64_16x16x4_f64, the
// lane maps verified by tools/mfma_probe), the 32x32 diagonal solves as
// register triangles (the k_trsm_*32 pattern).  Each block owns a stripe
// of X exclusively and stages panels through LDS, so the blocked chain's
// 31 launches and its 16 HBM re-writes of X collapse into one kernel.
// (A VALU version of this idea lost 3x to the blocked chain — the updates
// belong on MFMA; see DESIGN.md ablations.)
// ---------------------------------------------------------------------------

// X (M x v, row-major) <- X * U^-1 (trans=0) or X * L^-T (trans=1).
#define TRM_ROWS 128
__global__ __launch_bounds__(256) void k_trsm_right_mfma(
    const double *__restrict__ U, int64_t ldu, double *__restrict__ X,
    int64_t ldx, int v, int64_t M, int trans) {
    __shared__ double sXj[TRM_ROWS][33];     // current panel
    __shared__ double sXp[2][TRM_ROWS][33];  // solved panels, double-buffered
    __shared__ double sU[2][32][33];
    const int tid = threadIdx.x;
    const int lane = tid & 63, wave = tid >> 6;
    const int frow = lane & 15, fk = lane >> 4;
    const int64_t r0 = (int64_t)blockIdx.x * TRM_ROWS;
    const int rows = (int)min((int64_t)TRM_ROWS, M - r0);
    // register staging (w8-GEMM style): next panel's loads issue while MFMA
    // consumes the current LDS buffer — one barrier per 32-panel, HBM
    // latency hidden behind the update
    double rx[16], ru[4];  // 128x32 X panel + 32x32 U block per iteration
    auto load_u_regs = [&](int pb, int jb) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 256;
            ru[i] = trans
                        ? U[(int64_t)(jb + (e & 31)) * ldu + pb + (e >> 5)]
                        : U[(int64_t)(pb + (e >> 5)) * ldu + jb + (e & 31)];
        }
    };
    auto load_x_regs = [&](int pb) {
#pragma unroll
        for (int i = 0; i < 16; ++i) {
            const int e = tid + i * 256;
            rx[i] = (e >> 5) < rows
                        ? X[(r0 + (e >> 5)) * ldx + pb + (e & 31)]
                        : 0.0;
        }
    };
    auto write_bufs = [&](int buf) {
#pragma unroll
        for (int i = 0; i < 16; ++i) {
            const int e = tid + i * 256;
            sXp[buf][e >> 5][e & 31] = rx[i];
        }
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 256;
            sU[buf][e >> 5][e & 31] = ru[i];
        }
    };
    for (int jb = 0; jb < v; jb += 32) {
        __syncthreads();
        for (int i = tid; i < rows * 32; i += 256)
            sXj[i >> 5][i & 31] = X[(r0 + (i >> 5)) * ldx + jb + (i & 31)];
        int cur = 0;
        if (jb > 0) {
            load_x_regs(0);
            load_u_regs(0, jb);
            write_bufs(0);
        }
        __syncthreads();  // also covers the sXj stage
        for (int pb = 0; pb < jb; pb += 32) {
            if (pb + 32 < jb) {
                load_x_regs(pb + 32);
                load_u_regs(pb + 32, jb);
            }
            // sXj[rows x 32] -= sXp[rows x 32] * sU[32 x 32] on MFMA:
            // wave w owns rows [32w, 32w+32): 2 row-tiles x 2 col-tiles
            if (wave * 32 < rows) {
#pragma unroll
                for (int rt = 0; rt < 2; ++rt) {
                    const int rbase = wave * 32 + rt * 16;
                    f64x4 acc[2] = {f64x4{0, 0, 0, 0}, f64x4{0, 0, 0, 0}};
#pragma unroll
                    for (int kk = 0; kk < 8; ++kk) {
                        const int k = kk * 4 + fk;
                        const double a = sXp[cur][rbase + frow][k];
#pragma unroll
                        for (int ct = 0; ct < 2; ++ct)
                            acc[ct] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                                a, sU[cur][k][ct * 16 + frow], acc[ct], 0, 0,
                                0);
                    }
#pragma unroll
                    for (int ct = 0; ct < 2; ++ct)
#pragma unroll
                        for (int q = 0; q < 4; ++q)
                            sXj[rbase + 4 * q + fk][ct * 16 + frow] -=
                                acc[ct][q];
                }
            }
            if (pb + 32 < jb) write_bufs(cur ^ 1);
            __syncthreads();
            cur ^= 1;
        }
        // diagonal block into sU[0] for the solve
        for (int i = tid; i < 32 * 32; i += 256)
            sU[0][i >> 5][i & 31] =
                trans ? U[(int64_t)(jb + (i & 31)) * ldu + jb + (i >> 5)]
                      : U[(int64_t)(jb + (i >> 5)) * ldu + jb + (i & 31)];
        __syncthreads();
        if (tid < rows) {  // diagonal solve: thread = row, register triangle
            double x[32];
#pragma unroll
            for (int c = 0; c < 32; ++c) x[c] = sXj[tid][c];
#pragma unroll
            for (int c = 0; c < 32; ++c) {
                double acc = x[c];
#pragma unroll
                for (int i = 0; i < 32; ++i)
                    if (i < c) acc -= x[i] * sU[0][i][c];
                x[c] = acc / sU[0][c][c];
            }
#pragma unroll
            for (int c = 0; c < 32; ++c) sXj[tid][c] = x[c];
        }
        __syncthreads();
        for (int i = tid; i < rows * 32; i += 256)
            X[(r0 + (i >> 5)) * ldx + jb + (i & 31)] = sXj[i >> 5][i & 31];
    }
}

// X (v x N, row-major) <- L^-1 * X, L v x v unit lower.
#define TRM_COLS 128
__global__ __launch_bounds__(256) void k_trsm_left_mfma(
    const double *__restrict__ L, int64_t ldl, double *__restrict__ X,
    int64_t ldx, int v, int64_t N) {
    __shared__ double sXj[32][TRM_COLS + 1];
    __shared__ double sXp[2][32][TRM_COLS + 1];  // double-buffered
    __shared__ double sL[2][32][33];
    const int tid = threadIdx.x;
    const int lane = tid & 63, wave = tid >> 6;
    const int frow = lane & 15, fk = lane >> 4;
    const int64_t c0 = (int64_t)blockIdx.x * TRM_COLS;
    const int cols = (int)min((int64_t)TRM_COLS, N - c0);
    double rx[16], rl[4];  // 32x128 X panel + 32x32 L block per iteration
    auto load_l_regs = [&](int pb, int jb) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 256;
            rl[i] = L[(int64_t)(jb + (e >> 5)) * ldl + pb + (e & 31)];
        }
    };
    auto load_x_regs = [&](int pb) {
#pragma unroll
        for (int i = 0; i < 16; ++i) {
            const int e = tid + i * 256;
            rx[i] = (e & (TRM_COLS - 1)) < cols
                        ? X[(int64_t)(pb + (e >> 7)) * ldx + c0 +
                            (e & (TRM_COLS - 1))]
                        : 0.0;
        }
    };
    auto write_bufs = [&](int buf) {
#pragma unroll
        for (int i = 0; i < 16; ++i) {
            const int e = tid + i * 256;
            sXp[buf][e >> 7][e & (TRM_COLS - 1)] = rx[i];
        }
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int e = tid + i * 256;
            sL[buf][e >> 5][e & 31] = rl[i];
        }
    };
    for (int jb = 0; jb < v; jb += 32) {
        __syncthreads();
        for (int i = tid; i < 32 * TRM_COLS; i += 256)
            if ((i & (TRM_COLS - 1)) < cols)
                sXj[i >> 7][i & (TRM_COLS - 1)] =
                    X[(int64_t)(jb + (i >> 7)) * ldx + c0 +
                      (i & (TRM_COLS - 1))];
        int cur = 0;
        if (jb > 0) {
            load_x_regs(0);
            load_l_regs(0, jb);
            write_bufs(0);
        }
        __syncthreads();  // also covers the sXj stage
        for (int pb = 0; pb < jb; pb += 32) {
            if (pb + 32 < jb) {
                load_x_regs(pb + 32);
                load_l_regs(pb + 32, jb);
            }
            // sXj[32 x cols] -= sL[32 x 32] * sXp[32 x cols] on MFMA:
            // wave w owns cols [32w, 32w+32): 2 row-tiles x 2 col-tiles
            if (wave * 32 < cols) {
#pragma unroll
                for (int ct = 0; ct < 2; ++ct) {
                    const int cbase = wave * 32 + ct * 16;
                    f64x4 acc[2] = {f64x4{0, 0, 0, 0}, f64x4{0, 0, 0, 0}};
#pragma unroll
                    for (int kk = 0; kk < 8; ++kk) {
                        const int k = kk * 4 + fk;
                        const double b = sXp[cur][k][cbase + frow];
#pragma unroll
                        for (int rt = 0; rt < 2; ++rt)
                            acc[rt] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                                sL[cur][rt * 16 + frow][k], b, acc[rt], 0, 0,
                                0);
                    }
#pragma unroll
                    for (int rt = 0; rt < 2; ++rt)
#pragma unroll
                        for (int q = 0; q < 4; ++q)
                            sXj[rt * 16 + 4 * q + fk][cbase + frow] -=
                                acc[rt][q];
                }
            }
            if (pb + 32 < jb) write_bufs(cur ^ 1);
            __syncthreads();
            cur ^= 1;
        }
        for (int i = tid; i < 32 * 32; i += 256)
            sL[0][i >> 5][i & 31] =
                L[(int64_t)(jb + (i >> 5)) * ldl + jb + (i & 31)];
        __syncthreads();
        if (tid < cols) {  // diagonal solve: thread = column, unit lower
            double x[32];
#pragma unroll
            for (int r = 0; r < 32; ++r) x[r] = sXj[r][tid];
#pragma unroll
            for (int k = 0; k < 32; ++k) {
                const double xk = x[k];
#pragma unroll
                for (int r = 0; r < 32; ++r)
                    if (r > k) x[r] -= sL[0][r][k] * xk;
            }
#pragma unroll
            for (int r = 0; r < 32; ++r) sXj[r][tid] = x[r];
        }
        __syncthreads();
        for (int i = tid; i < 32 * TRM_COLS; i += 256)
            if ((i & (TRM_COLS - 1)) < cols)
                X[(int64_t)(jb + (i >> 7)) * ldx + c0 + (i & (TRM_COLS - 1))] =
                    sXj[i >> 7][i & (TRM_COLS - 1)];
    }
}

void launch_trsm_right_mfma(const double *U, int64_t ldu, double *X,
                            int64_t ldx, int v, int64_t M, int trans,
                            hipStream_t s) {
    if (M <= 0 || v <= 0) return;
    hipLaunchKernelGGL(k_trsm_right_mfma, dim3(cdiv64(M, TRM_ROWS)),
                       dim3(256), 0, s, U, ldu, X, ldx, v, M, trans);
}

void launch_trsm_left_mfma(const double *L, int64_t ldl, double *X,
                           int64_t ldx, int v, int64_t N, hipStream_t s) {
    if (N <= 0 || v <= 0) return;
    hipLaunchKernelGGL(k_trsm_left_mfma, dim3(cdiv64(N, TRM_COLS)), dim3(256),
                       0, s, L, ldl, X, ldx, v, N);
}

// validation helpers (the reference's CONFLUX_WITH_VALIDATION
// ||PA-LU||_F check, conflux_miniapp.cpp:169-507, without ScaLAPACK) --------
__global__ void k_tril_unit(const double *__restrict__ F, double *__restrict__ L,
                            int64_t n) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n * n; i += stride) {
        const int64_t r = i / n, c = i % n;
        L[i] = (r > c) ? F[i] : (r == c ? 1.0 : 0.0);
    }
}

__global__ void k_tril(const double *__restrict__ F, double *__restrict__ L,
                       int64_t n) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n * n; i += stride) {
        const int64_t r = i / n, c = i % n;
        L[i] = (r >= c) ? F[i] : 0.0;
    }
}

__global__ void k_triu(const double *__restrict__ F, double *__restrict__ U,
                       int64_t n) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n * n; i += stride) {
        const int64_t r = i / n, c = i % n;
        U[i] = (r <= c) ? F[i] : 0.0;
    }
}

__global__ void k_frob2(const double *__restrict__ A, int64_t n,
                        double *__restrict__ out) {
    __shared__ double part[256];
    const int tid = threadIdx.x;
    double s = 0;
    for (int64_t i = (int64_t)blockIdx.x * 256 + tid; i < n;
         i += (int64_t)gridDim.x * 256)
        s += A[i] * A[i];
    part[tid] = s;
    __syncthreads();
    for (int w = 128; w > 0; w >>= 1) {
        if (tid < w) part[tid] += part[tid + w];
        __syncthreads();
    }
    if (tid == 0) (void)atomicAdd(out, part[0]);
}

void launch_tril_unit(const double *F, double *L, int64_t n, hipStream_t s) {
    hipLaunchKernelGGL(k_tril_unit, dim3(cap_grid(n * n)), dim3(256), 0, s,
                       F, L, n);
}

// stripe variants for the streamed ||PA-LU|| validation: operate on a
// `rows`-row horizontal stripe whose first row is GLOBAL row `row0` of the
// factored matrix, so validation peaks at one N^2 buffer instead of five
// (bench-scale single-GPU sizes fit in HBM this way).
__global__ void k_tril_unit_rows(const double *__restrict__ F, int64_t ldf,
                                 double *__restrict__ L, int64_t ldl,
                                 int rows, int64_t row0, int64_t ncols) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < (int64_t)rows * ncols; i += stride) {
        const int64_t r = i / ncols, c = i % ncols, g = row0 + r;
        L[r * ldl + c] = (g > c) ? F[r * ldf + c] : (g == c ? 1.0 : 0.0);
    }
}

// zero the strict lower part of stripe rows in place (rows become pure U)
__global__ void k_triu_rows(double *__restrict__ F, int64_t ldf, int rows,
                            int64_t row0) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const int64_t w = row0 + rows;  // strict-lower cols are < global row < w
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < (int64_t)rows * w; i += stride) {
        const int64_t r = i / w, c = i % w;
        if (c < row0 + r) F[r * ldf + c] = 0.0;
    }
}

void launch_tril_unit_rows(const double *F, int64_t ldf, double *L,
                           int64_t ldl, int rows, int64_t row0, int64_t ncols,
                           hipStream_t s) {
    if (rows <= 0 || ncols <= 0) return;
    hipLaunchKernelGGL(k_tril_unit_rows, dim3(cap_grid((int64_t)rows * ncols)),
                       dim3(256), 0, s, F, ldf, L, ldl, rows, row0, ncols);
}

void launch_triu_rows(double *F, int64_t ldf, int rows, int64_t row0,
                      hipStream_t s) {
    if (rows <= 0) return;
    hipLaunchKernelGGL(k_triu_rows,
                       dim3(cap_grid((int64_t)rows * (row0 + rows))),
                       dim3(256), 0, s, F, ldf, rows, row0);
}
// A <- A + tril(A,-1)^T, i.e. mirror the strict lower triangle up (makes a
// lower-stored symmetric matrix explicit).  One thread per upper element.
__global__ void k_sym_mirror_up(double *__restrict__ A, int64_t n) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n * n; i += stride) {
        const int64_t r = i / n, c = i % n;
        if (r < c) A[i] = A[c * n + r];
    }
}

void launch_transpose_add_lower(double *A, int64_t n, hipStream_t s) {
    hipLaunchKernelGGL(k_sym_mirror_up, dim3(cap_grid(n * n)), dim3(256),
                       0, s, A, n);
}

void launch_tril(const double *F, double *L, int64_t n, hipStream_t s) {
    hipLaunchKernelGGL(k_tril, dim3(cap_grid(n * n)), dim3(256), 0, s, F,
                       L, n);
}
void launch_triu(const double *F, double *U, int64_t n, hipStream_t s) {
    hipLaunchKernelGGL(k_triu, dim3(cap_grid(n * n)), dim3(256), 0, s, F,
                       U, n);
}
void launch_frob2(const double *A, int64_t nelem, double *out, hipStream_t s) {
    int blocks = (int)cdiv64(nelem, 256 * 16);
    if (blocks > 4096) blocks = 4096;
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(k_frob2, dim3(blocks), dim3(256), 0, s, A, nelem, out);
}

void launch_potrf32(double *A, int64_t lda, int nb, hipStream_t s) {
    if (nb <= 0) return;
    hipLaunchKernelGGL(k_potrf32, dim3(1), dim3(256), 0, s, A, lda, nb);
}

void launch_getrf32_nopiv(double *A, int64_t lda, int nb, hipStream_t s) {
    if (nb <= 0) return;
    hipLaunchKernelGGL(k_getrf32_nopiv, dim3(1), dim3(256), 0, s, A, lda, nb);
}

int g_dgemm_variant = -1;  // 0 = 4-wave, 1 = 8-wave; env CONFLUX_GEMM_VARIANT

static int gemm_strip_w() {  // strip width in tiles; 0/1 = flat order
    static int w = -1;
    if (w < 0) {
        const char *e = getenv("CONFLUX_GEMM_STRIP");
        w = e ? atoi(e) : 8;
    }
    return w;
}

static int gemm_ntc() {  // non-temporal C epilogue (w8 kernel)
    static int v = -1;
    if (v < 0) {
        const char *e = getenv("CONFLUX_GEMM_NTC");
        v = e ? atoi(e) : 0;
    }
    return v;
}

void launch_dgemm_f64(const double *A, int64_t lda, const double *B,
                      int64_t ldb, double *C, int64_t ldc, int M, int64_t N,
                      int K, hipStream_t s, int maxwg) {
    if (M <= 0 || N <= 0 || K <= 0) return;
    if (g_dgemm_variant < 0) {
        const char *e = getenv("CONFLUX_GEMM_VARIANT");
        g_dgemm_variant = e ? atoi(e) : 1;
    }
    const int ntm = (int)cdiv64(M, GEMM_BM);
    const int ntn = (int)cdiv64(N, GEMM_BN);
    const int sw = gemm_strip_w();
    int nwg = ntm * ntn;
    // cap (persistent w8 kernel only): leave CUs free for a concurrent panel
    if (maxwg > 0 && g_dgemm_variant == 1 && nwg > maxwg) nwg = maxwg;
    if (g_dgemm_variant == 3) {
        const int ntm2 = (int)cdiv64(M, 256);
        hipLaunchKernelGGL(k_dgemm_f64_bm256, dim3(ntm2 * ntn), dim3(512), 0,
                           s, A, lda, B, ldb, C, ldc, M, N, K, ntm2, ntn, sw);
        return;
    }
    if (g_dgemm_variant == 2)
        hipLaunchKernelGGL(k_dgemm_f64_glds, dim3(ntm * ntn), dim3(512), 0, s,
                           A, lda, B, ldb, C, ldc, M, N, K, ntm, ntn, sw);
    else if (g_dgemm_variant == 1)
        hipLaunchKernelGGL(k_dgemm_f64_w8, dim3(nwg), dim3(512), 0, s,
                           A, lda, B, ldb, C, ldc, M, N, K, ntm, ntn, sw,
                           gemm_ntc());
    else
        hipLaunchKernelGGL(k_dgemm_f64, dim3(ntm * ntn), dim3(GEMM_TPB), 0, s,
                           A, lda, B, ldb, C, ldc, M, N, K, ntm, ntn, sw);
}

void launch_dgemm_f64_nt(const double *A, int64_t lda, const double *B,
                         int64_t ldb, double *C, int64_t ldc, int M, int64_t N,
                         int K, hipStream_t s) {
    if (M <= 0 || N <= 0 || K <= 0) return;
    const int ntm = (int)cdiv64(M, GEMM_BM);
    const int ntn = (int)cdiv64(N, GEMM_BN);
    hipLaunchKernelGGL(k_dgemm_f64_w8_nt, dim3(ntm * ntn), dim3(512), 0, s, A,
                       lda, B, ldb, C, ldc, M, N, K, ntm, ntn, gemm_strip_w(),
                       TriMask{0, 0, 0, 0, 0, 0, 0});
}

// Cholesky c4 single-launch variant: the trapezoid of v-tiles with global
// tile row >= tile col, launched as one rectangle with the TriMask.
// Requires v % 128 == 0 (caller falls back to per-tile launches otherwise).
void launch_dgemm_f64_nt_tril(const double *A, int64_t lda, const double *B,
                              int64_t ldb, double *C, int64_t ldc, int M,
                              int64_t N, int K, int v, int r0off,
                              int64_t c0off, int Px, int Py, int pi, int pj,
                              hipStream_t s) {
    if (M <= 0 || N <= 0 || K <= 0) return;
    const int ntm = (int)cdiv64(M, GEMM_BM);
    const int ntn = (int)cdiv64(N, GEMM_BN);
    hipLaunchKernelGGL(k_dgemm_f64_w8_nt, dim3(ntm * ntn), dim3(512), 0, s, A,
                       lda, B, ldb, C, ldc, M, N, K, ntm, ntn, gemm_strip_w(),
                       TriMask{v, r0off, c0off, Px, Py, pi, pj});
}

void launch_pack_candidate(const double *A10, int64_t lda, const int *gri,
                           int f, int n_src, int n_out, int v, const int *idx,
                           double *cand, hipStream_t s) {
    if (n_out <= 0) return;
    hipLaunchKernelGGL(k_pack_candidate,
                       dim3(cdiv64((int64_t)n_out * (v + 1), 256)), dim3(256),
                       0, s, A10, lda, gri, f, n_src, n_out, v, idx, cand);
}

void launch_row_move(const double *src, int64_t lds, double *dst, int64_t ldd,
                     const int *src_idx, const int *dst_idx, int n_rows,
                     int64_t cols, hipStream_t s) {
    if (n_rows <= 0 || cols <= 0) return;
    hipLaunchKernelGGL(k_row_move, dim3(cap_grid((int64_t)n_rows * cols)),
                       dim3(256), 0, s, src, lds, dst, ldd, src_idx, dst_idx,
                       n_rows, cols);
}

void launch_extract_col0_int(const double *cand, int stride, int n, int *out,
                             hipStream_t s) {
    if (n <= 0) return;
    hipLaunchKernelGGL(k_extract_col0_int, dim3(cdiv64(n, 256)), dim3(256), 0,
                       s, cand, stride, n, out);
}

void launch_slab_pack(const double *X, int64_t ldx, int n, int nlayr, int Pz,
                      double *out, hipStream_t s) {
    if (n <= 0) return;
    hipLaunchKernelGGL(k_slab_pack,
                       dim3(cdiv64((int64_t)n * nlayr * Pz, 256)), dim3(256),
                       0, s, X, ldx, n, nlayr, Pz, out);
}
