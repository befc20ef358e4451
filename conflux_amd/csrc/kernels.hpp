// Host-side launch API of the HIP kernels (kernels.hip).
#pragma once
#include <hip/hip_runtime.h>

#include <cstdint>

#define CONFLUX_PANEL_MAX_BLOCKS 256

void launch_init_matrix(double *A, int Ml, int Nl, int v, int Px, int Py,
                        int pi, int pj, int zero_layer, uint64_t seed,
                        hipStream_t s);
void launch_copy2d(const double *src, int64_t lds, double *dst, int64_t ldd,
                   int rows, int64_t cols, hipStream_t s);
void launch_zero2d(double *dst, int64_t ldd, int rows, int64_t cols,
                   hipStream_t s);
void launch_add2d(const double *src, int64_t lds, double *dst, int64_t ldd,
                  int rows, int64_t cols, hipStream_t s);
void launch_row_gather(const double *src, int64_t lds, double *dst,
                       int64_t ldd, const int *idx, int n_rows, int64_t cols,
                       hipStream_t s);
void launch_row_scatter(const double *src, int64_t lds, double *dst,
                        int64_t ldd, const int *idx, int n_rows, int64_t cols,
                        hipStream_t s);
void launch_rowperm_skip(double *mat, int64_t ld, const int *dst_idx,
                         const int *src_idx, int row_base, int n_rows,
                         int64_t skip0, int64_t skipn, int64_t tot_cols,
                         double *tmp, hipStream_t s);
int launch_panel_factor(double *panel, int64_t ldp, int m, int nb, void *sync,
                        int *ipiv, unsigned int epoch0, int *swap_dst,
                        int *swap_src, hipStream_t s);
int conflux_panel_sync_bytes();
void conflux_panel_spin_read(void *sync, unsigned long long *out,
                             hipStream_t s);
int conflux_panel_nb();
int conflux_panel_rpb();
int conflux_panel_blocks_per_cu();
void launch_trsm_left_lower_unit32(const double *L, int64_t ldl, double *X,
                                   int64_t ldx, int nb, int64_t N,
                                   hipStream_t s);
void launch_trsm_right_upper32(const double *U, int64_t ldu, double *X,
                               int64_t ldx, int nb, int64_t M, int trans,
                               hipStream_t s);
void launch_trsm_right_mfma(const double *U, int64_t ldu, double *X,
                            int64_t ldx, int v, int64_t M, int trans,
                            hipStream_t s);
void launch_trsm_left_mfma(const double *L, int64_t ldl, double *X,
                           int64_t ldx, int v, int64_t N, hipStream_t s);
void launch_tril_unit(const double *F, double *L, int64_t n, hipStream_t s);
void launch_tril_unit_rows(const double *F, int64_t ldf, double *L,
                           int64_t ldl, int rows, int64_t row0, int64_t ncols,
                           hipStream_t s);
void launch_triu_rows(double *F, int64_t ldf, int rows, int64_t row0,
                      hipStream_t s);
void launch_transpose_add_lower(double *A, int64_t n, hipStream_t s);
void launch_tril(const double *F, double *L, int64_t n, hipStream_t s);
void launch_triu(const double *F, double *U, int64_t n, hipStream_t s);
void launch_frob2(const double *A, int64_t nelem, double *out, hipStream_t s);
void launch_potrf32(double *A, int64_t lda, int nb, hipStream_t s);
void launch_getrf32_nopiv(double *A, int64_t lda, int nb, hipStream_t s);
void launch_dgemm_f64_nt(const double *A, int64_t lda, const double *B,
                         int64_t ldb, double *C, int64_t ldc, int M, int64_t N,
                         int K, hipStream_t s);
void launch_dgemm_f64_nt_tril(const double *A, int64_t lda, const double *B,
                              int64_t ldb, double *C, int64_t ldc, int M,
                              int64_t N, int K, int v, int r0off,
                              int64_t c0off, int Px, int Py, int pi, int pj,
                              hipStream_t s);
void launch_init_matrix_spd(double *A, int Ml, int Nl, int v, int Px, int Py,
                            int pi, int pj, int zero_layer, uint64_t seed,
                            int64_t Nglob, hipStream_t s);
void launch_dgemm_f64(const double *A, int64_t lda, const double *B,
                      int64_t ldb, double *C, int64_t ldc, int M, int64_t N,
                      int K, hipStream_t s,
                      int maxwg = 0);
void launch_pack_candidate(const double *A10, int64_t lda, const int *gri,
                           int f, int n_src, int n_out, int v, const int *idx,
                           double *cand, hipStream_t s);
void launch_row_move(const double *src, int64_t lds, double *dst, int64_t ldd,
                     const int *src_idx, const int *dst_idx, int n_rows,
                     int64_t cols, hipStream_t s);
void launch_extract_col0_int(const double *cand, int stride, int n, int *out,
                             hipStream_t s);
void launch_slab_pack(const double *X, int64_t ldx, int n, int nlayr, int Pz,
                      double *out, hipStream_t s);
