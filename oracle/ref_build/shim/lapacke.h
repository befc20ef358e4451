/* Minimal LAPACKE prototypes for libmkl_rt.so (LP64).  Only what
 * conflux_opt.hpp uses: LAPACKE_dgetrf (:158). */
#pragma once
#ifdef __cplusplus
extern "C" {
#endif

#define LAPACK_ROW_MAJOR 101
#define LAPACK_COL_MAJOR 102
typedef int lapack_int;

lapack_int LAPACKE_dpotrf(int matrix_layout, char uplo, lapack_int n,
                          double *a, lapack_int lda);

lapack_int LAPACKE_dgetrf(int matrix_layout, lapack_int m, lapack_int n,
                          double *a, lapack_int lda, lapack_int *ipiv);

#ifdef __cplusplus
}
#endif
