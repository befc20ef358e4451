/* Minimal CBLAS prototypes for linking the reference LU loop against
 * /opt/conda/lib/libmkl_rt.so (no cblas.h ships in this container).
 * Only the symbols conflux_opt.hpp uses: cblas_dgemm (:1628), cblas_dtrsm
 * (:1347, :1539).  Enum values are the standard CBLAS ABI constants. */
#pragma once
/* the reference's non-MKL branch includes only <cblas.h> but also calls
 * LAPACKE_* (it was evidently built with MKL upstream) — chain the shim */
#include "lapacke.h"
#ifdef __cplusplus
extern "C" {
#endif

typedef enum { CblasRowMajor = 101, CblasColMajor = 102 } CBLAS_LAYOUT;
typedef enum { CblasNoTrans = 111, CblasTrans = 112, CblasConjTrans = 113 } CBLAS_TRANSPOSE;
typedef enum { CblasUpper = 121, CblasLower = 122 } CBLAS_UPLO;
typedef enum { CblasNonUnit = 131, CblasUnit = 132 } CBLAS_DIAG;
typedef enum { CblasLeft = 141, CblasRight = 142 } CBLAS_SIDE;
typedef CBLAS_LAYOUT CBLAS_ORDER;

void cblas_dgemm(CBLAS_LAYOUT layout, CBLAS_TRANSPOSE TransA,
                 CBLAS_TRANSPOSE TransB, int M, int N, int K, double alpha,
                 const double *A, int lda, const double *B, int ldb,
                 double beta, double *C, int ldc);

void cblas_dsyrk(CBLAS_LAYOUT layout, CBLAS_UPLO Uplo, CBLAS_TRANSPOSE Trans,
                 int N, int K, double alpha, const double *A, int lda,
                 double beta, double *C, int ldc);

void cblas_dtrsm(CBLAS_LAYOUT layout, CBLAS_SIDE Side, CBLAS_UPLO Uplo,
                 CBLAS_TRANSPOSE TransA, CBLAS_DIAG Diag, int M, int N,
                 double alpha, const double *A, int lda, double *B, int ldb);

#ifdef __cplusplus
}
#endif
