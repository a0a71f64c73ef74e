/* Minimal Fortran-convention BLAS for linking the reference's CPU
 * libdirac objects (myblas.c / lbfgs.c / predict.c) as a correctness
 * oracle. The image has no CPU BLAS/LAPACK; the oracle
 * paths exercised (LBFGS Rosenbrock, coherency predict) only use the
 * level-1 routines implemented here. LAPACK factorizations referenced
 * by unused myblas wrappers are abort stubs so the link succeeds but
 * accidental use fails loudly. Test harness only — not framework code.
 */
#include <stdio.h>
#include <stdlib.h>
#include <math.h>
#include <complex.h>

double dasum_(int *n, double *x, int *inc) {
  double s = 0.0;
  for (int i = 0; i < *n; i++) s += fabs(x[i * *inc]);
  return s;
}
void daxpy_(int *n, double *a, double *x, int *incx, double *y,
            int *incy) {
  for (int i = 0; i < *n; i++) y[i * *incy] += *a * x[i * *incx];
}
void dcopy_(int *n, double *x, int *incx, double *y, int *incy) {
  for (int i = 0; i < *n; i++) y[i * *incy] = x[i * *incx];
}
double ddot_(int *n, double *x, int *incx, double *y, int *incy) {
  double s = 0.0;
  for (int i = 0; i < *n; i++) s += x[i * *incx] * y[i * *incy];
  return s;
}
double dnrm2_(int *n, double *x, int *inc) {
  double s = 0.0;
  for (int i = 0; i < *n; i++) s += x[i * *inc] * x[i * *inc];
  return sqrt(s);
}
void dscal_(int *n, double *a, double *x, int *inc) {
  for (int i = 0; i < *n; i++) x[i * *inc] *= *a;
}
int idamax_(int *n, double *x, int *inc) {
  int best = 1; double bv = -1.0;
  for (int i = 0; i < *n; i++) {
    double v = fabs(x[i * *inc]);
    if (v > bv) { bv = v; best = i + 1; }   /* 1-based */
  }
  return best;
}
int isamax_(int *n, float *x, int *inc) {
  int best = 1; float bv = -1.0f;
  for (int i = 0; i < *n; i++) {
    float v = fabsf(x[i * *inc]);
    if (v > bv) { bv = v; best = i + 1; }
  }
  return best;
}
float sasum_(int *n, float *x, int *inc) {
  float s = 0.0f;
  for (int i = 0; i < *n; i++) s += fabsf(x[i * *inc]);
  return s;
}
void saxpy_(int *n, float *a, float *x, int *incx, float *y, int *incy) {
  for (int i = 0; i < *n; i++) y[i * *incy] += *a * x[i * *incx];
}
void scopy_(int *n, float *x, int *incx, float *y, int *incy) {
  for (int i = 0; i < *n; i++) y[i * *incy] = x[i * *incx];
}
float snrm2_(int *n, float *x, int *inc) {
  float s = 0.0f;
  for (int i = 0; i < *n; i++) s += x[i * *inc] * x[i * *inc];
  return sqrtf(s);
}
void sscal_(int *n, float *a, float *x, int *inc) {
  for (int i = 0; i < *n; i++) x[i * *inc] *= *a;
}
double dznrm2_(int *n, double complex *x, int *inc) {
  double s = 0.0;
  for (int i = 0; i < *n; i++) {
    double re = creal(x[i * *inc]), im = cimag(x[i * *inc]);
    s += re * re + im * im;
  }
  return sqrt(s);
}
void zcopy_(int *n, double complex *x, int *incx, double complex *y,
            int *incy) {
  for (int i = 0; i < *n; i++) y[i * *incy] = x[i * *incx];
}
void zaxpy_(int *n, double complex *a, double complex *x, int *incx,
            double complex *y, int *incy) {
  for (int i = 0; i < *n; i++) y[i * *incy] += *a * x[i * *incx];
}
void zscal_(int *n, double complex *a, double complex *x, int *inc) {
  for (int i = 0; i < *n; i++) x[i * *inc] *= *a;
}
double complex zdotc_(int *n, double complex *x, int *incx,
                      double complex *y, int *incy) {
  double complex s = 0.0;
  for (int i = 0; i < *n; i++) s += conj(x[i * *incx]) * y[i * *incy];
  return s;
}
double dlamch_(char *c) {
  return (*c == 'E' || *c == 'e') ? 2.220446049250313e-16 : 0.0;
}
/* simple reference dgemm/dgemv (column-major) — rarely hot in oracle */
void dgemv_(char *trans, int *m, int *n, double *alpha, double *a,
            int *lda, double *x, int *incx, double *beta, double *y,
            int *incy) {
  int M = *m, N = *n;
  if (*trans == 'N' || *trans == 'n') {
    for (int i = 0; i < M; i++) {
      double s = 0.0;
      for (int j = 0; j < N; j++) s += a[i + j * *lda] * x[j * *incx];
      y[i * *incy] = *alpha * s + *beta * y[i * *incy];
    }
  } else {
    for (int j = 0; j < N; j++) {
      double s = 0.0;
      for (int i = 0; i < M; i++) s += a[i + j * *lda] * x[i * *incx];
      y[j * *incy] = *alpha * s + *beta * y[j * *incy];
    }
  }
}
void dgemm_(char *ta, char *tb, int *m, int *n, int *k, double *alpha,
            double *a, int *lda, double *b, int *ldb, double *beta,
            double *c, int *ldc) {
  int M = *m, N = *n, K = *k;
  int na = (*ta == 'N' || *ta == 'n'), nb = (*tb == 'N' || *tb == 'n');
  for (int j = 0; j < N; j++)
    for (int i = 0; i < M; i++) {
      double s = 0.0;
      for (int l = 0; l < K; l++) {
        double av = na ? a[i + l * *lda] : a[l + i * *lda];
        double bv = nb ? b[l + j * *ldb] : b[j + l * *ldb];
        s += av * bv;
      }
      c[i + j * *ldc] = *alpha * s + *beta * c[i + j * *ldc];
    }
}
/* weak: a real implementation linked alongside (e.g. shapelet.o for
 * oracle_misc) overrides the stub */
#define STUB(name) __attribute__((weak)) void name() { \
  fprintf(stderr, "miniblas: " #name " not implemented (oracle)\n"); \
  abort(); }
STUB(cgels_) STUB(dgels_) STUB(dgeqrf_) STUB(dgesvd_) STUB(dorgqr_)
STUB(dpotrf_) STUB(dpotrs_) STUB(dsyevx_) STUB(dtrtrs_) STUB(zgels_)
STUB(zgemm_) STUB(zgesvd_) STUB(zher_)
/* predict.c references shapelet_contrib (shapelet.c) — oracle covers
 * point/gaussian/disk/ring only */
STUB(shapelet_contrib)
/* pngoutput.c dependency of shapelet.c's plot helper — unused in oracle */
STUB(convert_tensor_to_image)
