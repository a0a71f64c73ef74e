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
/* small-matrix real SVD via one-sided Jacobi (column-major, m==n only;
 * enough for the Npoly x Npoly pseudo-inverses of consensus_poly.c) */
int dgesvd_(char *jobu, char *jobvt, int *m_, int *n_, double *a,
            int *lda_, double *s, double *u, int *ldu_, double *vt,
            int *ldvt_, double *work, int *lwork_, int *info) {
  int m = *m_, n = *n_, lda = *lda_, ldu = *ldu_, ldvt = *ldvt_;
  *info = 0;
  if (*lwork_ == -1) { work[0] = 64; return 0; }
  if (m != n || n > 32) { *info = -1; return 0; }
  double V[32 * 32];
  for (int i = 0; i < n; i++)
    for (int j = 0; j < n; j++) V[i + j * n] = (i == j) ? 1.0 : 0.0;
  for (int sweep = 0; sweep < 60; sweep++) {
    double off = 0.0;
    for (int p = 0; p < n - 1; p++)
      for (int q = p + 1; q < n; q++) {
        double app = 0, aqq = 0, apq = 0;
        for (int i = 0; i < m; i++) {
          app += a[i + p * lda] * a[i + p * lda];
          aqq += a[i + q * lda] * a[i + q * lda];
          apq += a[i + p * lda] * a[i + q * lda];
        }
        off += apq * apq;
        if (fabs(apq) < 1e-300) continue;
        double tau = (aqq - app) / (2.0 * apq);
        double t = (tau >= 0 ? 1.0 : -1.0)
                   / (fabs(tau) + sqrt(1.0 + tau * tau));
        double c = 1.0 / sqrt(1.0 + t * t), sn = c * t;
        for (int i = 0; i < m; i++) {
          double x = a[i + p * lda], y = a[i + q * lda];
          a[i + p * lda] = c * x - sn * y;
          a[i + q * lda] = sn * x + c * y;
        }
        for (int i = 0; i < n; i++) {
          double x = V[i + p * n], y = V[i + q * n];
          V[i + p * n] = c * x - sn * y;
          V[i + q * n] = sn * x + c * y;
        }
      }
    if (off < 1e-30) break;
  }
  /* singular values + U columns; sort descending */
  int idx[32];
  for (int j = 0; j < n; j++) {
    double nn = 0;
    for (int i = 0; i < m; i++) nn += a[i + j * lda] * a[i + j * lda];
    s[j] = sqrt(nn);
    idx[j] = j;
  }
  for (int i = 0; i < n - 1; i++)
    for (int j = i + 1; j < n; j++)
      if (s[idx[j]] > s[idx[i]]) { int t2 = idx[i]; idx[i] = idx[j];
                                   idx[j] = t2; }
  double stmp[32];
  for (int k = 0; k < n; k++) stmp[k] = s[idx[k]];
  for (int k = 0; k < n; k++) {
    int j = idx[k];
    double inv = stmp[k] > 1e-300 ? 1.0 / stmp[k] : 0.0;
    for (int i = 0; i < m; i++) u[i + k * ldu] = a[i + j * lda] * inv;
    /* VT row k = V column j transposed */
    for (int i = 0; i < n; i++) vt[k + i * ldvt] = V[i + j * n];
  }
  for (int k = 0; k < n; k++) s[k] = stmp[k];
  (void)jobu; (void)jobvt;
  return 0;
}

#define STUB(name) __attribute__((weak)) void name() { \
  fprintf(stderr, "miniblas: " #name " not implemented (oracle)\n"); \
  abort(); }
STUB(cgels_) STUB(dgels_) STUB(dgeqrf_) STUB(dorgqr_)
STUB(dpotrf_) STUB(dpotrs_) STUB(dsyevx_) STUB(dtrtrs_) STUB(zgels_)
STUB(zgemm_) STUB(zgesvd_) STUB(zher_)
/* predict.c references shapelet_contrib (shapelet.c) — oracle covers
 * point/gaussian/disk/ring only */
STUB(shapelet_contrib)
/* pngoutput.c dependency of shapelet.c's plot helper — unused in oracle */
STUB(convert_tensor_to_image)
