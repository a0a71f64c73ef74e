// Kernel launch wrappers + kernels in ONE translation unit (no -fgpu-rdc
// cross-TU device linking needed).
#include "common.h"
#include "predict.hip"
#include "jtj.hip"
#include "cholesky.hip"

extern "C" {

hipError_t launch_predict_coh(
    const double* u, const double* v, const double* w, const double* ll,
    const double* mm, const double* nn1, const float* sI, const float* sQ,
    const float* sU, const float* sV, const float* eX, const float* eY,
    const float* eP, const float* cxi, const float* sxi, const float* cphi,
    const float* sphi, const float* r1, const int* stype,
    const int* cluster_off, int M, int R, double freq, double fdelta2,
    double tdelta, const float* beam, const int* pairs, int Nbase,
    int Ktot, int Nsta, float2* out, hipStream_t stream) {
  const int tb = 128;
  const int nb = (R + tb - 1) / tb;
  hipLaunchKernelGGL(k_predict_coh, dim3(nb), dim3(tb), 0, stream,
      u, v, w, ll, mm, nn1, sI, sQ, sU, sV, eX, eY, eP, cxi, sxi, cphi,
      sphi, r1, stype, cluster_off, M, R, freq, fdelta2, tdelta,
      beam, pairs, Nbase, Ktot, Nsta, out);
  return hipGetLastError();
}

hipError_t launch_jtj_accum(
    const float2* x, const float2* coh, const float2* J, const int* pairs,
    const int* chunk_tab, const float* wts, int Nbase, int T, int N,
    int nseg, float2* D, float2* g, float2* Cx, float* cost, int npair,
    int grad_only, hipStream_t stream) {
  hipLaunchKernelGGL(k_jtj_accum, dim3(Nbase, nseg), dim3(64), 0, stream,
      x, coh, J, pairs, chunk_tab, wts, Nbase, T, N, D, g, Cx, cost, npair,
      grad_only);
  return hipGetLastError();
}

hipError_t launch_jtj_expand(
    const float2* D, const float2* g, const float2* Cx, const int* pidx,
    int N, int npair, int Mt, float* JtJ, float* Jtr, hipStream_t stream) {
  hipLaunchKernelGGL(k_jtj_expand, dim3(N, N, Mt), dim3(64), 0, stream,
      D, g, Cx, pidx, N, npair, JtJ, Jtr);
  return hipGetLastError();
}

hipError_t launch_model_cost(
    const float2* x, const float2* coh, const float2* J, const int* pairs,
    const int* chunk_tab, const float* wts, int Nbase, int T, int N,
    int nseg, float* cost, hipStream_t stream) {
  const long long B = (long long)nseg * T * Nbase;
  const int tb = 256;
  const int nb = (int)((B + tb - 1) / tb);
  hipLaunchKernelGGL(k_model_cost, dim3(nb), dim3(tb), 0, stream,
      x, coh, J, pairs, chunk_tab, wts, Nbase, T, N, nseg, cost);
  return hipGetLastError();
}

hipError_t launch_apply_jones(
    const float2* x, const float2* cohs, const float2* J, const int* pairs,
    const int* chunk_tab, int Nbase, int T, int N, int nseg, int M, int sub,
    float2* out, hipStream_t stream) {
  const long long B = (long long)nseg * T * Nbase;
  const int tb = 256;
  const int nb = (int)((B + tb - 1) / tb);
  hipLaunchKernelGGL(k_apply_jones, dim3(nb), dim3(tb), 0, stream,
      x, cohs, J, pairs, chunk_tab, Nbase, T, N, nseg, M, sub, out);
  return hipGetLastError();
}

hipError_t launch_chol_solve(
    const float* JtJ, const float* Jtr, const float* mu, int n, int batch,
    float* Lbuf, float* dp, int* info, int stages, hipStream_t stream) {
  const size_t shmem = (size_t)(n + 40) * PST * sizeof(float);
  hipLaunchKernelGGL(k_chol_solve, dim3(batch), dim3(512), shmem, stream,
      JtJ, Jtr, mu, n, Lbuf, dp, info, stages);
  return hipGetLastError();
}

// multi-workgroup right-looking variant: kernel sequence on one stream
// (graph-capturable). See cholesky.hip for the rationale.
hipError_t launch_chol_mw(
    const float* JtJ, const float* Jtr, const float* mu, int n, int batch,
    float* Lbuf, float* dp, int* info, int stages, hipStream_t stream) {
  hipLaunchKernelGGL(k_cholmw_init, dim3(batch, INIT_SLICES), dim3(512), 0,
      stream, JtJ, Jtr, mu, n, Lbuf, dp);
  if (stages < 2) return hipGetLastError();
  // n <= PANEL_CAP: the hardware-validated single-pass panel kernel;
  // larger n: the chunked-panel variant (row passes through LDS)
  const int cap = n < PANEL_CAP ? n : PANEL_CAP;
  const size_t shmem = (size_t)(cap + 8) * PST * sizeof(float);
  for (int k = 0; k < n; k += NB) {
    if (n <= PANEL_CAP) {
      hipLaunchKernelGGL(k_cholmw_panel, dim3(batch), dim3(512), shmem,
          stream, n, k, Lbuf, dp, info);
    } else {
      hipLaunchKernelGGL(k_cholmw_panel_big, dim3(batch), dim3(512),
          shmem, stream, n, k, Lbuf, dp, info);
    }
    const int tcnt = (n - k - NB) / NB;
    const int ntiles = tcnt * (tcnt + 1) / 2;
    if (ntiles > 0 && stages >= 3) {
      hipLaunchKernelGGL(k_cholmw_syrk, dim3(batch * ntiles), dim3(64), 0,
          stream, n, k, ntiles, Lbuf);
    }
  }
  if (stages < 4) return hipGetLastError();
  hipLaunchKernelGGL(k_cholmw_subst, dim3(batch), dim3(512), 0, stream,
      n, Lbuf, dp, info);
  return hipGetLastError();
}
}  // extern "C"
