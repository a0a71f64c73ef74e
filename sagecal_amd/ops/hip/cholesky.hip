// Fused batched damped-Cholesky solve for gfx950:
//   dp = (JtJ + mu*I)^-1 Jtr   per problem, one workgroup per problem.
//
// Replaces rocSOLVER potrf/potrs for the LM normal equations (the
// reference used cusolverDn potrf/potrs per cluster, clmfit_cuda.c:364) —
// rocSOLVER's batched fp32 potrf measures ~2 ms for [2,512,512] on MI355X
// (latency-bound internal loop); this kernel targets <0.15 ms.
//
// Algorithm: right-looking blocked Cholesky, panel width 32.
//   - copy lower triangle of JtJ + mu on diag into scratch L (global)
//   - per panel: factor 32x32 diag block in LDS; row-solve the sub-panel;
//     trailing SYRK update with the panel's J-tile staged in LDS
//   - blocked forward/backward substitution for the single RHS.
// Non-SPD pivots are clamped and flagged (info) — the LM accept/reject
// logic rejects the resulting step (mu grows), matching cholesky_ex
// semantics without host sync.
#include "common.h"

#define NB 32
#define NTH 256
#define JT 128   // J-tile rows staged in LDS for the trailing update

extern "C" __global__ void __launch_bounds__(NTH)
k_chol_solve(const float* __restrict__ JtJ, const float* __restrict__ Jtr,
             const float* __restrict__ mu, int n,
             float* __restrict__ Lbuf, float* __restrict__ dp,
             int* __restrict__ info) {
  const int bid = blockIdx.x;
  const int tid = threadIdx.x;
  const float* A = JtJ + (size_t)bid * n * n;
  float* L = Lbuf + (size_t)bid * n * n;
  const float* b = Jtr + (size_t)bid * n;
  float* xo = dp + (size_t)bid * n;

  __shared__ float dlds[NB][NB + 1];
  __shared__ float jl[JT][NB + 1];
  __shared__ float yv[NB];
  __shared__ int bad;

  if (tid == 0) bad = 0;
  const float m = mu[bid];
  // copy lower triangle + damping
  for (int idx = tid; idx < n * n; idx += NTH) {
    const int r = idx / n, c = idx - r * n;
    if (c <= r) L[idx] = A[idx] + (c == r ? m : 0.0f);
  }
  __syncthreads();

  for (int k = 0; k < n; k += NB) {
    const int nb = min(NB, n - k);
    // load diag block
    for (int idx = tid; idx < nb * nb; idx += NTH) {
      const int r = idx / nb, c = idx - r * nb;
      dlds[r][c] = L[(size_t)(k + r) * n + k + c];
    }
    __syncthreads();
    // factor 32x32 block (unblocked, in LDS)
    for (int c = 0; c < nb; ++c) {
      if (tid == 0) {
        float d = dlds[c][c];
        if (d <= 1e-30f) { bad = 1; d = 1e-30f; }
        dlds[c][c] = sqrtf(d);
      }
      __syncthreads();
      const float piv = dlds[c][c];
      for (int r = c + 1 + tid; r < nb; r += NTH) dlds[r][c] /= piv;
      __syncthreads();
      const int rem = nb - c - 1;
      for (int idx = tid; idx < rem * rem; idx += NTH) {
        const int rr = idx / rem, cc = idx - rr * rem;
        if (cc <= rr) {
          dlds[c + 1 + rr][c + 1 + cc] -=
              dlds[c + 1 + rr][c] * dlds[c + 1 + cc][c];
        }
      }
      __syncthreads();
    }
    // write factored diag block back
    for (int idx = tid; idx < nb * nb; idx += NTH) {
      const int r = idx / nb, c = idx - r * nb;
      if (c <= r) L[(size_t)(k + r) * n + k + c] = dlds[r][c];
    }
    // panel row-solve: L[k+nb:n, k:k+nb] = A_panel * Lkk^-T
    // row r: forward substitution against dlds (lower, transposed solve)
    for (int r = k + nb + tid; r < n; r += NTH) {
      float row[NB];
#pragma unroll 8
      for (int c = 0; c < nb; ++c) row[c] = L[(size_t)r * n + k + c];
      for (int c = 0; c < nb; ++c) {
        float s = row[c];
        for (int c2 = 0; c2 < c; ++c2) s -= row[c2] * dlds[c][c2];
        row[c] = s / dlds[c][c];
      }
#pragma unroll 8
      for (int c = 0; c < nb; ++c) L[(size_t)r * n + k + c] = row[c];
    }
    __syncthreads();
    // trailing update: for row-tiles J (cols) staged in LDS,
    // L[i, j] -= dot(panel[i], panel[j]) for k+nb <= j <= i < n
    for (int j0 = k + nb; j0 < n; j0 += JT) {
      const int jt = min(JT, n - j0);
      for (int idx = tid; idx < jt * nb; idx += NTH) {
        const int r = idx / nb, c = idx - r * nb;
        jl[r][c] = L[(size_t)(j0 + r) * n + k + c];
      }
      __syncthreads();
      // entries: i from j0.., j in tile, i >= j
      for (int i = j0 + tid / 32; i < n; i += NTH / 32) {
        float pr[NB];
#pragma unroll 8
        for (int c = 0; c < nb; ++c) pr[c] = L[(size_t)i * n + k + c];
        const int lane8 = tid % 32;
        const int jmax = min(jt, i - j0 + 1);
        for (int jj = lane8; jj < jmax; jj += 32) {
          float s = 0.f;
#pragma unroll 8
          for (int c = 0; c < nb; ++c) s += pr[c] * jl[jj][c];
          L[(size_t)i * n + j0 + jj] -= s;
        }
      }
      __syncthreads();
    }
  }

  // ---- forward substitution: solve L y = b (y kept in xo) ----
  for (int idx = tid; idx < n; idx += NTH) xo[idx] = b[idx];
  __syncthreads();
  for (int k = 0; k < n; k += NB) {
    const int nb = min(NB, n - k);
    // stage Lkk into LDS, then serial in-LDS triangular solve (fast)
    for (int idx = tid; idx < nb * nb; idx += NTH) {
      const int r = idx / nb, c = idx - r * nb;
      if (c <= r) dlds[r][c] = L[(size_t)(k + r) * n + k + c];
    }
    if (tid < nb) yv[tid] = xo[k + tid];
    __syncthreads();
    if (tid == 0) {
      for (int c = 0; c < nb; ++c) {
        float s = yv[c];
        for (int c2 = 0; c2 < c; ++c2) s -= dlds[c][c2] * yv[c2];
        yv[c] = s / dlds[c][c];
      }
    }
    __syncthreads();
    if (tid < nb) xo[k + tid] = yv[tid];
    for (int i = k + nb + tid; i < n; i += NTH) {
      float s = 0.f;
#pragma unroll 8
      for (int c = 0; c < nb; ++c) s += L[(size_t)i * n + k + c] * yv[c];
      xo[i] -= s;
    }
    __syncthreads();
  }
  // ---- backward substitution: solve L^T x = y ----
  for (int k = ((n - 1) / NB) * NB; k >= 0; k -= NB) {
    const int nb = min(NB, n - k);
    for (int idx = tid; idx < nb * nb; idx += NTH) {
      const int r = idx / nb, c = idx - r * nb;
      if (c <= r) dlds[r][c] = L[(size_t)(k + r) * n + k + c];
    }
    if (tid < nb) yv[tid] = xo[k + tid];
    __syncthreads();
    if (tid == 0) {
      for (int c = nb - 1; c >= 0; --c) {
        float s = yv[c];
        for (int c2 = c + 1; c2 < nb; ++c2) s -= dlds[c2][c] * yv[c2];
        yv[c] = s / dlds[c][c];
      }
    }
    __syncthreads();
    if (tid < nb) xo[k + tid] = yv[tid];
    __syncthreads();
    // update rows above: xo[i] -= L[k+c, i] * x[k+c] for i < k
    for (int i = tid; i < k; i += NTH) {
      float s = 0.f;
      for (int c = 0; c < nb; ++c) s += L[(size_t)(k + c) * n + i] * yv[c];
      xo[i] -= s;
    }
    __syncthreads();
  }
  if (tid == 0 && bad) atomicOr(&info[bid], 1);
}
