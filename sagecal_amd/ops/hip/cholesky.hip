// Fused batched damped-Cholesky solve for gfx950:
//   dp = (JtJ + mu*I)^-1 Jtr   per problem, one workgroup per problem.
//
// Replaces rocSOLVER potrf/potrs for the LM normal equations (the
// reference used cusolverDn potrf/potrs per cluster, clmfit_cuda.c:364):
// rocSOLVER's batched fp32 potrf measures ~2 ms for [2,512,512] on MI355X;
// this kernel is built for low LATENCY at small batch (the LM inner loop
// is a serial chain of solves):
//   - 32x32 diagonal blocks factored WAVE-SYNCHRONOUSLY in registers:
//     lane r holds row r, pivot/column broadcasts via __shfl — zero
//     barriers, zero LDS traffic in the factor;
//   - the whole panel (rows k..n x 32) staged in LDS once per panel;
//     row-solve reads the diag block from LDS (broadcast-friendly);
//   - trailing SYRK update: per-thread 4x4 register tiles over the
//     LDS panel with float4 K-chunks.
// Requires n <= MAXN_CHOL (LDS panel); larger systems (512-station joint
// solves) go to rocSOLVER on the host side.
#include "common.h"

#define NB 32
#define PST 36   // padded panel row stride (floats): bank-conflict-free float4 reads
#define NTH 256
#define MAXN_CHOL 1024

extern "C" __global__ void __launch_bounds__(NTH)
k_chol_solve(const float* __restrict__ JtJ, const float* __restrict__ Jtr,
             const float* __restrict__ mu, int n,
             float* __restrict__ Lbuf, float* __restrict__ dp,
             int* __restrict__ info, int stages) {
  const int bid = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const float* A = JtJ + (size_t)bid * n * n;
  float* L = Lbuf + (size_t)bid * n * n;
  const float* b = Jtr + (size_t)bid * n;
  float* xo = dp + (size_t)bid * n;

  // dynamic LDS: pan[n][NB] (panel incl. diag block) + yv[NB]
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* pan = smem;                  // [rows][NB] rows = n - k
  __shared__ float yv[NB];
  __shared__ int bad;

  if (tid == 0) bad = 0;
  const float m = mu[bid];
  for (int idx = tid; idx < n * n; idx += NTH) {
    const int r = idx / n, c = idx - r * n;
    if (c <= r) L[idx] = A[idx] + (c == r ? m : 0.0f);
  }
  __syncthreads();
  if (stages == 0) return;       // ablation: copy only

  for (int k = 0; k < n; k += NB) {
    const int nb = NB;            // host pads n to a multiple of NB
    const int rows = n - k;
    if (stages == 1 && k > 0) break;   // ablation: one panel
    // stage panel rows k..n, cols k..k+nb into LDS
    for (int idx = tid; idx < rows * nb; idx += NTH) {
      const int r = idx / nb, c = idx - r * nb;
      pan[r * PST + c] = L[(size_t)(k + r) * n + k + c];
    }
    __syncthreads();
    // ---- wave-synchronous 32x32 factor: lanes 0..31 of wave 0 hold rows
    if (tid < 64) {
      const int r = lane & 31;
      float row[NB];
      if (lane < 32) {
#pragma unroll
        for (int c = 0; c < nb; ++c) row[c] = pan[r * PST + c];
      }
#pragma unroll
      for (int c = 0; c < NB; ++c) {
        // pivot from lane c
        float pv = __shfl(row[c], c, 64);
        if (lane == c) {
          if (pv <= 1e-30f) { bad = 1; pv = 1e-30f; }
          pv = sqrtf(pv);
          row[c] = pv;
        }
        pv = __shfl(row[c], c, 64);
        if (lane < 32 && r > c) {
          row[c] /= pv;
          // rank-1 update needs L[cc][c] for cc in (c, r]; get from the
          // lanes' own row[c] via shfl inside the cc loop
        }
        // all lanes update their trailing cols cc>c (only rows r>cc matter)
#pragma unroll
        for (int cc = c + 1; cc < NB; ++cc) {
          const float lcc = __shfl(row[c], cc, 64);
          if (lane < 32 && r >= cc) row[cc] -= row[c] * lcc;
        }
      }
      if (lane < 32) {
#pragma unroll
        for (int c = 0; c < nb; ++c) pan[r * PST + c] = row[c];
      }
    }
    __syncthreads();
    // ---- row-solve sub-panel rows nb..rows against the diag block
    for (int r = nb + tid; r < rows; r += NTH) {
      float rw[NB];
#pragma unroll
      for (int c = 0; c < NB; ++c) rw[c] = pan[r * PST + c];
#pragma unroll
      for (int c = 0; c < NB; ++c) {
        float s = rw[c];
#pragma unroll
        for (int c2 = 0; c2 < c; ++c2) s -= rw[c2] * pan[c * PST + c2];
        rw[c] = s / pan[c * PST + c];
      }
#pragma unroll
      for (int c = 0; c < NB; ++c) pan[r * PST + c] = rw[c];
    }
    __syncthreads();
    // write panel back (final L values)
    for (int idx = tid; idx < rows * nb; idx += NTH) {
      const int r = idx / nb, c = idx - r * nb;
      L[(size_t)(k + r) * n + k + c] = pan[r * PST + c];
    }
    // ---- trailing SYRK: L[i,j] -= dot(pan[i], pan[j]), 4x4 reg tiles
    const int rows2 = rows - nb;
    if (rows2 > 0) {
      const int ntI = (rows2 + 3) >> 2;
      const int ntiles = ntI * (ntI + 1) / 2;
      for (int tile = tid; tile < ntiles; tile += NTH) {
        // triangular decode: I = row-tile, Jt = col-tile (Jt <= I)
        int I = (int)((sqrtf(8.0f * tile + 1.0f) - 1.0f) * 0.5f);
        while (I * (I + 1) / 2 > tile) --I;
        while ((I + 1) * (I + 2) / 2 <= tile) ++I;
        const int Jt = tile - I * (I + 1) / 2;
        const float* pi = pan + (nb + I * 4) * PST;
        const float* pj = pan + (nb + Jt * 4) * PST;
        float acc[4][4] = {};
#pragma unroll
        for (int c = 0; c < NB; c += 4) {
          float4 a0 = *(const float4*)(pi + 0 * PST + c);
          float4 a1 = *(const float4*)(pi + 1 * PST + c);
          float4 a2 = *(const float4*)(pi + 2 * PST + c);
          float4 a3 = *(const float4*)(pi + 3 * PST + c);
          float4 b0 = *(const float4*)(pj + 0 * PST + c);
          float4 b1 = *(const float4*)(pj + 1 * PST + c);
          float4 b2 = *(const float4*)(pj + 2 * PST + c);
          float4 b3 = *(const float4*)(pj + 3 * PST + c);
          const float4 aa[4] = {a0, a1, a2, a3};
          const float4 bbv[4] = {b0, b1, b2, b3};
#pragma unroll
          for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              acc[i][j] += aa[i].x * bbv[j].x + aa[i].y * bbv[j].y
                         + aa[i].z * bbv[j].z + aa[i].w * bbv[j].w;
            }
        }
        const int gi0 = k + nb + I * 4, gj0 = k + nb + Jt * 4;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int gi = gi0 + i;
          if (gi >= n) break;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const int gj = gj0 + j;
            if (gj <= gi && gj < n)
              L[(size_t)gi * n + gj] -= acc[i][j];
          }
        }
      }
    }
    __syncthreads();
  }
  if (stages <= 2) return;       // ablation: factorization only

  // ---- forward substitution: L y = b (y in xo) ----
  for (int idx = tid; idx < n; idx += NTH) xo[idx] = b[idx];
  __syncthreads();
  for (int k = 0; k < n; k += NB) {
    const int nb = NB;
    for (int idx = tid; idx < nb * nb; idx += NTH) {
      const int r = idx / nb, c = idx - r * nb;
      if (c <= r) pan[r * PST + c] = L[(size_t)(k + r) * n + k + c];
    }
    if (tid < nb) yv[tid] = xo[k + tid];
    __syncthreads();
    if (tid == 0) {
      for (int c = 0; c < nb; ++c) {
        float s = yv[c];
        for (int c2 = 0; c2 < c; ++c2) s -= pan[c * PST + c2] * yv[c2];
        yv[c] = s / pan[c * PST + c];
      }
    }
    __syncthreads();
    if (tid < nb) xo[k + tid] = yv[tid];
    for (int i = k + nb + tid; i < n; i += NTH) {
      float s = 0.f;
#pragma unroll
      for (int c = 0; c < NB; ++c) s += L[(size_t)i * n + k + c] * yv[c];
      xo[i] -= s;
    }
    __syncthreads();
  }
  // ---- backward substitution: L^T x = y ----
  for (int k = ((n - 1) / NB) * NB; k >= 0; k -= NB) {
    const int nb = NB;
    for (int idx = tid; idx < nb * nb; idx += NTH) {
      const int r = idx / nb, c = idx - r * nb;
      if (c <= r) pan[r * PST + c] = L[(size_t)(k + r) * n + k + c];
    }
    if (tid < nb) yv[tid] = xo[k + tid];
    __syncthreads();
    if (tid == 0) {
      for (int c = nb - 1; c >= 0; --c) {
        float s = yv[c];
        for (int c2 = c + 1; c2 < nb; ++c2) s -= pan[c2 * PST + c] * yv[c2];
        yv[c] = s / pan[c * PST + c];
      }
    }
    __syncthreads();
    if (tid < nb) xo[k + tid] = yv[tid];
    __syncthreads();
    // update rows above: xo[i] -= L[k+c, i] * x[k+c] for i < k
    for (int i = tid; i < k; i += NTH) {
      float s = 0.f;
#pragma unroll 8
      for (int c = 0; c < nb; ++c) s += L[(size_t)(k + c) * n + i] * yv[c];
      xo[i] -= s;
    }
    __syncthreads();
  }
  if (bad) {
    // poison the step so the LM accept mask rejects it (no host sync)
    const float qn = __int_as_float(0x7fc00000);
    for (int idx = tid; idx < n; idx += NTH) xo[idx] = qn;
    if (tid == 0) atomicOr(&info[bid], 1);
  }
}
