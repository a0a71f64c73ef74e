// Fused batched damped-Cholesky solve for gfx950:
//   dp = (JtJ + mu*I)^-1 Jtr   per problem, one workgroup per problem.
//
// Replaces rocSOLVER potrf/potrs for the LM normal equations (the
// reference used cusolverDn potrf/potrs per cluster, clmfit_cuda.c:364):
// rocSOLVER's batched fp32 potrf measures ~2 ms for [2,512,512] on MI355X.
// This kernel is built for low LATENCY at small batch (the LM inner loop is
// a serial chain of solves). A single workgroup is bandwidth-limited, so
// the design minimizes global traffic and keeps every hot loop
// compile-time unrolled (n padded to a multiple of 32 by the host):
//   - float4-vectorized triangle copy with fused damping;
//   - 32x32 diagonal blocks factored WAVE-SYNCHRONOUSLY in registers
//     (lane r holds row r; pivot broadcasts via __shfl; no barriers);
//   - whole panel staged in LDS (padded stride, conflict-free float4);
//   - trailing SYRK in 8x8 register tiles with float4 K-chunks and
//     vectorized read-modify-write of the trailing matrix;
//   - both triangular substitutions wave-synchronous in-panel, with an
//     L^T mirror written during the panel write-back so the backward pass
//     reads rows (coalesced) instead of columns.
// Requires n <= MAXN_CHOL; larger systems go to rocSOLVER host-side.
// NOTE: Lbuf scratch is 2*n*n floats per problem (L and L^T).
#include "common.h"

#define NB 32
#define PST 36   // padded LDS panel row stride (floats)
#define NTH 512
#define MAXN_CHOL 1024
// multi-workgroup path row-chunking: panels taller than PANEL_CAP rows
// stream through LDS in passes, lifting the mw size cap to MAXN_CHOL_MW
// (the 512-station 8N=4096 normal equations, clmfit_cuda.c:1624-1674)
#define PANEL_CAP 1024
#define MAXN_CHOL_MW 4096

extern "C" __global__ void __launch_bounds__(NTH)
k_chol_solve(const float* __restrict__ JtJ, const float* __restrict__ Jtr,
             const float* __restrict__ mu, int n,
             float* __restrict__ Lbuf, float* __restrict__ dp,
             int* __restrict__ info, int stages) {
  const int bid = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const float* A = JtJ + (size_t)bid * n * n;
  float* L = Lbuf + (size_t)bid * 2 * n * n;       // [L | L^T] scratch
  float* LT = L + (size_t)n * n;
  const float* b = Jtr + (size_t)bid * n;
  float* xo = dp + (size_t)bid * n;

  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* pan = smem;                  // [rows][PST]
  __shared__ float yv[NB];
  __shared__ int bad;

  if (tid == 0) bad = 0;
  const float m = mu[bid];
  // no A->L copy: left-looking panels stage straight from the damped A;
  // L/LT only ever hold FACTORED panels.
  if (stages == 0) return;

  for (int k = 0; k < n; k += NB) {
    const int rows = n - k;
    if (stages == 1 && k > 0) break;
    // stage S = damped A panel rows k..n, cols k..k+NB into LDS (float4)
    for (int idx = tid; idx < rows * (NB / 4); idx += NTH) {
      const int r = idx >> 3, c4 = (idx & 7) << 2;
      float4 v = *(const float4*)(A + (size_t)(k + r) * n + k + c4);
      const int d = r - c4;
      if (d >= 0 && d < 4) {     // diagonal falls in this quad
        if (d == 0) v.x += m;
        else if (d == 1) v.y += m;
        else if (d == 2) v.z += m;
        else v.w += m;
      }
      *(float4*)(pan + r * PST + c4) = v;
    }
    __syncthreads();
    // ---- LEFT-LOOKING update: S -= Lp(rows x k) * Lc(32 x k)^T
    // MFMA f32 16x16x4 (exact f32, 157 TF class): both fragments load
    // COALESCED from the LT mirror (A[i][k'] = LT[k'][k+I16+i],
    // B[k'][c] = LT[k'][k+c]) — one dword per lane per operand per MFMA,
    // ~6x fewer load-issue slots than scalar float4 tiles (the previous
    // version was load-issue-bound on a single CU).
    if (k > 0 && stages != 4) {
      const int w = tid >> 6;              // wave id (NTH/64 waves)
      const int l15 = lane & 15, l4 = lane >> 4;
      const int ntI = (rows + 15) >> 4;
      const int ntiles = ntI * 2;          // 2 col-tiles of 16
      for (int tile = w; tile < ntiles; tile += NTH / 64) {
        const int I = tile >> 1, Jt = tile & 1;
        const int arow = k + I * 16 + l15;       // S row this lane loads
        const int bcol = k + Jt * 16 + l15;
        const bool aok = (I * 16 + l15) < rows;
        using f32x4 = __attribute__((ext_vector_type(4))) float;
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        const float* LTb = LT + (size_t)l4 * n;  // + kk*n walks K
        // software-pipelined K loop: batch 4 MFMAs' loads ahead so L2
        // latency hides under the MFMA chain
        int kk = 0;
        for (; kk + 16 <= k; kk += 16) {
          float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
          if (aok) {
            a0 = LTb[(size_t)kk * n + arow];
            a1 = LTb[(size_t)(kk + 4) * n + arow];
            a2 = LTb[(size_t)(kk + 8) * n + arow];
            a3 = LTb[(size_t)(kk + 12) * n + arow];
          }
          const float b0 = LTb[(size_t)kk * n + bcol];
          const float b1 = LTb[(size_t)(kk + 4) * n + bcol];
          const float b2 = LTb[(size_t)(kk + 8) * n + bcol];
          const float b3 = LTb[(size_t)(kk + 12) * n + bcol];
          acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc, 0, 0, 0);
          acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc, 0, 0, 0);
          acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a2, b2, acc, 0, 0, 0);
          acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a3, b3, acc, 0, 0, 0);
        }
        for (; kk < k; kk += 4) {
          const float a = aok ? LTb[(size_t)kk * n + arow] : 0.f;
          const float bfrag = LTb[(size_t)kk * n + bcol];
          acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bfrag, acc, 0, 0, 0);
        }
        // C/D layout: col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int r = I * 16 + l4 * 4 + reg;
          if (r < rows) pan[r * PST + Jt * 16 + l15] -= acc[reg];
        }
      }
      __syncthreads();
    }
    // ---- wave-synchronous 32x32 factor on wave 0: lane r holds row r;
    // the scaled pivot column is published through LDS (yv) once per step
    // (in-wave LDS write->read ordering; broadcast reads are 1 cycle) —
    // ~10x fewer DS ops than a shuffle-based rank-1 update.
    if (tid < 64 && lane < 32) {
      const int r = lane;
      float row[NB];
#pragma unroll
      for (int c = 0; c < NB; ++c) row[c] = pan[r * PST + c];
#pragma unroll
      for (int c = 0; c < NB; ++c) {
        // pivot broadcast via shfl (a divergent LDS write followed by a
        // uniform read is NOT ordered: the compiler emits the else-path
        // read before the taken-path write)
        float pivraw = __shfl(row[c], c, 64);
        if (pivraw <= 1e-30f) {
          if (lane == 0) bad = 1;
          pivraw = 1e-30f;
        }
        const float pv = sqrtf(pivraw);
        if (r >= c) row[c] /= pv;
        yv[r] = row[c];              // non-divergent publish (one ds_write)
        const float lrc = row[c];
#pragma unroll
        for (int cc = c + 1; cc < NB; ++cc) {
          if (r >= cc) row[cc] -= lrc * yv[cc];
        }
      }
#pragma unroll
      for (int c = 0; c < NB; ++c) pan[r * PST + c] = row[c];
    }
    __syncthreads();
    // ---- row-solve sub-panel rows NB..rows against the diag block
    for (int r = NB + tid; stages != 5 && r < rows; r += NTH) {
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
    // write panel back: L (row-major, float4) and the L^T mirror
    // (LT writes coalesced: r varies fastest within a column)
    for (int idx = tid; idx < rows * (NB / 4); idx += NTH) {
      const int r = idx >> 3, c4 = (idx & 7) << 2;
      *(float4*)(L + (size_t)(k + r) * n + k + c4) =
          *(const float4*)(pan + r * PST + c4);
    }
    if (stages != 6) {
#pragma unroll
      for (int c = 0; c < NB; ++c) {
        for (int r = tid; r < rows; r += NTH) {
          LT[(size_t)(k + c) * n + k + r] = pan[r * PST + c];
        }
      }
    }
    __syncthreads();
  }
  if (stages <= 2 || stages >= 4) return;

  // ---- forward substitution: L y = b (y in xo), wave-sync in-panel ----
  for (int idx = tid; idx < n; idx += NTH) xo[idx] = b[idx];
  __syncthreads();
  for (int k = 0; k < n; k += NB) {
    // stage diag block
    for (int idx = tid; idx < NB * NB; idx += NTH) {
      const int r = idx >> 5, c = idx & 31;
      if (c <= r) pan[r * PST + c] = L[(size_t)(k + r) * n + k + c];
    }
    __syncthreads();
    if (tid < 64) {
      const int r = lane & 31;
      float rv[NB];
      float bv = (lane < 32) ? xo[k + r] : 0.f;
      if (lane < 32) {
#pragma unroll
        for (int c = 0; c < NB; ++c) rv[c] = (c <= r) ? pan[r * PST + c] : 0.f;
      }
#pragma unroll
      for (int c = 0; c < NB; ++c) {
        float yc;
        if (lane == c) bv /= rv[c];
        yc = __shfl(bv, c, 64);
        if (lane < 32 && r > c) bv -= rv[c] * yc;
        if (lane == c) yv[c] = bv;
      }
      if (lane < 32) xo[k + r] = bv;
    }
    __syncthreads();
    for (int i = k + NB + tid; i < n; i += NTH) {
      float s = 0.f;
      const float* Lr = L + (size_t)i * n + k;
#pragma unroll
      for (int c = 0; c < NB; ++c) s += Lr[c] * yv[c];
      xo[i] -= s;
    }
    __syncthreads();
  }
  // ---- backward substitution: L^T x = y, reading LT rows (coalesced) ----
  for (int k = ((n - 1) / NB) * NB; k >= 0; k -= NB) {
    // stage diag block of LT (upper-tri rows: LT[r][c]=L[c][r], c >= r)
    for (int idx = tid; idx < NB * NB; idx += NTH) {
      const int r = idx >> 5, c = idx & 31;
      if (c >= r) pan[r * PST + c] = LT[(size_t)(k + r) * n + k + c];
    }
    __syncthreads();
    if (tid < 64) {
      const int r = lane & 31;
      float rv[NB];
      float bv = (lane < 32) ? xo[k + r] : 0.f;
      if (lane < 32) {
#pragma unroll
        for (int c = 0; c < NB; ++c) rv[c] = (c >= r) ? pan[r * PST + c] : 0.f;
      }
#pragma unroll
      for (int ci = 0; ci < NB; ++ci) {
        const int c = NB - 1 - ci;
        float xc;
        if (lane == c) bv /= rv[c];
        xc = __shfl(bv, c, 64);
        if (lane < 32 && r < c) bv -= rv[c] * xc;
        if (lane == c) yv[c] = bv;
      }
      if (lane < 32) xo[k + r] = bv;
    }
    __syncthreads();
    // update rows above: xo[i] -= LT[i, k..k+NB) . x[k..k+NB)  for i < k
    for (int i = tid; i < k; i += NTH) {
      float s = 0.f;
      const float* Lr = LT + (size_t)i * n + k;
#pragma unroll
      for (int c = 0; c < NB; ++c) s += Lr[c] * yv[c];
      xo[i] -= s;
    }
    __syncthreads();
  }
  if (bad) {
    const float qn = __int_as_float(0x7fc00000);
    for (int idx = tid; idx < n; idx += NTH) xo[idx] = qn;
    if (tid == 0) atomicOr(&info[bid], 1);
  }
}

// ======================================================================
// Multi-workgroup right-looking variant.
//
// The single-kernel solver above runs ONE workgroup per problem: at the
// LM batch sizes of the headline config (B≈5, n=512) that is <2% of the
// 256 CUs and k_chol_solve measures ~65% of all GPU time (753 us/call,
// profiles/lm_kernel_stats_final.csv). Here the factorization is split
// into a kernel SEQUENCE on one stream (graph-captured, so launch cost
// is node dispatch only):
//   k_cholmw_init   — L := A + mu*I (working copy; right-looking mutates)
//   per panel k:
//     k_cholmw_panel — factor 32x32 diag + row-solve the panel
//                      (1 WG/problem; serial critical path)
//     k_cholmw_syrk  — trailing update: ONE 32x32 MFMA tile per 64-lane
//                      wave, grid = batch * n_lower_tiles — the O(n^3)
//                      work spreads over the whole chip instead of 1 CU
//   k_cholmw_subst  — both triangular substitutions + NaN poison
// Same scratch contract as k_chol_solve (2*n*n: L | L^T mirror).
// ======================================================================

// grid (batch, SLICES): B workgroups alone leave the copy bandwidth-bound
// on B CUs (~75 us for [5,512,512]); sliced over 32 WGs/problem it is ~8 us.
// Also seeds dp := b (the panel kernels run the forward substitution
// in-place as each panel factors).
#define INIT_SLICES 32
extern "C" __global__ void __launch_bounds__(NTH)
k_cholmw_init(const float* __restrict__ JtJ, const float* __restrict__ Jtr,
              const float* __restrict__ mu, int n, float* __restrict__ Lbuf,
              float* __restrict__ dp) {
  const int bid = blockIdx.x, slice = blockIdx.y, tid = threadIdx.x;
  const float* A = JtJ + (size_t)bid * n * n;
  float* L = Lbuf + (size_t)bid * 2 * n * n;
  const float m = mu[bid];
  const int nq = n * n / 4;
  for (int idx = slice * NTH + tid; idx < nq; idx += NTH * INIT_SLICES) {
    const size_t p = 4ull * idx;
    float4 v = *(const float4*)(A + p);
    const int r = (int)(p / n), c = (int)(p % n);
    const int d = r - c;
    if (d >= 0 && d < 4) {
      if (d == 0) v.x += m;
      else if (d == 1) v.y += m;
      else if (d == 2) v.z += m;
      else v.w += m;
    }
    *(float4*)(L + p) = v;
  }
  if (slice == 0) {
    for (int i = tid; i < n; i += NTH) dp[(size_t)bid * n + i] = Jtr[(size_t)bid * n + i];
  }
}

extern "C" __global__ void __launch_bounds__(NTH)
k_cholmw_panel(int n, int k, float* __restrict__ Lbuf,
               float* __restrict__ dp, int* __restrict__ info) {
  const int bid = blockIdx.x, tid = threadIdx.x, lane = tid & 63;
  float* L = Lbuf + (size_t)bid * 2 * n * n;
  float* LT = L + (size_t)n * n;
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* pan = smem;
  __shared__ float yv[NB];
  __shared__ float rdg[NB];   // reciprocal diagonal (div -> mul downstream)
  __shared__ int bad;
  if (tid == 0) bad = 0;
  const int rows = n - k;
  // stage panel rows k..n, cols k..k+NB from the (already-updated) L copy
  for (int idx = tid; idx < rows * (NB / 4); idx += NTH) {
    const int r = idx >> 3, c4 = (idx & 7) << 2;
    *(float4*)(pan + r * PST + c4) =
        *(const float4*)(L + (size_t)(k + r) * n + k + c4);
  }
  __syncthreads();
  // wave-synchronous 32x32 factor (same scheme as k_chol_solve)
  if (tid < 64 && lane < 32) {
    const int r = lane;
    float row[NB];
#pragma unroll
    for (int c = 0; c < NB; ++c) row[c] = pan[r * PST + c];
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      float pivraw = __shfl(row[c], c, 64);
      if (pivraw <= 1e-30f) {
        if (lane == 0) bad = 1;
        pivraw = 1e-30f;
      }
      const float pv = sqrtf(pivraw);
      const float rpv = 1.0f / pv;
      if (r >= c) row[c] *= rpv;
      yv[r] = row[c];
      if (lane == c) rdg[c] = rpv;
      const float lrc = row[c];
#pragma unroll
      for (int cc = c + 1; cc < NB; ++cc) {
        if (r >= cc) row[cc] -= lrc * yv[cc];
      }
    }
#pragma unroll
    for (int c = 0; c < NB; ++c) pan[r * PST + c] = row[c];
    // publish 1/pv on the panel diagonal: the LT mirror's diagonal is read
    // ONLY by the backward substitution's divide (syrk/fwd never touch it),
    // so storing the reciprocal there turns that divide into a multiply
    pan[r * PST + r] = rdg[r];
  }
  __syncthreads();
  // row-solve sub-panel rows NB..rows (multiply by reciprocal diag: the
  // 32 serial divides per row-thread were the panel's longest chain)
  for (int r = NB + tid; r < rows; r += NTH) {
    float rw[NB];
#pragma unroll
    for (int c = 0; c < NB; ++c) rw[c] = pan[r * PST + c];
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      float s = rw[c];
#pragma unroll
      for (int c2 = 0; c2 < c; ++c2) s -= rw[c2] * pan[c * PST + c2];
      rw[c] = s * rdg[c];
    }
#pragma unroll
    for (int c = 0; c < NB; ++c) pan[r * PST + c] = rw[c];
  }
  __syncthreads();
  // write back the LT mirror ONLY: every consumer of the factored panel
  // (syrk fragments, fused forward subst via LDS, backward subst) reads
  // L^T; the L working copy keeps unfactored trailing values, which is
  // all later panel stages ever read.
#pragma unroll
  for (int c = 0; c < NB; ++c) {
    for (int r = tid; r < rows; r += NTH) {
      LT[(size_t)(k + c) * n + k + r] = pan[r * PST + c];
    }
  }
  __syncthreads();
  // fused forward-substitution step: y_k = L_kk^-1 xo[k..k+NB] on wave 0,
  // then xo[k+NB..n] -= panel . y_k straight from LDS (the separate subst
  // kernel then only runs the backward pass)
  float* xo = dp + (size_t)bid * n;
  if (tid < 64) {
    const int r = lane & 31;
    float rv[NB];
    float bv = (lane < 32) ? xo[k + r] : 0.f;
    if (lane < 32) {
#pragma unroll
      for (int c = 0; c < NB; ++c)
        rv[c] = (c <= r) ? pan[r * PST + c] : 0.f;
    }
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      float yc;
      if (lane == c) bv *= rdg[c];
      yc = __shfl(bv, c, 64);
      if (lane < 32 && r > c) bv -= rv[c] * yc;
      if (lane == c) yv[c] = bv;
    }
    if (lane < 32) xo[k + r] = bv;
  }
  __syncthreads();
  for (int r = NB + tid; r < rows; r += NTH) {
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < NB; ++c) s += pan[r * PST + c] * yv[c];
    xo[k + r] -= s;
  }
  if (bad && tid == 0) atomicOr(&info[bid], 1);
}

extern "C" __global__ void __launch_bounds__(NTH)
k_cholmw_panel_big(int n, int k, float* __restrict__ Lbuf,
               float* __restrict__ dp, int* __restrict__ info) {
  const int bid = blockIdx.x, tid = threadIdx.x, lane = tid & 63;
  float* L = Lbuf + (size_t)bid * 2 * n * n;
  float* LT = L + (size_t)n * n;
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* pan = smem;
  // factored 32x32 diag block kept resident across row chunks so panels
  // taller than the LDS budget (n > PANEL_CAP, up to MAXN_CHOL_MW) are
  // processed in passes: the row-solve/LT-writeback/fwd-subst of a row
  // depend only on that block + rdg/yv, never on other rows.
  __shared__ float diag[NB * PST];
  __shared__ float yv[NB];
  __shared__ float rdg[NB];   // reciprocal diagonal (div -> mul downstream)
  __shared__ int bad;
  if (tid == 0) bad = 0;
  const int rows = n - k;
  const int cap = rows < PANEL_CAP ? rows : PANEL_CAP;
  // ---- pass 0: rows [0, cap) ----
  // stage panel rows k..k+cap, cols k..k+NB from the (already-updated) L
  for (int idx = tid; idx < cap * (NB / 4); idx += NTH) {
    const int r = idx >> 3, c4 = (idx & 7) << 2;
    *(float4*)(pan + r * PST + c4) =
        *(const float4*)(L + (size_t)(k + r) * n + k + c4);
  }
  __syncthreads();
  // wave-synchronous 32x32 factor (same scheme as k_chol_solve)
  if (tid < 64 && lane < 32) {
    const int r = lane;
    float row[NB];
#pragma unroll
    for (int c = 0; c < NB; ++c) row[c] = pan[r * PST + c];
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      float pivraw = __shfl(row[c], c, 64);
      if (pivraw <= 1e-30f) {
        if (lane == 0) bad = 1;
        pivraw = 1e-30f;
      }
      const float pv = sqrtf(pivraw);
      const float rpv = 1.0f / pv;
      if (r >= c) row[c] *= rpv;
      yv[r] = row[c];
      if (lane == c) rdg[c] = rpv;
      const float lrc = row[c];
#pragma unroll
      for (int cc = c + 1; cc < NB; ++cc) {
        if (r >= cc) row[cc] -= lrc * yv[cc];
      }
    }
#pragma unroll
    for (int c = 0; c < NB; ++c) pan[r * PST + c] = row[c];
    // publish 1/pv on the panel diagonal: the LT mirror's diagonal is read
    // ONLY by the backward substitution's divide (syrk/fwd never touch it),
    // so storing the reciprocal there turns that divide into a multiply
    pan[r * PST + r] = rdg[r];
  }
  __syncthreads();
  // keep the factored diag block for later row chunks
  for (int idx = tid; idx < NB * NB; idx += NTH) {
    const int r = idx >> 5, c = idx & 31;
    diag[r * PST + c] = pan[r * PST + c];
  }
  __syncthreads();
  // row-solve sub-panel rows NB..cap (multiply by reciprocal diag: the
  // 32 serial divides per row-thread were the panel's longest chain)
  for (int r = NB + tid; r < cap; r += NTH) {
    float rw[NB];
#pragma unroll
    for (int c = 0; c < NB; ++c) rw[c] = pan[r * PST + c];
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      float s = rw[c];
#pragma unroll
      for (int c2 = 0; c2 < c; ++c2) s -= rw[c2] * diag[c * PST + c2];
      rw[c] = s * rdg[c];
    }
#pragma unroll
    for (int c = 0; c < NB; ++c) pan[r * PST + c] = rw[c];
  }
  __syncthreads();
  // write back the LT mirror ONLY: every consumer of the factored panel
  // (syrk fragments, fused forward subst via LDS, backward subst) reads
  // L^T; the L working copy keeps unfactored trailing values, which is
  // all later panel stages ever read.
#pragma unroll
  for (int c = 0; c < NB; ++c) {
    for (int r = tid; r < cap; r += NTH) {
      LT[(size_t)(k + c) * n + k + r] = pan[r * PST + c];
    }
  }
  __syncthreads();
  // fused forward-substitution step: y_k = L_kk^-1 xo[k..k+NB] on wave 0,
  // then xo[k+NB..n] -= panel . y_k straight from LDS (the separate subst
  // kernel then only runs the backward pass)
  float* xo = dp + (size_t)bid * n;
  if (tid < 64) {
    const int r = lane & 31;
    float rv[NB];
    float bv = (lane < 32) ? xo[k + r] : 0.f;
    if (lane < 32) {
#pragma unroll
      for (int c = 0; c < NB; ++c)
        rv[c] = (c <= r) ? diag[r * PST + c] : 0.f;
    }
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      float yc;
      if (lane == c) bv *= rdg[c];
      yc = __shfl(bv, c, 64);
      if (lane < 32 && r > c) bv -= rv[c] * yc;
      if (lane == c) yv[c] = bv;
    }
    if (lane < 32) xo[k + r] = bv;
  }
  __syncthreads();
  for (int r = NB + tid; r < cap; r += NTH) {
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < NB; ++c) s += pan[r * PST + c] * yv[c];
    xo[k + r] -= s;
  }
  // ---- passes 1..: remaining row chunks (uniform trip count) ----
  for (int r0 = cap; r0 < rows; r0 += PANEL_CAP) {
    const int nr = (rows - r0) < PANEL_CAP ? (rows - r0) : PANEL_CAP;
    __syncthreads();
    for (int idx = tid; idx < nr * (NB / 4); idx += NTH) {
      const int r = idx >> 3, c4 = (idx & 7) << 2;
      *(float4*)(pan + r * PST + c4) =
          *(const float4*)(L + (size_t)(k + r0 + r) * n + k + c4);
    }
    __syncthreads();
    for (int r = tid; r < nr; r += NTH) {
      float rw[NB];
#pragma unroll
      for (int c = 0; c < NB; ++c) rw[c] = pan[r * PST + c];
#pragma unroll
      for (int c = 0; c < NB; ++c) {
        float s = rw[c];
#pragma unroll
        for (int c2 = 0; c2 < c; ++c2) s -= rw[c2] * diag[c * PST + c2];
        rw[c] = s * rdg[c];
      }
      float sxo = 0.f;
#pragma unroll
      for (int c = 0; c < NB; ++c) {
        pan[r * PST + c] = rw[c];
        sxo += rw[c] * yv[c];
      }
      xo[k + r0 + r] -= sxo;
    }
    __syncthreads();
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      for (int r = tid; r < nr; r += NTH) {
        LT[(size_t)(k + c) * n + k + r0 + r] = pan[r * PST + c];
      }
    }
  }
  if (bad && tid == 0) atomicOr(&info[bid], 1);
}

// trailing SYRK: tile t of the lower-triangular 32x32 tile grid of the
// trailing matrix (rows/cols k+NB..n). One 64-lane wave per tile: 2x2
// fragments of mfma_f32_16x16x4, K = NB, operands read COALESCED from
// the LT mirror (same fragment scheme as the left-looking update above).
extern "C" __global__ void __launch_bounds__(64)
k_cholmw_syrk(int n, int k, int ntiles, float* __restrict__ Lbuf) {
  const int g = blockIdx.x;
  const int bid = g / ntiles, t = g % ntiles;
  int I = (int)((sqrtf(8.0f * t + 1.0f) - 1.0f) * 0.5f + 1e-4f);
  while ((I + 1) * (I + 2) / 2 <= t) ++I;
  while (I * (I + 1) / 2 > t) --I;
  const int J = t - I * (I + 1) / 2;
  float* L = Lbuf + (size_t)bid * 2 * n * n;
  const float* LT = L + (size_t)n * n;
  const int r0 = k + NB + I * NB;
  const int c0 = k + NB + J * NB;
  const int lane = threadIdx.x & 63;
  const int l15 = lane & 15, l4 = lane >> 4;
  const float* LTb = LT + (size_t)(k + l4) * n;
  using f32x4 = __attribute__((ext_vector_type(4))) float;
  f32x4 a00 = {0.f, 0.f, 0.f, 0.f}, a01 = a00, a10 = a00, a11 = a00;
#pragma unroll
  for (int kk = 0; kk < NB; kk += 4) {
    const float x0 = LTb[(size_t)kk * n + r0 + l15];
    const float x1 = LTb[(size_t)kk * n + r0 + 16 + l15];
    const float y0 = LTb[(size_t)kk * n + c0 + l15];
    const float y1 = LTb[(size_t)kk * n + c0 + 16 + l15];
    a00 = __builtin_amdgcn_mfma_f32_16x16x4f32(x0, y0, a00, 0, 0, 0);
    a01 = __builtin_amdgcn_mfma_f32_16x16x4f32(x0, y1, a01, 0, 0, 0);
    a10 = __builtin_amdgcn_mfma_f32_16x16x4f32(x1, y0, a10, 0, 0, 0);
    a11 = __builtin_amdgcn_mfma_f32_16x16x4f32(x1, y1, a11, 0, 0, 0);
  }
  // D layout: col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int ra = r0 + l4 * 4 + reg, rb = ra + 16;
    const int ca = c0 + l15, cb = ca + 16;
    L[(size_t)ra * n + ca] -= a00[reg];
    L[(size_t)ra * n + cb] -= a01[reg];
    L[(size_t)rb * n + ca] -= a10[reg];
    L[(size_t)rb * n + cb] -= a11[reg];
  }
}

// backward substitution only: dp already holds y = L^-1 b (forward pass
// fused into the panel kernels)
extern "C" __global__ void __launch_bounds__(NTH)
k_cholmw_subst(int n, const float* __restrict__ Lbuf,
               float* __restrict__ dp, const int* __restrict__ info) {
  const int bid = blockIdx.x, tid = threadIdx.x, lane = tid & 63;
  const float* L = Lbuf + (size_t)bid * 2 * n * n;
  const float* LT = L + (size_t)n * n;
  float* xo = dp + (size_t)bid * n;
  __shared__ float pan[NB * PST];
  __shared__ float yv[NB];
  // backward: L^T x = y (LT rows coalesced)
  for (int k = ((n - 1) / NB) * NB; k >= 0; k -= NB) {
    for (int idx = tid; idx < NB * NB; idx += NTH) {
      const int r = idx >> 5, c = idx & 31;
      if (c >= r) pan[r * PST + c] = LT[(size_t)(k + r) * n + k + c];
    }
    __syncthreads();
    if (tid < 64) {
      const int r = lane & 31;
      float rv[NB];
      float bv = (lane < 32) ? xo[k + r] : 0.f;
      if (lane < 32) {
#pragma unroll
        for (int c = 0; c < NB; ++c)
          rv[c] = (c >= r) ? pan[r * PST + c] : 0.f;
      }
#pragma unroll
      for (int ci = 0; ci < NB; ++ci) {
        const int c = NB - 1 - ci;
        float xc;
        if (lane == c) bv *= rv[c];   // diag holds 1/pv (panel kernel)
        xc = __shfl(bv, c, 64);
        if (lane < 32 && r < c) bv -= rv[c] * xc;
        if (lane == c) yv[c] = bv;
      }
      if (lane < 32) xo[k + r] = bv;
    }
    __syncthreads();
    for (int i = tid; i < k; i += NTH) {
      float s = 0.f;
      const float* Lr = LT + (size_t)i * n + k;
#pragma unroll
      for (int c = 0; c < NB; ++c) s += Lr[c] * yv[c];
      xo[i] -= s;
    }
    __syncthreads();
  }
  if (info[bid]) {
    const float qn = __int_as_float(0x7fc00000);
    for (int idx = tid; idx < n; idx += NTH) xo[idx] = qn;
  }
}
