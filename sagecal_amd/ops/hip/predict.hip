// Coherency predict kernel for gfx950 (MI355X).
//
// MI355X-native replacement for the reference's kernel_coherencies
// (predict_model.cu:1059): one thread per baseline-row, sources staged
// through LDS in tiles, fp64 phase arithmetic with double-precision arg
// reduction followed by fast float sincos. Per-channel fluxes are
// precomputed on the host (torch), so the kernel carries no spectral-index
// math; shapelet envelopes are folded in by the host path (shapelet.py).
//
// Layout: rows r = t*Nbase + b (time-major); output coh [M, R, 4] complex64.
//
// Optional fused station-beam path (-B 1, the reference's DOBEAM_ARRAY:
// predict_model.cu:843-852 'scalef *= beam1*beam2'): when `beam` is
// non-null it holds the REAL scalar array-factor gain per
// (timeslot, source, station) [T, K, N] (beamgain = |sum phasors|/K,
// stationbeam.c:318 — the reference's array factor is real), and each
// source's contribution is scaled by beam[t,g,sta1]*beam[t,g,sta2].
#include "common.h"

#define SRC_TILE 64

struct SrcTile {
  double ll[SRC_TILE], mm[SRC_TILE], nn1[SRC_TILE];
  float sI[SRC_TILE], sQ[SRC_TILE], sU[SRC_TILE], sV[SRC_TILE];
  float eX[SRC_TILE], eY[SRC_TILE], eP[SRC_TILE];
  float cxi[SRC_TILE], sxi[SRC_TILE], cphi[SRC_TILE], sphi[SRC_TILE];
  float r1[SRC_TILE];  // time-smear source-distance term (precomputed host)
  int   stype[SRC_TILE];
};

extern "C" __global__ void __launch_bounds__(128)
k_predict_coh(const double* __restrict__ u, const double* __restrict__ v,
              const double* __restrict__ w,
              const double* __restrict__ ll, const double* __restrict__ mm,
              const double* __restrict__ nn1,
              const float* __restrict__ sI, const float* __restrict__ sQ,
              const float* __restrict__ sU, const float* __restrict__ sV,
              const float* __restrict__ eX, const float* __restrict__ eY,
              const float* __restrict__ eP, const float* __restrict__ cxi,
              const float* __restrict__ sxi, const float* __restrict__ cphi,
              const float* __restrict__ sphi, const float* __restrict__ r1t,
              const int* __restrict__ stype,
              const int* __restrict__ cluster_off, int M, int R,
              double freq, double fdelta2,      // channel halfwidth [Hz]
              double tdelta,
              const float* __restrict__ beam,   // [T, K, N] or null
              const int* __restrict__ pairs,    // [Nbase, 2] (with beam)
              int Nbase, int Ktot, int Nsta,
              float2* __restrict__ out) {
  __shared__ SrcTile tile;
  const int r = blockIdx.x * blockDim.x + threadIdx.x;
  const bool live = r < R;
  // beam indexing state (only touched when beam != null)
  int b_sta1 = 0, b_sta2 = 0;
  size_t b_toff = 0;
  if (beam != nullptr && live) {
    const int t = r / Nbase, bl = r - t * Nbase;
    b_sta1 = pairs[2 * bl];
    b_sta2 = pairs[2 * bl + 1];
    b_toff = (size_t)t * Ktot;
  }
  double ur = 0, vr = 0, wr = 0;
  float blf = 0.f;
  if (live) {
    ur = u[r]; vr = v[r]; wr = w[r];
    blf = (float)(sqrt(ur * ur + vr * vr + wr * wr) * freq);
  }
  const float uf = (float)(ur * freq), vf = (float)(vr * freq),
              wf = (float)(wr * freq);
  const float tsm_c = 7.2921150e-5f * (float)tdelta * blf;

  for (int ci = 0; ci < M; ++ci) {
    const int s0 = cluster_off[ci], s1 = cluster_off[ci + 1];
    // accumulators: IIl/QQl/UUl/VVl sums as complex
    cf aI = {0.f, 0.f}, aQ = {0.f, 0.f}, aU = {0.f, 0.f}, aV = {0.f, 0.f};
    for (int base = s0; base < s1; base += SRC_TILE) {
      const int nt = min(SRC_TILE, s1 - base);
      __syncthreads();
      for (int k = threadIdx.x; k < nt; k += blockDim.x) {
        const int g = base + k;
        tile.ll[k] = ll[g];   tile.mm[k] = mm[g];   tile.nn1[k] = nn1[g];
        tile.sI[k] = sI[g];   tile.sQ[k] = sQ[g];
        tile.sU[k] = sU[g];   tile.sV[k] = sV[g];
        tile.eX[k] = eX[g];   tile.eY[k] = eY[g];   tile.eP[k] = eP[g];
        tile.cxi[k] = cxi[g]; tile.sxi[k] = sxi[g];
        tile.cphi[k] = cphi[g]; tile.sphi[k] = sphi[g];
        tile.r1[k] = r1t[g];  tile.stype[k] = stype[g];
      }
      __syncthreads();
      if (!live) continue;
      for (int k = 0; k < nt; ++k) {
        // fp64 phase term G = 2*pi*(u l + v m + w (n-1)) (seconds * none)
        const double G = 6.283185307179586476925286766559 *
            (ur * tile.ll[k] + vr * tile.mm[k] + wr * tile.nn1[k]);
        // phase at this channel, arg-reduced in fp64 then fast float sincos
        const double ph = fma(G, freq, 0.0);
        const float phr = (float)fmod(ph, 6.283185307179586476925286766559);
        float sp, cp;
        __sincosf(phr, &sp, &cp);
        // freq smearing |sinc(G*fdelta/2)| (predict.c:178-189)
        float sm = 1.0f;
        const float smf = (float)(G * fdelta2);
        if (fabsf(smf) > 1e-9f) sm = fabsf(__sinf(smf) / smf);
        // time smearing (predict.c:94-107); r1 precomputed per source
        const float prod = tsm_c * tile.r1[k];
        if (prod > 1e-9f) sm *= 1.0645f * erff(0.8326f * prod) / prod;
        cf phc = {cp * sm, sp * sm};
        // extended-source envelope at wavelength-scaled uvw
        const int st = tile.stype[k];
        if (st != 0) {
          float up = uf, vp = vf;
          // projection rotation (predict.c:36-44); host bakes use_projection
          // into cxi/sxi/cphi/sphi (identity values when off)
          const float c_xi = tile.cxi[k], s_xi = tile.sxi[k];
          const float c_ph = tile.cphi[k], s_ph = tile.sphi[k];
          up = uf * c_xi - vf * c_ph * s_xi + wf * s_ph * s_xi;
          vp = uf * s_xi + vf * c_ph * c_xi - wf * s_ph * c_xi;
          if (st == 1) {  // gaussian
            float cpp, spp;
            __sincosf(tile.eP[k], &spp, &cpp);
            const float ut = tile.eX[k] * (cpp * up - spp * vp);
            const float vt = tile.eY[k] * (spp * up + cpp * vp);
            phc = cscale(phc, __expf(-19.739208802178716f * (ut * ut + vt * vt)));
          } else if (st == 2) {  // disk: j1
            const float b = sqrtf(up * up + vp * vp) * tile.eX[k] *
                            6.2831853071795865f;
            phc = cscale(phc, j1f(b));
          } else if (st == 3) {  // ring: j0
            const float b = sqrtf(up * up + vp * vp) * tile.eX[k] *
                            6.2831853071795865f;
            phc = cscale(phc, j0f(b));
          }
          // st==4 (shapelet): envelope folded in on the host path
        }
        if (beam != nullptr) {
          const float* bg = beam +
              (b_toff + (size_t)(base + k)) * (size_t)Nsta;
          phc = cscale(phc, bg[b_sta1] * bg[b_sta2]);
        }
        aI = cadd(aI, cscale(phc, tile.sI[k]));
        aQ = cadd(aQ, cscale(phc, tile.sQ[k]));
        aU = cadd(aU, cscale(phc, tile.sU[k]));
        aV = cadd(aV, cscale(phc, tile.sV[k]));
      }
    }
    if (live) {
      // Stokes -> coherency [[I+Q, U+jV],[U-jV, I-Q]] (predict.c:228-235)
      float2* o = out + ((size_t)ci * R + r) * 4;
      o[0] = cadd(aI, aQ);
      o[1] = {aU.x - aV.y, aU.y + aV.x};
      o[2] = {aU.x + aV.y, aU.y - aV.x};
      o[3] = csub(aI, aQ);
    }
    __syncthreads();
  }
}
