// Shared helpers for the sagecal_amd CDNA4 (gfx950) kernels.
// Complex arithmetic on float2, 2x2 complex-matrix micro-ops, and the
// lane/accumulation conventions used by predict.hip and jtj.hip.
#pragma once
#include <hip/hip_runtime.h>

#define WAVE 64

using cf = float2;

__device__ __forceinline__ cf cadd(cf a, cf b) { return {a.x + b.x, a.y + b.y}; }
__device__ __forceinline__ cf csub(cf a, cf b) { return {a.x - b.x, a.y - b.y}; }
__device__ __forceinline__ cf cmul(cf a, cf b) {
  return {a.x * b.x - a.y * b.y, a.x * b.y + a.y * b.x};
}
// a * conj(b)
__device__ __forceinline__ cf cmulc(cf a, cf b) {
  return {a.x * b.x + a.y * b.y, a.y * b.x - a.x * b.y};
}
__device__ __forceinline__ cf conjf2(cf a) { return {a.x, -a.y}; }
__device__ __forceinline__ cf cscale(cf a, float s) { return {a.x * s, a.y * s}; }
__device__ __forceinline__ float cabs2(cf a) { return a.x * a.x + a.y * a.y; }

// 2x2 complex matrix as cf[4], row-major [m00, m01, m10, m11].
// C = A * B
__device__ __forceinline__ void m2mul(const cf* A, const cf* B, cf* C) {
  C[0] = cadd(cmul(A[0], B[0]), cmul(A[1], B[2]));
  C[1] = cadd(cmul(A[0], B[1]), cmul(A[1], B[3]));
  C[2] = cadd(cmul(A[2], B[0]), cmul(A[3], B[2]));
  C[3] = cadd(cmul(A[2], B[1]), cmul(A[3], B[3]));
}
// C = A * B^H
__device__ __forceinline__ void m2mulh(const cf* A, const cf* B, cf* C) {
  C[0] = cadd(cmulc(A[0], B[0]), cmulc(A[1], B[1]));
  C[1] = cadd(cmulc(A[0], B[2]), cmulc(A[1], B[3]));
  C[2] = cadd(cmulc(A[2], B[0]), cmulc(A[3], B[1]));
  C[3] = cadd(cmulc(A[2], B[2]), cmulc(A[3], B[3]));
}
// C = A^H * B
__device__ __forceinline__ void m2hmul(const cf* A, const cf* B, cf* C) {
  C[0] = cadd(cmulc(B[0], A[0]), cmulc(B[2], A[2]));
  C[1] = cadd(cmulc(B[1], A[0]), cmulc(B[3], A[2]));
  C[2] = cadd(cmulc(B[0], A[1]), cmulc(B[2], A[3]));
  C[3] = cadd(cmulc(B[1], A[1]), cmulc(B[3], A[3]));
}

__device__ __forceinline__ void atomicAddCf(cf* dst, cf v) {
  atomicAdd(&dst->x, v.x);
  atomicAdd(&dst->y, v.y);
}
