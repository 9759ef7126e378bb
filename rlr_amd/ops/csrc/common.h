// Shared helpers for rlr_amd CDNA4 (gfx950) kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdio.h>

#define HIP_CHECK(expr)                                                      \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(_e),      \
              __FILE__, __LINE__);                                           \
      abort();                                                               \
    }                                                                        \
  } while (0)

// MI355X: 256 CUs; memory-bound grid-stride kernels cap the grid and stride
// (cdna_hip_programming.md Guideline 11).
constexpr int kWave = 64;
constexpr int kBlock = 256;
constexpr int kMaxBlocks = 2048;

static inline int grid_for(long n, int block = kBlock, int cap = kMaxBlocks) {
  long b = (n + block - 1) / block;
  return (int)(b < cap ? (b > 0 ? b : 1) : cap);
}

// ------------------------------------------------------------------ philox
// Philox4x32-10 counter-based RNG (deterministic: (seed, offset, idx) ->
// 4 x uint32).  Used for dropout masks and server DP noise so results are
// world-size- and replay-invariant.
struct Philox4 {
  uint32_t x, y, z, w;
};

__device__ __forceinline__ uint32_t mulhilo(uint32_t a, uint32_t b,
                                            uint32_t* hi) {
  uint64_t p = (uint64_t)a * b;
  *hi = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

__device__ __forceinline__ Philox4 philox4(uint64_t seed, uint64_t offset,
                                           uint32_t idx) {
  uint32_t c0 = (uint32_t)offset, c1 = (uint32_t)(offset >> 32);
  uint32_t c2 = idx, c3 = 0;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    uint32_t hi0, hi1;
    uint32_t lo0 = mulhilo(0xD2511F53u, c0, &hi0);
    uint32_t lo1 = mulhilo(0xCD9E8D57u, c2, &hi1);
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  return {c0, c1, c2, c3};
}

// ------------------------------------------------- bf16 <-> f32 helpers
// bf16 is carried as raw ushort; loads/stores go through these so kernels
// can be templated on the storage type (float or ushort-bf16).
__device__ __forceinline__ float bf2f_(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}
__device__ __forceinline__ unsigned short f2bf_(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  unsigned int r = v.i + 0x7FFF + ((v.i >> 16) & 1);  // round-nearest-even
  return (unsigned short)(r >> 16);
}
template <typename T>
__device__ __forceinline__ float ldv(const T* p);
template <>
__device__ __forceinline__ float ldv<float>(const float* p) { return *p; }
template <>
__device__ __forceinline__ float ldv<unsigned short>(
    const unsigned short* p) { return bf2f_(*p); }
template <typename T>
__device__ __forceinline__ void stv(T* p, float v);
template <>
__device__ __forceinline__ void stv<float>(float* p, float v) { *p = v; }
template <>
__device__ __forceinline__ void stv<unsigned short>(unsigned short* p,
                                                    float v) {
  *p = f2bf_(v);
}

__device__ __forceinline__ float u32_to_uniform(uint32_t v) {
  // (0,1]: matches the usual counter-RNG convention
  return (v >> 8) * (1.0f / 16777216.0f) + (0.5f / 16777216.0f);
}
