// Common device utilities for the NoisyNet-MI355X kernels (gfx950 / CDNA4).
//
// Wave width on CDNA4 is 64; all kernels use 256-thread blocks (4 waves).
// RNG is counter-based Philox4x32-10 keyed by (seed, linear element index),
// so results are deterministic per seed and independent of launch geometry.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <cmath>
#include <cstdint>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// ---------------------------------------------------------------------------
// dtype conversion helpers
// ---------------------------------------------------------------------------
template <typename T> DEV_INLINE float to_f32(T v);
template <> DEV_INLINE float to_f32<float>(float v) { return v; }
template <> DEV_INLINE float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <> DEV_INLINE float to_f32<_Float16>(_Float16 v) { return (float)v; }

template <typename T> DEV_INLINE T from_f32(float v);
template <> DEV_INLINE float from_f32<float>(float v) { return v; }
template <> DEV_INLINE __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}
template <> DEV_INLINE _Float16 from_f32<_Float16>(float v) {
  return (_Float16)v;
}

// ---------------------------------------------------------------------------
// Philox4x32-10 counter-based RNG
// ---------------------------------------------------------------------------
struct Philox4 {
  uint32_t x, y, z, w;
};

DEV_INLINE uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* hip) {
  uint64_t p = (uint64_t)a * (uint64_t)b;
  *hip = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

DEV_INLINE Philox4 philox4x32(uint64_t seed, uint64_t counter) {
  uint32_t c0 = (uint32_t)counter, c1 = (uint32_t)(counter >> 32);
  uint32_t c2 = 0u, c3 = 0u;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
#pragma unroll
  for (int i = 0; i < 10; ++i) {
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

// uniform in [0, 1)
DEV_INLINE float u01(uint32_t v) { return (float)v * (1.0f / 4294967296.0f); }
// uniform in (0, 1]  (for log() in Box-Muller)
DEV_INLINE float u01_open(uint32_t v) {
  return ((float)v + 1.0f) * (1.0f / 4294967296.0f);
}

// two N(0,1) samples from one Philox draw (Box-Muller)
DEV_INLINE void gauss2(uint64_t seed, uint64_t ctr, float* g0, float* g1) {
  Philox4 p = philox4x32(seed, ctr);
  float r = sqrtf(-2.0f * logf(u01_open(p.x)));
  float theta = 6.2831853071795864f * u01(p.y);
  float s, c;
  __sincosf(theta, &s, &c);
  *g0 = r * c;
  *g1 = r * s;
}

DEV_INLINE float gauss1(uint64_t seed, uint64_t ctr) {
  Philox4 p = philox4x32(seed, ctr);
  float r = sqrtf(-2.0f * logf(u01_open(p.x)));
  float theta = 6.2831853071795864f * u01(p.y);
  return r * __cosf(theta);
}

// four N(0,1) samples from ONE Philox draw (two Box-Muller pairs) with the
// hardware log (v_log_f32). The per-element gauss1 epilogue was the top
// cost of every noisy kernel (VALU:MFMA = 99:1 on the conv1 GEMM — one
// 10-round Philox plus a ~40-op software logf per OUTPUT ELEMENT,
// discarding 3/4 of the draw); amortizing over the 4 consecutive
// accumulator rows cuts the RNG VALU work ~8x. Noise quality is
// unaffected: counters stay unique per element-quad and __logf's ~1 ulp
// on (0,1] is far below the sampled distribution's own width.
DEV_INLINE void gauss4(uint64_t seed, uint64_t ctr, float g[4]) {
  Philox4 p = philox4x32(seed, ctr);
  float s, c;
  float r0 = sqrtf(-2.0f * __logf(u01_open(p.x)));
  __sincosf(6.2831853071795864f * u01(p.y), &s, &c);
  g[0] = r0 * c;
  g[1] = r0 * s;
  float r1 = sqrtf(-2.0f * __logf(u01_open(p.z)));
  __sincosf(6.2831853071795864f * u01(p.w), &s, &c);
  g[2] = r1 * c;
  g[3] = r1 * s;
}

// uniform in [-a, a]
DEV_INLINE float uniform_pm(uint64_t seed, uint64_t ctr, float a) {
  Philox4 p = philox4x32(seed, ctr);
  return (2.0f * u01(p.x) - 1.0f) * a;
}

// ---------------------------------------------------------------------------
// hipGraph-replay seed indirection
//
// Under graph capture every kernel argument is frozen into the graph, so a
// by-value seed would replay the SAME noise each step. Python installs a
// 1-element int64 device buffer (the step counter, incremented by a node
// inside the captured region); RNG kernels receive its pointer and mix the
// live counter with the per-launch salt captured at record time. base ==
// nullptr (eager mode) keeps the original by-value behaviour bit-for-bit.
// ---------------------------------------------------------------------------
DEV_INLINE uint64_t graph_seed(const int64_t* base, uint64_t salt) {
  if (base == nullptr) return salt;
  uint64_t s = (uint64_t)(*base) * 0x9E3779B97F4A7C15ull + salt;
  s ^= s >> 31;
  s *= 0xBF58476D1CE4E5B9ull;
  s ^= s >> 29;
  return s;
}

// device pointer to the live step counter (nullptr = eager mode);
// defined in elementwise.hip, set via set_seed_buffer()/clear_seed_buffer()
extern int64_t* g_seed_base;

// ---------------------------------------------------------------------------
// wave/block reductions
// ---------------------------------------------------------------------------
DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return v;
}

DEV_INLINE float wave_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  return v;
}

// block reduce using LDS; blockDim.x <= 1024, result valid on thread 0
DEV_INLINE float block_sum(float v, float* lds /* >= blockDim/WAVE floats */) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  v = wave_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  int nw = blockDim.x / WAVE;
  v = (threadIdx.x < nw) ? lds[threadIdx.x] : 0.0f;
  if (wid == 0) v = wave_sum(v);
  return v;
}

#define HIP_CHECK_LAST()                                                   \
  do {                                                                     \
    hipError_t e = hipGetLastError();                                      \
    TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",             \
                hipGetErrorString(e));                                     \
  } while (0)

// dispatch over {float, bf16, fp16} only (no double on the GPU path)
#define NN_DISPATCH(TYPE, NAME, ...)                       \
  AT_DISPATCH_SWITCH(TYPE, NAME,                           \
    AT_DISPATCH_CASE(at::kFloat, __VA_ARGS__)              \
    AT_DISPATCH_CASE(at::kBFloat16, __VA_ARGS__)           \
    AT_DISPATCH_CASE(at::kHalf, __VA_ARGS__))
