// Elementwise kernels: fake-quant (UniformQuantize chain), STE mask,
// multiplicative uniform weight noise, ReLU+clip, dropout.
//
// All memory-bound: grid-stride loops, 256-thread blocks, vectorized where
// the dtype permits (bf16 processed as 8-wide packs via float conversion --
// cdna_hip_programming.md G13). Semantics match
// noisynet_amd/ops/reference.py (the fp32 oracle in tests/test_ops_gpu.py).

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;

inline int grid_1d(int64_t n, int per_thread = 1) {
  int64_t blocks = (n + (int64_t)kBlock * per_thread - 1) / ((int64_t)kBlock * per_thread);
  return (int)std::min<int64_t>(blocks, 256 * 8 * 4);  // cap; grid-stride
}

// --------------------------------------------------------------------------
// fake_quant: out = round(clamp((x-min)/scale + U(-s,s), 0, qmax))*scale + min
// --------------------------------------------------------------------------
// 4 elements per thread: ONE Philox draw covers all four stochastic-
// rounding offsets (the per-element draw made quantization RNG-bound).
template <typename T, bool STOCH>
__global__ void fake_quant_kernel(const T* __restrict__ x, T* __restrict__ out,
                                  int64_t n, float min_value, float inv_scale,
                                  float scale, float qmax, float stoch,
                                  uint64_t seed,
                                  const int64_t* __restrict__ seed_base) {
  seed = graph_seed(seed_base, seed);
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       i0 < n; i0 += stride) {
    float u[4] = {0.0f, 0.0f, 0.0f, 0.0f};
    if (STOCH) {
      Philox4 ph = philox4x32(seed, (uint64_t)(i0 >> 2));
      u[0] = (2.0f * u01(ph.x) - 1.0f) * stoch;
      u[1] = (2.0f * u01(ph.y) - 1.0f) * stoch;
      u[2] = (2.0f * u01(ph.z) - 1.0f) * stoch;
      u[3] = (2.0f * u01(ph.w) - 1.0f) * stoch;
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int64_t i = i0 + j;
      if (i >= n) continue;
      float q = (to_f32(x[i]) - min_value) * inv_scale;
      if (STOCH) q += u[j];
      q = fminf(fmaxf(q, 0.0f), qmax);
      q = nearbyintf(q);
      out[i] = from_f32<T>(q * scale + min_value);
    }
  }
}

// 4-wide vector-load form (16-bit dtypes, n % 4 == 0): one 8-B load/store
// per quad instead of four 2-B ones; same Philox counter scheme, so the
// stochastic-rounding draws are bit-identical to the scalar form
template <typename T, bool STOCH>
__global__ void fake_quant_vec_kernel(const T* __restrict__ x,
                                      T* __restrict__ out, int64_t n_vec,
                                      float min_value, float inv_scale,
                                      float scale, float qmax, float stoch,
                                      uint64_t seed,
                                      const int64_t* __restrict__ seed_base) {
  struct alignas(8) V4 { T v[4]; };
  seed = graph_seed(seed_base, seed);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_vec;
       i += (int64_t)gridDim.x * blockDim.x) {
    V4 xv = ((const V4*)x)[i];
    float u[4] = {0.0f, 0.0f, 0.0f, 0.0f};
    if (STOCH) {
      Philox4 ph = philox4x32(seed, (uint64_t)i);
      u[0] = (2.0f * u01(ph.x) - 1.0f) * stoch;
      u[1] = (2.0f * u01(ph.y) - 1.0f) * stoch;
      u[2] = (2.0f * u01(ph.z) - 1.0f) * stoch;
      u[3] = (2.0f * u01(ph.w) - 1.0f) * stoch;
    }
    V4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float q = (to_f32(xv.v[j]) - min_value) * inv_scale;
      if (STOCH) q += u[j];
      q = fminf(fmaxf(q, 0.0f), qmax);
      q = nearbyintf(q);
      o.v[j] = from_f32<T>(q * scale + min_value);
    }
    ((V4*)out)[i] = o;
  }
}

// --------------------------------------------------------------------------
// STE mask: grad_in = grad_out * (min <= x <= max)
// --------------------------------------------------------------------------
template <typename T>
__global__ void ste_mask_kernel(const T* __restrict__ g, const T* __restrict__ x,
                                T* __restrict__ out, int64_t n, float min_value,
                                float max_value) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float xv = to_f32(x[i]);
    out[i] = (xv >= min_value && xv <= max_value) ? g[i] : from_f32<T>(0.0f);
  }
}

// 8-wide 16-B vector form (16-bit dtypes, n % 8 == 0): the scalar form was
// issue-bound at 2 B per lane per load
template <typename T>
struct alignas(16) Pack8 { T v[8]; };

template <typename T>
__global__ void ste_mask_vec_kernel(const T* __restrict__ g,
                                    const T* __restrict__ x,
                                    T* __restrict__ out, int64_t n_vec,
                                    float min_value, float max_value) {
  using V = Pack8<T>;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_vec;
       i += (int64_t)gridDim.x * blockDim.x) {
    V gv = ((const V*)g)[i];
    V xv = ((const V*)x)[i];
    V o;
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      float f = to_f32(xv.v[u]);
      o.v[u] = (f >= min_value && f <= max_value) ? gv.v[u]
                                                  : from_f32<T>(0.0f);
    }
    ((V*)out)[i] = o;
  }
}

// --------------------------------------------------------------------------
// mult uniform noise: out = x * (1 + U(-a, a))
// --------------------------------------------------------------------------
template <typename T>
__global__ void mult_uniform_kernel(const T* __restrict__ x, T* __restrict__ out,
                                    int64_t n, float a, uint64_t seed,
                                    const int64_t* __restrict__ seed_base) {
  seed = graph_seed(seed_base, seed);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = to_f32(x[i]);
    out[i] = from_f32<T>(v + v * uniform_pm(seed, (uint64_t)i, a));
  }
}

// --------------------------------------------------------------------------
// relu + upper clip
// --------------------------------------------------------------------------
template <typename T>
__global__ void relu_clip_kernel(const T* __restrict__ x, T* __restrict__ out,
                                 int64_t n, int do_relu, float act_max) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = to_f32(x[i]);
    if (do_relu) v = fmaxf(v, 0.0f);
    if (act_max > 0.0f) v = fminf(v, act_max);
    out[i] = from_f32<T>(v);
  }
}

// --------------------------------------------------------------------------
// dropout: mask = (u >= p)/(1-p); out = x*mask
// --------------------------------------------------------------------------
template <typename T>
__global__ void dropout_kernel(const T* __restrict__ x, T* __restrict__ out,
                               T* __restrict__ mask, int64_t n, float p,
                               float inv_keep, uint64_t seed,
                               const int64_t* __restrict__ seed_base) {
  seed = graph_seed(seed_base, seed);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    Philox4 r = philox4x32(seed, (uint64_t)i);
    float m = (u01(r.x) >= p) ? inv_keep : 0.0f;
    mask[i] = from_f32<T>(m);
    out[i] = from_f32<T>(to_f32(x[i]) * m);
  }
}

template <typename T>
__global__ void relu_clip_bwd_kernel(const T* __restrict__ g,
                                     const T* __restrict__ y,
                                     T* __restrict__ out, int64_t n,
                                     int do_relu, float act_max) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float yv = to_f32(y[i]);
    float mask = 1.0f;
    if (do_relu && yv <= 0.0f) mask = 0.0f;
    if (act_max > 0.0f && yv >= act_max) mask = 0.0f;
    out[i] = from_f32<T>(to_f32(g[i]) * mask);
  }
}

template <typename scalar_t> struct DevT { using type = scalar_t; };
template <> struct DevT<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct DevT<at::Half> { using type = _Float16; };

}  // namespace

// ===========================================================================
// host wrappers
// ===========================================================================

// hipGraph-replay seed indirection (see common.h graph_seed): Python keeps
// the 1-element int64 step-counter tensor alive for the lifetime of the
// captured graph and increments it inside the captured region.
int64_t* g_seed_base = nullptr;

void set_seed_buffer(torch::Tensor t) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kInt64 &&
                  t.numel() == 1 && t.is_contiguous(),
              "set_seed_buffer: need a 1-element contiguous int64 GPU tensor");
  g_seed_base = t.data_ptr<int64_t>();
}

void clear_seed_buffer() { g_seed_base = nullptr; }

torch::Tensor fake_quant_fwd(torch::Tensor x, int64_t num_bits, double min_value,
                             double max_value, double stochastic, int64_t seed) {
  TORCH_CHECK(x.is_cuda(), "fake_quant_fwd: expected GPU tensor");
  auto xc = x.contiguous(x.suggest_memory_format());
  auto out = torch::empty_like(xc);
  int64_t n = xc.numel();
  if (n == 0) return out;
  float qmax = ::powf(2.0f, (float)num_bits) - 1.0f;
  float scale = std::max(((float)max_value - (float)min_value) / qmax, 1e-6f);
  NN_DISPATCH(xc.scalar_type(),
                                  "fake_quant_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    auto stream = c10::hip::getCurrentHIPStream();
    if (sizeof(T) == 2 && (n & 3) == 0) {
      if (stochastic > 0)
        hipLaunchKernelGGL((fake_quant_vec_kernel<T, true>),
                           dim3(grid_1d(n / 4)), dim3(kBlock), 0, stream,
                           (const T*)xc.data_ptr(), (T*)out.data_ptr(), n / 4,
                           (float)min_value, 1.0f / scale, scale, qmax,
                           (float)stochastic, (uint64_t)seed, g_seed_base);
      else
        hipLaunchKernelGGL((fake_quant_vec_kernel<T, false>),
                           dim3(grid_1d(n / 4)), dim3(kBlock), 0, stream,
                           (const T*)xc.data_ptr(), (T*)out.data_ptr(), n / 4,
                           (float)min_value, 1.0f / scale, scale, qmax, 0.0f,
                           (uint64_t)seed, g_seed_base);
    } else if (stochastic > 0) {
      hipLaunchKernelGGL((fake_quant_kernel<T, true>), dim3(grid_1d((n + 3) / 4)),
                         dim3(kBlock), 0, stream,
                         (const T*)xc.data_ptr(), (T*)out.data_ptr(), n,
                         (float)min_value, 1.0f / scale, scale, qmax,
                         (float)stochastic, (uint64_t)seed, g_seed_base);
    } else {
      hipLaunchKernelGGL((fake_quant_kernel<T, false>), dim3(grid_1d((n + 3) / 4)),
                         dim3(kBlock), 0, stream,
                         (const T*)xc.data_ptr(), (T*)out.data_ptr(), n,
                         (float)min_value, 1.0f / scale, scale, qmax, 0.0f,
                         (uint64_t)seed, g_seed_base);
    }
  });
  HIP_CHECK_LAST();
  return out;
}

torch::Tensor ste_mask(torch::Tensor grad, torch::Tensor x, double min_value,
                       double max_value) {
  auto g = grad.contiguous(x.suggest_memory_format());
  auto xc = x.contiguous(x.suggest_memory_format());
  auto out = torch::empty_like(g);
  int64_t n = g.numel();
  NN_DISPATCH(g.scalar_type(),
                                  "ste_mask", [&] {
    using T = typename DevT<scalar_t>::type;
    if (sizeof(T) == 2 && (n & 7) == 0) {
      hipLaunchKernelGGL((ste_mask_vec_kernel<T>), dim3(grid_1d(n / 8)),
                         dim3(kBlock), 0, c10::hip::getCurrentHIPStream(),
                         (const T*)g.data_ptr(), (const T*)xc.data_ptr(),
                         (T*)out.data_ptr(), n / 8, (float)min_value,
                         (float)max_value);
    } else {
      hipLaunchKernelGGL((ste_mask_kernel<T>), dim3(grid_1d(n)), dim3(kBlock),
                         0, c10::hip::getCurrentHIPStream(),
                         (const T*)g.data_ptr(), (const T*)xc.data_ptr(),
                         (T*)out.data_ptr(), n, (float)min_value,
                         (float)max_value);
    }
  });
  HIP_CHECK_LAST();
  return out;
}

torch::Tensor mult_uniform_noise(torch::Tensor x, double a, int64_t seed) {
  auto xc = x.contiguous(x.suggest_memory_format());
  auto out = torch::empty_like(xc);
  int64_t n = xc.numel();
  NN_DISPATCH(xc.scalar_type(),
                                  "mult_uniform_noise", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((mult_uniform_kernel<T>), dim3(grid_1d(n)), dim3(kBlock),
                       0, c10::hip::getCurrentHIPStream(),
                       (const T*)xc.data_ptr(), (T*)out.data_ptr(), n, (float)a,
                       (uint64_t)seed, g_seed_base);
  });
  HIP_CHECK_LAST();
  return out;
}

torch::Tensor relu_clip_fwd(torch::Tensor x, bool relu, double act_max) {
  auto xc = x.contiguous(x.suggest_memory_format());
  auto out = torch::empty_like(xc);
  int64_t n = xc.numel();
  NN_DISPATCH(xc.scalar_type(),
                                  "relu_clip_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((relu_clip_kernel<T>), dim3(grid_1d(n)), dim3(kBlock), 0,
                       c10::hip::getCurrentHIPStream(), (const T*)xc.data_ptr(),
                       (T*)out.data_ptr(), n, relu ? 1 : 0, (float)act_max);
  });
  HIP_CHECK_LAST();
  return out;
}

std::vector<torch::Tensor> dropout_fwd(torch::Tensor x, double p, int64_t seed) {
  auto xc = x.contiguous(x.suggest_memory_format());
  auto out = torch::empty_like(xc);
  auto mask = torch::empty_like(xc);
  int64_t n = xc.numel();
  float inv_keep = 1.0f / (1.0f - (float)p);
  NN_DISPATCH(xc.scalar_type(),
                                  "dropout_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((dropout_kernel<T>), dim3(grid_1d(n)), dim3(kBlock), 0,
                       c10::hip::getCurrentHIPStream(), (const T*)xc.data_ptr(),
                       (T*)out.data_ptr(), (T*)mask.data_ptr(), n, (float)p,
                       inv_keep, (uint64_t)seed, g_seed_base);
  });
  HIP_CHECK_LAST();
  return {out, mask};
}

// grad of relu+clip re-derived from the saved OUTPUT y
torch::Tensor relu_clip_bwd(torch::Tensor g, torch::Tensor y, bool relu,
                            double act_max) {
  auto yc = y.contiguous(y.suggest_memory_format());
  auto gc = g.contiguous(y.suggest_memory_format());
  auto out = torch::empty_like(gc);
  int64_t n = gc.numel();
  NN_DISPATCH(gc.scalar_type(), "relu_clip_bwd", [&] {
    using T = typename DevT<scalar_t>::type;
    float lo = relu ? 0.0f : -INFINITY;
    float hi = act_max > 0 ? (float)act_max : INFINITY;
    // reuse ste_mask semantics: zero where y <= lo or y >= hi
    hipLaunchKernelGGL((relu_clip_bwd_kernel<T>), dim3(grid_1d(n)),
                       dim3(kBlock), 0, c10::hip::getCurrentHIPStream(),
                       (const T*)gc.data_ptr(), (const T*)yc.data_ptr(),
                       (T*)out.data_ptr(), n, relu ? 1 : 0,
                       act_max > 0 ? (float)act_max : 0.0f);
  });
  HIP_CHECK_LAST();
  return out;
}
