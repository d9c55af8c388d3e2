// EfficientNet-family activations with analytic backward (the reference's
// memory-efficient JIT Swish/Mish, timm/models/activations.py:10-66,
// preserved as the recompute-sigmoid trick: backward re-derives sigmoid(x)
// from the saved INPUT instead of storing both x and sigmoid(x)).

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;

inline int grid_1d(int64_t n) {
  return (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 8192);
}

enum Act { SWISH = 0, MISH = 1, HARDSWISH = 2, HARDSIGMOID = 3, SIGMOID = 4 };

DEV_INLINE float act_fwd_f(int act, float x) {
  switch (act) {
    case SWISH: return x / (1.0f + __expf(-x));
    case MISH: {
      float sp = logf(1.0f + __expf(x));
      return x * tanhf(sp);
    }
    case HARDSWISH: return x * fminf(fmaxf(x + 3.0f, 0.0f), 6.0f) * (1.0f / 6.0f);
    case HARDSIGMOID: return fminf(fmaxf(x + 3.0f, 0.0f), 6.0f) * (1.0f / 6.0f);
    default: return 1.0f / (1.0f + __expf(-x));
  }
}

DEV_INLINE float act_bwd_f(int act, float x) {
  switch (act) {
    case SWISH: {
      float s = 1.0f / (1.0f + __expf(-x));
      return s * (1.0f + x * (1.0f - s));
    }
    case MISH: {
      float sp = logf(1.0f + __expf(x));
      float tsp = tanhf(sp);
      float s = 1.0f / (1.0f + __expf(-x));
      return tsp + x * s * (1.0f - tsp * tsp);
    }
    case HARDSWISH:
      if (x <= -3.0f) return 0.0f;
      if (x >= 3.0f) return 1.0f;
      return (2.0f * x + 3.0f) * (1.0f / 6.0f);
    case HARDSIGMOID:
      return (x > -3.0f && x < 3.0f) ? (1.0f / 6.0f) : 0.0f;
    default: {
      float s = 1.0f / (1.0f + __expf(-x));
      return s * (1.0f - s);
    }
  }
}

template <typename T>
__global__ void act_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                               int64_t n, int act) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    y[i] = from_f32<T>(act_fwd_f(act, to_f32(x[i])));
}

template <typename T>
__global__ void act_bwd_kernel(const T* __restrict__ g, const T* __restrict__ x,
                               T* __restrict__ gx, int64_t n, int act) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    gx[i] = from_f32<T>(to_f32(g[i]) * act_bwd_f(act, to_f32(x[i])));
}

template <typename scalar_t> struct DevT { using type = scalar_t; };
template <> struct DevT<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct DevT<at::Half> { using type = _Float16; };

}  // namespace

torch::Tensor act_fwd(torch::Tensor x, int64_t act) {
  auto xc = x.contiguous(x.suggest_memory_format());
  auto y = torch::empty_like(xc);
  int64_t n = xc.numel();
  NN_DISPATCH(xc.scalar_type(), "act_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((act_fwd_kernel<T>), dim3(grid_1d(n)), dim3(kBlock), 0,
                       c10::hip::getCurrentHIPStream(),
                       (const T*)xc.data_ptr(), (T*)y.data_ptr(), n, (int)act);
  });
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor act_bwd(torch::Tensor g, torch::Tensor x, int64_t act) {
  auto xc = x.contiguous(x.suggest_memory_format());
  auto gc = g.contiguous(x.suggest_memory_format());
  auto gx = torch::empty_like(gc);
  int64_t n = gc.numel();
  NN_DISPATCH(gc.scalar_type(), "act_bwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((act_bwd_kernel<T>), dim3(grid_1d(n)), dim3(kBlock), 0,
                       c10::hip::getCurrentHIPStream(),
                       (const T*)gc.data_ptr(), (const T*)xc.data_ptr(),
                       (T*)gx.data_ptr(), n, (int)act);
  });
  HIP_CHECK_LAST();
  return gx;
}
