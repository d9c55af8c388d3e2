// Fused optimizer update kernels: SGD (momentum/nesterov/L2) and AdamW,
// each with the post-step weight clamp folded in (one pass over the param
// instead of the reference's eager chain + separate clamp_,
// noisynet.py:1520-1542). Parameters may be bf16/fp16/fp32; optimizer
// state (momentum / exp_avg / exp_avg_sq) is kept in the param's dtype to
// mirror torch.optim semantics under model.half()/bfloat16().

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;

inline int grid_1d(int64_t n) {
  int64_t blocks = (n + kBlock - 1) / kBlock;
  return (int)std::min<int64_t>(blocks, 256 * 16);
}

template <typename T, bool MOM, bool NESTEROV, bool CLAMP>
__global__ void sgd_kernel(T* __restrict__ p, const T* __restrict__ g,
                           T* __restrict__ buf, int64_t n, float lr,
                           float momentum, float wd, float cmin, float cmax) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float pv = to_f32(p[i]);
    float gv = to_f32(g[i]);
    if (wd != 0.0f) gv += wd * pv;
    if (MOM) {
      float b = to_f32(buf[i]) * momentum + gv;
      buf[i] = from_f32<T>(b);
      gv = NESTEROV ? (gv + momentum * b) : b;
    }
    pv -= lr * gv;
    if (CLAMP) pv = fminf(fmaxf(pv, cmin), cmax);
    p[i] = from_f32<T>(pv);
  }
}

template <typename T, bool CLAMP>
__global__ void adamw_kernel(T* __restrict__ p, const T* __restrict__ g,
                             T* __restrict__ m, T* __restrict__ v, int64_t n,
                             float lr, float beta1, float beta2, float eps,
                             float wd, float inv_bc1, float inv_bc2, float cmin,
                             float cmax) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float pv = to_f32(p[i]) * (1.0f - lr * wd);
    float gv = to_f32(g[i]);
    float mv = beta1 * to_f32(m[i]) + (1.0f - beta1) * gv;
    float vv = beta2 * to_f32(v[i]) + (1.0f - beta2) * gv * gv;
    m[i] = from_f32<T>(mv);
    v[i] = from_f32<T>(vv);
    float denom = sqrtf(vv * inv_bc2) + eps;
    pv -= lr * inv_bc1 * mv / denom;
    if (CLAMP) pv = fminf(fmaxf(pv, cmin), cmax);
    p[i] = from_f32<T>(pv);
  }
}

template <typename scalar_t> struct DevT { using type = scalar_t; };
template <> struct DevT<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct DevT<at::Half> { using type = _Float16; };

}  // namespace

void sgd_step(torch::Tensor p, torch::Tensor g, torch::Tensor buf, double lr,
              double momentum, double wd, bool nesterov, double cmin,
              double cmax) {
  int64_t n = p.numel();
  bool mom = momentum != 0.0;
  bool clamp = cmax > cmin;
  TORCH_CHECK(!mom || buf.numel() == n, "sgd_step: momentum buffer size");
  // align the grad to the PARAM's raw layout: a plain .contiguous() turned
  // channels_last grads into NCHW order, silently pairing every 4-D conv
  // param element with the WRONG grad element (models trained through the
  // channels_last path learned drastically worse; the buffers from
  // zeros_like(p) already share p's layout)
  auto gc = g.contiguous(p.suggest_memory_format());
  NN_DISPATCH(p.scalar_type(),
                                  "sgd_step", [&] {
    using T = typename DevT<scalar_t>::type;
    auto stream = c10::hip::getCurrentHIPStream();
    auto launch = [&](auto mom_t, auto nest_t, auto clamp_t) {
      hipLaunchKernelGGL((sgd_kernel<T, decltype(mom_t)::value,
                          decltype(nest_t)::value, decltype(clamp_t)::value>),
                         dim3(grid_1d(n)), dim3(kBlock), 0, stream,
                         (T*)p.data_ptr(), (const T*)gc.data_ptr(),
                         (T*)(mom ? buf.data_ptr() : p.data_ptr()), n,
                         (float)lr, (float)momentum, (float)wd, (float)cmin,
                         (float)cmax);
    };
    using TT = std::true_type; using FF = std::false_type;
    if (mom && nesterov && clamp) launch(TT{}, TT{}, TT{});
    else if (mom && nesterov) launch(TT{}, TT{}, FF{});
    else if (mom && clamp) launch(TT{}, FF{}, TT{});
    else if (mom) launch(TT{}, FF{}, FF{});
    else if (clamp) launch(FF{}, FF{}, TT{});
    else launch(FF{}, FF{}, FF{});
  });
  HIP_CHECK_LAST();
}

void adamw_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, int64_t step, double lr, double beta1,
                double beta2, double eps, double wd, double cmin, double cmax) {
  int64_t n = p.numel();
  bool clamp = cmax > cmin;
  float bc1 = 1.0f - ::powf((float)beta1, (float)step);
  float bc2 = 1.0f - ::powf((float)beta2, (float)step);
  auto gc = g.contiguous(p.suggest_memory_format());  // see sgd_step
  NN_DISPATCH(p.scalar_type(),
                                  "adamw_step", [&] {
    using T = typename DevT<scalar_t>::type;
    auto stream = c10::hip::getCurrentHIPStream();
    if (clamp) {
      hipLaunchKernelGGL((adamw_kernel<T, true>), dim3(grid_1d(n)),
                         dim3(kBlock), 0, stream, (T*)p.data_ptr(),
                         (const T*)gc.data_ptr(), (T*)m.data_ptr(),
                         (T*)v.data_ptr(), n, (float)lr, (float)beta1,
                         (float)beta2, (float)eps, (float)wd, 1.0f / bc1,
                         1.0f / bc2, (float)cmin, (float)cmax);
    } else {
      hipLaunchKernelGGL((adamw_kernel<T, false>), dim3(grid_1d(n)),
                         dim3(kBlock), 0, stream, (T*)p.data_ptr(),
                         (const T*)gc.data_ptr(), (T*)m.data_ptr(),
                         (T*)v.data_ptr(), n, (float)lr, (float)beta1,
                         (float)beta2, (float)eps, (float)wd, 1.0f / bc1,
                         1.0f / bc2, 0.0f, 0.0f);
    }
  });
  HIP_CHECK_LAST();
}
