// Fused softmax + cross-entropy (mean reduction): one block per row batch,
// saves the softmax for the backward (grad = (softmax - onehot)/N).
// Row widths here are small (10 for CIFAR, 1000 for ImageNet): one wave
// handles a row with a grid-stride over rows.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

template <typename T>
__global__ void softmax_xent_fwd_kernel(const T* __restrict__ logits,
                                        const int64_t* __restrict__ target,
                                        T* __restrict__ softmax,
                                        float* __restrict__ loss_partial,
                                        int64_t nrows, int ncols) {
  // one wave per row; blockDim.x == 256 -> 4 rows per block iteration
  int wid = threadIdx.x / WAVE;
  int lane = threadIdx.x & (WAVE - 1);
  float acc_loss = 0.0f;
  for (int64_t row = (int64_t)blockIdx.x * (blockDim.x / WAVE) + wid;
       row < nrows; row += (int64_t)gridDim.x * (blockDim.x / WAVE)) {
    const T* lr = logits + row * ncols;
    float mx = -INFINITY;
    for (int c = lane; c < ncols; c += WAVE) mx = fmaxf(mx, to_f32(lr[c]));
    mx = wave_max(mx);
    mx = __shfl(mx, 0, WAVE);
    float sum = 0.0f;
    for (int c = lane; c < ncols; c += WAVE) sum += __expf(to_f32(lr[c]) - mx);
    sum = wave_sum(sum);
    sum = __shfl(sum, 0, WAVE);
    float inv_sum = 1.0f / sum;
    T* sr = softmax + row * ncols;
    for (int c = lane; c < ncols; c += WAVE)
      sr[c] = from_f32<T>(__expf(to_f32(lr[c]) - mx) * inv_sum);
    if (lane == 0) {
      int64_t t = target[row];
      acc_loss += -(to_f32(lr[t]) - mx - logf(sum));
    }
  }
  // block-level partial loss (256 threads = 4 waves)
  __shared__ float lds[4];
  if (lane == 0) lds[wid] = acc_loss;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.0f;
    for (int w = 0; w < blockDim.x / WAVE; ++w) s += lds[w];
    atomicAdd(loss_partial, s);
  }
}

template <typename T>
__global__ void softmax_xent_bwd_kernel(const T* __restrict__ softmax,
                                        const int64_t* __restrict__ target,
                                        T* __restrict__ grad, int64_t nrows,
                                        int ncols, float scale,
                                        const float* __restrict__ scale_p
                                        = nullptr) {
  // device-resident upstream grad scale (graph capture / no host sync)
  if (scale_p != nullptr) scale *= scale_p[0];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < nrows * (int64_t)ncols; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = i / ncols;
    int col = (int)(i - row * ncols);
    float s = to_f32(softmax[i]);
    float onehot = (col == (int)target[row]) ? 1.0f : 0.0f;
    grad[i] = from_f32<T>((s - onehot) * scale);
  }
}

template <typename scalar_t> struct DevT { using type = scalar_t; };
template <> struct DevT<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct DevT<at::Half> { using type = _Float16; };

}  // namespace

std::vector<torch::Tensor> softmax_xent_fwd(torch::Tensor logits,
                                            torch::Tensor target) {
  TORCH_CHECK(logits.dim() == 2);
  int64_t nrows = logits.size(0);
  int ncols = (int)logits.size(1);
  auto softmax = torch::empty_like(logits);
  auto loss = torch::zeros({}, logits.options().dtype(torch::kFloat32));
  auto tgt = target.contiguous();
  int blocks = (int)std::min<int64_t>((nrows + 3) / 4, 2048);
  NN_DISPATCH(logits.scalar_type(), "softmax_xent_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((softmax_xent_fwd_kernel<T>), dim3(blocks), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(),
                       (const T*)logits.data_ptr(),
                       tgt.data_ptr<int64_t>(), (T*)softmax.data_ptr(),
                       loss.data_ptr<float>(), nrows, ncols);
  });
  HIP_CHECK_LAST();
  return {loss / (double)nrows, softmax};
}

torch::Tensor softmax_xent_bwd(torch::Tensor softmax, torch::Tensor target,
                               double gscale) {
  int64_t nrows = softmax.size(0);
  int ncols = (int)softmax.size(1);
  auto grad = torch::empty_like(softmax);
  auto tgt = target.contiguous();
  int64_t n = nrows * ncols;
  int blocks = (int)std::min<int64_t>((n + 255) / 256, 8192);
  float scale = (float)(gscale / (double)nrows);
  NN_DISPATCH(softmax.scalar_type(), "softmax_xent_bwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((softmax_xent_bwd_kernel<T>), dim3(blocks), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(),
                       (const T*)softmax.data_ptr(), tgt.data_ptr<int64_t>(),
                       (T*)grad.data_ptr(), nrows, ncols, scale);
  });
  HIP_CHECK_LAST();
  return grad;
}

// gscale as a device-resident f32 scalar: no host read of the upstream
// grad (required under hipGraph capture; also drops a per-step sync)
torch::Tensor softmax_xent_bwd_t(torch::Tensor softmax, torch::Tensor target,
                                 torch::Tensor gscale) {
  TORCH_CHECK(gscale.is_cuda() && gscale.scalar_type() == torch::kFloat32 &&
                  gscale.numel() == 1,
              "softmax_xent_bwd_t: gscale must be a 1-element f32 GPU tensor");
  int64_t nrows = softmax.size(0);
  int ncols = (int)softmax.size(1);
  auto grad = torch::empty_like(softmax);
  auto tgt = target.contiguous();
  int64_t n = nrows * ncols;
  int blocks = (int)std::min<int64_t>((n + 255) / 256, 8192);
  float inv_rows = (float)(1.0 / (double)nrows);
  NN_DISPATCH(softmax.scalar_type(), "softmax_xent_bwd_t", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((softmax_xent_bwd_kernel<T>), dim3(blocks), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(),
                       (const T*)softmax.data_ptr(), tgt.data_ptr<int64_t>(),
                       (T*)grad.data_ptr(), nrows, ncols, inv_rows,
                       gscale.data_ptr<float>());
  });
  HIP_CHECK_LAST();
  return grad;
}
