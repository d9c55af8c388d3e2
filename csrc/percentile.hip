// Exact k-th order statistic via 4-pass radix select over monotone float
// keys (the percentile calibration of QuantMeasure, reference
// hardware_model.py:233-249 torch.kthvalue). Runs only during the 5-batch
// calibration window, so a host-synced 256-bin histogram loop is fine.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;

DEV_INLINE uint32_t f32_key(float f) {
  uint32_t u = __float_as_uint(f);
  return u ^ (((int32_t)u >> 31) | 0x80000000u);
}

template <typename T>
__global__ void radix_hist_kernel(const T* __restrict__ x, int64_t n,
                                  uint32_t prefix, uint32_t prefix_mask,
                                  int shift,
                                  unsigned long long* __restrict__ hist) {
  __shared__ unsigned int lh[256];
  for (int i = threadIdx.x; i < 256; i += blockDim.x) lh[i] = 0;
  __syncthreads();
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t key = f32_key(to_f32(x[i]));
    if ((key & prefix_mask) == prefix) {
      atomicAdd(&lh[(key >> shift) & 0xFF], 1u);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 256; i += blockDim.x)
    if (lh[i]) atomicAdd(&hist[i], (unsigned long long)lh[i]);
}

template <typename scalar_t> struct DevT { using type = scalar_t; };
template <> struct DevT<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct DevT<at::Half> { using type = _Float16; };

}  // namespace

torch::Tensor kth_percentile(torch::Tensor x, double pctl) {
  TORCH_CHECK(x.dim() == 1 && x.is_contiguous());
  int64_t n = x.numel();
  TORCH_CHECK(n > 0, "kth_percentile: empty tensor");
  int64_t k = (int64_t)(n * pctl / 100.0);
  if (k < 1) k = 1;
  if (k > n) k = n;

  auto hist_t = torch::zeros({256}, x.options().dtype(torch::kInt64));
  auto stream = c10::hip::getCurrentHIPStream();
  int blocks = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 4096);

  uint32_t prefix = 0, prefix_mask = 0;
  int64_t remaining_k = k;
  for (int pass = 0; pass < 4; ++pass) {
    int shift = 24 - 8 * pass;
    hist_t.zero_();
    NN_DISPATCH(x.scalar_type(), "radix_hist", [&] {
      using T = typename DevT<scalar_t>::type;
      hipLaunchKernelGGL((radix_hist_kernel<T>), dim3(blocks), dim3(kBlock), 0,
                         stream, (const T*)x.data_ptr(), n, prefix,
                         prefix_mask, shift,
                         (unsigned long long*)hist_t.data_ptr<int64_t>());
    });
    HIP_CHECK_LAST();
    auto h = hist_t.cpu();
    auto* hp = h.data_ptr<int64_t>();
    int64_t cum = 0;
    int bucket = 255;
    for (int b = 0; b < 256; ++b) {
      if (cum + hp[b] >= remaining_k) { bucket = b; break; }
      cum += hp[b];
    }
    remaining_k -= cum;
    prefix |= ((uint32_t)bucket) << shift;
    prefix_mask |= 0xFFu << shift;
  }

  // invert the monotone key transform
  uint32_t key = prefix;
  uint32_t u = (key & 0x80000000u) ? (key ^ 0x80000000u) : ~key;
  float val;
  std::memcpy(&val, &u, 4);
  auto out = torch::tensor(val, x.options().dtype(torch::kFloat32));
  return out.to(x.scalar_type());
}
