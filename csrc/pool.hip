// MaxPool 2x2 stride 2, NHWC. Forward stores a 2-bit argmax code per output
// (packed in uint8) so backward is a gather-free scatter.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;

template <typename T>
__global__ void maxpool2x2_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                      uint8_t* __restrict__ code, int64_t n_out,
                                      int C, int H, int W, int OH, int OW) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_out;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t t = i / C;
    int ow = (int)(t % OW);
    t /= OW;
    int oh = (int)(t % OH);
    int64_t nb = t / OH;
    const T* base = x + ((nb * H + 2 * oh) * W + 2 * ow) * C + c;
    float v00 = to_f32(base[0]);
    float v01 = to_f32(base[C]);
    float v10 = to_f32(base[(int64_t)W * C]);
    float v11 = to_f32(base[(int64_t)W * C + C]);
    float m = v00;
    int k = 0;
    if (v01 > m) { m = v01; k = 1; }
    if (v10 > m) { m = v10; k = 2; }
    if (v11 > m) { m = v11; k = 3; }
    y[i] = from_f32<T>(m);
    code[i] = (uint8_t)k;
  }
}

// Windows are non-overlapping, so each thread owns one (window, channel)
// and writes ALL four input cells (grad to the argmax cell, zero to the
// rest): full coverage, no separate zero-fill pass over gx.
template <typename T>
__global__ void maxpool2x2_bwd_kernel(const T* __restrict__ g,
                                      const uint8_t* __restrict__ code,
                                      T* __restrict__ gx, int64_t n_out, int C,
                                      int H, int W, int OH, int OW) {
  const T zero = from_f32<T>(0.0f);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_out;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t t = i / C;
    int ow = (int)(t % OW);
    t /= OW;
    int oh = (int)(t % OH);
    int64_t nb = t / OH;
    int k = code[i];
    T gv = g[i];
    T* base = gx + ((nb * H + 2 * oh) * W + 2 * ow) * C + c;
    base[0] = (k == 0) ? gv : zero;
    base[C] = (k == 1) ? gv : zero;
    base[(int64_t)W * C] = (k == 2) ? gv : zero;
    base[(int64_t)W * C + C] = (k == 3) ? gv : zero;
    // odd input dims: the last column/row belong to no window; zero them
    if ((W & 1) && ow == OW - 1) {
      base[2 * C] = zero;
      base[(int64_t)W * C + 2 * C] = zero;
    }
    if ((H & 1) && oh == OH - 1) {
      base[2 * (int64_t)W * C] = zero;
      base[2 * (int64_t)W * C + C] = zero;
      if ((W & 1) && ow == OW - 1) base[2 * (int64_t)W * C + 2 * C] = zero;
    }
  }
}

template <typename scalar_t> struct DevT { using type = scalar_t; };
template <> struct DevT<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct DevT<at::Half> { using type = _Float16; };

}  // namespace

std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor x) {
  TORCH_CHECK(x.dim() == 4 && x.is_contiguous(at::MemoryFormat::ChannelsLast));
  int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2), W = (int)x.size(3);
  int OH = H / 2, OW = W / 2;
  auto y = torch::empty({N, C, OH, OW},
                        x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto code = torch::empty({N, OH, OW, C}, x.options().dtype(torch::kUInt8));
  int64_t n_out = (int64_t)N * C * OH * OW;
  int blocks = (int)std::min<int64_t>((n_out + kBlock - 1) / kBlock, 8192);
  NN_DISPATCH(x.scalar_type(),
                                  "maxpool2x2_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((maxpool2x2_fwd_kernel<T>), dim3(blocks), dim3(kBlock), 0,
                       c10::hip::getCurrentHIPStream(), (const T*)x.data_ptr(),
                       (T*)y.data_ptr(), code.data_ptr<uint8_t>(), n_out, C, H,
                       W, OH, OW);
  });
  HIP_CHECK_LAST();
  return {y, code};
}

torch::Tensor maxpool2x2_bwd(torch::Tensor g, torch::Tensor code, int64_t H,
                             int64_t W) {
  TORCH_CHECK(g.dim() == 4 && g.is_contiguous(at::MemoryFormat::ChannelsLast));
  int N = (int)g.size(0), C = (int)g.size(1), OH = (int)g.size(2), OW = (int)g.size(3);
  // empty() (not zeros(): that factory drops memory_format from
  // TensorOptions) -- the kernel writes every element, no fill needed.
  auto gx = torch::empty({N, C, (int)H, (int)W},
                         g.options().memory_format(at::MemoryFormat::ChannelsLast));
  int64_t n_out = (int64_t)N * C * OH * OW;
  int blocks = (int)std::min<int64_t>((n_out + kBlock - 1) / kBlock, 8192);
  NN_DISPATCH(g.scalar_type(),
                                  "maxpool2x2_bwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((maxpool2x2_bwd_kernel<T>), dim3(blocks), dim3(kBlock), 0,
                       c10::hip::getCurrentHIPStream(), (const T*)g.data_ptr(),
                       code.data_ptr<uint8_t>(), (T*)gx.data_ptr(), n_out, C,
                       (int)H, (int)W, OH, OW);
  });
  HIP_CHECK_LAST();
  return gx;
}

// ---------------------------------------------------------------------------
// Generic pooling (kernel k x k, stride s, padding p), NHWC.
// Max pool stores a k*k argmax code (uint8, k <= 15) for the backward
// scatter; avg pool backward distributes g/(k*k) (count_include_pad=True,
// matching nn.AvgPool2d defaults used by the reference ResNet).
// ---------------------------------------------------------------------------

namespace {

template <typename T>
__global__ void maxpool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   uint8_t* __restrict__ code, int64_t n_out,
                                   int C, int H, int W, int OH, int OW, int k,
                                   int stride, int pad) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_out;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t t = i / C;
    int ow = (int)(t % OW);
    t /= OW;
    int oh = (int)(t % OH);
    int64_t nb = t / OH;
    float best = -INFINITY;
    int bestk = 0;
    for (int r = 0; r < k; ++r) {
      int ih = oh * stride - pad + r;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < k; ++s) {
        int iw = ow * stride - pad + s;
        if (iw < 0 || iw >= W) continue;
        float v = to_f32(x[((nb * H + ih) * W + iw) * C + c]);
        if (v > best) { best = v; bestk = r * k + s; }
      }
    }
    y[i] = from_f32<T>(best);
    code[i] = (uint8_t)bestk;
  }
}

template <typename T>
__global__ void maxpool_bwd_kernel(const T* __restrict__ g,
                                   const uint8_t* __restrict__ code,
                                   float* __restrict__ gx, int64_t n_out, int C,
                                   int H, int W, int OH, int OW, int k,
                                   int stride, int pad) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_out;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t t = i / C;
    int ow = (int)(t % OW);
    t /= OW;
    int oh = (int)(t % OH);
    int64_t nb = t / OH;
    int kk = code[i];
    int r = kk / k, s = kk % k;
    int ih = oh * stride - pad + r;
    int iw = ow * stride - pad + s;
    // overlapping windows (stride < k) need atomic accumulation
    atomicAdd(&gx[((nb * H + ih) * W + iw) * C + c], to_f32(g[i]));
  }
}

template <typename T>
__global__ void avgpool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   int64_t n_out, int C, int H, int W, int OH,
                                   int OW, int k, int stride, int pad) {
  float inv = 1.0f / (k * k);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_out;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t t = i / C;
    int ow = (int)(t % OW);
    t /= OW;
    int oh = (int)(t % OH);
    int64_t nb = t / OH;
    float acc = 0.0f;
    for (int r = 0; r < k; ++r) {
      int ih = oh * stride - pad + r;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < k; ++s) {
        int iw = ow * stride - pad + s;
        if (iw < 0 || iw >= W) continue;
        acc += to_f32(x[((nb * H + ih) * W + iw) * C + c]);
      }
    }
    y[i] = from_f32<T>(acc * inv);
  }
}

template <typename T>
__global__ void avgpool_bwd_kernel(const T* __restrict__ g,
                                   float* __restrict__ gx, int64_t n_out, int C,
                                   int H, int W, int OH, int OW, int k,
                                   int stride, int pad) {
  float inv = 1.0f / (k * k);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_out;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t t = i / C;
    int ow = (int)(t % OW);
    t /= OW;
    int oh = (int)(t % OH);
    int64_t nb = t / OH;
    float gv = to_f32(g[i]) * inv;
    for (int r = 0; r < k; ++r) {
      int ih = oh * stride - pad + r;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < k; ++s) {
        int iw = ow * stride - pad + s;
        if (iw < 0 || iw >= W) continue;
        atomicAdd(&gx[((nb * H + ih) * W + iw) * C + c], gv);
      }
    }
  }
}

}  // namespace

std::vector<torch::Tensor> maxpool_fwd(torch::Tensor x, int64_t k,
                                       int64_t stride, int64_t pad) {
  TORCH_CHECK(x.dim() == 4 && x.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(k <= 15);
  int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2), W = (int)x.size(3);
  int OH = (H + 2 * (int)pad - (int)k) / (int)stride + 1;
  int OW = (W + 2 * (int)pad - (int)k) / (int)stride + 1;
  auto y = torch::empty({N, C, OH, OW},
                        x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto code = torch::empty({N, OH, OW, C}, x.options().dtype(torch::kUInt8));
  int64_t n_out = (int64_t)N * C * OH * OW;
  int blocks = (int)std::min<int64_t>((n_out + 255) / 256, 8192);
  NN_DISPATCH(x.scalar_type(), "maxpool_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((maxpool_fwd_kernel<T>), dim3(blocks), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(), (const T*)x.data_ptr(),
                       (T*)y.data_ptr(), code.data_ptr<uint8_t>(), n_out, C, H,
                       W, OH, OW, (int)k, (int)stride, (int)pad);
  });
  HIP_CHECK_LAST();
  return {y, code};
}

torch::Tensor maxpool_bwd(torch::Tensor g, torch::Tensor code, int64_t H,
                          int64_t W, int64_t k, int64_t stride, int64_t pad) {
  TORCH_CHECK(g.dim() == 4 && g.is_contiguous(at::MemoryFormat::ChannelsLast));
  int N = (int)g.size(0), C = (int)g.size(1), OH = (int)g.size(2), OW = (int)g.size(3);
  auto gx_f = torch::zeros({N, (int)H, (int)W, C},
                           g.options().dtype(torch::kFloat32));
  int64_t n_out = (int64_t)N * C * OH * OW;
  int blocks = (int)std::min<int64_t>((n_out + 255) / 256, 8192);
  NN_DISPATCH(g.scalar_type(), "maxpool_bwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((maxpool_bwd_kernel<T>), dim3(blocks), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(), (const T*)g.data_ptr(),
                       code.data_ptr<uint8_t>(), gx_f.data_ptr<float>(), n_out,
                       C, (int)H, (int)W, OH, OW, (int)k, (int)stride, (int)pad);
  });
  HIP_CHECK_LAST();
  // raw NHWC f32 -> logical NCHW channels_last in g's dtype
  return gx_f.permute({0, 3, 1, 2}).to(g.scalar_type())
      .contiguous(at::MemoryFormat::ChannelsLast);
}

torch::Tensor avgpool_fwd(torch::Tensor x, int64_t k, int64_t stride,
                          int64_t pad) {
  TORCH_CHECK(x.dim() == 4 && x.is_contiguous(at::MemoryFormat::ChannelsLast));
  int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2), W = (int)x.size(3);
  int OH = (H + 2 * (int)pad - (int)k) / (int)stride + 1;
  int OW = (W + 2 * (int)pad - (int)k) / (int)stride + 1;
  auto y = torch::empty({N, C, OH, OW},
                        x.options().memory_format(at::MemoryFormat::ChannelsLast));
  int64_t n_out = (int64_t)N * C * OH * OW;
  int blocks = (int)std::min<int64_t>((n_out + 255) / 256, 8192);
  NN_DISPATCH(x.scalar_type(), "avgpool_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((avgpool_fwd_kernel<T>), dim3(blocks), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(), (const T*)x.data_ptr(),
                       (T*)y.data_ptr(), n_out, C, H, W, OH, OW, (int)k,
                       (int)stride, (int)pad);
  });
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor avgpool_bwd(torch::Tensor g, int64_t H, int64_t W, int64_t k,
                          int64_t stride, int64_t pad) {
  TORCH_CHECK(g.dim() == 4 && g.is_contiguous(at::MemoryFormat::ChannelsLast));
  int N = (int)g.size(0), C = (int)g.size(1), OH = (int)g.size(2), OW = (int)g.size(3);
  auto gx_f = torch::zeros({N, (int)H, (int)W, C},
                           g.options().dtype(torch::kFloat32));
  int64_t n_out = (int64_t)N * C * OH * OW;
  int blocks = (int)std::min<int64_t>((n_out + 255) / 256, 8192);
  NN_DISPATCH(g.scalar_type(), "avgpool_bwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((avgpool_bwd_kernel<T>), dim3(blocks), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(), (const T*)g.data_ptr(),
                       gx_f.data_ptr<float>(), n_out, C, (int)H, (int)W, OH,
                       OW, (int)k, (int)stride, (int)pad);
  });
  HIP_CHECK_LAST();
  return gx_f.permute({0, 3, 1, 2}).to(g.scalar_type())
      .contiguous(at::MemoryFormat::ChannelsLast);
}
