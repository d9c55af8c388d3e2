// Depthwise convolution (groups == channels), NHWC, register-tiled.
//
// MobileNetV2 / EfficientNet dw-convs are memory-bound elementwise-ish ops
// (each output reads k*k inputs of ONE channel): no MFMA, one thread per
// output element vectorized 2-wide over channels where possible, filter taps
// unrolled in registers (cdna_hip_programming.md Appendix B "element-wise").
// Forward, dgrad and wgrad (atomic f32 per-channel-tap reduce).

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;

// w layout: [C, 1, R, S] contiguous == raw [C, R, S]
template <typename T>
__global__ void dwconv_fwd_kernel(const T* __restrict__ x,
                                  const T* __restrict__ w,
                                  const float* __restrict__ bias,
                                  T* __restrict__ y, int64_t n_out, int C,
                                  int H, int W, int OH, int OW, int R, int S,
                                  int stride, int pad, int has_bias) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_out;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t t = i / C;
    int ow = (int)(t % OW);
    t /= OW;
    int oh = (int)(t % OH);
    int64_t nb = t / OH;
    float acc = has_bias ? bias[c] : 0.0f;
    const T* wc = w + (int64_t)c * R * S;
    for (int r = 0; r < R; ++r) {
      int ih = oh * stride - pad + r;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < S; ++s) {
        int iw = ow * stride - pad + s;
        if (iw < 0 || iw >= W) continue;
        acc += to_f32(x[((nb * H + ih) * W + iw) * C + c])
               * to_f32(wc[r * S + s]);
      }
    }
    y[i] = from_f32<T>(acc);
  }
}

template <typename T>
__global__ void dwconv_dgrad_kernel(const T* __restrict__ gy,
                                    const T* __restrict__ w,
                                    T* __restrict__ dx, int64_t n_in, int C,
                                    int H, int W, int OH, int OW, int R, int S,
                                    int stride, int pad) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_in;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t t = i / C;
    int iw = (int)(t % W);
    t /= W;
    int ih = (int)(t % H);
    int64_t nb = t / H;
    float acc = 0.0f;
    const T* wc = w + (int64_t)c * R * S;
    for (int r = 0; r < R; ++r) {
      int ohs = ih + pad - r;
      if (ohs < 0 || ohs % stride) continue;
      int oh = ohs / stride;
      if (oh >= OH) continue;
      for (int s = 0; s < S; ++s) {
        int ows = iw + pad - s;
        if (ows < 0 || ows % stride) continue;
        int ow = ows / stride;
        if (ow >= OW) continue;
        acc += to_f32(gy[((nb * OH + oh) * OW + ow) * C + c])
               * to_f32(wc[r * S + s]);
      }
    }
    dx[i] = from_f32<T>(acc);
  }
}

// Each thread owns ONE (channel, pixel-stream) pair, accumulating all
// R*S taps in registers; one atomicAdd per tap per thread (instead of
// one per OUTPUT ELEMENT, which serialized on the same dw[c][r][s] word
// ~N*OH*OW deep). Channels ride the FAST lane index so gy/x reads stay
// coalesced; for C < kBlock several pixel streams share a block so no
// lanes idle (the one-channel-per-thread layout left 224 of 256 lanes
// dead on MobileNet's 32-channel stem).
template <typename T, int MAXTAPS>
__global__ void dwconv_wgrad_kernel(const T* __restrict__ gy,
                                    const T* __restrict__ x,
                                    float* __restrict__ dw, int C,
                                    int H, int W, int OH, int OW, int R,
                                    int S, int stride, int pad,
                                    int64_t npix, int cspan, int pgrp) {
  // cspan = min(C, kBlock) rounded context: threads [0, cspan*pgrp)
  int c_in_span = threadIdx.x % cspan;
  int grp = threadIdx.x / cspan;
  if (grp >= pgrp) return;
  int c = blockIdx.x * cspan + c_in_span;
  if (c >= C) return;
  float acc[MAXTAPS];
  int taps = R * S;
#pragma unroll
  for (int t = 0; t < MAXTAPS; ++t) acc[t] = 0.0f;
  // pixel streams walk CONSECUTIVE pixels in chunks of 8 so the tap
  // windows of successive outputs overlap in cache (grid-strided single
  // pixels gave zero x reuse across iterations)
  constexpr int CHUNK = 8;
  int64_t nchunks = (npix + CHUNK - 1) / CHUNK;
  for (int64_t chunk = (int64_t)blockIdx.y * pgrp + grp; chunk < nchunks;
       chunk += (int64_t)gridDim.y * pgrp) {
#pragma unroll
    for (int u = 0; u < CHUNK; ++u) {
      int64_t pix = chunk * CHUNK + u;
      if (pix >= npix) break;
      int64_t t = pix;
      int ow = (int)(t % OW);
      t /= OW;
      int oh = (int)(t % OH);
      int64_t nb = t / OH;
      float g = to_f32(gy[pix * C + c]);
      for (int r = 0; r < R; ++r) {
        int ih = oh * stride - pad + r;
        if (ih < 0 || ih >= H) continue;
        for (int s = 0; s < S; ++s) {
          int iw = ow * stride - pad + s;
          if (iw < 0 || iw >= W) continue;
          acc[r * S + s] += g * to_f32(x[((nb * H + ih) * W + iw) * C + c]);
        }
      }
    }
  }
  for (int t = 0; t < taps; ++t)
    if (acc[t] != 0.0f) atomicAdd(&dw[(int64_t)c * taps + t], acc[t]);
}

template <typename scalar_t> struct DevT { using type = scalar_t; };
template <> struct DevT<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct DevT<at::Half> { using type = _Float16; };

}  // namespace

torch::Tensor dwconv_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         int64_t stride, int64_t pad) {
  TORCH_CHECK(x.dim() == 4 && x.is_contiguous(at::MemoryFormat::ChannelsLast));
  int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2), W = (int)x.size(3);
  int R = (int)w.size(2), S = (int)w.size(3);
  int OH = (H + 2 * (int)pad - R) / (int)stride + 1;
  int OW = (W + 2 * (int)pad - S) / (int)stride + 1;
  auto wc = w.contiguous();
  bool has_bias = bias.numel() > 0;
  torch::Tensor bias_f;
  if (has_bias) bias_f = bias.to(torch::kFloat32).contiguous();
  auto y = torch::empty({N, C, OH, OW},
                        x.options().memory_format(at::MemoryFormat::ChannelsLast));
  int64_t n_out = (int64_t)N * C * OH * OW;
  int blocks = (int)std::min<int64_t>((n_out + kBlock - 1) / kBlock, 8192);
  NN_DISPATCH(x.scalar_type(), "dwconv_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((dwconv_fwd_kernel<T>), dim3(blocks), dim3(kBlock), 0,
                       c10::hip::getCurrentHIPStream(), (const T*)x.data_ptr(),
                       (const T*)wc.data_ptr(),
                       has_bias ? bias_f.data_ptr<float>() : nullptr,
                       (T*)y.data_ptr(), n_out, C, H, W, OH, OW, R, S,
                       (int)stride, (int)pad, has_bias ? 1 : 0);
  });
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor dwconv_dgrad(torch::Tensor gy, torch::Tensor w, int64_t stride,
                           int64_t pad, int64_t H, int64_t W) {
  TORCH_CHECK(gy.dim() == 4 && gy.is_contiguous(at::MemoryFormat::ChannelsLast));
  int N = (int)gy.size(0), C = (int)gy.size(1), OH = (int)gy.size(2), OW = (int)gy.size(3);
  int R = (int)w.size(2), S = (int)w.size(3);
  auto wc = w.contiguous();
  auto dx = torch::empty({N, C, (int)H, (int)W},
                         gy.options().memory_format(at::MemoryFormat::ChannelsLast));
  int64_t n_in = (int64_t)N * C * H * W;
  int blocks = (int)std::min<int64_t>((n_in + kBlock - 1) / kBlock, 8192);
  NN_DISPATCH(gy.scalar_type(), "dwconv_dgrad", [&] {
    using T = typename DevT<scalar_t>::type;
    hipLaunchKernelGGL((dwconv_dgrad_kernel<T>), dim3(blocks), dim3(kBlock), 0,
                       c10::hip::getCurrentHIPStream(),
                       (const T*)gy.data_ptr(), (const T*)wc.data_ptr(),
                       (T*)dx.data_ptr(), n_in, C, (int)H, (int)W, OH, OW, R,
                       S, (int)stride, (int)pad);
  });
  HIP_CHECK_LAST();
  return dx;
}

torch::Tensor dwconv_wgrad(torch::Tensor gy, torch::Tensor x, int64_t stride,
                           int64_t pad, int64_t R, int64_t S) {
  TORCH_CHECK(gy.dim() == 4 && gy.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(x.dim() == 4 && x.is_contiguous(at::MemoryFormat::ChannelsLast));
  int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2), W = (int)x.size(3);
  int OH = (int)gy.size(2), OW = (int)gy.size(3);
  auto dw_f = torch::zeros({C, (int)R, (int)S}, x.options().dtype(torch::kFloat32));
  int64_t npix = (int64_t)N * OH * OW;
  int cspan = std::min(C, kBlock);
  int pgrp = kBlock / cspan;  // pixel streams sharing one block
  int cblocks = (C + cspan - 1) / cspan;
  int64_t nchunks = (npix + 7) / 8;
  int mslices = (int)std::min<int64_t>(
      (nchunks + pgrp - 1) / pgrp,
      std::max<int64_t>(1, 2048 / std::max(1, cblocks)));
  TORCH_CHECK(R * S <= 64, "dwconv_wgrad: filter too large");
  NN_DISPATCH(gy.scalar_type(), "dwconv_wgrad", [&] {
    using T = typename DevT<scalar_t>::type;
    auto stream = c10::hip::getCurrentHIPStream();
    if (R * S <= 9)
      hipLaunchKernelGGL((dwconv_wgrad_kernel<T, 9>), dim3(cblocks, mslices),
                         dim3(kBlock), 0, stream, (const T*)gy.data_ptr(),
                         (const T*)x.data_ptr(), dw_f.data_ptr<float>(), C, H,
                         W, OH, OW, (int)R, (int)S, (int)stride, (int)pad,
                         npix, cspan, pgrp);
    else if (R * S <= 25)
      hipLaunchKernelGGL((dwconv_wgrad_kernel<T, 25>), dim3(cblocks, mslices),
                         dim3(kBlock), 0, stream, (const T*)gy.data_ptr(),
                         (const T*)x.data_ptr(), dw_f.data_ptr<float>(), C, H,
                         W, OH, OW, (int)R, (int)S, (int)stride, (int)pad,
                         npix, cspan, pgrp);
    else
      hipLaunchKernelGGL((dwconv_wgrad_kernel<T, 64>), dim3(cblocks, mslices),
                         dim3(kBlock), 0, stream, (const T*)gy.data_ptr(),
                         (const T*)x.data_ptr(), dw_f.data_ptr<float>(), C, H,
                         W, OH, OW, (int)R, (int)S, (int)stride, (int)pad,
                         npix, cspan, pgrp);
  });
  HIP_CHECK_LAST();
  return dw_f.view({C, 1, (int)R, (int)S}).to(x.scalar_type());
}
