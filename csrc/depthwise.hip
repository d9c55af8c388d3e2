// Depthwise convolution (groups == channels), NHWC, channel-vectorized.
//
// MobileNetV2 / EfficientNet dw-convs are memory-bound (each output reads
// R*S inputs of ONE channel). The scalar one-thread-per-element form was
// ISSUE-bound, not bandwidth-bound: 9 bounds-checked 2-byte loads plus
// address math per output element (measured 30 ms of a 62 ms MNv2 step).
// NHWC makes channels the contiguous axis, so each thread now owns a
// VEC-channel vector (VEC=8 -> 16-byte loads) of one output pixel: 8x
// fewer address computations and full-width memory transactions. C is a
// multiple of 8 for every MNv2/EffNet dw layer; a scalar fallback covers
// the rest.
//
// wgrad is deterministic: each (block, pixel-group) writes its own partial
// filter image (plain stores), summed on the host in fixed order.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;

template <typename T, int VEC>
struct alignas(sizeof(T) * VEC) VecT {
  T v[VEC];
};

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------
template <typename T, int VEC>
__global__ void dwconv_fwd_vec_kernel(const T* __restrict__ x,
                                      const T* __restrict__ w,
                                      const float* __restrict__ bias,
                                      T* __restrict__ y, int64_t n_vec, int Cv,
                                      int C, int H, int W, int OH, int OW,
                                      int R, int S, int stride, int pad,
                                      int has_bias) {
  using V = VecT<T, VEC>;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_vec;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cv = (int)(i % Cv);
    int64_t t = i / Cv;
    int ow = (int)(t % OW);
    t /= OW;
    int oh = (int)(t % OH);
    int64_t nb = t / OH;
    int c0 = cv * VEC;
    float acc[VEC];
#pragma unroll
    for (int u = 0; u < VEC; ++u) acc[u] = has_bias ? bias[c0 + u] : 0.0f;
    const T* wc = w + (int64_t)c0;  // w stored [R, S, C] (see wrapper)
    for (int r = 0; r < R; ++r) {
      int ih = oh * stride - pad + r;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < S; ++s) {
        int iw = ow * stride - pad + s;
        if (iw < 0 || iw >= W) continue;
        V xv = *(const V*)&x[((nb * H + ih) * W + iw) * C + c0];
        V wv = *(const V*)&wc[((int64_t)r * S + s) * C];
#pragma unroll
        for (int u = 0; u < VEC; ++u)
          acc[u] += to_f32(xv.v[u]) * to_f32(wv.v[u]);
      }
    }
    V out;
#pragma unroll
    for (int u = 0; u < VEC; ++u) out.v[u] = from_f32<T>(acc[u]);
    *(V*)&y[i * VEC] = out;
  }
}

// scalar fallback (C not a multiple of VEC); w layout [C, R, S]
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

// ---------------------------------------------------------------------------
// dgrad
// ---------------------------------------------------------------------------
template <typename T, int VEC>
__global__ void dwconv_dgrad_vec_kernel(const T* __restrict__ gy,
                                        const T* __restrict__ w,
                                        T* __restrict__ dx, int64_t n_vec,
                                        int Cv, int C, int H, int W, int OH,
                                        int OW, int R, int S, int stride,
                                        int pad) {
  using V = VecT<T, VEC>;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_vec;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cv = (int)(i % Cv);
    int64_t t = i / Cv;
    int iw = (int)(t % W);
    t /= W;
    int ih = (int)(t % H);
    int64_t nb = t / H;
    int c0 = cv * VEC;
    float acc[VEC] = {};
    const T* wc = w + (int64_t)c0;  // [R, S, C]
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
        V gv = *(const V*)&gy[((nb * OH + oh) * OW + ow) * C + c0];
        V wv = *(const V*)&wc[((int64_t)r * S + s) * C];
#pragma unroll
        for (int u = 0; u < VEC; ++u)
          acc[u] += to_f32(gv.v[u]) * to_f32(wv.v[u]);
      }
    }
    V out;
#pragma unroll
    for (int u = 0; u < VEC; ++u) out.v[u] = from_f32<T>(acc[u]);
    *(V*)&dx[i * VEC] = out;
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

// ---------------------------------------------------------------------------
// wgrad: thread = (channel-vector, pixel stream). Each (block, stream)
// writes its own partial [C, R*S] image -- no atomics anywhere, host sums
// partials in fixed order (deterministic under --seed).
// ---------------------------------------------------------------------------
template <typename T, int VEC, int MAXTAPS>
__global__ void dwconv_wgrad_vec_kernel(const T* __restrict__ gy,
                                        const T* __restrict__ x,
                                        float* __restrict__ partials, int Cv,
                                        int C, int H, int W, int OH, int OW,
                                        int R, int S, int stride, int pad,
                                        int64_t npix, int cspan_v, int pgrp) {
  using V = VecT<T, VEC>;
  // dynamic LDS: [cspan_v][taps][VEC] f32 block accumulation region
  extern __shared__ float lregion[];
  int cv_l = threadIdx.x % cspan_v;
  int grp = threadIdx.x / cspan_v;
  int cv = blockIdx.x * cspan_v + cv_l;
  bool active = (grp < pgrp) && (cv < Cv);
  int c0 = cv * VEC;
  int taps = R * S;
  float acc[MAXTAPS][VEC];
#pragma unroll
  for (int t = 0; t < MAXTAPS; ++t)
#pragma unroll
    for (int u = 0; u < VEC; ++u) acc[t][u] = 0.0f;

  constexpr int CHUNK = 8;
  int64_t nchunks = (npix + CHUNK - 1) / CHUNK;
  if (active) {
    for (int64_t chunk = (int64_t)blockIdx.y * pgrp + grp; chunk < nchunks;
         chunk += (int64_t)gridDim.y * pgrp) {
#pragma unroll
      for (int uu = 0; uu < CHUNK; ++uu) {
        int64_t pix = chunk * CHUNK + uu;
        if (pix >= npix) break;
        int64_t t = pix;
        int ow = (int)(t % OW);
        t /= OW;
        int oh = (int)(t % OH);
        int64_t nb = t / OH;
        V gv = *(const V*)&gy[pix * C + c0];
        for (int r = 0; r < R; ++r) {
          int ih = oh * stride - pad + r;
          if (ih < 0 || ih >= H) continue;
          for (int s = 0; s < S; ++s) {
            int iw = ow * stride - pad + s;
            if (iw < 0 || iw >= W) continue;
            V xv = *(const V*)&x[((nb * H + ih) * W + iw) * C + c0];
#pragma unroll
            for (int u = 0; u < VEC; ++u)
              acc[r * S + s][u] += to_f32(gv.v[u]) * to_f32(xv.v[u]);
          }
        }
      }
    }
  }
  // ordered in-block combine across the pixel groups (deterministic:
  // group index order), so each BLOCK emits one partial row -- pgrp x
  // fewer partial rows lets the grid grow to occupancy-filling sizes
  // without exploding the partials reduce
  for (int g2 = 0; g2 < pgrp; ++g2) {
    if (grp == g2 && cv < Cv) {
      for (int t = 0; t < taps; ++t)
#pragma unroll
        for (int u = 0; u < VEC; ++u) {
          int slot = (cv_l * taps + t) * VEC + u;
          lregion[slot] = (g2 == 0 ? 0.0f : lregion[slot]) + acc[t][u];
        }
    }
    __syncthreads();
  }
  if (grp == 0 && cv < Cv) {
    float* out = partials + (int64_t)blockIdx.y * ((int64_t)C * taps);
    for (int t = 0; t < taps; ++t)
#pragma unroll
      for (int u = 0; u < VEC; ++u)
        out[(int64_t)(c0 + u) * taps + t] =
            lregion[(cv_l * taps + t) * VEC + u];
  }
}

// Sliding-window form for the dominant 3x3/stride-1 case: each thread
// walks a whole OUTPUT ROW of one image, keeping the 3x3 input window of
// VEC channels in registers -- 4 vector loads per output pixel (3 new
// window columns + gy) instead of 10, and VEC=4 keeps the accumulator
// bank at ~36 VGPRs so ~7 waves/SIMD hide the walk's load latency.
// Same one-partial-row-per-block in-block ordered combine as the generic
// vec kernel.
template <typename T, int VEC>
__global__ void dwconv_wgrad_win3_kernel(const T* __restrict__ gy,
                                         const T* __restrict__ x,
                                         float* __restrict__ partials,
                                         int Cv, int C, int H, int W, int OH,
                                         int OW, int pad, int64_t nrows,
                                         int cspan_v, int pgrp) {
  using V = VecT<T, VEC>;
  extern __shared__ float lregion3[];
  int cv_l = threadIdx.x % cspan_v;
  int grp = threadIdx.x / cspan_v;
  int cv = blockIdx.x * cspan_v + cv_l;
  bool active = (grp < pgrp) && (cv < Cv);
  int c0 = cv * VEC;
  float acc[9][VEC];
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int u = 0; u < VEC; ++u) acc[t][u] = 0.0f;

  if (active) {
    for (int64_t row = (int64_t)blockIdx.y * pgrp + grp; row < nrows;
         row += (int64_t)gridDim.y * pgrp) {
      int oh = (int)(row % OH);
      int64_t nb = row / OH;
      const T* xb = x + nb * ((int64_t)H * W) * C + c0;
      const T* gb = gy + (row * OW) * C + c0;
      int ih0 = oh - pad;  // input rows ih0 .. ih0+2
      // window columns w0|w1|w2 = input cols (ow-1, ow, ow+1) x 3 rows
      float w0[3][VEC], w1[3][VEC], w2[3][VEC];
#pragma unroll
      for (int r = 0; r < 3; ++r)
#pragma unroll
        for (int u = 0; u < VEC; ++u) { w0[r][u] = 0.0f; w1[r][u] = 0.0f; }
      // preload column for iw = 0 into w1 (iw = -1 stays zero in w0)
#pragma unroll
      for (int r = 0; r < 3; ++r) {
        int ih = ih0 + r;
        if (ih >= 0 && ih < H) {
          V v = *(const V*)&xb[((int64_t)ih * W + 0) * C];
#pragma unroll
          for (int u = 0; u < VEC; ++u) w1[r][u] = to_f32(v.v[u]);
        }
      }
      for (int ow = 0; ow < OW; ++ow) {
        int iw2 = ow + 1;  // rightmost window column (pad=1)
#pragma unroll
        for (int r = 0; r < 3; ++r) {
          int ih = ih0 + r;
          if (ih >= 0 && ih < H && iw2 < W) {
            V v = *(const V*)&xb[((int64_t)ih * W + iw2) * C];
#pragma unroll
            for (int u = 0; u < VEC; ++u) w2[r][u] = to_f32(v.v[u]);
          } else {
#pragma unroll
            for (int u = 0; u < VEC; ++u) w2[r][u] = 0.0f;
          }
        }
        V gv = *(const V*)&gb[(int64_t)ow * C];
#pragma unroll
        for (int r = 0; r < 3; ++r)
#pragma unroll
          for (int u = 0; u < VEC; ++u) {
            float g = to_f32(gv.v[u]);
            acc[r * 3 + 0][u] += g * w0[r][u];
            acc[r * 3 + 1][u] += g * w1[r][u];
            acc[r * 3 + 2][u] += g * w2[r][u];
          }
#pragma unroll
        for (int r = 0; r < 3; ++r)
#pragma unroll
          for (int u = 0; u < VEC; ++u) {
            w0[r][u] = w1[r][u];
            w1[r][u] = w2[r][u];
          }
      }
    }
  }
  // ordered in-block combine (see dwconv_wgrad_vec_kernel)
  for (int g2 = 0; g2 < pgrp; ++g2) {
    if (grp == g2 && cv < Cv) {
      for (int t = 0; t < 9; ++t)
#pragma unroll
        for (int u = 0; u < VEC; ++u) {
          int slot = (cv_l * 9 + t) * VEC + u;
          lregion3[slot] = (g2 == 0 ? 0.0f : lregion3[slot]) + acc[t][u];
        }
    }
    __syncthreads();
  }
  if (grp == 0 && cv < Cv) {
    float* out = partials + (int64_t)blockIdx.y * ((int64_t)C * 9);
    for (int t = 0; t < 9; ++t)
#pragma unroll
      for (int u = 0; u < VEC; ++u)
        out[(int64_t)(c0 + u) * 9 + t] = lregion3[(cv_l * 9 + t) * VEC + u];
  }
}

template <typename T, int MAXTAPS>
__global__ void dwconv_wgrad_kernel(const T* __restrict__ gy,
                                    const T* __restrict__ x,
                                    float* __restrict__ partials, int C,
                                    int H, int W, int OH, int OW, int R,
                                    int S, int stride, int pad,
                                    int64_t npix, int cspan, int pgrp) {
  int c_in_span = threadIdx.x % cspan;
  int grp = threadIdx.x / cspan;
  if (grp >= pgrp) return;
  int c = blockIdx.x * cspan + c_in_span;
  if (c >= C) return;
  float acc[MAXTAPS];
  int taps = R * S;
#pragma unroll
  for (int t = 0; t < MAXTAPS; ++t) acc[t] = 0.0f;
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
  int64_t row = (int64_t)blockIdx.y * pgrp + grp;
  float* out = partials + row * ((int64_t)C * taps);
  for (int t = 0; t < taps; ++t) out[(int64_t)c * taps + t] = acc[t];
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
  bool has_bias = bias.numel() > 0;
  torch::Tensor bias_f;
  if (has_bias) bias_f = bias.to(torch::kFloat32).contiguous();
  auto y = torch::empty({N, C, OH, OW},
                        x.options().memory_format(at::MemoryFormat::ChannelsLast));
  constexpr int VEC = 8;
  NN_DISPATCH(x.scalar_type(), "dwconv_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    auto stream = c10::hip::getCurrentHIPStream();
    if (C % VEC == 0 && sizeof(T) == 2) {
      // [C,1,R,S] -> [R,S,C] so the filter vector load matches x's layout
      auto wrsc = w.view({C, R * S}).t().contiguous();
      int Cv = C / VEC;
      int64_t n_vec = (int64_t)N * OH * OW * Cv;
      int blocks = (int)std::min<int64_t>((n_vec + kBlock - 1) / kBlock, 8192);
      hipLaunchKernelGGL((dwconv_fwd_vec_kernel<T, VEC>), dim3(blocks),
                         dim3(kBlock), 0, stream, (const T*)x.data_ptr(),
                         (const T*)wrsc.data_ptr(),
                         has_bias ? bias_f.data_ptr<float>() : nullptr,
                         (T*)y.data_ptr(), n_vec, Cv, C, H, W, OH, OW, R, S,
                         (int)stride, (int)pad, has_bias ? 1 : 0);
    } else {
      auto wc = w.contiguous();
      int64_t n_out = (int64_t)N * C * OH * OW;
      int blocks = (int)std::min<int64_t>((n_out + kBlock - 1) / kBlock, 8192);
      hipLaunchKernelGGL((dwconv_fwd_kernel<T>), dim3(blocks), dim3(kBlock), 0,
                         stream, (const T*)x.data_ptr(),
                         (const T*)wc.data_ptr(),
                         has_bias ? bias_f.data_ptr<float>() : nullptr,
                         (T*)y.data_ptr(), n_out, C, H, W, OH, OW, R, S,
                         (int)stride, (int)pad, has_bias ? 1 : 0);
    }
  });
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor dwconv_dgrad(torch::Tensor gy, torch::Tensor w, int64_t stride,
                           int64_t pad, int64_t H, int64_t W) {
  TORCH_CHECK(gy.dim() == 4 && gy.is_contiguous(at::MemoryFormat::ChannelsLast));
  int N = (int)gy.size(0), C = (int)gy.size(1), OH = (int)gy.size(2), OW = (int)gy.size(3);
  int R = (int)w.size(2), S = (int)w.size(3);
  auto dx = torch::empty({N, C, (int)H, (int)W},
                         gy.options().memory_format(at::MemoryFormat::ChannelsLast));
  constexpr int VEC = 8;
  NN_DISPATCH(gy.scalar_type(), "dwconv_dgrad", [&] {
    using T = typename DevT<scalar_t>::type;
    auto stream = c10::hip::getCurrentHIPStream();
    if (C % VEC == 0 && sizeof(T) == 2) {
      auto wrsc = w.view({C, R * S}).t().contiguous();
      int Cv = C / VEC;
      int64_t n_vec = (int64_t)N * H * W * Cv;
      int blocks = (int)std::min<int64_t>((n_vec + kBlock - 1) / kBlock, 8192);
      hipLaunchKernelGGL((dwconv_dgrad_vec_kernel<T, VEC>), dim3(blocks),
                         dim3(kBlock), 0, stream, (const T*)gy.data_ptr(),
                         (const T*)wrsc.data_ptr(), (T*)dx.data_ptr(), n_vec,
                         Cv, C, (int)H, (int)W, OH, OW, R, S, (int)stride,
                         (int)pad);
    } else {
      auto wc = w.contiguous();
      int64_t n_in = (int64_t)N * C * H * W;
      int blocks = (int)std::min<int64_t>((n_in + kBlock - 1) / kBlock, 8192);
      hipLaunchKernelGGL((dwconv_dgrad_kernel<T>), dim3(blocks), dim3(kBlock),
                         0, stream, (const T*)gy.data_ptr(),
                         (const T*)wc.data_ptr(), (T*)dx.data_ptr(), n_in, C,
                         (int)H, (int)W, OH, OW, R, S, (int)stride, (int)pad);
    }
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
  int taps = (int)(R * S);
  int64_t npix = (int64_t)N * OH * OW;
  TORCH_CHECK(taps <= 64, "dwconv_wgrad: filter too large");
  constexpr int VEC = 8;
  // NOISYNET_DW_WGRAD_VEC4=1: force the narrower accumulator form for
  // 3x3 too (occupancy experiment; acc[9][8] is 72 VGPRs)
  static const bool force4 = [] {
    const char* e = getenv("NOISYNET_DW_WGRAD_VEC4");
    return e && e[0] == '1';
  }();
  bool vec = !force4 && (C % VEC == 0) && gy.element_size() == 2 && taps <= 9;
  // 5x5 filters: acc[25][8] would spill; a 4-wide vector still beats the
  // scalar form 4x on address math
  bool vec4 = !vec && (C % 4 == 0) && gy.element_size() == 2 && taps <= 25;
  torch::Tensor parts;
  static const bool no_win3 = [] {
    const char* e = getenv("NOISYNET_DW_NO_WIN3");
    return e && e[0] == '1';
  }();
  bool win3 = !no_win3 && taps == 9 && stride == 1 && pad == 1 &&
              (C % 4 == 0) && gy.element_size() == 2;
  NN_DISPATCH(gy.scalar_type(), "dwconv_wgrad", [&] {
    using T = typename DevT<scalar_t>::type;
    auto stream = c10::hip::getCurrentHIPStream();
    if (win3) {
      constexpr int V4 = 4;
      int Cv = C / V4;
      int cspan_v = std::min(Cv, kBlock);
      int pgrp = kBlock / cspan_v;
      int cblocks = (Cv + cspan_v - 1) / cspan_v;
      int64_t nrows = (int64_t)N * OH;
      int mslices = (int)std::min<int64_t>(
          (nrows + pgrp - 1) / pgrp,
          std::max<int64_t>(1, 1024 / std::max(1, cblocks)));
      parts = torch::zeros({(int64_t)mslices, (int64_t)C * 9},
                           x.options().dtype(torch::kFloat32));
      size_t lds = (size_t)cspan_v * 9 * V4 * sizeof(float);
      hipLaunchKernelGGL((dwconv_wgrad_win3_kernel<T, V4>),
                         dim3(cblocks, mslices), dim3(kBlock), lds, stream,
                         (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                         parts.data_ptr<float>(), Cv, C, H, W, OH, OW,
                         (int)pad, nrows, cspan_v, pgrp);
    } else if (vec || vec4) {
      int V = vec ? VEC : 4;
      int Cv = C / V;
      int cspan_v = std::min(Cv, kBlock);
      int pgrp = kBlock / cspan_v;
      int cblocks = (Cv + cspan_v - 1) / cspan_v;
      int64_t nchunks = (npix + 7) / 8;
      // one partial row PER BLOCK (in-block ordered combine) keeps the
      // partials reduce 21x smaller; 512 blocks measured best (2048
      // thrashed L2: each block's 8-pixel tap windows are only reused
      // while resident)
      int mslices = (int)std::min<int64_t>(
          (nchunks + pgrp - 1) / pgrp,
          std::max<int64_t>(1, 512 / std::max(1, cblocks)));
      parts = torch::zeros({(int64_t)mslices, (int64_t)C * taps},
                           x.options().dtype(torch::kFloat32));
      size_t lds = (size_t)cspan_v * taps * V * sizeof(float);
      if (vec) {
        auto* kfn = &dwconv_wgrad_vec_kernel<T, VEC, 9>;
        if (lds > 64 * 1024)
          hipFuncSetAttribute((const void*)kfn,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)lds);
        hipLaunchKernelGGL(kfn,
                           dim3(cblocks, mslices), dim3(kBlock), lds, stream,
                           (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                           parts.data_ptr<float>(), Cv, C, H, W, OH, OW,
                           (int)R, (int)S, (int)stride, (int)pad, npix,
                           cspan_v, pgrp);
      } else {
        auto* kfn = &dwconv_wgrad_vec_kernel<T, 4, 25>;
        if (lds > 64 * 1024)
          hipFuncSetAttribute((const void*)kfn,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)lds);
        hipLaunchKernelGGL(kfn,
                           dim3(cblocks, mslices), dim3(kBlock), lds, stream,
                           (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                           parts.data_ptr<float>(), Cv, C, H, W, OH, OW,
                           (int)R, (int)S, (int)stride, (int)pad, npix,
                           cspan_v, pgrp);
      }
    } else {
      int cspan = std::min(C, kBlock);
      int pgrp = kBlock / cspan;
      int cblocks = (C + cspan - 1) / cspan;
      int64_t nchunks = (npix + 7) / 8;
      int mslices = (int)std::min<int64_t>(
          (nchunks + pgrp - 1) / pgrp,
          std::max<int64_t>(1, 512 / std::max(1, cblocks)));
      int64_t rows = (int64_t)mslices * pgrp;
      parts = torch::zeros({rows, (int64_t)C * taps},
                           x.options().dtype(torch::kFloat32));
      if (taps <= 9)
        hipLaunchKernelGGL((dwconv_wgrad_kernel<T, 9>),
                           dim3(cblocks, mslices), dim3(kBlock), 0, stream,
                           (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                           parts.data_ptr<float>(), C, H, W, OH, OW, (int)R,
                           (int)S, (int)stride, (int)pad, npix, cspan, pgrp);
      else if (taps <= 25)
        hipLaunchKernelGGL((dwconv_wgrad_kernel<T, 25>),
                           dim3(cblocks, mslices), dim3(kBlock), 0, stream,
                           (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                           parts.data_ptr<float>(), C, H, W, OH, OW, (int)R,
                           (int)S, (int)stride, (int)pad, npix, cspan, pgrp);
      else
        hipLaunchKernelGGL((dwconv_wgrad_kernel<T, 64>),
                           dim3(cblocks, mslices), dim3(kBlock), 0, stream,
                           (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                           parts.data_ptr<float>(), C, H, W, OH, OW, (int)R,
                           (int)S, (int)stride, (int)pad, npix, cspan, pgrp);
    }
  });
  HIP_CHECK_LAST();
  auto dw_f = parts.sum(0);
  return dw_f.view({C, 1, (int)R, (int)S}).to(x.scalar_type());
}
