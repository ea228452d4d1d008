// Pooling kernels for CDNA4 (gfx950): MaxPool2d / AvgPool2d (small
// windows) and global average pool, NCHW, fp32 + bf16.
//
// All three are pure-bandwidth ops; the design rule is one pass and no
// atomics. MaxPool backward is a GATHER: each input element walks the
// <=ceil(K/S)^2 output windows that cover it and accumulates dy where
// the saved argmax picked it — deterministic, unlike the scatter-atomic
// formulation (overlapping 3x3-stride-2 windows collide).
//
// Workload parity: reference ResNet/Inception/CNN pooling
// (SURVEY.md section 2.3 "MaxPool/AvgPool/AdaptiveAvgPool" row).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <typename T>
DEVINL float load1(const T* p);
template <>
DEVINL float load1<bf16_t>(const bf16_t* p) { return bf2f(*p); }
template <>
DEVINL float load1<float>(const float* p) { return *p; }
template <typename T>
DEVINL void store1(T* p, float v);
template <>
DEVINL void store1<bf16_t>(bf16_t* p, float v) { *p = f2bf(v); }
template <>
DEVINL void store1<float>(float* p, float v) { *p = v; }

template <typename T>
__global__ void maxpool_fwd_kernel(const T* __restrict__ x,
                                   T* __restrict__ y,
                                   int* __restrict__ idx, int C, int H,
                                   int W, int Ho, int Wo, int K, int S,
                                   int P, long total) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  const int wo = (int)(i % Wo);
  const int ho = (int)((i / Wo) % Ho);
  const long nc = i / ((long)Wo * Ho);
  const T* xp = x + nc * H * W;
  const int h0 = ho * S - P, w0 = wo * S - P;
  float best = -INFINITY;
  int bi = -1;
  for (int kh = 0; kh < K; ++kh) {
    const int h = h0 + kh;
    if (h < 0 || h >= H) continue;
    for (int kw = 0; kw < K; ++kw) {
      const int w = w0 + kw;
      if (w < 0 || w >= W) continue;
      const float v = load1(xp + h * W + w);
      if (v > best) {
        best = v;
        bi = h * W + w;
      }
    }
  }
  store1(y + i, best);
  idx[i] = bi;
}

template <typename T>
__global__ void maxpool_bwd_kernel(const T* __restrict__ dy,
                                   const int* __restrict__ idx,
                                   T* __restrict__ dx, int C, int H, int W,
                                   int Ho, int Wo, int K, int S, int P,
                                   long total_in) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total_in) return;
  const int w = (int)(i % W);
  const int h = (int)((i / W) % H);
  const long nc = i / ((long)W * H);
  const int me = h * W + w;
  const T* dyp = dy + nc * (long)Ho * Wo;
  const int* ip = idx + nc * (long)Ho * Wo;
  float acc = 0.f;
  // output windows covering (h, w): ho in [ceil((h+P-K+1)/S), (h+P)/S]
  const int ho_lo = max(0, (h + P - K + S) / S);  // ceil for positives
  const int ho_hi = min(Ho - 1, (h + P) / S);
  const int wo_lo = max(0, (w + P - K + S) / S);
  const int wo_hi = min(Wo - 1, (w + P) / S);
  for (int ho = ho_lo; ho <= ho_hi; ++ho)
    for (int wo = wo_lo; wo <= wo_hi; ++wo)
      if (ip[ho * Wo + wo] == me) acc += load1(dyp + ho * Wo + wo);
  store1(dx + i, acc);
}

template <typename T>
__global__ void avgpool_fwd_kernel(const T* __restrict__ x,
                                   T* __restrict__ y, int C, int H, int W,
                                   int Ho, int Wo, int K, int S, int P,
                                   int include_pad, long total) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  const int wo = (int)(i % Wo);
  const int ho = (int)((i / Wo) % Ho);
  const long nc = i / ((long)Wo * Ho);
  const T* xp = x + nc * H * W;
  const int h0 = ho * S - P, w0 = wo * S - P;
  float acc = 0.f;
  int cnt = 0;
  for (int kh = 0; kh < K; ++kh) {
    const int h = h0 + kh;
    if (h < 0 || h >= H) continue;
    for (int kw = 0; kw < K; ++kw) {
      const int w = w0 + kw;
      if (w < 0 || w >= W) continue;
      acc += load1(xp + h * W + w);
      ++cnt;
    }
  }
  const int div = include_pad ? K * K : max(cnt, 1);
  store1(y + i, acc / div);
}

template <typename T>
__global__ void avgpool_bwd_kernel(const T* __restrict__ dy,
                                   T* __restrict__ dx, int C, int H, int W,
                                   int Ho, int Wo, int K, int S, int P,
                                   int include_pad, long total_in) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total_in) return;
  const int w = (int)(i % W);
  const int h = (int)((i / W) % H);
  const long nc = i / ((long)W * H);
  const T* dyp = dy + nc * (long)Ho * Wo;
  float acc = 0.f;
  const int ho_lo = max(0, (h + P - K + S) / S);
  const int ho_hi = min(Ho - 1, (h + P) / S);
  const int wo_lo = max(0, (w + P - K + S) / S);
  const int wo_hi = min(Wo - 1, (w + P) / S);
  for (int ho = ho_lo; ho <= ho_hi; ++ho) {
    for (int wo = wo_lo; wo <= wo_hi; ++wo) {
      int div = K * K;
      if (!include_pad) {
        const int h0 = ho * S - P, w0 = wo * S - P;
        const int hc = min(h0 + K, H) - max(h0, 0);
        const int wc = min(w0 + K, W) - max(w0, 0);
        div = max(hc * wc, 1);
      }
      acc += load1(dyp + ho * Wo + wo) / div;
    }
  }
  store1(dx + i, acc);
}

// global average pool to (1,1): one wave per (n,c) plane
template <typename T>
__global__ void gap_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                               int HW, long NC) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long nc = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (nc >= NC) return;
  const T* xp = x + nc * (long)HW;
  float acc = 0.f;
  for (int j = lane; j < HW; j += WAVE) acc += load1(xp + j);
  acc = wave_sum(acc);
  if (lane == 0) store1(y + nc, acc / HW);
}

template <typename T>
__global__ void gap_bwd_kernel(const T* __restrict__ dy, T* __restrict__ dx,
                               int HW, long total) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  store1(dx + i, load1(dy + i / HW) / HW);
}

int nblocks(long total, int block) {
  return (int)((total + block - 1) / block);
}

}  // namespace

#define POOL_DISPATCH(T_CALL_F32, T_CALL_BF16, t)                     \
  do {                                                                \
    if ((t).scalar_type() == at::kBFloat16) {                         \
      T_CALL_BF16;                                                    \
    } else {                                                          \
      T_CALL_F32;                                                     \
    }                                                                 \
  } while (0)

std::vector<at::Tensor> maxpool2d_fwd(at::Tensor x, long K, long S, long P) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  const int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3);
  const int Ho = (int)((H + 2 * P - K) / S + 1);
  const int Wo = (int)((W + 2 * P - K) / S + 1);
  auto y = at::empty({N, C, Ho, Wo}, x.options());
  auto idx = at::empty({N, C, Ho, Wo}, x.options().dtype(at::kInt));
  const long total = (long)N * C * Ho * Wo;
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid(nblocks(total, 256));
  POOL_DISPATCH(
      hipLaunchKernelGGL((maxpool_fwd_kernel<float>), grid, block, 0, stream,
                         x.data_ptr<float>(), y.data_ptr<float>(),
                         idx.data_ptr<int>(), C, H, W, Ho, Wo, (int)K,
                         (int)S, (int)P, total),
      hipLaunchKernelGGL((maxpool_fwd_kernel<bf16_t>), grid, block, 0,
                         stream,
                         reinterpret_cast<const bf16_t*>(x.data_ptr()),
                         reinterpret_cast<bf16_t*>(y.data_ptr()),
                         idx.data_ptr<int>(), C, H, W, Ho, Wo, (int)K,
                         (int)S, (int)P, total),
      x);
  HIP_CHECK_LAST();
  return {y, idx};
}

at::Tensor maxpool2d_bwd(at::Tensor dy, at::Tensor idx, long H, long W,
                         long K, long S, long P) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  const int N = (int)dy.size(0), C = (int)dy.size(1),
            Ho = (int)dy.size(2), Wo = (int)dy.size(3);
  auto dx = at::empty({N, C, H, W}, dy.options());
  const long total_in = (long)N * C * H * W;
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid(nblocks(total_in, 256));
  POOL_DISPATCH(
      hipLaunchKernelGGL((maxpool_bwd_kernel<float>), grid, block, 0, stream,
                         dy.data_ptr<float>(), idx.data_ptr<int>(),
                         dx.data_ptr<float>(), C, (int)H, (int)W, Ho, Wo,
                         (int)K, (int)S, (int)P, total_in),
      hipLaunchKernelGGL((maxpool_bwd_kernel<bf16_t>), grid, block, 0,
                         stream,
                         reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                         idx.data_ptr<int>(),
                         reinterpret_cast<bf16_t*>(dx.data_ptr()), C, (int)H,
                         (int)W, Ho, Wo, (int)K, (int)S, (int)P, total_in),
      dy);
  HIP_CHECK_LAST();
  return dx;
}

at::Tensor avgpool2d_fwd(at::Tensor x, long K, long S, long P,
                         bool include_pad) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  const int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3);
  const int Ho = (int)((H + 2 * P - K) / S + 1);
  const int Wo = (int)((W + 2 * P - K) / S + 1);
  auto y = at::empty({N, C, Ho, Wo}, x.options());
  const long total = (long)N * C * Ho * Wo;
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid(nblocks(total, 256));
  POOL_DISPATCH(
      hipLaunchKernelGGL((avgpool_fwd_kernel<float>), grid, block, 0, stream,
                         x.data_ptr<float>(), y.data_ptr<float>(), C, H, W,
                         Ho, Wo, (int)K, (int)S, (int)P, include_pad ? 1 : 0,
                         total),
      hipLaunchKernelGGL((avgpool_fwd_kernel<bf16_t>), grid, block, 0,
                         stream,
                         reinterpret_cast<const bf16_t*>(x.data_ptr()),
                         reinterpret_cast<bf16_t*>(y.data_ptr()), C, H, W,
                         Ho, Wo, (int)K, (int)S, (int)P, include_pad ? 1 : 0,
                         total),
      x);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor avgpool2d_bwd(at::Tensor dy, long H, long W, long K, long S,
                         long P, bool include_pad) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  const int N = (int)dy.size(0), C = (int)dy.size(1),
            Ho = (int)dy.size(2), Wo = (int)dy.size(3);
  auto dx = at::empty({N, C, H, W}, dy.options());
  const long total_in = (long)N * C * H * W;
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid(nblocks(total_in, 256));
  POOL_DISPATCH(
      hipLaunchKernelGGL((avgpool_bwd_kernel<float>), grid, block, 0, stream,
                         dy.data_ptr<float>(), dx.data_ptr<float>(), C,
                         (int)H, (int)W, Ho, Wo, (int)K, (int)S, (int)P,
                         include_pad ? 1 : 0, total_in),
      hipLaunchKernelGGL((avgpool_bwd_kernel<bf16_t>), grid, block, 0,
                         stream,
                         reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                         reinterpret_cast<bf16_t*>(dx.data_ptr()), C, (int)H,
                         (int)W, Ho, Wo, (int)K, (int)S, (int)P,
                         include_pad ? 1 : 0, total_in),
      dy);
  HIP_CHECK_LAST();
  return dx;
}

at::Tensor global_avgpool_fwd(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  const long NC = x.size(0) * x.size(1);
  const int HW = (int)(x.size(2) * x.size(3));
  auto y = at::empty({x.size(0), x.size(1), 1, 1}, x.options());
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid(nblocks(NC, 4));
  POOL_DISPATCH(
      hipLaunchKernelGGL((gap_fwd_kernel<float>), grid, block, 0, stream,
                         x.data_ptr<float>(), y.data_ptr<float>(), HW, NC),
      hipLaunchKernelGGL((gap_fwd_kernel<bf16_t>), grid, block, 0, stream,
                         reinterpret_cast<const bf16_t*>(x.data_ptr()),
                         reinterpret_cast<bf16_t*>(y.data_ptr()), HW, NC),
      x);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor global_avgpool_bwd(at::Tensor dy, long H, long W) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  auto dx = at::empty({dy.size(0), dy.size(1), H, W}, dy.options());
  const long total = dx.numel();
  const int HW = (int)(H * W);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid(nblocks(total, 256));
  POOL_DISPATCH(
      hipLaunchKernelGGL((gap_bwd_kernel<float>), grid, block, 0, stream,
                         dy.data_ptr<float>(), dx.data_ptr<float>(), HW,
                         total),
      hipLaunchKernelGGL((gap_bwd_kernel<bf16_t>), grid, block, 0, stream,
                         reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                         reinterpret_cast<bf16_t*>(dx.data_ptr()), HW,
                         total),
      dy);
  HIP_CHECK_LAST();
  return dx;
}
