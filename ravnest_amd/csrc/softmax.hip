// Standalone softmax fwd/bwd over the last dim (CDNA4, fp32 + bf16).
//
// Wave-per-row online pass: single read computes (max, sum) with the
// merge trick (same as ce_loss.hip), second read normalizes. Backward
// is dx = (dy - sum(dy*y)) * y. The fused-attention kernels keep their
// own in-register softmax; this op covers the standalone call sites
// (CNN head, sampling) — SURVEY.md section 2.3 "standalone softmax".
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <typename T>
DEVINL float ld1(const T* p);
template <>
DEVINL float ld1<bf16_t>(const bf16_t* p) { return bf2f(*p); }
template <>
DEVINL float ld1<float>(const float* p) { return *p; }
template <typename T>
DEVINL void st1(T* p, float v);
template <>
DEVINL void st1<bf16_t>(bf16_t* p, float v) { *p = f2bf(v); }
template <>
DEVINL void st1<float>(float* p, float v) { *p = v; }

template <typename T>
__global__ void softmax_fwd_kernel(const T* __restrict__ x,
                                   T* __restrict__ y, int D, long N) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (row >= N) return;
  const T* xr = x + row * (long)D;
  // online (m, s) merge in one read
  float m = -INFINITY, s = 0.f;
  for (int j = lane; j < D; j += WAVE) {
    const float v = ld1(xr + j);
    if (v > m) {
      s = s * __expf(m - v) + 1.f;
      m = v;
    } else {
      s += __expf(v - m);
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float mo = __shfl_xor(m, off, WAVE);
    const float so = __shfl_xor(s, off, WAVE);
    const float mn = fmaxf(m, mo);
    // lanes past a short row carry m = -inf: exp(-inf - -inf) is NaN,
    // so zero those contributions explicitly
    s = (m == -INFINITY ? 0.f : s * __expf(m - mn)) +
        (mo == -INFINITY ? 0.f : so * __expf(mo - mn));
    m = mn;
  }
  const float inv = 1.f / s;
  T* yr = y + row * (long)D;
  for (int j = lane; j < D; j += WAVE)
    st1(yr + j, __expf(ld1(xr + j) - m) * inv);
}

template <typename T>
__global__ void softmax_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ y,
                                   T* __restrict__ dx, int D, long N) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (row >= N) return;
  const T* dr = dy + row * (long)D;
  const T* yr = y + row * (long)D;
  float dot = 0.f;
  for (int j = lane; j < D; j += WAVE) dot += ld1(dr + j) * ld1(yr + j);
  dot = wave_sum(dot);
  T* xr = dx + row * (long)D;
  for (int j = lane; j < D; j += WAVE)
    st1(xr + j, (ld1(dr + j) - dot) * ld1(yr + j));
}

}  // namespace

at::Tensor softmax_fwd(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int D = (int)x.size(-1);
  const long N = x.numel() / D;
  auto y = at::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid((unsigned)((N + 3) / 4));
  if (x.scalar_type() == at::kBFloat16)
    hipLaunchKernelGGL((softmax_fwd_kernel<bf16_t>), grid, block, 0, stream,
                       reinterpret_cast<const bf16_t*>(x.data_ptr()),
                       reinterpret_cast<bf16_t*>(y.data_ptr()), D, N);
  else
    hipLaunchKernelGGL((softmax_fwd_kernel<float>), grid, block, 0, stream,
                       x.data_ptr<float>(), y.data_ptr<float>(), D, N);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor softmax_bwd(at::Tensor dy, at::Tensor y) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && y.is_contiguous());
  const int D = (int)dy.size(-1);
  const long N = dy.numel() / D;
  auto dx = at::empty_like(dy);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid((unsigned)((N + 3) / 4));
  if (dy.scalar_type() == at::kBFloat16)
    hipLaunchKernelGGL((softmax_bwd_kernel<bf16_t>), grid, block, 0, stream,
                       reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                       reinterpret_cast<const bf16_t*>(y.data_ptr()),
                       reinterpret_cast<bf16_t*>(dx.data_ptr()), D, N);
  else
    hipLaunchKernelGGL((softmax_bwd_kernel<float>), grid, block, 0, stream,
                       dy.data_ptr<float>(), y.data_ptr<float>(),
                       dx.data_ptr<float>(), D, N);
  HIP_CHECK_LAST();
  return dx;
}
