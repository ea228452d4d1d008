// Fused BatchNorm2d (NCHW) forward/backward for CDNA4.
//
// "standalone BN fwd/bwd with Welford" (ResNet /
// Inception / CNN workloads). Layout: one block per channel for the
// reductions (contiguous HW runs per (n,c) keep accesses coalesced),
// elementwise grid-stride kernels for the apply passes. fp32 statistics;
// x in fp32 or bf16.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <typename T>
DEVINL float ldb(const T* p);
template <>
DEVINL float ldb<bf16_t>(const bf16_t* p) { return bf2f(*p); }
template <>
DEVINL float ldb<float>(const float* p) { return *p; }

template <typename T>
DEVINL void stb(T* p, float v);
template <>
DEVINL void stb<bf16_t>(bf16_t* p, float v) { *p = f2bf(v); }
template <>
DEVINL void stb<float>(float* p, float v) { *p = v; }

// per-channel sum + sumsq (one block per channel)
template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x,
                                float* __restrict__ mean,
                                float* __restrict__ rstd,
                                float* __restrict__ running_mean,
                                float* __restrict__ running_var,
                                int C, long N, long HW, float eps,
                                float momentum, int update_running) {
  __shared__ float scratch[16];
  const int c = blockIdx.x;
  if (c >= C) return;
  const long cnt = N * HW;
  float s = 0.f, s2 = 0.f;
  for (long i = threadIdx.x; i < cnt; i += blockDim.x) {
    const long n = i / HW, hw = i % HW;
    const float v = ldb(x + (n * C + c) * HW + hw);
    s += v;
    s2 += v * v;
  }
  s = block_sum(s, scratch);
  s2 = block_sum(s2, scratch);
  if (threadIdx.x == 0) {
    const float mu = s / cnt;
    const float var = fmaxf(s2 / cnt - mu * mu, 0.f);
    mean[c] = mu;
    rstd[c] = rsqrtf(var + eps);
    if (update_running) {
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
      // running_var uses the unbiased estimate (torch semantics)
      const float ub = (cnt > 1) ? var * cnt / (cnt - 1) : var;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * ub;
    }
  }
}

template <typename T>
__global__ void bn_apply_kernel(const T* __restrict__ x,
                                const float* __restrict__ mean,
                                const float* __restrict__ rstd,
                                const float* __restrict__ w,
                                const float* __restrict__ b,
                                T* __restrict__ y, int C, long total,
                                long HW) {
  const long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < total; i += stride) {
    const int c = (int)((i / HW) % C);
    const float v = (ldb(x + i) - mean[c]) * rstd[c] * w[c] + b[c];
    stb(const_cast<T*>(y) + i, v);
  }
}

// per-channel sum(dy) and sum(dy * xhat)
template <typename T>
__global__ void bn_bwd_stats_kernel(const T* __restrict__ dy,
                                    const T* __restrict__ x,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    float* __restrict__ sum_dy,
                                    float* __restrict__ sum_dyxh, int C,
                                    long N, long HW) {
  __shared__ float scratch[16];
  const int c = blockIdx.x;
  if (c >= C) return;
  const long cnt = N * HW;
  const float mu = mean[c], rs = rstd[c];
  float s1 = 0.f, s2 = 0.f;
  for (long i = threadIdx.x; i < cnt; i += blockDim.x) {
    const long n = i / HW, hw = i % HW;
    const long off = (n * C + c) * HW + hw;
    const float g = ldb(dy + off);
    s1 += g;
    s2 += g * (ldb(x + off) - mu) * rs;
  }
  s1 = block_sum(s1, scratch);
  s2 = block_sum(s2, scratch);
  if (threadIdx.x == 0) {
    sum_dy[c] = s1;
    sum_dyxh[c] = s2;
  }
}

template <typename T>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ dy,
                                    const T* __restrict__ x,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    const float* __restrict__ w,
                                    const float* __restrict__ sum_dy,
                                    const float* __restrict__ sum_dyxh,
                                    T* __restrict__ dx, int C, long total,
                                    long HW, long cnt) {
  const long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < total; i += stride) {
    const int c = (int)((i / HW) % C);
    const float xh = (ldb(x + i) - mean[c]) * rstd[c];
    const float g = ldb(dy + i);
    const float d = w[c] * rstd[c] *
        (g - sum_dy[c] / cnt - xh * sum_dyxh[c] / cnt);
    stb(const_cast<T*>(dx) + i, d);
  }
}

int bn_grid(long total) {
  return (int)std::min<long>((total + 255) / 256, 2048);
}

}  // namespace

std::vector<at::Tensor> batchnorm_fwd(at::Tensor x, at::Tensor w,
                                      at::Tensor b, at::Tensor running_mean,
                                      at::Tensor running_var, bool training,
                                      double momentum, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  TORCH_CHECK(w.scalar_type() == at::kFloat && b.scalar_type() == at::kFloat);
  const long N = x.size(0), HW = x.size(2) * x.size(3);
  const int C = x.size(1);
  auto y = at::empty_like(x);
  auto mean = at::empty({C}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({C}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  const long total = x.numel();

#define BN_FWD(T)                                                           \
  do {                                                                      \
    if (training) {                                                         \
      hipLaunchKernelGGL((bn_stats_kernel<T>), dim3(C), dim3(256), 0,       \
                         stream, reinterpret_cast<const T*>(x.data_ptr()),  \
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),    \
                         running_mean.data_ptr<float>(),                    \
                         running_var.data_ptr<float>(), C, N, HW,           \
                         (float)eps, (float)momentum, 1);                   \
    } else {                                                                \
      mean.copy_(running_mean);                                             \
      rstd.copy_((running_var + eps).rsqrt());                              \
    }                                                                       \
    hipLaunchKernelGGL((bn_apply_kernel<T>), dim3(bn_grid(total)),          \
                       dim3(256), 0, stream,                                \
                       reinterpret_cast<const T*>(x.data_ptr()),            \
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),      \
                       w.data_ptr<float>(), b.data_ptr<float>(),            \
                       reinterpret_cast<T*>(y.data_ptr()), C, total, HW);   \
  } while (0)

  if (x.scalar_type() == at::kBFloat16) {
    BN_FWD(bf16_t);
  } else {
    BN_FWD(float);
  }
  HIP_CHECK_LAST();
  return {y, mean, rstd};
}

std::vector<at::Tensor> batchnorm_bwd(at::Tensor dy, at::Tensor x,
                                      at::Tensor w, at::Tensor mean,
                                      at::Tensor rstd) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  const long N = x.size(0), HW = x.size(2) * x.size(3);
  const int C = x.size(1);
  const long total = x.numel();
  const long cnt = N * HW;
  auto dx = at::empty_like(x);
  auto sum_dy = at::empty({C}, x.options().dtype(at::kFloat));
  auto sum_dyxh = at::empty({C}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();

#define BN_BWD(T)                                                           \
  do {                                                                      \
    hipLaunchKernelGGL((bn_bwd_stats_kernel<T>), dim3(C), dim3(256), 0,     \
                       stream, reinterpret_cast<const T*>(dy.data_ptr()),   \
                       reinterpret_cast<const T*>(x.data_ptr()),            \
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),      \
                       sum_dy.data_ptr<float>(),                            \
                       sum_dyxh.data_ptr<float>(), C, N, HW);               \
    hipLaunchKernelGGL((bn_bwd_apply_kernel<T>), dim3(bn_grid(total)),      \
                       dim3(256), 0, stream,                                \
                       reinterpret_cast<const T*>(dy.data_ptr()),           \
                       reinterpret_cast<const T*>(x.data_ptr()),            \
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),      \
                       w.data_ptr<float>(), sum_dy.data_ptr<float>(),       \
                       sum_dyxh.data_ptr<float>(),                          \
                       reinterpret_cast<T*>(dx.data_ptr()), C, total, HW,   \
                       cnt);                                                \
  } while (0)

  if (x.scalar_type() == at::kBFloat16) {
    BN_BWD(bf16_t);
  } else {
    BN_BWD(float);
  }
  HIP_CHECK_LAST();
  // dw = sum_dyxh, db = sum_dy (already per-channel)
  return {dx, sum_dyxh, sum_dy};
}
