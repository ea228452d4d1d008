// Dropout with counter-based Philox RNG (replayable + graph-safe).
//
// The (seed, offset) pair comes from the HOST (drawn from the torch CPU
// generator by the python wrapper), so the versioned-recompute engine's
// RNG-state capture/restore replays the identical mask — SURVEY.md
// section 2.3 "Dropout (with replayable RNG)". Each thread generates 4
// uniforms per philox call; mask stored as uint8.
//
// hipGraph support: the kernel additionally XORs in a value read from a
// DEVICE counter buffer at run time. Outside graph mode the counter is 0
// (identical behavior); inside a captured step the engine increments the
// counter once per replay, so each replay of the frozen host seed draws
// a fresh mask (engine/graphstep.py).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

// vectorized: 8 elements/thread/iter (two philox quads, 16B x/y for
// bf16, 8B mask store); scalar tail for the last partial octet
template <typename T>
__global__ void dropout_fwd_kernel(const T* __restrict__ x,
                                   T* __restrict__ y,
                                   unsigned char* __restrict__ mask, long n,
                                   float p, float scale,
                                   unsigned long long seed,
                                   const long long* __restrict__ seed_buf) {
  const Philox4 ph(seed_buf ? seed ^ (unsigned long long)*seed_buf : seed);
  const long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  const long stride = (long)gridDim.x * blockDim.x;
  const float thresh = p;
  const long no = n >> 3;  // full octets
  for (long o = i0; o < no; o += stride) {
    unsigned int r[8];
    ph.gen((unsigned long long)(o * 2), 0ull, r);
    ph.gen((unsigned long long)(o * 2 + 1), 0ull, r + 4);
    unsigned long long mbits = 0;
    float vals[8];
    if constexpr (sizeof(T) == 2) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(x + o * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[j] = us2f((unsigned short)v[j]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[j] = ((const float*)x)[o * 8 + j];
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float u = (r[j] >> 8) * (1.0f / 16777216.0f);
      const bool keep = u >= thresh;
      mbits |= ((unsigned long long)(keep ? 1 : 0)) << (8 * j);
      vals[j] = keep ? vals[j] * scale : 0.f;
    }
    *reinterpret_cast<unsigned long long*>(mask + o * 8) = mbits;
    if constexpr (sizeof(T) == 2) {
      bf16x8 ov;
#pragma unroll
      for (int j = 0; j < 8; ++j) ov[j] = (short)f2us(vals[j]);
      *reinterpret_cast<bf16x8*>(y + o * 8) = ov;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) ((float*)y)[o * 8 + j] = vals[j];
    }
  }
  // tail (same philox indexing as the vector body)
  for (long i = (no << 3) + i0; i < n; i += stride) {
    unsigned int r[4];
    ph.gen((unsigned long long)(i >> 2), 0ull, r);
    const float u = (r[i & 3] >> 8) * (1.0f / 16777216.0f);
    const bool keep = u >= thresh;
    mask[i] = keep ? 1 : 0;
    float v;
    if constexpr (sizeof(T) == 2)
      v = bf2f(x[i]);
    else
      v = x[i];
    v = keep ? v * scale : 0.f;
    if constexpr (sizeof(T) == 2)
      y[i] = f2bf(v);
    else
      y[i] = v;
  }
}

template <typename T>
__global__ void dropout_bwd_kernel(const T* __restrict__ dy,
                                   const unsigned char* __restrict__ mask,
                                   T* __restrict__ dx, long n, float scale) {
  const long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  const long stride = (long)gridDim.x * blockDim.x;
  const long no = n >> 3;
  for (long o = i0; o < no; o += stride) {
    const unsigned long long mbits =
        *reinterpret_cast<const unsigned long long*>(mask + o * 8);
    if constexpr (sizeof(T) == 2) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(dy + o * 8);
      bf16x8 ov;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = us2f((unsigned short)v[j]);
        ov[j] = (short)f2us(((mbits >> (8 * j)) & 1) ? f * scale : 0.f);
      }
      *reinterpret_cast<bf16x8*>(dx + o * 8) = ov;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = ((const float*)dy)[o * 8 + j];
        ((float*)dx)[o * 8 + j] =
            ((mbits >> (8 * j)) & 1) ? f * scale : 0.f;
      }
    }
  }
  for (long i = (no << 3) + i0; i < n; i += stride) {
    float v;
    if constexpr (sizeof(T) == 2)
      v = bf2f(dy[i]);
    else
      v = dy[i];
    v = mask[i] ? v * scale : 0.f;
    if constexpr (sizeof(T) == 2)
      dx[i] = f2bf(v);
    else
      dx[i] = v;
  }
}

}  // namespace

static int grid_cap(long work, int block) {
  long g = (work + block - 1) / block;
  return (int)std::min<long>(g, 2048);
}

std::vector<at::Tensor> dropout_fwd(at::Tensor x, double p, int64_t seed,
                                    c10::optional<at::Tensor> seed_buf) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const long long* sb = nullptr;
  if (seed_buf.has_value()) {
    TORCH_CHECK(seed_buf->is_cuda() &&
                seed_buf->scalar_type() == at::kLong);
    sb = reinterpret_cast<const long long*>(seed_buf->data_ptr<int64_t>());
  }
  const long n = x.numel();
  auto y = at::empty_like(x);
  auto mask = at::empty({n}, x.options().dtype(at::kByte));
  const float scale = 1.0f / (1.0f - (float)p);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid(grid_cap((n + 3) / 4, 256));
  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((dropout_fwd_kernel<bf16_t>), grid, block, 0, stream,
                       reinterpret_cast<const bf16_t*>(x.data_ptr()),
                       reinterpret_cast<bf16_t*>(y.data_ptr()),
                       mask.data_ptr<unsigned char>(), n, (float)p, scale,
                       (unsigned long long)seed, sb);
  } else {
    hipLaunchKernelGGL((dropout_fwd_kernel<float>), grid, block, 0, stream,
                       x.data_ptr<float>(), y.data_ptr<float>(),
                       mask.data_ptr<unsigned char>(), n, (float)p, scale,
                       (unsigned long long)seed, sb);
  }
  HIP_CHECK_LAST();
  return {y, mask};
}

at::Tensor dropout_bwd(at::Tensor dy, at::Tensor mask, double p) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  const long n = dy.numel();
  auto dx = at::empty_like(dy);
  const float scale = 1.0f / (1.0f - (float)p);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid(grid_cap(n, 256));
  if (dy.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((dropout_bwd_kernel<bf16_t>), grid, block, 0, stream,
                       reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                       mask.data_ptr<unsigned char>(),
                       reinterpret_cast<bf16_t*>(dx.data_ptr()), n, scale);
  } else {
    hipLaunchKernelGGL((dropout_bwd_kernel<float>), grid, block, 0, stream,
                       dy.data_ptr<float>(), mask.data_ptr<unsigned char>(),
                       dx.data_ptr<float>(), n, scale);
  }
  HIP_CHECK_LAST();
  return dx;
}
