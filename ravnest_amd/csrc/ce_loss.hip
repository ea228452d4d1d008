// Fused softmax cross-entropy over large vocabularies (BERT V=30522).
//
// fwd: one 256-thread block per row chunk computes max and log-sum-exp in
// one pass each (block reduction), accumulates the summed loss with one
// atomic per row. bwd: dlogits = (softmax - onehot) * gscale, elementwise
// from (logits, lse). SURVEY.md section 2.3 loss row (split-K style row
// passes; vectorized bf16 loads).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <typename T>
DEVINL float ld(const T* p);
template <>
DEVINL float ld<bf16_t>(const bf16_t* p) { return bf2f(*p); }
template <>
DEVINL float ld<float>(const float* p) { return *p; }

// one block per row, ONE pass over V: per-thread online (max, sum)
// state merged wave- then block-wide ((m,s) pairs combine as
// s = s1*exp(m1-m) + s2*exp(m2-m)). Halves the V reads of the naive
// two-pass version (the 30522-vocab read is the whole cost).
DEVINL void ms_merge(float& m, float& s, float m2, float s2) {
  const float mn = fmaxf(m, m2);
  if (mn == -INFINITY) { m = mn; return; }
  s = s * __expf(m - mn) + s2 * __expf(m2 - mn);
  m = mn;
}

template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ lse_out,
                              float* __restrict__ loss_accum,
                              int* __restrict__ valid_accum, int V, long N,
                              long ignore_index) {
  __shared__ float scratch_m[16];
  __shared__ float scratch_s[16];
  const long row = blockIdx.x;
  if (row >= N) return;
  const T* lr = logits + row * V;
  const long tgt = targets[row];

  float m = -INFINITY, s = 0.f;
  if constexpr (sizeof(T) == 2) {
    // vectorized: 8 bf16 per thread per iteration (16B loads)
    const int nv = V >> 3;
    for (int g = threadIdx.x; g < nv; g += blockDim.x) {
      const bf16x8 v8 = *reinterpret_cast<const bf16x8*>(lr + g * 8);
      float vals[8], vmax = -INFINITY;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        vals[j] = us2f((unsigned short)v8[j]);
        vmax = fmaxf(vmax, vals[j]);
      }
      float ls = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) ls += __expf(vals[j] - vmax);
      if (vmax > m) {
        s = s * __expf(m - vmax) + ls;
        m = vmax;
      } else {
        s += ls * __expf(vmax - m);
      }
    }
    for (int i = (nv << 3) + threadIdx.x; i < V; i += blockDim.x) {
      const float v = ld(lr + i);
      if (v > m) {
        s = s * __expf(m - v) + 1.f;
        m = v;
      } else {
        s += __expf(v - m);
      }
    }
  } else {
    for (int i = threadIdx.x; i < V; i += blockDim.x) {
      const float v = ld(lr + i);
      if (v > m) {
        s = s * __expf(m - v) + 1.f;
        m = v;
      } else {
        s += __expf(v - m);
      }
    }
  }
  // wave merge
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float m2 = __shfl_down(m, off, WAVE);
    float s2 = __shfl_down(s, off, WAVE);
    ms_merge(m, s, m2, s2);
  }
  // block merge via LDS
  {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int nw = (blockDim.x + WAVE - 1) / WAVE;
    if (lane == 0) {
      scratch_m[wid] = m;
      scratch_s[wid] = s;
    }
    __syncthreads();
    if (wid == 0) {
      m = (threadIdx.x < nw) ? scratch_m[threadIdx.x] : -INFINITY;
      s = (threadIdx.x < nw) ? scratch_s[threadIdx.x] : 0.f;
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) {
        float m2 = __shfl_down(m, off, WAVE);
        float s2 = __shfl_down(s, off, WAVE);
        ms_merge(m, s, m2, s2);
      }
      if (threadIdx.x == 0) {
        scratch_m[0] = m;
        scratch_s[0] = s;
      }
    }
    __syncthreads();
    m = scratch_m[0];
    s = scratch_s[0];
  }
  const float lse = m + __logf(s);

  if (threadIdx.x == 0) {
    lse_out[row] = lse;
    if (tgt != ignore_index) {
      atomicAdd(loss_accum, lse - ld(lr + tgt));
      atomicAdd(valid_accum, 1);
    }
  }
}

template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const long* __restrict__ targets,
                              const float* __restrict__ lse,
                              const float* __restrict__ gscale,
                              T* __restrict__ dlogits, int V, long N,
                              long ignore_index) {
  const long row = blockIdx.x;
  if (row >= N) return;
  const T* lr = logits + row * V;
  T* dr = dlogits + row * V;
  const long tgt = targets[row];
  const float g = gscale[0];
  const float l = lse[row];
  if (tgt == ignore_index) {
    for (int i = threadIdx.x; i < V; i += blockDim.x) {
      if constexpr (sizeof(T) == 2)
        dr[i] = f2bf(0.f);
      else
        dr[i] = 0.f;
    }
    return;
  }
  if constexpr (sizeof(T) == 2) {
    const int nv = V >> 3;
    for (int gi = threadIdx.x; gi < nv; gi += blockDim.x) {
      const bf16x8 v8 = *reinterpret_cast<const bf16x8*>(lr + gi * 8);
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int i = gi * 8 + j;
        float p = __expf(us2f((unsigned short)v8[j]) - l);
        o[j] = (short)f2us((p - (i == (int)tgt ? 1.f : 0.f)) * g);
      }
      *reinterpret_cast<bf16x8*>(dr + gi * 8) = o;
    }
    for (int i = (nv << 3) + threadIdx.x; i < V; i += blockDim.x) {
      float p = __expf(ld(lr + i) - l);
      dr[i] = f2bf((p - (i == (int)tgt ? 1.f : 0.f)) * g);
    }
  } else {
    for (int i = threadIdx.x; i < V; i += blockDim.x) {
      float p = __expf(ld(lr + i) - l);
      dr[i] = (p - (i == (int)tgt ? 1.f : 0.f)) * g;
    }
  }
}

}  // namespace

std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor targets,
                               int64_t ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous());
  TORCH_CHECK(logits.dim() == 2);
  TORCH_CHECK(targets.scalar_type() == at::kLong);
  const long N = logits.size(0);
  const int V = logits.size(1);
  auto lse = at::empty({N}, logits.options().dtype(at::kFloat));
  auto loss = at::zeros({1}, logits.options().dtype(at::kFloat));
  auto nvalid = at::zeros({1}, logits.options().dtype(at::kInt));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid(N);
  if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((ce_fwd_kernel<bf16_t>), grid, block, 0, stream,
                       reinterpret_cast<const bf16_t*>(logits.data_ptr()),
                       targets.data_ptr<long>(), lse.data_ptr<float>(),
                       loss.data_ptr<float>(), nvalid.data_ptr<int>(), V, N,
                       ignore_index);
  } else {
    hipLaunchKernelGGL((ce_fwd_kernel<float>), grid, block, 0, stream,
                       logits.data_ptr<float>(), targets.data_ptr<long>(),
                       lse.data_ptr<float>(), loss.data_ptr<float>(),
                       nvalid.data_ptr<int>(), V, N, ignore_index);
  }
  HIP_CHECK_LAST();
  return {loss.squeeze(0), lse, nvalid.squeeze(0)};
}

at::Tensor ce_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse,
                  at::Tensor gscale, int64_t ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous());
  const long N = logits.size(0);
  const int V = logits.size(1);
  auto dlogits = at::empty_like(logits);
  auto gs = gscale.to(at::kFloat).reshape({1}).contiguous();
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid(N);
  if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((ce_bwd_kernel<bf16_t>), grid, block, 0, stream,
                       reinterpret_cast<const bf16_t*>(logits.data_ptr()),
                       targets.data_ptr<long>(), lse.data_ptr<float>(),
                       gs.data_ptr<float>(),
                       reinterpret_cast<bf16_t*>(dlogits.data_ptr()), V, N,
                       ignore_index);
  } else {
    hipLaunchKernelGGL((ce_bwd_kernel<float>), grid, block, 0, stream,
                       logits.data_ptr<float>(), targets.data_ptr<long>(),
                       lse.data_ptr<float>(), gs.data_ptr<float>(),
                       dlogits.data_ptr<float>(), V, N, ignore_index);
  }
  HIP_CHECK_LAST();
  return dlogits;
}
