// Fused embedding kernels for CDNA4 (gfx950).
//
// BERT/GPT stems do gather(word) + gather(pos) [+ LayerNorm]; stock
// torch runs that as 3-4 HBM round trips over a (B,S,H) tensor. The
// fused forward does ONE pass (two row gathers -> sum -> optional LN ->
// store) and saves only (mean, rstd): the backward RE-GATHERS the summed
// input instead of keeping the (B,S,H) activation resident (HBM3E
// capacity rule — recompute a 2-row gather rather than park 100 MB).
//
// Backward: dword is scatter-add over token ids (fp32 atomics into a
// workspace — random ids rarely collide), dpos is a no-atomic column
// reduction over the batch (every batch row hits the same pos row, so
// atomics there would serialize B-way).
//
// Workload parity: reference BERT/minGPT embedding stems (SURVEY.md
// section 2.3 "Embedding + positional lookup" row).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

// one wave per (b,s) token row; rows_per_block = blockDim.x/64
template <bool LN>
__global__ void emb2_fwd_kernel(const long* __restrict__ ids,
                                const bf16_t* __restrict__ word,
                                const bf16_t* __restrict__ pos,
                                const bf16_t* __restrict__ w,
                                const bf16_t* __restrict__ b,
                                bf16_t* __restrict__ y,
                                float* __restrict__ mean_out,
                                float* __restrict__ rstd_out,
                                int H, int S, long N, long V, float eps) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (row >= N) return;
  long id = ids[row];
  // out-of-range token id: defined behavior (row := 0, stats := identity)
  // instead of OOB reads — torch device-asserts, but an abort would take
  // the whole capture/graph down; a zero row is diagnosable
  if (id < 0 || id >= V) {
    bf16x8 z = {};
    for (int base = lane * 8; base < H; base += WAVE * 8)
      *reinterpret_cast<bf16x8*>(y + row * (long)H + base) = z;
    if (LN && lane == 0) { mean_out[row] = 0.f; rstd_out[row] = 1.f; }
    return;
  }
  const int s = (int)(row % S);
  const bf16_t* wr = word + id * (long)H;
  const bf16_t* pr = pos + (long)s * H;
  bf16_t* yr = y + row * (long)H;

  // up to 2 bf16x8 vectors per lane covers H <= 1024; generic loop above
  float v[2][8];
  int nvec = 0;
  float sum = 0.f, sumsq = 0.f;
  for (int base = lane * 8; base < H && nvec < 2; base += WAVE * 8, ++nvec) {
    bf16x8 a = *reinterpret_cast<const bf16x8*>(wr + base);
    bf16x8 p = *reinterpret_cast<const bf16x8*>(pr + base);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = us2f((unsigned short)a[j]) + us2f((unsigned short)p[j]);
      v[nvec][j] = f;
      sum += f;
      sumsq += f * f;
    }
  }
  if (!LN) {
    int k = 0;
    for (int base = lane * 8; base < H && k < 2; base += WAVE * 8, ++k) {
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o[j] = (short)f2us(v[k][j]);
      *reinterpret_cast<bf16x8*>(yr + base) = o;
    }
    return;
  }
  sum = wave_sum(sum);
  sumsq = wave_sum(sumsq);
  const float mean = sum / H;
  const float var = sumsq / H - mean * mean;
  const float rstd = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  int k = 0;
  for (int base = lane * 8; base < H && k < 2; base += WAVE * 8, ++k) {
    bf16x8 ww = *reinterpret_cast<const bf16x8*>(w + base);
    bf16x8 bb = *reinterpret_cast<const bf16x8*>(b + base);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float nh = (v[k][j] - mean) * rstd;
      o[j] = (short)f2us(nh * us2f((unsigned short)ww[j]) +
                         us2f((unsigned short)bb[j]));
    }
    *reinterpret_cast<bf16x8*>(yr + base) = o;
  }
}

// dword scatter-add: one wave per token row, fp32 atomics
__global__ void emb_scatter_word_kernel(const bf16_t* __restrict__ dx,
                                        const long* __restrict__ ids,
                                        float* __restrict__ dword,
                                        int H, long N, long V) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (row >= N) return;
  const long id = ids[row];
  if (id < 0 || id >= V) return;  // OOB id: no scatter (see fwd note)
  const bf16_t* dr = dx + row * (long)H;
  float* wr = dword + id * (long)H;
  for (int h = lane; h < H; h += WAVE)
    atomicAdd(wr + h, bf2f(dr[h]));
}

// dpos column reduction: block (256) owns a (s, h-chunk) tile, loops B
__global__ void emb_dpos_kernel(const bf16_t* __restrict__ dx,
                                float* __restrict__ dpos,
                                int H, int S, int B) {
  const int s = blockIdx.y;
  const int h = blockIdx.x * blockDim.x + threadIdx.x;
  if (h >= H) return;
  float acc = 0.f;
  const bf16_t* p = dx + (long)s * H + h;
  const long stride = (long)S * H;
  for (int b = 0; b < B; ++b, p += stride) acc += bf2f(*p);
  dpos[(long)s * H + h] = acc;
}

}  // namespace

std::vector<at::Tensor> emb2_ln_fwd(at::Tensor ids, at::Tensor word,
                                    at::Tensor pos, at::Tensor w,
                                    at::Tensor b, double eps) {
  TORCH_CHECK(ids.is_cuda() && ids.scalar_type() == at::kLong &&
              ids.is_contiguous());
  TORCH_CHECK(word.scalar_type() == at::kBFloat16 &&
              pos.scalar_type() == at::kBFloat16 &&
              word.is_contiguous() && pos.is_contiguous());
  const int H = (int)word.size(1);
  TORCH_CHECK(H % 8 == 0 && H <= 1024, "emb2_ln_fwd: H % 8 == 0, H <= 1024");
  const int S = (int)ids.size(-1);
  TORCH_CHECK((long)S <= pos.size(0),
              "sequence longer than the position table");
  const long N = ids.numel();
  auto y = at::empty({ids.size(0), ids.size(1), (long)H}, word.options());
  auto mean = at::empty({N}, word.options().dtype(at::kFloat));
  auto rstd = at::empty({N}, word.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid((unsigned)((N + 3) / 4));
  hipLaunchKernelGGL((emb2_fwd_kernel<true>), grid, block, 0, stream,
                     ids.data_ptr<long>(),
                     reinterpret_cast<const bf16_t*>(word.data_ptr()),
                     reinterpret_cast<const bf16_t*>(pos.data_ptr()),
                     reinterpret_cast<const bf16_t*>(w.data_ptr()),
                     reinterpret_cast<const bf16_t*>(b.data_ptr()),
                     reinterpret_cast<bf16_t*>(y.data_ptr()),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), H, S, N,
                     word.size(0), (float)eps);
  HIP_CHECK_LAST();
  return {y, mean, rstd};
}

at::Tensor emb2_add_fwd(at::Tensor ids, at::Tensor word, at::Tensor pos) {
  TORCH_CHECK(ids.is_cuda() && ids.scalar_type() == at::kLong &&
              ids.is_contiguous());
  TORCH_CHECK(word.scalar_type() == at::kBFloat16 && word.is_contiguous() &&
              pos.is_contiguous());
  const int H = (int)word.size(1);
  TORCH_CHECK(H % 8 == 0 && H <= 1024);
  const int S = (int)ids.size(-1);
  TORCH_CHECK((long)S <= pos.size(0),
              "sequence longer than the position table");
  const long N = ids.numel();
  auto y = at::empty({ids.size(0), ids.size(1), (long)H}, word.options());
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid((unsigned)((N + 3) / 4));
  hipLaunchKernelGGL((emb2_fwd_kernel<false>), grid, block, 0, stream,
                     ids.data_ptr<long>(),
                     reinterpret_cast<const bf16_t*>(word.data_ptr()),
                     reinterpret_cast<const bf16_t*>(pos.data_ptr()),
                     nullptr, nullptr,
                     reinterpret_cast<bf16_t*>(y.data_ptr()), nullptr,
                     nullptr, H, S, N, word.size(0), 0.f);
  HIP_CHECK_LAST();
  return y;
}

std::vector<at::Tensor> emb2_bwd(at::Tensor dx, at::Tensor ids, long V,
                                 long P) {
  TORCH_CHECK(dx.is_cuda() && dx.scalar_type() == at::kBFloat16 &&
              dx.is_contiguous() && ids.is_contiguous());
  const int H = (int)dx.size(-1);
  const int S = (int)ids.size(-1);
  const int B = (int)ids.size(0);
  const long N = ids.numel();
  auto dword = at::zeros({V, (long)H}, dx.options().dtype(at::kFloat));
  auto dpos = at::zeros({P, (long)H}, dx.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  {
    dim3 block(256);
    dim3 grid((unsigned)((N + 3) / 4));
    hipLaunchKernelGGL(emb_scatter_word_kernel, grid, block, 0, stream,
                       reinterpret_cast<const bf16_t*>(dx.data_ptr()),
                       ids.data_ptr<long>(), dword.data_ptr<float>(), H, N,
                       V);
  }
  {
    dim3 block(256);
    dim3 grid((H + 255) / 256, S);
    hipLaunchKernelGGL(emb_dpos_kernel, grid, block, 0, stream,
                       reinterpret_cast<const bf16_t*>(dx.data_ptr()),
                       dpos.data_ptr<float>(), H, S, B);
  }
  HIP_CHECK_LAST();
  return {dword, dpos};
}
