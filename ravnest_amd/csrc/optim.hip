// Fused multi-tensor optimizers (Adam / SGD+momentum / LAMB) for CDNA4.
//
// ONE kernel launch applies the update across every parameter tensor
// (apex-style chunking: host builds a chunk table, each 256-thread block
// owns one chunk) — SURVEY.md section 2.3 "fused multi-tensor apply
// kernels (one launch per bucket), incl. LAMB trust-ratio". fp32 master
// params/grads/states.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <vector>

#include "common.h"

namespace {

constexpr int CHUNK = 1 << 16;

struct Chunk {
  float* p;
  float* g;
  float* m;
  float* v;
  int n;
  int tensor_idx;
};

__global__ void adam_mt_kernel(const Chunk* __restrict__ chunks, float lr,
                               float b1, float b2, float eps, float wd,
                               float bc1, float bc2) {
  const Chunk c = chunks[blockIdx.x];
  for (int i = threadIdx.x; i < c.n; i += blockDim.x) {
    float g = c.g[i];
    float p = c.p[i];
    if (wd != 0.f) g += wd * p;
    float m = c.m[i] = b1 * c.m[i] + (1.f - b1) * g;
    float v = c.v[i] = b2 * c.v[i] + (1.f - b2) * g * g;
    const float denom = sqrtf(v / bc2) + eps;
    c.p[i] = p - lr * (m / bc1) / denom;
  }
}

__global__ void sgd_mt_kernel(const Chunk* __restrict__ chunks, float lr,
                              float momentum, float wd, int nesterov) {
  const Chunk c = chunks[blockIdx.x];
  for (int i = threadIdx.x; i < c.n; i += blockDim.x) {
    float g = c.g[i];
    float p = c.p[i];
    if (wd != 0.f) g += wd * p;
    if (momentum != 0.f) {
      float b = c.m[i] = momentum * c.m[i] + g;
      g = nesterov ? g + momentum * b : b;
    }
    c.p[i] = p - lr * g;
  }
}

// LAMB phase 1: update m/v, accumulate ||w||^2 and ||update||^2 per tensor
__global__ void lamb_phase1_kernel(const Chunk* __restrict__ chunks,
                                   float* __restrict__ norms,  // [T][2]
                                   float b1, float b2, float eps, float wd,
                                   float bc1, float bc2) {
  __shared__ float scratch[16];
  const Chunk c = chunks[blockIdx.x];
  float wsq = 0.f, usq = 0.f;
  for (int i = threadIdx.x; i < c.n; i += blockDim.x) {
    float g = c.g[i];
    float p = c.p[i];
    float m = c.m[i] = b1 * c.m[i] + (1.f - b1) * g;
    float v = c.v[i] = b2 * c.v[i] + (1.f - b2) * g * g;
    float upd = (m / bc1) / (sqrtf(v / bc2) + eps);
    if (wd != 0.f) upd += wd * p;
    wsq += p * p;
    usq += upd * upd;
  }
  wsq = block_sum(wsq, scratch);
  usq = block_sum(usq, scratch);
  if (threadIdx.x == 0) {
    atomicAdd(&norms[c.tensor_idx * 2 + 0], wsq);
    atomicAdd(&norms[c.tensor_idx * 2 + 1], usq);
  }
}

__global__ void lamb_phase2_kernel(const Chunk* __restrict__ chunks,
                                   const float* __restrict__ norms, float lr,
                                   float b1, float b2, float eps, float wd,
                                   float bc1, float bc2, float clamp_trust) {
  const Chunk c = chunks[blockIdx.x];
  const float wn = sqrtf(norms[c.tensor_idx * 2 + 0]);
  const float un = sqrtf(norms[c.tensor_idx * 2 + 1]);
  float trust = 1.f;
  if (wn > 0.f && un > 0.f) trust = fminf(wn / un, clamp_trust);
  const float step = lr * trust;
  for (int i = threadIdx.x; i < c.n; i += blockDim.x) {
    float m = c.m[i], v = c.v[i];
    float upd = (m / bc1) / (sqrtf(v / bc2) + eps);
    if (wd != 0.f) upd += wd * c.p[i];
    c.p[i] -= step * upd;
  }
}

// build the device-side chunk table
at::Tensor build_chunks(const std::vector<at::Tensor>& ps,
                        const std::vector<at::Tensor>& gs,
                        const std::vector<at::Tensor>& ms,
                        const std::vector<at::Tensor>& vs, int& nchunks) {
  std::vector<Chunk> chunks;
  for (size_t t = 0; t < ps.size(); ++t) {
    TORCH_CHECK(ps[t].scalar_type() == at::kFloat,
                "fused optimizers expect fp32 master params");
    TORCH_CHECK(ps[t].is_contiguous() && gs[t].is_contiguous());
    long n = ps[t].numel();
    float* p = ps[t].data_ptr<float>();
    float* g = gs[t].data_ptr<float>();
    float* m = ms.empty() ? nullptr : ms[t].data_ptr<float>();
    float* v = vs.empty() ? nullptr : vs[t].data_ptr<float>();
    for (long off = 0; off < n; off += CHUNK) {
      Chunk c;
      c.p = p + off;
      c.g = g + off;
      c.m = m ? m + off : nullptr;
      c.v = v ? v + off : nullptr;
      c.n = (int)std::min<long>(CHUNK, n - off);
      c.tensor_idx = (int)t;
      chunks.push_back(c);
    }
  }
  nchunks = (int)chunks.size();
  auto host = at::from_blob(chunks.data(),
                            {(long)(chunks.size() * sizeof(Chunk))},
                            at::TensorOptions().dtype(at::kByte));
  return host.to(ps[0].device());  // blocking copy: host buffer dies here
}

}  // namespace

void fused_adam(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                double lr, double b1, double b2, double eps, double wd,
                double bc1, double bc2) {
  if (ps.empty()) return;
  int nchunks = 0;
  auto dev_chunks = build_chunks(ps, gs, ms, vs, nchunks);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(adam_mt_kernel, dim3(nchunks), dim3(256), 0, stream,
                     reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()),
                     (float)lr, (float)b1, (float)b2, (float)eps, (float)wd,
                     (float)bc1, (float)bc2);
  HIP_CHECK_LAST();
}

void fused_sgd(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
               std::vector<at::Tensor> bufs, double lr, double momentum,
               double wd, bool nesterov) {
  if (ps.empty()) return;
  int nchunks = 0;
  auto dev_chunks = build_chunks(ps, gs, bufs, {}, nchunks);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(sgd_mt_kernel, dim3(nchunks), dim3(256), 0, stream,
                     reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()),
                     (float)lr, (float)momentum, (float)wd,
                     nesterov ? 1 : 0);
  HIP_CHECK_LAST();
}

void fused_lamb(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                double lr, double b1, double b2, double eps, double wd,
                double bc1, double bc2, double clamp_trust) {
  if (ps.empty()) return;
  int nchunks = 0;
  auto dev_chunks = build_chunks(ps, gs, ms, vs, nchunks);
  auto norms = at::zeros({(long)ps.size(), 2},
                         ps[0].options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(lamb_phase1_kernel, dim3(nchunks), dim3(256), 0, stream,
                     reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()),
                     norms.data_ptr<float>(), (float)b1, (float)b2,
                     (float)eps, (float)wd, (float)bc1, (float)bc2);
  hipLaunchKernelGGL(lamb_phase2_kernel, dim3(nchunks), dim3(256), 0, stream,
                     reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()),
                     norms.data_ptr<float>(), (float)lr, (float)b1, (float)b2,
                     (float)eps, (float)wd, (float)bc1, (float)bc2,
                     (float)clamp_trust);
  HIP_CHECK_LAST();
}
