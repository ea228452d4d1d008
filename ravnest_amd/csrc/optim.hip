// Fused multi-tensor optimizers (Adam / SGD+momentum / LAMB) for CDNA4.
//
// ONE kernel launch applies the update across every parameter tensor
// (apex-style chunking: host builds a chunk table, each 256-thread block
// owns one chunk) — SURVEY.md section 2.3 "fused multi-tensor apply
// kernels (one launch per bucket), incl. LAMB trust-ratio".
//
// Two parameter modes:
//  * fp32 params + fp32 grads (CPU-parity training)
//  * bf16 params + bf16 grads with an fp32 MASTER copy in optimizer
//    state (the MI355X-native mode: bf16 weights halve HBM traffic for
//    every GEMM; the master preserves convergence).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <vector>

#include "common.h"

namespace {

constexpr int CHUNK = 1 << 16;

struct Chunk {
  void* p;        // param (fp32 or bf16)
  void* g;        // grad  (same dtype as p)
  float* m;
  float* v;
  float* master;  // fp32 master (bf16 mode) or nullptr
  int n;
  int tensor_idx;
};

template <typename T>
DEVINL float ldv(const void* p, int i) {
  if constexpr (sizeof(T) == 2)
    return bf2f(reinterpret_cast<const bf16_t*>(p)[i]);
  else
    return reinterpret_cast<const float*>(p)[i];
}

template <typename T>
DEVINL void stv(void* p, int i, float v) {
  if constexpr (sizeof(T) == 2)
    reinterpret_cast<bf16_t*>(p)[i] = f2bf(v);
  else
    reinterpret_cast<float*>(p)[i] = v;
}

template <typename T>
DEVINL void adam_one(float& p, float g, float& m, float& v, float lr,
                     float b1, float b2, float eps, float wd, float bc1,
                     float bc2) {
  if (wd != 0.f) g += wd * p;
  m = b1 * m + (1.f - b1) * g;
  v = b2 * v + (1.f - b2) * g * g;
  p -= lr * (m / bc1) / (sqrtf(v / bc2) + eps);
}

// vectorized: 4 elements/thread/iter with 16-byte fp32 state accesses
// (m/v/master are the traffic; chunks are 64Ki elements so the body is
// always aligned and the tail only exists in the last chunk)
template <typename T>
__global__ void adam_mt_kernel(const Chunk* __restrict__ chunks, float lr,
                               float b1, float b2, float eps, float wd,
                               float bc1, float bc2,
                               const float* __restrict__ lr_buf,
                               const long long* __restrict__ step_buf) {
  // hipGraph-capturable mode: lr and the bias-correction step come from
  // DEVICE buffers so replays see live values (host scalars would
  // freeze at capture — engine/graphstep.py)
  if (lr_buf) {
    lr = *lr_buf;
    const float st = (float)*step_buf;
    bc1 = 1.f - __powf(b1, st);
    bc2 = 1.f - __powf(b2, st);
  }
  const Chunk c = chunks[blockIdx.x];
  const int n4 = c.n >> 2;
  float4* m4 = reinterpret_cast<float4*>(c.m);
  float4* v4 = reinterpret_cast<float4*>(c.v);
  float4* w4 = reinterpret_cast<float4*>(c.master);
  for (int i = threadIdx.x; i < n4; i += blockDim.x) {
    float4 m = m4[i], v = v4[i];
    float4 p;
    float g[4];
    if constexpr (sizeof(T) == 2) {
      unsigned long long graw =
          reinterpret_cast<const unsigned long long*>(c.g)[i];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        g[j] = us2f((unsigned short)(graw >> (16 * j)));
      p = w4[i];
    } else {
      const float4 gv = reinterpret_cast<const float4*>(c.g)[i];
      g[0] = gv.x; g[1] = gv.y; g[2] = gv.z; g[3] = gv.w;
      p = reinterpret_cast<const float4*>(c.p)[i];
    }
    adam_one<T>(p.x, g[0], m.x, v.x, lr, b1, b2, eps, wd, bc1, bc2);
    adam_one<T>(p.y, g[1], m.y, v.y, lr, b1, b2, eps, wd, bc1, bc2);
    adam_one<T>(p.z, g[2], m.z, v.z, lr, b1, b2, eps, wd, bc1, bc2);
    adam_one<T>(p.w, g[3], m.w, v.w, lr, b1, b2, eps, wd, bc1, bc2);
    m4[i] = m;
    v4[i] = v;
    if constexpr (sizeof(T) == 2) {
      w4[i] = p;
      unsigned long long praw =
          ((unsigned long long)f2us(p.x)) |
          ((unsigned long long)f2us(p.y) << 16) |
          ((unsigned long long)f2us(p.z) << 32) |
          ((unsigned long long)f2us(p.w) << 48);
      reinterpret_cast<unsigned long long*>(c.p)[i] = praw;
    } else {
      reinterpret_cast<float4*>(c.p)[i] = p;
    }
  }
  // scalar tail
  for (int i = (n4 << 2) + threadIdx.x; i < c.n; i += blockDim.x) {
    float g = ldv<T>(c.g, i);
    float p = c.master ? c.master[i] : ldv<T>(c.p, i);
    float m = c.m[i], v = c.v[i];
    adam_one<T>(p, g, m, v, lr, b1, b2, eps, wd, bc1, bc2);
    c.m[i] = m;
    c.v[i] = v;
    if (c.master) c.master[i] = p;
    stv<T>(c.p, i, p);
  }
}

template <typename T>
__global__ void sgd_mt_kernel(const Chunk* __restrict__ chunks, float lr,
                              float momentum, float wd, int nesterov) {
  const Chunk c = chunks[blockIdx.x];
  for (int i = threadIdx.x; i < c.n; i += blockDim.x) {
    float g = ldv<T>(c.g, i);
    float p = c.master ? c.master[i] : ldv<T>(c.p, i);
    if (wd != 0.f) g += wd * p;
    if (momentum != 0.f) {
      float b = c.m[i] = momentum * c.m[i] + g;
      g = nesterov ? g + momentum * b : b;
    }
    p -= lr * g;
    if (c.master) c.master[i] = p;
    stv<T>(c.p, i, p);
  }
}

// LAMB phase 1: update m/v, accumulate ||w||^2 and ||update||^2 per tensor
template <typename T>
__global__ void lamb_phase1_kernel(const Chunk* __restrict__ chunks,
                                   float* __restrict__ norms,  // [T][2]
                                   float b1, float b2, float eps, float wd,
                                   float bc1, float bc2) {
  __shared__ float scratch[16];
  const Chunk c = chunks[blockIdx.x];
  float wsq = 0.f, usq = 0.f;
  for (int i = threadIdx.x; i < c.n; i += blockDim.x) {
    float g = ldv<T>(c.g, i);
    float p = c.master ? c.master[i] : ldv<T>(c.p, i);
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

template <typename T>
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
    float p = c.master ? c.master[i] : ldv<T>(c.p, i);
    float upd = (m / bc1) / (sqrtf(v / bc2) + eps);
    if (wd != 0.f) upd += wd * p;
    p -= step * upd;
    if (c.master) c.master[i] = p;
    stv<T>(c.p, i, p);
  }
}

// build the device-side chunk table; returns element dtype size (2|4)
at::Tensor build_chunks(const std::vector<at::Tensor>& ps,
                        const std::vector<at::Tensor>& gs,
                        const std::vector<at::Tensor>& ms,
                        const std::vector<at::Tensor>& vs,
                        const std::vector<at::Tensor>& masters, int& nchunks,
                        int& esize) {
  std::vector<Chunk> chunks;
  esize = ps[0].scalar_type() == at::kBFloat16 ? 2 : 4;
  for (size_t t = 0; t < ps.size(); ++t) {
    // any DENSE layout is fine (channels_last conv weights included):
    // the update is elementwise over raw storage, so p/g/state only need
    // to share one layout
    const bool dense =
        ps[t].is_contiguous() ||
        ps[t].is_contiguous(at::MemoryFormat::ChannelsLast) ||
        ps[t].is_contiguous(at::MemoryFormat::ChannelsLast3d);
    TORCH_CHECK(dense, "fused optimizer: param must be dense");
    TORCH_CHECK(gs[t].strides() == ps[t].strides(),
                "grad layout must match param layout");
    TORCH_CHECK(gs[t].scalar_type() == ps[t].scalar_type(),
                "grad dtype must match param dtype");
    long n = ps[t].numel();
    void* p = ps[t].data_ptr();
    void* g = gs[t].data_ptr();
    float* m = (ms.empty() || ms[t].numel() == 0)
                   ? nullptr : ms[t].data_ptr<float>();
    float* v = (vs.empty() || vs[t].numel() == 0)
                   ? nullptr : vs[t].data_ptr<float>();
    float* master = masters.empty() ? nullptr
                                    : masters[t].data_ptr<float>();
    const int es = ps[t].scalar_type() == at::kBFloat16 ? 2 : 4;
    TORCH_CHECK(es == esize, "mixed param dtypes in one group");
    TORCH_CHECK(es == 4 || master != nullptr,
                "bf16 params require fp32 masters");
    for (long off = 0; off < n; off += CHUNK) {
      Chunk c;
      c.p = (char*)p + off * es;
      c.g = (char*)g + off * es;
      c.m = m ? m + off : nullptr;
      c.v = v ? v + off : nullptr;
      c.master = master ? master + off : nullptr;
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

// multi-tensor device copy: ONE launch clones every parameter into its
// snapshot slot (the versioning engine's per-step param snapshot was ~200
// separate hipMemcpy calls — SURVEY.md section 2.3 "multi-tensor
// clone/copy kernels ... versioned weight arena").
__global__ void copy_mt_kernel(const Chunk* __restrict__ chunks) {
  const Chunk c = chunks[blockIdx.x];
  // p = src, m = dst (fp32 elems) OR 2-byte mode via tensor_idx flag
  if (c.tensor_idx == 2) {  // 2-byte elements
    const int n4 = c.n >> 3;  // 16B groups of 8 bf16
    const ulong2* src = reinterpret_cast<const ulong2*>(c.p);
    ulong2* dst = reinterpret_cast<ulong2*>(c.g);
    for (int i = threadIdx.x; i < n4; i += blockDim.x) dst[i] = src[i];
    for (int i = (n4 << 3) + threadIdx.x; i < c.n; i += blockDim.x)
      reinterpret_cast<bf16_t*>(c.g)[i] =
          reinterpret_cast<const bf16_t*>(c.p)[i];
  } else {
    const int n4 = c.n >> 2;
    const float4* src = reinterpret_cast<const float4*>(c.p);
    float4* dst = reinterpret_cast<float4*>(c.g);
    for (int i = threadIdx.x; i < n4; i += blockDim.x) dst[i] = src[i];
    for (int i = (n4 << 2) + threadIdx.x; i < c.n; i += blockDim.x)
      reinterpret_cast<float*>(c.g)[i] =
          reinterpret_cast<const float*>(c.p)[i];
  }
}

}  // namespace

void fused_adam(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                std::vector<at::Tensor> masters, double lr, double b1,
                double b2, double eps, double wd, double bc1, double bc2) {
  if (ps.empty()) return;
  int nchunks = 0, esize = 4;
  auto dev_chunks = build_chunks(ps, gs, ms, vs, masters, nchunks, esize);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  if (esize == 2)
    hipLaunchKernelGGL((adam_mt_kernel<bf16_t>), dim3(nchunks), dim3(256), 0,
                       stream,
                       reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()),
                       (float)lr, (float)b1, (float)b2, (float)eps, (float)wd,
                       (float)bc1, (float)bc2, nullptr, nullptr);
  else
    hipLaunchKernelGGL((adam_mt_kernel<float>), dim3(nchunks), dim3(256), 0,
                       stream,
                       reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()),
                       (float)lr, (float)b1, (float)b2, (float)eps, (float)wd,
                       (float)bc1, (float)bc2, nullptr, nullptr);
  HIP_CHECK_LAST();
}

// graph-capturable Adam: the chunk table is prebuilt ONCE (pointers are
// stable inside a captured step: params/state outside the graph pool,
// grads at fixed pool addresses) and lr/step live in device buffers.
std::tuple<at::Tensor, long, long> adam_build_table(
    std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
    std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
    std::vector<at::Tensor> masters) {
  TORCH_CHECK(!ps.empty());
  int nchunks = 0, esize = 4;
  auto table = build_chunks(ps, gs, ms, vs, masters, nchunks, esize);
  return {table, (long)nchunks, (long)esize};
}

void fused_adam_graph(at::Tensor table, long nchunks, long esize,
                      at::Tensor lr_buf, double b1, double b2, double eps,
                      double wd, at::Tensor step_buf) {
  TORCH_CHECK(table.is_cuda() && lr_buf.is_cuda() && step_buf.is_cuda());
  TORCH_CHECK(lr_buf.scalar_type() == at::kFloat &&
              step_buf.scalar_type() == at::kLong);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  if (esize == 2)
    hipLaunchKernelGGL((adam_mt_kernel<bf16_t>), dim3((int)nchunks),
                       dim3(256), 0, stream,
                       reinterpret_cast<const Chunk*>(table.data_ptr()), 0.f,
                       (float)b1, (float)b2, (float)eps, (float)wd, 1.f, 1.f,
                       lr_buf.data_ptr<float>(),
                       reinterpret_cast<const long long*>(
                           step_buf.data_ptr<int64_t>()));
  else
    hipLaunchKernelGGL((adam_mt_kernel<float>), dim3((int)nchunks),
                       dim3(256), 0, stream,
                       reinterpret_cast<const Chunk*>(table.data_ptr()), 0.f,
                       (float)b1, (float)b2, (float)eps, (float)wd, 1.f, 1.f,
                       lr_buf.data_ptr<float>(),
                       reinterpret_cast<const long long*>(
                           step_buf.data_ptr<int64_t>()));
  HIP_CHECK_LAST();
}

void fused_sgd(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
               std::vector<at::Tensor> bufs, std::vector<at::Tensor> masters,
               double lr, double momentum, double wd, bool nesterov) {
  if (ps.empty()) return;
  int nchunks = 0, esize = 4;
  auto dev_chunks = build_chunks(ps, gs, bufs, {}, masters, nchunks, esize);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  if (esize == 2)
    hipLaunchKernelGGL((sgd_mt_kernel<bf16_t>), dim3(nchunks), dim3(256), 0,
                       stream,
                       reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()),
                       (float)lr, (float)momentum, (float)wd, nesterov);
  else
    hipLaunchKernelGGL((sgd_mt_kernel<float>), dim3(nchunks), dim3(256), 0,
                       stream,
                       reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()),
                       (float)lr, (float)momentum, (float)wd, nesterov);
  HIP_CHECK_LAST();
}

void fused_lamb(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                std::vector<at::Tensor> masters, double lr, double b1,
                double b2, double eps, double wd, double bc1, double bc2,
                double clamp_trust) {
  if (ps.empty()) return;
  int nchunks = 0, esize = 4;
  auto dev_chunks = build_chunks(ps, gs, ms, vs, masters, nchunks, esize);
  auto norms = at::zeros({(long)ps.size(), 2},
                         ps[0].options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
#define LAMB_LAUNCH(T)                                                       \
  do {                                                                       \
    hipLaunchKernelGGL((lamb_phase1_kernel<T>), dim3(nchunks), dim3(256), 0, \
                       stream,                                               \
                       reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()),\
                       norms.data_ptr<float>(), (float)b1, (float)b2,        \
                       (float)eps, (float)wd, (float)bc1, (float)bc2);       \
    hipLaunchKernelGGL((lamb_phase2_kernel<T>), dim3(nchunks), dim3(256), 0, \
                       stream,                                               \
                       reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()),\
                       norms.data_ptr<float>(), (float)lr, (float)b1,        \
                       (float)b2, (float)eps, (float)wd, (float)bc1,         \
                       (float)bc2, (float)clamp_trust);                      \
  } while (0)
  if (esize == 2)
    LAMB_LAUNCH(bf16_t);
  else
    LAMB_LAUNCH(float);
  HIP_CHECK_LAST();
}


void fused_copy(std::vector<at::Tensor> srcs, std::vector<at::Tensor> dsts) {
  if (srcs.empty()) return;
  std::vector<Chunk> chunks;
  for (size_t t = 0; t < srcs.size(); ++t) {
    TORCH_CHECK(srcs[t].strides() == dsts[t].strides(),
                "fused_copy: layouts must match");
    TORCH_CHECK(srcs[t].scalar_type() == dsts[t].scalar_type());
    const int es = srcs[t].element_size();
    TORCH_CHECK(es == 2 || es == 4, "fused_copy: 2- or 4-byte elems");
    long n = srcs[t].numel();
    char* sp = (char*)srcs[t].data_ptr();
    char* dp = (char*)dsts[t].data_ptr();
    for (long off = 0; off < n; off += CHUNK) {
      Chunk c{};
      c.p = sp + off * es;
      c.g = dp + off * es;
      c.n = (int)std::min<long>(CHUNK, n - off);
      c.tensor_idx = es;  // element size flag
      chunks.push_back(c);
    }
  }
  auto host = at::from_blob(chunks.data(),
                            {(long)(chunks.size() * sizeof(Chunk))},
                            at::TensorOptions().dtype(at::kByte));
  auto dev_chunks = host.to(srcs[0].device());
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(copy_mt_kernel, dim3((int)chunks.size()), dim3(256), 0,
                     stream,
                     reinterpret_cast<const Chunk*>(dev_chunks.data_ptr()));
  HIP_CHECK_LAST();
}
