// Implicit-GEMM spatial convolution on MFMA (gfx950), bf16 NCHW.
//
// SURVEY.md section 2.3 conv row ("implicit-GEMM fwd/dgrad/wgrad MFMA
// kernels"). v1 structure: one wave per 32x32 output tile, the
// contiguous operand (weights / dy) read as direct bf16x8 fragments and
// the im2col operand gathered element-wise with out-of-bounds zeros —
// correctness-first; the LDS-staged ladder is the round-2 upgrade. The
// default ResNet path stays fp32-MIOpen (measured faster at fp32); this
// kernel targets the bf16 conv gap (MIOpen bf16 measured ~4x slower
// than fp32 on this stack — BASELINE.md).
//
// MFMA 32x32x16 bf16 fragment algebra: see attention.hip header.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8c __attribute__((ext_vector_type(8)));
typedef float f32x16c __attribute__((ext_vector_type(16)));

namespace {

struct ConvGeom {
  int N, Ci, H, W;     // input
  int Co, R, S;        // weights
  int Ho, Wo;          // output
  int st, pad;
};

// ---- forward: out[n][co][ho][wo] = sum_{ci,r,s} x * w -------------
// GEMM view: i = co (A, contiguous k), j = flat (n,ho,wo) position,
// k = ci*R*S + r*S + s (B gathered from x with OOB zeros).
__global__ __launch_bounds__(256) void conv_fwd_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
    const float* __restrict__ bias, bf16_t* __restrict__ y, ConvGeom g,
    int relu) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int hi = lane >> 5;
  const int j32 = lane & 31;
  const long P = (long)g.N * g.Ho * g.Wo;       // output positions
  const int K = g.Ci * g.R * g.S;
  const int co_tiles = (g.Co + 31) / 32;
  const long tile = (long)blockIdx.x * (blockDim.x / WAVE) +
                    threadIdx.x / WAVE;
  const int co0 = (int)(tile % co_tiles) * 32;
  const long p0 = (tile / co_tiles) * 32;
  if (p0 >= P) return;

  // this lane's B position (column j32)
  const long p = p0 + j32;
  const bool live = p < P;
  const int wo = (int)(p % g.Wo);
  const int ho = (int)((p / g.Wo) % g.Ho);
  const int n = (int)(p / ((long)g.Wo * g.Ho));
  const int h0 = ho * g.st - g.pad, w0 = wo * g.st - g.pad;
  const bf16_t* xn = x + (long)n * g.Ci * g.H * g.W;
  // this lane's A row (weights row co0 + j32, contiguous k)
  const int co_a = min(co0 + j32, g.Co - 1);
  const bf16_t* wrow = w + (long)co_a * K;

  f32x16c acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;

  for (int kk = 0; kk < K; kk += 16) {
    bf16x8c af, bf;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int k = kk + hi * 8 + e;
      af[e] = (k < K) ? wrow[k] : (bf16_t)f2bf(0.f);
      float bv = 0.f;
      if (live && k < K) {
        const int ci = k / (g.R * g.S);
        const int rs = k % (g.R * g.S);
        const int hh = h0 + rs / g.S;
        const int ww = w0 + rs % g.S;
        if (hh >= 0 && hh < g.H && ww >= 0 && ww < g.W)
          bv = bf2f(xn[((long)ci * g.H + hh) * g.W + ww]);
      }
      bf[e] = f2bf(bv);
    }
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
  }
  // D[row = co pattern][col = position j32]
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int co = co0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    const long pc = p0 + j32;
    if (co < g.Co && pc < P) {
      float v = acc[r] + (bias ? bias[co] : 0.f);
      if (relu) v = fmaxf(v, 0.f);
      const int wo2 = (int)(pc % g.Wo);
      const int ho2 = (int)((pc / g.Wo) % g.Ho);
      const int n2 = (int)(pc / ((long)g.Wo * g.Ho));
      y[(((long)n2 * g.Co + co) * g.Ho + ho2) * g.Wo + wo2] = f2bf(v);
    }
  }
}

// ---- dgrad: dx[n][ci][h][w] = sum_{co,r,s | aligned} dy * w --------
// i = ci, j = flat (n,h,w), k = co*R*S + r*S + s; both operands
// gathered (w with channel-major stride, dy with stride-alignment
// zeros).
__global__ __launch_bounds__(256) void conv_dgrad_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ w,
    bf16_t* __restrict__ dx, ConvGeom g) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int hi = lane >> 5;
  const int j32 = lane & 31;
  const long P = (long)g.N * g.H * g.W;  // input positions
  const int K = g.Co * g.R * g.S;
  const int ci_tiles = (g.Ci + 31) / 32;
  const long tile = (long)blockIdx.x * (blockDim.x / WAVE) +
                    threadIdx.x / WAVE;
  const int ci0 = (int)(tile % ci_tiles) * 32;
  const long p0 = (tile / ci_tiles) * 32;
  if (p0 >= P) return;

  const long p = p0 + j32;
  const bool live = p < P;
  const int ww = (int)(p % g.W);
  const int hh = (int)((p / g.W) % g.H);
  const int n = (int)(p / ((long)g.W * g.H));
  const bf16_t* dyn = dy + (long)n * g.Co * g.Ho * g.Wo;
  const int ci_a = min(ci0 + j32, g.Ci - 1);

  f32x16c acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;

  for (int kk = 0; kk < K; kk += 16) {
    bf16x8c af, bf;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int k = kk + hi * 8 + e;
      float av = 0.f, bv = 0.f;
      if (k < K) {
        const int co = k / (g.R * g.S);
        const int rs = k % (g.R * g.S);
        const int r2 = rs / g.S, s2 = rs % g.S;
        av = bf2f(w[(((long)co * g.Ci + ci_a) * g.R + r2) * g.S + s2]);
        if (live) {
          const int hnum = hh + g.pad - r2;
          const int wnum = ww + g.pad - s2;
          if (hnum >= 0 && wnum >= 0 && hnum % g.st == 0 &&
              wnum % g.st == 0) {
            const int ho = hnum / g.st, wo = wnum / g.st;
            if (ho < g.Ho && wo < g.Wo)
              bv = bf2f(dyn[((long)co * g.Ho + ho) * g.Wo + wo]);
          }
        }
      }
      af[e] = f2bf(av);
      bf[e] = f2bf(bv);
    }
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int ci = ci0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    const long pc = p0 + j32;
    if (ci < g.Ci && pc < P) {
      const int ww2 = (int)(pc % g.W);
      const int hh2 = (int)((pc / g.W) % g.H);
      const int n2 = (int)(pc / ((long)g.W * g.H));
      dx[(((long)n2 * g.Ci + ci) * g.H + hh2) * g.W + ww2] =
          f2bf(acc[r]);
    }
  }
}

// ---- wgrad: dw[co][ci][r][s] = sum_{n,ho,wo} dy * x ----------------
// i = co (A = dy, contiguous over (ho,wo)), j = ci*R*S + r*S + s,
// k = flat (ho,wo) within ONE image; grid.z = n with fp32 atomic
// accumulation (fills the chip for the long-K skinny shape).
__global__ __launch_bounds__(256) void conv_wgrad_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ x,
    float* __restrict__ dw, ConvGeom g) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int hi = lane >> 5;
  const int j32 = lane & 31;
  const int KRS = g.Ci * g.R * g.S;
  const int co_tiles = (g.Co + 31) / 32;
  const int j_tiles = (KRS + 31) / 32;
  const int tile = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  if (tile >= co_tiles * j_tiles) return;
  const int co0 = (tile % co_tiles) * 32;
  const int j0 = (tile / co_tiles) * 32;
  const int n = blockIdx.z;

  const int co_a = min(co0 + j32, g.Co - 1);
  const bf16_t* dyrow = dy + ((long)n * g.Co + co_a) * g.Ho * g.Wo;
  // this lane's B column: (ci, r, s)
  const int j = min(j0 + j32, KRS - 1);
  const int ci = j / (g.R * g.S);
  const int rs = j % (g.R * g.S);
  const int r2 = rs / g.S, s2 = rs % g.S;
  const bf16_t* xn = x + ((long)n * g.Ci + ci) * g.H * g.W;

  const int HW = g.Ho * g.Wo;
  f32x16c acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  for (int kk = 0; kk < HW; kk += 16) {
    bf16x8c af, bf;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int k = kk + hi * 8 + e;
      float av = 0.f, bv = 0.f;
      if (k < HW) {
        av = bf2f(dyrow[k]);
        const int ho = k / g.Wo, wo = k % g.Wo;
        const int hh = ho * g.st - g.pad + r2;
        const int ww = wo * g.st - g.pad + s2;
        if (hh >= 0 && hh < g.H && ww >= 0 && ww < g.W)
          bv = bf2f(xn[(long)hh * g.W + ww]);
      }
      af[e] = f2bf(av);
      bf[e] = f2bf(bv);
    }
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int co = co0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    const int jc = j0 + j32;
    if (co < g.Co && jc < KRS)
      atomicAdd(&dw[(long)co * KRS + jc], acc[r]);
  }
}

ConvGeom make_geom(const at::Tensor& x, const at::Tensor& w, long st,
                   long pad) {
  ConvGeom g;
  g.N = (int)x.size(0);
  g.Ci = (int)x.size(1);
  g.H = (int)x.size(2);
  g.W = (int)x.size(3);
  g.Co = (int)w.size(0);
  g.R = (int)w.size(2);
  g.S = (int)w.size(3);
  g.st = (int)st;
  g.pad = (int)pad;
  g.Ho = (g.H + 2 * g.pad - g.R) / g.st + 1;
  g.Wo = (g.W + 2 * g.pad - g.S) / g.st + 1;
  return g;
}

}  // namespace

at::Tensor conv2d_fwd(at::Tensor x, at::Tensor w,
                      c10::optional<at::Tensor> bias, long st, long pad,
                      bool relu) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16, "conv2d_fwd: bf16 NCHW");
  TORCH_CHECK((int)w.size(1) == (int)x.size(1), "channel mismatch");
  auto g = make_geom(x, w, st, pad);
  auto y = at::empty({g.N, g.Co, g.Ho, g.Wo}, x.options());
  const float* bp = nullptr;
  at::Tensor bf32;
  if (bias.has_value() && bias->defined()) {
    bf32 = bias->to(at::kFloat).contiguous();
    bp = bf32.data_ptr<float>();
  }
  const long P = (long)g.N * g.Ho * g.Wo;
  const long tiles = ((P + 31) / 32) * ((g.Co + 31) / 32);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(conv_fwd_kernel, dim3((unsigned)((tiles + 3) / 4)),
                     dim3(256), 0, stream,
                     reinterpret_cast<const bf16_t*>(x.data_ptr()),
                     reinterpret_cast<const bf16_t*>(w.data_ptr()), bp,
                     reinterpret_cast<bf16_t*>(y.data_ptr()), g,
                     relu ? 1 : 0);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor conv2d_dgrad(at::Tensor dy, at::Tensor w, long N, long H,
                        long W, long st, long pad) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && w.is_contiguous());
  auto x_shape = at::empty({N, w.size(1), H, W}, dy.options());
  auto g = make_geom(x_shape, w, st, pad);
  TORCH_CHECK(g.Ho == (int)dy.size(2) && g.Wo == (int)dy.size(3));
  const long P = (long)g.N * g.H * g.W;
  const long tiles = ((P + 31) / 32) * ((g.Ci + 31) / 32);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(conv_dgrad_kernel, dim3((unsigned)((tiles + 3) / 4)),
                     dim3(256), 0, stream,
                     reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                     reinterpret_cast<const bf16_t*>(w.data_ptr()),
                     reinterpret_cast<bf16_t*>(x_shape.data_ptr()), g);
  HIP_CHECK_LAST();
  return x_shape;
}

at::Tensor conv2d_wgrad(at::Tensor dy, at::Tensor x, long R, long S,
                        long st, long pad) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  auto wshape = at::empty({dy.size(1), x.size(1), R, S},
                          x.options());
  auto g = make_geom(x, wshape, st, pad);
  TORCH_CHECK(g.Ho == (int)dy.size(2) && g.Wo == (int)dy.size(3));
  auto dw = at::zeros({dy.size(1), x.size(1) * R * S},
                      x.options().dtype(at::kFloat));
  const int tiles =
      ((g.Co + 31) / 32) * ((g.Ci * g.R * g.S + 31) / 32);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 grid((unsigned)((tiles + 3) / 4), 1, (unsigned)g.N);
  hipLaunchKernelGGL(conv_wgrad_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                     reinterpret_cast<const bf16_t*>(x.data_ptr()),
                     dw.data_ptr<float>(), g);
  HIP_CHECK_LAST();
  return dw.view({dy.size(1), x.size(1), R, S});
}
