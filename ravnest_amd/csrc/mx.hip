// MX-scaled fp8 (OCP e4m3) path for CDNA4 (gfx950).
//
// gfx950's only large-K low-precision MFMA is the block-scaled
// `mfma_scale_f32_32x32x64_f8f6f4`: per-32-element k-blocks carry an
// e8m0 scale and the hardware fuses dequant+matmul at ~2x the bf16 MFMA
// rate (guide section 4 "FP8/FP6/FP4 — block-scaled MX").
//
// Fragment/scale semantics established ON-DEVICE (mx_scale_probe with
// unique-contribution chunks): the k-pairing of A/B is positional (any
// loading works if A and B use the same logical order — matmul is
// k-permutation invariant), but scale byte 0 of lane (i, s) applies to
// element POSITIONS e in [16s, 16s+16) of BOTH half-lanes of row i.
// Aligning MX 32-blocks with that partition requires the k-INTERLEAVED
// loading  pos(hi, e) <-> k = (e>>4)*32 + hi*16 + (e&15):
// then lane (i, hi) supplies exactly block hi's e8m0 scale.
//
// Pieces: mx_quant (bf16 -> e4m3 bytes + per-32-block e8m0 scales,
// round-to-nearest-even), mx_gemm (y = x @ W^T over quantized inputs —
// both operands K-contiguous row-major, which is exactly the TN Linear
// layout), and a scale-semantics probe.
//
// Workload parity: BASELINE.json config "GPT-Sorter fp8 CDNA4 MFMA
// path"; ops/mx.py wires it as MXLinear.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef int i32x8v __attribute__((ext_vector_type(8)));
typedef float f32x16v __attribute__((ext_vector_type(16)));

namespace {

// float -> OCP e4m3fn byte, round-to-nearest-even, saturate to ±448
DEVINL unsigned char f32_to_e4m3(float v) {
  if (v != v) return 0x7F;  // NaN
  const unsigned int bits = __float_as_uint(v);
  const unsigned char sign = (bits >> 24) & 0x80;
  float a = fabsf(v);
  if (a > 448.f) return sign | 0x7E;  // saturate to max finite
  if (a < 0.0009765625f)  // < 2^-10 = half of the min subnormal
    return sign;
  // scale into e4m3: exponent bias 7, 3 mantissa bits
  int e;
  float m = frexpf(a, &e);  // a = m * 2^e, m in [0.5, 1)
  int exp = e - 1 + 7;      // e4m3 biased exponent of a
  unsigned int mant;
  if (exp >= 1) {
    // normal: mant from m in [1,2)
    mant = (unsigned int)rintf((m * 2.f - 1.f) * 8.f);
    if (mant == 8) {
      mant = 0;
      exp += 1;
    }
    if (exp > 15 || (exp == 15 && mant == 7))
      return sign | 0x7E;  // overflow -> max finite (0x7E = 448)
    return sign | (unsigned char)((exp << 3) | mant);
  }
  // subnormal: value = mant * 2^-9
  mant = (unsigned int)rintf(a * 512.f);
  if (mant > 7) mant = 7;
  return sign | (unsigned char)mant;
}

// bf16 (R, K) -> e4m3 bytes (R, K) + e8m0 scales (R, K/32).
// one wave per 2 blocks: lane handles 1 elem of a 32-block per half.
__global__ void mx_quant_kernel(const bf16_t* __restrict__ x,
                                unsigned char* __restrict__ q,
                                unsigned char* __restrict__ scales,
                                long nblocks) {
  const long blk = ((long)blockIdx.x * blockDim.x + threadIdx.x) / 32;
  const int e = threadIdx.x & 31;
  if (blk >= nblocks) return;
  const bf16_t* xp = x + blk * 32;
  const float v = bf2f(xp[e]);
  float amax = fabsf(v);
#pragma unroll
  for (int off = 16; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, WAVE));
  // e8m0 scale 2^s with amax/2^s <= 448
  int s = 0;
  if (amax > 0.f) {
    int ex;
    frexpf(amax / 448.f, &ex);  // amax/448 in [2^(ex-1), 2^ex)
    s = ex;                      // smallest s with amax/2^s <= 448
    if (s < -127) s = -127;
    if (s > 127) s = 127;
  } else {
    s = -127;
  }
  q[blk * 32 + e] = f32_to_e4m3(ldexpf(v, -s));
  if (e == 0) scales[blk] = (unsigned char)(s + 127);
}

// y[M,N] (bf16) = x[M,K]e4m3 @ W[N,K]e4m3^T, both with (.., K/32) e8m0
// scales. One wave per 64x64 output tile (2x2 register tiling of the
// 32x32 MFMA): 2 A + 2 B fragment loads feed 4 MFMAs per k-step, 2x the
// arithmetic intensity of a 32x32 tile. K streamed from L2; the LDS/
// 8-phase ladder is the round-2 upgrade.
typedef int i32x4v __attribute__((ext_vector_type(4)));

DEVINL i32x8v load_frag_ilv(const unsigned char* p, int hi) {
  // k-interleaved fragment halves (see header comment): positions
  // e<16 <- block0 cols [hi*16,+16), e>=16 <- block1 cols [hi*16,+16)
  i32x8v f;
  *reinterpret_cast<i32x4v*>(&f) =
      *reinterpret_cast<const i32x4v*>(p + hi * 16);
  *(reinterpret_cast<i32x4v*>(&f) + 1) =
      *reinterpret_cast<const i32x4v*>(p + 32 + hi * 16);
  return f;
}

__global__ __launch_bounds__(256, 1) void mx_gemm_kernel(
    const unsigned char* __restrict__ x, const unsigned char* __restrict__ xs,
    const unsigned char* __restrict__ w, const unsigned char* __restrict__ ws,
    bf16_t* __restrict__ y, int M, int N, int K) {
  constexpr int MI = 4, NJ = 2;  // 128x64 per wave: 6 loads, 8 MFMAs
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int hi = lane >> 5;
  const int j32 = lane & 31;
  const int tiles_n = (N + NJ * 32 - 1) / (NJ * 32);
  const long tile = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  const int m0 = (int)(tile / tiles_n) * (MI * 32);
  const int n0 = (int)(tile % tiles_n) * (NJ * 32);
  if (m0 >= M) return;
  const int kb = K / 32;
  const unsigned char* xp[MI];
  const unsigned char* wp[NJ];
  const unsigned char* xsp[MI];
  const unsigned char* wsp[NJ];
#pragma unroll
  for (int t = 0; t < MI; ++t) {
    const int ra = min(m0 + t * 32 + j32, M - 1);
    xp[t] = x + (long)ra * K;
    xsp[t] = xs + (long)ra * kb;
  }
#pragma unroll
  for (int t = 0; t < NJ; ++t) {
    const int rb = min(n0 + t * 32 + j32, N - 1);
    wp[t] = w + (long)rb * K;
    wsp[t] = ws + (long)rb * kb;
  }

  f32x16v acc[MI][NJ];
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[i][j][r] = 0.f;

  for (int k0 = 0; k0 < K; k0 += 64) {
    i32x8v af[MI], bf[NJ];
    int sa[MI], sb[NJ];
#pragma unroll
    for (int t = 0; t < MI; ++t) {
      af[t] = load_frag_ilv(xp[t] + k0, hi);
      sa[t] = xsp[t][k0 / 32 + hi];
    }
#pragma unroll
    for (int t = 0; t < NJ; ++t) {
      bf[t] = load_frag_ilv(wp[t] + k0, hi);
      sb[t] = wsp[t][k0 / 32 + hi];
    }
#pragma unroll
    for (int i = 0; i < MI; ++i)
#pragma unroll
      for (int j = 0; j < NJ; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            af[i], bf[j], acc[i][j], 0, 0, 0, sa[i], 0, sb[j]);
  }
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = m0 + i * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const int col = n0 + j * 32 + j32;
        if (row < M && col < N) y[(long)row * N + col] = f2bf(acc[i][j][r]);
      }
}

// scale-semantics probe: per-lane scale operands
__global__ void mx_scale_probe_kernel(const unsigned char* __restrict__ a,
                                      const unsigned char* __restrict__ b,
                                      const int* __restrict__ sa_l,
                                      const int* __restrict__ sb_l,
                                      float* __restrict__ d) {
  if (threadIdx.x >= 64) return;
  const int lane = threadIdx.x;
  const int hi = lane >> 5;
  i32x8v af, bf;
  unsigned char* afb = reinterpret_cast<unsigned char*>(&af);
  unsigned char* bfb = reinterpret_cast<unsigned char*>(&bf);
#pragma unroll
  for (int e = 0; e < 32; ++e) {
    afb[e] = a[(lane & 31) * 64 + hi * 32 + e];
    bfb[e] = b[(hi * 32 + e) * 32 + (lane & 31)];
  }
  f32x16v acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
      af, bf, acc, 0, 0, 0, sa_l[lane], 0, sb_l[lane]);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    d[row * 32 + (lane & 31)] = acc[r];
  }
}

}  // namespace

std::vector<at::Tensor> mx_quant(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == at::kBFloat16);
  const long K = x.size(-1);
  TORCH_CHECK(K % 32 == 0, "mx_quant: last dim % 32 == 0");
  const long R = x.numel() / K;
  auto q = at::empty_like(x, x.options().dtype(at::kByte));
  auto s = at::empty({R, K / 32}, x.options().dtype(at::kByte));
  const long nblocks = R * (K / 32);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid((unsigned)((nblocks * 32 + 255) / 256));
  hipLaunchKernelGGL(mx_quant_kernel, grid, block, 0, stream,
                     reinterpret_cast<const bf16_t*>(x.data_ptr()),
                     q.data_ptr<unsigned char>(),
                     s.data_ptr<unsigned char>(), nblocks);
  HIP_CHECK_LAST();
  return {q, s};
}

at::Tensor mx_gemm(at::Tensor x, at::Tensor xs, at::Tensor w,
                   at::Tensor ws) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kByte &&
              w.scalar_type() == at::kByte);
  const int K = (int)x.size(-1);
  const int M = (int)(x.numel() / K);
  const int N = (int)w.size(0);
  TORCH_CHECK((int)w.size(1) == K && K % 64 == 0,
              "mx_gemm: K must match and be a multiple of 64");
  auto y = at::empty({(long)M, (long)N},
                     x.options().dtype(at::kBFloat16));
  const long tiles = (long)((M + 127) / 128) * ((N + 63) / 64);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid((unsigned)((tiles + 3) / 4));
  hipLaunchKernelGGL(mx_gemm_kernel, grid, block, 0, stream,
                     x.data_ptr<unsigned char>(),
                     xs.data_ptr<unsigned char>(),
                     w.data_ptr<unsigned char>(),
                     ws.data_ptr<unsigned char>(),
                     reinterpret_cast<bf16_t*>(y.data_ptr()), M, N, K);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor mx_scale_probe(at::Tensor a, at::Tensor b, at::Tensor sa,
                          at::Tensor sb) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kByte);
  TORCH_CHECK(sa.scalar_type() == at::kInt && sa.numel() == 64);
  auto d = at::zeros({32, 32}, a.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(mx_scale_probe_kernel, dim3(1), dim3(64), 0, stream,
                     a.data_ptr<unsigned char>(),
                     b.data_ptr<unsigned char>(), sa.data_ptr<int>(),
                     sb.data_ptr<int>(), d.data_ptr<float>());
  HIP_CHECK_LAST();
  return d;
}
