// Fused LayerNorm forward/backward for CDNA4 (gfx950).
//
// One 64-lane wave per row; fp32 statistics; vectorized 16-byte loads
// (bf16x8) when the hidden size allows (guide: scalar bf16 loads are
// ~2x slower). dgamma/dbeta use a column-parallel reduction kernel
// (coalesced across lanes) instead of atomics.
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

// ---------------- forward ----------------
template <typename T, typename PT>
__global__ void ln_fwd_kernel(const T* __restrict__ x,
                              const PT* __restrict__ w,
                              const PT* __restrict__ b,
                              T* __restrict__ y,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out,
                              int H, long N, float eps) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (row >= N) return;
  const T* xr = x + row * H;
  T* yr = y + row * H;

  float s = 0.f, s2 = 0.f;
  if constexpr (sizeof(T) == 2) {
    if ((H & 7) == 0) {
      const int G = H >> 3;
      for (int g = lane; g < G; g += WAVE) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + g * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = us2f((unsigned short)v[j]);
          s += f;
          s2 += f * f;
        }
      }
      s = wave_sum(s);
      s2 = wave_sum(s2);
      const float mean = s / H;
      const float var = fmaxf(s2 / H - mean * mean, 0.f);
      const float rstd = rsqrtf(var + eps);
      if (lane == 0) {
        mean_out[row] = mean;
        rstd_out[row] = rstd;
      }
      for (int g = lane; g < G; g += WAVE) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + g * 8);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = us2f((unsigned short)v[j]);
          float wj = load1(w + g * 8 + j);
          float bj = load1(b + g * 8 + j);
          o[j] = (short)f2us((f - mean) * rstd * wj + bj);
        }
        *reinterpret_cast<bf16x8*>(yr + g * 8) = o;
      }
      return;
    }
  }
  // generic path
  for (int i = lane; i < H; i += WAVE) {
    float f = load1(xr + i);
    s += f;
    s2 += f * f;
  }
  s = wave_sum(s);
  s2 = wave_sum(s2);
  const float mean = s / H;
  const float var = fmaxf(s2 / H - mean * mean, 0.f);
  const float rstd = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  for (int i = lane; i < H; i += WAVE) {
    float f = load1(xr + i);
    store1(yr + i, (f - mean) * rstd * load1(w + i) + load1(b + i));
  }
}

// ---------------- fused residual-add + LayerNorm forward -------------
// y = LN(a + b); also writes s = a + b (saved for backward and as the
// next block's residual input). Saves one full elementwise add pass
// through HBM per LN (profile: the separate adds cost ~10ms/7 steps).
template <typename T, typename PT>
__global__ void ln_add_fwd_kernel(const T* __restrict__ a,
                                  const T* __restrict__ b,
                                  const PT* __restrict__ w,
                                  const PT* __restrict__ bias,
                                  T* __restrict__ y, T* __restrict__ s_out,
                                  float* __restrict__ mean_out,
                                  float* __restrict__ rstd_out,
                                  int H, long N, float eps) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (row >= N) return;
  const T* ar = a + row * H;
  const T* br = b + row * H;
  T* yr = y + row * H;
  T* sr = s_out + row * H;

  float s1 = 0.f, s2 = 0.f;
  if constexpr (sizeof(T) == 2) {
    if ((H & 7) == 0) {
      const int G = H >> 3;
      for (int g = lane; g < G; g += WAVE) {
        bf16x8 va = *reinterpret_cast<const bf16x8*>(ar + g * 8);
        bf16x8 vb = *reinterpret_cast<const bf16x8*>(br + g * 8);
        bf16x8 vs;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = us2f((unsigned short)va[j]) + us2f((unsigned short)vb[j]);
          vs[j] = (short)f2us(f);
          f = us2f((unsigned short)vs[j]);  // quantized sum (bitwise saved)
          s1 += f;
          s2 += f * f;
        }
        *reinterpret_cast<bf16x8*>(sr + g * 8) = vs;
      }
      s1 = wave_sum(s1);
      s2 = wave_sum(s2);
      const float mean = s1 / H;
      const float var = fmaxf(s2 / H - mean * mean, 0.f);
      const float rstd = rsqrtf(var + eps);
      if (lane == 0) {
        mean_out[row] = mean;
        rstd_out[row] = rstd;
      }
      for (int g = lane; g < G; g += WAVE) {
        bf16x8 vs = *reinterpret_cast<const bf16x8*>(sr + g * 8);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = us2f((unsigned short)vs[j]);
          o[j] = (short)f2us((f - mean) * rstd * load1(w + g * 8 + j) +
                             load1(bias + g * 8 + j));
        }
        *reinterpret_cast<bf16x8*>(yr + g * 8) = o;
      }
      return;
    }
  }
  for (int i = lane; i < H; i += WAVE) {
    float f = load1(ar + i) + load1(br + i);
    store1(sr + i, f);
    f = load1(sr + i);
    s1 += f;
    s2 += f * f;
  }
  s1 = wave_sum(s1);
  s2 = wave_sum(s2);
  const float mean = s1 / H;
  const float var = fmaxf(s2 / H - mean * mean, 0.f);
  const float rstd = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  for (int i = lane; i < H; i += WAVE) {
    float f = load1(sr + i);
    store1(yr + i, (f - mean) * rstd * load1(w + i) + load1(bias + i));
  }
}

// ---------------- backward: dx ----------------
template <typename T, typename PT>
__global__ void ln_bwd_dx_kernel(const T* __restrict__ dy,
                                 const T* __restrict__ x,
                                 const PT* __restrict__ w,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 T* __restrict__ dx, int H, long N) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (row >= N) return;
  const T* dyr = dy + row * H;
  const T* xr = x + row * H;
  T* dxr = dx + row * H;
  const float mu = mean[row], rs = rstd[row];

  float c1 = 0.f, c2 = 0.f;
  if constexpr (sizeof(T) == 2) {
    if ((H & 7) == 0) {
      const int G = H >> 3;
      for (int g8 = lane; g8 < G; g8 += WAVE) {
        bf16x8 vd = *reinterpret_cast<const bf16x8*>(dyr + g8 * 8);
        bf16x8 vx = *reinterpret_cast<const bf16x8*>(xr + g8 * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = us2f((unsigned short)vd[j]) * load1(w + g8 * 8 + j);
          float xhat = (us2f((unsigned short)vx[j]) - mu) * rs;
          c1 += g * xhat;
          c2 += g;
        }
      }
      c1 = wave_sum(c1) / H;
      c2 = wave_sum(c2) / H;
      for (int g8 = lane; g8 < G; g8 += WAVE) {
        bf16x8 vd = *reinterpret_cast<const bf16x8*>(dyr + g8 * 8);
        bf16x8 vx = *reinterpret_cast<const bf16x8*>(xr + g8 * 8);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = us2f((unsigned short)vd[j]) * load1(w + g8 * 8 + j);
          float xhat = (us2f((unsigned short)vx[j]) - mu) * rs;
          o[j] = (short)f2us(rs * (g - xhat * c1 - c2));
        }
        *reinterpret_cast<bf16x8*>(dxr + g8 * 8) = o;
      }
      return;
    }
  }
  for (int i = lane; i < H; i += WAVE) {
    float g = load1(dyr + i) * load1(w + i);
    float xhat = (load1(xr + i) - mu) * rs;
    c1 += g * xhat;
    c2 += g;
  }
  c1 = wave_sum(c1) / H;
  c2 = wave_sum(c2) / H;
  for (int i = lane; i < H; i += WAVE) {
    float g = load1(dyr + i) * load1(w + i);
    float xhat = (load1(xr + i) - mu) * rs;
    store1(dxr + i, rs * (g - xhat * c1 - c2));
  }
}

// ---------------- backward: dgamma/dbeta ------------------------------
// Phase 1: row-chunked partial column sums (coalesced across lanes, the
// grid covers chunks x columns so all 256 CUs stay busy). Phase 2: small
// reduce over chunks.
template <typename T>
__global__ void ln_bwd_dwdb_partial(const T* __restrict__ dy,
                                    const T* __restrict__ x,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    float* __restrict__ part_w,
                                    float* __restrict__ part_b,
                                    int H, long N, long rows_per_chunk) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  const int chunk = blockIdx.y;
  if (col >= H) return;
  const long r0 = (long)chunk * rows_per_chunk;
  const long r1 = min(N, r0 + rows_per_chunk);
  float sw = 0.f, sb = 0.f;
  long r = r0;
  for (; r + 4 <= r1; r += 4) {  // keep several loads in flight
    float g[4], xh[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      g[j] = load1(dy + (r + j) * H + col);
      xh[j] = (load1(x + (r + j) * H + col) - mean[r + j]) * rstd[r + j];
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      sw += g[j] * xh[j];
      sb += g[j];
    }
  }
  for (; r < r1; ++r) {
    float g = load1(dy + r * H + col);
    float xhat = (load1(x + r * H + col) - mean[r]) * rstd[r];
    sw += g * xhat;
    sb += g;
  }
  // few hundred chunk-blocks per column: atomics are cheap here and
  // remove the second kernel + workspace (profile: the final reduce cost
  // 70us/call at bad occupancy)
  atomicAdd(&part_w[col], sw);
  atomicAdd(&part_b[col], sb);
}

}  // namespace

std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w,
                                      at::Tensor b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int H = x.size(-1);
  const long N = x.numel() / H;
  auto y = at::empty_like(x);
  auto mean = at::empty({N}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({N}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  const int waves_per_block = 4;
  dim3 block(WAVE * waves_per_block);
  dim3 grid((N + waves_per_block - 1) / waves_per_block);

#define DISPATCH_LN_FWD(T, PT)                                              \
  hipLaunchKernelGGL((ln_fwd_kernel<T, PT>), grid, block, 0, stream,        \
                     reinterpret_cast<const T*>(x.data_ptr()),              \
                     reinterpret_cast<const PT*>(w.data_ptr()),             \
                     reinterpret_cast<const PT*>(b.data_ptr()),             \
                     reinterpret_cast<T*>(y.data_ptr()),                    \
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), H, N,  \
                     (float)eps)

  if (x.scalar_type() == at::kBFloat16 && w.scalar_type() == at::kFloat) {
    DISPATCH_LN_FWD(bf16_t, float);
  } else if (x.scalar_type() == at::kBFloat16) {
    DISPATCH_LN_FWD(bf16_t, bf16_t);
  } else if (w.scalar_type() == at::kFloat) {
    DISPATCH_LN_FWD(float, float);
  } else {
    TORCH_CHECK(false, "layernorm_fwd: unsupported dtype combo");
  }
  HIP_CHECK_LAST();
  return {y, mean, rstd};
}

std::vector<at::Tensor> layernorm_bwd(at::Tensor dy, at::Tensor x,
                                      at::Tensor w, at::Tensor mean,
                                      at::Tensor rstd) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  TORCH_CHECK(dy.scalar_type() == x.scalar_type());
  const int H = x.size(-1);
  const long N = x.numel() / H;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({H}, x.options().dtype(at::kFloat));
  auto db = at::zeros({H}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  const int waves_per_block = 4;
  dim3 block(WAVE * waves_per_block);
  dim3 grid((N + waves_per_block - 1) / waves_per_block);

  // dw/db: pick chunks so chunks x col-blocks fills the 256-CU chip
  const int col_blocks = (H + 255) / 256;
  int nchunks = (int)std::min<long>((N + 63) / 64,
                                    std::max(1L, (long)(768 / col_blocks)));
  nchunks = std::max(nchunks, 1);
  const long rows_per_chunk = (N + nchunks - 1) / nchunks;
  dim3 cblock(256);
  dim3 cgrid(col_blocks, nchunks);

#define DISPATCH_LN_BWD(T, PT)                                              \
  do {                                                                      \
    hipLaunchKernelGGL((ln_bwd_dx_kernel<T, PT>), grid, block, 0, stream,   \
                       reinterpret_cast<const T*>(dy.data_ptr()),           \
                       reinterpret_cast<const T*>(x.data_ptr()),            \
                       reinterpret_cast<const PT*>(w.data_ptr()),           \
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),      \
                       reinterpret_cast<T*>(dx.data_ptr()), H, N);          \
    hipLaunchKernelGGL((ln_bwd_dwdb_partial<T>), cgrid, cblock, 0, stream,  \
                       reinterpret_cast<const T*>(dy.data_ptr()),           \
                       reinterpret_cast<const T*>(x.data_ptr()),            \
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),      \
                       dw.data_ptr<float>(), db.data_ptr<float>(),          \
                       H, N, rows_per_chunk);                               \
  } while (0)

  if (x.scalar_type() == at::kBFloat16 && w.scalar_type() == at::kFloat) {
    DISPATCH_LN_BWD(bf16_t, float);
  } else if (x.scalar_type() == at::kBFloat16) {
    DISPATCH_LN_BWD(bf16_t, bf16_t);
  } else if (w.scalar_type() == at::kFloat) {
    DISPATCH_LN_BWD(float, float);
  } else {
    TORCH_CHECK(false, "layernorm_bwd: unsupported dtype combo");
  }
  HIP_CHECK_LAST();
  // grads in the weight dtype
  auto dw_c = dw.to(w.scalar_type());
  auto db_c = db.to(w.scalar_type());
  return {dx, dw_c, db_c};
}


std::vector<at::Tensor> layernorm_add_fwd(at::Tensor a, at::Tensor b,
                                          at::Tensor w, at::Tensor bias,
                                          double eps) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(a.sizes() == b.sizes() && a.scalar_type() == b.scalar_type());
  const int H = a.size(-1);
  const long N = a.numel() / H;
  auto y = at::empty_like(a);
  auto s_out = at::empty_like(a);
  auto mean = at::empty({N}, a.options().dtype(at::kFloat));
  auto rstd = at::empty({N}, a.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  const int waves_per_block = 4;
  dim3 block(WAVE * waves_per_block);
  dim3 grid((N + waves_per_block - 1) / waves_per_block);

#define DISPATCH_LN_ADD_FWD(T, PT)                                          \
  hipLaunchKernelGGL((ln_add_fwd_kernel<T, PT>), grid, block, 0, stream,    \
                     reinterpret_cast<const T*>(a.data_ptr()),              \
                     reinterpret_cast<const T*>(b.data_ptr()),              \
                     reinterpret_cast<const PT*>(w.data_ptr()),             \
                     reinterpret_cast<const PT*>(bias.data_ptr()),          \
                     reinterpret_cast<T*>(y.data_ptr()),                    \
                     reinterpret_cast<T*>(s_out.data_ptr()),                \
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), H, N,  \
                     (float)eps)

  if (a.scalar_type() == at::kBFloat16 && w.scalar_type() == at::kFloat) {
    DISPATCH_LN_ADD_FWD(bf16_t, float);
  } else if (a.scalar_type() == at::kBFloat16) {
    DISPATCH_LN_ADD_FWD(bf16_t, bf16_t);
  } else if (w.scalar_type() == at::kFloat) {
    DISPATCH_LN_ADD_FWD(float, float);
  } else {
    TORCH_CHECK(false, "layernorm_add_fwd: unsupported dtype combo");
  }
  HIP_CHECK_LAST();
  return {y, s_out, mean, rstd};
}
