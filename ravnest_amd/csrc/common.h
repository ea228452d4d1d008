// Shared helpers for the ravnest_amd CDNA4 (gfx950) kernel library.
// Wave width is 64 on CDNA4 — hard-coded per the HIP programming guide.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

typedef __hip_bfloat16 bf16_t;

// ---- vector types for 16-byte loads (8 x bf16 / 4 x f32) -------------
typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

DEVINL float bf2f(bf16_t v) { return __bfloat162float(v); }
DEVINL bf16_t f2bf(float v) { return __float2bfloat16(v); }

DEVINL float us2f(unsigned short u) {
  unsigned int x = ((unsigned int)u) << 16;
  return __builtin_bit_cast(float, x);
}
DEVINL unsigned short f2us(float f) {
  // round-to-nearest-even bf16 truncation
  unsigned int x = __builtin_bit_cast(unsigned int, f);
  unsigned int lsb = (x >> 16) & 1;
  x += 0x7fffu + lsb;
  return (unsigned short)(x >> 16);
}

// ---- wave-level reductions (64 lanes) --------------------------------
DEVINL float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return __shfl(v, 0, WAVE);
}

DEVINL float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  return __shfl(v, 0, WAVE);
}

// block reduction across up to 16 waves through LDS
DEVINL float block_sum(float v, float* lds_scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = (blockDim.x + WAVE - 1) / WAVE;
  v = wave_sum(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  float r = (threadIdx.x < nw) ? lds_scratch[threadIdx.x] : 0.f;
  if (wid == 0) {
    r = wave_sum(r);
    if (lane == 0) lds_scratch[0] = r;
  }
  __syncthreads();
  r = lds_scratch[0];
  __syncthreads();
  return r;
}

DEVINL float block_max(float v, float* lds_scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = (blockDim.x + WAVE - 1) / WAVE;
  v = wave_max(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  float r = (threadIdx.x < nw) ? lds_scratch[threadIdx.x] : -INFINITY;
  if (wid == 0) {
    r = wave_max(r);
    if (lane == 0) lds_scratch[0] = r;
  }
  __syncthreads();
  r = lds_scratch[0];
  __syncthreads();
  return r;
}

// ---- philox4x32-10 (counter-based, replayable) -----------------------
struct Philox4 {
  unsigned int k0, k1;
  DEVINL Philox4(unsigned long long seed) {
    k0 = (unsigned int)(seed & 0xffffffffu);
    k1 = (unsigned int)(seed >> 32);
  }
  DEVINL static unsigned int mulhi(unsigned int a, unsigned int b) {
    return (unsigned int)(((unsigned long long)a * b) >> 32);
  }
  DEVINL void round_(unsigned int& c0, unsigned int& c1, unsigned int& c2,
                     unsigned int& c3, unsigned int key0,
                     unsigned int key1) const {
    const unsigned int M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
    unsigned int h0 = mulhi(M0, c0), l0 = M0 * c0;
    unsigned int h1 = mulhi(M1, c2), l1 = M1 * c2;
    unsigned int n0 = h1 ^ c1 ^ key0, n1 = l1;
    unsigned int n2 = h0 ^ c3 ^ key1, n3 = l0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
  }
  // 4 uniform u32 from a 128-bit counter
  DEVINL void gen(unsigned long long ctr_lo, unsigned long long ctr_hi,
                  unsigned int out[4]) const {
    unsigned int c0 = (unsigned int)ctr_lo, c1 = (unsigned int)(ctr_lo >> 32);
    unsigned int c2 = (unsigned int)ctr_hi, c3 = (unsigned int)(ctr_hi >> 32);
    unsigned int key0 = k0, key1 = k1;
    const unsigned int B0 = 0x9E3779B9u, B1 = 0xBB67AE85u;
#pragma unroll
    for (int i = 0; i < 9; ++i) {
      round_(c0, c1, c2, c3, key0, key1);
      key0 += B0; key1 += B1;
    }
    round_(c0, c1, c2, c3, key0, key1);
    out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
  }
};

#define HIP_CHECK_LAST()                                                    \
  do {                                                                      \
    hipError_t e_ = hipGetLastError();                                      \
    TORCH_CHECK(e_ == hipSuccess, "HIP kernel launch failed: ",             \
                hipGetErrorString(e_));                                     \
  } while (0)
