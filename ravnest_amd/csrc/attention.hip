// Flash-style fused attention forward for CDNA4 (gfx950), bf16, D=64/128.
//
// Structure (per cdna_hip_programming.md section B "fused attention
// prefill" ladder, adapted to head_dim 64): one 64-lane wave owns a
// 32-row Q tile; swapped QK^T (mfma(K, Q)) puts each q-row's scores
// lane-local so the online-softmax row reduction is 15 register ops +
// one __shfl_xor(32); P->bf16 repacking for the PV MFMA A-operand uses
// v_cvt_pk_bf16_f32-equivalent packing + __builtin_amdgcn_permlane32_swap
// (guide T12). K/V tiles are read straight from L2 (at S<=2k and D<=128
// the K/V working set per (b,h) is L2-resident; LDS staging is pure
// overhead — guide common-mistake #7).
//
// MFMA C/D layout used (guide section 3, verified by mfma_probe_*):
//   v_mfma_f32_32x32x16_bf16: D[row][col], col = lane&31,
//     row = (reg&3) + 8*(reg>>2) + 4*(lane>>5), reg in [0,16)
//   A[i][k]: i = lane&31, k = (lane>>5)*8 + e, e in [0,8)
//   B[k][j]: j = lane&31, k = (lane>>5)*8 + e
//
// Workload parity: minGPT causal attention and BERT padding-mask
// attention (SURVEY.md section 2.3 attention row).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8v __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4v __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

// hardware transpose read (guide T10; exact semantics verified on-device
// by tr16_probe): every lane supplies its OWN 8-B-aligned address and
// loads 4 contiguous bf16; per 16-lane group those 16 loads (in lane
// order) form a [4][16] row-major block M, and lane q of the group
// receives column q, i.e. M[0..3][q].
#define DS_TR16(p) __builtin_amdgcn_ds_read_tr16_b64_v4bf16( \
    (__attribute__((address_space(3))) bf16x4v*)( \
        (__attribute__((address_space(3))) void*)(p)))

namespace {

DEVINL float bfv2f(__bf16 v) { return (float)v; }

// pack two f32 into one u32 of 2 bf16 (lo, hi)
DEVINL unsigned int pack_bf2(float lo, float hi) {
  return (unsigned int)f2us(lo) | ((unsigned int)f2us(hi) << 16);
}

// addressing: element offset of (b, h, row) = b*bs + h*hs + row*rs.
// separate (B,H,S,D) tensors: bs=H*S*D, hs=S*D, rs=D.
// packed qkv (B,S,3,H,D): bs=S*3*H*D, hs=D (+ part*H*D folded into the
// base pointer), rs=3*H*D. o packed (B,S,H,D): bs=S*H*D, hs=D, rs=H*D.
struct Strides { long bs, hs, rs; };

template <int D>  // head_dim: 64 or 128
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const float* __restrict__ mask,
    bf16_t* __restrict__ o, float* __restrict__ lse, int S, int causal,
    float scale, int has_mask, long H,
    Strides sq, Strides sk, Strides sv, Strides so) {
  constexpr int DSTEPS = D / 16;   // QK^T k-steps
  constexpr int DHALF = D / 32;    // PV column halves
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int hi = lane >> 5;        // 0 or 1 (lane half)
  const int j32 = lane & 31;

  const int qtile = blockIdx.x * 4 + wid;  // 32-row q tile index
  const int q0 = qtile * 32;
  if (q0 >= S) return;
  const long bh = blockIdx.y;  // fused batch*head index
  const long b = bh / H, h = bh % H;
  const bf16_t* qp = q + b * sq.bs + h * sq.hs;
  const bf16_t* kp = k + b * sk.bs + h * sk.hs;
  const bf16_t* vp = v + b * sv.bs + h * sv.hs;
  bf16_t* op = o + b * so.bs + h * so.hs;
  const float* mp = has_mask ? (mask + b * (long)S) : nullptr;

  // ---- load Q fragments: B-operand, j = qrow = j32, k = d ----
  // frag[s] covers d in [s*16 + hi*8, +8)
  bf16x8v qf[DSTEPS];
  const int qrow = q0 + j32;
  const bf16_t* qrp = qp + (long)min(qrow, S - 1) * sq.rs;
#pragma unroll
  for (int s = 0; s < DSTEPS; ++s)
    qf[s] = *reinterpret_cast<const bf16x8v*>(qrp + s * 16 + hi * 8);

  // ---- state ----
  f32x16 oacc[DHALF];
#pragma unroll
  for (int h = 0; h < DHALF; ++h)
#pragma unroll
    for (int r = 0; r < 16; ++r) oacc[h][r] = 0.f;
  float m_run = -INFINITY;  // running max for qrow j32
  float l_run = 0.f;        // running sum for qrow j32

  const int kv_end = causal ? min(S, q0 + 32) : S;

  for (int k0 = 0; k0 < kv_end; k0 += 32) {
    // ---- QK^T: A = K tile (i = key = j32 local), k = d ----
    f32x16 s_acc;
#pragma unroll
    for (int r = 0; r < 16; ++r) s_acc[r] = 0.f;
    const int krow = k0 + j32;
    const bf16_t* krp = kp + (long)min(krow, S - 1) * sk.rs;
#pragma unroll
    for (int s = 0; s < DSTEPS; ++s) {
      bf16x8v kf = *reinterpret_cast<const bf16x8v*>(krp + s * 16 + hi * 8);
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], s_acc,
                                                      0, 0, 0);
    }

    // ---- scale + mask + causal; track tile max ----
    // reg r holds score for key kq = k0 + (r&3) + 8*(r>>2) + 4*hi,
    // qrow = q0 + j32
    float tile_max = -INFINITY;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = k0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      float sv = s_acc[r] * scale;
      if (key >= S) sv = -INFINITY;
      if (causal && key > qrow) sv = -INFINITY;
      if (has_mask) sv += mp[key];  // additive (B,1,1,S) mask, see host
      s_acc[r] = sv;
      tile_max = fmaxf(tile_max, sv);
    }
    // combine the two lane halves (same qrow, different keys)
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, WAVE));

    const float m_new = fmaxf(m_run, tile_max);
    // alpha is per Q-ROW. This lane's softmax state tracks qrow = j32,
    // but the PV accumulator's rows follow the MFMA C layout
    // (row = (r&3)+8*(r>>2)+4*hi), so the O rescale needs each row's
    // alpha broadcast from the lane that owns it. On random data the
    // running max stops moving after the first tiles: skip the whole
    // rescale (16 shfl + 32 mul) when NO row in the wave changed.
    float alpha = 1.f;
    if (m_new != m_run) {
      alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
      m_run = m_new;
    }
    l_run *= alpha;
    // note: guarding this with __any(alpha != 1) measured SLOWER
    // (-17% fwd): the wave-uniform branch defeats the compiler's
    // load/MFMA pipelining across the tile boundary
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row_local = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const float a_r = __shfl(alpha, row_local, WAVE);
#pragma unroll
      for (int h = 0; h < DHALF; ++h) oacc[h][r] *= a_r;
    }

    // ---- exponentiate + row-sum ----
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      float p = (s_acc[r] == -INFINITY) ? 0.f : __expf(s_acc[r] - m_run);
      s_acc[r] = p;
      psum += p;
    }
    psum += __shfl_xor(psum, 32, WAVE);
    l_run += psum;

    // ---- pack P (f32, acc layout) -> PV A-operand fragments ----
    // lane needs A[i=qrow][k=key]: keys (hi*8 + e) + 16*step.
    // cvt_pk pairs + permlane32_swap redistribute the halves (T12).
    unsigned int pa[2][4];  // [k-step][4 u32 = 8 bf16]
#pragma unroll
    for (int step = 0; step < 2; ++step) {
      const int b0 = step * 8;
      unsigned int c01 = pack_bf2(s_acc[b0 + 0], s_acc[b0 + 1]);
      unsigned int c23 = pack_bf2(s_acc[b0 + 2], s_acc[b0 + 3]);
      unsigned int c45 = pack_bf2(s_acc[b0 + 4], s_acc[b0 + 5]);
      unsigned int c67 = pack_bf2(s_acc[b0 + 6], s_acc[b0 + 7]);
      {
        auto r2 = __builtin_amdgcn_permlane32_swap(c01, c45, false, false);
        pa[step][0] = r2[0];  // keys (0,1)|(8,9) -> e0,e1
        pa[step][2] = r2[1];  // keys (4,5)|(12,13) -> e4,e5
      }
      {
        auto r2 = __builtin_amdgcn_permlane32_swap(c23, c67, false, false);
        pa[step][1] = r2[0];  // e2,e3
        pa[step][3] = r2[1];  // e6,e7
      }
    }

    // ---- PV: O[qrow][d] += P A-frag x V B-frag ----
    // B[k=key][j=d]: lane reads V[k0 + step*16 + hi*8 + e][dhalf*32 + j32]
#pragma unroll
    for (int hh = 0; hh < DHALF; ++hh) {
#pragma unroll
      for (int step = 0; step < 2; ++step) {
        bf16x8v vf;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int key = k0 + step * 16 + hi * 8 + e;
          vf[e] = *reinterpret_cast<const __bf16*>(
              vp + (long)min(key, S - 1) * sv.rs + hh * 32 + j32);
        }
        oacc[hh] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            *reinterpret_cast<const bf16x8v*>(&pa[step][0]), vf, oacc[hh],
            0, 0, 0);
      }
    }
  }

  // ---- epilogue: O /= l, store O + lse ----
  // broadcast l (and guard empty rows)
  const float l_safe = (l_run > 0.f) ? l_run : 1.f;
  // per-reg qrow for the PV acc layout
#pragma unroll
  for (int hh = 0; hh < DHALF; ++hh) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row_local = (r & 3) + 8 * (r >> 2) + 4 * hi;
      // l of that row lives in lane row_local (and row_local+32)
      const float l_row = __shfl(l_safe, row_local, WAVE);
      const float val = oacc[hh][r] / l_row;
      const int row = q0 + row_local;
      if (row < S)
        op[(long)row * so.rs + hh * 32 + j32] = f2bf(val);
    }
  }
  if (qrow < S && hi == 0)
    lse[bh * (long)S + qrow] =
        (l_run > 0.f) ? m_run + __logf(l_run) : -INFINITY;
}

// ---------------------------------------------------------------------
// fwd v2 (D=64): K/V tiles staged once per BLOCK through double-buffered
// LDS (one barrier per tile; the next tile's stage write lands in the
// other slot under the current tile's compute).
// The v1 kernel's four waves each re-read every K/V tile from L2 (16B
// fragment loads + 32 scalar V loads per tile) and sat 38% of wave
// cycles in memory waits (PMC r01). Here 256 threads cooperatively load
// the 4KB K and V tiles, K stored with the ((row&15)<<4) XOR swizzle so
// the wave's ds_read_b128 fragment reads are bank-conflict-free (guide
// G4/T2: the b128 16-lane groups read rows distinct mod 16), V stored
// linear and consumed as 2B LDS reads (16 consecutive banks per group).
// ---------------------------------------------------------------------
template <bool TR16, bool DROP = false>
// TR16: V B-fragment reads via tr16 hardware-transpose gather vs eight
// 2-B reads (A/B, env-selected). DROP: attention-PROB dropout fused in
// (HF BertSelfAttention semantics): the per-(qrow,key) philox keep mask
// scales the UNNORMALIZED exp values fed to PV while l_run keeps the
// undropped softmax normalizer (so O = dropout(P) @ V and lse is the
// true logsumexp). Counter scheme shared with the bwd kernels: one
// philox gen per 2x2 (qrow,key) square — ctr = (bh<<32 | qrow>>1,
// key>>1), output index (qrow&1)*2 + (key&1); keep iff u >= p (same
// convention as csrc/dropout.hip), seed XORed with the device graph
// counter at run time.
__global__ __launch_bounds__(256, 4) void attn_fwd_lds_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const float* __restrict__ mask,
    bf16_t* __restrict__ o, float* __restrict__ lse, int S, int causal,
    float scale, int has_mask, long H,
    Strides sq, Strides sk, Strides sv, Strides so,
    float pdrop = 0.f, float inv_keep = 1.f,
    unsigned long long dseed = 0,
    const long long* __restrict__ seed_buf = nullptr,
    unsigned* __restrict__ mbits_out = nullptr) {
  constexpr int D = 64;
  // per slot: K swizzled 4KB + V linear 4KB (row stride 128B) or, for
  // the tr16 path, V padded to 144B rows (4608B) so the 4-lanes-per-row
  // group gather hits distinct bank pairs.
  constexpr int VSTRIDE = TR16 ? 144 : 128;
  constexpr int SLOT = 4096 + (TR16 ? 4608 : 4096);
  __shared__ __attribute__((aligned(16))) char smem[2 * SLOT];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int hi = lane >> 5;
  const int j32 = lane & 31;

  const int q0 = (blockIdx.x * 4 + wid) * 32;
  const long bh = blockIdx.y;
  const long b = bh / H, h = bh % H;
  const bf16_t* qp = q + b * sq.bs + h * sq.hs;
  const bf16_t* kp = k + b * sk.bs + h * sk.hs;
  const bf16_t* vp = v + b * sv.bs + h * sv.hs;
  bf16_t* op = o + b * so.bs + h * so.hs;
  const float* mp = has_mask ? (mask + b * (long)S) : nullptr;
  const bool live_wave = q0 < S;

  bf16x8v qf[4];
  const int qrow = q0 + j32;
  if (live_wave) {
    const bf16_t* qrp = qp + (long)min(qrow, S - 1) * sq.rs;
#pragma unroll
    for (int s = 0; s < 4; ++s)
      qf[s] = *reinterpret_cast<const bf16x8v*>(qrp + s * 16 + hi * 8);
  }

  f32x16 oacc[2];
#pragma unroll
  for (int hh = 0; hh < 2; ++hh)
#pragma unroll
    for (int r = 0; r < 16; ++r) oacc[hh][r] = 0.f;
  float m_run = -INFINITY, l_run = 0.f;

  // staging indices: thread t loads 16B of row (t>>3), col bytes (t&7)*16
  const int st_row = threadIdx.x >> 3;
  const int st_c16 = (threadIdx.x & 7) * 16;  // byte col
  const int k_dst = (st_row * 128 + st_c16) ^ ((st_row & 15) << 4);
  const int v_dst = st_row * VSTRIDE + st_c16;
  // ds_read_b64_tr_b16 semantics (verified by tr16_probe on gfx950):
  // each lane supplies its OWN 8-B address; per 16-lane group the 16
  // loads (lane q -> 4 contiguous bf16) form a [4][16] row-major block
  // M, and lane q receives column q (M[0..3][q]). For the PV B-fragment
  // lane l needs V[kbase+e][hh*32 + (l&31)], e=0..7 -> lane q=l&15 loads
  // row kbase+(q>>2), cols 4*(q&3) of the group's 16-col window.
  const int q4 = lane & 15;
  const int v_tr_base = (8 * hi + (q4 >> 2)) * VSTRIDE +
                        (((lane >> 4) & 1) * 16 + 4 * (q4 & 3)) * 2;

  // every wave must loop over ALL tiles (barriers are block-wide); a
  // wave past the causal horizon just skips its compute.
  // T14 split (guide G15): tile t+1's global loads ISSUE before tile t's
  // MFMA/softmax phase; double-buffered slots make the LDS write safe
  // after compute (the slot's last readers passed the barrier at the top
  // of THIS iteration).
  const int kv_all = causal ? min(S, blockIdx.x * 128 + 128) : S;
  // K tile rides global_load_lds (lane-linear dest; the XOR swizzle
  // becomes the closed-form source inverse — see attention_bwd.hip).
  // V cannot: its TR16 image uses 144-byte padded rows, which a
  // contiguous lane-linear DMA cannot produce, so V keeps the register
  // round-trip.
  const int wdst = ((int)threadIdx.x >> 6) * 1024;
  const int p16 = (int)threadIdx.x * 16;
  const int inv_r7 = ((p16 >> 7) & 1) ^ ((p16 >> 10) & 1);
  const int p_src = (p16 & ~0xF0) | ((((p16 >> 4) & 1) ^ inv_r7) << 4) |
                    ((((p16 >> 5) & 1) ^ ((p16 >> 8) & 1)) << 5) |
                    ((((p16 >> 6) & 1) ^ ((p16 >> 9) & 1)) << 6) |
                    (inv_r7 << 7);
  auto stage_k = [&](int kt, char* sl) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            reinterpret_cast<const char*>(kp) +
            (long)min(kt + (p_src >> 7), S - 1) * sk.rs * 2 + (p_src & 127)),
        (__attribute__((address_space(3))) void*)(sl + wdst), 16, 0, 0);
  };
  bf16x8v vstage = *reinterpret_cast<const bf16x8v*>(
      vp + (long)min(st_row, S - 1) * sv.rs + st_c16 / 2);
  stage_k(0, smem);  // prologue: stage tile 0 into slot 0
  *reinterpret_cast<bf16x8v*>(smem + 4096 + v_dst) = vstage;
  for (int k0 = 0; k0 < kv_all; k0 += 32) {
    const int slot = (k0 >> 5) & 1;
    char* k_lds = smem + slot * SLOT;
    char* v_lds = k_lds + 4096;
    __syncthreads();  // this slot staged; other slot's readers done
    const bool have_next = k0 + 32 < kv_all;
    if (have_next) {  // issue next tile's loads under this compute
      stage_k(k0 + 32, smem + (slot ^ 1) * SLOT);
      vstage = *reinterpret_cast<const bf16x8v*>(
          vp + (long)min(k0 + 32 + st_row, S - 1) * sv.rs + st_c16 / 2);
    }
    const bool compute = live_wave && (!causal || k0 <= q0 + 31);
    if (compute) {
      // ---- QK^T: K A-frags from swizzled LDS (ds_read_b128) ----
      f32x16 s_acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) s_acc[r] = 0.f;
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        const int koff = (j32 * 128 + (s * 16 + hi * 8) * 2) ^
                         ((j32 & 15) << 4);
        bf16x8v kf = *reinterpret_cast<const bf16x8v*>(k_lds + koff);
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], s_acc,
                                                        0, 0, 0);
      }
      // additive mask: ONE lane-load for the tile's 32 keys, values
      // redistributed by wave shuffle (vs 16 scattered loads per lane)
      float mload = 0.f;
      if (has_mask) mload = mp[min(k0 + j32, S - 1)];
      float tile_max = -INFINITY;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int key = k0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float sv2 = s_acc[r] * scale;
        if (key >= S) sv2 = -INFINITY;
        if (causal && key > qrow) sv2 = -INFINITY;
        if (has_mask) sv2 += __shfl(mload, key - k0, WAVE);
        s_acc[r] = sv2;
        tile_max = fmaxf(tile_max, sv2);
      }
      tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, WAVE));
      // T13 defer-max (guide §5.5): skip the whole rescale when this
      // tile's max is within THR of the running max — P values are then
      // bounded by e^THR instead of 1, which the fp32 accumulator
      // tolerates (measured accuracy cost ~3x vs THR=0, still inside
      // the bf16 output rounding class; forced-branch test in
      // tests/test_kernels_gpu.py). The exponentiation below happens
      // AFTER this decision (the safe textbook order).
      constexpr float DEFER_THR = 8.f;
      float alpha = 1.f;
      if (!__all(tile_max - m_run <= DEFER_THR)) {
        const float m_new = fmaxf(m_run, tile_max);
        if (m_new != m_run) {
          alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
          m_run = m_new;
        }
      }
      l_run *= alpha;
      // skip the 32-shfl/32-mul O-rescale whenever NO row's running
      // max moved this tile (the common case once maxima settle)
      if (__any(alpha != 1.f)) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int row_local = (r & 3) + 8 * (r >> 2) + 4 * hi;
          const float a_r = __shfl(alpha, row_local, WAVE);
#pragma unroll
          for (int hh = 0; hh < 2; ++hh) oacc[hh][r] *= a_r;
        }
      }
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float pv = (s_acc[r] == -INFINITY) ? 0.f : __expf(s_acc[r] - m_run);
        s_acc[r] = pv;
        psum += pv;
      }
      psum += __shfl_xor(psum, 32, WAVE);
      l_run += psum;

      if constexpr (DROP) {
        const Philox4 ph(seed_buf
                             ? dseed ^ (unsigned long long)*seed_buf
                             : dseed);
        const long bh_ctr = ((unsigned long long)bh << 32) |
                            (unsigned)(qrow >> 1);
        const int qb = (qrow & 1) << 1;
        unsigned mloc = 0;
#pragma unroll
        for (int rq = 0; rq < 8; ++rq) {
          const int r0 = 2 * rq;
          const int key0 = k0 + (r0 & 3) + 8 * (r0 >> 2) + 4 * hi;  // even
          unsigned int rr[4];
          ph.gen((unsigned long long)bh_ctr,
                 (unsigned long long)(key0 >> 1), rr);
          if ((rr[qb] >> 8) * (1.0f / 16777216.0f) >= pdrop)
            mloc |= 1u << r0;
          if ((rr[qb | 1] >> 8) * (1.0f / 16777216.0f) >= pdrop)
            mloc |= 1u << (r0 + 1);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r)
          s_acc[r] *= ((mloc >> r) & 1) ? inv_keep : 0.f;
        // publish the 32-key word for the backward kernels: bit
        // (key - k0) of this qrow; lane pairs (hi halves) interleave
        // 4-key nibbles, merged via permlane-free shfl
        unsigned w_half = 0;
#pragma unroll
        for (int r = 0; r < 16; ++r)
          w_half |= ((mloc >> r) & 1)
                    << ((r & 3) + 8 * (r >> 2) + 4 * hi);
        const unsigned w32 = w_half | __shfl_xor(w_half, 32, WAVE);
        if (hi == 0 && qrow < S && mbits_out)
          mbits_out[(bh * (long)S + qrow) * ((S + 31) >> 5) + (k0 >> 5)] =
              w32;
      }

      unsigned int pa[2][4];
#pragma unroll
      for (int step = 0; step < 2; ++step) {
        const int b0 = step * 8;
        unsigned int c01 = pack_bf2(s_acc[b0 + 0], s_acc[b0 + 1]);
        unsigned int c23 = pack_bf2(s_acc[b0 + 2], s_acc[b0 + 3]);
        unsigned int c45 = pack_bf2(s_acc[b0 + 4], s_acc[b0 + 5]);
        unsigned int c67 = pack_bf2(s_acc[b0 + 6], s_acc[b0 + 7]);
        {
          auto r2 = __builtin_amdgcn_permlane32_swap(c01, c45, false, false);
          pa[step][0] = r2[0];
          pa[step][2] = r2[1];
        }
        {
          auto r2 = __builtin_amdgcn_permlane32_swap(c23, c67, false, false);
          pa[step][1] = r2[0];
          pa[step][3] = r2[1];
        }
      }
      // ---- PV: V B-frags. TR16: two ds_read_b64_tr_b16 per fragment
      // (T10) replace eight 2-B reads (16-bank groups) ----
#pragma unroll
      for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
        for (int step = 0; step < 2; ++step) {
          bf16x8v vf;
          if constexpr (TR16) {
            char* vtp = v_lds + v_tr_base + step * 16 * VSTRIDE + hh * 64;
            bf16x4v lo = DS_TR16(vtp);
            bf16x4v hi4 = DS_TR16(vtp + 4 * VSTRIDE);
#pragma unroll
            for (int e = 0; e < 4; ++e) {
              vf[e] = lo[e];
              vf[e + 4] = hi4[e];
            }
          } else {
#pragma unroll
            for (int e = 0; e < 8; ++e) {
              const int key_l = step * 16 + hi * 8 + e;
              vf[e] = *reinterpret_cast<const bf16_t*>(
                  v_lds + key_l * VSTRIDE + (hh * 32 + j32) * 2);
            }
          }
          oacc[hh] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<const bf16x8v*>(&pa[step][0]), vf,
              oacc[hh], 0, 0, 0);
        }
      }
    }
    if (have_next) {  // stage V t+1 into the other slot (safe: its last
                      // readers passed this iteration's barrier; K's
                      // DMA was issued at the top of this iteration)
      char* nk = smem + (slot ^ 1) * SLOT;
      *reinterpret_cast<bf16x8v*>(nk + 4096 + v_dst) = vstage;
    }
  }

  if (!live_wave) return;
  const float l_safe = (l_run > 0.f) ? l_run : 1.f;
#pragma unroll
  for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row_local = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const float l_row = __shfl(l_safe, row_local, WAVE);
      const int row = q0 + row_local;
      if (row < S)
        op[(long)row * so.rs + hh * 32 + j32] = f2bf(oacc[hh][r] / l_row);
    }
  }
  if (qrow < S && hi == 0)
    lse[bh * (long)S + qrow] =
        (l_run > 0.f) ? m_run + __logf(l_run) : -INFINITY;
}

// ---- layout probe: one wave computes one 32x32x16 tile --------------
__global__ void mfma_probe_kernel(const bf16_t* __restrict__ a,
                                  const bf16_t* __restrict__ b,
                                  float* __restrict__ d) {
  if (threadIdx.x >= 64) return;
  const int lane = threadIdx.x;
  const int hi = lane >> 5;
  bf16x8v af, bf;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    // A[i][k] i=lane&31, k=hi*8+e ; A stored row-major 32x16
    af[e] = *reinterpret_cast<const __bf16*>(a + (lane & 31) * 16 + hi * 8 + e);
    // B[k][j] j=lane&31, k=hi*8+e ; B stored row-major 16x32
    bf[e] = *reinterpret_cast<const __bf16*>(b + (hi * 8 + e) * 32 + (lane & 31));
  }
  f32x16 acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    d[row * 32 + (lane & 31)] = acc[r];
  }
}

}  // namespace

static const bool use_lds_fwd = [] {
  const char* e = getenv("RAVNEST_ATTN_FWD_V1");
  return !(e && e[0] == '1');  // default: the LDS kernel for D=64
}();

static const bool use_tr16 = [] {
  const char* e = getenv("RAVNEST_ATTN_TR16");
  return !(e && e[0] == '0');  // default: hardware-transpose V reads
}();

std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor mask, bool causal, double scale) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
              v.is_contiguous());
  TORCH_CHECK(q.dim() == 4, "q must be (B,H,S,D)");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16,
              "attn_fwd: bf16 only (autocast provides bf16)");
  const long B = q.size(0), H = q.size(1);
  const int S = q.size(2), D = q.size(3);
  TORCH_CHECK(D == 64 || D == 128, "attn_fwd: head_dim 64 or 128");

  auto o = at::empty_like(q);
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));

  const bool has_mask = mask.defined() && mask.numel() > 0;
  at::Tensor mask_f;
  const float* mask_ptr = nullptr;
  if (has_mask) {
    mask_f = mask.to(at::kFloat).reshape({B, S}).contiguous();
    mask_ptr = mask_f.data_ptr<float>();
  }

  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid((S + 127) / 128, B * H);
  const Strides sep{H * (long)S * D, (long)S * D, (long)D};
  const Strides so = sep;

#define LAUNCH_ATTN_FWD(DD, QP, KP, VP, OP, SQ, SK, SV, SO)                 \
  do {                                                                      \
    if (DD == 64 && use_lds_fwd && use_tr16)                                \
      hipLaunchKernelGGL((attn_fwd_lds_kernel<true>), grid, block, 0,       \
                         stream, QP, KP, VP, mask_ptr,                      \
                         OP, lse.data_ptr<float>(), S,                      \
                         causal ? 1 : 0, (float)scale, has_mask ? 1 : 0,    \
                         (long)H, SQ, SK, SV, SO);                          \
    else if (DD == 64 && use_lds_fwd)                                       \
      hipLaunchKernelGGL((attn_fwd_lds_kernel<false>), grid, block, 0,      \
                         stream, QP,                                        \
                         KP, VP, mask_ptr, OP, lse.data_ptr<float>(), S,    \
                         causal ? 1 : 0, (float)scale, has_mask ? 1 : 0,    \
                         (long)H, SQ, SK, SV, SO);                          \
    else                                                                    \
      hipLaunchKernelGGL((attn_fwd_kernel<DD>), grid, block, 0, stream, QP, \
                         KP, VP, mask_ptr, OP, lse.data_ptr<float>(), S,    \
                         causal ? 1 : 0, (float)scale, has_mask ? 1 : 0,    \
                         (long)H, SQ, SK, SV, SO);                          \
  } while (0)

  if (D == 64) {
    LAUNCH_ATTN_FWD(64, reinterpret_cast<const bf16_t*>(q.data_ptr()),
                    reinterpret_cast<const bf16_t*>(k.data_ptr()),
                    reinterpret_cast<const bf16_t*>(v.data_ptr()),
                    reinterpret_cast<bf16_t*>(o.data_ptr()), sep, sep, sep,
                    so);
  } else {
    LAUNCH_ATTN_FWD(128, reinterpret_cast<const bf16_t*>(q.data_ptr()),
                    reinterpret_cast<const bf16_t*>(k.data_ptr()),
                    reinterpret_cast<const bf16_t*>(v.data_ptr()),
                    reinterpret_cast<bf16_t*>(o.data_ptr()), sep, sep, sep,
                    so);
  }
  HIP_CHECK_LAST();
  return {o, lse};
}

// packed layout: qkv (B,S,3,H,D) contiguous -> o (B,S,H,D); no
// permute/contiguous copies around the attention core.
std::vector<at::Tensor> attn_fwd_qkv(at::Tensor qkv, at::Tensor mask,
                                     bool causal, double scale,
                                     double pdrop, int64_t dseed,
                                     c10::optional<at::Tensor> seed_buf) {
  TORCH_CHECK(qkv.is_cuda() && qkv.is_contiguous());
  TORCH_CHECK(qkv.dim() == 5 && qkv.size(2) == 3,
              "qkv must be (B,S,3,H,D)");
  TORCH_CHECK(qkv.scalar_type() == at::kBFloat16);
  const long B = qkv.size(0), H = qkv.size(3);
  const int S = qkv.size(1), D = qkv.size(4);
  TORCH_CHECK(D == 64 || D == 128, "attn_fwd_qkv: head_dim 64 or 128");

  auto o = at::empty({B, (long)S, H, (long)D}, qkv.options());
  auto lse = at::empty({B, H, S}, qkv.options().dtype(at::kFloat));

  const bool has_mask = mask.defined() && mask.numel() > 0;
  at::Tensor mask_f;
  const float* mask_ptr = nullptr;
  if (has_mask) {
    mask_f = mask.to(at::kFloat).reshape({B, S}).contiguous();
    mask_ptr = mask_f.data_ptr<float>();
  }

  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 block(256);
  dim3 grid((S + 127) / 128, B * H);
  const long HD = H * (long)D;
  const Strides sp{(long)S * 3 * HD, (long)D, 3 * HD};
  const Strides so{(long)S * HD, (long)D, HD};
  const bf16_t* base = reinterpret_cast<const bf16_t*>(qkv.data_ptr());

  if (pdrop > 0.0) {
    TORCH_CHECK(D == 64 && use_lds_fwd && use_tr16,
                "attn_fwd_qkv: fused prob-dropout needs the D=64 LDS+tr16 "
                "path");
    const long long* sb = nullptr;
    if (seed_buf.has_value()) sb = reinterpret_cast<const long long*>(seed_buf->data_ptr<int64_t>());
    // per-(bh,qrow) 32-key mask words, read back by the bwd kernels
    auto mbits = at::empty({B * H * (long)S, (long)((S + 31) / 32)},
                           qkv.options().dtype(at::kInt));
    hipLaunchKernelGGL((attn_fwd_lds_kernel<true, true>), grid, block, 0,
                       stream, base, base + HD, base + 2 * HD, mask_ptr,
                       reinterpret_cast<bf16_t*>(o.data_ptr()),
                       lse.data_ptr<float>(), S, causal ? 1 : 0,
                       (float)scale, has_mask ? 1 : 0, (long)H, sp, sp, sp,
                       so, (float)pdrop, 1.f / (1.f - (float)pdrop),
                       (unsigned long long)dseed, sb,
                       reinterpret_cast<unsigned*>(mbits.data_ptr<int>()));
    HIP_CHECK_LAST();
    return {o, lse, mbits};
  } else if (D == 64) {
    LAUNCH_ATTN_FWD(64, base, base + HD, base + 2 * HD,
                    reinterpret_cast<bf16_t*>(o.data_ptr()), sp, sp, sp, so);
  } else {
    LAUNCH_ATTN_FWD(128, base, base + HD, base + 2 * HD,
                    reinterpret_cast<bf16_t*>(o.data_ptr()), sp, sp, sp, so);
  }
  HIP_CHECK_LAST();
  return {o, lse};
}

// ---- MX-scaled fp8 MFMA probe (gfx950 f8f6f4 path seed) -------------
// v_mfma_scale_f32_32x32x64_f8f6f4: A 32x64 fp8 (8 VGPR/lane = 32 vals),
// B 64x32 fp8, C/D f32x16 (same C layout as bf16 32x32). Assumed input
// layout (extends the verified bf16 pattern): lane l holds
// A[i=l&31][k=(l>>5)*32 + e], e=0..31 (bytes of 8 dwords), B mirrored.
// Scales: e8m0 bytes, 127 = 1.0 (identity) — per-32-block scale operand.
namespace {
typedef int i32x8 __attribute__((ext_vector_type(8)));

__global__ void mfma_mx_probe_kernel(const unsigned char* __restrict__ a,
                                     const unsigned char* __restrict__ b,
                                     float* __restrict__ d, int sa, int sb) {
  if (threadIdx.x >= 64) return;
  const int lane = threadIdx.x;
  const int hi = lane >> 5;
  i32x8 af, bf;
  unsigned char* afb = reinterpret_cast<unsigned char*>(&af);
  unsigned char* bfb = reinterpret_cast<unsigned char*>(&bf);
#pragma unroll
  for (int e = 0; e < 32; ++e) {
    // A stored row-major 32x64; B stored 64x32 col-read: B[k][j]
    afb[e] = a[(lane & 31) * 64 + hi * 32 + e];
    bfb[e] = b[(hi * 32 + e) * 32 + (lane & 31)];
  }
  f32x16 acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
      af, bf, acc, 0, 0, 0, sa, 0, sb);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    d[row * 32 + (lane & 31)] = acc[r];
  }
}
}  // namespace

at::Tensor mfma_mx_probe(at::Tensor a, at::Tensor b, int64_t sa, int64_t sb) {
  // a: (32,64) uint8 fp8-e4m3 bytes; b: (64,32) uint8
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kByte);
  TORCH_CHECK(a.sizes() == at::IntArrayRef({32, 64}));
  TORCH_CHECK(b.sizes() == at::IntArrayRef({64, 32}));
  auto d = at::zeros({32, 32}, a.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(mfma_mx_probe_kernel, dim3(1), dim3(64), 0, stream,
                     a.data_ptr<unsigned char>(), b.data_ptr<unsigned char>(),
                     d.data_ptr<float>(), (int)sa, (int)sb);
  HIP_CHECK_LAST();
  return d;
}

namespace {
// empirical mapping probe for ds_read_b64_tr_b16: LDS filled with
// identity values; each lane passes addr = base + lane*8B; dump outputs.
__global__ void tr16_probe_kernel(short* __restrict__ out) {
  __shared__ __bf16 buf[512];
  if (threadIdx.x < 64) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      // store the element INDEX as raw bf16 bits
      reinterpret_cast<short*>(buf)[threadIdx.x * 8 + j] =
          (short)(threadIdx.x * 8 + j);
    }
  }
  __syncthreads();
  if (threadIdx.x >= 64) return;
  bf16x4v r = DS_TR16(reinterpret_cast<char*>(buf) + threadIdx.x * 8);
#pragma unroll
  for (int j = 0; j < 4; ++j)
    out[threadIdx.x * 4 + j] = reinterpret_cast<short*>(&r)[j];
}
}  // namespace

at::Tensor tr16_probe() {
  auto out = at::zeros({64, 4},
                       at::TensorOptions().dtype(at::kShort).device(at::kCUDA));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     out.data_ptr<short>());
  HIP_CHECK_LAST();
  return out;
}

at::Tensor mfma_probe(at::Tensor a, at::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.sizes() == at::IntArrayRef({32, 16}));
  TORCH_CHECK(b.sizes() == at::IntArrayRef({16, 32}));
  auto d = at::zeros({32, 32}, a.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     reinterpret_cast<const bf16_t*>(a.data_ptr()),
                     reinterpret_cast<const bf16_t*>(b.data_ptr()),
                     d.data_ptr<float>());
  HIP_CHECK_LAST();
  return d;
}
