// Flash-style fused attention BACKWARD for CDNA4 (gfx950), bf16, D=64.
//
// Three kernels (flash2-style split, no atomics):
//   attn_bwd_delta: Delta[row] = dO[row] . O[row]            (rowwise)
//   attn_bwd_dkv:   each wave owns a 32-KEY tile, loops Q tiles,
//                   recomputes P from (Q,K,lse), accumulates dV,dK
//   attn_bwd_dq:    each wave owns a 32-ROW Q tile, loops KV tiles,
//                   recomputes P^T, accumulates dQ
//
// Both matmul kernels reuse the forward's fragment algebra
// (attention.hip header comment): the MFMA C layout [row-pattern][col =
// lane&31] is turned into the next MFMA's A operand by bf16 pair packing
// + __builtin_amdgcn_permlane32_swap (guide T12/T21 primitives).
//
// dS_ij = P_ij * (dP_ij - Delta_i) * scale;  dV = P^T dO; dK = dS^T Q;
// dQ = dS K.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8v __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4v __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

// hardware transpose read (guide T10; semantics verified by tr16_probe in
// attention.hip): each lane loads 4 contiguous bf16 at its OWN 8-B
// address; per 16-lane group the loads form a [4][16] block M (lane
// order) and lane q receives column q.
#define DS_TR16(p) __builtin_amdgcn_ds_read_tr16_b64_v4bf16( \
    (__attribute__((address_space(3))) bf16x4v*)( \
        (__attribute__((address_space(3))) void*)(p)))

namespace {

// fused attention-prob dropout: the fwd kernel published one 32-key
// mask word per (bh, qrow) k-block (bit = key - k0, philox drawn once
// in the fwd); the bwd kernels READ those words instead of re-running
// philox (regenerating measured ~4x the masking cost). Two readers for
// the two lane layouts: key fixed per lane (dv/dk) and qrow fixed per
// lane (dq).
DEVINL unsigned drop_bits_keyfixed(const unsigned* __restrict__ mb,
                                   long bh, int S, int q0, int hi,
                                   int key, int k0) {
  // lane l loads the word of qrow q0 + (l&31); each lane then gathers
  // its 16 r-elements' bits via wave shuffles (wave-uniform branch)
  const int lane = threadIdx.x & (WAVE - 1);
  const int wpr = (S + 31) >> 5;
  const int myrow = min(q0 + (lane & 31), S - 1);
  const unsigned w = mb[(bh * (long)S + myrow) * wpr + (k0 >> 5)];
  unsigned bits = 0;
  const int kb = key - k0;  // bit position within the word (0..31)
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qoff = (r & 3) + 8 * (r >> 2) + 4 * hi;  // qrow - q0
    const unsigned wr = __shfl(w, qoff, WAVE);
    bits |= ((wr >> kb) & 1) << r;
  }
  return bits;
}

DEVINL unsigned drop_bits_rowfixed(const unsigned* __restrict__ mb,
                                   long bh, int S, int k0, int hi,
                                   int qrow) {
  const int wpr = (S + 31) >> 5;
  const unsigned w =
      mb[(bh * (long)S + min(qrow, S - 1)) * wpr + (k0 >> 5)];
  unsigned bits = 0;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kb = (r & 3) + 8 * (r >> 2) + 4 * hi;  // key - k0
    bits |= ((w >> kb) & 1) << r;
  }
  return bits;
}


// element offset of (b, h, row) = b*bs + h*hs + row*rs (see attention.hip)
struct StridesB { long bs, hs, rs; };

DEVINL unsigned int pack_bf2b(float lo, float hi) {
  return (unsigned int)f2us(lo) | ((unsigned int)f2us(hi) << 16);
}

// transform a [row-pattern][col=lane] f32 acc (16 regs) into the A-operand
// fragments covering k=rows: pa[step] holds rows 16*step + (lane>>5)*8+e.
// per-lane byte offset for a tr16 B-fragment read of
// B[k = kbase + 0..3][j = hh*32 + (lane&31)] from a 128-B-row LDS image,
// optionally XOR-swizzled ((row&15)<<4, 16-B granularity — an 8-B-aligned
// read never crosses the flipped boundary). The row-dependent XOR also
// de-conflicts the 4-lanes-per-row gather (rows r and r+2 land 32 B
// apart), so the swizzled images need no extra padding.
template <bool SWZ>
DEVINL int tr16_b_off(int lane, int kbase, int hh) {
  const int q4 = lane & 15;
  const int row = kbase + (q4 >> 2);
  int off = row * 128 + hh * 64 + ((lane >> 4) & 1) * 32 + 8 * (q4 & 3);
  if (SWZ) off ^= (row & 15) << 4;
  return off;
}

DEVINL void acc_to_afrag(const f32x16& acc, unsigned int pa[2][4]) {
#pragma unroll
  for (int step = 0; step < 2; ++step) {
    const int b0 = step * 8;
    unsigned int c01 = pack_bf2b(acc[b0 + 0], acc[b0 + 1]);
    unsigned int c23 = pack_bf2b(acc[b0 + 2], acc[b0 + 3]);
    unsigned int c45 = pack_bf2b(acc[b0 + 4], acc[b0 + 5]);
    unsigned int c67 = pack_bf2b(acc[b0 + 6], acc[b0 + 7]);
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
}

// Delta[row] = sum_d dO[row][d] * O[row][d]. D/8 lanes per row, bf16x8
// loads (the one-wave-per-row 2-B-scalar version measured 7x off the
// bandwidth bound at 2.5% of the BERT step — rocprofv3 r01 final).
template <int D>
__global__ void attn_bwd_delta_kernel(const bf16_t* __restrict__ dout,
                                      const bf16_t* __restrict__ o,
                                      float* __restrict__ delta, int S,
                                      long H, long NR, StridesB sdo) {
  constexpr int LPR = D / 8;  // lanes per row
  const int lane = threadIdx.x & (WAVE - 1);
  const long t = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long id = t / LPR;
  if (id >= NR) return;
  const long bh = id / S;
  const long row = id % S;
  const long off = (bh / H) * sdo.bs + (bh % H) * sdo.hs + row * sdo.rs +
                   (t % LPR) * 8;
  bf16x8v a = *reinterpret_cast<const bf16x8v*>(dout + off);
  bf16x8v b = *reinterpret_cast<const bf16x8v*>(o + off);
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) s += (float)a[j] * (float)b[j];
#pragma unroll
  for (int m = 1; m < LPR; m <<= 1) s += __shfl_xor(s, m, WAVE);
  if ((lane & (LPR - 1)) == 0) delta[id] = s;
}

// -------------------------------------------------------------------
// dv kernel: wave owns keys; S -> P -> dV only (~110 VGPR, 4 waves/SIMD;
// the combined dkv kernel sat at 180 VGPR = 2 waves/SIMD and 56% of wave
// cycles parked on memory waits — PMC profile r01)
// -------------------------------------------------------------------
template <bool DROP = false>
__global__ __launch_bounds__(256, DROP ? 4 : 5) void attn_bwd_dv_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ mask, bf16_t* __restrict__ dv, int S,
    int causal, float scale, int has_mask, long H, StridesB sio,
    StridesB sdo, StridesB sg, float inv_keep = 1.f,
    const unsigned* __restrict__ mbits = nullptr) {
  constexpr int D = 64;
  // block-shared tiles: dO (linear: 2B B-operand reads) and Q
  // (((row&15)<<4)-swizzled: conflict-free b128 A-fragment reads).
  // After staging, a tile iteration does NO per-wave global reads;
  // lse staged once (same rationale as dk).
  __shared__ __attribute__((aligned(16))) char smem[16384 + 8192];
  float* lse_lds = reinterpret_cast<float*>(smem + 16384);
  // two 8KB slots: [slot][dO linear 4KB | Q swz 4KB]
  const int st_row = threadIdx.x >> 3;
  const int st_c16 = (threadIdx.x & 7) * 16;
  const int lane = threadIdx.x & (WAVE - 1);
  const int hi = lane >> 5;
  const int j32 = lane & 31;
  const int k0 = (blockIdx.x * 4 + (int)(threadIdx.x / WAVE)) * 32;
  const bool live_wave = k0 < S;
  const long bh = blockIdx.y;
  const long b = bh / H, h = bh % H;
  const bf16_t* qp = q + b * sio.bs + h * sio.hs;
  const bf16_t* kp = k + b * sio.bs + h * sio.hs;
  const bf16_t* dop = dout + b * sdo.bs + h * sdo.hs;
  bf16_t* dvp = dv + b * sg.bs + h * sg.hs;
  const float* lsep = lse + bh * (long)S;
  const float* mp = has_mask ? (mask + b * (long)S) : nullptr;
  const bool lse_in_lds = S <= 2048;
  if (lse_in_lds) {
    for (int i = threadIdx.x; i < S; i += blockDim.x) lse_lds[i] = lsep[i];
  }

  const int key = k0 + j32;
  const bf16_t* krp = kp + (long)min(key, S - 1) * sio.rs;
  bf16x8v kf[4];
  if (live_wave) {
#pragma unroll
    for (int s = 0; s < 4; ++s)
      kf[s] = *reinterpret_cast<const bf16x8v*>(krp + s * 16 + hi * 8);
  }
  const float mask_val = (has_mask && key < S) ? mp[min(key, S - 1)] : 0.f;

  f32x16 dv_acc[2];
#pragma unroll
  for (int hh = 0; hh < 2; ++hh)
#pragma unroll
    for (int r = 0; r < 16; ++r) dv_acc[hh][r] = 0.f;

  const int q_start = causal ? blockIdx.x * 128 : 0;
  // glds staging (LDS-DMA, rule 21): dest is lane-linear, so dO (linear
  // image) gldses with the plain addressing and Q's XOR swizzle moves
  // to the SOURCE side via the closed-form inverse of
  // p ^ (((p>>7)&15)<<4) (verified bijective + 16B-aligned). Replaces
  // the register round-trip (2 global loads + 2 ds_writes per thread
  // per tile) and frees the 16 staging VGPRs; the loop-top
  // __syncthreads drains the in-flight DMA (vmcnt) before first read.
  const int wdst = ((int)threadIdx.x >> 6) * 1024;
  const int p16 = (int)threadIdx.x * 16;
  const int inv_r7 = ((p16 >> 7) & 1) ^ ((p16 >> 10) & 1);
  const int p_src = (p16 & ~0xF0) | ((((p16 >> 4) & 1) ^ inv_r7) << 4) |
                    ((((p16 >> 5) & 1) ^ ((p16 >> 8) & 1)) << 5) |
                    ((((p16 >> 6) & 1) ^ ((p16 >> 9) & 1)) << 6) |
                    (inv_r7 << 7);
  const int s_row = p_src >> 7, s_col = p_src & 127;
  auto stage2 = [&](int qt, char* sl) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            reinterpret_cast<const char*>(dop) +
            (long)min(qt + st_row, S - 1) * sdo.rs * 2 + st_c16),
        (__attribute__((address_space(3))) void*)(sl + wdst), 16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            reinterpret_cast<const char*>(qp) +
            (long)min(qt + s_row, S - 1) * sio.rs * 2 + s_col),
        (__attribute__((address_space(3))) void*)(sl + 4096 + wdst),
        16, 0, 0);
  };
  stage2(q_start, smem);
  for (int q0 = q_start; q0 < S; q0 += 32) {
    const int slot = ((q0 - q_start) >> 5) & 1;
    const char* do_lds = smem + slot * 8192;
    const char* q_lds = do_lds + 4096;
    __syncthreads();
    const bool have_next = q0 + 32 < S;
    if (have_next)  // T14: next tile's LDS-DMA under this compute
      stage2(q0 + 32, smem + (slot ^ 1) * 8192);
    const bool compute = live_wave && (!causal || q0 + 31 >= k0);
    if (compute) {
    f32x16 s_acc;
#pragma unroll
    for (int r = 0; r < 16; ++r) s_acc[r] = 0.f;
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int qoff = (j32 * 128 + (s * 16 + hi * 8) * 2) ^
                       ((j32 & 15) << 4);
      bf16x8v qf = *reinterpret_cast<const bf16x8v*>(q_lds + qoff);
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kf[s], s_acc,
                                                      0, 0, 0);
    }
    unsigned dbits = 0;
    if constexpr (DROP)
      dbits = drop_bits_keyfixed(mbits, bh, S, q0, hi, key, k0);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = q0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      const int qclmp = min(qrow, S - 1);
      const float l = lse_in_lds ? lse_lds[qclmp] : lsep[qclmp];
      float sv = s_acc[r] * scale + mask_val;
      bool dead = (qrow >= S) || (key >= S) || (causal && key > qrow) ||
                  !isfinite(l);
      s_acc[r] = dead ? 0.f : __expf(sv - l);  // P
      if constexpr (DROP)  // dV uses the DROPPED probabilities
        s_acc[r] *= ((dbits >> r) & 1) ? inv_keep : 0.f;
    }
    unsigned int pa_p[2][4];
    acc_to_afrag(s_acc, pa_p);
#pragma unroll
    for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
      for (int step = 0; step < 2; ++step) {
        bf16x8v dof;
        {
          const int kb = step * 16 + 8 * hi;
          bf16x4v lo = DS_TR16(do_lds + tr16_b_off<false>(lane, kb, hh));
          bf16x4v h4 = DS_TR16(do_lds + tr16_b_off<false>(lane, kb + 4, hh));
#pragma unroll
          for (int e = 0; e < 4; ++e) { dof[e] = lo[e]; dof[e + 4] = h4[e]; }
        }
        dv_acc[hh] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            *reinterpret_cast<const bf16x8v*>(&pa_p[step][0]), dof,
            dv_acc[hh], 0, 0, 0);
      }
    }
    }  // compute
  }
  if (!live_wave) return;
#pragma unroll
  for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krow = k0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (krow < S)
        dvp[(long)krow * sg.rs + hh * 32 + j32] = f2bf(dv_acc[hh][r]);
    }
  }
}

// -------------------------------------------------------------------
// dk kernel: wave owns keys; S, dP -> dS -> dK (~150 VGPR, 3 waves/SIMD)
// -------------------------------------------------------------------
template <bool DROP = false>
__global__ __launch_bounds__(256, DROP ? 3 : 4) void attn_bwd_dk_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    const float* __restrict__ mask, bf16_t* __restrict__ dk, int S,
    int causal, float scale, int has_mask, long H, StridesB sio,
    StridesB sdo, StridesB sg, float inv_keep = 1.f,
    const unsigned* __restrict__ mbits = nullptr) {
  constexpr int D = 64;
  // Q tile swizzled (b128 A-frags for S, XOR-adjusted u16 B-reads for
  // dK) + dO tile swizzled (b128 A-frags for dP); double-buffered.
  // + lse/delta staged once (PMC: the per-reg L2 broadcast loads of
  // lse/delta every tile kept 58% of wave cycles parked)
  __shared__ __attribute__((aligned(16))) char smem[16384 + 16384];
  float* lse_lds = reinterpret_cast<float*>(smem + 16384);
  float* dlt_lds = lse_lds + 2048;
  const int st_row = threadIdx.x >> 3;
  const int st_c16 = (threadIdx.x & 7) * 16;
  const int lane = threadIdx.x & (WAVE - 1);
  const int hi = lane >> 5;
  const int j32 = lane & 31;
  const int k0 = (blockIdx.x * 4 + (int)(threadIdx.x / WAVE)) * 32;
  const bool live_wave = k0 < S;
  const long bh = blockIdx.y;
  const long b = bh / H, h = bh % H;
  const bf16_t* qp = q + b * sio.bs + h * sio.hs;
  const bf16_t* kp = k + b * sio.bs + h * sio.hs;
  const bf16_t* vp = v + b * sio.bs + h * sio.hs;
  const bf16_t* dop = dout + b * sdo.bs + h * sdo.hs;
  bf16_t* dkp = dk + b * sg.bs + h * sg.hs;
  const float* lsep = lse + bh * (long)S;
  const float* dltp = delta + bh * (long)S;
  const float* mp = has_mask ? (mask + b * (long)S) : nullptr;
  const bool lse_in_lds = S <= 2048;
  if (lse_in_lds) {
    for (int i = threadIdx.x; i < S; i += blockDim.x) {
      lse_lds[i] = lsep[i];
      dlt_lds[i] = dltp[i];
    }
  }

  const int key = k0 + j32;
  const bf16_t* krp = kp + (long)min(key, S - 1) * sio.rs;
  const bf16_t* vrp = vp + (long)min(key, S - 1) * sio.rs;
  bf16x8v kf[4], vf[4];
  if (live_wave) {
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      kf[s] = *reinterpret_cast<const bf16x8v*>(krp + s * 16 + hi * 8);
      vf[s] = *reinterpret_cast<const bf16x8v*>(vrp + s * 16 + hi * 8);
    }
  }
  const float mask_val = (has_mask && key < S) ? mp[min(key, S - 1)] : 0.f;

  f32x16 dk_acc[2];
#pragma unroll
  for (int hh = 0; hh < 2; ++hh)
#pragma unroll
    for (int r = 0; r < 16; ++r) dk_acc[hh][r] = 0.f;

  const int q_start = causal ? blockIdx.x * 128 : 0;
  // glds staging: both tiles are XOR-swizzled images, so both use the
  // source-side inverse permutation (see the dv kernel's comment)
  const int wdst = ((int)threadIdx.x >> 6) * 1024;
  const int p16 = (int)threadIdx.x * 16;
  const int inv_r7 = ((p16 >> 7) & 1) ^ ((p16 >> 10) & 1);
  const int p_src = (p16 & ~0xF0) | ((((p16 >> 4) & 1) ^ inv_r7) << 4) |
                    ((((p16 >> 5) & 1) ^ ((p16 >> 8) & 1)) << 5) |
                    ((((p16 >> 6) & 1) ^ ((p16 >> 9) & 1)) << 6) |
                    (inv_r7 << 7);
  const int s_row = p_src >> 7, s_col = p_src & 127;
  auto stage2 = [&](int qt, char* sl) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            reinterpret_cast<const char*>(qp) +
            (long)min(qt + s_row, S - 1) * sio.rs * 2 + s_col),
        (__attribute__((address_space(3))) void*)(sl + wdst), 16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            reinterpret_cast<const char*>(dop) +
            (long)min(qt + s_row, S - 1) * sdo.rs * 2 + s_col),
        (__attribute__((address_space(3))) void*)(sl + 4096 + wdst),
        16, 0, 0);
  };
  stage2(q_start, smem);
  for (int q0 = q_start; q0 < S; q0 += 32) {
    const int slot = ((q0 - q_start) >> 5) & 1;
    const char* q_lds = smem + slot * 8192;
    const char* do_lds = q_lds + 4096;
    __syncthreads();
    const bool have_next = q0 + 32 < S;
    if (have_next)  // T14: next tile's LDS-DMA under this compute
      stage2(q0 + 32, smem + (slot ^ 1) * 8192);
    const bool compute = live_wave && (!causal || q0 + 31 >= k0);
    if (compute) {
    f32x16 s_acc, dp_acc;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      s_acc[r] = 0.f;
      dp_acc[r] = 0.f;
    }
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int foff = (j32 * 128 + (s * 16 + hi * 8) * 2) ^
                       ((j32 & 15) << 4);
      bf16x8v qf = *reinterpret_cast<const bf16x8v*>(q_lds + foff);
      bf16x8v dof = *reinterpret_cast<const bf16x8v*>(do_lds + foff);
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kf[s], s_acc,
                                                      0, 0, 0);
      dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vf[s], dp_acc,
                                                       0, 0, 0);
    }
    unsigned dbits = 0;
    if constexpr (DROP)
      dbits = drop_bits_keyfixed(mbits, bh, S, q0, hi, key, k0);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = q0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      const int qclmp = min(qrow, S - 1);
      const float l = lse_in_lds ? lse_lds[qclmp] : lsep[qclmp];
      const float dlt = lse_in_lds ? dlt_lds[qclmp] : dltp[qclmp];
      float sv = s_acc[r] * scale + mask_val;
      bool dead = (qrow >= S) || (key >= S) || (causal && key > qrow) ||
                  !isfinite(l);
      const float p = dead ? 0.f : __expf(sv - l);
      float dpr = dp_acc[r];
      if constexpr (DROP)  // dP = mask * d(P_drop); delta is unchanged
        dpr *= ((dbits >> r) & 1) ? inv_keep : 0.f;
      dp_acc[r] = dead ? 0.f : p * (dpr - dlt) * scale;
    }
    unsigned int pa_ds[2][4];
    acc_to_afrag(dp_acc, pa_ds);
#pragma unroll
    for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
      for (int step = 0; step < 2; ++step) {
        bf16x8v qf2;
        {
          const int kb = step * 16 + 8 * hi;
          bf16x4v lo = DS_TR16(q_lds + tr16_b_off<true>(lane, kb, hh));
          bf16x4v h4 = DS_TR16(q_lds + tr16_b_off<true>(lane, kb + 4, hh));
#pragma unroll
          for (int e = 0; e < 4; ++e) { qf2[e] = lo[e]; qf2[e + 4] = h4[e]; }
        }
        dk_acc[hh] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            *reinterpret_cast<const bf16x8v*>(&pa_ds[step][0]), qf2,
            dk_acc[hh], 0, 0, 0);
      }
    }
    }  // compute
  }
  if (!live_wave) return;
#pragma unroll
  for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krow = k0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (krow < S)
        dkp[(long)krow * sg.rs + hh * 32 + j32] = f2bf(dk_acc[hh][r]);
    }
  }
}

// -------------------------------------------------------------------
// merged dv+dk kernel (A/B LOSER, kept behind RAVNEST_ATTN_MERGED_DKV=1
// for reference): one S-matmul + one staging feeds both accumulators,
// but the combined register set lands at 210 VGPRs = 2 waves/SIMD and
// measures 161 TF vs the split pair's 270 TF — the occupancy tier
// outweighs the saved matmul pass on this latency-bound structure.
// -------------------------------------------------------------------
__global__ __launch_bounds__(256) void attn_bwd_dvdk_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    const float* __restrict__ mask, bf16_t* __restrict__ dk,
    bf16_t* __restrict__ dv, int S, int causal, float scale, int has_mask,
    long H, StridesB sio, StridesB sdo, StridesB sg) {
  constexpr int D = 64;
  __shared__ __attribute__((aligned(16))) char smem[16384 + 16384];
  float* lse_lds = reinterpret_cast<float*>(smem + 16384);
  float* dlt_lds = lse_lds + 2048;
  const int st_row = threadIdx.x >> 3;
  const int st_c16 = (threadIdx.x & 7) * 16;
  const int lane = threadIdx.x & (WAVE - 1);
  const int hi = lane >> 5;
  const int j32 = lane & 31;
  const int k0 = (blockIdx.x * 4 + (int)(threadIdx.x / WAVE)) * 32;
  const bool live_wave = k0 < S;
  const long bh = blockIdx.y;
  const long b = bh / H, h = bh % H;
  const bf16_t* qp = q + b * sio.bs + h * sio.hs;
  const bf16_t* kp = k + b * sio.bs + h * sio.hs;
  const bf16_t* vp = v + b * sio.bs + h * sio.hs;
  const bf16_t* dop = dout + b * sdo.bs + h * sdo.hs;
  bf16_t* dkp = dk + b * sg.bs + h * sg.hs;
  bf16_t* dvp = dv + b * sg.bs + h * sg.hs;
  const float* lsep = lse + bh * (long)S;
  const float* dltp = delta + bh * (long)S;
  const float* mp = has_mask ? (mask + b * (long)S) : nullptr;
  const bool lse_in_lds = S <= 2048;
  if (lse_in_lds) {
    for (int i = threadIdx.x; i < S; i += blockDim.x) {
      lse_lds[i] = lsep[i];
      dlt_lds[i] = dltp[i];
    }
  }

  const int key = k0 + j32;
  const bf16_t* krp = kp + (long)min(key, S - 1) * sio.rs;
  const bf16_t* vrp = vp + (long)min(key, S - 1) * sio.rs;
  bf16x8v kf[4], vf[4];
  if (live_wave) {
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      kf[s] = *reinterpret_cast<const bf16x8v*>(krp + s * 16 + hi * 8);
      vf[s] = *reinterpret_cast<const bf16x8v*>(vrp + s * 16 + hi * 8);
    }
  }
  const float mask_val = (has_mask && key < S) ? mp[min(key, S - 1)] : 0.f;

  f32x16 dv_acc[2], dk_acc[2];
#pragma unroll
  for (int hh = 0; hh < 2; ++hh)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      dv_acc[hh][r] = 0.f;
      dk_acc[hh][r] = 0.f;
    }

  const int q_start = causal ? blockIdx.x * 128 : 0;
  // glds staging: both tiles are XOR-swizzled images, so both use the
  // source-side inverse permutation (see the dv kernel's comment)
  const int wdst = ((int)threadIdx.x >> 6) * 1024;
  const int p16 = (int)threadIdx.x * 16;
  const int inv_r7 = ((p16 >> 7) & 1) ^ ((p16 >> 10) & 1);
  const int p_src = (p16 & ~0xF0) | ((((p16 >> 4) & 1) ^ inv_r7) << 4) |
                    ((((p16 >> 5) & 1) ^ ((p16 >> 8) & 1)) << 5) |
                    ((((p16 >> 6) & 1) ^ ((p16 >> 9) & 1)) << 6) |
                    (inv_r7 << 7);
  const int s_row = p_src >> 7, s_col = p_src & 127;
  auto stage2 = [&](int qt, char* sl) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            reinterpret_cast<const char*>(qp) +
            (long)min(qt + s_row, S - 1) * sio.rs * 2 + s_col),
        (__attribute__((address_space(3))) void*)(sl + wdst), 16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            reinterpret_cast<const char*>(dop) +
            (long)min(qt + s_row, S - 1) * sdo.rs * 2 + s_col),
        (__attribute__((address_space(3))) void*)(sl + 4096 + wdst),
        16, 0, 0);
  };
  stage2(q_start, smem);
  for (int q0 = q_start; q0 < S; q0 += 32) {
    const int slot = ((q0 - q_start) >> 5) & 1;
    const char* q_lds = smem + slot * 8192;
    const char* do_lds = q_lds + 4096;
    __syncthreads();
    const bool have_next = q0 + 32 < S;
    if (have_next)  // T14: next tile's LDS-DMA under this compute
      stage2(q0 + 32, smem + (slot ^ 1) * 8192);
    const bool compute = live_wave && (!causal || q0 + 31 >= k0);
    if (compute) {
      f32x16 s_acc, dp_acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        s_acc[r] = 0.f;
        dp_acc[r] = 0.f;
      }
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        const int foff = (j32 * 128 + (s * 16 + hi * 8) * 2) ^
                         ((j32 & 15) << 4);
        bf16x8v qf = *reinterpret_cast<const bf16x8v*>(q_lds + foff);
        bf16x8v dof = *reinterpret_cast<const bf16x8v*>(do_lds + foff);
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kf[s], s_acc,
                                                        0, 0, 0);
        dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vf[s], dp_acc,
                                                         0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = q0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const int qclmp = min(qrow, S - 1);
        const float l = lse_in_lds ? lse_lds[qclmp] : lsep[qclmp];
        const float dlt = lse_in_lds ? dlt_lds[qclmp] : dltp[qclmp];
        float sv = s_acc[r] * scale + mask_val;
        bool dead = (qrow >= S) || (key >= S) || (causal && key > qrow) ||
                    !isfinite(l);
        const float p = dead ? 0.f : __expf(sv - l);
        s_acc[r] = p;                                          // P
        dp_acc[r] = dead ? 0.f : p * (dp_acc[r] - dlt) * scale;  // dS
      }
      unsigned int pa_p[2][4], pa_ds[2][4];
      acc_to_afrag(s_acc, pa_p);
      acc_to_afrag(dp_acc, pa_ds);
#pragma unroll
      for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
        for (int step = 0; step < 2; ++step) {
          bf16x8v dof2, qf2;
          {
            const int kb = step * 16 + 8 * hi;
            const int o0 = tr16_b_off<true>(lane, kb, hh);
            const int o1 = tr16_b_off<true>(lane, kb + 4, hh);
            bf16x4v a0 = DS_TR16(do_lds + o0), a1 = DS_TR16(do_lds + o1);
            bf16x4v b0 = DS_TR16(q_lds + o0), b1 = DS_TR16(q_lds + o1);
#pragma unroll
            for (int e = 0; e < 4; ++e) {
              dof2[e] = a0[e]; dof2[e + 4] = a1[e];
              qf2[e] = b0[e]; qf2[e + 4] = b1[e];
            }
          }
          dv_acc[hh] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<const bf16x8v*>(&pa_p[step][0]), dof2,
              dv_acc[hh], 0, 0, 0);
          dk_acc[hh] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<const bf16x8v*>(&pa_ds[step][0]), qf2,
              dk_acc[hh], 0, 0, 0);
        }
      }
    }
  }
  if (!live_wave) return;
#pragma unroll
  for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krow = k0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (krow < S) {
        const long off = (long)krow * sg.rs + hh * 32 + j32;
        dvp[off] = f2bf(dv_acc[hh][r]);
        dkp[off] = f2bf(dk_acc[hh][r]);
      }
    }
  }
}

// -------------------------------------------------------------------
// dq kernel: wave owns q rows [q0, q0+32); loops kv tiles (forward
// orientation: acc = [key-pattern][qrow=lane&31]).
// -------------------------------------------------------------------
template <bool DROP = false>
__global__ __launch_bounds__(256, DROP ? 3 : 4) void attn_bwd_dq_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    const float* __restrict__ mask, bf16_t* __restrict__ dq, int S,
    int causal, float scale, int has_mask, long H, StridesB sio,
    StridesB sdo, StridesB sg, float inv_keep = 1.f,
    const unsigned* __restrict__ mbits = nullptr) {
  constexpr int D = 64;
  // K tile swizzled (b128 A-frags for S^T, XOR-adjusted u16 B-reads for
  // dQ) + V tile swizzled (b128 A-frags for dP^T); double-buffered
  __shared__ __attribute__((aligned(16))) char smem[16384];
  const int st_row = threadIdx.x >> 3;
  const int st_c16 = (threadIdx.x & 7) * 16;
  const int lane = threadIdx.x & (WAVE - 1);
  const int hi = lane >> 5;
  const int j32 = lane & 31;
  const int qtile = blockIdx.x * 4 + (int)(threadIdx.x / WAVE);
  const int q0 = qtile * 32;
  const bool live_wave = q0 < S;
  const long bh = blockIdx.y;
  const long b = bh / H, h = bh % H;
  const bf16_t* qp = q + b * sio.bs + h * sio.hs;
  const bf16_t* kp = k + b * sio.bs + h * sio.hs;
  const bf16_t* vp = v + b * sio.bs + h * sio.hs;
  const bf16_t* dop = dout + b * sdo.bs + h * sdo.hs;
  bf16_t* dqp = dq + b * sg.bs + h * sg.hs;
  const float* mp = has_mask ? (mask + b * (long)S) : nullptr;

  const int qrow = q0 + j32;
  const bf16_t* qrp = qp + (long)min(qrow, S - 1) * sio.rs;
  const bf16_t* dorp = dop + (long)min(qrow, S - 1) * sdo.rs;
  bf16x8v qf[4], dof[4];
  if (live_wave) {
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      qf[s] = *reinterpret_cast<const bf16x8v*>(qrp + s * 16 + hi * 8);
      dof[s] = *reinterpret_cast<const bf16x8v*>(dorp + s * 16 + hi * 8);
    }
  }
  const float l_row = (qrow < S) ? lse[bh * (long)S + qrow] : -INFINITY;
  const float dlt_row = (qrow < S) ? delta[bh * (long)S + qrow] : 0.f;

  f32x16 dq_acc[2];
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[h][r] = 0.f;

  // block-wide kv range (barriers are block-wide); waves guard compute
  const int kv_end = causal ? min(S, blockIdx.x * 128 + 128) : S;
  // glds staging: K and V tiles, source-side inverse of the XOR
  // swizzle (see the dv kernel's comment)
  const int wdst = ((int)threadIdx.x >> 6) * 1024;
  const int p16 = (int)threadIdx.x * 16;
  const int inv_r7 = ((p16 >> 7) & 1) ^ ((p16 >> 10) & 1);
  const int p_src = (p16 & ~0xF0) | ((((p16 >> 4) & 1) ^ inv_r7) << 4) |
                    ((((p16 >> 5) & 1) ^ ((p16 >> 8) & 1)) << 5) |
                    ((((p16 >> 6) & 1) ^ ((p16 >> 9) & 1)) << 6) |
                    (inv_r7 << 7);
  const int s_row = p_src >> 7, s_col = p_src & 127;
  auto stage2 = [&](int kt, char* sl) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            reinterpret_cast<const char*>(kp) +
            (long)min(kt + s_row, S - 1) * sio.rs * 2 + s_col),
        (__attribute__((address_space(3))) void*)(sl + wdst), 16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            reinterpret_cast<const char*>(vp) +
            (long)min(kt + s_row, S - 1) * sio.rs * 2 + s_col),
        (__attribute__((address_space(3))) void*)(sl + 4096 + wdst),
        16, 0, 0);
  };
  stage2(0, smem);
  for (int k0 = 0; k0 < kv_end; k0 += 32) {
    const int slot = (k0 >> 5) & 1;
    const char* k_ldsb = smem + slot * 8192;
    const char* v_ldsb = k_ldsb + 4096;
    __syncthreads();
    const bool have_next = k0 + 32 < kv_end;
    if (have_next)  // T14: next tile's LDS-DMA under this compute
      stage2(k0 + 32, smem + (slot ^ 1) * 8192);
    const bool compute = live_wave && (!causal || k0 <= q0 + 31);
    if (compute) {
    f32x16 s_acc, dp_acc;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      s_acc[r] = 0.f;
      dp_acc[r] = 0.f;
    }
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int foff = (j32 * 128 + (s * 16 + hi * 8) * 2) ^
                       ((j32 & 15) << 4);
      bf16x8v kfr = *reinterpret_cast<const bf16x8v*>(k_ldsb + foff);
      bf16x8v vfr = *reinterpret_cast<const bf16x8v*>(v_ldsb + foff);
      // S^T[key][qrow], dP^T[key][qrow]
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfr, qf[s], s_acc,
                                                      0, 0, 0);
      dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfr, dof[s], dp_acc,
                                                       0, 0, 0);
    }
    unsigned dbits = 0;
    if constexpr (DROP)
      dbits = drop_bits_rowfixed(mbits, bh, S, k0, hi, qrow);
    float mload = 0.f;
    if (has_mask) mload = mp[min(k0 + j32, S - 1)];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kk = k0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      float sv = s_acc[r] * scale;
      if (has_mask && kk < S) sv += __shfl(mload, kk - k0, WAVE);
      bool dead = (qrow >= S) || (kk >= S) || (causal && kk > qrow) ||
                  !isfinite(l_row);
      const float p = dead ? 0.f : __expf(sv - l_row);
      float dpr = dp_acc[r];
      if constexpr (DROP)
        dpr *= ((dbits >> r) & 1) ? inv_keep : 0.f;
      dp_acc[r] = dead ? 0.f : p * (dpr - dlt_row) * scale;  // dS
    }
    // dQ += dS K : A = transform(dS^T) over k=keys; B[k=key][j=d] = K rows
    unsigned int pa_ds[2][4];
    acc_to_afrag(dp_acc, pa_ds);
#pragma unroll
    for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
      for (int step = 0; step < 2; ++step) {
        bf16x8v kcol;
        {
          const int kb = step * 16 + 8 * hi;
          bf16x4v lo = DS_TR16(k_ldsb + tr16_b_off<true>(lane, kb, hh));
          bf16x4v h4 = DS_TR16(k_ldsb + tr16_b_off<true>(lane, kb + 4, hh));
#pragma unroll
          for (int e = 0; e < 4; ++e) { kcol[e] = lo[e]; kcol[e + 4] = h4[e]; }
        }
        dq_acc[hh] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            *reinterpret_cast<const bf16x8v*>(&pa_ds[step][0]), kcol,
            dq_acc[hh], 0, 0, 0);
      }
    }
    }  // compute
  }
  if (!live_wave) return;

#pragma unroll
  for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = q0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (row < S)
        dqp[(long)row * sg.rs + hh * 32 + j32] = f2bf(dq_acc[hh][r]);
    }
  }
}

}  // namespace

namespace {

std::vector<at::Tensor> attn_bwd_impl(
    const bf16_t* qb, const bf16_t* kb, const bf16_t* vb, const bf16_t* ob,
    const bf16_t* dob, at::Tensor lse, const float* mask_ptr, bool has_mask,
    bool causal, double scale, long B, long H, int S, StridesB sio,
    StridesB sdo, StridesB sg, bf16_t* dqb, bf16_t* dkb, bf16_t* dvb,
    const at::TensorOptions& fopt, float pdrop = 0.f,
    const unsigned* mbits = nullptr) {
  const float inv_keep = 1.f / (1.f - pdrop);
  auto delta = at::empty({B * H * (long)S}, fopt);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  const long NR = B * H * (long)S;
  hipLaunchKernelGGL((attn_bwd_delta_kernel<64>),
                     dim3((NR * 8 + 255) / 256), dim3(256), 0, stream, dob,
                     ob, delta.data_ptr<float>(), S, H, NR, sdo);
  dim3 block(256);
  dim3 gridk((S + 127) / 128, B * H);
  // split is the measured default: the merged kernel lands at 210 VGPRs
  // (2 waves/SIMD) and runs at 161 TF vs the split pair's 270 TF
  static const bool split_dkv = [] {
    const char* e = getenv("RAVNEST_ATTN_MERGED_DKV");
    return !(e && e[0] == '1');
  }();
  TORCH_CHECK(pdrop == 0.f || (split_dkv && mbits != nullptr),
              "attn_bwd: fused prob-dropout needs the split dv/dk path "
              "and the fwd's mask words");
  if (pdrop > 0.f) {
    hipLaunchKernelGGL((attn_bwd_dv_kernel<true>), gridk, block, 0, stream,
                       qb, kb, dob, lse.data_ptr<float>(), mask_ptr, dvb, S,
                       causal ? 1 : 0, (float)scale, has_mask ? 1 : 0, H,
                       sio, sdo, sg, inv_keep, mbits);
    hipLaunchKernelGGL((attn_bwd_dk_kernel<true>), gridk, block, 0, stream,
                       qb, kb, vb, dob, lse.data_ptr<float>(),
                       delta.data_ptr<float>(), mask_ptr, dkb, S,
                       causal ? 1 : 0, (float)scale, has_mask ? 1 : 0, H,
                       sio, sdo, sg, inv_keep, mbits);
    hipLaunchKernelGGL((attn_bwd_dq_kernel<true>), gridk, block, 0, stream,
                       qb, kb, vb, dob, lse.data_ptr<float>(),
                       delta.data_ptr<float>(), mask_ptr, dqb, S,
                       causal ? 1 : 0, (float)scale, has_mask ? 1 : 0, H,
                       sio, sdo, sg, inv_keep, mbits);
    HIP_CHECK_LAST();
    return {};
  }
  if (split_dkv) {
    hipLaunchKernelGGL((attn_bwd_dv_kernel<false>), gridk, block, 0, stream,
                       qb, kb,
                       dob, lse.data_ptr<float>(), mask_ptr, dvb, S,
                       causal ? 1 : 0, (float)scale, has_mask ? 1 : 0, H,
                       sio, sdo, sg);
    hipLaunchKernelGGL((attn_bwd_dk_kernel<false>), gridk, block, 0, stream,
                       qb, kb,
                       vb, dob, lse.data_ptr<float>(),
                       delta.data_ptr<float>(), mask_ptr, dkb, S,
                       causal ? 1 : 0, (float)scale, has_mask ? 1 : 0, H,
                       sio, sdo, sg);
  } else {
    hipLaunchKernelGGL(attn_bwd_dvdk_kernel, gridk, block, 0, stream, qb,
                       kb, vb, dob, lse.data_ptr<float>(),
                       delta.data_ptr<float>(), mask_ptr, dkb, dvb, S,
                       causal ? 1 : 0, (float)scale, has_mask ? 1 : 0, H,
                       sio, sdo, sg);
  }
  hipLaunchKernelGGL((attn_bwd_dq_kernel<false>), gridk, block, 0, stream,
                     qb, kb,
                     vb, dob, lse.data_ptr<float>(), delta.data_ptr<float>(),
                     mask_ptr, dqb, S, causal ? 1 : 0, (float)scale,
                     has_mask ? 1 : 0, H, sio, sdo, sg);
  HIP_CHECK_LAST();
  return {};
}

}  // namespace

std::vector<at::Tensor> attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor o, at::Tensor dout,
                                 at::Tensor lse, at::Tensor mask, bool causal,
                                 double scale) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
              v.is_contiguous() && o.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  const long B = q.size(0), H = q.size(1);
  const int S = q.size(2), D = q.size(3);
  TORCH_CHECK(D == 64, "attn_bwd kernel: head_dim 64");
  auto dc = dout.contiguous();
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);

  const bool has_mask = mask.defined() && mask.numel() > 0;
  at::Tensor mask_f;
  const float* mask_ptr = nullptr;
  if (has_mask) {
    mask_f = mask.to(at::kFloat).reshape({B, S}).contiguous();
    mask_ptr = mask_f.data_ptr<float>();
  }
  const StridesB sep{H * (long)S * D, (long)S * D, (long)D};
  attn_bwd_impl(reinterpret_cast<const bf16_t*>(q.data_ptr()),
                reinterpret_cast<const bf16_t*>(k.data_ptr()),
                reinterpret_cast<const bf16_t*>(v.data_ptr()),
                reinterpret_cast<const bf16_t*>(o.data_ptr()),
                reinterpret_cast<const bf16_t*>(dc.data_ptr()), lse,
                mask_ptr, has_mask, causal, scale, B, H, S, sep, sep, sep,
                reinterpret_cast<bf16_t*>(dq.data_ptr()),
                reinterpret_cast<bf16_t*>(dk.data_ptr()),
                reinterpret_cast<bf16_t*>(dv.data_ptr()),
                q.options().dtype(at::kFloat));
  return {dq, dk, dv};
}

// packed layout: qkv (B,S,3,H,D), o/dout (B,S,H,D) -> dqkv (B,S,3,H,D)
at::Tensor attn_bwd_qkv(at::Tensor qkv, at::Tensor o, at::Tensor dout,
                        at::Tensor lse, at::Tensor mask, bool causal,
                        double scale, double pdrop,
                        c10::optional<at::Tensor> mbits) {
  TORCH_CHECK(qkv.is_cuda() && qkv.is_contiguous() && o.is_contiguous());
  TORCH_CHECK(qkv.dim() == 5 && qkv.size(2) == 3);
  const long B = qkv.size(0), H = qkv.size(3);
  const int S = qkv.size(1), D = qkv.size(4);
  TORCH_CHECK(D == 64, "attn_bwd_qkv kernel: head_dim 64");
  auto dc = dout.contiguous();
  auto dqkv = at::empty_like(qkv);

  const bool has_mask = mask.defined() && mask.numel() > 0;
  at::Tensor mask_f;
  const float* mask_ptr = nullptr;
  if (has_mask) {
    mask_f = mask.to(at::kFloat).reshape({B, S}).contiguous();
    mask_ptr = mask_f.data_ptr<float>();
  }
  const long HD = H * (long)D;
  const StridesB sp{(long)S * 3 * HD, (long)D, 3 * HD};   // qkv & dqkv
  const StridesB sod{(long)S * HD, (long)D, HD};          // o, dout
  const bf16_t* base = reinterpret_cast<const bf16_t*>(qkv.data_ptr());
  bf16_t* dbase = reinterpret_cast<bf16_t*>(dqkv.data_ptr());
  const unsigned* mb = nullptr;
  if (mbits.has_value())
    mb = reinterpret_cast<const unsigned*>(mbits->data_ptr<int>());
  attn_bwd_impl(base, base + HD, base + 2 * HD,
                reinterpret_cast<const bf16_t*>(o.data_ptr()),
                reinterpret_cast<const bf16_t*>(dc.data_ptr()), lse,
                mask_ptr, has_mask, causal, scale, B, H, S, sp, sod, sp,
                dbase, dbase + HD, dbase + 2 * HD,
                qkv.options().dtype(at::kFloat), (float)pdrop, mb);
  return dqkv;
}
