// Hand-written bf16 MFMA GEMM for CDNA4 (gfx950) — the transformer
// projection hot path (SURVEY.md §2.3 Linear/GEMM row; reference runs
// these through stock GPU GEMMs, models.py:18,22 and every attention/
// MLP block).
//
// C[M,N] = A[M,K] @ B[N,K]^T  (the nn.Linear "TN" layout: both operands
// row-major with K contiguous), bf16 in / fp32 accumulate / bf16 out,
// with fused epilogues: none / +bias / +bias+GELU (the GELU variant also
// stores the pre-activation for the exact backward).
//
// Structure: the 256x256-tile phase schedule from the CDNA4 guide
// (guide §5 "The 256² 8-phase template"), built for the chip:
//   * 512 threads = 8 waves as 2(M) x 4(N); per-wave output 128x64 via
//     8x4 fragments of v_mfma_f32_16x16x32_bf16 (the 32x32x16 shape was
//     measured SLOWER here: its 2-accumulator clusters hit the
//     dependent-MFMA latency);
//   * K-tiles of 64 staged HBM->LDS with global_load_lds dwordx4 (the
//     async LDS-DMA path; 2 ops per 16 KiB half-tile), double-buffered:
//     128 KiB LDS total, one block per CU;
//   * conflict-free 3-bit XOR swizzle (col bits 4-6 ^= row bits 1-3)
//     applied to the glds SOURCE address and the ds_read_b128 address —
//     the LDS image itself stays lane-linear (guide rule 21); the
//     guide's 1-bit st_16x32 left uniform 2-way conflicts on THIS
//     fragment map (measured LDS_BANK_CONFLICT == #reads; now 0);
//   * counted s_waitcnt vmcnt(6) once per K-tile (3 half-tiles stay in
//     flight across barriers), raw s_barrier only at phase ends —
//     __syncthreads would drain the LDS-DMA queue (vmcnt(0)) and
//     serialize the pipeline (guide §5 "pipelining across barriers");
//   * s_setprio(1) around each 16-MFMA cluster (guide T5);
//   * bijective XCD-aware remap + M-grouped supertiles (A-band stays
//     cache-resident) + tpb consecutive column tiles per block (the
//     staging pipeline runs across tiles, so the cold-start prologue is
//     paid once per block, and A restages hit L2);
//   * operand lanes SWAPPED vs the textbook mapping: the kernel computes
//     mfma(B-frag, A-frag) so each lane's 4 accumulator values are 4
//     CONSECUTIVE N-columns of one M-row -> the epilogue stores 8 B per
//     lane per fragment instead of 4 scalar stores (store-issue tail).
//
// Numerics: fp32 accumulation over the full K, one bf16 rounding at the
// store — same class as hipBLASLt's bf16 GEMM.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8v __attribute__((ext_vector_type(8)));
typedef unsigned short u16x4 __attribute__((ext_vector_type(4)));

#define LDS_V8(p)                                            \
  *reinterpret_cast<__attribute__((address_space(3))) const  \
                        bf16x8v*>(                           \
      (__attribute__((address_space(3))) const void*)(p))

namespace {

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int HALF_BYTES = 128 * BK * 2;          // 16 KiB per half-tile
constexpr int TILE_BYTES = 2 * HALF_BYTES;        // 32 KiB per operand
constexpr int BUF_BYTES = 2 * TILE_BYTES;         // A+B per buffer
// LDS: [buf(2)][op A|B][half(2)][128 rows][64 cols] bf16 = 128 KiB

DEVINL float gelu_f(float x) {
  const float k = 0.7978845608028654f;  // sqrt(2/pi)
  const float t = __expf(2.f * k * (x + 0.044715f * x * x * x));
  return x * t / (t + 1.f);  // 0.5*x*(1+tanh(.)) rewritten in one exp
}

// one lane's glds SOURCE offset for op i (0/1) of a [128 x 64]
// half-tile: rows gR0..+127 of a row-major (ld-element) bf16 source at
// k-window 0. Advancing one K-tile is +128 bytes (done per stage call).
// LDS dest is lane-linear (glds requirement); the swizzle therefore
// moves to the per-lane SOURCE byte offset (involution: same XOR the
// ds_read side applies). Returned as a 32-bit byte offset from the
// tensor base (operands < 4 GiB, host-checked) so the 8 precomputed
// staging addresses cost 8 VGPRs, not 16.
DEVINL unsigned stage_off(long ld, long gR0, long gRmax, int i) {
  const int tid = threadIdx.x;
  const int p = i * 8192 + (tid >> 6) * 1024 + (tid & 63) * 16;
  const int rr = p >> 7;
  const int cb = (p & 127) ^ (((p >> 8) & 7) << 4);
  long g = gR0 + rr;
  if (g > gRmax) g = gRmax;
  return (unsigned)(g * ld * 2 + cb);
}

// fragment read offset inside one [128][64] half-tile image (bytes),
// swizzled. rr = row in half, kc = which K-32 chunk.
//
// Swizzle choice: a b128 lane group reads 16 distinct rows r at
// col-bytes cb in {c, c+16}; bank = (r&1)*32 + cb'/4, so only row bit 0
// reaches the bank. XORing col bits 4-6 with row bits 1-3 makes all 16
// (row, cb) pairs of every lane group land on distinct banks: same-cb
// rows differ in (r&1, (r>>1)&7); cross-cb collisions would need
// (r>>1)^(r'>>1) == cb-delta/16, which no group's row sets satisfy.
DEVINL int frag_off(int rr, int kc, int lane) {
  const int cb = ((lane >> 4) << 4) + (kc << 6);
  const int p = (rr << 7) + cb;
  return p ^ (((p >> 8) & 7) << 4);
}

// EPI: 0 = store acc; 1 = +bias; 2 = +bias, store pre-act H and GELU Y
template <int EPI>
__global__ __launch_bounds__(512, 2) void gemm_nt_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    const bf16_t* __restrict__ bias, bf16_t* __restrict__ C,
    bf16_t* __restrict__ H, long M, long N, long K, int MT, int NTb,
    int tpb) {
  __shared__ char smem[2 * BUF_BYTES];

  const int nwg = MT * NTb;
  int bid = blockIdx.x, wg;
  if ((nwg & 7) == 0) {
    wg = (bid & 7) * (nwg >> 3) + (bid >> 3);
  } else {
    const int q = nwg >> 3, r = nwg & 7, x = bid & 7, lin = bid >> 3;
    wg = (x < r ? x * (q + 1) : r * (q + 1) + (x - r) * q) + lin;
  }
  constexpr int GM = 8;
  const int group = wg / (GM * NTb);
  const int lid = wg % (GM * NTb);
  const int gsz = min(GM, MT - group * GM);
  const int bm = group * GM + lid % gsz;
  const int bn0 = (lid / gsz) * tpb;  // first of this block's column tiles

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2, wn = wid & 3;  // 2 x 4 wave grid

  const long mBase = (long)bm * BM;
  const long nBase = (long)bn0 * BN;
  const int nKT = (int)(K / BK);
  const unsigned SJ = (unsigned)(BN * K * 2);  // B stride per column tile

  // kind: 0/1 = A half, 2/3 = B half.
  unsigned sp[4][2];
  unsigned capB[2];  // row-(N-1) cap per op (ragged-N clamp)
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    sp[0][i] = stage_off(K, mBase, M - 1, i);
    sp[1][i] = stage_off(K, mBase + 128, M - 1, i);
    sp[2][i] = stage_off(K, nBase, N - 1, i);
    sp[3][i] = stage_off(K, nBase + 128, N - 1, i);
    capB[i] = stage_off(K, 0, 0, i) + (unsigned)((N - 1) * K * 2);
  }
  const int slo[4] = {0, HALF_BYTES, TILE_BYTES, TILE_BYTES + HALF_BYTES};
  const int wbase = (threadIdx.x >> 6) * 1024;
  // stage half-tile `kind` of virtual tile (jj = column tile, tt = K-tile)
  auto stage = [&](int jj, int tt, int kind) {
    if (jj >= tpb) {  // clamp: re-stages the final tile's bytes
      jj = tpb - 1;
      tt = nKT - 1;
    }
    const unsigned off = (unsigned)tt * (BK * 2);
    const int b = (jj * nKT + tt) & 1;
    char* const lb = smem + b * BUF_BYTES + slo[kind] + wbase;
    const char* const base =
        reinterpret_cast<const char*>(kind >= 2 ? B : A);
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      unsigned o = sp[kind][i];
      if (kind >= 2) {
        o += (unsigned)jj * SJ;
        if (o > capB[i]) o = capB[i];
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(base + o + off),
          (__attribute__((address_space(3))) void*)(lb + i * 8192),
          16, 0, 0);
    }
  };
  // (jj, tt) of virtual tile v+dt given current (j, t); dt in {1, 2}
  auto norm = [&](int j, int t, int dt, int& jj, int& tt) {
    tt = t + dt;
    jj = j;
    if (tt >= nKT) {
      tt -= nKT;
      ++jj;
    }
  };

  // ---- prologue: tile (0,0) fully + 3 half-tiles of tile (0,1) ------
  // In-tile read map: B halves are ds_read at ph0+ph1; each wave reads
  // its OWN A half at ph0 AND ph2 (rows 0-63 then 64-127). So a tile's
  // phases may stage into the live buffer: B0 at ph2, B1+A0 at ph3, and
  // A1 of the NEXT tile at ph0 (the other buffer, whose reads finished
  // last tile). The prologue below ends in exactly the steady state the
  // boundary vmcnt(6) maintains: 3 newest half-tiles in flight,
  // everything older landed. Requires nKT >= 2 (host checks K >= 128).
  stage(0, 0, 0); stage(0, 0, 2); stage(0, 0, 3); stage(0, 0, 1);
  stage(0, 1, 2); stage(0, 1, 3); stage(0, 1, 0);
  asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // fragment LDS offsets (bytes, within a half image)
  const int l15 = lane & 15;

  bf16x8v aF[4][2], bF[4][2];

  for (int j = 0; j < tpb; ++j) {
  f32x4 acc[8][4];
#pragma unroll
  for (int m = 0; m < 8; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = f32x4{0.f, 0.f, 0.f, 0.f};

  // ---- main loop over K-tiles ---------------------------------------
  for (int t = 0; t < nKT; ++t) {
    const int b = (j * nKT + t) & 1;
    int jj, tt;
    // this wave's A half image = wm; B half image = wn>>1
    char* const aH0 = smem + b * BUF_BYTES + wm * HALF_BYTES;
    char* const bH =
        smem + b * BUF_BYTES + TILE_BYTES + (wn >> 1) * HALF_BYTES;
    const int bRow0 = (wn & 1) * 64;          // B row base within half

    // ph0: read B fragments 0-1 + A fragments of M-half 0 (12 reads)
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc)
        bF[nf][kc] = LDS_V8(bH + frag_off(bRow0 + nf * 16 + l15, kc, lane));
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc)
        aF[mf][kc] = LDS_V8(aH0 + frag_off(mf * 16 + l15, kc, lane));
    norm(j, t, 1, jj, tt);
    stage(jj, tt, 1);  // next tile's A-half-1
    asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
#pragma unroll
        for (int kc = 0; kc < 2; ++kc)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              bF[nf][kc], aF[mf][kc], acc[mf][nf], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ph1: read B fragments 2-3 (4 reads); MFMA M-half 0 x N-frags 2,3
#pragma unroll
    for (int nf = 2; nf < 4; ++nf)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc)
        bF[nf][kc] = LDS_V8(bH + frag_off(bRow0 + nf * 16 + l15, kc, lane));
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 2; nf < 4; ++nf)
#pragma unroll
        for (int kc = 0; kc < 2; ++kc)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              bF[nf][kc], aF[mf][kc], acc[mf][nf], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ph2: read A fragments of M-half 1 (8 reads)
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc)
        aF[mf][kc] = LDS_V8(aH0 + frag_off(64 + mf * 16 + l15, kc, lane));
    norm(j, t, 2, jj, tt);
    stage(jj, tt, 2);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
#pragma unroll
        for (int kc = 0; kc < 2; ++kc)
          acc[4 + mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              bF[nf][kc], aF[mf][kc], acc[4 + mf][nf], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ph3: no reads; MFMA M-half 1 x N-frags 2,3; tile-boundary vmcnt
    stage(jj, tt, 3);
    stage(jj, tt, 0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 2; nf < 4; ++nf)
#pragma unroll
        for (int kc = 0; kc < 2; ++kc)
          acc[4 + mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              bF[nf][kc], aF[mf][kc], acc[4 + mf][nf], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue for column tile j ----------------------------------
  // Runs while the next tile's staged half-tiles are still in flight
  // (the epilogue touches no LDS, so no wait). Swapped-operand C/D
  // layout: lane holds M-row mBase+wm*128+mf*16+l15, N-cols
  // nBase+j*256+wn*64+nf*16+(lane>>4)*4 + v, v = 0..3 -> one 8 B store
  // per fragment (4 consecutive bf16 of one row). Interior fast path:
  // ONE divergence test for the whole tile, then unpredicated stores.
  const long mRow0 = mBase + wm * 128 + l15;
  const long nCol0 = nBase + (long)j * BN + wn * 64 + ((lane >> 4) << 2);
  auto epilogue = [&](auto interior) {
#pragma unroll
    for (int mf = 0; mf < 8; ++mf) {
      const long m = mRow0 + mf * 16;
      if constexpr (!interior.value) {
        if (m >= M) continue;
      }
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const long n = nCol0 + nf * 16;
        bool full = true;
        if constexpr (!interior.value) {
          if (n >= N) continue;
          full = (n + 4 <= N);
        }
        float v0 = acc[mf][nf][0], v1 = acc[mf][nf][1];
        float v2 = acc[mf][nf][2], v3 = acc[mf][nf][3];
        if (EPI >= 1) {
          if (full) {
            const u16x4 bb = *reinterpret_cast<const u16x4*>(bias + n);
            v0 += us2f(bb[0]); v1 += us2f(bb[1]);
            v2 += us2f(bb[2]); v3 += us2f(bb[3]);
          } else {
            v0 += bf2f(bias[n]);
            if (n + 1 < N) v1 += bf2f(bias[n + 1]);
            if (n + 2 < N) v2 += bf2f(bias[n + 2]);
          }
        }
        if (EPI == 2) {
          u16x4 h{f2us(v0), f2us(v1), f2us(v2), f2us(v3)};
          if (full) {
            *reinterpret_cast<u16x4*>(H + m * N + n) = h;
          } else {
            for (int v = 0; v < 4; ++v)
              if (n + v < N) H[m * N + n + v] = f2bf(us2f(h[v]));
          }
          v0 = gelu_f(v0); v1 = gelu_f(v1);
          v2 = gelu_f(v2); v3 = gelu_f(v3);
        }
        const u16x4 o{f2us(v0), f2us(v1), f2us(v2), f2us(v3)};
        if (full) {
          *reinterpret_cast<u16x4*>(C + m * N + n) = o;
        } else {
          for (int v = 0; v < 4; ++v)
            if (n + v < N) C[m * N + n + v] = f2bf(us2f(o[v]));
        }
      }
    }
  };
  if (mRow0 + 7 * 16 < M && nCol0 + 3 * 16 + 4 <= N)
    epilogue(std::true_type{});
  else
    epilogue(std::false_type{});
  }  // for j (column tiles of this block)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}

}  // namespace

std::vector<at::Tensor> gemm_nt_bf16(at::Tensor A2, at::Tensor B,
                                     c10::optional<at::Tensor> bias,
                                     int64_t epi) {
  TORCH_CHECK(A2.is_cuda() && A2.scalar_type() == at::kBFloat16 &&
              A2.is_contiguous() && A2.dim() == 2);
  TORCH_CHECK(B.is_cuda() && B.scalar_type() == at::kBFloat16 &&
              B.is_contiguous() && B.dim() == 2);
  const long M = A2.size(0), K = A2.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "gemm_nt_bf16: K mismatch");
  TORCH_CHECK(K % BK == 0 && K >= 2 * BK,
              "gemm_nt_bf16: K % 64 == 0 and K >= 128 required");
  TORCH_CHECK(A2.numel() * 2 < (1ll << 32) && B.numel() * 2 < (1ll << 32),
              "gemm_nt_bf16: operands must be < 4 GiB (32-bit staging "
              "offsets)");
  TORCH_CHECK(epi == 0 || bias.has_value(), "bias required for epi>=1");
  const bf16_t* bp = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->is_contiguous() &&
                bias->scalar_type() == at::kBFloat16 && bias->numel() == N);
    bp = reinterpret_cast<const bf16_t*>(bias->data_ptr());
  }
  auto C = at::empty({M, N}, A2.options());
  at::Tensor Hp;
  bf16_t* hp = nullptr;
  if (epi == 2) {
    Hp = at::empty({M, N}, A2.options());
    hp = reinterpret_cast<bf16_t*>(Hp.data_ptr());
  }
  const int MT = (int)((M + BM - 1) / BM), NT = (int)((N + BN - 1) / BN);
  // column tiles per block: amortizes the per-block staging cold start;
  // capped so the grid still fills the chip (>= ~2 block-waves)
  int tpb = 1;
  for (int cand : {8, 6, 4, 3, 2}) {
    if (NT % cand == 0 && (long)MT * (NT / cand) >= 512) {
      tpb = cand;
      break;
    }
  }
  const int NTb = NT / tpb;
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 grid((unsigned)(MT * NTb)), block(512);
  auto* a = reinterpret_cast<const bf16_t*>(A2.data_ptr());
  auto* b = reinterpret_cast<const bf16_t*>(B.data_ptr());
  auto* c = reinterpret_cast<bf16_t*>(C.data_ptr());
  switch (epi) {
    case 0:
      hipLaunchKernelGGL((gemm_nt_kernel<0>), grid, block, 0, stream, a, b,
                         bp, c, hp, M, N, K, MT, NTb, tpb);
      break;
    case 1:
      hipLaunchKernelGGL((gemm_nt_kernel<1>), grid, block, 0, stream, a, b,
                         bp, c, hp, M, N, K, MT, NTb, tpb);
      break;
    default:
      hipLaunchKernelGGL((gemm_nt_kernel<2>), grid, block, 0, stream, a, b,
                         bp, c, hp, M, N, K, MT, NTb, tpb);
  }
  HIP_CHECK_LAST();
  if (epi == 2) return {C, Hp};
  return {C};
}
