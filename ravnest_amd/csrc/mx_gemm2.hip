// LDS-staged block-scaled MX fp8 GEMM for CDNA4 (gfx950) — v2 of the
// MX path (csrc/mx.hip's register-tiled L2-streaming kernel measured
// 534 TF; this adapts the 256^2 staged structure of csrc/gemm.hip,
// whose bf16 twin measures ~1050 TF, to the 2x-rate scaled MFMA).
//
//   y[M,N] (bf16) = (xq[M,K] e4m3 . xs[M,K/32] e8m0)
//                 @ (wq[N,K] e4m3 . ws[N,K/32] e8m0)^T
//
// Same skeleton as gemm.hip: 512 threads = 8 waves (2M x 4N), 256x256
// tile, glds-staged double-buffered K-tiles with the conflict-free
// 3-bit source/read XOR swizzle, counted vmcnt(6) across raw barriers,
// 4-phase read striping (12/4/8/0 ds_read_b128), multi-column-tile
// blocks, M-grouped XCD-contiguous mapping. Differences:
//   * a K-tile is 128 fp8 ELEMENTS = the same 128-byte rows, feeding
//     two v_mfma_scale_f32_32x32x64_f8f6f4 k-steps;
//   * fragments are 32-row (32 B/lane, k-interleaved halves per the
//     mx.hip on-device scale-semantics probe: lane (row, hi) loads
//     bytes [ks*64 + hi*16) and [ks*64 + 32 + hi*16) so its scale
//     operand byte covers exactly MX block (4*kt + 2*ks + hi));
//   * e8m0 scales ride as ONE u32 global load per 32-row fragment per
//     K-tile (4 blocks of scale = 4 bytes; K % 128 == 0 keeps them
//     aligned). These ordinary loads next to glds cost hipcc a
//     vmcnt(0) at their use (guide trap (b)) — measured acceptable;
//   * operands SWAPPED (mfma(w-frag, x-frag)): each lane's accumulator
//     quad-runs are 4 consecutive N columns of one M row -> 8 B stores.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef int i32x8v __attribute__((ext_vector_type(8)));
typedef int i32x4v __attribute__((ext_vector_type(4)));
typedef float f32x16v __attribute__((ext_vector_type(16)));
typedef unsigned short u16x4 __attribute__((ext_vector_type(4)));

#define LDS_I4(p)                                            \
  *reinterpret_cast<__attribute__((address_space(3))) const  \
                        i32x4v*>(                            \
      (__attribute__((address_space(3))) const void*)(p))

namespace {

constexpr int BM = 256, BN = 256, BKE = 128;  // K-tile in fp8 elements
constexpr int HALF_BYTES = 128 * BKE;         // 16 KiB per half-tile
constexpr int TILE_BYTES = 2 * HALF_BYTES;
constexpr int BUF_BYTES = 2 * TILE_BYTES;
// e8m0 scales ride the SAME glds pipeline as the data (guide trap b:
// a plain global scale load issued between glds batches forces hipcc
// to a vmcnt(0) at the scale's use, draining the whole staging
// pipeline). The host repacks scales tile-major — (nKT, Mp)/(nKT, Np)
// u32, one u32 = 4 MX-block exponents per row per 128-element K-tile,
// rows padded to the 256 grid — so one wave can glds each side's 1 KiB
// per tile; consumers then read them from LDS (lgkm-counted, no vm
// interaction). 2 KiB per buffer: 2 x 66 KiB still one block per CU.
constexpr int SC_BYTES = 2048;
constexpr int BUF_FULL = BUF_BYTES + SC_BYTES;

DEVINL unsigned stage_off8(long ld, long gR0, long gRmax, int i) {
  const int tid = threadIdx.x;
  const int p = i * 8192 + (tid >> 6) * 1024 + (tid & 63) * 16;
  const int rr = p >> 7;
  const int cb = (p & 127) ^ (((p >> 8) & 7) << 4);
  long g = gR0 + rr;
  if (g > gRmax) g = gRmax;
  return (unsigned)(g * ld + cb);  // ld in BYTES per row == K elements
}

// fragment piece offset: row rr of a half image, k-step ks (64 elems),
// half hi, piece pc (0: +0, 1: +32) — 16 B each, swizzled
DEVINL int frag_off8(int rr, int ks, int hi, int pc) {
  const int cb = ks * 64 + pc * 32 + hi * 16;
  const int p = (rr << 7) + cb;
  return p ^ (((p >> 8) & 7) << 4);
}

__global__ __launch_bounds__(512, 2) void mx2_kernel(
    const unsigned char* __restrict__ XQ,
    const unsigned char* __restrict__ XS_T,
    const unsigned char* __restrict__ WQ,
    const unsigned char* __restrict__ WS_T,
    bf16_t* __restrict__ C, long M, long N, long K, long Mp, long Np,
    int MT, int NTb, int tpb) {
  __shared__ char smem[2 * BUF_FULL];

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
  const int bn0 = (lid / gsz) * tpb;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2, wn = wid & 3;
  const int l31 = lane & 31, hi = lane >> 5;

  const long mBase = (long)bm * BM;
  const long nBase = (long)bn0 * BN;
  const int nKT = (int)(K / BKE);
  const unsigned SJ = (unsigned)(BN * K);  // bytes per column tile of W

  unsigned sp[4][2];
  unsigned capB[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    sp[0][i] = stage_off8(K, mBase, M - 1, i);
    sp[1][i] = stage_off8(K, mBase + 128, M - 1, i);
    sp[2][i] = stage_off8(K, nBase, N - 1, i);
    sp[3][i] = stage_off8(K, nBase + 128, N - 1, i);
    capB[i] = stage_off8(K, 0, 0, i) + (unsigned)((N - 1) * K);
  }
  const int slo[4] = {0, HALF_BYTES, TILE_BYTES, TILE_BYTES + HALF_BYTES};
  const int wbase = (threadIdx.x >> 6) * 1024;
  auto stage = [&](int jj, int tt, int kind) {
    if (jj >= tpb) {
      jj = tpb - 1;
      tt = nKT - 1;
    }
    const unsigned off = (unsigned)tt * BKE;
    const int b = (jj * nKT + tt) & 1;
    char* const lb = smem + b * BUF_FULL + slo[kind] + wbase;
    const char* const base =
        reinterpret_cast<const char*>(kind >= 2 ? WQ : XQ);
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
  auto norm = [&](int j, int t, int dt, int& jj, int& tt) {
    tt = t + dt;
    jj = j;
    if (tt >= nKT) {
      tt -= nKT;
      ++jj;
    }
  };
  // one wave gldses each side's 1 KiB of tile-major scales (dst is the
  // area base: glds writes lane L's 16 B at dst + 16L; rows padded on
  // the host so no clamp is needed)
  auto stage_sc = [&](int jj, int tt) {
    if (jj >= tpb) {
      jj = tpb - 1;
      tt = nKT - 1;
    }
    const int b = (jj * nKT + tt) & 1;
    if (wid == 0) {
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(
              XS_T + (long)tt * Mp * 4 + mBase * 4 + lane * 16),
          (__attribute__((address_space(3))) void*)(
              smem + b * BUF_FULL + BUF_BYTES),
          16, 0, 0);
    } else if (wid == 1) {
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(
              WS_T + (long)tt * Np * 4 + (nBase + (long)jj * BN) * 4 +
              lane * 16),
          (__attribute__((address_space(3))) void*)(
              smem + b * BUF_FULL + BUF_BYTES + 1024),
          16, 0, 0);
    }
  };

  stage_sc(0, 0);
  stage(0, 0, 0); stage(0, 0, 2); stage(0, 0, 3); stage(0, 0, 1);
  stage(0, 1, 2); stage(0, 1, 3); stage(0, 1, 0);
  asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  i32x8v aF[2][2], bF[2][2];  // [frag][ks] current strip
  unsigned aSc[4], bSc[2];    // u32 scale words (4 blocks each)

  for (int j = 0; j < tpb; ++j) {
  f32x16v acc[4][2];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 2; ++n)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[m][n][r] = 0.f;

  for (int t = 0; t < nKT; ++t) {
    const int b = (j * nKT + t) & 1;
    int jj, tt;
    char* const aH0 = smem + b * BUF_FULL + wm * HALF_BYTES;
    char* const bH =
        smem + b * BUF_FULL + TILE_BYTES + (wn >> 1) * HALF_BYTES;
    const int bRow0 = (wn & 1) * 64;

    // scale words for this K-tile: LDS reads of the glds-staged area
    // (lgkm-counted — no vmcnt interaction with the data pipeline)
    {
      const char* const scA = smem + b * BUF_FULL + BUF_BYTES;
      const char* const scB = scA + 1024;
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
        aSc[mf] = *reinterpret_cast<
            const __attribute__((address_space(3))) unsigned*>(
            (const __attribute__((address_space(3))) void*)(
                scA + (wm * 128 + mf * 32 + l31) * 4));
      bSc[0] = *reinterpret_cast<
          const __attribute__((address_space(3))) unsigned*>(
          (const __attribute__((address_space(3))) void*)(
              scB + (wn * 64 + l31) * 4));
      bSc[1] = *reinterpret_cast<
          const __attribute__((address_space(3))) unsigned*>(
          (const __attribute__((address_space(3))) void*)(
              scB + (wn * 64 + 32 + l31) * 4));
    }

    // ph0: read B strip 0 + A strips of M-half 0 (12 b128 reads)
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      i32x8v f;
      *reinterpret_cast<i32x4v*>(&f) =
          LDS_I4(bH + frag_off8(bRow0 + l31, ks, hi, 0));
      *(reinterpret_cast<i32x4v*>(&f) + 1) =
          LDS_I4(bH + frag_off8(bRow0 + l31, ks, hi, 1));
      bF[0][ks] = f;
    }
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        i32x8v f;
        *reinterpret_cast<i32x4v*>(&f) =
            LDS_I4(aH0 + frag_off8(mf * 32 + l31, ks, hi, 0));
        *(reinterpret_cast<i32x4v*>(&f) + 1) =
            LDS_I4(aH0 + frag_off8(mf * 32 + l31, ks, hi, 1));
        aF[mf][ks] = f;
      }
    norm(j, t, 1, jj, tt);
    stage(jj, tt, 1);
    stage_sc(jj, tt);  // next tile's scales ride the same vmcnt window
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        acc[mf][0] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            bF[0][ks], aF[mf][ks], acc[mf][0], 0, 0, 0,
            (bSc[0] >> (8 * (2 * ks + hi))) & 0xff, 0,
            (aSc[mf] >> (8 * (2 * ks + hi))) & 0xff);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ph1: read B strip 1 (4 reads); MFMA M-half 0 x strip 1
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      i32x8v f;
      *reinterpret_cast<i32x4v*>(&f) =
          LDS_I4(bH + frag_off8(bRow0 + 32 + l31, ks, hi, 0));
      *(reinterpret_cast<i32x4v*>(&f) + 1) =
          LDS_I4(bH + frag_off8(bRow0 + 32 + l31, ks, hi, 1));
      bF[1][ks] = f;
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        acc[mf][1] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            bF[1][ks], aF[mf][ks], acc[mf][1], 0, 0, 0,
            (bSc[1] >> (8 * (2 * ks + hi))) & 0xff, 0,
            (aSc[mf] >> (8 * (2 * ks + hi))) & 0xff);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ph2: read A strips of M-half 1 (8 reads); MFMA x strip 0
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        i32x8v f;
        *reinterpret_cast<i32x4v*>(&f) =
            LDS_I4(aH0 + frag_off8(64 + mf * 32 + l31, ks, hi, 0));
        *(reinterpret_cast<i32x4v*>(&f) + 1) =
            LDS_I4(aH0 + frag_off8(64 + mf * 32 + l31, ks, hi, 1));
        aF[mf][ks] = f;
      }
    norm(j, t, 2, jj, tt);
    stage(jj, tt, 2);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        acc[2 + mf][0] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            bF[0][ks], aF[mf][ks], acc[2 + mf][0], 0, 0, 0,
            (bSc[0] >> (8 * (2 * ks + hi))) & 0xff, 0,
            (aSc[2 + mf] >> (8 * (2 * ks + hi))) & 0xff);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ph3: MFMA M-half 1 x strip 1; tile-boundary vmcnt
    stage(jj, tt, 3);
    stage(jj, tt, 0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        acc[2 + mf][1] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            bF[1][ks], aF[mf][ks], acc[2 + mf][1], 0, 0, 0,
            (bSc[1] >> (8 * (2 * ks + hi))) & 0xff, 0,
            (aSc[2 + mf] >> (8 * (2 * ks + hi))) & 0xff);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue (swapped-operand 32x32 layout): lane holds M-row
  // mBase+wm*128+mf*32+l31, N-cols (reg&3)+8*(reg>>2)+4*hi per strip
  const long mRow0 = mBase + wm * 128 + l31;
  const long nCol0 = nBase + (long)j * BN + wn * 64 + 4 * hi;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
    const long m = mRow0 + mf * 32;
    if (m >= M) continue;
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
      for (int qd = 0; qd < 4; ++qd) {
        const long n = nCol0 + nf * 32 + qd * 8;
        if (n >= N) continue;
        const u16x4 o{f2us(acc[mf][nf][4 * qd + 0]),
                      f2us(acc[mf][nf][4 * qd + 1]),
                      f2us(acc[mf][nf][4 * qd + 2]),
                      f2us(acc[mf][nf][4 * qd + 3])};
        if (n + 4 <= N) {
          *reinterpret_cast<u16x4*>(C + m * N + n) = o;
        } else {
          for (int v = 0; v < 4; ++v)
            if (n + v < N) C[m * N + n + v] = f2bf(us2f(o[v]));
        }
      }
    }
  }
  }  // for j
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}

}  // namespace

at::Tensor mx_gemm2(at::Tensor xq, at::Tensor xs, at::Tensor wq,
                    at::Tensor ws) {
  TORCH_CHECK(xq.is_cuda() && xq.scalar_type() == at::kByte &&
              xq.is_contiguous() && wq.scalar_type() == at::kByte &&
              wq.is_contiguous());
  const long K = xq.size(-1);
  const long M = xq.numel() / K, N = wq.size(0);
  TORCH_CHECK(wq.size(1) == K, "mx_gemm2: K mismatch");
  TORCH_CHECK(K % BKE == 0 && K >= 2 * BKE,
              "mx_gemm2: K % 128 == 0 and K >= 256 required");
  TORCH_CHECK(xq.numel() < (1ll << 32) && wq.numel() < (1ll << 32));
  TORCH_CHECK(xs.is_contiguous() && ws.is_contiguous() &&
              xs.numel() == M * (K / 32) && ws.numel() == N * (K / 32));
  auto C = at::empty({M, N}, xq.options().dtype(at::kBFloat16));
  const int MT = (int)((M + BM - 1) / BM), NT = (int)((N + BN - 1) / BN);
  int tpb = 1;
  for (int cand : {8, 6, 4, 3, 2}) {
    if (NT % cand == 0 && (long)MT * (NT / cand) >= 512) {
      tpb = cand;
      break;
    }
  }
  const int NTb = NT / tpb;
  // tile-major scale repack for the kernel's glds staging: (rows, K/32)
  // u8 -> (nKT, rowsPadded) u32 (one u32 = the 4 MX-block exponents of
  // one row's 128-element K-tile), rows zero-padded to the 256 grid
  // (e8m0 0x00 = 2^-127: pad rows contribute ~0 and their outputs are
  // dropped by the epilogue bounds checks anyway).
  const long nKT = K / BKE, Mp = (long)MT * BM, Np = (long)NT * BN;
  auto iopt = xq.options().dtype(at::kInt);
  auto pack = [&](const at::Tensor& sc, long rows, long rowsP) {
    auto t = at::zeros({nKT, rowsP}, iopt);
    t.narrow(1, 0, rows).copy_(
        sc.reshape({rows, nKT, 4}).view(at::kInt).reshape({rows, nKT})
            .transpose(0, 1));
    return t;
  };
  auto xs_t = pack(xs, M, Mp);
  auto ws_t = pack(ws, N, Np);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 grid((unsigned)(MT * NTb)), block(512);
  hipLaunchKernelGGL(mx2_kernel, grid, block, 0, stream,
                     xq.data_ptr<unsigned char>(),
                     reinterpret_cast<unsigned char*>(xs_t.data_ptr<int>()),
                     wq.data_ptr<unsigned char>(),
                     reinterpret_cast<unsigned char*>(ws_t.data_ptr<int>()),
                     reinterpret_cast<bf16_t*>(C.data_ptr()), M, N, K, Mp,
                     Np, MT, NTb, tpb);
  HIP_CHECK_LAST();
  return C;
}
