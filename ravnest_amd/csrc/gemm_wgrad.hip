// Split-K weight-gradient GEMM for CDNA4 (gfx950):
//   dW[N,K] = dy[M,N]^T @ x[M,K]   (reduction over the huge token dim)
//
// The wgrad is the "skinny TN" case: both operands are stored M-major,
// so the MFMA fragments need 8 consecutive REDUCTION (m) elements per
// lane at a fixed column — a transposed read. Rather than pre-transpose
// 400 MB operands, tiles are staged row-major into LDS and the
// fragments are gathered with ds_read_b64_tr_b16, gfx950's hardware
// transpose-read (semantics verified on-device by attention's
// tr16_probe: per 16-lane group, lane q loads row (q>>2), cols 4*(q&3)
// of a [4][16] window and RECEIVES column q).
//
// v2 staging: global_load_lds into a LINEAR [64][256 B] image whose
// bank conflicts are killed by a column XOR swizzle applied to the
// glds SOURCE address and the tr16 read address (guide rule 21):
//   cb' = cb ^ ((row & 7) << 5)
// A tr16 32-lane group covers 8 rows x 4 8-B col-quads; bank =
// (colwin/4 + 2*(q&3)) ^ (8*(row&7)) — all 32 combinations distinct,
// so the gather is conflict-free with no row padding (v1's padded rows
// + register staging was LDS-write-bound: 8 ds_write_b128 per thread
// per tile at 13 cycles each exceeded the MFMA time).
// Tail blocks (ragged N/K) keep a register-staged masked path writing
// the SAME swizzled image (glds cannot zero out-of-range columns).
//
// Geometry: 256 threads = 4 waves as 2(N) x 2(K); output tile 128x128;
// 16x16x32 bf16 MFMA; each block owns one output tile x one M-slice of
// 64-deep m-tiles, double-buffered (2 x 32 KiB LDS -> 2 blocks/CU, the
// cross-block overlap hides the simple vmcnt(0)-per-tile pipeline).
// SPLITK partial sums land as fp32 atomicAdd into a zeroed workspace.
//
// Parity: SURVEY.md §2.3 Linear/GEMM row (wgrad).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x4v __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8v __attribute__((ext_vector_type(8)));
typedef short s16x8 __attribute__((ext_vector_type(8)));

#define DS_TR16(p)                                   \
  __builtin_amdgcn_ds_read_tr16_b64_v4bf16(          \
      (__attribute__((address_space(3))) bf16x4v*)(  \
          (__attribute__((address_space(3))) void*)(p)))

namespace {

constexpr int BN = 128, BK = 128, BM = 64;   // tile: n x k x m-depth
constexpr int ROWB = 256;                    // image row bytes (linear)
constexpr int DY_BYTES = BM * ROWB;          // 16 KiB per operand image
constexpr int PAIR_BYTES = 2 * DY_BYTES;
// LDS: [buf 2][dy | x] images = 64 KiB -> 2 blocks/CU
// (a BK=64 / 3-blocks-per-CU variant measured SLOWER: halving BK
// doubles the dy re-reads per FLOP and the kernel goes HBM-bound)

DEVINL int swz(int row, int cb) { return cb ^ ((row & 7) << 5); }

template <bool TAIL>
__global__ __launch_bounds__(256, 2) void wgrad_kernel(
    const bf16_t* __restrict__ DY, const bf16_t* __restrict__ X,
    float* __restrict__ WS, long M, long N, long K, int NT, int KT,
    int splitk, long mTilesPer) {
  __shared__ char smem[2 * PAIR_BYTES];

  // block -> (output tile, m-slice); slices of one tile stay adjacent
  // so (after the XCD remap) a tile's partials land on one XCD's L2
  const int ntiles = NT * KT;
  const int nwg = ntiles * splitk;
  int bid = blockIdx.x, wg;
  if ((nwg & 7) == 0) {
    wg = (bid & 7) * (nwg >> 3) + (bid >> 3);
  } else {
    const int q = nwg >> 3, r = nwg & 7, x = bid & 7, lin = bid >> 3;
    wg = (x < r ? x * (q + 1) : r * (q + 1) + (x - r) * q) + lin;
  }
  const int tile = wg / splitk, slice = wg % splitk;
  const int bn = tile % NT, bk = tile / NT;
  const long n0 = (long)bn * BN, k0 = (long)bk * BK;
  const bool is_tail = (n0 + BN > N) || (k0 + BK > K);
  if (TAIL != is_tail) return;  // the two specializations split the grid
  const long mt0 = slice * mTilesPer;
  const long mt1 = min(mt0 + mTilesPer, M / BM);
  if (mt0 >= mt1) return;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wn = wid >> 1, wk = wid & 1;  // 2 x 2 wave grid

  auto img = [&](int b, int op) {
    return smem + b * PAIR_BYTES + op * DY_BYTES;
  };

  // ---- staging ------------------------------------------------------
  // 8 pieces of 16 B per thread cover the 32 KiB image pair. Piece i of
  // wave w lands lane-linear at (i*4 KiB + w*1 KiB + lane*16) — row =
  // off/256, cb = off%256 — and its SOURCE column carries the inverse
  // swizzle. 64-bit source offsets (operands can approach 4 GiB).
  long srcOff[8];          // byte offset into DY/X at m-tile 0
  int dstOff[8];           // byte offset into the image pair
  int shiftv[8], validv[8];  // tail masking (register path only)
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const int o = i * 4096 + wid * 1024 + lane * 16;
    const int op = o >= DY_BYTES;       // 0 = dy, 1 = x
    const int lo = op ? o - DY_BYTES : o;
    const int row = lo >> 8;
    const int cb = swz(row, lo & 255);
    const long cols = op ? K : N;
    const long cbase = op ? k0 : n0;
    long c = cbase + cb / 2;
    int sh = 0, va = 8;
    if (TAIL && c + 8 > cols) {
      const long cl = max(0l, cols - 8);
      sh = (int)(c - cl);
      va = (int)max(0l, min(8l, cols - c));
      c = cl;
    }
    srcOff[i] = ((long)row * cols + c) * 2;
    dstOff[i] = o;
    shiftv[i] = sh;
    validv[i] = va;
  }
  const long dyStep = (long)BM * N * 2, xStep = (long)BM * K * 2;

  auto stage_glds = [&](long mt, int b) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int op = dstOff[i] >= DY_BYTES;
      const char* src = reinterpret_cast<const char*>(op ? X : DY) +
                        srcOff[i] + mt * (op ? xStep : dyStep);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(
              smem + b * PAIR_BYTES + (dstOff[i] - lane * 16)),
          16, 0, 0);
    }
  };

  s16x8 stg[8];
  auto issue_loads = [&](long mt) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int op = dstOff[i] >= DY_BYTES;
      const char* s = reinterpret_cast<const char*>(op ? X : DY) +
                      srcOff[i] + mt * (op ? xStep : dyStep);
      stg[i] = *reinterpret_cast<const s16x8*>(s);
    }
  };
  auto write_stage = [&](int b) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      s16x8 v = stg[i];
      if (shiftv[i] | (validv[i] ^ 8)) {
        s16x8 w;
        for (int j = 0; j < 8; ++j)
          w[j] = (j < validv[i]) ? v[j + shiftv[i]] : short(0);
        v = w;
      }
      *reinterpret_cast<s16x8*>(smem + b * PAIR_BYTES + dstOff[i]) = v;
    }
  };

  // ---- tr16 fragment addressing (swizzled) -------------------------
  // lane l reads operand[m = mc*32+g*8+r*4+(q>>2)][colwin + 4*(q&3)..]
  const int q4 = lane & 15, g = lane >> 4;
  auto tr_off = [&](int mc, int r, int colwin) {
    const int row = mc * 32 + g * 8 + r * 4 + (q4 >> 2);
    const int cb = (colwin + 4 * (q4 & 3)) * 2;
    return row * ROWB + swz(row, cb);
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 4; ++b) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

  // prologue: stage tile mt0 into buf 0
  if (TAIL) {
    issue_loads(mt0);
    write_stage(0);
    __syncthreads();
  } else {
    stage_glds(mt0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  for (long mt = mt0; mt < mt1; ++mt) {
    const int b = (int)((mt - mt0) & 1);
    if (mt + 1 < mt1) {
      if (TAIL)
        issue_loads(mt + 1);
      else
        stage_glds(mt + 1, b ^ 1);
    }

    // fragments: dyT[nf][mc] from the dy image, x[kf][mc] from x image
    bf16x8v aF[4][2], bF[4][2];
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
#pragma unroll
      for (int mc = 0; mc < 2; ++mc) {
        bf16x4v lo = DS_TR16(img(b, 0) + tr_off(mc, 0, wn * 64 + nf * 16));
        bf16x4v hi = DS_TR16(img(b, 0) + tr_off(mc, 1, wn * 64 + nf * 16));
        bf16x8v f;
        f[0] = lo[0]; f[1] = lo[1]; f[2] = lo[2]; f[3] = lo[3];
        f[4] = hi[0]; f[5] = hi[1]; f[6] = hi[2]; f[7] = hi[3];
        aF[nf][mc] = f;
      }
#pragma unroll
    for (int kf = 0; kf < 4; ++kf)
#pragma unroll
      for (int mc = 0; mc < 2; ++mc) {
        bf16x4v lo = DS_TR16(img(b, 1) + tr_off(mc, 0, wk * 64 + kf * 16));
        bf16x4v hi = DS_TR16(img(b, 1) + tr_off(mc, 1, wk * 64 + kf * 16));
        bf16x8v f;
        f[0] = lo[0]; f[1] = lo[1]; f[2] = lo[2]; f[3] = lo[3];
        f[4] = hi[0]; f[5] = hi[1]; f[6] = hi[2]; f[7] = hi[3];
        bF[kf][mc] = f;
      }

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mc = 0; mc < 2; ++mc)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
#pragma unroll
        for (int kf = 0; kf < 4; ++kf)
          acc[nf][kf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              aF[nf][mc], bF[kf][mc], acc[nf][kf], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    if (TAIL) {
      if (mt + 1 < mt1) write_stage(b ^ 1);
      __syncthreads();
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  // ---- epilogue: fp32 atomic accumulation into the workspace -------
  // D layout: row i = n = (lane>>4)*4 + v, col j = k = lane&15
#pragma unroll
  for (int nf = 0; nf < 4; ++nf)
#pragma unroll
    for (int kf = 0; kf < 4; ++kf) {
      const long k = k0 + wk * 64 + kf * 16 + (lane & 15);
      if (k >= K) continue;
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        const long n = n0 + wn * 64 + nf * 16 + ((lane >> 4) << 2) + v;
        if (n < N) atomicAdd(WS + n * K + k, acc[nf][kf][v]);
      }
    }
}

}  // namespace

at::Tensor gemm_wgrad_bf16(at::Tensor dy, at::Tensor x) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16 &&
              dy.is_contiguous() && dy.dim() == 2);
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 &&
              x.is_contiguous() && x.dim() == 2);
  const long M = dy.size(0), N = dy.size(1), K = x.size(1);
  TORCH_CHECK(x.size(0) == M, "gemm_wgrad: M mismatch");
  TORCH_CHECK(M % BM == 0 && M >= 2 * BM,
              "gemm_wgrad: M % 64 == 0 and M >= 128 required");
  TORCH_CHECK(N >= 8 && K >= 8);
  auto ws = at::zeros({N, K}, dy.options().dtype(at::kFloat));
  const int NT = (int)((N + BN - 1) / BN), KT = (int)((K + BK - 1) / BK);
  const long mTiles = M / BM;
  const int ntiles = NT * KT;
  // split the reduction so the grid reaches ~3 block-waves (2/CU occ)
  int splitk = (int)std::min<long>(
      mTiles, std::max<long>(1, (1536 + ntiles - 1) / ntiles));
  const long mTilesPer = (mTiles + splitk - 1) / splitk;
  splitk = (int)((mTiles + mTilesPer - 1) / mTilesPer);
  auto stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 grid((unsigned)(ntiles * splitk)), block(256);
  const bool has_tail = (N % BN) || (K % BK);
  hipLaunchKernelGGL((wgrad_kernel<false>), grid, block, 0, stream,
                     reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                     reinterpret_cast<const bf16_t*>(x.data_ptr()),
                     ws.data_ptr<float>(), M, N, K, NT, KT, splitk,
                     mTilesPer);
  if (has_tail)
    hipLaunchKernelGGL((wgrad_kernel<true>), grid, block, 0, stream,
                       reinterpret_cast<const bf16_t*>(dy.data_ptr()),
                       reinterpret_cast<const bf16_t*>(x.data_ptr()),
                       ws.data_ptr<float>(), M, N, K, NT, KT, splitk,
                       mTilesPer);
  HIP_CHECK_LAST();
  return ws;
}
