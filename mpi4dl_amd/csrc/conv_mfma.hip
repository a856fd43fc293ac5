// Implicit-GEMM MFMA convolution for gfx950 (CDNA4) — forward,
// backward-weight; backward-data reuses the forward kernel on
// transformed weights (stride 1).
//
// This is the hand-written compute path behind HaloConv2d: conv on
// halo-pre-padded NCHW bf16 tiles (padding usually 0 after the
// exchange; plain padding also supported). Design notes
// (/opt/skills/guides/cdna_hip_programming.md):
//
// * GEMM view: OUT[Kout][N*OH*OW] = W[Kout][C*R*S] x X[C*R*S][N*OH*OW].
//   With NCHW, pixels are contiguous for fixed (c,r,s), so the X
//   ("B") operand loads coalesce; weights are (c,r,s)-contiguous, so
//   the W ("A") operand is directly fragment-shaped.
// * N-tiles are OUTPUT-ROW SEGMENTS (one (n,oh) row per block), so the
//   per-k input address is base + c*HW + r*W + s and never wraps —
//   no per-pixel div/mod in the hot loop.
// * mfma_f32_16x16x32_bf16; block = 256 threads = 4 waves (2x2), tile
//   BM=128 (out-ch) x BN=128 (pixels) x BK=32; wave tile 64x64,
//   acc 4x4 fragments (64 f32/lane).
// * A (weights) LDS image [BM][BK] row-major -> ds_read_b128 fragments.
//   B (pixels) LDS image stores element (k, px) of each 16-px block at
//   off = px + (k&3)*16 + (k>>3)*64 + ((k>>2)&1)*256 (bf16 units), so
//   ONE ds_read_b64_tr_b16 pair per fragment delivers the full k=32
//   per-lane B fragment (the attention-V recipe, guide S2 tr-read).
// * fp32 accumulate; epilogue adds bias and stores bf16.
//
// Reference behaviour being replaced: cuDNN/MIOpen conv under
// torchgems' conv_spatial (/root/reference/src/torchgems/spatial.py:1027).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cstdint>

namespace conv_mfma {

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(8))) short s16x8;

#define BM 128
#define BN 128
#define BK 32
#define WAVES 4

struct ConvGeom {
  int N, C, H, W;        // input
  int K, R, S;           // weights
  int OH, OW;            // output
  int sh, sw, ph, pw;    // stride / pad
  int CRS;               // C*R*S
  int row_tiles;         // ceil(OW / BN)
};

// ---------------------------------------------------------------------------
// Forward kernel
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void conv_fwd_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    const float* __restrict__ bias, bf16* __restrict__ out, ConvGeom g) {
  // block -> (m_tile, n, oh, ow_tile); XCD-aware bijective swizzle so
  // consecutive tiles spread across the 8 XCDs' L2s (guide T1)
  const int m_tiles = (g.K + BM - 1) / BM;
  const int nwg = gridDim.x;
  const int q8 = nwg >> 3, r8 = nwg & 7;
  const int xcd = blockIdx.x & 7, sub = blockIdx.x >> 3;
  int bid = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + sub;
  if (bid >= nwg) bid = blockIdx.x;  // safety (never taken: mapping bijective)
  const int mt = bid % m_tiles;
  bid /= m_tiles;
  const int owt = bid % g.row_tiles;
  bid /= g.row_tiles;
  const int oh = bid % g.OH;
  const int n = bid / g.OH;

  const int k0out = mt * BM;           // first out-channel of tile
  const int ow0 = owt * BN;            // first output col of tile
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;            // 4 waves: (wm, wn) = (wid>>1, wid&1)
  const int wm = wid >> 1, wn = wid & 1;

  // LDS: A image [BM][BK] bf16 (+8 pad per row vs bank conflicts),
  //      B image (BN/16) blocks x 512 elems, double buffered.
  __shared__ __attribute__((aligned(16))) short lds[2 * (BM * (BK + 8)) +
                                                    2 * ((BN / 16) * 520)];
  auto ldsA = [&](int buf) { return lds + buf * (BM * (BK + 8)); };
  auto ldsB = [&](int buf) {
    return lds + 2 * (BM * (BK + 8)) + buf * ((BN / 16) * 520);
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)g.H * g.W;
  const int64_t in_n = (int64_t)n * g.C * HW;
  const int RS = g.R * g.S;

  // ---- staging helpers ----------------------------------------------------
  // A: 256 threads load BM*BK = 4096 bf16 in 2 passes of dwordx4 (8 bf16).
  // thread -> (row = idx/4, kchunk = idx%4)
  // B: 512 (k, px8) chunks -> 2 per thread.

  auto stage = [&](int buf, int kk0) {
    // ---- A (weights) ----
#pragma unroll
    for (int pass = 0; pass < 2; ++pass) {
      const int idx = pass * 256 + tid;       // 0..511
      const int row = idx >> 2;               // 0..127
      const int kc = (idx & 3) * 8;           // 0,8,16,24
      const int kout = k0out + row;
      short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      if (kout < g.K) {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int k = kk0 + kc + e;
          if (k < g.CRS) v[e] = ((const short*)w)[(int64_t)kout * g.CRS + k];
        }
      }
      short* dst = ldsA(buf) + row * (BK + 8) + kc;
#pragma unroll
      for (int e = 0; e < 8; ++e) dst[e] = v[e];
    }
    // ---- B (input pixels) ----
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int idx = it * 256 + tid;         // 0..511
      const int kk = idx & 31;                // k within tile
      const int pxc = idx >> 5;               // 16 chunks of 8 px
      const int px0 = pxc * 8;
      const int k = kk0 + kk;
      short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      if (k < g.CRS) {
        const int c = k / RS;
        const int rs = k - c * RS;
        const int r = rs / g.S;
        const int s = rs - r * g.S;
        const int ih = oh * g.sh - g.ph + r;
        if (ih >= 0 && ih < g.H) {
          const bf16* src = x + in_n + c * HW + (int64_t)ih * g.W;
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int ow = ow0 + px0 + e;
            const int iw = ow * g.sw - g.pw + s;
            if (ow < g.OW && iw >= 0 && iw < g.W)
              v[e] = ((const short*)src)[iw];
          }
        }
      }
      // B image (tr-read layout, semantics measured by tr16_probe):
      // element (k, px) of 16-px block pb lives at block-local offset
      //   (k>>3)*128 + ((k>>2)&1)*64 + (k&3)*16 + px
      // so this 8-pixel chunk is 16 CONTIGUOUS bytes (one ds_write_b128),
      // and the fragment read is one ds_read_b64_tr_b16 pair per k=32.
      const int base = (px0 >> 4) * 520 + ((kk >> 3) << 7) +
                       (((kk >> 2) & 1) << 6) + ((kk & 3) << 4) + (px0 & 15);
      short* dstB = ldsB(buf) + base;
#pragma unroll
      for (int e = 0; e < 8; ++e) dstB[e] = v[e];
    }
  };

  stage(0, 0);
  __syncthreads();

  const int a_row0 = wm * 64;     // my wave's 64 out-channels
  const int b_px0 = wn * 64;      // my wave's 64 pixels

  for (int kk0 = 0; kk0 < g.CRS; kk0 += BK) {
    const int buf = (kk0 / BK) & 1;
    if (kk0 + BK < g.CRS) {
      stage(buf ^ 1, kk0 + BK);
    }
    // fragments + MFMA
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
      const short* arow =
          ldsA(buf) + (a_row0 + mf * 16 + (lane & 15)) * (BK + 8) +
          ((lane >> 4) << 3);
      s16x8 afrag = *(const s16x8*)arow;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int pb = (b_px0 >> 4) + nf;
        // per-lane tr address: group base (lane>>4)*128 + (lane&15)*4;
        // read pair covers k = (lane>>4)*8 .. +8 for pixel col lane&15
        __attribute__((address_space(3))) short* bbase =
            (__attribute__((address_space(3))) short*)(ldsB(buf)) + pb * 520 +
            ((lane >> 4) << 7) + ((lane & 15) << 2);
        s16x4 b0 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (__attribute__((address_space(3))) s16x4*)bbase);
        s16x4 b1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (__attribute__((address_space(3))) s16x4*)(bbase + 64));
        s16x8 bfrag;
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          bfrag[e] = b0[e];
          bfrag[e + 4] = b1[e];
        }
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[mf][nf], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: bias + bf16 store ---------------------------------------
  const int64_t out_n = ((int64_t)n * g.K) * g.OH * g.OW;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int kout = k0out + a_row0 + mf * 16 + ((lane >> 4) << 2) + reg;
      if (kout >= g.K) continue;
      const float b = bias ? bias[kout] : 0.f;
      const int64_t orow = out_n + (int64_t)kout * g.OH * g.OW +
                           (int64_t)oh * g.OW;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int ow = ow0 + b_px0 + nf * 16 + (lane & 15);
        if (ow < g.OW)
          out[orow + ow] = (bf16)(acc[mf][nf][reg] + b);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Forward v2 (stride-1 in W): R passes over kernel rows with weights
// pre-permuted to [K][R][C*S] on the host, so
//  * the A (weight) tile stays dwordx4-contiguous per pass, and
//  * the B (pixel) loader loads ONE row segment per (c, px-chunk) and
//    writes its S shifted copies — S x fewer global loads than v1 and
//    16-byte vector loads instead of u16 scalars (guide G13).
// Accumulators persist across the R passes.
//
// Round 2: the pixel axis is GLOBAL (n*OH*OW flattened, one 8-px chunk
// never crosses an output row — OW % 8 == 0), so OW < BN shapes
// (AmoebaNet's 1x7/7x1 at 64..256 px) fill whole tiles; and the M tile
// adapts (MFRAG template: BM 32/64/128) so K=26/52/104 layers don't
// burn 60-80% of the MFMA work on masked rows.
// ---------------------------------------------------------------------------

template <int SS, int MFRAG>
__global__ __launch_bounds__(256) void conv_fwd_v2_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w2,
    const float* __restrict__ bias, bf16* __restrict__ out, ConvGeom g) {
  // Software-pipelined staging (guide T14): global loads for tile t+2
  // are issued while tile t computes; the LDS write of tile t+1 sits
  // right after the single per-iteration barrier, so its vmcnt wait is
  // hidden under the previous iteration's MFMAs.
  const int S = SS > 0 ? SS : g.S;
  constexpr int UNITS = (SS == 1) ? 2 : 1;  // S==1: 32 c's -> 512 units
  constexpr int BMV = 32 * MFRAG;
  constexpr int ACH = BMV * 4;              // A chunks (8 shorts each)

  const int m_tiles = (g.K + BMV - 1) / BMV;
  const int nwg = gridDim.x;
  const int q8 = nwg >> 3, r8 = nwg & 7;
  const int xcd = blockIdx.x & 7, sub = blockIdx.x >> 3;
  int bid = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + sub;
  const int mt = bid % m_tiles;
  const int pt = bid / m_tiles;
  const int OHW = g.OH * g.OW;
  const int ptiles_per_img = OHW / BN;
  const int n = pt / ptiles_per_img;
  const int q0 = (pt - n * ptiles_per_img) * BN;  // first output pixel

  const int k0out = mt * BMV;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;

  __shared__ __attribute__((aligned(16))) short lds[2 * (BMV * (BK + 8)) +
                                                    2 * ((BN / 16) * 520)];
  auto ldsA = [&](int buf) { return lds + buf * (BMV * (BK + 8)); };
  auto ldsB = [&](int buf) {
    return lds + 2 * (BMV * (BK + 8)) + buf * ((BN / 16) * 520);
  };

  f32x4 acc[MFRAG][4];
#pragma unroll
  for (int i = 0; i < MFRAG; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)g.H * g.W;
  const int64_t in_n = (int64_t)n * g.C * HW;
  const int CS = g.C * S;
  const int KT = (CS + BK - 1) / BK;
  const int total_it = g.R * KT;

  // pipelined register state
  s16x8 aReg[2];
  s16x8 bReg[UNITS][3];

  auto stage_load = [&](int it) {
    const int r = it / KT;
    const int kk0 = (it % KT) * BK;
#pragma unroll
    for (int pass = 0; pass < (ACH + 255) / 256; ++pass) {
      const int idx = pass * 256 + tid;
      if (ACH < 256 && idx >= ACH) break;
      const int row = idx >> 2;
      const int kc = (idx & 3) * 8;
      const int kout = k0out + row;
      if (kout < g.K && CS - (kk0 + kc) >= 8) {
        aReg[pass] = *(const s16x8*)((const short*)w2 +
                                     ((int64_t)kout * g.R + r) * CS + kk0 + kc);
      }
    }
    const int c_lo = kk0 / S;
#pragma unroll
    for (int u0 = 0; u0 < UNITS; ++u0) {
      const int u = u0 * 256 + tid;
      const int ci = u >> 4;
      const int pxc = u & 15;
      const int c = c_lo + ci;
      const int q = q0 + pxc * 8;
      const int oh = q / g.OW;
      const int ow00 = q - oh * g.OW;
      const int a0 = ow00 - g.pw;
      const int a0a = a0 & ~7;  // aligned floor; shift resolved at write
      const int ih = oh * g.sh - g.ph + r;
      if (ih >= 0 && ih < g.H && c < g.C && a0a >= 0 &&
          a0a + 16 + ((SS > 1) ? 8 : 0) <= g.W) {
        const short* src =
            (const short*)(x + in_n + (int64_t)c * HW + (int64_t)ih * g.W);
        bReg[u0][0] = *(const s16x8*)(src + a0a);
        bReg[u0][1] = *(const s16x8*)(src + a0a + 8);
        if (SS > 1) bReg[u0][2] = *(const s16x8*)(src + a0a + 16);
      }
    }
  };

  auto stage_write = [&](int it) {
    const int buf = it & 1;
    const int r = it / KT;
    const int kk0 = (it % KT) * BK;
#pragma unroll
    for (int pass = 0; pass < (ACH + 255) / 256; ++pass) {
      const int idx = pass * 256 + tid;
      if (ACH < 256 && idx >= ACH) break;
      const int row = idx >> 2;
      const int kc = (idx & 3) * 8;
      const int kout = k0out + row;
      short v[8];
      const int rem = CS - (kk0 + kc);
      if (kout < g.K && rem >= 8) {
        *(s16x8*)v = aReg[pass];
      } else if (kout < g.K && rem > 0) {
        const int64_t b = ((int64_t)kout * g.R + r) * CS + kk0 + kc;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          v[e] = (e < rem) ? ((const short*)w2)[b + e] : (short)0;
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = 0;
      }
      *(s16x8*)(ldsA(buf) + row * (BK + 8) + kc) = *(const s16x8*)v;
    }
    const int c_lo = kk0 / S;
#pragma unroll
    for (int u0 = 0; u0 < UNITS; ++u0) {
      const int u = u0 * 256 + tid;
      const int ci = u >> 4;
      const int pxc = u & 15;
      const int c = c_lo + ci;
      const int px0 = pxc * 8;
      const int q = q0 + px0;
      const int oh = q / g.OW;
      const int ow00 = q - oh * g.OW;
      const int a0 = ow00 - g.pw;
      const int a0a = a0 & ~7;
      const int d = a0 - a0a;
      const int ih = oh * g.sh - g.ph + r;
      const bool row_ok = (ih >= 0 && ih < g.H);
      const bool fast = row_ok && c < g.C && a0a >= 0 &&
                        a0a + 16 + ((SS > 1) ? 8 : 0) <= g.W;
      if (fast && ow00 + 8 <= g.OW) {
        short raw24[24];
        *(s16x8*)raw24 = bReg[u0][0];
        *(s16x8*)(raw24 + 8) = bReg[u0][1];
        if (SS > 1) *(s16x8*)(raw24 + 16) = bReg[u0][2];
        // uniform-scalar switch turns the runtime alignment shift into
        // compile-time indices (keeps raw/seg in registers)
        short raw[16];
        switch (d) {
#define SHIFT_CASE(D)                                                      \
  case D:                                                                  \
    _Pragma("unroll") for (int j = 0; j < 16; ++j) raw[j] = raw24[j + D];  \
    break;
          SHIFT_CASE(0)
          SHIFT_CASE(1)
          SHIFT_CASE(2)
          SHIFT_CASE(3)
          SHIFT_CASE(4)
          SHIFT_CASE(5)
          SHIFT_CASE(6)
          SHIFT_CASE(7)
#undef SHIFT_CASE
          default:
            break;
        }
#pragma unroll
        for (int ss = 0; ss < (SS > 0 ? SS : 1); ++ss) {
          const int kk = c * S + ss - kk0;
          if (kk < 0 || kk >= BK) continue;
          short v[8];
#pragma unroll
          for (int e = 0; e < 8; ++e) v[e] = raw[ss + e];
          const int base = (px0 >> 4) * 520 + ((kk >> 3) << 7) +
                           (((kk >> 2) & 1) << 6) + ((kk & 3) << 4) +
                           (px0 & 15);
          *(s16x8*)(ldsB(buf) + base) = *(const s16x8*)v;
        }
      } else {
        // edge path: synchronous scalar load+write (row borders)
        const short* src =
            (const short*)(x + in_n + (int64_t)c * HW + (int64_t)ih * g.W);
        const bool ok = row_ok && c < g.C;
        for (int ss = 0; ss < S; ++ss) {
          const int kk = c * S + ss - kk0;
          if (kk < 0 || kk >= BK) continue;
          short v[8];
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int col = a0 + ss + e;
            v[e] = (ok && col >= 0 && col < g.W) ? src[col] : (short)0;
          }
          const int base = (px0 >> 4) * 520 + ((kk >> 3) << 7) +
                           (((kk >> 2) & 1) << 6) + ((kk & 3) << 4) +
                           (px0 & 15);
          *(s16x8*)(ldsB(buf) + base) = *(const s16x8*)v;
        }
      }
    }
  };

  const int a_row0 = wm * 16 * MFRAG;
  const int b_px0 = wn * 64;

  stage_load(0);
  stage_write(0);
  if (total_it > 1) stage_load(1);

  for (int it = 0; it < total_it; ++it) {
    const int buf = it & 1;
    __syncthreads();
    if (it + 1 < total_it) stage_write(it + 1);
    if (it + 2 < total_it) stage_load(it + 2);
#pragma unroll
    for (int mf = 0; mf < MFRAG; ++mf) {
      const short* arow =
          ldsA(buf) + (a_row0 + mf * 16 + (lane & 15)) * (BK + 8) +
          ((lane >> 4) << 3);
      s16x8 afrag = *(const s16x8*)arow;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int pb = (b_px0 >> 4) + nf;
        __attribute__((address_space(3))) short* bbase =
            (__attribute__((address_space(3))) short*)(ldsB(buf)) + pb * 520 +
            ((lane >> 4) << 7) + ((lane & 15) << 2);
        s16x4 b0 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (__attribute__((address_space(3))) s16x4*)bbase);
        s16x4 b1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (__attribute__((address_space(3))) s16x4*)(bbase + 64));
        s16x8 bfrag;
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          bfrag[e] = b0[e];
          bfrag[e + 4] = b1[e];
        }
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[mf][nf], 0, 0, 0);
      }
    }
  }

  const int64_t out_n = (int64_t)n * g.K * OHW;
#pragma unroll
  for (int mf = 0; mf < MFRAG; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int kout = k0out + a_row0 + mf * 16 + ((lane >> 4) << 2) + reg;
      if (kout >= g.K) continue;
      const float b = bias ? bias[kout] : 0.f;
      const int64_t obase = out_n + (int64_t)kout * OHW;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int q = q0 + b_px0 + nf * 16 + (lane & 15);
        out[obase + q] = (bf16)(acc[mf][nf][reg] + b);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Backward-weight kernel: GW[Kout][CRS] += GO[Kout][px] * X[CRS][px]^T
// Both operands are pixel-contiguous rows -> plain b128 fragments.
// Tile 64x64, BKpx = 64 pixels per step, one (n, oh-row-chunk) per
// blockIdx.y slice, fp32 atomicAdd into the fp32 workspace.
// ---------------------------------------------------------------------------

#define WBM 64
#define WBN 64
#define WBK 64

__global__ __launch_bounds__(256) void conv_bwdw_kernel(
    const bf16* __restrict__ go, const bf16* __restrict__ x,
    float* __restrict__ gw, ConvGeom g) {
  const int m_tiles = (g.K + WBM - 1) / WBM;
  const int n_tiles = (g.CRS + WBN - 1) / WBN;
  int bid = blockIdx.x;
  const int mt = bid % m_tiles;
  bid /= m_tiles;
  const int nt = bid % n_tiles;
  const int slice = bid / n_tiles;  // (n, oh) slice index
  const int n = slice / g.OH;
  const int oh = slice % g.OH;

  const int kout0 = mt * WBM;
  const int crs0 = nt * WBN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;  // 4 waves (2x2): wave tile 32x32? -> use 2x2 of 16
  const int wm = wid >> 1, wn = wid & 1;

  __shared__ __attribute__((aligned(16))) short lds[WBM * (WBK + 8) +
                                                    WBN * (WBK + 8)];
  short* ldsA = lds;                       // GO [kout][px]
  short* ldsB = lds + WBM * (WBK + 8);     // X  [crs][px]

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)g.H * g.W;
  const int64_t in_n = (int64_t)n * g.C * HW;
  const int64_t go_row = ((int64_t)n * g.K) * g.OH * g.OW +
                         (int64_t)oh * g.OW;
  const int RS = g.R * g.S;

  for (int px0 = 0; px0 < g.OW; px0 += WBK) {
    // stage GO tile [WBM][WBK]: idx -> (row, px8)
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int idx = it * 256 + tid;  // 512 chunks of 8
      const int row = idx >> 3;
      const int pc = (idx & 7) * 8;
      short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      const int kout = kout0 + row;
      if (kout < g.K) {
        const bf16* src = go + go_row + (int64_t)kout * g.OH * g.OW;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int ow = px0 + pc + e;
          if (ow < g.OW) v[e] = ((const short*)src)[ow];
        }
      }
      short* dst = ldsA + row * (WBK + 8) + pc;
#pragma unroll
      for (int e = 0; e < 8; ++e) dst[e] = v[e];
    }
    // stage X tile [WBN][WBK]
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int idx = it * 256 + tid;
      const int row = idx >> 3;
      const int pc = (idx & 7) * 8;
      short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      const int k = crs0 + row;
      if (k < g.CRS) {
        const int c = k / RS;
        const int rs = k - c * RS;
        const int r = rs / g.S;
        const int s = rs - r * g.S;
        const int ih = oh * g.sh - g.ph + r;
        if (ih >= 0 && ih < g.H) {
          const bf16* src = x + in_n + c * HW + (int64_t)ih * g.W;
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int ow = px0 + pc + e;
            const int iw = ow * g.sw - g.pw + s;
            if (ow < g.OW && iw >= 0 && iw < g.W)
              v[e] = ((const short*)src)[iw];
          }
        }
      }
      short* dst = ldsB + row * (WBK + 8) + pc;
#pragma unroll
      for (int e = 0; e < 8; ++e) dst[e] = v[e];
    }
    __syncthreads();

    // wave (wm, wn) computes 32x32: 2x2 fragments of 16x16, k = WBK
#pragma unroll
    for (int kk = 0; kk < WBK; kk += 32) {
#pragma unroll
      for (int mf = 0; mf < 2; ++mf) {
        const short* arow = ldsA + (wm * 32 + mf * 16 + (lane & 15)) *
                                       (WBK + 8) +
                            kk + ((lane >> 4) << 3);
        s16x8 afrag = *(const s16x8*)arow;
#pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
          const short* brow = ldsB + (wn * 32 + nf * 16 + (lane & 15)) *
                                         (WBK + 8) +
                              kk + ((lane >> 4) << 3);
          s16x8 bfrag = *(const s16x8*)brow;
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, bfrag, acc[mf][nf], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // accumulate into gw: D[m][n] position: col = lane&15, row = (lane>>4)*4+reg
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int kout = kout0 + wm * 32 + mf * 16 + ((lane >> 4) << 2) + reg;
      if (kout >= g.K) continue;
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        const int crs = crs0 + wn * 32 + nf * 16 + (lane & 15);
        if (crs < g.CRS)
          atomicAdd(&gw[(int64_t)kout * g.CRS + crs], acc[mf][nf][reg]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

static ConvGeom make_geom(const torch::Tensor& x, const torch::Tensor& w,
                          int64_t sh, int64_t sw, int64_t ph, int64_t pw) {
  ConvGeom g;
  g.N = (int)x.size(0);
  g.C = (int)x.size(1);
  g.H = (int)x.size(2);
  g.W = (int)x.size(3);
  g.K = (int)w.size(0);
  g.R = (int)w.size(2);
  g.S = (int)w.size(3);
  g.sh = (int)sh; g.sw = (int)sw; g.ph = (int)ph; g.pw = (int)pw;
  g.OH = (g.H + 2 * g.ph - g.R) / g.sh + 1;
  g.OW = (g.W + 2 * g.pw - g.S) / g.sw + 1;
  g.CRS = g.C * g.R * g.S;
  g.row_tiles = (g.OW + BN - 1) / BN;
  return g;
}

torch::Tensor conv_fwd(torch::Tensor x, torch::Tensor w,
                       c10::optional<torch::Tensor> bias, int64_t sh,
                       int64_t sw, int64_t ph, int64_t pw) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16,
              "conv_fwd: bf16 only");
  TORCH_CHECK(x.size(1) == w.size(1), "channel mismatch");
  ConvGeom g = make_geom(x, w, sh, sw, ph, pw);
  auto out = torch::empty({g.N, g.K, g.OH, g.OW},
                          x.options().dtype(torch::kBFloat16));
  const float* bptr = nullptr;
  torch::Tensor b32;
  if (bias.has_value()) {
    b32 = bias->to(torch::kFloat).contiguous();
    bptr = b32.data_ptr<float>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  const int64_t OHW = (int64_t)g.OH * g.OW;
  if (g.sh == 1 && g.sw == 1 && g.C * g.S >= BK && OHW % BN == 0 &&
      g.OW % 8 == 0) {
    // v2: row-pass kernel with [K][R][C*S]-permuted weights; global
    // pixel axis + M-adaptive BM (32/64/128 by K)
    auto w2 = w.view({g.K, g.C, g.R, g.S})
                  .permute({0, 2, 1, 3})
                  .reshape({g.K, g.R, (int64_t)g.C * g.S})
                  .contiguous();
    const int mfrag = (g.K <= 32) ? 1 : (g.K <= 64) ? 2 : 4;
    const int m_tiles2 = (g.K + 32 * mfrag - 1) / (32 * mfrag);
    const int64_t blocks2 = (int64_t)m_tiles2 * (OHW / BN) * g.N;
    TORCH_CHECK(blocks2 > 0 && blocks2 < (1LL << 31), "grid too large");
    auto launch_v2 = [&](auto* kern) {
      hipLaunchKernelGGL(kern, dim3((uint32_t)blocks2), dim3(256), 0,
                         stream.stream(), (const bf16*)x.data_ptr(),
                         (const bf16*)w2.data_ptr(), bptr,
                         (bf16*)out.data_ptr(), g);
    };
    switch (g.S * 10 + mfrag) {
      case 31: launch_v2(conv_fwd_v2_kernel<3, 1>); break;
      case 32: launch_v2(conv_fwd_v2_kernel<3, 2>); break;
      case 34: launch_v2(conv_fwd_v2_kernel<3, 4>); break;
      case 71: launch_v2(conv_fwd_v2_kernel<7, 1>); break;
      case 72: launch_v2(conv_fwd_v2_kernel<7, 2>); break;
      case 74: launch_v2(conv_fwd_v2_kernel<7, 4>); break;
      case 11: launch_v2(conv_fwd_v2_kernel<1, 1>); break;
      case 12: launch_v2(conv_fwd_v2_kernel<1, 2>); break;
      case 14: launch_v2(conv_fwd_v2_kernel<1, 4>); break;
      default:
        if (mfrag == 1) launch_v2(conv_fwd_v2_kernel<0, 1>);
        else if (mfrag == 2) launch_v2(conv_fwd_v2_kernel<0, 2>);
        else launch_v2(conv_fwd_v2_kernel<0, 4>);
    }
    return out;
  }
  const int m_tiles = (g.K + BM - 1) / BM;
  const int64_t blocks = (int64_t)m_tiles * g.row_tiles * g.OH * g.N;
  TORCH_CHECK(blocks < (1LL << 31), "grid too large");
  hipLaunchKernelGGL(conv_fwd_kernel, dim3((uint32_t)blocks), dim3(256), 0,
                     stream.stream(), (const bf16*)x.data_ptr(),
                     (const bf16*)w.data_ptr(), bptr, (bf16*)out.data_ptr(),
                     g);
  return out;
}

torch::Tensor conv_bwd_weight(torch::Tensor go, torch::Tensor x,
                              int64_t R, int64_t S, int64_t sh, int64_t sw,
                              int64_t ph, int64_t pw) {
  TORCH_CHECK(go.is_cuda() && go.is_contiguous() && x.is_contiguous());
  TORCH_CHECK(go.scalar_type() == torch::kBFloat16 &&
              x.scalar_type() == torch::kBFloat16);
  const int K = (int)go.size(1), C = (int)x.size(1);
  ConvGeom g;
  g.N = (int)x.size(0); g.C = C; g.H = (int)x.size(2); g.W = (int)x.size(3);
  g.K = K; g.R = (int)R; g.S = (int)S;
  g.sh = (int)sh; g.sw = (int)sw; g.ph = (int)ph; g.pw = (int)pw;
  g.OH = (int)go.size(2); g.OW = (int)go.size(3);
  g.CRS = C * (int)R * (int)S;
  g.row_tiles = 0;
  auto gw = torch::zeros({(int64_t)K, (int64_t)g.CRS},
                         x.options().dtype(torch::kFloat));
  const int m_tiles = (K + WBM - 1) / WBM;
  const int n_tiles = (g.CRS + WBN - 1) / WBN;
  const int64_t blocks = (int64_t)m_tiles * n_tiles * g.N * g.OH;
  TORCH_CHECK(blocks < (1LL << 31), "grid too large");
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(conv_bwdw_kernel, dim3((uint32_t)blocks), dim3(256), 0,
                     stream.stream(), (const bf16*)go.data_ptr(),
                     (const bf16*)x.data_ptr(), gw.data_ptr<float>(), g);
  return gw.view({(int64_t)K, (int64_t)C, R, S});
}

// ---------------------------------------------------------------------------
// ds_read_b64_tr_b16 semantics probe: LDS holds arange shorts; each lane
// does one tr read with a configurable per-lane address pattern; output
// [64][4] shows exactly which LDS elements land in which lane/elem.
// ---------------------------------------------------------------------------

__global__ void tr16_probe_kernel(short* __restrict__ out, int mode) {
  __shared__ __attribute__((aligned(16))) short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  if (threadIdx.x < 64) {
    const int l = threadIdx.x;
    int off;
    if (mode == 0) off = 0;                               // uniform
    else if (mode == 1) off = (l & 15) + ((l >> 4) << 6); // my assumed map
    else if (mode == 2) off = l * 4;                      // linear x4
    else off = (l & 15) * 4 + ((l >> 4) << 6);            // 4-elem rows
    s16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (__attribute__((address_space(3))) s16x4*)(
            (__attribute__((address_space(3))) short*)lds + off));
#pragma unroll
    for (int e = 0; e < 4; ++e) out[l * 4 + e] = v[e];
  }
}

torch::Tensor tr16_probe(int64_t mode) {
  auto out = torch::zeros({64, 4}, torch::TensorOptions()
                                       .dtype(torch::kShort)
                                       .device(torch::kCUDA));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream.stream(),
                     out.data_ptr<short>(), (int)mode);
  return out;
}

}  // namespace conv_mfma

void register_conv_mfma(pybind11::module_& m) {
  m.def("conv_fwd", &conv_mfma::conv_fwd,
        "implicit-GEMM MFMA conv forward (bf16 NCHW)");
  m.def("conv_bwd_weight", &conv_mfma::conv_bwd_weight,
        "implicit-GEMM MFMA conv weight gradient (fp32 out)");
  m.def("tr16_probe", &conv_mfma::tr16_probe,
        "ds_read_b64_tr_b16 lane-mapping probe");
}
