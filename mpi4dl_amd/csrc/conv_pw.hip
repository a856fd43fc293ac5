// Pointwise (1x1) convolution as a streaming MFMA GEMM for gfx950.
//
// OUT[K][P] = W[K][C] @ X[C][P],  P = N*OH*OW (NCHW: pixels contiguous
// per channel). These shapes are BANDWIDTH-bound on MI355X (arithmetic
// intensity K*C/(K+C) ~ 50-200 FLOP/B vs the ~400 machine balance), so
// the design goal is streaming efficiency at high occupancy, not MFMA
// peak:
//  * B (pixels) is staged global->LDS with `global_load_lds` dwordx4
//    (16 B/lane), source addresses pre-swizzled so the lane-linear LDS
//    image IS the ds_read_b64_tr_b16 fragment layout (guide §5 "swizzle
//    on the SOURCE address" rule) — no ds_write pass, no repack;
//  * A (weights) is tiny and L2-resident: each lane loads its MFMA
//    fragment straight from global (no LDS round-trip — guide §5
//    "operand streamed once and not shared" row);
//  * M-adaptive tiles: BM in {32, 64, 128} via MFRAG template so K=26
//    or K=52 layers (AmoebaNet c/4 bottlenecks) don't waste 60-80% of
//    the M tile;
//  * small LDS (2 x 8.3 KB) -> 8 waves/SIMD occupancy does the latency
//    hiding (guide: "pure HBM streaming at high occupancy" regime).
//
// Also used for the 1x1 stride-1 BACKWARD-DATA (gx = pw(go, w^T)) and
// the stride-2 backward-data via the scatter2 epilogue
// (gx[:, :, ::2, ::2] = pw(go, w^T), rest exact zeros) — VERDICT r1
// item 9: the conv triple fully in-house.
//
// Reference behaviour replaced: cuDNN/MIOpen 1x1 conv under torchgems
// (/root/reference/src/torchgems/spatial.py:1027); in the round-1
// profile these were ~20-30% of the timed window as hipBLASLt Cijk_*
// GEMMs + batched_transpose glue.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cstdint>

namespace conv_pw {

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(8))) short s16x8;

#define PW_BN 128       // pixels per block
#define PW_BK 32        // channels per k-step
#define LDSB_BLK 520    // shorts per 16-px LDS block (512 + 8 pad)

struct PwGeom {
  int N, C, K;
  int HW_in;        // input pixels per image (H*W)
  int OH, OW;       // output dims (per image)
  int sh, sw;       // stride (1 or 2); pad is always 0 for 1x1
  int W_in;         // input row length
  int osh, osw;     // output scatter strides (1 = dense)
  int oW;           // output row length in elements (for scatter)
  int oHW;          // output pixels per image in the OUT TENSOR
};

// lane -> (k, b) map such that LDS offset l*8 shorts == tr16 layout
// offset of (k, px0=8b): off/8 = 16*(k>>3) + 8*((k>>2)&1) + 2*(k&3) + b
__device__ __forceinline__ void lane_kb(int l, int& k, int& b) {
  k = ((l >> 4) << 3) + (((l >> 3) & 1) << 2) + ((l >> 1) & 3);
  b = l & 1;
}

// ---------------------------------------------------------------------------
// Forward kernel. MFRAG = 16-row fragments per wave (BM = 2*16*MFRAG).
// 256 threads = 4 waves as 2x2; wave tile (16*MFRAG) x 64.
// ---------------------------------------------------------------------------

template <int MFRAG, int SW>
__global__ __launch_bounds__(256) void pw_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    const float* __restrict__ bias, bf16* __restrict__ out, PwGeom g) {
  constexpr int BM = 32 * MFRAG;
  const int m_tiles = (g.K + BM - 1) / BM;
  // XCD-aware bijective swizzle (guide T1)
  const int nwg = gridDim.x;
  const int q8 = nwg >> 3, r8 = nwg & 7;
  const int xcd = blockIdx.x & 7, sub = blockIdx.x >> 3;
  int bid = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + sub;
  const int mt = bid % m_tiles;
  const int pt = bid / m_tiles;          // pixel-tile index
  const int OHW = g.OH * g.OW;
  const int ptiles_per_img = OHW / PW_BN;
  const int n = pt / ptiles_per_img;
  const int q0 = (pt - n * ptiles_per_img) * PW_BN;  // first out pixel

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;
  const int k0out = mt * BM;

  __shared__ __attribute__((aligned(16))) short lds[2 * 8 * LDSB_BLK];
  auto ldsB = [&](int buf) { return lds + buf * 8 * LDSB_BLK; };

  f32x4 acc[MFRAG][4];
#pragma unroll
  for (int i = 0; i < MFRAG; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int64_t in_n = (int64_t)n * g.C * g.HW_in;
  const int ksteps = (g.C + PW_BK - 1) / PW_BK;
  const int full_ksteps = g.C / PW_BK;

  int lk, lb;
  lane_kb(lane, lk, lb);

  // per-lane source pixel offset for the glds path (stride 1 only):
  // wave wid stages blocks pb = wid*2 and wid*2+1
  // source = x[n][c0+lk][q0 + pb*16 + 8*lb .. +8]
  const int64_t src_base =
      in_n + (int64_t)lk * g.HW_in + q0 + lb * 8;  // + c0*HW + pb*16

  auto stage_glds = [&](int buf, int c0) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int pb = wid * 2 + i;
      const bf16* src = x + src_base + (int64_t)c0 * g.HW_in + pb * 16;
      auto* dst = (__attribute__((address_space(3))) void*)(ldsB(buf) +
                                                            pb * LDSB_BLK);
      __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1)))
                                        unsigned int*)src,
                                       (__attribute__((address_space(3)))
                                        unsigned int*)dst,
                                       16, 0, 0);
    }
  };

  // register-staged path: tail k-step (c0+32 > C) or stride-2 gather.
  // 512 (k, px8) chunks -> 2 per thread, zero-filled beyond C.
  auto stage_reg = [&](int buf, int c0) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int idx = it * 256 + tid;
      const int kk = idx & 31;
      const int pxc = idx >> 5;
      const int px0 = pxc * 8;
      const int c = c0 + kk;
      short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      if (c < g.C) {
        if (SW == 1) {
          const short* src =
              (const short*)(x + in_n + (int64_t)c * g.HW_in + q0 + px0);
          *(s16x8*)v = *(const s16x8*)src;
        } else {
          // stride 2: out pixel q -> input (2*oh)*W_in + 2*ow.
          // chunk stays inside one out row (OW % 8 == 0).
          const int q = q0 + px0;
          const int oh = q / g.OW;
          const int ow = q - oh * g.OW;
          const short* src = (const short*)(x + in_n + (int64_t)c * g.HW_in +
                                            (int64_t)(oh * g.sh) * g.W_in +
                                            ow * 2);
          short raw[16];
          *(s16x8*)raw = *(const s16x8*)src;
          *(s16x8*)(raw + 8) = *(const s16x8*)(src + 8);
#pragma unroll
          for (int e = 0; e < 8; ++e) v[e] = raw[2 * e];
        }
      }
      const int base = (px0 >> 4) * LDSB_BLK + ((kk >> 3) << 7) +
                       (((kk >> 2) & 1) << 6) + ((kk & 3) << 4) + (px0 & 15);
      *(s16x8*)(ldsB(buf) + base) = *(const s16x8*)v;
    }
  };

  auto stage = [&](int buf, int step) {
    const int c0 = step * PW_BK;
    if (SW == 1 && step < full_ksteps)
      stage_glds(buf, c0);
    else
      stage_reg(buf, c0);
  };

  // A fragments straight from global (L2-resident weights).
  // fragment (mf): rows k0out + wm*16*MFRAG + mf*16 + (lane&15),
  // k-chunk c0 + (lane>>4)*8.
  const int arow_base = k0out + wm * 16 * MFRAG + (lane & 15);
  const int acol = (lane >> 4) << 3;

  stage(0, 0);

  for (int step = 0; step < ksteps; ++step) {
    const int buf = step & 1;
    const int c0 = step * PW_BK;
    __syncthreads();
    if (step + 1 < ksteps) stage(buf ^ 1, step + 1);

    s16x8 afrag[MFRAG];
    const bool full = (c0 + PW_BK <= g.C);
#pragma unroll
    for (int mf = 0; mf < MFRAG; ++mf) {
      int row = arow_base + mf * 16;
      if (row >= g.K) row = g.K - 1;  // clamp: masked at epilogue
      const short* ap = (const short*)w + (int64_t)row * g.C + c0 + acol;
      if (full) {
        afrag[mf] = *(const s16x8*)ap;
      } else {
        short v[8];
#pragma unroll
        for (int e = 0; e < 8; ++e)
          v[e] = (c0 + acol + e < g.C) ? ap[e] : (short)0;
        afrag[mf] = *(const s16x8*)v;
      }
    }
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int pb = wn * 4 + nf;
      __attribute__((address_space(3))) short* bbase =
          (__attribute__((address_space(3))) short*)(ldsB(buf)) +
          pb * LDSB_BLK + ((lane >> 4) << 7) + ((lane & 15) << 2);
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
#pragma unroll
      for (int mf = 0; mf < MFRAG; ++mf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[mf], bfrag, acc[mf][nf], 0, 0, 0);
    }
  }

  // epilogue: bias + bf16 store (dense or scatter2)
  const int64_t out_n = (int64_t)n * g.K * g.oHW;
#pragma unroll
  for (int mf = 0; mf < MFRAG; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int kout = k0out + wm * 16 * MFRAG + mf * 16 +
                       ((lane >> 4) << 2) + reg;
      if (kout >= g.K) continue;
      const float bv = bias ? bias[kout] : 0.f;
      const int64_t obase = out_n + (int64_t)kout * g.oHW;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int q = q0 + wn * 64 + nf * 16 + (lane & 15);
        int64_t oq;
        if (g.osh == 1 && g.osw == 1) {
          oq = q;
        } else {
          const int oh = q / g.OW;
          const int ow = q - oh * g.OW;
          oq = (int64_t)(oh * g.osh) * g.oW + ow * g.osw;
        }
        out[obase + oq] = (bf16)(acc[mf][nf][reg] + bv);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Fat-shape kernel: C >= 256 and K >= 128 make the GEMM compute-bound,
// so this variant uses the guide's "step 3" GEMM structure (§5 ladder,
// ~874 TF class at 4096^3): 128x128 tile, BK=64, BOTH operands staged
// global->LDS with `global_load_lds` dwordx4:
//  * A (weights) lands lane-linear as [128][64] with the st_16x32 XOR
//    swizzle applied on the SOURCE address (k-chunk ^= 2 for lanes in
//    the upper 512B of each 1 KB group) so ds_read_b128 fragment reads
//    are bank-conflict-free;
//  * B (pixels) lands in the same tr16 block layout as the skinny
//    kernel (16 blocks of 512 shorts + 8 pad per buffer).
// ---------------------------------------------------------------------------

template <int BUFS>
__global__ __launch_bounds__(256) void pw_fat_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    const float* __restrict__ bias, bf16* __restrict__ out, PwGeom g) {
  constexpr int BM = 128;
  constexpr int FBK = 64;  // channels per K-step (2 x 32 sub-steps)
  const int m_tiles = (g.K + BM - 1) / BM;
  const int nwg = gridDim.x;
  const int q8 = nwg >> 3, r8 = nwg & 7;
  const int xcd = blockIdx.x & 7, sub = blockIdx.x >> 3;
  int bid = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + sub;
  const int mt = bid % m_tiles;
  const int pt = bid / m_tiles;
  const int OHW = g.OH * g.OW;
  const int ptiles_per_img = OHW / PW_BN;
  const int n = pt / ptiles_per_img;
  const int q0 = (pt - n * ptiles_per_img) * PW_BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;
  const int k0out = mt * BM;

  // LDS: A BUFS x [128][64] linear (16 KB each), B BUFS x 16 blk x 520
  __shared__ __attribute__((aligned(16))) short lds[BUFS * (BM * FBK) +
                                                    BUFS * 16 * LDSB_BLK];
  auto ldsA = [&](int buf) { return lds + buf * (BM * FBK); };
  auto ldsB = [&](int buf) {
    return lds + BUFS * (BM * FBK) + buf * 16 * LDSB_BLK;
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int64_t in_n = (int64_t)n * g.C * g.HW_in;
  const int ksteps = (g.C + FBK - 1) / FBK;
  const int full_ksteps = g.C / FBK;

  int lk, lb;
  lane_kb(lane, lk, lb);
  const int64_t bsrc_base = in_n + (int64_t)lk * g.HW_in + q0 + lb * 8;

  // A glds: 16 instructions (4 per wave); instruction i covers rows
  // [i*8, i*8+8); lane l -> row i*8 + (l>>3), k-chunk (l&7) with the
  // XOR source swizzle (chunk ^= 2 on the upper half-group).
  const int a_row_l = (lane >> 3);
  const int a_chunk = (lane & 7) ^ (((lane >> 5) & 1) << 1);

  auto stage_glds = [&](int buf, int c0) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int inst = wid * 4 + i;  // 0..15
      int row = k0out + inst * 8 + a_row_l;
      if (row >= g.K) row = g.K - 1;  // clamp; masked at epilogue
      const bf16* srcA = w + (int64_t)row * g.C + c0 + a_chunk * 8;
      auto* dstA = (__attribute__((address_space(3))) void*)(ldsA(buf) +
                                                             inst * 512);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcA,
          (__attribute__((address_space(3))) unsigned int*)dstA, 16, 0, 0);
    }
#pragma unroll
    for (int ksub = 0; ksub < 2; ++ksub) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int pb = wid * 2 + i;
        const bf16* srcB =
            x + bsrc_base + (int64_t)(c0 + ksub * 32) * g.HW_in + pb * 16;
        auto* dstB = (__attribute__((address_space(3))) void*)(
            ldsB(buf) + (pb * 2 + ksub) * LDSB_BLK);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)srcB,
            (__attribute__((address_space(3))) unsigned int*)dstB, 16, 0, 0);
      }
    }
  };

  // register tail path (C % 64): zero-filled, plain stores
  auto stage_tail = [&](int buf, int c0) {
    // A: 512 chunks of 8 shorts per 32-sub-step x2 -> 4 per thread
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = it * 256 + tid;  // 0..1023 over [128 rows][8 chunks]
      const int row = idx >> 3;
      const int ch = idx & 7;
      int krow = k0out + row;
      if (krow >= g.K) krow = g.K - 1;
      short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      const int cbase = c0 + ch * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        if (cbase + e < g.C)
          v[e] = ((const short*)w)[(int64_t)krow * g.C + cbase + e];
      // match the glds image incl. swizzle: within each 8-row group,
      // rows 4..7 store chunk ch at position ch^2
      const int grp = row >> 3;
      const int rl = row & 7;
      const int sch = (rl >= 4) ? (ch ^ 2) : ch;
      *(s16x8*)(ldsA(buf) + grp * 512 + rl * 64 + sch * 8) = *(const s16x8*)v;
    }
    // B: two 32-sub-steps of 512 chunks -> 4 per thread
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = it * 256 + tid;
      const int ksub = idx >> 9;
      const int kk = idx & 31;
      const int pxc = (idx >> 5) & 15;
      const int px0 = pxc * 8;
      const int c = c0 + ksub * 32 + kk;
      short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      if (c < g.C) {
        const short* src =
            (const short*)(x + in_n + (int64_t)c * g.HW_in + q0 + px0);
        *(s16x8*)v = *(const s16x8*)src;
      }
      const int base = ((px0 >> 4) * 2 + ksub) * LDSB_BLK + ((kk >> 3) << 7) +
                       (((kk >> 2) & 1) << 6) + ((kk & 3) << 4) + (px0 & 15);
      *(s16x8*)(ldsB(buf) + base) = *(const s16x8*)v;
    }
  };

  auto stage = [&](int buf, int step) {
    if (step < full_ksteps)
      stage_glds(buf, step * FBK);
    else
      stage_tail(buf, step * FBK);
  };

  const int a_row0 = wm * 64 + (lane & 15);
  const int a_ch_rd = (lane >> 4);  // 16-B chunk within 32-k sub-step

  // BUFS == 2: plain double-buffer (glds flies during compute, drained
  // by the vmcnt(0) inside __syncthreads).
  // BUFS == 3: the guide's span pipeline — one whole K-step stays in
  // flight ACROSS the raw barrier. Each wave issues 8 glds per step.
  // Ordering per iteration: wait MY glds for this step (counted vmcnt,
  // leaving the next step's in flight) -> raw barrier (now EVERY
  // wave's loads for this buffer have landed) -> issue step+2's glds
  // into the buffer all waves just stopped reading -> compute.
  if (BUFS == 3) {
    stage(0, 0);
    if (1 < full_ksteps) stage(1, 1);
  } else {
    stage(0, 0);
  }

  for (int step = 0; step < ksteps; ++step) {
    const int buf = (BUFS == 3) ? (step % 3) : (step & 1);
    if (BUFS == 3 && step < full_ksteps) {
      if (step + 1 < full_ksteps)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      if (step + 2 < full_ksteps) stage((step + 2) % 3, step + 2);
    } else if (BUFS == 3) {
      // tail step (C % 64): plain fences around the register staging
      __syncthreads();
      stage(buf, step);
      __syncthreads();
    } else {
      __syncthreads();
      if (step + 1 < ksteps) stage(buf ^ 1, step + 1);
    }
#pragma unroll
    for (int ksub = 0; ksub < 2; ++ksub) {
      s16x8 afrag[4];
#pragma unroll
      for (int mf = 0; mf < 4; ++mf) {
        const int row = a_row0 + mf * 16;
        int ch = ksub * 4 + a_ch_rd;
        ch ^= ((row >> 2) & 1) << 1;  // st_16x32 read-side swizzle
        afrag[mf] = *(const s16x8*)(ldsA(buf) + ((row >> 3) * 512) +
                                    (row & 7) * 64 + ch * 8);
      }
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int pb = wn * 4 + nf;
        __attribute__((address_space(3))) short* bbase =
            (__attribute__((address_space(3))) short*)(ldsB(buf)) +
            (pb * 2 + ksub) * LDSB_BLK + ((lane >> 4) << 7) +
            ((lane & 15) << 2);
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
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mf], bfrag, acc[mf][nf], 0, 0, 0);
      }
    }
  }

  const int64_t out_n = (int64_t)n * g.K * g.oHW;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int kout = k0out + wm * 64 + mf * 16 + ((lane >> 4) << 2) + reg;
      if (kout >= g.K) continue;
      const float bv = bias ? bias[kout] : 0.f;
      const int64_t obase = out_n + (int64_t)kout * g.oHW;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int q = q0 + wn * 64 + nf * 16 + (lane & 15);
        out[obase + q] = (bf16)(acc[mf][nf][reg] + bv);
      }
    }
  }
}


// ---------------------------------------------------------------------------
// 256x256-tile variant for the BIGGEST fat shapes (C >= 512, K >= 256,
// e.g. the concat-input reduce convs, C up to 6656): doubling the tile
// doubles MFMA work per staged byte (128 vs 64 FLOP/B) and halves the
// A re-reads. 512 threads = 8 waves (2M x 4N), wave tile 128x64,
// BK=64, 2-buffer dual glds (A linear + XOR swizzle, B tr16 blocks).
// LDS 130.6 KB -> 1 block/CU; these shapes' x operand is L2/L3
// resident, so the next-step prefetch hides the shorter latencies.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(512) void pw_fat256_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    const float* __restrict__ bias, bf16* __restrict__ out, PwGeom g) {
  constexpr int BM = 256;
  constexpr int BN2 = 256;
  constexpr int FBK = 64;
  const int m_tiles = (g.K + BM - 1) / BM;
  const int nwg = gridDim.x;
  const int q8 = nwg >> 3, r8 = nwg & 7;
  const int xcd = blockIdx.x & 7, sub = blockIdx.x >> 3;
  int bid = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + sub;
  const int mt = bid % m_tiles;
  const int pt = bid / m_tiles;
  const int OHW = g.OH * g.OW;
  const int ptiles_per_img = OHW / BN2;
  const int n = pt / ptiles_per_img;
  const int q0 = (pt - n * ptiles_per_img) * BN2;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;        // 0..7
  const int wm = wid >> 2, wn = wid & 3;
  const int k0out = mt * BM;

  __shared__ __attribute__((aligned(16))) short lds[2 * (BM * FBK) +
                                                    2 * 32 * LDSB_BLK];
  auto ldsA = [&](int buf) { return lds + buf * (BM * FBK); };
  auto ldsB = [&](int buf) {
    return lds + 2 * (BM * FBK) + buf * 32 * LDSB_BLK;
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int64_t in_n = (int64_t)n * g.C * g.HW_in;
  const int ksteps = (g.C + FBK - 1) / FBK;
  const int full_ksteps = g.C / FBK;

  int lk, lb;
  lane_kb(lane, lk, lb);
  const int64_t bsrc_base = in_n + (int64_t)lk * g.HW_in + q0 + lb * 8;
  const int a_row_l = (lane >> 3);
  const int a_chunk = (lane & 7) ^ (((lane >> 5) & 1) << 1);

  // one paired (A, B) glds issue — slot i of this wave's 4. Issues are
  // SPREAD one per MFMA cluster inside the compute loop: 8 back-to-back
  // LDS-DMA issues per wave cost ~as much as the whole MFMA phase
  // (guide price list), and the PMC probe showed 50% parked waves with
  // the batched form.
  auto stage_one = [&](int buf, int c0, int i) {
    const int inst = wid * 4 + i;  // 0..31: A rows inst*8..+8
    int row = k0out + inst * 8 + a_row_l;
    if (row >= g.K) row = g.K - 1;
    const bf16* srcA = w + (int64_t)row * g.C + c0 + a_chunk * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)srcA,
        (__attribute__((address_space(3))) unsigned int*)(
            (__attribute__((address_space(3))) void*)(ldsA(buf) +
                                                      inst * 512)),
        16, 0, 0);
    // B: inst -> (pxblk, ksub)
    const int pxblk = inst >> 1;
    const int ksub = inst & 1;
    const bf16* srcB =
        x + bsrc_base + (int64_t)(c0 + ksub * 32) * g.HW_in + pxblk * 16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)srcB,
        (__attribute__((address_space(3))) unsigned int*)(
            (__attribute__((address_space(3))) void*)(
                ldsB(buf) + (pxblk * 2 + ksub) * LDSB_BLK)),
        16, 0, 0);
  };

  auto stage_glds = [&](int buf, int c0) {
#pragma unroll
    for (int i = 0; i < 4; ++i) stage_one(buf, c0, i);
  };

  auto stage_tail = [&](int buf, int c0) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = it * 512 + tid;  // 2048 A chunks
      const int row = idx >> 3;
      const int ch = idx & 7;
      int krow = k0out + row;
      if (krow >= g.K) krow = g.K - 1;
      short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      const int cbase = c0 + ch * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        if (cbase + e < g.C)
          v[e] = ((const short*)w)[(int64_t)krow * g.C + cbase + e];
      const int grp = row >> 3;
      const int rl = row & 7;
      const int sch = (rl >= 4) ? (ch ^ 2) : ch;
      *(s16x8*)(ldsA(buf) + grp * 512 + rl * 64 + sch * 8) = *(const s16x8*)v;
    }
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = it * 512 + tid;  // 2048 B chunks
      const int ksub = idx >> 10;
      const int kk = idx & 31;
      const int pxc = (idx >> 5) & 31;
      const int px0 = pxc * 8;
      const int c = c0 + ksub * 32 + kk;
      short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      if (c < g.C) {
        const short* src =
            (const short*)(x + in_n + (int64_t)c * g.HW_in + q0 + px0);
        *(s16x8*)v = *(const s16x8*)src;
      }
      const int base = ((px0 >> 4) * 2 + ksub) * LDSB_BLK + ((kk >> 3) << 7) +
                       (((kk >> 2) & 1) << 6) + ((kk & 3) << 4) + (px0 & 15);
      *(s16x8*)(ldsB(buf) + base) = *(const s16x8*)v;
    }
  };

  auto stage = [&](int buf, int step) {
    if (step < full_ksteps)
      stage_glds(buf, step * FBK);
    else
      stage_tail(buf, step * FBK);
  };

  stage(0, 0);

  const int a_row0 = wm * 128 + (lane & 15);
  const int a_ch_rd = (lane >> 4);

  for (int step = 0; step < ksteps; ++step) {
    const int buf = step & 1;
    __syncthreads();
    // measured: spreading the 8 glds issues one-per-MFMA-cluster was
    // 0-9% SLOWER (the runtime branch inside the unrolled loop defeats
    // hipcc's scheduling — guide §5.4 trap 4c); batched staging stays
    if (step + 1 < ksteps) stage(buf ^ 1, step + 1);
#pragma unroll
    for (int ksub = 0; ksub < 2; ++ksub) {
      s16x8 afrag[8];
#pragma unroll
      for (int mf = 0; mf < 8; ++mf) {
        const int row = a_row0 + mf * 16;
        int ch = ksub * 4 + a_ch_rd;
        ch ^= ((row >> 2) & 1) << 1;
        afrag[mf] = *(const s16x8*)(ldsA(buf) + ((row >> 3) * 512) +
                                    (row & 7) * 64 + ch * 8);
      }
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int pb = wn * 4 + nf;
        __attribute__((address_space(3))) short* bbase =
            (__attribute__((address_space(3))) short*)(ldsB(buf)) +
            (pb * 2 + ksub) * LDSB_BLK + ((lane >> 4) << 7) +
            ((lane & 15) << 2);
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
#pragma unroll
        for (int mf = 0; mf < 8; ++mf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mf], bfrag, acc[mf][nf], 0, 0, 0);
      }
    }
  }

  const int64_t out_n = (int64_t)n * g.K * g.oHW;
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int kout = k0out + wm * 128 + mf * 16 + ((lane >> 4) << 2) + reg;
      if (kout >= g.K) continue;
      const float bv = bias ? bias[kout] : 0.f;
      const int64_t obase = out_n + (int64_t)kout * g.oHW;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int q = q0 + wn * 64 + nf * 16 + (lane & 15);
        out[obase + q] = (bf16)(acc[mf][nf][reg] + bv);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 1x1 backward-weight: GW[K][C] = sum_p GO[K][p] * X[C][p].
// Both operands are PIXEL-contiguous rows, so both stage exactly like
// the fat kernel's A operand (linear [128][64] glds + XOR swizzle) and
// both fragments are plain ds_read_b128 — MFMA with B read as A-style
// k-major. Pixel-slab split-K: each block reduces one slab of one
// image and atomically adds its fp32 128x128 tile once.
// Replaces conv_bwdw for 1x1 (whose per-output-row grid does
// m*n*N*OH blocks of 64-px reductions with per-element atomics).
// ---------------------------------------------------------------------------

struct BwGeom {
  int Nimg, K, C;
  int OHW;     // pixels per image
  int slab;    // pixels per block (divides OHW)
};

template <int BUFS>
__global__ __launch_bounds__(256) void pw_bwdw_kernel(
    const bf16* __restrict__ go, const bf16* __restrict__ x,
    float* __restrict__ gw, BwGeom g) {
  constexpr int BM = 128, BNW = 128, FBK = 64;
  const int m_tiles = (g.K + BM - 1) / BM;
  const int n_tiles = (g.C + BNW - 1) / BNW;
  int bid = blockIdx.x;
  const int mt = bid % m_tiles;
  bid /= m_tiles;
  const int nt = bid % n_tiles;
  bid /= n_tiles;
  const int slabs_per_img = g.OHW / g.slab;
  const int img = bid / slabs_per_img;
  const int p0 = (bid - img * slabs_per_img) * g.slab;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;
  const int k0out = mt * BM;
  const int c0out = nt * BNW;

  __shared__ __attribute__((aligned(16))) short lds[2 * BUFS * (BM * FBK)];
  auto ldsA = [&](int buf) { return lds + buf * (BM * FBK); };
  auto ldsB = [&](int buf) { return lds + (BUFS + buf) * (BM * FBK); };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int64_t go_img = (int64_t)img * g.K * g.OHW;
  const int64_t x_img = (int64_t)img * g.C * g.OHW;
  const int a_row_l = (lane >> 3);
  const int a_chunk = (lane & 7) ^ (((lane >> 5) & 1) << 1);

  auto stage = [&](int buf, int p) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int inst = wid * 4 + i;
      int row = k0out + inst * 8 + a_row_l;
      if (row >= g.K) row = g.K - 1;
      const bf16* srcA = go + go_img + (int64_t)row * g.OHW + p + a_chunk * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcA,
          (__attribute__((address_space(3))) unsigned int*)(
              (__attribute__((address_space(3))) void*)(ldsA(buf) +
                                                        inst * 512)),
          16, 0, 0);
      int rowB = c0out + inst * 8 + a_row_l;
      if (rowB >= g.C) rowB = g.C - 1;
      const bf16* srcB = x + x_img + (int64_t)rowB * g.OHW + p + a_chunk * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcB,
          (__attribute__((address_space(3))) unsigned int*)(
              (__attribute__((address_space(3))) void*)(ldsB(buf) +
                                                        inst * 512)),
          16, 0, 0);
    }
  };

  const int ksteps = g.slab / FBK;
  stage(0, p0);
  if (BUFS == 3 && 1 < ksteps) stage(1, p0 + FBK);
  const int a_row0 = wm * 64 + (lane & 15);
  const int b_row0 = wn * 64 + (lane & 15);
  const int ch_rd = (lane >> 4);

  for (int step = 0; step < ksteps; ++step) {
    const int buf = (BUFS == 3) ? (step % 3) : (step & 1);
    if (BUFS == 3) {
      if (step + 1 < ksteps)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      if (step + 2 < ksteps) stage((step + 2) % 3, p0 + (step + 2) * FBK);
    } else {
      __syncthreads();
      if (step + 1 < ksteps) stage(buf ^ 1, p0 + (step + 1) * FBK);
    }
#pragma unroll
    for (int ksub = 0; ksub < 2; ++ksub) {
      s16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int ra = a_row0 + f * 16;
        int cha = (ksub * 4 + ch_rd) ^ (((ra >> 2) & 1) << 1);
        afrag[f] = *(const s16x8*)(ldsA(buf) + ((ra >> 3) * 512) +
                                   (ra & 7) * 64 + cha * 8);
        const int rb = b_row0 + f * 16;
        int chb = (ksub * 4 + ch_rd) ^ (((rb >> 2) & 1) << 1);
        bfrag[f] = *(const s16x8*)(ldsB(buf) + ((rb >> 3) * 512) +
                                   (rb & 7) * 64 + chb * 8);
      }
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mf], bfrag[nf], acc[mf][nf], 0, 0, 0);
    }
  }

  // epilogue: fp32 atomic add (slabs/images race on the same tile).
  // MFMA D layout: row = m-frag row block + (lane>>4)*4 + reg, col = lane&15
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int kout = k0out + wm * 64 + mf * 16 + ((lane >> 4) << 2) + reg;
      if (kout >= g.K) continue;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int c = c0out + wn * 64 + nf * 16 + (lane & 15);
        if (c < g.C) atomicAdd(&gw[(int64_t)kout * g.C + c], acc[mf][nf][reg]);
      }
    }
  }
}


// 256x256-tile bwdw for big (K, C): 512 threads, 8 waves (2M x 4N),
// wave tile 128x64; same dual linear-glds staging. 2x MFMA per staged
// byte vs the 128 tile and 1/4 the (mt, nt) re-reads.
__global__ __launch_bounds__(512) void pw_bwdw256_kernel(
    const bf16* __restrict__ go, const bf16* __restrict__ x,
    float* __restrict__ gw, BwGeom g) {
  constexpr int BM = 256, BNW = 256, FBK = 64;
  const int m_tiles = (g.K + BM - 1) / BM;
  const int n_tiles = (g.C + BNW - 1) / BNW;
  int bid = blockIdx.x;
  const int mt = bid % m_tiles;
  bid /= m_tiles;
  const int nt = bid % n_tiles;
  bid /= n_tiles;
  const int slabs_per_img = g.OHW / g.slab;
  const int img = bid / slabs_per_img;
  const int p0 = (bid - img * slabs_per_img) * g.slab;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2, wn = wid & 3;
  const int k0out = mt * BM;
  const int c0out = nt * BNW;

  __shared__ __attribute__((aligned(16))) short lds[4 * (BM * FBK)];
  auto ldsA = [&](int buf) { return lds + buf * (BM * FBK); };
  auto ldsB = [&](int buf) { return lds + (2 + buf) * (BM * FBK); };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int64_t go_img = (int64_t)img * g.K * g.OHW;
  const int64_t x_img = (int64_t)img * g.C * g.OHW;
  const int a_row_l = (lane >> 3);
  const int a_chunk = (lane & 7) ^ (((lane >> 5) & 1) << 1);

  auto stage = [&](int buf, int p) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int inst = wid * 4 + i;  // 0..31
      int row = k0out + inst * 8 + a_row_l;
      if (row >= g.K) row = g.K - 1;
      const bf16* srcA = go + go_img + (int64_t)row * g.OHW + p + a_chunk * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcA,
          (__attribute__((address_space(3))) unsigned int*)(
              (__attribute__((address_space(3))) void*)(ldsA(buf) +
                                                        inst * 512)),
          16, 0, 0);
      int rowB = c0out + inst * 8 + a_row_l;
      if (rowB >= g.C) rowB = g.C - 1;
      const bf16* srcB = x + x_img + (int64_t)rowB * g.OHW + p + a_chunk * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcB,
          (__attribute__((address_space(3))) unsigned int*)(
              (__attribute__((address_space(3))) void*)(ldsB(buf) +
                                                        inst * 512)),
          16, 0, 0);
    }
  };

  const int ksteps = g.slab / FBK;
  stage(0, p0);
  const int a_row0 = wm * 128 + (lane & 15);
  const int b_row0 = wn * 64 + (lane & 15);
  const int ch_rd = (lane >> 4);

  for (int step = 0; step < ksteps; ++step) {
    const int buf = step & 1;
    __syncthreads();
    if (step + 1 < ksteps) stage(buf ^ 1, p0 + (step + 1) * FBK);
#pragma unroll
    for (int ksub = 0; ksub < 2; ++ksub) {
      s16x8 afrag[8], bfrag[4];
#pragma unroll
      for (int f = 0; f < 8; ++f) {
        const int ra = a_row0 + f * 16;
        int cha = (ksub * 4 + ch_rd) ^ (((ra >> 2) & 1) << 1);
        afrag[f] = *(const s16x8*)(ldsA(buf) + ((ra >> 3) * 512) +
                                   (ra & 7) * 64 + cha * 8);
      }
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int rb = b_row0 + f * 16;
        int chb = (ksub * 4 + ch_rd) ^ (((rb >> 2) & 1) << 1);
        bfrag[f] = *(const s16x8*)(ldsB(buf) + ((rb >> 3) * 512) +
                                   (rb & 7) * 64 + chb * 8);
      }
#pragma unroll
      for (int mf = 0; mf < 8; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mf], bfrag[nf], acc[mf][nf], 0, 0, 0);
    }
  }

#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int kout = k0out + wm * 128 + mf * 16 + ((lane >> 4) << 2) + reg;
      if (kout >= g.K) continue;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int c = c0out + wn * 64 + nf * 16 + (lane & 15);
        if (c < g.C) atomicAdd(&gw[(int64_t)kout * g.C + c], acc[mf][nf][reg]);
      }
    }
  }
}


// Pipelined bwdw256: BK=32 so THREE (A,B) buffer pairs fit in 96 KB —
// at this kernel's 1-block/CU occupancy the in-kernel span is the only
// latency hiding available (the 2-buffer __syncthreads form parks
// waves on the glds drain; PMC probe in profiles/PERF_NOTES.md).
// Protocol per step: wait MY 4 glds for this step (counted vmcnt(4) —
// the 4 newer ones stay in flight) -> raw barrier (now EVERY wave's
// shares landed) -> issue step+2 into the buffer all waves finished
// reading two steps ago -> 32 MFMA. [256][32]-short images need no
// swizzle: 64-byte rows decorrelate the b128 lane-group banks.
__global__ __launch_bounds__(512) void pw_bwdw256p_kernel(
    const bf16* __restrict__ go, const bf16* __restrict__ x,
    float* __restrict__ gw, BwGeom g) {
  constexpr int BM = 256, BNW = 256, FBK = 32;
  const int m_tiles = (g.K + BM - 1) / BM;
  const int n_tiles = (g.C + BNW - 1) / BNW;
  int bid = blockIdx.x;
  const int mt = bid % m_tiles;
  bid /= m_tiles;
  const int nt = bid % n_tiles;
  bid /= n_tiles;
  const int slabs_per_img = g.OHW / g.slab;
  const int img = bid / slabs_per_img;
  const int p0 = (bid - img * slabs_per_img) * g.slab;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2, wn = wid & 3;
  const int k0out = mt * BM;
  const int c0out = nt * BNW;

  __shared__ __attribute__((aligned(16))) short lds[6 * (BM * FBK)];
  auto ldsA = [&](int buf) { return lds + buf * (BM * FBK); };
  auto ldsB = [&](int buf) { return lds + (3 + buf) * (BM * FBK); };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int64_t go_img = (int64_t)img * g.K * g.OHW;
  const int64_t x_img = (int64_t)img * g.C * g.OHW;
  // glds lane map for [256][32]-short images: 1 KB instr = 16 rows;
  // lane l -> row (l>>2), 8-short chunk (l&3)
  const int g_row = lane >> 2;
  const int g_ch = lane & 3;

  auto stage = [&](int buf, int p) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int inst = wid * 2 + i;  // 0..15: rows inst*16..+16
      int row = k0out + inst * 16 + g_row;
      if (row >= g.K) row = g.K - 1;
      const bf16* srcA = go + go_img + (int64_t)row * g.OHW + p + g_ch * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcA,
          (__attribute__((address_space(3))) unsigned int*)(
              (__attribute__((address_space(3))) void*)(ldsA(buf) +
                                                        inst * 512)),
          16, 0, 0);
      int rowB = c0out + inst * 16 + g_row;
      if (rowB >= g.C) rowB = g.C - 1;
      const bf16* srcB = x + x_img + (int64_t)rowB * g.OHW + p + g_ch * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcB,
          (__attribute__((address_space(3))) unsigned int*)(
              (__attribute__((address_space(3))) void*)(ldsB(buf) +
                                                        inst * 512)),
          16, 0, 0);
    }
  };

  const int ksteps = g.slab / FBK;
  stage(0, p0);
  if (1 < ksteps) stage(1, p0 + FBK);
  const int a_row0 = wm * 128 + (lane & 15);
  const int b_row0 = wn * 64 + (lane & 15);
  const int ch_rd = (lane >> 4);  // 8-short k-chunk within the 32-k step

  for (int step = 0; step < ksteps; ++step) {
    const int buf = step % 3;
    if (step + 1 < ksteps)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (step + 2 < ksteps) stage((step + 2) % 3, p0 + (step + 2) * FBK);
    s16x8 afrag[8], bfrag[4];
#pragma unroll
    for (int f = 0; f < 8; ++f) {
      const int ra = a_row0 + f * 16;
      afrag[f] = *(const s16x8*)(ldsA(buf) + ra * 32 + ch_rd * 8);
    }
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      const int rb = b_row0 + f * 16;
      bfrag[f] = *(const s16x8*)(ldsB(buf) + rb * 32 + ch_rd * 8);
    }
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[mf], bfrag[nf], acc[mf][nf], 0, 0, 0);
  }
  // the last step's readers finish before the epilogue (no barrier
  // needed: each wave only reads LDS it waited for, and the epilogue
  // touches no LDS)

#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int kout = k0out + wm * 128 + mf * 16 + ((lane >> 4) << 2) + reg;
      if (kout >= g.K) continue;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int c = c0out + wn * 64 + nf * 16 + (lane & 15);
        if (c < g.C) atomicAdd(&gw[(int64_t)kout * g.C + c], acc[mf][nf][reg]);
      }
    }
  }
}

torch::Tensor pw_bwdw(torch::Tensor go, torch::Tensor x) {
  TORCH_CHECK(go.is_cuda() && go.is_contiguous() && x.is_contiguous());
  TORCH_CHECK(go.scalar_type() == torch::kBFloat16 &&
              x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(go.size(0) == x.size(0) && go.size(2) == x.size(2) &&
              go.size(3) == x.size(3), "pw_bwdw: stride-1 1x1 only");
  BwGeom g;
  g.Nimg = (int)go.size(0);
  g.K = (int)go.size(1);
  g.C = (int)x.size(1);
  g.OHW = (int)(go.size(2) * go.size(3));
  TORCH_CHECK(g.OHW % 64 == 0, "pw_bwdw: OHW % 64");
  // 256 tile only when the pixel dim is deep enough that k-steps stay
  // long after slab splitting (small-P shapes measured 0.5x with deep
  // splits: 2-kstep blocks are staging-latency bound and the 256x256
  // fp32 atomic tiles dominate — gpurun_out/bwdw_ab.log)
  const bool big = (g.K >= 256 && g.C >= 256 && g.OHW >= 16384);
  const int tile = big ? 256 : 128;
  const int mn = ((g.K + tile - 1) / tile) * ((g.C + tile - 1) / tile);
  int slab = g.OHW;
  while (slab > 512 && (int64_t)mn * g.Nimg * (g.OHW / slab) < 384 &&
         slab % 2 == 0 && (slab / 2) % 64 == 0)
    slab /= 2;
  auto gw = torch::zeros({(int64_t)g.K, (int64_t)g.C},
                         x.options().dtype(torch::kFloat));
  g.slab = slab;
  const int64_t blocks =
      (int64_t)mn * g.Nimg * (g.OHW / slab);
  TORCH_CHECK(blocks > 0 && blocks < (1LL << 31), "pw_bwdw grid");
  auto stream = at::cuda::getCurrentCUDAStream();
  if (big) {
    // measured (gpurun_out/bwdwp.log): the BK=32 3-buffer span is
    // correct but 0.96x of the BK=64 2-buffer (halving the MFMA work
    // per barrier cancels the span win). Kept for round-3 reference.
    static const bool pipe = []() {
      const char* e = getenv("MPI4DL_BWDW_PIPE");
      return e && e[0] == '3';
    }();
    if (pipe)
      hipLaunchKernelGGL(pw_bwdw256p_kernel, dim3((uint32_t)blocks),
                         dim3(512), 0, stream.stream(),
                         (const bf16*)go.data_ptr(),
                         (const bf16*)x.data_ptr(), gw.data_ptr<float>(), g);
    else
      hipLaunchKernelGGL(pw_bwdw256_kernel, dim3((uint32_t)blocks), dim3(512),
                         0, stream.stream(), (const bf16*)go.data_ptr(),
                         (const bf16*)x.data_ptr(), gw.data_ptr<float>(), g);
    return gw;
  }
  static const int bufs = []() {
    const char* e = getenv("MPI4DL_PW_PIPE");
    return (e && e[0] == '3') ? 3 : 2;
  }();
  if (bufs == 3)
    hipLaunchKernelGGL(pw_bwdw_kernel<3>, dim3((uint32_t)blocks), dim3(256),
                       0, stream.stream(), (const bf16*)go.data_ptr(),
                       (const bf16*)x.data_ptr(), gw.data_ptr<float>(), g);
  else
    hipLaunchKernelGGL(pw_bwdw_kernel<2>, dim3((uint32_t)blocks), dim3(256),
                       0, stream.stream(), (const bf16*)go.data_ptr(),
                       (const bf16*)x.data_ptr(), gw.data_ptr<float>(), g);
  return gw;
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

static void launch_pw(const torch::Tensor& x, const torch::Tensor& w,
                      const float* bias, torch::Tensor& out, PwGeom g) {
  const int K = g.K;
  auto stream = at::cuda::getCurrentCUDAStream();
  // biggest fat shapes: 256x256 tile (2x MFMA per staged byte)
  if (g.sw == 1 && g.osw == 1 && g.C >= 512 && K >= 256 &&
      (g.OH * g.OW) % 256 == 0 &&
      [] { const char* e = getenv("MPI4DL_PW_FAT256");
           return !(e && e[0] == '0'); }()) {
    const int m_tiles = (K + 255) / 256;
    const int64_t blocks =
        (int64_t)g.N * (g.OH * g.OW / 256) * m_tiles;
    TORCH_CHECK(blocks > 0 && blocks < (1LL << 31), "pw grid size");
    hipLaunchKernelGGL(pw_fat256_kernel, dim3((uint32_t)blocks), dim3(512),
                       0, stream.stream(), (const bf16*)x.data_ptr(),
                       (const bf16*)w.data_ptr(), bias,
                       (bf16*)out.data_ptr(), g);
    return;
  }
  // fat shapes (compute-bound): 128x128xBK64 dual-glds kernel
  if (g.sw == 1 && g.osw == 1 && g.C >= 256 && K >= 128) {
    const int m_tiles = (K + 127) / 128;
    const int64_t blocks =
        (int64_t)g.N * (g.OH * g.OW / PW_BN) * m_tiles;
    TORCH_CHECK(blocks > 0 && blocks < (1LL << 31), "pw grid size");
    // measured (gpurun_out/pipe_ab.log): 2-buffer + 2 blocks/CU beats
    // the 3-buffer span pipeline ~2x here (the span lever only pays in
    // the 1-block/CU VGPR~250 regime — guide §5 glds table)
    static const int bufs = []() {
      const char* e = getenv("MPI4DL_PW_PIPE");
      return (e && e[0] == '3') ? 3 : 2;
    }();
    if (bufs == 3)
      hipLaunchKernelGGL(pw_fat_kernel<3>, dim3((uint32_t)blocks), dim3(256),
                         0, stream.stream(), (const bf16*)x.data_ptr(),
                         (const bf16*)w.data_ptr(), bias,
                         (bf16*)out.data_ptr(), g);
    else
      hipLaunchKernelGGL(pw_fat_kernel<2>, dim3((uint32_t)blocks), dim3(256),
                         0, stream.stream(), (const bf16*)x.data_ptr(),
                         (const bf16*)w.data_ptr(), bias,
                         (bf16*)out.data_ptr(), g);
    return;
  }
  // measured: MFRAG=8 (BM=256, acc 128 f32/lane) LOSES ~1.4x on the
  // skinny shapes — register pressure halves occupancy and these
  // few-kstep kernels live on cross-block latency hiding. Keep BM<=128.
  const int mfrag = (K <= 32) ? 1 : (K <= 64) ? 2 : 4;
  const int BM = 32 * mfrag;
  const int m_tiles = (K + BM - 1) / BM;
  const int64_t ptiles = (int64_t)g.N * (g.OH * g.OW / PW_BN);
  const int64_t blocks = ptiles * m_tiles;
  TORCH_CHECK(blocks > 0 && blocks < (1LL << 31), "pw grid size");
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3((uint32_t)blocks), dim3(256), 0,
                       stream.stream(), (const bf16*)x.data_ptr(),
                       (const bf16*)w.data_ptr(), bias,
                       (bf16*)out.data_ptr(), g);
  };
  if (g.sw == 1) {
    if (mfrag == 1) launch(pw_kernel<1, 1>);
    else if (mfrag == 2) launch(pw_kernel<2, 1>);
    else launch(pw_kernel<4, 1>);
  } else {
    if (mfrag == 1) launch(pw_kernel<1, 2>);
    else if (mfrag == 2) launch(pw_kernel<2, 2>);
    else launch(pw_kernel<4, 2>);
  }
}

// forward: x [N,C,H,W] bf16, w [K,C,1,1] bf16, stride (sh,sw), pad 0
torch::Tensor pw_fwd(torch::Tensor x, torch::Tensor w,
                     c10::optional<torch::Tensor> bias, int64_t sh,
                     int64_t sw) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16, "pw_fwd: bf16 only");
  PwGeom g;
  g.N = (int)x.size(0);
  g.C = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  g.K = (int)w.size(0);
  TORCH_CHECK(w.numel() == (int64_t)g.K * g.C, "pw_fwd: w must be [K,C,1,1]");
  g.HW_in = H * W;
  g.W_in = W;
  g.sh = (int)sh;
  g.sw = (int)sw;
  g.OH = (H - 1) / g.sh + 1;
  g.OW = (W - 1) / g.sw + 1;
  g.osh = 1; g.osw = 1;
  g.oW = g.OW;
  g.oHW = g.OH * g.OW;
  TORCH_CHECK((g.OH * g.OW) % PW_BN == 0, "pw_fwd: OH*OW % 128 != 0");
  if (g.sw != 1) TORCH_CHECK(g.OW % 8 == 0, "pw_fwd: OW % 8 != 0 at stride 2");
  auto out = torch::empty({(int64_t)g.N, (int64_t)g.K, (int64_t)g.OH,
                           (int64_t)g.OW},
                          x.options());
  const float* bptr = nullptr;
  torch::Tensor b32;
  if (bias.has_value()) {
    b32 = bias->to(torch::kFloat).contiguous();
    bptr = b32.data_ptr<float>();
  }
  launch_pw(x, w, bptr, out, g);
  return out;
}

// stride-2 backward-data: gx[:, :, ::sh, ::sw] = pw(go, w^T), rest 0.
// go [N,K,OH,OW]; wt [C,K] (w transposed, contiguous); returns [N,C,H,W].
torch::Tensor pw_bwd_data_strided(torch::Tensor go, torch::Tensor wt,
                                  int64_t H, int64_t W, int64_t sh,
                                  int64_t sw) {
  TORCH_CHECK(go.is_cuda() && go.is_contiguous() && wt.is_contiguous());
  TORCH_CHECK(go.scalar_type() == torch::kBFloat16 &&
              wt.scalar_type() == torch::kBFloat16);
  PwGeom g;
  g.N = (int)go.size(0);
  g.C = (int)go.size(1);            // reduction dim = K of fwd
  g.K = (int)wt.size(0);            // output channels = C of fwd
  g.OH = (int)go.size(2);
  g.OW = (int)go.size(3);
  g.HW_in = g.OH * g.OW;
  g.W_in = g.OW;
  g.sh = 1; g.sw = 1;               // reading go densely
  g.osh = (int)sh; g.osw = (int)sw; // scattering into gx
  g.oW = (int)W;
  g.oHW = (int)(H * W);
  TORCH_CHECK((g.OH * g.OW) % PW_BN == 0,
              "pw_bwd_data: OH*OW % 128 != 0");
  auto gx = torch::zeros({(int64_t)g.N, (int64_t)g.K, H, W}, go.options());
  launch_pw(go, wt, nullptr, gx, g);
  return gx;
}

}  // namespace conv_pw

void register_conv_pw(pybind11::module_& m) {
  m.def("pw_fwd", &conv_pw::pw_fwd,
        "1x1 conv forward (streaming MFMA GEMM, bf16 NCHW)");
  m.def("pw_bwd_data_strided", &conv_pw::pw_bwd_data_strided,
        "1x1 strided conv backward-data (scatter epilogue)");
  m.def("pw_bwdw", &conv_pw::pw_bwdw,
        "1x1 stride-1 conv backward-weight (NT MFMA GEMM, fp32 out)");
}
