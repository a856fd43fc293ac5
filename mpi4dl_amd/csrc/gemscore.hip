// gemscore — MI355X-native (gfx950/CDNA4) kernels for mpi4dl_amd.
//
// This translation unit holds the hand-written HIP kernels plus their
// torch bindings. Everything is written for CDNA4 directly (wave64,
// no CUDA-compat shims): see /opt/skills/guides/cdna_hip_programming.md
// for the hardware model the tiling decisions cite.
//
// Kernels:
//   * halo pack / unpack / unpack-add — one launch moves ALL strips of
//     a halo exchange between the padded tile and a flat RCCL staging
//     buffer (replaces the reference's 8 slice+clone launches,
//     /root/reference/src/torchgems/spatial.py:336-413).
//   * bn_stats — per-channel sum + centred sum-of-squares in one pass
//     (feeds TileBatchNorm2d's cross-tile reduction).
//
// Conv/pool implicit-GEMM MFMA kernels live in conv_mfma.hip.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cstdint>
#include <vector>

typedef __attribute__((ext_vector_type(8))) short s16x8_;

#define MAX_STRIPS 8
#define DEVCHECK(x) TORCH_CHECK(x, #x " failed")

namespace {

struct StripDesc {
  int64_t plane_off;  // rs * Wp + cs into a (Hp, Wp) plane
  int32_t rows;
  int32_t cols;
  int64_t buf_off;    // element offset into the flat staging buffer
};

struct StripArgs {
  StripDesc s[MAX_STRIPS];
  int32_t nstrips;
  int32_t nc;           // N*C planes
  int64_t plane_stride; // Hp*Wp
  int64_t wp;           // padded width
};

// mode: 0 = pack (tile->buf), 1 = unpack (buf->tile), 2 = unpack-add
template <typename T, int MODE>
__global__ void halo_copy_kernel(T* __restrict__ tile, T* __restrict__ buf,
                                 StripArgs a, int64_t total) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int64_t rem = i;
    int si = 0;
    for (; si < a.nstrips; ++si) {  // <=8 strips: register scan
      const int64_t sz = (int64_t)a.s[si].rows * a.s[si].cols * a.nc;
      if (rem < sz) break;
      rem -= sz;
    }
    const StripDesc d = a.s[si];
    const int64_t cols = d.cols;
    const int64_t rc = (int64_t)d.rows * cols;
    const int64_t plane = rem / rc;
    const int64_t r = (rem % rc) / cols;
    const int64_t c = rem % cols;
    const int64_t tidx = plane * a.plane_stride + d.plane_off + r * a.wp + c;
    const int64_t bidx = d.buf_off + plane * rc + r * cols + c;
    if (MODE == 0)
      buf[bidx] = tile[tidx];
    else if (MODE == 1)
      tile[tidx] = buf[bidx];
    else
      tile[tidx] += buf[bidx];
  }
}

StripArgs build_args(const torch::Tensor& tile, const torch::Tensor& desc,
                     int64_t* total_out) {
  TORCH_CHECK(tile.dim() == 4, "tile must be (N,C,Hp,Wp)");
  TORCH_CHECK(desc.device().is_cpu() && desc.dtype() == torch::kInt64,
              "desc must be a CPU int64 tensor [nstrips, 4]");
  auto d = desc.accessor<int64_t, 2>();
  StripArgs a;
  a.nstrips = (int32_t)desc.size(0);
  TORCH_CHECK(a.nstrips <= MAX_STRIPS, "too many strips");
  a.nc = (int32_t)(tile.size(0) * tile.size(1));
  a.plane_stride = tile.size(2) * tile.size(3);
  a.wp = tile.size(3);
  int64_t total = 0;
  for (int i = 0; i < a.nstrips; ++i) {
    const int64_t rs = d[i][0], cs = d[i][1];
    a.s[i].rows = (int32_t)d[i][2];
    a.s[i].cols = (int32_t)d[i][3];
    a.s[i].plane_off = rs * a.wp + cs;
    a.s[i].buf_off = total;
    total += (int64_t)a.s[i].rows * a.s[i].cols * a.nc;
  }
  *total_out = total;
  return a;
}

template <int MODE>
void halo_copy(torch::Tensor tile, torch::Tensor buf, torch::Tensor desc) {
  TORCH_CHECK(tile.is_cuda() && buf.is_cuda(), "tensors must be on GPU");
  TORCH_CHECK(tile.is_contiguous() && buf.is_contiguous());
  TORCH_CHECK(tile.scalar_type() == buf.scalar_type());
  int64_t total = 0;
  StripArgs a = build_args(tile, desc, &total);
  if (total == 0) return;
  TORCH_CHECK(buf.numel() >= total, "staging buffer too small");
  const int block = 256;
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, tile.scalar_type(),
      "halo_copy", [&] {
        if (MODE == 2) {
          // grad strips may OVERLAP at corners (each neighbour received a
          // replicated copy of the corner in forward, so each contributes
          // a gradient): launch per strip, stream-ordered, so the adds
          // into the overlap are sequential instead of racing.
          for (int i = 0; i < a.nstrips; ++i) {
            StripArgs one;
            one.nstrips = 1;
            one.nc = a.nc;
            one.plane_stride = a.plane_stride;
            one.wp = a.wp;
            one.s[0] = a.s[i];
            const int64_t t1 = (int64_t)a.s[i].rows * a.s[i].cols * a.nc;
            one.s[0].buf_off = a.s[i].buf_off;
            const int grid1 =
                (int)std::min<int64_t>((t1 + block - 1) / block, 2048);
            hipLaunchKernelGGL((halo_copy_kernel<scalar_t, MODE>), dim3(grid1),
                               dim3(block), 0, stream.stream(),
                               tile.data_ptr<scalar_t>(),
                               buf.data_ptr<scalar_t>() + a.s[i].buf_off -
                                   one.s[0].buf_off,
                               one, t1);
          }
          return;
        }
        const int grid =
            (int)std::min<int64_t>((total + block - 1) / block, 2048);
        hipLaunchKernelGGL((halo_copy_kernel<scalar_t, MODE>), dim3(grid),
                           dim3(block), 0, stream.stream(),
                           tile.data_ptr<scalar_t>(), buf.data_ptr<scalar_t>(),
                           a, total);
      });
}

// ---------------------------------------------------------------------------
// bn_stats: per-channel sum and sum-of-squares in ONE pass over (N,C,H,W).
// One workgroup per channel-chunk; wave-level reduction then LDS combine.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x, float* __restrict__ out,
                                int64_t N, int64_t C, int64_t HW) {
  // block.x = 256 threads; one block per channel; grid-stride over channels
  for (int64_t ch = blockIdx.x; ch < C; ch += gridDim.x) {
    float s = 0.f, ss = 0.f;
    for (int64_t n = 0; n < N; ++n) {
      const T* p = x + (n * C + ch) * HW;
      for (int64_t i = threadIdx.x; i < HW; i += blockDim.x) {
        const float v = (float)p[i];
        s += v;
        ss += v * v;
      }
    }
    // wave64 reduction then cross-wave via LDS
    for (int off = 32; off > 0; off >>= 1) {
      s += __shfl_down(s, off, 64);
      ss += __shfl_down(ss, off, 64);
    }
    __shared__ float ls[8], lss[8];
    const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
    if (lane == 0) { ls[wid] = s; lss[wid] = ss; }
    __syncthreads();
    if (threadIdx.x == 0) {
      float ts = 0.f, tss = 0.f;
      for (int w = 0; w < (int)(blockDim.x >> 6); ++w) { ts += ls[w]; tss += lss[w]; }
      out[ch] = ts;
      out[C + ch] = tss;
    }
    __syncthreads();
  }
}

torch::Tensor bn_stats(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous());
  const int64_t N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto out = torch::empty({2 * C}, x.options().dtype(torch::kFloat));
  const int grid = (int)std::min<int64_t>(C, 2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, x.scalar_type(),
      "bn_stats", [&] {
        hipLaunchKernelGGL((bn_stats_kernel<scalar_t>), dim3(grid), dim3(256),
                           0, stream.stream(), x.data_ptr<scalar_t>(),
                           out.data_ptr<float>(), N, C, HW);
      });
  return out;
}


// ---------------------------------------------------------------------------
// Pooling (NCHW), padding-aware, global-geometry divisors.
//
// Replaces torch's eager max_pool NCHW kernels (15.5% of step time in
// profiles/r01_bench_single_gpu_kernel_stats.txt). Backward is
// gather-formulated (each input position scans the <=ceil(k/s)^2 windows
// that contain it) — no atomics.
//
// Geometry: output (oh,ow) covers input rows [oh*s-p, oh*s-p+k). For
// tile-parallel exact avg (count_include_pad=False) the divisor counts
// window cells inside the GLOBAL image: global row = gr0 + oh*s + kh
// with gr0 = tile_row_offset - p; plain case: gr0 = -p, Hg = H.
// ---------------------------------------------------------------------------

// 8 consecutive outputs per thread: the input row segment is read once
// into a register sliding window (guide G13 — vectorize ALWAYS).
template <typename T, int KK, int SS>
__global__ void maxpool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   uint8_t* __restrict__ idx, int64_t NC,
                                   int H, int W, int OH, int OW, int k_, int s_,
                                   int p) {
  const int k = KK > 0 ? KK : k_;
  const int s = KK > 0 ? SS : s_;
  const int OW8 = (OW + 7) / 8;
  const int64_t total = NC * OH * OW8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += stride) {
    const int owc = (int)(t % OW8);
    const int oh = (int)((t / OW8) % OH);
    const int64_t plane = t / ((int64_t)OW8 * OH);
    const T* xp = x + plane * H * W;
    const int ow0 = owc * 8;
    const int h0 = oh * s - p;
    float best[8];
    uint8_t bidx[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) { best[e] = -INFINITY; bidx[e] = 0; }
    const int w_lo = ow0 * s - p;
    for (int kh = 0; kh < (KK > 0 ? KK : k); ++kh) {
      const int h = h0 + kh;
      if (h < 0 || h >= H) continue;
      const T* row = xp + h * W;
      if (KK > 0) {
        // compile-time (k,s): span stays in registers, loops fully unroll
        constexpr int SPAN = KK > 0 ? 8 * SS + KK - SS : 1;
        float seg[SPAN];
        // NOTE: a vectorized span-load variant (bf16 reinterpret, then a
        // type-generic memcpy form) was measured and REVERTED: the
        // reinterpret broke fp32 instantiations, and the memcpy form
        // regressed the bench ~5% (the local span array drops to
        // scratch). Scalar loads + L1 hitting the overlapped spans stay.
#pragma unroll
        for (int j = 0; j < SPAN; ++j) {
          const int w = w_lo + j;
          seg[j] = (w >= 0 && w < W) ? (float)row[w] : -INFINITY;
        }
#pragma unroll
        for (int e = 0; e < 8; ++e) {
#pragma unroll
          for (int kw = 0; kw < KK; ++kw) {
            const float v = seg[e * SS + kw];
            if (v > best[e]) {
              best[e] = v;
              bidx[e] = (uint8_t)(kh * KK + kw);
            }
          }
        }
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int wbase = w_lo + e * s;
          for (int kw = 0; kw < k; ++kw) {
            const int w = wbase + kw;
            if (w < 0 || w >= W) continue;
            const float v = (float)row[w];
            if (v > best[e]) { best[e] = v; bidx[e] = (uint8_t)(kh * k + kw); }
          }
        }
      }
    }
    const int64_t obase = plane * (int64_t)OH * OW + (int64_t)oh * OW;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int ow = ow0 + e;
      if (ow < OW) { y[obase + ow] = (T)best[e]; idx[obase + ow] = bidx[e]; }
    }
  }
}

// 8 consecutive input columns per thread: the candidate output windows
// of the 8 inputs overlap, so go/idx rows are read once per (oh, thread).
// (KK, SS) compile-time specialisations fully unroll the window loops
// and fold the offset div/mod (same trick as the forward).
template <typename T, int KK, int SS>
__global__ void maxpool_bwd_kernel(const T* __restrict__ go,
                                   const uint8_t* __restrict__ idx,
                                   T* __restrict__ gi, int64_t NC, int H,
                                   int W, int OH, int OW, int k_, int s_,
                                   int p) {
  const int k = KK > 0 ? KK : k_;
  const int s = KK > 0 ? SS : s_;
  const int W8 = (W + 7) / 8;
  const int64_t total = NC * H * W8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += stride) {
    const int wc = (int)(t % W8);
    const int h = (int)((t / W8) % H);
    const int64_t plane = t / ((int64_t)W8 * H);
    const int w0 = wc * 8;
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    int oh_lo = (h + p - k + 1 + s - 1) / s; if (oh_lo < 0) oh_lo = 0;
    int oh_hi = (h + p) / s; if (oh_hi > OH - 1) oh_hi = OH - 1;
    int ow_lo = (w0 + p - k + 1 + s - 1) / s; if (ow_lo < 0) ow_lo = 0;
    int ow_hi = (w0 + 7 + p) / s; if (ow_hi > OW - 1) ow_hi = OW - 1;
    const T* gop = go + plane * OH * OW;
    const uint8_t* ip = idx + plane * OH * OW;
    // at most ceil((8 + k - 1)/s) + 1 candidate columns
    constexpr int NOW = KK > 0 ? (8 + KK - 1) / SS + 1 : 0;
    for (int oh = oh_lo; oh <= oh_hi; ++oh) {
      if (KK > 0) {
#pragma unroll
        for (int j = 0; j < NOW; ++j) {
          const int ow = ow_lo + j;
          if (ow > ow_hi) break;
          const int o = oh * OW + ow;
          const int off = ip[o];
          const int r = oh * SS - p + off / KK;  // winner's input row/col
          if (r != h) continue;
          const int dw = ow * SS - p + off % KK - w0;
          const float gval = (float)gop[o];
          // acc[dw] with a RUNTIME index would force acc[] to scratch;
          // the unrolled compare keeps it in registers
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (dw == e) acc[e] += gval;
        }
      } else {
        for (int ow = ow_lo; ow <= ow_hi; ++ow) {
          const int o = oh * OW + ow;
          const int off = ip[o];
          const int r = oh * s - p + off / k;
          if (r != h) continue;
          const int dw = ow * s - p + off % k - w0;
          const float gval = (float)gop[o];
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (dw == e) acc[e] += gval;
        }
      }
    }
    const int64_t ibase = plane * (int64_t)H * W + (int64_t)h * W;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      if (w0 + e < W) gi[ibase + w0 + e] = (T)acc[e];
  }
}

__device__ __forceinline__ float avg_div(int oh, int ow, int s, int k,
                                         long gr0, long gc0, long Hg, long Wg,
                                         int include_pad) {
  if (include_pad) return (float)(k * k);
  long r0 = gr0 + (long)oh * s, c0 = gc0 + (long)ow * s;
  long r1 = r0 + k, c1 = c0 + k;
  if (r0 < 0) r0 = 0; if (c0 < 0) c0 = 0;
  if (r1 > Hg) r1 = Hg; if (c1 > Wg) c1 = Wg;
  long cnt = (r1 - r0) * (c1 - c0);
  return cnt > 0 ? (float)cnt : 1.f;
}

// 8 outputs per thread; compile-time (k,s) keeps the row segment in
// registers (KK=0 -> generic scalar path).
template <typename T, int KK, int SS>
__global__ void avgpool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   int64_t NC, int H, int W, int OH, int OW,
                                   int k_, int s_, int p, long gr0, long gc0,
                                   long Hg, long Wg, int include_pad) {
  const int k = KK > 0 ? KK : k_;
  const int s = KK > 0 ? SS : s_;
  const int OW8 = (OW + 7) / 8;
  const int64_t total = NC * OH * OW8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += stride) {
    const int owc = (int)(t % OW8);
    const int oh = (int)((t / OW8) % OH);
    const int64_t plane = t / ((int64_t)OW8 * OH);
    const T* xp = x + plane * H * W;
    const int ow0 = owc * 8;
    const int h0 = oh * s - p;
    const int w_lo = ow0 * s - p;
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    for (int kh = 0; kh < (KK > 0 ? KK : k); ++kh) {
      const int h = h0 + kh;
      if (h < 0 || h >= H) continue;
      const T* row = xp + h * W;
      if (KK > 0) {
        constexpr int SPAN = KK > 0 ? 8 * SS + KK - SS : 1;
        float seg[SPAN];
#pragma unroll
        for (int j = 0; j < SPAN; ++j) {
          const int w = w_lo + j;
          seg[j] = (w >= 0 && w < W) ? (float)row[w] : 0.f;
        }
#pragma unroll
        for (int e = 0; e < 8; ++e)
#pragma unroll
          for (int kw = 0; kw < KK; ++kw) acc[e] += seg[e * SS + kw];
      } else {
        for (int e = 0; e < 8; ++e) {
          const int wb = w_lo + e * s;
          for (int kw = 0; kw < k; ++kw) {
            const int w = wb + kw;
            if (w >= 0 && w < W) acc[e] += (float)row[w];
          }
        }
      }
    }
    const int64_t obase = plane * (int64_t)OH * OW + (int64_t)oh * OW;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int ow = ow0 + e;
      if (ow < OW)
        y[obase + ow] =
            (T)(acc[e] / avg_div(oh, ow, s, k, gr0, gc0, Hg, Wg, include_pad));
    }
  }
}

// 8 inputs per thread; interior windows (the overwhelming majority)
// divide by the constant k*k, skipping the per-window divisor math.
template <typename T, int KK, int SS>
__global__ void avgpool_bwd_kernel(const T* __restrict__ go, T* __restrict__ gi,
                                   int64_t NC, int H, int W, int OH, int OW,
                                   int k_, int s_, int p, long gr0, long gc0,
                                   long Hg, long Wg, int include_pad) {
  const int k = KK > 0 ? KK : k_;
  const int s = KK > 0 ? SS : s_;
  const int W8 = (W + 7) / 8;
  const int64_t total = NC * H * W8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += stride) {
    const int wc = (int)(t % W8);
    const int h = (int)((t / W8) % H);
    const int64_t plane = t / ((int64_t)W8 * H);
    const int w0 = wc * 8;
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    int oh_lo = (h + p - k + 1 + s - 1) / s; if (oh_lo < 0) oh_lo = 0;
    int oh_hi = (h + p) / s; if (oh_hi > OH - 1) oh_hi = OH - 1;
    int ow_lo = (w0 + p - k + 1 + s - 1) / s; if (ow_lo < 0) ow_lo = 0;
    int ow_hi = (w0 + 7 + p) / s; if (ow_hi > OW - 1) ow_hi = OW - 1;
    const T* gop = go + plane * OH * OW;
    // interior iff every touched window lies fully inside the global image
    const bool interior =
        include_pad ||
        ((gr0 + (long)oh_lo * s) >= 0 && (gr0 + (long)oh_hi * s + k) <= Hg &&
         (gc0 + (long)ow_lo * s) >= 0 && (gc0 + (long)ow_hi * s + k) <= Wg);
    const float inv_kk = 1.f / (float)(k * k);
    if (KK == 3 && SS == 1) {
      // stride-1 k3: input w is covered by ow in [w-2+p, w+p]; compute
      // the <=10 per-row contributions once, then a 3-tap sliding sum
      // (5x fewer VALU than the predicated 8x scan)
      for (int oh = oh_lo; oh <= oh_hi; ++oh) {
        float gd[12];
#pragma unroll
        for (int j = 0; j < 10; ++j) {
          const int ow = ow_lo + j;
          float v = 0.f;
          if (ow <= ow_hi) {
            const float g = (float)gop[oh * OW + ow];
            v = interior ? g * inv_kk
                         : g / avg_div(oh, ow, 1, 3, gr0, gc0, Hg, Wg,
                                       include_pad);
          }
          gd[j] = v;
        }
        gd[10] = 0.f;
        gd[11] = 0.f;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          // ow range covering input w0+e: [w0+e+p-2, w0+e+p]
          const int j0 = w0 + e + p - 2 - ow_lo;
          float sum = 0.f;
#pragma unroll
          for (int d = 0; d < 3; ++d) {
            const int j = j0 + d;
            if (j >= 0 && j < 10) sum += gd[j];
          }
          acc[e] += sum;
        }
      }
    } else {
      for (int oh = oh_lo; oh <= oh_hi; ++oh) {
        for (int ow = ow_lo; ow <= ow_hi; ++ow) {
          const float g = (float)gop[oh * OW + ow];
          const float gd =
              interior ? g * inv_kk
                       : g / avg_div(oh, ow, s, k, gr0, gc0, Hg, Wg,
                                     include_pad);
          // which of my 8 inputs does window (oh, ow) cover?
          const int wlo = ow * s - p;
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int w = w0 + e;
            if (w >= wlo && w < wlo + k && w < W) acc[e] += gd;
          }
        }
      }
    }
    const int64_t ibase = plane * (int64_t)H * W + (int64_t)h * W;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      if (w0 + e < W) gi[ibase + w0 + e] = (T)acc[e];
  }
}

// ---------------------------------------------------------------------------
// Fused BatchNorm (NCHW): fp64-accumulated channel stats, one
// normalize+affine(+ReLU) pass, gather-formulated backward.
// Replaces MIOpenBatchNorm{Fwd,Bwd}Spatial (14.3% of step time) and the
// surrounding unfused elementwise casts; doubles as the local-stats leg
// of TileBatchNorm2d's cross-tile sync (python does one allreduce of the
// [sum,sumsq] / [gsum,gxsum] vectors between the two kernels).
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(8))) short short8v;

template <typename T>
__device__ __forceinline__ float to_f32(T v) { return (float)v; }

// 2D grid: x = channel, y = HW chunk (the chip needs >>256 workgroups —
// guide G1/G11); 16-byte vector loads (G13); fp64 accumulation with one
// fp64 atomicAdd per block per output.
template <typename T>
__global__ void bn_stats64_kernel(const T* __restrict__ x,
                                  double* __restrict__ out, int64_t N,
                                  int64_t C, int64_t HW, int64_t chunk) {
  const int64_t ch = blockIdx.x;
  const int64_t i0 = (int64_t)blockIdx.y * chunk;
  const int64_t i1 = min(i0 + chunk, HW);
  double s = 0.0, ss = 0.0;
  const int VEC = sizeof(T) == 2 ? 8 : 4;
  for (int64_t n = 0; n < N; ++n) {
    const T* p = x + (n * C + ch) * HW;
    int64_t i = i0 + (int64_t)threadIdx.x * VEC;
    if (sizeof(T) == 2) {
      for (; i + 8 <= i1; i += (int64_t)blockDim.x * 8) {
        short8v v = *(const short8v*)(p + i);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float f = to_f32(((const T*)&v)[e]);
          s += (double)f;
          ss += (double)(f * f);
        }
      }
    } else {
      for (; i + 4 <= i1; i += (int64_t)blockDim.x * 4) {
        const float4 v = *(const float4*)((const float*)p + i);
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          const float f = ((const float*)&v)[e];
          s += (double)f;
          ss += (double)(f * f);
        }
      }
    }
    // tail (only the last chunk can be ragged)
    if (blockIdx.y == gridDim.y - 1) {
      int64_t tail0 = i1 - (i1 % VEC);
      for (int64_t t = tail0 + threadIdx.x; t < i1; t += blockDim.x) {
        const float f = to_f32(p[t]);
        s += (double)f;
        ss += (double)(f * f);
      }
    }
  }
  for (int off = 32; off > 0; off >>= 1) {
    s += __shfl_down(s, off, 64);
    ss += __shfl_down(ss, off, 64);
  }
  __shared__ double ls[8], lss[8];
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  if (lane == 0) { ls[wid] = s; lss[wid] = ss; }
  __syncthreads();
  if (threadIdx.x == 0) {
    double ts = 0.0, tss = 0.0;
    for (int wv = 0; wv < (int)(blockDim.x >> 6); ++wv) { ts += ls[wv]; tss += lss[wv]; }
    atomicAdd(&out[ch], ts);
    atomicAdd(&out[C + ch], tss);
  }
}

template <typename T, bool RELU>
__global__ void bn_apply_kernel(const T* __restrict__ x, T* __restrict__ y,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ wgt,
                                const float* __restrict__ bias, int64_t N,
                                int64_t C, int64_t HW) {
  // 8 elements per thread (HW % 8 == 0 fast path enforced host-side)
  const int64_t total8 = N * C * HW / 8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total8; i8 += stride) {
    const int64_t i = i8 * 8;
    const int64_t ch = (i / HW) % C;
    const float m = mean[ch], inv = invstd[ch], w = wgt[ch], b = bias[ch];
    if (sizeof(T) == 2) {
      short8v vx = *(const short8v*)((const short*)x + i);
      short8v vy;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float v = (to_f32(((const T*)&vx)[e]) - m) * inv * w + b;
        if (RELU) v = v > 0.f ? v : 0.f;
        ((T*)&vy)[e] = (T)v;
      }
      *(short8v*)((short*)y + i) = vy;
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float v = ((float)x[i + e] - m) * inv * w + b;
        if (RELU) v = v > 0.f ? v : 0.f;
        y[i + e] = (T)v;
      }
    }
  }
}

// per-channel [gsum, gxsum] where gxsum = sum(go * xhat); RELU masks go
// by (y > 0) — pass y=nullptr when no fusion. 2D grid + vector loads.
template <typename T, bool RELU>
__global__ void bn_bwd_stats_kernel(const T* __restrict__ go,
                                    const T* __restrict__ x,
                                    const T* __restrict__ y,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    double* __restrict__ out, int64_t N,
                                    int64_t C, int64_t HW, int64_t chunk) {
  const int64_t ch = blockIdx.x;
  const int64_t i0 = (int64_t)blockIdx.y * chunk;
  const int64_t i1 = min(i0 + chunk, HW);
  const float m = mean[ch], inv = invstd[ch];
  double gs = 0.0, gx = 0.0;
  for (int64_t n = 0; n < N; ++n) {
    const int64_t off = (n * C + ch) * HW;
    if (sizeof(T) == 2) {
      int64_t i = i0 + (int64_t)threadIdx.x * 8;
      for (; i + 8 <= i1; i += (int64_t)blockDim.x * 8) {
        short8v vg = *(const short8v*)((const short*)go + off + i);
        short8v vx = *(const short8v*)((const short*)x + off + i);
        short8v vy;
        if (RELU) vy = *(const short8v*)((const short*)y + off + i);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float g = to_f32(((const T*)&vg)[e]);
          if (RELU && to_f32(((const T*)&vy)[e]) <= 0.f) g = 0.f;
          const float xh = (to_f32(((const T*)&vx)[e]) - m) * inv;
          gs += (double)g;
          gx += (double)(g * xh);
        }
      }
      if (blockIdx.y == gridDim.y - 1) {
        int64_t tail0 = i1 - (i1 % 8);
        for (int64_t t = tail0 + threadIdx.x; t < i1; t += blockDim.x) {
          float g = (float)go[off + t];
          if (RELU && (float)y[off + t] <= 0.f) g = 0.f;
          const float xh = ((float)x[off + t] - m) * inv;
          gs += (double)g;
          gx += (double)(g * xh);
        }
      }
    } else {
      for (int64_t i = i0 + threadIdx.x; i < i1; i += blockDim.x) {
        float g = (float)go[off + i];
        if (RELU && (float)y[off + i] <= 0.f) g = 0.f;
        const float xh = ((float)x[off + i] - m) * inv;
        gs += (double)g;
        gx += (double)(g * xh);
      }
    }
  }
  for (int off2 = 32; off2 > 0; off2 >>= 1) {
    gs += __shfl_down(gs, off2, 64);
    gx += __shfl_down(gx, off2, 64);
  }
  __shared__ double l1[8], l2[8];
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  if (lane == 0) { l1[wid] = gs; l2[wid] = gx; }
  __syncthreads();
  if (threadIdx.x == 0) {
    double t1 = 0.0, t2 = 0.0;
    for (int wv = 0; wv < (int)(blockDim.x >> 6); ++wv) { t1 += l1[wv]; t2 += l2[wv]; }
    atomicAdd(&out[ch], t1);
    atomicAdd(&out[C + ch], t2);
  }
}

template <typename T, bool RELU>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ go,
                                    const T* __restrict__ x,
                                    const T* __restrict__ y,
                                    T* __restrict__ gi,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ wgt,
                                    const float* __restrict__ gsum,
                                    const float* __restrict__ gxsum,
                                    double inv_n, int64_t N, int64_t C,
                                    int64_t HW) {
  const int64_t total8 = N * C * HW / 8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total8; i8 += stride) {
    const int64_t i = i8 * 8;
    const int64_t ch = (i / HW) % C;
    const float m = mean[ch], inv = invstd[ch], w = wgt[ch];
    const float gsn = (float)(gsum[ch] * inv_n);
    const float gxn = (float)(gxsum[ch] * inv_n);
    if (sizeof(T) == 2) {
      short8v vg = *(const short8v*)((const short*)go + i);
      short8v vx = *(const short8v*)((const short*)x + i);
      short8v vy;
      if (RELU) vy = *(const short8v*)((const short*)y + i);
      short8v vo;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float g = to_f32(((const T*)&vg)[e]);
        if (RELU && to_f32(((const T*)&vy)[e]) <= 0.f) g = 0.f;
        const float xh = (to_f32(((const T*)&vx)[e]) - m) * inv;
        ((T*)&vo)[e] = (T)((g - gsn - xh * gxn) * w * inv);
      }
      *(short8v*)((short*)gi + i) = vo;
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float g = (float)go[i + e];
        if (RELU && (float)y[i + e] <= 0.f) g = 0.f;
        const float xh = ((float)x[i + e] - m) * inv;
        gi[i + e] = (T)((g - gsn - xh * gxn) * w * inv);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Fused momentum-SGD on flat fp32 buffers: one kernel for the whole
// model step (replaces ~3 torch launches per parameter — the profile
// showed 5.8k tiny CUDAFunctor_add kernels per step).
//   v = mu*v + g (+ wd*p);  p -= lr*v;  then g = 0.
// ---------------------------------------------------------------------------

__global__ void sgd_momentum_kernel(float* __restrict__ p,
                                    float* __restrict__ g,
                                    float* __restrict__ v, float lr, float mu,
                                    float wd, int64_t n4) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 pv = *(float4*)(p + i * 4);
    float4 gv = *(float4*)(g + i * 4);
    float4 vv = *(float4*)(v + i * 4);
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      float grad = ((float*)&gv)[e] + wd * ((float*)&pv)[e];
      float vel = mu * ((float*)&vv)[e] + grad;
      ((float*)&vv)[e] = vel;
      ((float*)&pv)[e] = ((float*)&pv)[e] - lr * vel;
      ((float*)&gv)[e] = 0.f;
    }
    *(float4*)(p + i * 4) = pv;
    *(float4*)(v + i * 4) = vv;
    *(float4*)(g + i * 4) = gv;
  }
}

void sgd_momentum(torch::Tensor p, torch::Tensor g, torch::Tensor v,
                  double lr, double mu, double wd) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous() && g.is_contiguous() &&
              v.is_contiguous());
  TORCH_CHECK(p.scalar_type() == torch::kFloat);
  TORCH_CHECK(p.numel() % 4 == 0, "flat buffer must be padded to 4");
  const int64_t n4 = p.numel() / 4;
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = (int)std::min<int64_t>((n4 + 255) / 256, 2048);
  hipLaunchKernelGGL(sgd_momentum_kernel, dim3(grid), dim3(256), 0,
                     stream.stream(), p.data_ptr<float>(),
                     g.data_ptr<float>(), v.data_ptr<float>(), (float)lr,
                     (float)mu, (float)wd, n4);
}

// bn_finalize: fold the fp64 [sum,sumsq] -> [mean, invstd] conversion,
// the running-stat EMA update and num_batches_tracked into ONE kernel
// (the python glue cost ~8 small launches per BN call).
__global__ void bn_finalize_kernel(double* __restrict__ stats,
                                   float* __restrict__ out,
                                   float* __restrict__ rmean,
                                   float* __restrict__ rvar,
                                   int64_t* __restrict__ tracked, float mom,
                                   double n, float eps, int64_t C,
                                   int rezero) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= C) return;
  const double mean = stats[i] / n;
  double var = stats[C + i] / n - mean * mean;
  if (var < 0.0) var = 0.0;
  if (rezero) {
    stats[i] = 0.0;       // restore the zero invariant for the next
    stats[C + i] = 0.0;   // bn_stats64_acc accumulation (no fill kernel)
  }
  out[i] = (float)mean;
  out[C + i] = rsqrtf((float)var + eps);
  if (rmean) {
    rmean[i] = (1.f - mom) * rmean[i] + mom * (float)mean;
    const float unb = (float)(var * (n / (n > 1.0 ? n - 1.0 : 1.0)));
    rvar[i] = (1.f - mom) * rvar[i] + mom * unb;
  }
  if (i == 0 && tracked) tracked[0] += 1;
}

torch::Tensor bn_finalize(torch::Tensor stats,
                          c10::optional<torch::Tensor> rmean,
                          c10::optional<torch::Tensor> rvar,
                          c10::optional<torch::Tensor> tracked, double mom,
                          double n, double eps, bool rezero = false) {
  const int64_t C = stats.numel() / 2;
  auto out = torch::empty({2 * C}, stats.options().dtype(torch::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(
      bn_finalize_kernel, dim3((uint32_t)((C + 255) / 256)), dim3(256), 0,
      stream.stream(), stats.data_ptr<double>(), out.data_ptr<float>(),
      rmean.has_value() ? rmean->data_ptr<float>() : nullptr,
      rvar.has_value() ? rvar->data_ptr<float>() : nullptr,
      tracked.has_value() ? tracked->data_ptr<int64_t>() : nullptr,
      (float)mom, n, (float)eps, C, rezero ? 1 : 0);
  return out;
}


// ---------------- torch-facing wrappers ----------------

static inline int grid_for(int64_t total, int block) {
  int64_t g = (total + block - 1) / block;
  return (int)std::min<int64_t>(g, 4096);
}

std::vector<torch::Tensor> maxpool_fwd(torch::Tensor x, int64_t k, int64_t s,
                                       int64_t p) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous());
  const int64_t N = x.size(0), C = x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int OH = (H + 2 * (int)p - (int)k) / (int)s + 1;
  const int OW = (W + 2 * (int)p - (int)k) / (int)s + 1;
  auto y = torch::empty({N, C, OH, OW}, x.options());
  auto idx = torch::empty({N, C, OH, OW}, x.options().dtype(torch::kByte));
  const int64_t total = N * C * (int64_t)OH * OW;
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, x.scalar_type(),
      "maxpool_fwd", [&] {
        auto launch = [&](auto kernel) {
          hipLaunchKernelGGL(kernel, dim3(grid_for(total, 256)), dim3(256), 0,
                             stream.stream(), x.data_ptr<scalar_t>(),
                             y.data_ptr<scalar_t>(), idx.data_ptr<uint8_t>(),
                             N * C, H, W, OH, OW, (int)k, (int)s, (int)p);
        };
        if (k == 3 && s == 2)
          launch(maxpool_fwd_kernel<scalar_t, 3, 2>);
        else if (k == 3 && s == 1)
          launch(maxpool_fwd_kernel<scalar_t, 3, 1>);
        else if (k == 2 && s == 2)
          launch(maxpool_fwd_kernel<scalar_t, 2, 2>);
        else
          launch(maxpool_fwd_kernel<scalar_t, 0, 0>);
      });
  return {y, idx};
}

torch::Tensor maxpool_bwd(torch::Tensor go, torch::Tensor idx, int64_t H,
                          int64_t W, int64_t k, int64_t s, int64_t p) {
  TORCH_CHECK(go.is_cuda() && go.is_contiguous());
  const int64_t N = go.size(0), C = go.size(1);
  const int OH = (int)go.size(2), OW = (int)go.size(3);
  auto gi = torch::empty({N, C, H, W}, go.options());
  const int64_t total = N * C * H * W;
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, go.scalar_type(),
      "maxpool_bwd", [&] {
        auto launch = [&](auto kernel) {
          hipLaunchKernelGGL(kernel, dim3(grid_for(total, 256)), dim3(256), 0,
                             stream.stream(), go.data_ptr<scalar_t>(),
                             idx.data_ptr<uint8_t>(), gi.data_ptr<scalar_t>(),
                             N * C, (int)H, (int)W, OH, OW, (int)k, (int)s,
                             (int)p);
        };
        if (k == 3 && s == 1)
          launch(maxpool_bwd_kernel<scalar_t, 3, 1>);
        else if (k == 3 && s == 2)
          launch(maxpool_bwd_kernel<scalar_t, 3, 2>);
        else if (k == 2 && s == 2)
          launch(maxpool_bwd_kernel<scalar_t, 2, 2>);
        else
          launch(maxpool_bwd_kernel<scalar_t, 0, 0>);
      });
  return gi;
}

torch::Tensor avgpool_fwd(torch::Tensor x, int64_t k, int64_t s, int64_t p,
                          int64_t gr0, int64_t gc0, int64_t Hg, int64_t Wg,
                          bool include_pad) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous());
  const int64_t N = x.size(0), C = x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int OH = (H + 2 * (int)p - (int)k) / (int)s + 1;
  const int OW = (W + 2 * (int)p - (int)k) / (int)s + 1;
  auto y = torch::empty({N, C, OH, OW}, x.options());
  const int64_t total = N * C * (int64_t)OH * OW;
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, x.scalar_type(),
      "avgpool_fwd", [&] {
        auto launch = [&](auto kern) {
          hipLaunchKernelGGL(kern, dim3(grid_for(total, 256)), dim3(256), 0,
                             stream.stream(), x.data_ptr<scalar_t>(),
                             y.data_ptr<scalar_t>(), N * C, H, W, OH, OW,
                             (int)k, (int)s, (int)p, (long)gr0, (long)gc0,
                             (long)Hg, (long)Wg, include_pad ? 1 : 0);
        };
        if (k == 3 && s == 2) launch(avgpool_fwd_kernel<scalar_t, 3, 2>);
        else if (k == 3 && s == 1) launch(avgpool_fwd_kernel<scalar_t, 3, 1>);
        else launch(avgpool_fwd_kernel<scalar_t, 0, 0>);
      });
  return y;
}

torch::Tensor avgpool_bwd(torch::Tensor go, int64_t H, int64_t W, int64_t k,
                          int64_t s, int64_t p, int64_t gr0, int64_t gc0,
                          int64_t Hg, int64_t Wg, bool include_pad) {
  TORCH_CHECK(go.is_cuda() && go.is_contiguous());
  const int64_t N = go.size(0), C = go.size(1);
  const int OH = (int)go.size(2), OW = (int)go.size(3);
  auto gi = torch::empty({N, C, H, W}, go.options());
  const int64_t total = N * C * H * W;
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, go.scalar_type(),
      "avgpool_bwd", [&] {
        auto launch = [&](auto kern) {
          hipLaunchKernelGGL(kern, dim3(grid_for(total, 256)), dim3(256), 0,
                             stream.stream(), go.data_ptr<scalar_t>(),
                             gi.data_ptr<scalar_t>(), N * C, (int)H, (int)W,
                             OH, OW, (int)k, (int)s, (int)p, (long)gr0,
                             (long)gc0, (long)Hg, (long)Wg,
                             include_pad ? 1 : 0);
        };
        if (k == 3 && s == 2) launch(avgpool_bwd_kernel<scalar_t, 3, 2>);
        else if (k == 3 && s == 1) launch(avgpool_bwd_kernel<scalar_t, 3, 1>);
        else launch(avgpool_bwd_kernel<scalar_t, 0, 0>);
      });
  return gi;
}

static int64_t stats_splits(int64_t C, int64_t HW) {
  // target ~2048 workgroups (256 CUs x 8); chunk in multiples of 2048
  int64_t splits = std::max<int64_t>(1, 2048 / std::max<int64_t>(C, 1));
  int64_t max_splits = std::max<int64_t>(1, (HW + 2047) / 2048);
  return std::min(splits, max_splits);
}

torch::Tensor bn_stats64(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous());
  const int64_t N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto out = torch::zeros({2 * C}, x.options().dtype(torch::kDouble));
  const int64_t splits = stats_splits(C, HW);
  int64_t chunk = (HW + splits - 1) / splits;
  chunk = ((chunk + 2047) / 2048) * 2048;
  const int64_t gy = (HW + chunk - 1) / chunk;
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, x.scalar_type(),
      "bn_stats64", [&] {
        hipLaunchKernelGGL((bn_stats64_kernel<scalar_t>),
                           dim3((uint32_t)C, (uint32_t)gy), dim3(256), 0,
                           stream.stream(), x.data_ptr<scalar_t>(),
                           out.data_ptr<double>(), N, C, HW, chunk);
      });
  return out;
}

// accumulate into a caller-owned ZEROED fp64 buffer (persistent across
// steps; bn_finalize(..., rezero=true) restores the invariant) — kills
// the per-BN-call torch.zeros fill kernel (~3k launches/step).
torch::Tensor bn_stats64_acc(torch::Tensor x, torch::Tensor out) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous());
  const int64_t N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  TORCH_CHECK(out.is_cuda() && out.is_contiguous() &&
              out.scalar_type() == torch::kDouble && out.numel() == 2 * C);
  const int64_t splits = stats_splits(C, HW);
  int64_t chunk = (HW + splits - 1) / splits;
  chunk = ((chunk + 2047) / 2048) * 2048;
  const int64_t gy = (HW + chunk - 1) / chunk;
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, x.scalar_type(),
      "bn_stats64_acc", [&] {
        hipLaunchKernelGGL((bn_stats64_kernel<scalar_t>),
                           dim3((uint32_t)C, (uint32_t)gy), dim3(256), 0,
                           stream.stream(), x.data_ptr<scalar_t>(),
                           out.data_ptr<double>(), N, C, HW, chunk);
      });
  return out;
}

torch::Tensor bn_apply(torch::Tensor x, torch::Tensor mean,
                       torch::Tensor invstd, torch::Tensor w, torch::Tensor b,
                       bool relu) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous());
  const int64_t N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto y = torch::empty_like(x);
  const int64_t total = N * C * HW;
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, x.scalar_type(),
      "bn_apply", [&] {
        if (relu)
          hipLaunchKernelGGL((bn_apply_kernel<scalar_t, true>),
                             dim3(grid_for(total, 256)), dim3(256), 0,
                             stream.stream(), x.data_ptr<scalar_t>(),
                             y.data_ptr<scalar_t>(), mean.data_ptr<float>(),
                             invstd.data_ptr<float>(), w.data_ptr<float>(),
                             b.data_ptr<float>(), N, C, HW);
        else
          hipLaunchKernelGGL((bn_apply_kernel<scalar_t, false>),
                             dim3(grid_for(total, 256)), dim3(256), 0,
                             stream.stream(), x.data_ptr<scalar_t>(),
                             y.data_ptr<scalar_t>(), mean.data_ptr<float>(),
                             invstd.data_ptr<float>(), w.data_ptr<float>(),
                             b.data_ptr<float>(), N, C, HW);
      });
  return y;
}

torch::Tensor bn_bwd_stats(torch::Tensor go, torch::Tensor x, torch::Tensor y,
                           torch::Tensor mean, torch::Tensor invstd,
                           bool relu) {
  const int64_t N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto out = torch::zeros({2 * C}, x.options().dtype(torch::kDouble));
  const int64_t splits = stats_splits(C, HW);
  int64_t chunk = (HW + splits - 1) / splits;
  chunk = ((chunk + 2047) / 2048) * 2048;
  const int64_t gy = (HW + chunk - 1) / chunk;
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, x.scalar_type(),
      "bn_bwd_stats", [&] {
        if (relu)
          hipLaunchKernelGGL((bn_bwd_stats_kernel<scalar_t, true>),
                             dim3((uint32_t)C, (uint32_t)gy), dim3(256), 0,
                             stream.stream(), go.data_ptr<scalar_t>(),
                             x.data_ptr<scalar_t>(), y.data_ptr<scalar_t>(),
                             mean.data_ptr<float>(), invstd.data_ptr<float>(),
                             out.data_ptr<double>(), N, C, HW, chunk);
        else
          hipLaunchKernelGGL((bn_bwd_stats_kernel<scalar_t, false>),
                             dim3((uint32_t)C, (uint32_t)gy), dim3(256), 0,
                             stream.stream(), go.data_ptr<scalar_t>(),
                             x.data_ptr<scalar_t>(), (scalar_t*)nullptr,
                             mean.data_ptr<float>(), invstd.data_ptr<float>(),
                             out.data_ptr<double>(), N, C, HW, chunk);
      });
  return out;
}

torch::Tensor bn_bwd_apply(torch::Tensor go, torch::Tensor x, torch::Tensor y,
                           torch::Tensor mean, torch::Tensor invstd,
                           torch::Tensor w, torch::Tensor gsum,
                           torch::Tensor gxsum, double n, bool relu) {
  const int64_t N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto gi = torch::empty_like(x);
  const int64_t total = N * C * HW;
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::Half, at::ScalarType::BFloat16, x.scalar_type(),
      "bn_bwd_apply", [&] {
        if (relu)
          hipLaunchKernelGGL(
              (bn_bwd_apply_kernel<scalar_t, true>),
              dim3(grid_for(total, 256)), dim3(256), 0, stream.stream(),
              go.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
              y.data_ptr<scalar_t>(), gi.data_ptr<scalar_t>(),
              mean.data_ptr<float>(), invstd.data_ptr<float>(),
              w.data_ptr<float>(), gsum.data_ptr<float>(),
              gxsum.data_ptr<float>(), 1.0 / n, N, C, HW);
        else
          hipLaunchKernelGGL(
              (bn_bwd_apply_kernel<scalar_t, false>),
              dim3(grid_for(total, 256)), dim3(256), 0, stream.stream(),
              go.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
              (scalar_t*)nullptr, gi.data_ptr<scalar_t>(),
              mean.data_ptr<float>(), invstd.data_ptr<float>(),
              w.data_ptr<float>(), gsum.data_ptr<float>(),
              gxsum.data_ptr<float>(), 1.0 / n, N, C, HW);
      });
  return gi;
}

}  // namespace

void register_conv_mfma(pybind11::module_& m);
void register_conv_pw(pybind11::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  register_conv_mfma(m);
  register_conv_pw(m);
  m.def("halo_pack", &halo_copy<0>, "pack halo strips tile->flat buffer");
  m.def("halo_unpack", &halo_copy<1>, "unpack halo strips buffer->tile");
  m.def("halo_unpack_add", &halo_copy<2>, "accumulate grad strips into tile");
  m.def("bn_stats", &bn_stats, "per-channel [sum, sumsq] in one pass");
  m.def("maxpool_fwd", &maxpool_fwd);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("avgpool_fwd", &avgpool_fwd);
  m.def("avgpool_bwd", &avgpool_bwd);
  m.def("bn_stats64", &bn_stats64);
  m.def("bn_apply", &bn_apply);
  m.def("bn_bwd_stats", &bn_bwd_stats);
  m.def("bn_bwd_apply", &bn_bwd_apply);
  m.def("sgd_momentum", &sgd_momentum);
  m.def("bn_finalize", &bn_finalize, pybind11::arg("stats"),
        pybind11::arg("rmean"), pybind11::arg("rvar"),
        pybind11::arg("tracked"), pybind11::arg("mom"), pybind11::arg("n"),
        pybind11::arg("eps"), pybind11::arg("rezero") = false);
  m.def("bn_stats64_acc", &bn_stats64_acc);
  m.attr("gfx_arch") = "gfx950";
}
