// Fused conv3x3(pad1) + ReLU + maxpool2x2 kernels for gfx950 (MI355X).
//
// Motivation (profiles/mnist_b1024_eager_kernel_stats.md): MIOpen executes
// the benchmark MNIST-CNN as im2col + hundreds of small batched GEMMs per
// step plus a pathological find phase. The model's tensors are tiny, so
// the MI355X-native design stages whole per-sample planes in LDS and
// computes each layer in ONE kernel per direction.
//
// Iteration history (each measured with rocprofv3 / cuda events on MI355X;
// numbers are the conv2 shape [16ch 14x14] at batch 16384):
//   v1  one thread per output, global scalar loads -> cache-amplified.
//   v2  LDS-staged per-sample workgroups           -> bwd_weight-dominated.
//   v3  halo-padded planes (branch-free taps), bank-skewed strides,
//       chunked bwd_weight: fwd 1439us / bwd_data 1199us / bwd_w 914us.
//   v4  (this file) instruction-count pass:
//       - fwd computes a 2x2 POOLED block per thread from a 6x6 window
//         read as ds_read_b128+b64 rows (12 LDS instructions per channel
//         instead of 64 b32), 4x fewer address computations;
//       - bwd_data computes a 2x4 din block per thread from a 4x6 dconv
//         window (vectorized rows, 16 reads per 8 outputs vs 144);
//       - bwd_weight drops the scattered dconv plane entirely: the pool
//         gradient is pre-gated into a dense per-cell (g, argmax) pair at
//         stage time, inner loop reads 2 LDS values per cell + 9 taps.
//       Padded geometry guarantees every vector read is 16B-aligned
//       (guide §17) and channel strides are bank-skewed (mod 64 != 0).
//
// ReLU/maxpool tie-breaking matches torch: first index wins ties, and a
// pooled value of exactly 0 (all-negative window) propagates no gradient.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "ops_common.h"

namespace dmlamd {

typedef float f4 __attribute__((ext_vector_type(4)));
typedef float f2 __attribute__((ext_vector_type(2)));

// Padded plane geometry. Guard covers the 6x6 (fwd) / 4x6 (bwd) window
// over-read of ragged 2x-blocks; PADW is a multiple of 4 so row starts at
// 4-aligned x are 16B-aligned; cstride keeps 4-alignment and avoids
// cstride % 64 == 0 (cross-channel bank collisions).
struct PlaneGeom {
  int PADH, PADW, cs;
};

__host__ __device__ __forceinline__ PlaneGeom plane_geom(int H, int W) {
  PlaneGeom g;
  const int PH = H / 2, PW = W / 2;
  g.PADH = H + 2 + ((PH & 1) ? 2 : 0);
  int padw = W + 2 + ((PW & 1) ? 2 : 0);
  g.PADW = (padw + 3) & ~3;
  g.cs = g.PADH * g.PADW;
  // keep cs a multiple of 4 (16B-aligned rows) with cs/4 odd, so the
  // per-channel bank offset (cs mod 64) has gcd(cs,64)=4 and 16
  // simultaneous cross-channel b32 reads at equal (y,x) span 16 distinct
  // banks (measured: cs=360 aliased ci and ci+8 -> 5.2 conflict
  // cycles/LDS instruction in bwd_weight, profiles/smallcnn_pmc.txt)
  if ((g.cs & 7) == 0) g.cs += 4; // cs is a multiple of 4 by construction
  return g;
}

// Fill the interior of an (already zeroed) padded LDS region.
__device__ __forceinline__ void fill_plane_padded(const float* __restrict__ g,
                                                  float* __restrict__ lds, int CIN, int H,
                                                  int W, const PlaneGeom& pg) {
  const int plane = H * W;
  for (int i = threadIdx.x; i < CIN * plane; i += kBlock) {
    const int ci = i / plane;
    const int rem = i - ci * plane;
    const int y = rem / W, x = rem - y * W;
    lds[ci * pg.cs + (y + 1) * pg.PADW + (x + 1)] = g[i];
  }
}

__device__ __forceinline__ void stage_to_lds(const float* __restrict__ g, float* __restrict__ l,
                                             int count) {
  for (int i = threadIdx.x; i < count; i += kBlock) l[i] = g[i];
}

// Read 6 consecutive padded floats at a 16B-aligned offset.
__device__ __forceinline__ void read_row6(const float* __restrict__ p, float* __restrict__ row) {
  const f4 a = *(const f4*)p;
  const f2 b = *(const f2*)(p + 4);
  row[0] = a.x;
  row[1] = a.y;
  row[2] = a.z;
  row[3] = a.w;
  row[4] = b.x;
  row[5] = b.y;
}

// --------------------------------------------------------------- forward

// Workgroup = one sample (grid-stride). Thread = (co, 2x2 pooled block).
// LDS: [CIN][cs] halo-padded input + [COUT*CIN*9] weights.
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_fwd_kernel(
    const float* __restrict__ in, const float* __restrict__ w, const float* __restrict__ bias,
    float* __restrict__ out, uint8_t* __restrict__ argmax, int N, int CIN, int COUT, int H,
    int W) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const PlaneGeom pg = plane_geom(H, W);
  float* ilds = (float*)smem; // [CIN][cs]
  float* wlds = ilds + CIN * pg.cs; // [COUT][CIN][9]

  const int plane = CIN * H * W;
  const int PH = H / 2, PW = W / 2;
  const int BH = (PH + 1) / 2, BW = (PW + 1) / 2; // 2x2 pooled blocks
  const int nblocks = COUT * BH * BW;

  stage_to_lds(w, wlds, COUT * CIN * 9);
  for (int i = threadIdx.x; i < CIN * pg.cs; i += kBlock) ilds[i] = 0.0f; // halo, once
  __syncthreads();

  for (int n = blockIdx.x; n < N; n += gridDim.x) {
    fill_plane_padded(in + (int64_t)n * plane, ilds, CIN, H, W, pg);
    __syncthreads();

    for (int t = threadIdx.x; t < nblocks; t += kBlock) {
      const int bx = t % BW;
      const int by = (t / BW) % BH;
      const int co = t / (BW * BH);
      const int y0 = 4 * by, x0 = 4 * bx; // padded top-left of 6x6 window

      float conv[4][4];
      const float bz = bias[co];
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) conv[i][j] = bz;

      const float* wc = wlds + co * CIN * 9;
      for (int ci = 0; ci < CIN; ++ci) {
        const float* ip = ilds + ci * pg.cs + y0 * pg.PADW + x0;
        float win[6][6];
#pragma unroll
        for (int r = 0; r < 6; ++r) read_row6(ip + r * pg.PADW, win[r]);
        const float* wk = wc + ci * 9;
#pragma unroll
        for (int ky = 0; ky < 3; ++ky) {
#pragma unroll
          for (int kx = 0; kx < 3; ++kx) {
            const float wv = wk[ky * 3 + kx];
#pragma unroll
            for (int cy = 0; cy < 4; ++cy) {
#pragma unroll
              for (int cx = 0; cx < 4; ++cx) {
                conv[cy][cx] = fmaf(wv, win[cy + ky][cx + kx], conv[cy][cx]);
              }
            }
          }
        }
      }

      // relu + 2x2 max per pooled cell; first index wins ties
#pragma unroll
      for (int dy = 0; dy < 2; ++dy) {
        const int py = 2 * by + dy;
        if (py >= PH) break;
#pragma unroll
        for (int dx = 0; dx < 2; ++dx) {
          const int px = 2 * bx + dx;
          if (px >= PW) continue;
          const float r0 = fmaxf(conv[2 * dy][2 * dx], 0.0f);
          const float r1 = fmaxf(conv[2 * dy][2 * dx + 1], 0.0f);
          const float r2 = fmaxf(conv[2 * dy + 1][2 * dx], 0.0f);
          const float r3 = fmaxf(conv[2 * dy + 1][2 * dx + 1], 0.0f);
          float m = r0;
          int arg = 0;
          if (r1 > m) { m = r1; arg = 1; }
          if (r2 > m) { m = r2; arg = 2; }
          if (r3 > m) { m = r3; arg = 3; }
          const int64_t oidx = (((int64_t)n * COUT + co) * PH + py) * PW + px;
          out[oidx] = m;
          argmax[oidx] = (uint8_t)arg;
        }
      }
    }
    __syncthreads(); // before overwriting ilds for the next sample
  }
}

// ------------------------------------------------------------- bwd data

// Scatter the relu+maxpool-folded conv gradient of one sample into a
// halo-padded LDS plane (zeroed beforehand).
__device__ __forceinline__ void scatter_dconv_padded(const float* __restrict__ dpooled,
                                                     const uint8_t* __restrict__ argmax,
                                                     const float* __restrict__ pooled,
                                                     float* __restrict__ dclds, int64_t n,
                                                     int COUT, int H, int W,
                                                     const PlaneGeom& pg) {
  const int PH = H / 2, PW = W / 2;
  const int cells = COUT * PH * PW;
  for (int cell = threadIdx.x; cell < cells; cell += kBlock) {
    const int64_t pidx = n * cells + cell;
    const float pv = pooled[pidx];
    if (pv <= 0.0f) continue;
    const int sub = argmax[pidx];
    const int px = cell % PW;
    const int py = (cell / PW) % PH;
    const int co = cell / (PW * PH);
    const int yy = 2 * py + (sub >> 1);
    const int xx = 2 * px + (sub & 1);
    dclds[co * pg.cs + (yy + 1) * pg.PADW + (xx + 1)] = dpooled[pidx];
  }
}

// Workgroup = one sample. Thread = (ci, 2x4 din block): reads a 4x6
// dconv window per cout (vectorized rows), 8 outputs per thread.
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_bwd_data_kernel(
    const float* __restrict__ dpooled, const uint8_t* __restrict__ argmax,
    const float* __restrict__ pooled, const float* __restrict__ w, float* __restrict__ din,
    int N, int CIN, int COUT, int H, int W) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const PlaneGeom pg = plane_geom(H, W);
  float* dclds = (float*)smem; // [COUT][cs]
  float* wlds = dclds + COUT * pg.cs; // [COUT][CIN][9]

  const int BH = H / 2, BW = (W + 3) / 4; // 2x4 output blocks
  const int nblocks = CIN * BH * BW;

  stage_to_lds(w, wlds, COUT * CIN * 9);

  for (int n = blockIdx.x; n < N; n += gridDim.x) {
    for (int i = threadIdx.x; i < COUT * pg.cs; i += kBlock) dclds[i] = 0.0f;
    __syncthreads();
    scatter_dconv_padded(dpooled, argmax, pooled, dclds, n, COUT, H, W, pg);
    __syncthreads();

    for (int t = threadIdx.x; t < nblocks; t += kBlock) {
      const int bx = t % BW;
      const int by = (t / BW) % BH;
      const int ci = t / (BW * BH);
      const int y0 = 2 * by, x0 = 4 * bx; // din block top-left (unpadded)

      float acc[2][4] = {{0, 0, 0, 0}, {0, 0, 0, 0}};
      for (int co = 0; co < COUT; ++co) {
        // dconv rows y0-1..y0+2 -> padded rows y0..y0+3; cols x0-1..x0+4
        // -> padded cols x0..x0+5 (aligned: x0 multiple of 4)
        const float* dp = dclds + co * pg.cs + y0 * pg.PADW + x0;
        float win[4][6];
#pragma unroll
        for (int r = 0; r < 4; ++r) read_row6(dp + r * pg.PADW, win[r]);
        const float* wk = wlds + (co * CIN + ci) * 9;
        // din(y,x) += w[ky][kx] * dconv(y-ky+1, x-kx+1)
        // local: dconv(y0+dy-ky+1, x0+dx-kx+1) = win[dy-ky+2][dx-kx+2]
#pragma unroll
        for (int ky = 0; ky < 3; ++ky) {
#pragma unroll
          for (int kx = 0; kx < 3; ++kx) {
            const float wv = wk[ky * 3 + kx];
#pragma unroll
            for (int dy = 0; dy < 2; ++dy) {
#pragma unroll
              for (int dx = 0; dx < 4; ++dx) {
                acc[dy][dx] = fmaf(wv, win[dy - ky + 2][dx - kx + 2], acc[dy][dx]);
              }
            }
          }
        }
      }
#pragma unroll
      for (int dy = 0; dy < 2; ++dy) {
#pragma unroll
        for (int dx = 0; dx < 4; ++dx) {
          const int x = x0 + dx;
          if (x < W) din[(((int64_t)n * CIN + ci) * H + y0 + dy) * W + x] = acc[dy][dx];
        }
      }
    }
    __syncthreads();
  }
}

// ----------------------------------------------------------- bwd weight

// Workgroup = chunk of `samples` samples looped through LDS. The pool
// gradient is pre-gated at stage time into dense per-cell (g, sub) pairs
// (g = 0 encodes "no gradient"). Thread owns ((co,ci), slice): 9 register
// partials across the chunk; LDS tree across slices; one atomicAdd per
// tap per workgroup.
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_bwd_weight_kernel(
    const float* __restrict__ dpooled, const uint8_t* __restrict__ argmax,
    const float* __restrict__ pooled, const float* __restrict__ in, float* __restrict__ dw,
    float* __restrict__ db, int N, int CIN, int COUT, int H, int W, int samples) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const PlaneGeom pg = plane_geom(H, W);
  const int iplane = CIN * H * W;
  const int PH = H / 2, PW = W / 2;
  const int pcells = PH * PW;
  const int pstride = pcells + ((pcells % 64) ? 0 : 4) + 1; // bank skew
  float* ilds = (float*)smem; // [CIN][cs]
  float* glds = ilds + CIN * pg.cs; // [COUT][pstride] pre-gated gradients
  float* slds = glds + COUT * pstride; // [COUT][pstride] argmax as float
  float* redlds = slds + COUT * pstride; // [kBlock]

  const int cells = COUT * pcells;

  const int pairs = COUT * CIN;
  const int nslices = max(1, kBlock / pairs);
  const int pair = threadIdx.x % pairs;
  const int slice = threadIdx.x / pairs;
  const int co = pair / CIN;
  const int ci = pair % CIN;
  const bool active = threadIdx.x < pairs * nslices;

  const int n0 = blockIdx.x * samples;
  const int nvalid = min(samples, N - n0);

  float acc[9] = {0, 0, 0, 0, 0, 0, 0, 0, 0};
  float accb = 0.0f;

  // zero the input halo ONCE; interiors are overwritten every sample
  for (int i = threadIdx.x; i < CIN * pg.cs; i += kBlock) ilds[i] = 0.0f;
  __syncthreads();

  for (int s = 0; s < nvalid; ++s) {
    const int64_t n = n0 + s;
    fill_plane_padded(in + n * iplane, ilds, CIN, H, W, pg);
    for (int cell = threadIdx.x; cell < cells; cell += kBlock) {
      const int64_t pidx = n * cells + cell;
      const int cco = cell / pcells;
      const int rem = cell - cco * pcells;
      const float pv = pooled[pidx];
      glds[cco * pstride + rem] = (pv > 0.0f) ? dpooled[pidx] : 0.0f;
      // pre-decode the argmax into a padded-plane offset so the hot loop
      // below needs no div/mod per cell (offsets < 2048 are exact floats)
      const int sub = argmax[pidx];
      const int px = rem % PW;
      const int py = rem / PW;
      const int yy = 2 * py + (sub >> 1);
      const int xx = 2 * px + (sub & 1);
      slds[cco * pstride + rem] = (float)(yy * pg.PADW + xx);
    }
    __syncthreads();

    if (active) {
      const float* ip = ilds + ci * pg.cs;
      const float* gp = glds + co * pstride;
      const float* sp = slds + co * pstride;
      for (int cell = slice; cell < pcells; cell += nslices) {
        // branchless: g == 0 cells contribute 0 to every accumulator, and
        // their stored offset still addresses in-bounds LDS — skipping
        // them would diverge the wave at 16-lane granularity
        const float g = gp[cell];
        const int off = (int)sp[cell];
        if (ci == 0) accb += g;
        // input rows yy-1..yy+1 -> padded rows yy..yy+2, cols likewise
        const float* iw = ip + off;
#pragma unroll
        for (int ky = 0; ky < 3; ++ky) {
#pragma unroll
          for (int kx = 0; kx < 3; ++kx) {
            acc[ky * 3 + kx] = fmaf(g, iw[ky * pg.PADW + kx], acc[ky * 3 + kx]);
          }
        }
      }
    }
    __syncthreads();
  }

  // cross-slice reduction (skipped when nslices == 1)
  if (nslices > 1) {
#pragma unroll
    for (int k = 0; k < 9; ++k) {
      redlds[threadIdx.x] = active ? acc[k] : 0.0f;
      __syncthreads();
      if (active && slice == 0) {
        float total = acc[k];
        for (int sl = 1; sl < nslices; ++sl) total += redlds[sl * pairs + pair];
        acc[k] = total;
      }
      __syncthreads();
    }
    redlds[threadIdx.x] = (active && ci == 0) ? accb : 0.0f;
    __syncthreads();
    if (active && slice == 0 && ci == 0) {
      float total = accb;
      for (int sl = 1; sl < nslices; ++sl) total += redlds[sl * pairs + pair];
      accb = total;
    }
  }

  if (active && slice == 0) {
#pragma unroll
    for (int k = 0; k < 9; ++k) atomicAdd(&dw[pair * 9 + k], acc[k]);
    if (ci == 0 && db != nullptr) atomicAdd(&db[co], accb);
  }
}

// ------------------------------------------------------------ launchers

static constexpr int kMaxLds = 160 * 1024;

void conv3x3_relu_pool_fwd(at::Tensor in, at::Tensor w, at::Tensor b, at::Tensor out,
                           at::Tensor argmax) {
  TORCH_CHECK(in.is_cuda() && in.scalar_type() == at::kFloat && in.is_contiguous(),
              "input must be contiguous fp32 on device");
  TORCH_CHECK(w.is_contiguous() && b.is_contiguous(), "weights must be contiguous");
  TORCH_CHECK(w.size(2) == 3 && w.size(3) == 3, "kernel must be 3x3");
  const int N = in.size(0), CIN = in.size(1), H = in.size(2), W = in.size(3);
  const int COUT = w.size(0);
  TORCH_CHECK(H % 2 == 0 && W % 2 == 0, "H and W must be even for 2x2 pooling");
  TORCH_CHECK(w.size(1) == CIN, "weight in-channels must match input");
  TORCH_CHECK(b.numel() == COUT, "bias must be [COUT]");
  TORCH_CHECK(out.numel() == (int64_t)N * COUT * (H / 2) * (W / 2) && out.is_contiguous(),
              "out must be contiguous [N,COUT,H/2,W/2]");
  TORCH_CHECK(argmax.numel() == out.numel() && argmax.scalar_type() == at::kByte,
              "argmax must be uint8 like out");
  const PlaneGeom pg = plane_geom(H, W);
  const int lds_bytes = (CIN * pg.cs + COUT * CIN * 9) * (int)sizeof(float);
  TORCH_CHECK(lds_bytes <= kMaxLds, "plane+weights exceed LDS (", lds_bytes, " B)");
  auto stream = c10::hip::getCurrentHIPStream();
  const int blocks = std::min(N, kMaxGrid);
  hipLaunchKernelGGL(conv3x3_relu_pool_fwd_kernel, dim3(blocks), dim3(kBlock), lds_bytes, stream,
                     in.data_ptr<float>(), w.data_ptr<float>(), b.data_ptr<float>(),
                     out.data_ptr<float>(), argmax.data_ptr<uint8_t>(), N, CIN, COUT, H, W);
}

void conv3x3_relu_pool_bwd_data(at::Tensor dpooled, at::Tensor argmax, at::Tensor pooled,
                                at::Tensor w, at::Tensor din) {
  const int N = din.size(0), CIN = din.size(1), H = din.size(2), W = din.size(3);
  const int COUT = w.size(0);
  TORCH_CHECK(w.size(1) == CIN && din.is_contiguous(), "weight/din geometry mismatch");
  TORCH_CHECK(dpooled.numel() == (int64_t)N * COUT * (H / 2) * (W / 2) &&
                  argmax.numel() == dpooled.numel() && pooled.numel() == dpooled.numel(),
              "dpooled/argmax/pooled must be [N,COUT,H/2,W/2]");
  const PlaneGeom pg = plane_geom(H, W);
  const int lds_bytes = (COUT * pg.cs + COUT * CIN * 9) * (int)sizeof(float);
  TORCH_CHECK(lds_bytes <= kMaxLds, "dconv plane exceeds LDS");
  auto stream = c10::hip::getCurrentHIPStream();
  const int blocks = std::min(N, kMaxGrid);
  hipLaunchKernelGGL(conv3x3_relu_pool_bwd_data_kernel, dim3(blocks), dim3(kBlock), lds_bytes,
                     stream, dpooled.data_ptr<float>(), argmax.data_ptr<uint8_t>(),
                     pooled.data_ptr<float>(), w.data_ptr<float>(), din.data_ptr<float>(), N,
                     CIN, COUT, H, W);
}

void conv3x3_relu_pool_bwd_weight(at::Tensor dpooled, at::Tensor argmax, at::Tensor pooled,
                                  at::Tensor in, at::Tensor dw, at::Tensor db) {
  const int N = in.size(0), CIN = in.size(1), H = in.size(2), W = in.size(3);
  const int COUT = dw.size(0);
  TORCH_CHECK(COUT * CIN <= kBlock, "bwd_weight supports COUT*CIN <= ", kBlock);
  const PlaneGeom pg = plane_geom(H, W);
  const int pcells = (H / 2) * (W / 2);
  const int pstride = pcells + ((pcells % 64) ? 0 : 4) + 1;
  const int lds_bytes =
      (CIN * pg.cs + 2 * COUT * pstride + kBlock) * (int)sizeof(float);
  TORCH_CHECK(lds_bytes <= kMaxLds, "bwd_weight staging exceeds LDS (", lds_bytes, " B)");
  // chunk so that the grid stays ~>= 512 blocks while cutting atomics
  // (DMLCLOUD_SMALLCNN_BWDW_CHUNK overrides for tuning experiments)
  int samples = 1;
  if (const char* env = std::getenv("DMLCLOUD_SMALLCNN_BWDW_CHUNK")) {
    samples = std::max(1, atoi(env));
  } else {
    while (samples < 16 && (N + samples * 2 - 1) / (samples * 2) >= 512) samples *= 2;
  }
  const int blocks = (N + samples - 1) / samples;
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(conv3x3_relu_pool_bwd_weight_kernel, dim3(blocks), dim3(kBlock), lds_bytes,
                     stream, dpooled.data_ptr<float>(), argmax.data_ptr<uint8_t>(),
                     pooled.data_ptr<float>(), in.data_ptr<float>(), dw.data_ptr<float>(),
                     db.defined() ? db.data_ptr<float>() : nullptr, N, CIN, COUT, H, W, samples);
}

} // namespace dmlamd
