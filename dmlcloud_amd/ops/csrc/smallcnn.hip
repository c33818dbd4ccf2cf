// Fused conv3x3(pad1) + ReLU + maxpool2x2 kernels for gfx950 (MI355X).
//
// Motivation (profiles/mnist_b1024_eager_kernel_stats.md): MIOpen executes
// the benchmark MNIST-CNN as im2col + hundreds of small batched GEMMs per
// step plus a pathological find phase. The model's tensors are tiny, so
// the MI355X-native design stages whole per-sample planes in LDS and
// computes each layer in ONE kernel per direction.
//
// Iteration history (each measured with rocprofv3 on MI355X):
//   v1  one thread per output, global scalar loads       -> cache-amplified,
//       slower than MIOpen.
//   v2  LDS-staged per-sample workgroups                 -> 3x faster, but
//       bwd_weight dominated (global atomics per 2-sample block) and the
//       in-loop bounds checks serialized the tap loops.
//   v3  (this file) adds: halo-padded LDS planes so every 3x3/4x4 tap loop
//       is branch-free; +4-float channel strides so simultaneous
//       cross-channel LDS reads land on different banks; bwd_weight loops
//       a multi-sample chunk inside one block (register accumulators live
//       across the chunk), cutting global atomics by the chunk factor.
//
// ReLU/maxpool tie-breaking matches torch: first index wins ties, and a
// pooled value of exactly 0 (all-negative window) propagates no gradient.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "ops_common.h"

namespace dmlamd {

// Padded channel stride: (H+2) halo rows x (W+2) halo cols, plus 4 floats
// so that simultaneous reads at equal (y,x) across channels hit different
// LDS banks ((H+2)*(W+2) is typically a multiple of 64).
__host__ __device__ __forceinline__ int padded_cstride(int H, int W) {
  return (H + 2) * (W + 2) + 4;
}

// Zero an LDS range then fill its interior with one input plane (halo stays 0).
__device__ __forceinline__ void stage_plane_padded(const float* __restrict__ g,
                                                   float* __restrict__ lds, int CIN, int H,
                                                   int W) {
  const int cs = padded_cstride(H, W), PADW = W + 2;
  for (int i = threadIdx.x; i < CIN * cs; i += kBlock) lds[i] = 0.0f;
  __syncthreads();
  const int plane = H * W;
  for (int i = threadIdx.x; i < CIN * plane; i += kBlock) {
    const int ci = i / plane;
    const int rem = i - ci * plane;
    const int y = rem / W, x = rem - y * W;
    lds[ci * cs + (y + 1) * PADW + (x + 1)] = g[i];
  }
}

__device__ __forceinline__ void stage_to_lds(const float* __restrict__ g, float* __restrict__ l,
                                             int count) {
  for (int i = threadIdx.x; i < count; i += kBlock) l[i] = g[i];
}

// --------------------------------------------------------------- forward

// Workgroup = one sample (grid-stride over samples).
// LDS: [CIN][cstride] halo-padded input plane + [COUT*CIN*9] weights.
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_fwd_kernel(
    const float* __restrict__ in, const float* __restrict__ w, const float* __restrict__ bias,
    float* __restrict__ out, uint8_t* __restrict__ argmax, int N, int CIN, int COUT, int H,
    int W) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int cs = padded_cstride(H, W), PADW = W + 2;
  float* ilds = (float*)smem; // [CIN][cs]
  float* wlds = ilds + CIN * cs; // [COUT][CIN][9]

  const int plane = CIN * H * W;
  const int PH = H / 2, PW = W / 2;
  const int outs = COUT * PH * PW;

  stage_to_lds(w, wlds, COUT * CIN * 9);

  for (int n = blockIdx.x; n < N; n += gridDim.x) {
    stage_plane_padded(in + (int64_t)n * plane, ilds, CIN, H, W);
    __syncthreads();

    for (int o = threadIdx.x; o < outs; o += kBlock) {
      const int px = o % PW;
      const int py = (o / PW) % PH;
      const int co = o / (PW * PH);
      const int y0 = 2 * py, x0 = 2 * px; // top-left of the padded 4x4 window

      float acc0 = bias[co], acc1 = acc0, acc2 = acc0, acc3 = acc0;
      const float* wc = wlds + co * CIN * 9;
      for (int ci = 0; ci < CIN; ++ci) {
        const float* ip = ilds + ci * cs + y0 * PADW + x0;
        const float* wk = wc + ci * 9;
        float win[4][4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
#pragma unroll
          for (int c = 0; c < 4; ++c) win[r][c] = ip[r * PADW + c];
        }
#pragma unroll
        for (int ky = 0; ky < 3; ++ky) {
#pragma unroll
          for (int kx = 0; kx < 3; ++kx) {
            const float wv = wk[ky * 3 + kx];
            acc0 = fmaf(wv, win[ky][kx], acc0);
            acc1 = fmaf(wv, win[ky][kx + 1], acc1);
            acc2 = fmaf(wv, win[ky + 1][kx], acc2);
            acc3 = fmaf(wv, win[ky + 1][kx + 1], acc3);
          }
        }
      }

      float r0 = fmaxf(acc0, 0.0f), r1 = fmaxf(acc1, 0.0f);
      float r2 = fmaxf(acc2, 0.0f), r3 = fmaxf(acc3, 0.0f);
      float m = r0;
      int arg = 0;
      if (r1 > m) { m = r1; arg = 1; }
      if (r2 > m) { m = r2; arg = 2; }
      if (r3 > m) { m = r3; arg = 3; }
      const int64_t oidx = (int64_t)n * outs + o;
      out[oidx] = m;
      argmax[oidx] = (uint8_t)arg;
    }
    __syncthreads(); // before overwriting ilds for the next sample
  }
}

// ------------------------------------------------------------- bwd data

// Scatter the relu+maxpool-folded conv gradient of one sample into a
// halo-padded LDS plane (zeroed by the caller's stage).
__device__ __forceinline__ void scatter_dconv_padded(const float* __restrict__ dpooled,
                                                     const uint8_t* __restrict__ argmax,
                                                     const float* __restrict__ pooled,
                                                     float* __restrict__ dclds, int64_t n,
                                                     int COUT, int H, int W) {
  const int cs = padded_cstride(H, W), PADW = W + 2;
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
    dclds[co * cs + (yy + 1) * PADW + (xx + 1)] = dpooled[pidx];
  }
}

// Workgroup = one sample. LDS: [COUT][cstride] dconv plane + weights.
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_bwd_data_kernel(
    const float* __restrict__ dpooled, const uint8_t* __restrict__ argmax,
    const float* __restrict__ pooled, const float* __restrict__ w, float* __restrict__ din,
    int N, int CIN, int COUT, int H, int W) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int cs = padded_cstride(H, W), PADW = W + 2;
  float* dclds = (float*)smem; // [COUT][cs]
  float* wlds = dclds + COUT * cs; // [COUT][CIN][9]

  const int iplane = CIN * H * W;

  stage_to_lds(w, wlds, COUT * CIN * 9);

  for (int n = blockIdx.x; n < N; n += gridDim.x) {
    for (int i = threadIdx.x; i < COUT * cs; i += kBlock) dclds[i] = 0.0f;
    __syncthreads();
    scatter_dconv_padded(dpooled, argmax, pooled, dclds, n, COUT, H, W);
    __syncthreads();

    for (int o = threadIdx.x; o < iplane; o += kBlock) {
      const int x = o % W;
      const int y = (o / W) % H;
      const int ci = o / (W * H);
      float acc = 0.0f;
      for (int co = 0; co < COUT; ++co) {
        // correlation with flipped kernel: dconv(y-ky+1, x-kx+1) ->
        // padded row (y-ky+2), col (x-kx+2); halo absorbs the bounds.
        const float* dp = dclds + co * cs + (y + 2) * PADW + (x + 2);
        const float* wk = wlds + (co * CIN + ci) * 9;
#pragma unroll
        for (int ky = 0; ky < 3; ++ky) {
#pragma unroll
          for (int kx = 0; kx < 3; ++kx) {
            acc = fmaf(wk[ky * 3 + kx], dp[-ky * PADW - kx], acc);
          }
        }
      }
      din[(int64_t)n * iplane + o] = acc;
    }
    __syncthreads();
  }
}

// ----------------------------------------------------------- bwd weight

// Workgroup = a chunk of `samples` samples, looped one at a time through
// LDS. Thread owns ((co,ci) pair, slice); its 9 register partials live
// across the whole chunk; LDS tree across slices; one global atomicAdd
// per tap per workgroup at the end.
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_bwd_weight_kernel(
    const float* __restrict__ dpooled, const uint8_t* __restrict__ argmax,
    const float* __restrict__ pooled, const float* __restrict__ in, float* __restrict__ dw,
    float* __restrict__ db, int N, int CIN, int COUT, int H, int W, int samples) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int cs = padded_cstride(H, W), PADW = W + 2;
  const int iplane = CIN * H * W;
  const int dplane = COUT * H * W; // dconv plane, unpadded (+4 bank skew)
  const int dstride = H * W + 4;
  float* ilds = (float*)smem; // [CIN][cs]
  float* dclds = ilds + CIN * cs; // [COUT][dstride]
  float* redlds = dclds + COUT * dstride; // [kBlock]

  const int PH = H / 2, PW = W / 2;
  const int cells = COUT * PH * PW;

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

  for (int s = 0; s < nvalid; ++s) {
    const int64_t n = n0 + s;
    stage_plane_padded(in + n * iplane, ilds, CIN, H, W);
    for (int i = threadIdx.x; i < COUT * dstride; i += kBlock) dclds[i] = 0.0f;
    __syncthreads();
    for (int cell = threadIdx.x; cell < cells; cell += kBlock) {
      const int64_t pidx = n * cells + cell;
      const float pv = pooled[pidx];
      if (pv <= 0.0f) continue;
      const int sub = argmax[pidx];
      const int px = cell % PW;
      const int py = (cell / PW) % PH;
      const int cco = cell / (PW * PH);
      const int yy = 2 * py + (sub >> 1);
      const int xx = 2 * px + (sub & 1);
      dclds[cco * dstride + yy * W + xx] = dpooled[pidx];
    }
    __syncthreads();

    if (active) {
      const float* ip = ilds + ci * cs;
      const float* dp = dclds + co * dstride;
      for (int yx = slice; yx < H * W; yx += nslices) {
        const float g = dp[yx];
        if (g == 0.0f) continue;
        const int yy = yx / W, xx = yx - yy * W;
        if (ci == 0) accb += g;
        // input window rows yy-1..yy+1 -> padded rows yy..yy+2
        const float* iw = ip + yy * PADW + xx;
#pragma unroll
        for (int ky = 0; ky < 3; ++ky) {
#pragma unroll
          for (int kx = 0; kx < 3; ++kx) {
            acc[ky * 3 + kx] = fmaf(g, iw[ky * PADW + kx], acc[ky * 3 + kx]);
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
  const int lds_bytes = (CIN * padded_cstride(H, W) + COUT * CIN * 9) * (int)sizeof(float);
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
  const int lds_bytes = (COUT * padded_cstride(H, W) + COUT * CIN * 9) * (int)sizeof(float);
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
  const int lds_bytes =
      (CIN * padded_cstride(H, W) + COUT * (H * W + 4) + kBlock) * (int)sizeof(float);
  TORCH_CHECK(lds_bytes <= kMaxLds, "bwd_weight staging exceeds LDS (", lds_bytes, " B)");
  // chunk so that the grid stays ~>= 512 blocks (occupancy) while cutting
  // the per-chunk global atomics by the chunk factor
  int samples = 1;
  while (samples < 16 && (N + samples * 2 - 1) / (samples * 2) >= 512) samples *= 2;
  const int blocks = (N + samples - 1) / samples;
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(conv3x3_relu_pool_bwd_weight_kernel, dim3(blocks), dim3(kBlock), lds_bytes,
                     stream, dpooled.data_ptr<float>(), argmax.data_ptr<uint8_t>(),
                     pooled.data_ptr<float>(), in.data_ptr<float>(), dw.data_ptr<float>(),
                     db.defined() ? db.data_ptr<float>() : nullptr, N, CIN, COUT, H, W, samples);
}

} // namespace dmlamd
