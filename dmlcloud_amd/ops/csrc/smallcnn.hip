// Fused conv3x3(pad1) + ReLU + maxpool2x2 kernels for gfx950 (MI355X).
//
// Motivation (profiles/mnist_b1024_eager_kernel_stats.md): MIOpen executes
// the benchmark MNIST-CNN as im2col + hundreds of small batched GEMMs per
// step plus a pathological find phase. The model's tensors are tiny, so
// the MI355X-native design stages whole per-sample planes in LDS and
// computes each layer in ONE kernel per direction:
//
//   forward     — one workgroup per sample: the input plane (CIN x H x W)
//                 and the weights are loaded into LDS once; every conv+
//                 relu+pool output of the sample is computed from LDS.
//                 Global traffic = read input once, write pooled+argmax
//                 once (the theoretical minimum).
//   bwd_data    — one workgroup per sample: the sparse conv-gradient
//                 plane (relu+maxpool gradient folded in via the saved
//                 argmax) is materialized in LDS, then correlated with
//                 the flipped weights from LDS.
//   bwd_weight  — one workgroup per S-sample chunk: samples' input and
//                 gradient planes staged in LDS; each thread owns one
//                 (cout,cin) pair slice and accumulates its 9 taps in
//                 registers; cross-slice LDS tree; one global atomicAdd
//                 per output per workgroup.
//
// A first version of these kernels used one thread per output with global
// scalar loads: numerically identical but ~64x cache-amplified and
// latency-bound — slower than MIOpen. The LDS-staged structure below is
// the fix (guide §2/§6: stage reused tiles in LDS, coalesce global).
//
// ReLU/maxpool tie-breaking matches torch: first index wins ties, and a
// pooled value of exactly 0 (all-negative window) propagates no gradient.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "ops_common.h"

namespace dmlamd {

// Cooperative coalesced copy of `count` floats global->LDS.
__device__ __forceinline__ void stage_to_lds(const float* __restrict__ g, float* __restrict__ l,
                                             int count) {
  for (int i = threadIdx.x; i < count; i += kBlock) l[i] = g[i];
}

// --------------------------------------------------------------- forward

// Workgroup = one sample. LDS: [CIN*H*W] input plane + [COUT*CIN*9] weights.
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_fwd_kernel(
    const float* __restrict__ in, const float* __restrict__ w, const float* __restrict__ bias,
    float* __restrict__ out, uint8_t* __restrict__ argmax, int N, int CIN, int COUT, int H,
    int W) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* ilds = (float*)smem; // [CIN][H][W]
  float* wlds = ilds + CIN * H * W; // [COUT][CIN][9]

  const int plane = CIN * H * W;
  const int PH = H / 2, PW = W / 2;
  const int outs = COUT * PH * PW;

  for (int n = blockIdx.x; n < N; n += gridDim.x) {
    stage_to_lds(in + (int64_t)n * plane, ilds, plane);
    if (n == blockIdx.x) stage_to_lds(w, wlds, COUT * CIN * 9); // once per block
    __syncthreads();

    for (int o = threadIdx.x; o < outs; o += kBlock) {
      const int px = o % PW;
      const int py = (o / PW) % PH;
      const int co = o / (PW * PH);
      const int y0 = 2 * py, x0 = 2 * px;

      float acc0 = bias[co], acc1 = acc0, acc2 = acc0, acc3 = acc0;
      const float* wc = wlds + co * CIN * 9;
      for (int ci = 0; ci < CIN; ++ci) {
        const float* ip = ilds + ci * H * W;
        const float* wk = wc + ci * 9;
        float win[4][4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int yy = y0 - 1 + r;
#pragma unroll
          for (int c = 0; c < 4; ++c) {
            const int xx = x0 - 1 + c;
            win[r][c] = (yy >= 0 && yy < H && xx >= 0 && xx < W) ? ip[yy * W + xx] : 0.0f;
          }
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

// Workgroup = one sample. LDS: [COUT*H*W] dconv plane + [COUT*CIN*9] weights.
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_bwd_data_kernel(
    const float* __restrict__ dpooled, const uint8_t* __restrict__ argmax,
    const float* __restrict__ pooled, const float* __restrict__ w, float* __restrict__ din,
    int N, int CIN, int COUT, int H, int W) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* dclds = (float*)smem; // [COUT][H][W]
  float* wlds = dclds + COUT * H * W; // [COUT][CIN][9]

  const int PH = H / 2, PW = W / 2;
  const int cells = COUT * PH * PW;
  const int dplane = COUT * H * W;
  const int iplane = CIN * H * W;

  for (int n = blockIdx.x; n < N; n += gridDim.x) {
    if (n == blockIdx.x) stage_to_lds(w, wlds, COUT * CIN * 9);
    // build the sparse dconv plane in LDS
    for (int i = threadIdx.x; i < dplane; i += kBlock) dclds[i] = 0.0f;
    __syncthreads();
    for (int cell = threadIdx.x; cell < cells; cell += kBlock) {
      const int64_t pidx = (int64_t)n * cells + cell;
      const float pv = pooled[pidx];
      if (pv <= 0.0f) continue;
      const float g = dpooled[pidx];
      const int sub = argmax[pidx];
      const int px = cell % PW;
      const int py = (cell / PW) % PH;
      const int co = cell / (PW * PH);
      const int yy = 2 * py + (sub >> 1);
      const int xx = 2 * px + (sub & 1);
      dclds[co * H * W + yy * W + xx] = g; // each cell owns its 2x2 patch
    }
    __syncthreads();

    for (int o = threadIdx.x; o < iplane; o += kBlock) {
      const int x = o % W;
      const int y = (o / W) % H;
      const int ci = o / (W * H);
      float acc = 0.0f;
      for (int co = 0; co < COUT; ++co) {
        const float* dp = dclds + co * H * W;
        const float* wk = wlds + (co * CIN + ci) * 9;
#pragma unroll
        for (int ky = 0; ky < 3; ++ky) {
          const int yy = y - ky + 1;
          if (yy < 0 || yy >= H) continue;
#pragma unroll
          for (int kx = 0; kx < 3; ++kx) {
            const int xx = x - kx + 1;
            if (xx < 0 || xx >= W) continue;
            acc = fmaf(wk[ky * 3 + kx], dp[yy * W + xx], acc);
          }
        }
      }
      din[(int64_t)n * iplane + o] = acc;
    }
    __syncthreads();
  }
}

// ----------------------------------------------------------- bwd weight

// Workgroup = chunk of SAMPLES samples. LDS: inputs [S][CIN*H*W] +
// dconv planes rebuilt sparse [S][COUT*H*W]. Thread owns (pair, slice):
// pair = (co,ci), slices split the cell space; 9 register partials each,
// LDS tree across slices, one global atomicAdd per (pair, tap).
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_bwd_weight_kernel(
    const float* __restrict__ dpooled, const uint8_t* __restrict__ argmax,
    const float* __restrict__ pooled, const float* __restrict__ in, float* __restrict__ dw,
    float* __restrict__ db, int N, int CIN, int COUT, int H, int W, int samples) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int iplane = CIN * H * W;
  const int dplane = COUT * H * W;
  float* ilds = (float*)smem; // [S][CIN*H*W]
  float* dclds = ilds + (int64_t)samples * iplane; // [S][COUT*H*W]
  float* redlds = dclds + (int64_t)samples * dplane; // [kBlock] scratch

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

  // stage inputs + rebuild dconv planes for the chunk
  for (int s = 0; s < nvalid; ++s) {
    stage_to_lds(in + (int64_t)(n0 + s) * iplane, ilds + (int64_t)s * iplane, iplane);
  }
  for (int64_t i = threadIdx.x; i < (int64_t)nvalid * dplane; i += kBlock) dclds[i] = 0.0f;
  __syncthreads();
  for (int64_t sc = threadIdx.x; sc < (int64_t)nvalid * cells; sc += kBlock) {
    const int s = sc / cells;
    const int cell = sc % cells;
    const int64_t pidx = (int64_t)(n0 + s) * cells + cell;
    const float pv = pooled[pidx];
    if (pv <= 0.0f) continue;
    const float g = dpooled[pidx];
    const int sub = argmax[pidx];
    const int px = cell % PW;
    const int py = (cell / PW) % PH;
    const int cco = cell / (PW * PH);
    const int yy = 2 * py + (sub >> 1);
    const int xx = 2 * px + (sub & 1);
    dclds[(int64_t)s * dplane + cco * H * W + yy * W + xx] = g;
  }
  __syncthreads();

  float acc[9] = {0, 0, 0, 0, 0, 0, 0, 0, 0};
  float accb = 0.0f;
  if (active) {
    for (int s = 0; s < nvalid; ++s) {
      const float* ip = ilds + (int64_t)s * iplane + ci * H * W;
      const float* dp = dclds + (int64_t)s * dplane + co * H * W;
      // iterate this co's conv positions, sliced across threads
      for (int yx = slice; yx < H * W; yx += nslices) {
        const float g = dp[yx];
        if (g == 0.0f) continue;
        const int yy = yx / W, xx = yx % W;
        if (ci == 0) accb += g;
#pragma unroll
        for (int ky = 0; ky < 3; ++ky) {
          const int iy = yy + ky - 1;
          if (iy < 0 || iy >= H) continue;
#pragma unroll
          for (int kx = 0; kx < 3; ++kx) {
            const int ix = xx + kx - 1;
            if (ix < 0 || ix >= W) continue;
            acc[ky * 3 + kx] = fmaf(g, ip[iy * W + ix], acc[ky * 3 + kx]);
          }
        }
      }
    }
  }

  // cross-slice reduction (skipped when nslices == 1)
  if (nslices > 1) {
#pragma unroll
    for (int k = 0; k < 9; ++k) {
      __syncthreads();
      redlds[threadIdx.x] = active ? acc[k] : 0.0f;
      __syncthreads();
      if (active && slice == 0) {
        float total = acc[k];
        for (int sl = 1; sl < nslices; ++sl) total += redlds[sl * pairs + pair];
        acc[k] = total;
      }
    }
    __syncthreads();
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
  const int lds_bytes = (CIN * H * W + COUT * CIN * 9) * (int)sizeof(float);
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
  const int lds_bytes = (COUT * H * W + COUT * CIN * 9) * (int)sizeof(float);
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
  const int iplane = CIN * H * W, dplane = COUT * H * W;
  // pick the largest chunk that fits a 64 KiB LDS budget (>= 2 blocks/CU
  // for latency hiding; kBlock floats of reduction scratch included)
  constexpr int kWeightLdsBudget = 64 * 1024;
  int samples = 1;
  while (samples < 16 &&
         ((int64_t)(samples * 2) * (iplane + dplane) + kBlock) * (int64_t)sizeof(float) <=
             kWeightLdsBudget)
    samples *= 2;
  const int lds_bytes = (samples * (iplane + dplane) + kBlock) * (int)sizeof(float);
  TORCH_CHECK(lds_bytes <= kMaxLds, "bwd_weight staging exceeds LDS");
  const int blocks = (N + samples - 1) / samples;
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(conv3x3_relu_pool_bwd_weight_kernel, dim3(blocks), dim3(kBlock), lds_bytes,
                     stream, dpooled.data_ptr<float>(), argmax.data_ptr<uint8_t>(),
                     pooled.data_ptr<float>(), in.data_ptr<float>(), dw.data_ptr<float>(),
                     db.defined() ? db.data_ptr<float>() : nullptr, N, CIN, COUT, H, W, samples);
}

} // namespace dmlamd
