// Fused conv3x3(pad1) + ReLU + maxpool2x2 kernels for gfx950 (MI355X).
//
// Motivation (profiles/mnist_b1024_eager_kernel_stats.md): MIOpen executes
// the benchmark MNIST-CNN as im2col + hundreds of small batched GEMMs per
// step plus a pathological find phase — the step is kernel-soup-bound. The
// model's math is trivial (< 1 GFLOP/step) and its tensors are small, so
// the right MI355X design is one fused kernel per layer: each thread
// computes one POOLED output (4 conv results + ReLU + max) directly from
// the input window, weights staged in LDS, with the pool argmax recorded
// for an exact backward. Memory traffic is the theoretical minimum (read
// input once, write pooled + argmax once).
//
// Backward splits into:
//   bwd_data:   din = dconv (*) flipped-W, where dconv is reconstructed
//               on the fly from (dpooled, argmax, pooled>0) — the
//               relu+maxpool gradient is folded in for free, nothing is
//               materialized.
//   bwd_weight: two-level reduction (registers -> LDS tree -> one
//               atomicAdd per block) over (batch x pooled cells) per
//               (cout, cin) pair; bias gradient fused into the cin==0
//               blocks.
//
// ReLU/maxpool tie-breaking matches torch: first index wins ties, and a
// pooled value of exactly 0 (all-negative window) propagates no gradient.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "ops_common.h"

namespace dmlamd {

// --------------------------------------------------------------- forward

template <int CIN_T>
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_fwd_kernel(
    const float* __restrict__ in, const float* __restrict__ w, const float* __restrict__ bias,
    float* __restrict__ out, uint8_t* __restrict__ argmax, int N, int CIN, int COUT, int H,
    int W) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* wlds = (float*)smem; // [COUT][CIN][9]
  const int cin = CIN_T > 0 ? CIN_T : CIN;
  const int wcount = COUT * cin * 9;
  for (int i = threadIdx.x; i < wcount; i += kBlock) wlds[i] = w[i];
  __syncthreads();

  const int PH = H / 2, PW = W / 2;
  const int64_t total = (int64_t)N * COUT * PH * PW;
  const int64_t stride = (int64_t)gridDim.x * kBlock;

  for (int64_t idx = (int64_t)blockIdx.x * kBlock + threadIdx.x; idx < total; idx += stride) {
    int px = idx % PW;
    int py = (idx / PW) % PH;
    int co = (idx / ((int64_t)PW * PH)) % COUT;
    int n = idx / ((int64_t)PW * PH * COUT);

    const int y0 = 2 * py, x0 = 2 * px;
    float acc0 = bias[co], acc1 = acc0, acc2 = acc0, acc3 = acc0;
    const float* wc = wlds + co * cin * 9;

#pragma unroll 4
    for (int ci = 0; ci < cin; ++ci) {
      const float* inp = in + (((int64_t)n * cin + ci) * H) * W;
      float win[4][4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int yy = y0 - 1 + r;
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          const int xx = x0 - 1 + c;
          win[r][c] = (yy >= 0 && yy < H && xx >= 0 && xx < W) ? inp[yy * W + xx] : 0.0f;
        }
      }
      const float* wk = wc + ci * 9;
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

    // relu + 2x2 max with first-index tie-break (torch semantics)
    float r0 = fmaxf(acc0, 0.0f), r1 = fmaxf(acc1, 0.0f);
    float r2 = fmaxf(acc2, 0.0f), r3 = fmaxf(acc3, 0.0f);
    float m = r0;
    int arg = 0;
    if (r1 > m) { m = r1; arg = 1; }
    if (r2 > m) { m = r2; arg = 2; }
    if (r3 > m) { m = r3; arg = 3; }
    out[idx] = m;
    argmax[idx] = (uint8_t)arg;
  }
}

// ------------------------------------------------------------- bwd data

// dconv(n,co,yy,xx) reconstructed from the pooled tensors: nonzero only at
// the argmax position of its 2x2 cell and only if the pooled max was > 0.
__device__ __forceinline__ float dconv_at(const float* __restrict__ dpooled,
                                          const uint8_t* __restrict__ argmax,
                                          const float* __restrict__ pooled, int64_t plane_off,
                                          int yy, int xx, int PW) {
  const int py = yy >> 1, px = xx >> 1;
  const int64_t pidx = plane_off + py * PW + px;
  const int sub = ((yy & 1) << 1) | (xx & 1);
  if (argmax[pidx] != sub) return 0.0f;
  if (pooled[pidx] <= 0.0f) return 0.0f;
  return dpooled[pidx];
}

template <int COUT_T>
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_bwd_data_kernel(
    const float* __restrict__ dpooled, const uint8_t* __restrict__ argmax,
    const float* __restrict__ pooled, const float* __restrict__ w, float* __restrict__ din,
    int N, int CIN, int COUT, int H, int W) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* wlds = (float*)smem; // [COUT][CIN][9]
  const int cout = COUT_T > 0 ? COUT_T : COUT;
  const int wcount = cout * CIN * 9;
  for (int i = threadIdx.x; i < wcount; i += kBlock) wlds[i] = w[i];
  __syncthreads();

  const int PH = H / 2, PW = W / 2;
  const int64_t total = (int64_t)N * CIN * H * W;
  const int64_t stride = (int64_t)gridDim.x * kBlock;

  for (int64_t idx = (int64_t)blockIdx.x * kBlock + threadIdx.x; idx < total; idx += stride) {
    int x = idx % W;
    int y = (idx / W) % H;
    int ci = (idx / ((int64_t)W * H)) % CIN;
    int n = idx / ((int64_t)W * H * CIN);

    float acc = 0.0f;
    for (int co = 0; co < cout; ++co) {
      const int64_t plane_off = ((int64_t)n * cout + co) * PH * PW;
      const float* wk = wlds + (co * CIN + ci) * 9;
#pragma unroll
      for (int ky = 0; ky < 3; ++ky) {
        const int yy = y - ky + 1;
        if (yy < 0 || yy >= H) continue;
#pragma unroll
        for (int kx = 0; kx < 3; ++kx) {
          const int xx = x - kx + 1;
          if (xx < 0 || xx >= W) continue;
          const float d = dconv_at(dpooled, argmax, pooled, plane_off, yy, xx, PW);
          acc = fmaf(wk[ky * 3 + kx], d, acc);
        }
      }
    }
    din[idx] = acc;
  }
}

// ----------------------------------------------------------- bwd weight

// Grid: (COUT*CIN) pairs x NCHUNK slices of the (n, py, px) cell space.
// Each thread keeps 9 dW partials (+1 db partial in ci==0 blocks) in
// registers, reduces through LDS, then ONE atomicAdd per output per block.
__global__ void __launch_bounds__(kBlock) conv3x3_relu_pool_bwd_weight_kernel(
    const float* __restrict__ dpooled, const uint8_t* __restrict__ argmax,
    const float* __restrict__ pooled, const float* __restrict__ in, float* __restrict__ dw,
    float* __restrict__ db, int N, int CIN, int COUT, int H, int W, int nchunk) {
  const int PH = H / 2, PW = W / 2;
  const int pair = blockIdx.x % (COUT * CIN);
  const int chunk = blockIdx.x / (COUT * CIN);
  const int co = pair / CIN;
  const int ci = pair % CIN;

  const int64_t cells = (int64_t)N * PH * PW;
  const int64_t per_chunk = (cells + nchunk - 1) / nchunk;
  const int64_t begin = chunk * per_chunk;
  const int64_t end = min(begin + per_chunk, cells);

  float acc[9] = {0, 0, 0, 0, 0, 0, 0, 0, 0};
  float accb = 0.0f;

  for (int64_t cell = begin + threadIdx.x; cell < end; cell += kBlock) {
    const int px = cell % PW;
    const int py = (cell / PW) % PH;
    const int n = cell / ((int64_t)PW * PH);
    const int64_t pidx = ((int64_t)n * COUT + co) * PH * PW + py * PW + px;
    const float pv = pooled[pidx];
    if (pv <= 0.0f) continue;
    const float g = dpooled[pidx];
    if (g == 0.0f) continue;
    const int sub = argmax[pidx];
    const int yy = 2 * py + (sub >> 1);
    const int xx = 2 * px + (sub & 1);
    accb += g;
    const float* inp = in + ((int64_t)n * CIN + ci) * H * W;
#pragma unroll
    for (int ky = 0; ky < 3; ++ky) {
      const int iy = yy + ky - 1;
      if (iy < 0 || iy >= H) continue;
#pragma unroll
      for (int kx = 0; kx < 3; ++kx) {
        const int ix = xx + kx - 1;
        if (ix < 0 || ix >= W) continue;
        acc[ky * 3 + kx] = fmaf(g, inp[iy * W + ix], acc[ky * 3 + kx]);
      }
    }
  }

  // block reduction of the 9 (+1) partials, one value at a time
  __shared__ float lds[kBlock / kWave];
  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
#pragma unroll
  for (int k = 0; k < 9; ++k) {
    float v = wave_reduce<float, OP_SUM>(acc[k]);
    if (lane == 0) lds[wave] = v;
    __syncthreads();
    if (threadIdx.x == 0) {
      float total = lds[0];
#pragma unroll
      for (int ww = 1; ww < kBlock / kWave; ++ww) total += lds[ww];
      atomicAdd(&dw[(co * CIN + ci) * 9 + k], total);
    }
    __syncthreads();
  }
  if (ci == 0 && db != nullptr) {
    float v = wave_reduce<float, OP_SUM>(accb);
    if (lane == 0) lds[wave] = v;
    __syncthreads();
    if (threadIdx.x == 0) {
      float total = lds[0];
#pragma unroll
      for (int ww = 1; ww < kBlock / kWave; ++ww) total += lds[ww];
      atomicAdd(&db[co], total);
    }
  }
}

// ------------------------------------------------------------ launchers

static void check_fwd_args(const at::Tensor& in, const at::Tensor& w, const at::Tensor& b) {
  TORCH_CHECK(in.is_cuda() && in.scalar_type() == at::kFloat && in.is_contiguous(),
              "input must be contiguous fp32 on device");
  TORCH_CHECK(w.is_contiguous() && b.is_contiguous(), "weights must be contiguous");
  TORCH_CHECK(w.size(2) == 3 && w.size(3) == 3, "kernel must be 3x3");
}

void conv3x3_relu_pool_fwd(at::Tensor in, at::Tensor w, at::Tensor b, at::Tensor out,
                           at::Tensor argmax) {
  check_fwd_args(in, w, b);
  const int N = in.size(0), CIN = in.size(1), H = in.size(2), W = in.size(3);
  const int COUT = w.size(0);
  TORCH_CHECK(H % 2 == 0 && W % 2 == 0, "H and W must be even for 2x2 pooling");
  const int64_t total = (int64_t)N * COUT * (H / 2) * (W / 2);
  const int lds_bytes = COUT * CIN * 9 * sizeof(float);
  TORCH_CHECK(lds_bytes <= 160 * 1024, "weights exceed LDS");
  auto stream = c10::hip::getCurrentHIPStream();
  const int blocks = grid_for(total, kBlock);

  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(kBlock), lds_bytes, stream,
                       in.data_ptr<float>(), w.data_ptr<float>(), b.data_ptr<float>(),
                       out.data_ptr<float>(), argmax.data_ptr<uint8_t>(), N, CIN, COUT, H, W);
  };
  if (CIN == 1)
    launch(conv3x3_relu_pool_fwd_kernel<1>);
  else if (CIN == 16)
    launch(conv3x3_relu_pool_fwd_kernel<16>);
  else if (CIN == 32)
    launch(conv3x3_relu_pool_fwd_kernel<32>);
  else
    launch(conv3x3_relu_pool_fwd_kernel<0>);
}

void conv3x3_relu_pool_bwd_data(at::Tensor dpooled, at::Tensor argmax, at::Tensor pooled,
                                at::Tensor w, at::Tensor din) {
  const int N = din.size(0), CIN = din.size(1), H = din.size(2), W = din.size(3);
  const int COUT = w.size(0);
  const int64_t total = (int64_t)N * CIN * H * W;
  const int lds_bytes = COUT * CIN * 9 * sizeof(float);
  auto stream = c10::hip::getCurrentHIPStream();
  const int blocks = grid_for(total, kBlock);

  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(kBlock), lds_bytes, stream,
                       dpooled.data_ptr<float>(), argmax.data_ptr<uint8_t>(),
                       pooled.data_ptr<float>(), w.data_ptr<float>(), din.data_ptr<float>(), N,
                       CIN, COUT, H, W);
  };
  if (COUT == 16)
    launch(conv3x3_relu_pool_bwd_data_kernel<16>);
  else if (COUT == 32)
    launch(conv3x3_relu_pool_bwd_data_kernel<32>);
  else
    launch(conv3x3_relu_pool_bwd_data_kernel<0>);
}

void conv3x3_relu_pool_bwd_weight(at::Tensor dpooled, at::Tensor argmax, at::Tensor pooled,
                                  at::Tensor in, at::Tensor dw, at::Tensor db) {
  const int N = in.size(0), CIN = in.size(1), H = in.size(2), W = in.size(3);
  const int COUT = dw.size(0);
  const int PH = H / 2, PW = W / 2;
  const int64_t cells = (int64_t)N * PH * PW;
  // aim for >= 2048 blocks total with >= ~4k cells per block
  int nchunk = (int)std::min<int64_t>((cells + 4095) / 4096, std::max<int64_t>(1, kMaxGrid / (COUT * CIN)));
  if (nchunk < 1) nchunk = 1;
  const int blocks = COUT * CIN * nchunk;
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(conv3x3_relu_pool_bwd_weight_kernel, dim3(blocks), dim3(kBlock), 0, stream,
                     dpooled.data_ptr<float>(), argmax.data_ptr<uint8_t>(),
                     pooled.data_ptr<float>(), in.data_ptr<float>(), dw.data_ptr<float>(),
                     db.defined() ? db.data_ptr<float>() : nullptr, N, CIN, COUT, H, W, nchunk);
}

} // namespace dmlamd
