// Fused LayerNorm (bf16 activations, fp32 statistics) for gfx950.
//
// GPT-2's profile (gpurun_out/profgpt2) shows torch's LayerNorm stack
// (vectorized_layer_norm + cuComputeGradInput + cuComputePartGradGammaBeta)
// at ~8% of the training step. These kernels process one ROW PER WAVE
// with short8 (16 B) vectorized loads, keep the row entirely in registers
// (one global read per tensor), reduce with wave shuffles, and accumulate
// dgamma/dbeta per-lane across the wave's rows with a single fp32
// atomicAdd per column per wave at the end.
//
// Layout: x [R, D] row-major bf16, D % 8 == 0, D <= 8 * 64 * kMaxVec.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_bf16.h>

#include "ops_common.h"

namespace dmlamd {

typedef short short8 __attribute__((ext_vector_type(8)));

constexpr int kMaxVec = 4; // up to 4 short8 per lane -> D <= 2048

__device__ __forceinline__ float bf2f(unsigned short b) {
  __hip_bfloat16 h;
  __builtin_memcpy(&h, &b, 2);
  return __bfloat162float(h);
}

__device__ __forceinline__ unsigned short f2bf(float f) {
  const __hip_bfloat16 h = __float2bfloat16(f);
  unsigned short b;
  __builtin_memcpy(&b, &h, 2);
  return b;
}

// --------------------------------------------------------------- forward

__global__ void __launch_bounds__(kBlock) layernorm_fwd_kernel(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ gamma,
    const __hip_bfloat16* __restrict__ beta, __hip_bfloat16* __restrict__ y,
    float* __restrict__ mean_out, float* __restrict__ rstd_out, int64_t R, int D, float eps) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
  const int waves_per_block = kBlock / kWave;
  const int nvec = D / 8;

  for (int64_t row = (int64_t)blockIdx.x * waves_per_block + wave; row < R;
       row += (int64_t)gridDim.x * waves_per_block) {
    const short8* xr = (const short8*)(x + row * D);
    unsigned short vals[kMaxVec][8];
    float sum = 0.0f, sumsq = 0.0f;
#pragma unroll
    for (int j = 0; j < kMaxVec; ++j) {
      const int idx = j * kWave + lane;
      if (idx < nvec) {
        *(short8*)vals[j] = xr[idx];
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          const float f = bf2f(vals[j][k]);
          sum += f;
          sumsq += f * f;
        }
      }
    }
    sum = wave_reduce<float, OP_SUM>(sum);
    sumsq = wave_reduce<float, OP_SUM>(sumsq);
    // broadcast from lane 0
    sum = __shfl(sum, 0, kWave);
    sumsq = __shfl(sumsq, 0, kWave);
    const float mean = sum / D;
    const float var = fmaxf(sumsq / D - mean * mean, 0.0f);
    const float rstd = __frsqrt_rn(var + eps);
    if (lane == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }

    short8* yr = (short8*)(y + row * D);
#pragma unroll
    for (int j = 0; j < kMaxVec; ++j) {
      const int idx = j * kWave + lane;
      if (idx < nvec) {
        unsigned short gv[8], bv[8], ov[8];
        *(short8*)gv = ((const short8*)gamma)[idx];
        *(short8*)bv = ((const short8*)beta)[idx];
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          const float xh = (bf2f(vals[j][k]) - mean) * rstd;
          ov[k] = f2bf(xh * bf2f(gv[k]) + bf2f(bv[k]));
        }
        yr[idx] = *(short8*)ov;
      }
    }
  }
}

// -------------------------------------------------------------- backward
//
// dgamma/dbeta strategy: each wave accumulates per-lane column partials
// in registers across ALL the rows it processes (grid is capped at
// kLnBwdBlocks so every wave owns many rows), then writes ONE fp32
// partial row per wave to a global workspace; a small second kernel
// reduces the <=1024 wave-partials per column and emits bf16 grads.
// (A first version did per-lane atomicAdds from 8192 waves: 12.6M
// atomics onto 2*D addresses serialized ~8000 deep — 1.7 ms/call, 40x
// slower than the forward. Deterministic partials fixed it.)

constexpr int kLnBwdBlocks = 256; // 1024 waves of dgamma/dbeta partials
__global__ void __launch_bounds__(kBlock) layernorm_bwd_kernel(
    const __hip_bfloat16* __restrict__ dy, const __hip_bfloat16* __restrict__ x,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    const __hip_bfloat16* __restrict__ gamma, __hip_bfloat16* __restrict__ dx,
    float* __restrict__ dgb_partials, int64_t R, int D) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
  const int waves_per_block = kBlock / kWave;
  const int nvec = D / 8;
  const float invD = 1.0f / D;

  float dgamma_acc[kMaxVec][8];
  float dbeta_acc[kMaxVec][8];
#pragma unroll
  for (int j = 0; j < kMaxVec; ++j)
#pragma unroll
    for (int k = 0; k < 8; ++k) dgamma_acc[j][k] = dbeta_acc[j][k] = 0.0f;

  for (int64_t row = (int64_t)blockIdx.x * waves_per_block + wave; row < R;
       row += (int64_t)gridDim.x * waves_per_block) {
    const short8* dyr = (const short8*)(dy + row * D);
    const short8* xr = (const short8*)(x + row * D);
    const float mean = mean_in[row];
    const float rstd = rstd_in[row];

    unsigned short dyv[kMaxVec][8], xv[kMaxVec][8], gv[kMaxVec][8];
    float s1 = 0.0f, s2 = 0.0f;
#pragma unroll
    for (int j = 0; j < kMaxVec; ++j) {
      const int idx = j * kWave + lane;
      if (idx < nvec) {
        *(short8*)dyv[j] = dyr[idx];
        *(short8*)xv[j] = xr[idx];
        *(short8*)gv[j] = ((const short8*)gamma)[idx];
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          const float d = bf2f(dyv[j][k]);
          const float xh = (bf2f(xv[j][k]) - mean) * rstd;
          const float a = d * bf2f(gv[j][k]);
          s1 += a;
          s2 += a * xh;
          dgamma_acc[j][k] += d * xh;
          dbeta_acc[j][k] += d;
        }
      }
    }
    s1 = wave_reduce<float, OP_SUM>(s1);
    s2 = wave_reduce<float, OP_SUM>(s2);
    s1 = __shfl(s1, 0, kWave) * invD;
    s2 = __shfl(s2, 0, kWave) * invD;

    short8* dxr = (short8*)(dx + row * D);
#pragma unroll
    for (int j = 0; j < kMaxVec; ++j) {
      const int idx = j * kWave + lane;
      if (idx < nvec) {
        unsigned short ov[8];
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          const float d = bf2f(dyv[j][k]);
          const float xh = (bf2f(xv[j][k]) - mean) * rstd;
          const float a = d * bf2f(gv[j][k]);
          ov[k] = f2bf(rstd * (a - s1 - xh * s2));
        }
        dxr[idx] = *(short8*)ov;
      }
    }
  }

  // cross-wave LDS reduction -> ONE fp32 partial row per BLOCK
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = (float*)smem; // [waves_per_block][2*D]
#pragma unroll
  for (int j = 0; j < kMaxVec; ++j) {
    const int idx = j * kWave + lane;
    if (idx < nvec) {
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        red[wave * 2 * D + idx * 8 + k] = dgamma_acc[j][k];
        red[wave * 2 * D + D + idx * 8 + k] = dbeta_acc[j][k];
      }
    }
  }
  __syncthreads();
  float* bp = dgb_partials + (int64_t)blockIdx.x * 2 * D;
  for (int i = threadIdx.x; i < 2 * D; i += kBlock) {
    float total = 0.0f;
#pragma unroll
    for (int w = 0; w < kBlock / kWave; ++w) total += red[w * 2 * D + i];
    bp[i] = total;
  }
}

// Reduce the per-block partial rows per column: one block per column,
// threads over partial rows, wave+LDS tree, one bf16 store.
__global__ void __launch_bounds__(kBlock) dgb_reduce_kernel(
    const float* __restrict__ dgb_partials, __hip_bfloat16* __restrict__ dgamma,
    __hip_bfloat16* __restrict__ dbeta, int D, int nrows) {
  const int col = blockIdx.x; // [0, 2*D)
  float local = 0.0f;
  for (int r = threadIdx.x; r < nrows; r += kBlock) {
    local += dgb_partials[(int64_t)r * 2 * D + col];
  }
  const float total = block_reduce<float, OP_SUM>(local);
  if (threadIdx.x == 0) {
    if (col < D) {
      dgamma[col] = __float2bfloat16(total);
    } else {
      dbeta[col - D] = __float2bfloat16(total);
    }
  }
}

// ------------------------------------------------------------ launchers

void layernorm_fwd(at::Tensor x, at::Tensor gamma, at::Tensor beta, at::Tensor y,
                   at::Tensor mean, at::Tensor rstd, double eps) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.is_contiguous(),
              "x must be contiguous bf16 on device");
  const int D = x.size(-1);
  const int64_t R = x.numel() / D;
  TORCH_CHECK(D % 8 == 0 && D <= 8 * kWave * kMaxVec, "D must be a multiple of 8 and <= 2048");
  TORCH_CHECK(gamma.numel() == D && beta.numel() == D &&
                  gamma.scalar_type() == at::kBFloat16 && beta.scalar_type() == at::kBFloat16,
              "gamma/beta must be bf16[D]");
  TORCH_CHECK(y.numel() == x.numel() && y.scalar_type() == at::kBFloat16 && y.is_contiguous(),
              "y must be contiguous bf16 like x");
  TORCH_CHECK(mean.numel() >= R && rstd.numel() >= R && mean.scalar_type() == at::kFloat &&
                  rstd.scalar_type() == at::kFloat,
              "mean/rstd must be fp32[R]");
  auto stream = c10::hip::getCurrentHIPStream();
  const int waves_per_block = kBlock / kWave;
  const int blocks = (int)std::min<int64_t>((R + waves_per_block - 1) / waves_per_block, kMaxGrid);
  hipLaunchKernelGGL(layernorm_fwd_kernel, dim3(blocks), dim3(kBlock), 0, stream,
                     (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)gamma.data_ptr(),
                     (const __hip_bfloat16*)beta.data_ptr(), (__hip_bfloat16*)y.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), R, D, (float)eps);
}

// dgb_ws: fp32 workspace of at least ln_bwd_partials_size(R, D) elements.
void layernorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor mean, at::Tensor rstd,
                   at::Tensor gamma, at::Tensor dx, at::Tensor dgb_ws, at::Tensor dgamma,
                   at::Tensor dbeta) {
  const int D = x.size(-1);
  const int64_t R = x.numel() / D;
  TORCH_CHECK(dy.numel() == x.numel() && dy.is_contiguous() && dx.numel() == x.numel(),
              "dy/dx must match x");
  TORCH_CHECK(mean.numel() >= R && rstd.numel() >= R, "mean/rstd must be fp32[R]");
  TORCH_CHECK(gamma.numel() == D && dgamma.numel() == D && dbeta.numel() == D,
              "gamma/dgamma/dbeta must be [D]");
  auto stream = c10::hip::getCurrentHIPStream();
  const int waves_per_block = kBlock / kWave;
  const int blocks =
      (int)std::min<int64_t>((R + waves_per_block - 1) / waves_per_block, kLnBwdBlocks);
  TORCH_CHECK(dgb_ws.numel() >= (int64_t)blocks * 2 * D && dgb_ws.scalar_type() == at::kFloat,
              "dgb workspace too small");
  const int lds_bytes = waves_per_block * 2 * D * (int)sizeof(float);
  TORCH_CHECK(lds_bytes <= 160 * 1024, "D too large for the cross-wave reduction");
  // no memset needed: every launched wave (idle ones included) contributes
  // zeros, and every block writes its full partial row
  hipLaunchKernelGGL(layernorm_bwd_kernel, dim3(blocks), dim3(kBlock), lds_bytes, stream,
                     (const __hip_bfloat16*)dy.data_ptr(), (const __hip_bfloat16*)x.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     (const __hip_bfloat16*)gamma.data_ptr(), (__hip_bfloat16*)dx.data_ptr(),
                     dgb_ws.data_ptr<float>(), R, D);
  hipLaunchKernelGGL(dgb_reduce_kernel, dim3(2 * D), dim3(kBlock), 0, stream,
                     dgb_ws.data_ptr<float>(), (__hip_bfloat16*)dgamma.data_ptr(),
                     (__hip_bfloat16*)dbeta.data_ptr(), D, blocks);
}

} // namespace dmlamd
