// Fused flat-buffer optimizers and gradient clipping for gfx950.
//
// The reference steps each optimizer param-by-param through the stock
// torch optimizer (reference dmlcloud/stage.py:287-288) and clips per
// param-group with torch.nn.utils.clip_grad_norm_ (stage.py:276-279).
// Here the whole model lives in ONE flat fp32 buffer (parallel/flat.py),
// so the optimizer update is a single vectorized elementwise kernel and
// gradient clipping is a deterministic two-stage L2 norm plus one scale
// kernel — all graph-capturable (the step counter and the clip scale live
// in device memory, so hipGraph replay stays correct).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_bf16.h>

#include "ops_common.h"

namespace dmlamd {

__global__ void increment_step_kernel(int32_t* step) {
  if (threadIdx.x == 0 && blockIdx.x == 0) step[0] += 1;
}

// ------------------------------------------------------------------- Adam
// Vectorized float4 grid-stride over the flat parameter buffer.
__global__ void __launch_bounds__(kBlock) fused_adam_kernel(
    float* __restrict__ param, const float* __restrict__ grad, float* __restrict__ exp_avg,
    float* __restrict__ exp_avg_sq, const int32_t* __restrict__ step_t, int64_t n, float lr,
    float beta1, float beta2, float eps, float weight_decay, float grad_scale) {
  const float t = (float)step_t[0];
  const float bc1 = 1.0f - __powf(beta1, t);
  const float bc2 = 1.0f - __powf(beta2, t);
  const float inv_bc1 = 1.0f / bc1;

  const int64_t nvec = n / 4;
  const int64_t stride = (int64_t)gridDim.x * kBlock;
  float4* p4 = (float4*)param;
  const float4* g4 = (const float4*)grad;
  float4* m4 = (float4*)exp_avg;
  float4* v4 = (float4*)exp_avg_sq;

  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < nvec; i += stride) {
    float4 p = p4[i], g = g4[i], m = m4[i], v = v4[i];
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float* pp = (&p.x) + k;
      float* gg = (&g.x) + k;
      float* mm = (&m.x) + k;
      float* vv = (&v.x) + k;
      float gi = *gg * grad_scale + weight_decay * *pp;
      float mi = beta1 * *mm + (1.0f - beta1) * gi;
      float vi = beta2 * *vv + (1.0f - beta2) * gi * gi;
      *mm = mi;
      *vv = vi;
      *pp -= lr * (mi * inv_bc1) / (__builtin_sqrtf(vi / bc2) + eps);
    }
    p4[i] = p;
    m4[i] = m;
    v4[i] = v;
  }
  // scalar tail
  for (int64_t i = nvec * 4 + (int64_t)blockIdx.x * kBlock + threadIdx.x; i < n; i += stride) {
    float gi = grad[i] * grad_scale + weight_decay * param[i];
    float mi = beta1 * exp_avg[i] + (1.0f - beta1) * gi;
    float vi = beta2 * exp_avg_sq[i] + (1.0f - beta2) * gi * gi;
    exp_avg[i] = mi;
    exp_avg_sq[i] = vi;
    param[i] -= lr * (mi * inv_bc1) / (__builtin_sqrtf(vi / bc2) + eps);
  }
}

void fused_adam(at::Tensor param, at::Tensor grad, at::Tensor exp_avg, at::Tensor exp_avg_sq,
                at::Tensor step_t, double lr, double beta1, double beta2, double eps,
                double weight_decay, double grad_scale) {
  TORCH_CHECK(param.is_cuda() && grad.is_cuda(), "device tensors required");
  TORCH_CHECK(param.scalar_type() == at::kFloat && grad.scalar_type() == at::kFloat,
              "flat Adam operates on fp32 buffers");
  TORCH_CHECK(step_t.scalar_type() == at::kInt, "step counter must be int32");
  const int64_t n = param.numel();
  TORCH_CHECK(grad.numel() == n && exp_avg.numel() == n && exp_avg_sq.numel() == n,
              "buffer size mismatch");
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(increment_step_kernel, dim3(1), dim3(64), 0, stream,
                     step_t.data_ptr<int32_t>());
  int blocks = grid_for(n / 4 + 1, kBlock);
  hipLaunchKernelGGL(fused_adam_kernel, dim3(blocks), dim3(kBlock), 0, stream,
                     param.data_ptr<float>(), grad.data_ptr<float>(), exp_avg.data_ptr<float>(),
                     exp_avg_sq.data_ptr<float>(), step_t.data_ptr<int32_t>(), n, (float)lr,
                     (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                     (float)grad_scale);
}

// -------------------------------------------------- Adam, bf16 mixed-precision
// True mixed precision for the flat replica: parameters and gradients live
// in bf16 HBM buffers (model computes in bf16, no autocast casting per
// layer), the optimizer keeps the fp32 master + moments. One kernel: read
// bf16 grad, update fp32 master/moments, write back bf16 params.
__global__ void __launch_bounds__(kBlock) fused_adam_bf16_kernel(
    __hip_bfloat16* __restrict__ param, const __hip_bfloat16* __restrict__ grad,
    float* __restrict__ master, float* __restrict__ exp_avg, float* __restrict__ exp_avg_sq,
    const int32_t* __restrict__ step_t, int64_t n, float lr, float beta1, float beta2,
    float eps, float weight_decay, float grad_scale) {
  const float t = (float)step_t[0];
  const float bc1 = 1.0f - __powf(beta1, t);
  const float bc2 = 1.0f - __powf(beta2, t);
  const float inv_bc1 = 1.0f / bc1;

  const int64_t nvec = n / 8; // 8 bf16 = 16B
  const int64_t stride = (int64_t)gridDim.x * kBlock;
  typedef short short8 __attribute__((ext_vector_type(8)));
  typedef float float4v __attribute__((ext_vector_type(4)));
  short8* p8 = (short8*)param;
  const short8* g8 = (const short8*)grad;

  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < nvec; i += stride) {
    unsigned short gv[8], pv[8];
    float m[8], v[8], w[8];
    *(short8*)gv = g8[i];
    *(float4v*)&m[0] = ((float4v*)exp_avg)[2 * i];
    *(float4v*)&m[4] = ((float4v*)exp_avg)[2 * i + 1];
    *(float4v*)&v[0] = ((float4v*)exp_avg_sq)[2 * i];
    *(float4v*)&v[4] = ((float4v*)exp_avg_sq)[2 * i + 1];
    *(float4v*)&w[0] = ((float4v*)master)[2 * i];
    *(float4v*)&w[4] = ((float4v*)master)[2 * i + 1];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      __hip_bfloat16 hg;
      __builtin_memcpy(&hg, &gv[k], 2);
      const float gi = __bfloat162float(hg) * grad_scale + weight_decay * w[k];
      const float mi = beta1 * m[k] + (1.0f - beta1) * gi;
      const float vi = beta2 * v[k] + (1.0f - beta2) * gi * gi;
      m[k] = mi;
      v[k] = vi;
      w[k] -= lr * (mi * inv_bc1) / (__builtin_sqrtf(vi / bc2) + eps);
      const __hip_bfloat16 h = __float2bfloat16(w[k]);
      __builtin_memcpy(&pv[k], &h, 2);
    }
    ((float4v*)exp_avg)[2 * i] = *(float4v*)&m[0];
    ((float4v*)exp_avg)[2 * i + 1] = *(float4v*)&m[4];
    ((float4v*)exp_avg_sq)[2 * i] = *(float4v*)&v[0];
    ((float4v*)exp_avg_sq)[2 * i + 1] = *(float4v*)&v[4];
    ((float4v*)master)[2 * i] = *(float4v*)&w[0];
    ((float4v*)master)[2 * i + 1] = *(float4v*)&w[4];
    p8[i] = *(short8*)pv;
  }
  // scalar tail
  for (int64_t i = nvec * 8 + (int64_t)blockIdx.x * kBlock + threadIdx.x; i < n; i += stride) {
    const float gi = __bfloat162float(grad[i]) * grad_scale + weight_decay * master[i];
    const float mi = beta1 * exp_avg[i] + (1.0f - beta1) * gi;
    const float vi = beta2 * exp_avg_sq[i] + (1.0f - beta2) * gi * gi;
    exp_avg[i] = mi;
    exp_avg_sq[i] = vi;
    master[i] -= lr * (mi * inv_bc1) / (__builtin_sqrtf(vi / bc2) + eps);
    param[i] = __float2bfloat16(master[i]);
  }
}

void fused_adam_bf16(at::Tensor param, at::Tensor grad, at::Tensor master, at::Tensor exp_avg,
                     at::Tensor exp_avg_sq, at::Tensor step_t, double lr, double beta1,
                     double beta2, double eps, double weight_decay, double grad_scale) {
  TORCH_CHECK(param.is_cuda() && param.scalar_type() == at::kBFloat16, "param must be bf16");
  TORCH_CHECK(grad.scalar_type() == at::kBFloat16, "grad must be bf16");
  TORCH_CHECK(master.scalar_type() == at::kFloat, "master must be fp32");
  const int64_t n = param.numel();
  TORCH_CHECK(grad.numel() == n && master.numel() == n && exp_avg.numel() == n &&
                  exp_avg_sq.numel() == n,
              "buffer size mismatch");
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(increment_step_kernel, dim3(1), dim3(64), 0, stream,
                     step_t.data_ptr<int32_t>());
  int blocks = grid_for(n / 8 + 1, kBlock);
  hipLaunchKernelGGL(fused_adam_bf16_kernel, dim3(blocks), dim3(kBlock), 0, stream,
                     (__hip_bfloat16*)param.data_ptr(), (const __hip_bfloat16*)grad.data_ptr(),
                     master.data_ptr<float>(), exp_avg.data_ptr<float>(),
                     exp_avg_sq.data_ptr<float>(), step_t.data_ptr<int32_t>(), n, (float)lr,
                     (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                     (float)grad_scale);
}

// -------------------------------------------------------------------- SGD
__global__ void __launch_bounds__(kBlock) fused_sgd_kernel(
    float* __restrict__ param, const float* __restrict__ grad, float* __restrict__ momentum_buf,
    int64_t n, float lr, float momentum, float weight_decay, float grad_scale,
    bool use_momentum) {
  const int64_t nvec = n / 4;
  const int64_t stride = (int64_t)gridDim.x * kBlock;
  float4* p4 = (float4*)param;
  const float4* g4 = (const float4*)grad;
  float4* m4 = (float4*)momentum_buf;

  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < nvec; i += stride) {
    float4 p = p4[i], g = g4[i];
    float4 m = use_momentum ? m4[i] : make_float4(0, 0, 0, 0);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float* pp = (&p.x) + k;
      float* gg = (&g.x) + k;
      float* mm = (&m.x) + k;
      float gi = *gg * grad_scale + weight_decay * *pp;
      if (use_momentum) {
        float mi = momentum * *mm + gi;
        *mm = mi;
        gi = mi;
      }
      *pp -= lr * gi;
    }
    p4[i] = p;
    if (use_momentum) m4[i] = m;
  }
  for (int64_t i = nvec * 4 + (int64_t)blockIdx.x * kBlock + threadIdx.x; i < n; i += stride) {
    float gi = grad[i] * grad_scale + weight_decay * param[i];
    if (use_momentum) {
      float mi = momentum * momentum_buf[i] + gi;
      momentum_buf[i] = mi;
      gi = mi;
    }
    param[i] -= lr * gi;
  }
}

void fused_sgd(at::Tensor param, at::Tensor grad, at::Tensor momentum_buf, double lr,
               double momentum, double weight_decay, double grad_scale, bool use_momentum) {
  TORCH_CHECK(param.is_cuda() && grad.is_cuda(), "device tensors required");
  TORCH_CHECK(param.scalar_type() == at::kFloat, "flat SGD operates on fp32 buffers");
  const int64_t n = param.numel();
  auto stream = c10::hip::getCurrentHIPStream();
  int blocks = grid_for(n / 4 + 1, kBlock);
  hipLaunchKernelGGL(fused_sgd_kernel, dim3(blocks), dim3(kBlock), 0, stream,
                     param.data_ptr<float>(), grad.data_ptr<float>(),
                     momentum_buf.defined() ? momentum_buf.data_ptr<float>() : nullptr, n,
                     (float)lr, (float)momentum, (float)weight_decay, (float)grad_scale,
                     use_momentum);
}

// ---------------------------------------------------- SGD, bf16 mixed-precision
__global__ void __launch_bounds__(kBlock) fused_sgd_bf16_kernel(
    __hip_bfloat16* __restrict__ param, const __hip_bfloat16* __restrict__ grad,
    float* __restrict__ master, float* __restrict__ momentum_buf, int64_t n, float lr,
    float momentum, float weight_decay, float grad_scale, bool use_momentum) {
  const int64_t stride = (int64_t)gridDim.x * kBlock;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < n; i += stride) {
    float gi = __bfloat162float(grad[i]) * grad_scale + weight_decay * master[i];
    if (use_momentum) {
      const float mi = momentum * momentum_buf[i] + gi;
      momentum_buf[i] = mi;
      gi = mi;
    }
    master[i] -= lr * gi;
    param[i] = __float2bfloat16(master[i]);
  }
}

void fused_sgd_bf16(at::Tensor param, at::Tensor grad, at::Tensor master,
                    at::Tensor momentum_buf, double lr, double momentum, double weight_decay,
                    double grad_scale, bool use_momentum) {
  TORCH_CHECK(param.is_cuda() && param.scalar_type() == at::kBFloat16, "param must be bf16");
  TORCH_CHECK(master.scalar_type() == at::kFloat, "master must be fp32");
  const int64_t n = param.numel();
  auto stream = c10::hip::getCurrentHIPStream();
  int blocks = grid_for(n, kBlock * 4);
  hipLaunchKernelGGL(fused_sgd_bf16_kernel, dim3(blocks), dim3(kBlock), 0, stream,
                     (__hip_bfloat16*)param.data_ptr(), (const __hip_bfloat16*)grad.data_ptr(),
                     master.data_ptr<float>(),
                     momentum_buf.defined() ? momentum_buf.data_ptr<float>() : nullptr, n,
                     (float)lr, (float)momentum, (float)weight_decay, (float)grad_scale,
                     use_momentum);
}

// ------------------------------------------------------- L2 norm and clip
__global__ void __launch_bounds__(kBlock) sqnorm_partials_kernel(
    const float* __restrict__ x, int64_t n, double* __restrict__ partials) {
  double local = 0.0;
  const int64_t nvec = n / 4;
  const int64_t stride = (int64_t)gridDim.x * kBlock;
  const float4* x4 = (const float4*)x;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < nvec; i += stride) {
    float4 v = x4[i];
    local += (double)v.x * v.x + (double)v.y * v.y + (double)v.z * v.z + (double)v.w * v.w;
  }
  for (int64_t i = nvec * 4 + (int64_t)blockIdx.x * kBlock + threadIdx.x; i < n; i += stride) {
    local += (double)x[i] * x[i];
  }
  double total = block_reduce<double, OP_SUM>(local);
  if (threadIdx.x == 0) partials[blockIdx.x] = total;
}

typedef short short8v __attribute__((ext_vector_type(8)));

__global__ void __launch_bounds__(kBlock) sqnorm_partials_bf16_kernel(
    const __hip_bfloat16* __restrict__ x, int64_t n, double* __restrict__ partials) {
  double local = 0.0;
  const int64_t nvec = n / 8;
  const int64_t stride = (int64_t)gridDim.x * kBlock;
  const short8v* x8 = (const short8v*)x;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < nvec; i += stride) {
    unsigned short v[8];
    *(short8v*)v = x8[i];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      __hip_bfloat16 h;
      __builtin_memcpy(&h, &v[k], 2);
      const float f = __bfloat162float(h);
      local += (double)f * f;
    }
  }
  for (int64_t i = nvec * 8 + (int64_t)blockIdx.x * kBlock + threadIdx.x; i < n; i += stride) {
    const float f = __bfloat162float(x[i]);
    local += (double)f * f;
  }
  double total = block_reduce<double, OP_SUM>(local);
  if (threadIdx.x == 0) partials[blockIdx.x] = total;
}

__global__ void __launch_bounds__(kBlock) scale_bf16_by_device_scalar_kernel(
    __hip_bfloat16* __restrict__ x, int64_t n, const float* __restrict__ scale_ptr) {
  const float s = scale_ptr[0];
  const int64_t stride = (int64_t)gridDim.x * kBlock;
  const int64_t nvec = n / 8;
  short8v* x8 = (short8v*)x;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < nvec; i += stride) {
    unsigned short v[8];
    *(short8v*)v = x8[i];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      __hip_bfloat16 h;
      __builtin_memcpy(&h, &v[k], 2);
      const __hip_bfloat16 r = __float2bfloat16(__bfloat162float(h) * s);
      __builtin_memcpy(&v[k], &r, 2);
    }
    x8[i] = *(short8v*)v;
  }
  for (int64_t i = nvec * 8 + (int64_t)blockIdx.x * kBlock + threadIdx.x; i < n; i += stride) {
    x[i] = __float2bfloat16(__bfloat162float(x[i]) * s);
  }
}

// Writes norm to out[0] and the clip scale min(1, max_norm/(norm+1e-6)) to
// out[1] (fp32). max_norm < 0 disables the scale computation (norm only).
// norm_scale pre-multiplies the computed norm: with a flat gradient buffer
// that holds the SUM over world ranks, norm_scale = 1/world makes the clip
// act on the AVERAGED gradient norm — matching torch DDP +
// clip_grad_norm_ semantics regardless of world size.
__global__ void __launch_bounds__(kBlock) norm_finalize_kernel(
    const double* __restrict__ partials, int nblocks, float* __restrict__ out, float max_norm,
    float norm_scale) {
  double local = 0.0;
  for (int i = threadIdx.x; i < nblocks; i += kBlock) local += partials[i];
  double total = block_reduce<double, OP_SUM>(local);
  if (threadIdx.x == 0) {
    float norm = (float)__builtin_sqrt(total) * norm_scale;
    out[0] = norm;
    if (max_norm >= 0.0f) {
      float scale = max_norm / (norm + 1e-6f);
      out[1] = scale < 1.0f ? scale : 1.0f;
    }
  }
}

__global__ void __launch_bounds__(kBlock) scale_by_device_scalar_kernel(
    float* __restrict__ x, int64_t n, const float* __restrict__ scale_ptr) {
  const float s = scale_ptr[0];
  const int64_t nvec = n / 4;
  const int64_t stride = (int64_t)gridDim.x * kBlock;
  float4* x4 = (float4*)x;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < nvec; i += stride) {
    float4 v = x4[i];
    v.x *= s;
    v.y *= s;
    v.z *= s;
    v.w *= s;
    x4[i] = v;
  }
  for (int64_t i = nvec * 4 + (int64_t)blockIdx.x * kBlock + threadIdx.x; i < n; i += stride) {
    x[i] *= s;
  }
}

// out: fp32[2] {norm, scale}. partials: fp64 workspace (>= grid blocks).
// flat may be fp32 or bf16. norm_scale: see norm_finalize_kernel.
void l2_norm_and_scale(at::Tensor flat, at::Tensor partials, at::Tensor out, double max_norm,
                       bool apply, double norm_scale) {
  TORCH_CHECK(flat.is_cuda(), "device tensor required");
  const bool is_bf16 = flat.scalar_type() == at::kBFloat16;
  TORCH_CHECK(is_bf16 || flat.scalar_type() == at::kFloat, "flat must be fp32 or bf16");
  TORCH_CHECK(partials.scalar_type() == at::kDouble, "partials must be fp64");
  TORCH_CHECK(out.scalar_type() == at::kFloat && out.numel() >= 2, "out must be fp32[2]");
  const int64_t n = flat.numel();
  auto stream = c10::hip::getCurrentHIPStream();
  int blocks = grid_for(n / 4 + 1, kBlock * 4);
  TORCH_CHECK(partials.numel() >= blocks, "partials workspace too small");
  if (is_bf16) {
    hipLaunchKernelGGL(sqnorm_partials_bf16_kernel, dim3(blocks), dim3(kBlock), 0, stream,
                       (const __hip_bfloat16*)flat.data_ptr(), n, partials.data_ptr<double>());
  } else {
    hipLaunchKernelGGL(sqnorm_partials_kernel, dim3(blocks), dim3(kBlock), 0, stream,
                       flat.data_ptr<float>(), n, partials.data_ptr<double>());
  }
  hipLaunchKernelGGL(norm_finalize_kernel, dim3(1), dim3(kBlock), 0, stream,
                     partials.data_ptr<double>(), blocks, out.data_ptr<float>(),
                     (float)max_norm, (float)norm_scale);
  if (apply) {
    int sblocks = grid_for(n / 4 + 1, kBlock);
    if (is_bf16) {
      hipLaunchKernelGGL(scale_bf16_by_device_scalar_kernel, dim3(sblocks), dim3(kBlock), 0,
                         stream, (__hip_bfloat16*)flat.data_ptr(), n, out.data_ptr<float>() + 1);
    } else {
      hipLaunchKernelGGL(scale_by_device_scalar_kernel, dim3(sblocks), dim3(kBlock), 0, stream,
                         flat.data_ptr<float>(), n, out.data_ptr<float>() + 1);
    }
  }
}

} // namespace dmlamd
