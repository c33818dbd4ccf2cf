// EXPERIMENTAL: flash-attention forward for gfx950 (MFMA bf16, D=64).
//
// Custom attention path (docs/ROADMAP.md item 1; backward follows in
// round 2 — this kernel already emits the per-row lse it needs). Tiling
// mirrors ops/_attention_ref.py; fragment maps are the empirically
// confirmed ones (profiles/mfma_16x16x32_bf16_layout.txt):
//
//   A (bf16x8): lane l elem j <-> A[l%16][8*(l/16)+j]
//   B (bf16x8): lane l elem j <-> B[8*(l/16)+j][l%16]
//   C/D (f32x4): lane l reg r <-> D[4*(l/16)+r][l%16]
//
// Iteration history (B=64 H=12 N=1024 D=64, vs AOTriton SDPA 213/425 TF
// causal/full; profiles/attention_fwd_experimental.txt):
//   v1 wave-per-q-tile, scattered global K/V:        97 / 186 TF
//   v2 4 waves share staged K/V LDS tiles:          130 / 250 TF
//   v3 write-late double buffer: unchanged — NOT staging-bound; causal
//      wall == full wall because block barriers pace all 4 waves by the
//      deepest-diagonal wave (active in every round).
//   v4 (this file) splits the KV dimension across the block's 4 waves
//      instead of Q: block = ONE 16-row q-tile; wave w owns kv tiles
//      w, w+4, ... in its own LDS buffer (staged cooperatively, 4 tiles
//      per round, one barrier pair per FOUR tiles); per-wave online
//      softmax partials merge at the end (flash split-K combine). Every
//      wave is busy in every round, causal included.
//
// S is computed TRANSPOSED (S^T = K Q^T) so each lane's 8 scores share a
// single query (softmax row reductions = 2 xor-shuffles); P^T re-enters
// the PV MFMA as the A operand through a small per-wave LDS slice.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_bf16.h>

#include "ops_common.h"

namespace dmlamd {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int kAttnD = 64; // head dim (fixed in v1..v4)
constexpr int kQT = 16; // query rows per block
constexpr int kKT = 32; // keys per kv tile
constexpr int kPStride = 40; // P_lds row stride in bf16 (16B-aligned rows)
constexpr int kWavesPerBlock = 4;
// K/V LDS row stride in bf16: multiple of 8 (16B-aligned vector rows) with
// gcd(stride/2 dwords, 64 banks) = 4 so 16 simultaneous row reads at one
// column offset span 16 distinct banks (88 bf16 = 44 dwords)
constexpr int kKVStride = 88;
constexpr int kOStride = 68; // merge buffer row stride in f32 (bank skew)

__global__ void __launch_bounds__(kWavesPerBlock * kWave) attn_fwd_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ o,
    float* __restrict__ lse, int BH, int N, float scale, bool causal) {
  // staging buffers and the merge scratch overlap (never live together)
  __shared__ __align__(16) char smem[kWavesPerBlock * 2 * kKT * kKVStride * 2];
  __shared__ __hip_bfloat16 p_lds_all[kWavesPerBlock][kQT][kPStride];
  __shared__ float ms_lds[kWavesPerBlock][2][kQT]; // per-wave m, s partials

  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
  const int row16 = lane & 15; // q index inside the tile (and key row for K frags)
  const int grp = lane >> 4; // 16-lane group 0..3
  __hip_bfloat16(*p_lds)[kPStride] = p_lds_all[wave];
  // buffer w holds wave w's current kv tile: [K|V][kKT][kKVStride]
  auto kbuf = [&](int w) { return (__hip_bfloat16*)smem + (int64_t)w * 2 * kKT * kKVStride; };
  auto vbuf = [&](int w) {
    return (__hip_bfloat16*)smem + (int64_t)w * 2 * kKT * kKVStride + kKT * kKVStride;
  };
  float* o_merge = (float*)smem; // [kWavesPerBlock][kQT][kOStride] on reuse

  // staging coords: thread t fills row t/8, bf16x8 chunk t%8 of one 32x64 tile
  const int st_row = threadIdx.x >> 3;
  const int st_col = (threadIdx.x & 7) * 8;

  const int tiles_per_bh = N / kQT;
  const int64_t total_qtiles = (int64_t)BH * tiles_per_bh;

  for (int64_t qtile = blockIdx.x; qtile < total_qtiles; qtile += gridDim.x) {
    const int bh = qtile / tiles_per_bh;
    const int i0 = (qtile - (int64_t)bh * tiles_per_bh) * kQT;
    const __hip_bfloat16* qp = q + (int64_t)bh * N * kAttnD;
    const __hip_bfloat16* kp = k + (int64_t)bh * N * kAttnD;
    const __hip_bfloat16* vp = v + (int64_t)bh * N * kAttnD;

    bf16x8 qf[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      qf[c] = *(const bf16x8*)(qp + (int64_t)(i0 + row16) * kAttnD + 32 * c + 8 * grp);
    }

    float m_run = -1e30f;
    float s_run = 0.0f;
    f32x4 o_acc[4];
#pragma unroll
    for (int db = 0; db < 4; ++db) o_acc[db] = f32x4{0, 0, 0, 0};

    const int kv_end = causal ? (i0 + kQT) : N;
    const int ntiles = (kv_end + kKT - 1) / kKT;
    const int nrounds = (ntiles + kWavesPerBlock - 1) / kWavesPerBlock;

    for (int ri = 0; ri < nrounds; ++ri) {
      // ---- stage up to 4 tiles, one per wave buffer (cooperative) ----
#pragma unroll
      for (int w = 0; w < kWavesPerBlock; ++w) {
        const int jt = ri * kWavesPerBlock + w;
        if (jt < ntiles) {
          const int j0 = jt * kKT;
          *(bf16x8*)(kbuf(w) + st_row * kKVStride + st_col) =
              *(const bf16x8*)(kp + (int64_t)(j0 + st_row) * kAttnD + st_col);
          *(bf16x8*)(vbuf(w) + st_row * kKVStride + st_col) =
              *(const bf16x8*)(vp + (int64_t)(j0 + st_row) * kAttnD + st_col);
        }
      }
      __syncthreads();

      const int jt = ri * kWavesPerBlock + wave;
      if (jt < ntiles) {
        const int j0 = jt * kKT;
        const __hip_bfloat16* kw = kbuf(wave);
        const __hip_bfloat16* vw = vbuf(wave);

        // ---- S^T = K Q^T for two 16-key halves ----
        float sv[8]; // h*4 + r: key = j0 + 16h + 4grp + r, query = row16
#pragma unroll
        for (int h = 0; h < 2; ++h) {
          f32x4 acc = {0, 0, 0, 0};
#pragma unroll
          for (int c = 0; c < 2; ++c) {
            const bf16x8 kf =
                *(const bf16x8*)(kw + (16 * h + row16) * kKVStride + 32 * c + 8 * grp);
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf, qf[c], acc, 0, 0, 0);
          }
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            float s = acc[r] * scale;
            if (causal) {
              const int key_g = j0 + 16 * h + 4 * grp + r;
              const int q_g = i0 + row16;
              if (key_g > q_g) s = -1e30f;
            }
            sv[h * 4 + r] = s;
          }
        }

        // ---- online softmax (per q = row16; reduce across grp groups) ----
        float mt = sv[0];
#pragma unroll
        for (int x = 1; x < 8; ++x) mt = fmaxf(mt, sv[x]);
        mt = fmaxf(mt, __shfl_xor(mt, 16, kWave));
        mt = fmaxf(mt, __shfl_xor(mt, 32, kWave));
        const float m_new = fmaxf(m_run, mt);
        const float alpha = __expf(m_run - m_new); // m_run=-1e30 -> 0

        float ps = 0.0f;
#pragma unroll
        for (int x = 0; x < 8; ++x) {
          sv[x] = __expf(sv[x] - m_new);
          ps += sv[x];
        }
        ps += __shfl_xor(ps, 16, kWave);
        ps += __shfl_xor(ps, 32, kWave);
        s_run = s_run * alpha + ps;
        m_run = m_new;

        // ---- P^T -> per-wave LDS slice (4 keys per 8-byte write) ----
#pragma unroll
        for (int h = 0; h < 2; ++h) {
          bf16x4 pw;
#pragma unroll
          for (int r = 0; r < 4; ++r) pw[r] = (__bf16)sv[h * 4 + r];
          *(bf16x4*)(&p_lds[row16][16 * h + 4 * grp]) = pw;
        }

        float a_o[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) a_o[r] = __shfl(alpha, 4 * grp + r, kWave);

        // ---- PV: A = P[q][key] from LDS, B = V[key][d] from LDS ----
        const bf16x8 pf = *(const bf16x8*)(&p_lds[row16][8 * grp]);
#pragma unroll
        for (int db = 0; db < 4; ++db) {
          bf16x8 vf;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            vf[j] = *(const __bf16*)(vw + (8 * grp + j) * kKVStride + 16 * db + row16);
          }
#pragma unroll
          for (int r = 0; r < 4; ++r) o_acc[db][r] *= a_o[r];
          o_acc[db] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, o_acc[db], 0, 0, 0);
        }
      }
      __syncthreads(); // all buffers consumed before restaging
    }

    // ---- flash split-K merge of the 4 per-wave partials ----
    // (the staging buffers are dead now; o_merge reuses their LDS)
    if (lane < kQT) {
      ms_lds[wave][0][lane] = m_run; // per-q m partial (q = row16)
      ms_lds[wave][1][lane] = s_run;
    }
#pragma unroll
    for (int db = 0; db < 4; ++db) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int q_o = 4 * grp + r;
        o_merge[(wave * kQT + q_o) * kOStride + 16 * db + row16] = o_acc[db][r];
      }
    }
    __syncthreads();

    // 256 threads cover the 16x64 output: thread t -> (q = t/16, d = t%16*4..+4)
    {
      const int q_o = threadIdx.x >> 4;
      const int d0 = (threadIdx.x & 15) * 4;
      float m_all = -1e30f;
#pragma unroll
      for (int w = 0; w < kWavesPerBlock; ++w) m_all = fmaxf(m_all, ms_lds[w][0][q_o]);
      float s_all = 0.0f;
      float ov[4] = {0, 0, 0, 0};
#pragma unroll
      for (int w = 0; w < kWavesPerBlock; ++w) {
        const float sc = __expf(ms_lds[w][0][q_o] - m_all);
        s_all += ms_lds[w][1][q_o] * sc;
#pragma unroll
        for (int x = 0; x < 4; ++x) {
          ov[x] += o_merge[(w * kQT + q_o) * kOStride + d0 + x] * sc;
        }
      }
      const float inv = 1.0f / s_all;
#pragma unroll
      for (int x = 0; x < 4; ++x) {
        o[(int64_t)bh * N * kAttnD + (int64_t)(i0 + q_o) * kAttnD + d0 + x] =
            __float2bfloat16(ov[x] * inv);
      }
      if (d0 == 0) {
        lse[(int64_t)bh * N + i0 + q_o] = m_all + __logf(s_all);
      }
    }
    __syncthreads(); // merge reads done before the next q-tile restages
  }
}

void attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o, at::Tensor lse,
              double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16 && q.is_contiguous(),
              "q must be contiguous bf16 [B,H,N,D]");
  TORCH_CHECK(q.dim() == 4 && q.size(3) == kAttnD, "v1 supports head dim 64");
  const int B = q.size(0), H = q.size(1), N = q.size(2);
  TORCH_CHECK(N % kKT == 0, "N must be a multiple of 32");
  const int BH = B * H;
  const int64_t total_qtiles = (int64_t)BH * (N / kQT);
  const int blocks = (int)std::min<int64_t>(total_qtiles, kMaxGrid);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(attn_fwd_kernel, dim3(blocks), dim3(kWavesPerBlock * kWave), 0, stream,
                     (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                     (const __hip_bfloat16*)v.data_ptr(), (__hip_bfloat16*)o.data_ptr(),
                     lse.data_ptr<float>(), BH, N, (float)scale, causal);
}

} // namespace dmlamd
