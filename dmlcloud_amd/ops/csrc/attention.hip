// EXPERIMENTAL: flash-attention forward for gfx950 (MFMA bf16, D=64).
//
// Correctness-first v1 of the custom attention path (docs/ROADMAP.md
// item 1; the backward follows in round 2 — this kernel already emits
// the per-row lse the backward needs). Tiling mirrors
// ops/_attention_ref.py exactly; fragment maps are the empirically
// confirmed ones (profiles/mfma_16x16x32_bf16_layout.txt):
//
//   A (bf16x8): lane l elem j <-> A[l%16][8*(l/16)+j]
//   B (bf16x8): lane l elem j <-> B[8*(l/16)+j][l%16]
//   C/D (f32x4): lane l reg r <-> D[4*(l/16)+r][l%16]
//
// Structure: ONE WAVE owns a 16-row Q tile and iterates 32-key KV tiles.
// S is computed TRANSPOSED (S^T = K Q^T) so each lane's 8 scores share a
// single query (softmax row reductions = 2 xor-shuffles); P^T goes
// through a small per-wave LDS slice to re-enter the PV MFMA as the A
// operand. Online softmax keeps m/sumexp in registers; the O accumulator
// (16x64) lives in 4 C fragments.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_bf16.h>

#include "ops_common.h"

namespace dmlamd {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int kAttnD = 64; // head dim (v1: fixed)
constexpr int kQT = 16; // query rows per wave
constexpr int kKT = 32; // keys per kv tile
constexpr int kPStride = 40; // P_lds row stride in bf16 (16B-aligned rows)
constexpr int kWavesPerBlock = 4;
// K/V LDS row stride in bf16: multiple of 8 (16B-aligned vector rows) with
// (stride/2 dwords, 64 banks) gcd = 4 so 16 simultaneous row reads at one
// column offset span 16 distinct banks (rows at stride 88 bf16 = 44 dwords)
constexpr int kKVStride = 88;

// v2: the block's 4 waves own consecutive q-tiles of ONE (b,h) and share
// cooperatively staged K/V LDS tiles (one bf16x8 global load per thread per
// tile — coalesced — instead of per-lane scattered loads; v1 measured
// 97-186 TF vs AOTriton's 213-458 at the GPT-2 shape).
// v5: each wave owns TWO 16-row q sub-tiles (32 q rows): the shared K
// fragments and the scalar V fragment reads amortize over twice the
// MFMAs, roughly doubling the per-tile MFMA:overhead ratio.
__global__ void __launch_bounds__(kWavesPerBlock * kWave) attn_fwd_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ o,
    float* __restrict__ lse, int BH, int N, float scale, bool causal) {
  __shared__ __hip_bfloat16 p_lds_all[kWavesPerBlock][kQT][kPStride];
  __shared__ __hip_bfloat16 k_lds[2][kKT][kKVStride];
  __shared__ __hip_bfloat16 v_lds[2][kKT][kKVStride];

  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
  const int row16 = lane & 15; // q index inside the tile (and key row for K frags)
  const int grp = lane >> 4; // 16-lane group 0..3
  __hip_bfloat16(*p_lds)[kPStride] = p_lds_all[wave];

  // staging coords: thread t loads row t/8, bf16x8 chunk t%8 of a 32x64 tile
  const int st_row = threadIdx.x >> 3;
  const int st_col = (threadIdx.x & 7) * 8;

  const int qrows_per_block = kWavesPerBlock * 2 * kQT; // 128 (2 sub-tiles per wave)
  const int nqb = (N + qrows_per_block - 1) / qrows_per_block;
  const int64_t total_blocks = (int64_t)BH * nqb;

  for (int64_t blk = blockIdx.x; blk < total_blocks; blk += gridDim.x) {
    const int bh = blk / nqb;
    const int qb0 = (blk - (int64_t)bh * nqb) * qrows_per_block;
    const int i0 = qb0 + wave * 2 * kQT; // this wave's 32 q rows
    const bool valid = i0 < N;
    const __hip_bfloat16* qp = q + (int64_t)bh * N * kAttnD;
    const __hip_bfloat16* kp = k + (int64_t)bh * N * kAttnD;
    const __hip_bfloat16* vp = v + (int64_t)bh * N * kAttnD;

    bf16x8 qf[2][2]; // [sub-tile][k-chunk]
    if (valid) {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          qf[sub][c] = *(const bf16x8*)(qp + (int64_t)(i0 + 16 * sub + row16) * kAttnD +
                                        32 * c + 8 * grp);
        }
      }
    }

    float m_run[2] = {-1e30f, -1e30f};
    float s_run[2] = {0.0f, 0.0f};
    f32x4 o_acc[2][4];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub)
#pragma unroll
      for (int db = 0; db < 4; ++db) o_acc[sub][db] = f32x4{0, 0, 0, 0};

    const int kv_end_block = causal ? min(qb0 + qrows_per_block, N) : N;
    const int my_kv_end = causal ? (i0 + 2 * kQT) : N;

    // write-late double buffer (guide §6 G15): the NEXT tile's global
    // loads stay in flight through the current tile's compute; their
    // ds_write targets the other buffer just before the single barrier.
    *(bf16x8*)(&k_lds[0][st_row][st_col]) =
        *(const bf16x8*)(kp + (int64_t)st_row * kAttnD + st_col);
    *(bf16x8*)(&v_lds[0][st_row][st_col]) =
        *(const bf16x8*)(vp + (int64_t)st_row * kAttnD + st_col);
    __syncthreads();

    const int ntiles = (kv_end_block + kKT - 1) / kKT;
    for (int jt = 0; jt < ntiles; ++jt) {
      const int j0 = jt * kKT;
      const int buf = jt & 1;
      bf16x8 knext, vnext;
      const bool has_next = jt + 1 < ntiles;
      if (has_next) {
        knext = *(const bf16x8*)(kp + (int64_t)(j0 + kKT + st_row) * kAttnD + st_col);
        vnext = *(const bf16x8*)(vp + (int64_t)(j0 + kKT + st_row) * kAttnD + st_col);
      }

      if (valid && j0 < my_kv_end) {
        // shared V fragments for both q sub-tiles (the amortization win)
        bf16x8 vf[4];
#pragma unroll
        for (int db = 0; db < 4; ++db) {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            vf[db][j] = *(const __bf16*)(&v_lds[buf][8 * grp + j][16 * db + row16]);
          }
        }
        typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));

#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
          const int qi0 = i0 + 16 * sub;
          if (causal && j0 >= qi0 + kQT) continue; // tile fully past diagonal

          // ---- S^T = K Q^T for two 16-key halves ----
          float sv[8]; // h*4 + r: key = j0 + 16h + 4grp + r, query = row16
#pragma unroll
          for (int h = 0; h < 2; ++h) {
            f32x4 acc = {0, 0, 0, 0};
#pragma unroll
            for (int c = 0; c < 2; ++c) {
              const bf16x8 kf = *(const bf16x8*)(&k_lds[buf][16 * h + row16][32 * c + 8 * grp]);
              acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf, qf[sub][c], acc, 0, 0, 0);
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              float s = acc[r] * scale;
              if (causal) {
                const int key_g = j0 + 16 * h + 4 * grp + r;
                const int q_g = qi0 + row16;
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
          const float m_new = fmaxf(m_run[sub], mt);
          const float alpha = __expf(m_run[sub] - m_new); // -1e30 -> 0

          float ps = 0.0f;
#pragma unroll
          for (int x = 0; x < 8; ++x) {
            sv[x] = __expf(sv[x] - m_new);
            ps += sv[x];
          }
          ps += __shfl_xor(ps, 16, kWave);
          ps += __shfl_xor(ps, 32, kWave);
          s_run[sub] = s_run[sub] * alpha + ps;
          m_run[sub] = m_new;

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

          // ---- PV: A = P[q][key] from LDS, B = shared V fragments ----
          const bf16x8 pf = *(const bf16x8*)(&p_lds[row16][8 * grp]);
#pragma unroll
          for (int db = 0; db < 4; ++db) {
#pragma unroll
            for (int r = 0; r < 4; ++r) o_acc[sub][db][r] *= a_o[r];
            o_acc[sub][db] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf[db], o_acc[sub][db], 0, 0, 0);
          }
        }
      }
      if (has_next) {
        *(bf16x8*)(&k_lds[buf ^ 1][st_row][st_col]) = knext;
        *(bf16x8*)(&v_lds[buf ^ 1][st_row][st_col]) = vnext;
      }
      __syncthreads(); // readers of buf done AND buf^1 writes visible
    }

    if (valid) {
      // ---- epilogue: O /= sumexp, store; lse = m + log(sumexp) ----
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
        for (int db = 0; db < 4; ++db) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int q_o = 4 * grp + r;
            const float denom = __shfl(s_run[sub], q_o, kWave);
            const float val = o_acc[sub][db][r] / denom;
            o[(int64_t)bh * N * kAttnD + (int64_t)(i0 + 16 * sub + q_o) * kAttnD + 16 * db +
              row16] = __float2bfloat16(val);
          }
        }
        if (lane < 16) {
          lse[(int64_t)bh * N + i0 + 16 * sub + row16] = m_run[sub] + __logf(s_run[sub]);
        }
      }
    }
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
  const int qrows_per_block = kWavesPerBlock * 2 * kQT;
  const int64_t total_blocks = (int64_t)BH * ((N + qrows_per_block - 1) / qrows_per_block);
  const int blocks = (int)std::min<int64_t>(total_blocks, kMaxGrid);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(attn_fwd_kernel, dim3(blocks), dim3(kWavesPerBlock * kWave), 0, stream,
                     (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                     (const __hip_bfloat16*)v.data_ptr(), (__hip_bfloat16*)o.data_ptr(),
                     lse.data_ptr<float>(), BH, N, (float)scale, causal);
}

} // namespace dmlamd
