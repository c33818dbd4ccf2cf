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
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int kAttnD = 64; // head dim (v1: fixed)
constexpr int kQT = 16; // query rows per wave
constexpr int kKT = 32; // keys per kv tile (backward kernels)
constexpr int kPStride = 40; // P_lds row stride in bf16 (16B-aligned rows, backward)
constexpr int kWavesPerBlock = 4;
// K/V LDS row stride in bf16: multiple of 8 (16B-aligned vector rows) with
// (stride/2 dwords, 64 banks) gcd = 4 so 16 simultaneous row reads at one
// column offset span 16 distinct banks (rows at stride 88 bf16 = 44 dwords)
constexpr int kKVStride = 88;
// Forward v7 uses 64-key tiles: keys/softmax-round doubles, so the
// per-round fixed costs (max/sum cross-lane reductions, running-stat
// updates, O-accumulator rescale, P LDS round-trip waits, the block-wide
// staging barrier) halve per key while the MFMA count per key stays
// identical. 16 rows at stride 72 bf16 = 36 dwords hit 16 distinct banks
// (gcd(36,64)=4).
constexpr int kKTF = 64; // fwd: keys per kv tile
constexpr int kPStrideF = 72; // fwd: P_lds row stride in bf16

// Tiled 32x64 LDS image index for ds_read_b64_tr_b16 consumption
// (tools/tr_probe.hip): elem[(col/16*2 + row/4%2)*256 + row/8*64 +
// row%4*16 + col%16] = T[row][col]. Row-major b128 reads also work on
// the image (any 8 contiguous cols within a 16-col sub-tile row).
__device__ __forceinline__ int vt_idx(int row, int col) {
  return (((col >> 4) * 2 + ((row >> 2) & 1)) * 4 + (row >> 3)) * 64 + (row & 3) * 16 +
         (col & 15);
}

// v2: the block's 4 waves own consecutive q-tiles of ONE (b,h) and share
// cooperatively staged K/V LDS tiles (one bf16x8 global load per thread per
// tile — coalesced — instead of per-lane scattered loads; v1 measured
// 97-186 TF vs AOTriton's 213-458 at the GPT-2 shape).
// v5: each wave owns TWO 16-row q sub-tiles (32 q rows): the shared K
// fragments and the scalar V fragment reads amortize over twice the
// MFMAs, roughly doubling the per-tile MFMA:overhead ratio.
// per-tensor strides (elements): batch, head, row; the last dim must be
// contiguous. Lets the model pass transpose views without materializing.
struct Strides {
  int64_t b, h, r;
};

__device__ __forceinline__ const __hip_bfloat16* tslice(const __hip_bfloat16* t,
                                                        const Strides& st, int bh, int H) {
  return t + (int64_t)(bh / H) * st.b + (int64_t)(bh % H) * st.h;
}

// Occupancy note: 204 VGPRs + 32 AGPRs -> 2 waves/SIMD. Forcing 3
// waves/SIMD via amdgpu_waves_per_eu(3) spills 164 B/lane and measured
// SLOWER (552 vs 508 us causal at the GPT-2 shape) — the spill traffic
// in the inner loop outweighs the extra latency hiding. Keep 2 waves.
__global__ void __launch_bounds__(kWavesPerBlock * kWave) attn_fwd_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ o,
    float* __restrict__ lse, int BH, int H, int N, float scale, bool causal, Strides sq,
    Strides sk, Strides sv) {
  __shared__ __hip_bfloat16 p_lds_all[kWavesPerBlock][kQT][kPStrideF];
  __shared__ __hip_bfloat16 k_lds[2][kKTF][kKVStride];
  // V lives in TILED images read by ds_read_b64_tr_b16 (empirically
  // probed semantics, tools/tr_probe.hip: 16 contiguous per-lane 8-byte
  // addresses cover a 64-element region transposed as 4x16). A 64-key
  // tile is TWO consecutive 32x64 images (key chunk c at offset c*2048):
  //   elem[c*2048 + (db*2+half)*256 + g*64 + jj*16 + cc] = V[32c+8g+4half+jj][16db+cc]
  __shared__ __hip_bfloat16 v_tr[2][2 * kKTF * kAttnD / 2]; // [2][4096]

  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
  const int row16 = lane & 15; // q index inside the tile (and key row for K frags)
  const int grp = lane >> 4; // 16-lane group 0..3
  __hip_bfloat16(*p_lds)[kPStrideF] = p_lds_all[wave];

  // staging coords: 64x64 tile = 512 bf16x8 pieces, 2 per thread:
  // (st_row, st_col) and (st_row + 32, st_col)
  const int st_row = threadIdx.x >> 3;
  const int st_col = (threadIdx.x & 7) * 8;
  // tiled V image offset for the (st_row, st_col) piece; the second
  // piece lands at the same offset in the second 32x64 image (+2048)
  const int st_vt = ((st_col >> 4) * 2 + ((st_row >> 2) & 1)) * 256 + (st_row >> 3) * 64 +
                    (st_row & 3) * 16 + (st_col & 15);

  const int qrows_per_block = kWavesPerBlock * 2 * kQT; // 128 (2 sub-tiles per wave)
  const int nqb = (N + qrows_per_block - 1) / qrows_per_block;
  const int64_t total_blocks = (int64_t)BH * nqb;

  for (int64_t blk = blockIdx.x; blk < total_blocks; blk += gridDim.x) {
    const int bh = blk / nqb;
    const int qb0 = (blk - (int64_t)bh * nqb) * qrows_per_block;
    const int i0 = qb0 + wave * 2 * kQT; // this wave's 32 q rows
    const bool valid = i0 < N;
    const __hip_bfloat16* qp = tslice(q, sq, bh, H);
    const __hip_bfloat16* kp = tslice(k, sk, bh, H);
    const __hip_bfloat16* vp = tslice(v, sv, bh, H);

    bf16x8 qf[2][2]; // [sub-tile][k-chunk]
    if (valid) {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          qf[sub][c] = *(const bf16x8*)(qp + (int64_t)(i0 + 16 * sub + row16) * sq.r + 32 * c +
                                        8 * grp);
        }
      }
    }

    const float scale2 = scale * 1.44269504f; // fold log2(e): exp -> v_exp_f32
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
    *(bf16x8*)(&k_lds[0][st_row][st_col]) = *(const bf16x8*)(kp + (int64_t)st_row * sk.r + st_col);
    *(bf16x8*)(&k_lds[0][st_row + 32][st_col]) =
        *(const bf16x8*)(kp + (int64_t)(st_row + 32) * sk.r + st_col);
    *(bf16x8*)(&v_tr[0][st_vt]) = *(const bf16x8*)(vp + (int64_t)st_row * sv.r + st_col);
    *(bf16x8*)(&v_tr[0][2048 + st_vt]) =
        *(const bf16x8*)(vp + (int64_t)(st_row + 32) * sv.r + st_col);
    __syncthreads();

    const int ntiles = (kv_end_block + kKTF - 1) / kKTF;
    for (int jt = 0; jt < ntiles; ++jt) {
      const int j0 = jt * kKTF;
      const int buf = jt & 1;
      bf16x8 knext0, knext1, vnext0, vnext1;
      const bool has_next = jt + 1 < ntiles;
      if (has_next) {
        knext0 = *(const bf16x8*)(kp + (int64_t)(j0 + kKTF + st_row) * sk.r + st_col);
        knext1 = *(const bf16x8*)(kp + (int64_t)(j0 + kKTF + 32 + st_row) * sk.r + st_col);
        vnext0 = *(const bf16x8*)(vp + (int64_t)(j0 + kKTF + st_row) * sv.r + st_col);
        vnext1 = *(const bf16x8*)(vp + (int64_t)(j0 + kKTF + 32 + st_row) * sv.r + st_col);
      }

      if (valid && j0 < my_kv_end) {
        // shared V fragments for both q sub-tiles: [key chunk][D chunk],
        // two hardware transpose reads per fragment
        typedef short short4v __attribute__((ext_vector_type(4)));
        bf16x8 vf[2][4];
#pragma unroll
        for (int c = 0; c < 2; ++c) {
#pragma unroll
          for (int db = 0; db < 4; ++db) {
            const short4v t0 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                (__attribute__((address_space(3))) short4v*)&v_tr[buf][c * 2048 +
                                                                       (db * 2 + 0) * 256 +
                                                                       lane * 4]);
            const short4v t1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                (__attribute__((address_space(3))) short4v*)&v_tr[buf][c * 2048 +
                                                                       (db * 2 + 1) * 256 +
                                                                       lane * 4]);
            __builtin_memcpy(&vf[c][db], &t0, 8);
            __builtin_memcpy((char*)&vf[c][db] + 8, &t1, 8);
          }
        }
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
          const int qi0 = i0 + 16 * sub;
          if (causal && j0 >= qi0 + kQT) continue; // tile fully past diagonal

          // ---- S^T = K Q^T for four 16-key quarters ----
          float sv[16]; // h*4 + r: key = j0 + 16h + 4grp + r, query = row16
#pragma unroll
          for (int h = 0; h < 4; ++h) {
            f32x4 acc = {0, 0, 0, 0};
#pragma unroll
            for (int c = 0; c < 2; ++c) {
              const bf16x8 kf = *(const bf16x8*)(&k_lds[buf][16 * h + row16][32 * c + 8 * grp]);
              acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf, qf[sub][c], acc, 0, 0, 0);
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              float s = acc[r] * scale2; // log2-domain scores
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
          for (int x = 1; x < 16; ++x) mt = fmaxf(mt, sv[x]);
          mt = fmaxf(mt, __shfl_xor(mt, 16, kWave));
          mt = fmaxf(mt, __shfl_xor(mt, 32, kWave));
          const float m_new = fmaxf(m_run[sub], mt);
          const float alpha = __builtin_amdgcn_exp2f(m_run[sub] - m_new); // -1e30 -> 0

          float ps = 0.0f;
#pragma unroll
          for (int x = 0; x < 16; ++x) {
            sv[x] = __builtin_amdgcn_exp2f(sv[x] - m_new); // raw v_exp_f32 rate
            ps += sv[x];
          }
          ps += __shfl_xor(ps, 16, kWave);
          ps += __shfl_xor(ps, 32, kWave);
          s_run[sub] = s_run[sub] * alpha + ps;
          m_run[sub] = m_new;

          // ---- P^T -> per-wave LDS slice (4 keys per 8-byte write) ----
#pragma unroll
          for (int h = 0; h < 4; ++h) {
            bf16x4 pw;
#pragma unroll
            for (int r = 0; r < 4; ++r) pw[r] = (__bf16)sv[h * 4 + r];
            *(bf16x4*)(&p_lds[row16][16 * h + 4 * grp]) = pw;
          }

          float a_o[4];
#pragma unroll
          for (int r = 0; r < 4; ++r) a_o[r] = __shfl(alpha, 4 * grp + r, kWave);

          // ---- PV: A = P[q][key] from LDS, B = shared V fragments ----
          const bf16x8 pf0 = *(const bf16x8*)(&p_lds[row16][8 * grp]);
          const bf16x8 pf1 = *(const bf16x8*)(&p_lds[row16][32 + 8 * grp]);
#pragma unroll
          for (int db = 0; db < 4; ++db) {
#pragma unroll
            for (int r = 0; r < 4; ++r) o_acc[sub][db][r] *= a_o[r];
            o_acc[sub][db] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf0, vf[0][db], o_acc[sub][db], 0, 0, 0);
            o_acc[sub][db] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf1, vf[1][db], o_acc[sub][db], 0, 0, 0);
          }
        }
      }
      if (has_next) {
        *(bf16x8*)(&k_lds[buf ^ 1][st_row][st_col]) = knext0;
        *(bf16x8*)(&k_lds[buf ^ 1][st_row + 32][st_col]) = knext1;
        *(bf16x8*)(&v_tr[buf ^ 1][st_vt]) = vnext0;
        *(bf16x8*)(&v_tr[buf ^ 1][2048 + st_vt]) = vnext1;
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
          // convert the log2-domain stats back to natural-log lse
          lse[(int64_t)bh * N + i0 + 16 * sub + row16] =
              0.69314718f * m_run[sub] + __logf(s_run[sub]);
        }
      }
    }
  }
}

static Strides strides_of(const at::Tensor& t) {
  TORCH_CHECK(t.stride(3) == 1, "last dim must be contiguous");
  return Strides{t.stride(0), t.stride(1), t.stride(2)};
}

void attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o, at::Tensor lse,
              double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16, "q must be bf16 [B,H,N,D]");
  TORCH_CHECK(q.dim() == 4 && q.size(3) == kAttnD, "v1 supports head dim 64");
  TORCH_CHECK(k.sizes() == q.sizes() && v.sizes() == q.sizes(),
              "self-attention geometry required: k/v shapes must equal q");
  TORCH_CHECK(k.scalar_type() == at::kBFloat16 && v.scalar_type() == at::kBFloat16,
              "k/v must be bf16");
  const int B = q.size(0), H = q.size(1), N = q.size(2);
  TORCH_CHECK(N % kKTF == 0, "N must be a multiple of 64");
  TORCH_CHECK(o.is_contiguous() && o.sizes() == q.sizes(), "o must be contiguous [B,H,N,D]");
  TORCH_CHECK(lse.scalar_type() == at::kFloat && lse.numel() >= (int64_t)B * H * N,
              "lse must be fp32[B*H*N]");
  const int BH = B * H;
  const int qrows_per_block = kWavesPerBlock * 2 * kQT;
  const int64_t total_blocks = (int64_t)BH * ((N + qrows_per_block - 1) / qrows_per_block);
  const int blocks = (int)std::min<int64_t>(total_blocks, kMaxGrid);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(attn_fwd_kernel, dim3(blocks), dim3(kWavesPerBlock * kWave), 0, stream,
                     (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                     (const __hip_bfloat16*)v.data_ptr(), (__hip_bfloat16*)o.data_ptr(),
                     lse.data_ptr<float>(), BH, H, N, (float)scale, causal, strides_of(q),
                     strides_of(k), strides_of(v));
}

// ===========================================================================
// Backward (flash recomputation, ops/_attention_ref.py::flash_attn_bwd_tiled)
// ===========================================================================
//
//   delta = rowsum(dO * O)                    (attn_bwd_delta_kernel)
//   P^T   = exp(K Q^T * scale - lse[q])       (recomputed per tile pair)
//   dP^T  = V dO^T
//   dS^T  = P^T * (dP^T - delta[q]) * scale
//   dV   += P^T  @ dO      dK += dS^T @ Q     (attn_bwd_dkdv_kernel,
//                                              wave owns a 16-key tile)
//   dQ   += dS @ K                            (attn_bwd_dq_kernel,
//                                              wave owns a 16-q tile)
//
// Every MFMA reuses the forward's fragment patterns; the only transposes
// (P^T/dS^T C-layout -> A operand) go through small per-wave LDS slices
// exactly like the forward's P.

// -------------------------------------------------- delta = rowsum(dO*O)
__global__ void __launch_bounds__(kBlock) attn_bwd_delta_kernel(
    const __hip_bfloat16* __restrict__ dout, const __hip_bfloat16* __restrict__ o,
    float* __restrict__ delta, int64_t rows, int H, int N, Strides sdo) {
  // one wave per 8 rows: lane l -> row l/8, 8-elem chunk l%8
  const int64_t stride = ((int64_t)gridDim.x * kBlock) / 8;
  for (int64_t r0 = ((int64_t)blockIdx.x * kBlock + threadIdx.x) / 8; r0 < rows; r0 += stride) {
    const int chunk = threadIdx.x & 7;
    const int bh = r0 / N;
    const int rr = r0 - (int64_t)bh * N;
    bf16x8 a = *(const bf16x8*)(tslice(dout, sdo, bh, H) + (int64_t)rr * sdo.r + 8 * chunk);
    bf16x8 b = *(const bf16x8*)(o + r0 * kAttnD + 8 * chunk);
    float s = 0.0f;
#pragma unroll
    for (int j = 0; j < 8; ++j) s += (float)a[j] * (float)b[j];
    // reduce the 8 chunks of this row (lanes l..l+7 within the wave)
#pragma unroll
    for (int off = 4; off > 0; off >>= 1) s += __shfl_xor(s, off, kWave);
    if (chunk == 0) delta[r0] = s;
  }
}

// ------------------------------------------------------------- dK and dV
// Block = 4 waves, each owning a 16-key tile of one (b,h); q-tiles of 32
// rows (Q and dO staged in shared LDS) stream past. Per q-tile, per wave:
// 2 MFMAs S^T, 2 MFMAs dP^T, 4+4 MFMAs dV/dK accumulation (k = 32 q rows).
__global__ void __launch_bounds__(kWavesPerBlock * kWave) attn_bwd_dkdv_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    __hip_bfloat16* __restrict__ dk, __hip_bfloat16* __restrict__ dv, int BH, int H, int N,
    float scale, bool causal, Strides sq, Strides sk, Strides sv, Strides sdo) {
  constexpr int kQStep = 32; // q rows per iteration (MFMA contraction width)
  __shared__ __hip_bfloat16 q_lds[kQStep * kAttnD]; // tiled (vt_idx) image
  __shared__ __hip_bfloat16 do_lds[kQStep * kAttnD]; // tiled (vt_idx) image
  // per-wave transpose slices, [key16][q32(+skew)] rows for A-operand reads
  __shared__ __hip_bfloat16 pt_lds_all[kWavesPerBlock][16][kPStride];
  __shared__ __hip_bfloat16 dst_lds_all[kWavesPerBlock][16][kPStride];

  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
  const int row16 = lane & 15;
  const int grp = lane >> 4;
  __hip_bfloat16(*pt_lds)[kPStride] = pt_lds_all[wave];
  __hip_bfloat16(*dst_lds)[kPStride] = dst_lds_all[wave];

  const int st_row = threadIdx.x >> 3; // 0..31
  const int st_col = (threadIdx.x & 7) * 8;

  const int keys_per_block = kWavesPerBlock * 16; // 64
  const int nkb = (N + keys_per_block - 1) / keys_per_block;
  const int64_t total_blocks = (int64_t)BH * nkb;

  for (int64_t blk = blockIdx.x; blk < total_blocks; blk += gridDim.x) {
    const int bh = blk / nkb;
    const int kb0 = (blk - (int64_t)bh * nkb) * keys_per_block;
    const int j0 = kb0 + wave * 16; // this wave's 16 keys
    const bool valid = j0 < N;
    const __hip_bfloat16* qp = tslice(q, sq, bh, H);
    const __hip_bfloat16* kp = tslice(k, sk, bh, H);
    const __hip_bfloat16* vp = tslice(v, sv, bh, H);
    const __hip_bfloat16* dop = tslice(dout, sdo, bh, H);
    const float* lsep = lse + (int64_t)bh * N;
    const float* delp = delta + (int64_t)bh * N;

    // wave-local K and V fragments for this key tile (row-major loads)
    bf16x8 kf[2], vf2[2];
    if (valid) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        kf[c] = *(const bf16x8*)(kp + (int64_t)(j0 + row16) * sk.r + 32 * c + 8 * grp);
        vf2[c] = *(const bf16x8*)(vp + (int64_t)(j0 + row16) * sv.r + 32 * c + 8 * grp);
      }
    }

    f32x4 dv_acc[4], dk_acc[4]; // rows = key (4grp+r), cols = 16*db + row16
#pragma unroll
    for (int db = 0; db < 4; ++db) {
      dv_acc[db] = f32x4{0, 0, 0, 0};
      dk_acc[db] = f32x4{0, 0, 0, 0};
    }

    const int i_start = causal ? (kb0 & ~(kQStep - 1)) : 0; // block-aligned diag
    for (int i0 = i_start; i0 < N; i0 += kQStep) {
      // ---- stage Q and dO into tiled images (one bf16x8 per thread) ----
      const int st_t = vt_idx(st_row, st_col);
      *(bf16x8*)(&q_lds[st_t]) = *(const bf16x8*)(qp + (int64_t)(i0 + st_row) * sq.r + st_col);
      *(bf16x8*)(&do_lds[st_t]) = *(const bf16x8*)(dop + (int64_t)(i0 + st_row) * sdo.r + st_col);
      __syncthreads();

      if (valid && (!causal || i0 + kQStep > j0)) {
        // two 16-q halves share this wave's key tile
#pragma unroll
        for (int hq = 0; hq < 2; ++hq) {
          const int q0 = i0 + 16 * hq; // this half's first q row
          const float lse2_q = lsep[q0 + row16] * 1.44269504f; // log2 domain
          const float del_q = delp[q0 + row16];

          // ---- S^T = K Q^T (rows = key, cols = q) ----
          f32x4 acc = {0, 0, 0, 0};
          f32x4 dpt = {0, 0, 0, 0};
#pragma unroll
          for (int c = 0; c < 2; ++c) {
            const int rm = vt_idx(16 * hq + row16, 32 * c + 8 * grp);
            const bf16x8 qfr = *(const bf16x8*)(&q_lds[rm]);
            const bf16x8 dof = *(const bf16x8*)(&do_lds[rm]);
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf[c], qfr, acc, 0, 0, 0);
            dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vf2[c], dof, dpt, 0, 0, 0);
          }

          // ---- P^T and dS^T in C layout; write transposed slices ----
          // lane holds keys 4grp+r (rows), q = row16 (col)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int key_g = j0 + 4 * grp + r;
            const int q_g = q0 + row16;
            float p = 0.0f;
            if (!causal || key_g <= q_g) {
              p = __builtin_amdgcn_exp2f(acc[r] * (scale * 1.44269504f) - lse2_q);
            }
            const float ds = p * (dpt[r] - del_q) * scale;
            // [key][q] slices: q column of this half = 16*hq + row16
            pt_lds[4 * grp + r][16 * hq + row16] = __float2bfloat16(p);
            dst_lds[4 * grp + r][16 * hq + row16] = __float2bfloat16(ds);
          }
        }
        // same-wave LDS visibility; A-operand reads below

        // ---- dV += P^T @ dO ; dK += dS^T @ Q  (contraction over 32 q) ----
        const bf16x8 ptf = *(const bf16x8*)(&pt_lds[row16][8 * grp]);
        const bf16x8 dstf = *(const bf16x8*)(&dst_lds[row16][8 * grp]);
        typedef short short4v __attribute__((ext_vector_type(4)));
#pragma unroll
        for (int db = 0; db < 4; ++db) {
          bf16x8 dob, qb;
#pragma unroll
          for (int half = 0; half < 2; ++half) {
            const int base = (db * 2 + half) * 256 + lane * 4;
            const short4v td = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                (__attribute__((address_space(3))) short4v*)&do_lds[base]);
            const short4v tq = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                (__attribute__((address_space(3))) short4v*)&q_lds[base]);
            __builtin_memcpy((char*)&dob + 8 * half, &td, 8);
            __builtin_memcpy((char*)&qb + 8 * half, &tq, 8);
          }
          dv_acc[db] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ptf, dob, dv_acc[db], 0, 0, 0);
          dk_acc[db] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dstf, qb, dk_acc[db], 0, 0, 0);
        }
      }
      __syncthreads(); // staged tiles consumed before restage
    }

    if (valid) {
#pragma unroll
      for (int db = 0; db < 4; ++db) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key_o = 4 * grp + r;
          dv[(int64_t)bh * N * kAttnD + (int64_t)(j0 + key_o) * kAttnD + 16 * db + row16] =
              __float2bfloat16(dv_acc[db][r]);
          dk[(int64_t)bh * N * kAttnD + (int64_t)(j0 + key_o) * kAttnD + 16 * db + row16] =
              __float2bfloat16(dk_acc[db][r]);
        }
      }
    }
  }
}

// ------------------------------------------------------------------- dQ
// Mirror of the forward: block = 4 waves x one 16-q tile; 32-key KV tiles
// (K and V staged shared). dq[q][d] += dS[q][key] @ K[key][d].
__global__ void __launch_bounds__(kWavesPerBlock * kWave) attn_bwd_dq_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    __hip_bfloat16* __restrict__ dq, int BH, int H, int N, float scale, bool causal, Strides sq,
    Strides sk, Strides sv, Strides sdo) {
  __shared__ __hip_bfloat16 k_lds[kKT * kAttnD]; // tiled (vt_idx) image
  __shared__ __hip_bfloat16 v_lds[kKT][kKVStride];
  __shared__ __hip_bfloat16 ds_lds_all[kWavesPerBlock][kQT][kPStride]; // [q][key32]

  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
  const int row16 = lane & 15;
  const int grp = lane >> 4;
  __hip_bfloat16(*ds_lds)[kPStride] = ds_lds_all[wave];

  const int st_row = threadIdx.x >> 3;
  const int st_col = (threadIdx.x & 7) * 8;

  const int qrows_per_block = kWavesPerBlock * kQT; // 64
  const int nqb = (N + qrows_per_block - 1) / qrows_per_block;
  const int64_t total_blocks = (int64_t)BH * nqb;

  for (int64_t blk = blockIdx.x; blk < total_blocks; blk += gridDim.x) {
    const int bh = blk / nqb;
    const int qb0 = (blk - (int64_t)bh * nqb) * qrows_per_block;
    const int i0 = qb0 + wave * kQT;
    const bool valid = i0 < N;
    const __hip_bfloat16* qp = tslice(q, sq, bh, H);
    const __hip_bfloat16* kp = tslice(k, sk, bh, H);
    const __hip_bfloat16* vp = tslice(v, sv, bh, H);
    const __hip_bfloat16* dop = tslice(dout, sdo, bh, H);
    const float* lsep = lse + (int64_t)bh * N;
    const float* delp = delta + (int64_t)bh * N;

    bf16x8 qf[2], dof[2];
    float lse_q = 0.0f, del_q = 0.0f;
    if (valid) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        qf[c] = *(const bf16x8*)(qp + (int64_t)(i0 + row16) * sq.r + 32 * c + 8 * grp);
        dof[c] = *(const bf16x8*)(dop + (int64_t)(i0 + row16) * sdo.r + 32 * c + 8 * grp);
      }
      lse_q = lsep[i0 + row16] * 1.44269504f; // log2 domain
      del_q = delp[i0 + row16];
    }

    f32x4 dq_acc[4];
#pragma unroll
    for (int db = 0; db < 4; ++db) dq_acc[db] = f32x4{0, 0, 0, 0};

    const int kv_end_block = causal ? min(qb0 + qrows_per_block, N) : N;
    const int my_kv_end = causal ? (i0 + kQT) : N;

    for (int j0 = 0; j0 < kv_end_block; j0 += kKT) {
      *(bf16x8*)(&k_lds[vt_idx(st_row, st_col)]) =
          *(const bf16x8*)(kp + (int64_t)(j0 + st_row) * sk.r + st_col);
      *(bf16x8*)(&v_lds[st_row][st_col]) =
          *(const bf16x8*)(vp + (int64_t)(j0 + st_row) * sv.r + st_col);
      __syncthreads();

      if (valid && j0 < my_kv_end) {
#pragma unroll
        for (int h = 0; h < 2; ++h) { // two 16-key halves
          // S^T rows = key (4grp+r), col = q (row16)
          f32x4 acc = {0, 0, 0, 0};
          f32x4 dpt = {0, 0, 0, 0};
#pragma unroll
          for (int c = 0; c < 2; ++c) {
            const bf16x8 kfr = *(const bf16x8*)(&k_lds[vt_idx(16 * h + row16, 32 * c + 8 * grp)]);
            const bf16x8 vfr = *(const bf16x8*)(&v_lds[16 * h + row16][32 * c + 8 * grp]);
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kfr, qf[c], acc, 0, 0, 0);
            dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfr, dof[c], dpt, 0, 0, 0);
          }
          // dS^T -> [q][key] slice (4 consecutive keys pack per write)
          bf16x4 dsw;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int key_g = j0 + 16 * h + 4 * grp + r;
            const int q_g = i0 + row16;
            float p = 0.0f;
            if (!causal || key_g <= q_g) {
              p = __builtin_amdgcn_exp2f(acc[r] * (scale * 1.44269504f) - lse_q); // lse_q in log2 domain
            }
            dsw[r] = (__bf16)(p * (dpt[r] - del_q) * scale);
          }
          *(bf16x4*)(&ds_lds[row16][16 * h + 4 * grp]) = dsw;
        }

        // ---- dq += dS @ K (contraction over the 32 keys) ----
        const bf16x8 dsf = *(const bf16x8*)(&ds_lds[row16][8 * grp]);
        typedef short short4v __attribute__((ext_vector_type(4)));
#pragma unroll
        for (int db = 0; db < 4; ++db) {
          bf16x8 kb;
#pragma unroll
          for (int half = 0; half < 2; ++half) {
            const short4v tk = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                (__attribute__((address_space(3))) short4v*)&k_lds[(db * 2 + half) * 256 +
                                                                   lane * 4]);
            __builtin_memcpy((char*)&kb + 8 * half, &tk, 8);
          }
          dq_acc[db] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, kb, dq_acc[db], 0, 0, 0);
        }
      }
      __syncthreads();
    }

    if (valid) {
#pragma unroll
      for (int db = 0; db < 4; ++db) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int q_o = 4 * grp + r;
          dq[(int64_t)bh * N * kAttnD + (int64_t)(i0 + q_o) * kAttnD + 16 * db + row16] =
              __float2bfloat16(dq_acc[db][r]);
        }
      }
    }
  }
}

void attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor dout, at::Tensor o,
              at::Tensor lse, at::Tensor dq, at::Tensor dk, at::Tensor dv, at::Tensor delta,
              double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16, "q must be bf16 [B,H,N,D]");
  TORCH_CHECK(q.dim() == 4 && q.size(3) == kAttnD, "bwd supports head dim 64");
  TORCH_CHECK(k.sizes() == q.sizes() && v.sizes() == q.sizes() && dout.sizes() == q.sizes() &&
                  o.sizes() == q.sizes(),
              "self-attention geometry required: k/v/dout/o shapes must equal q");
  TORCH_CHECK(dout.stride(3) == 1 && o.is_contiguous(), "o contiguous; dout last-dim contig");
  TORCH_CHECK(dq.is_contiguous() && dk.is_contiguous() && dv.is_contiguous(),
              "grad outputs must be contiguous");
  TORCH_CHECK(dq.sizes() == q.sizes() && dk.sizes() == q.sizes() && dv.sizes() == q.sizes(),
              "grad output shapes must equal q");
  const int B = q.size(0), H = q.size(1), N = q.size(2);
  TORCH_CHECK(N % 64 == 0, "bwd requires N to be a multiple of 64");
  const int BH = B * H;
  TORCH_CHECK(lse.scalar_type() == at::kFloat && lse.numel() >= (int64_t)BH * N,
              "lse must be fp32[BH*N]");
  TORCH_CHECK(delta.numel() >= (int64_t)BH * N && delta.scalar_type() == at::kFloat,
              "delta workspace must be fp32[BH*N]");
  auto stream = c10::hip::getCurrentHIPStream();

  const int64_t rows = (int64_t)BH * N;
  hipLaunchKernelGGL(attn_bwd_delta_kernel, dim3(grid_for(rows * 8, kBlock)), dim3(kBlock), 0,
                     stream, (const __hip_bfloat16*)dout.data_ptr(),
                     (const __hip_bfloat16*)o.data_ptr(), delta.data_ptr<float>(), rows, H, N,
                     strides_of(dout));

  const int64_t kv_blocks = (int64_t)BH * ((N + 63) / 64);
  hipLaunchKernelGGL(attn_bwd_dkdv_kernel, dim3((int)std::min<int64_t>(kv_blocks, kMaxGrid)),
                     dim3(kWavesPerBlock * kWave), 0, stream,
                     (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                     (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                     lse.data_ptr<float>(), delta.data_ptr<float>(),
                     (__hip_bfloat16*)dk.data_ptr(), (__hip_bfloat16*)dv.data_ptr(), BH, H, N,
                     (float)scale, causal, strides_of(q), strides_of(k), strides_of(v),
                     strides_of(dout));

  const int64_t q_blocks = (int64_t)BH * ((N + 63) / 64);
  hipLaunchKernelGGL(attn_bwd_dq_kernel, dim3((int)std::min<int64_t>(q_blocks, kMaxGrid)),
                     dim3(kWavesPerBlock * kWave), 0, stream,
                     (const __hip_bfloat16*)q.data_ptr(), (const __hip_bfloat16*)k.data_ptr(),
                     (const __hip_bfloat16*)v.data_ptr(), (const __hip_bfloat16*)dout.data_ptr(),
                     lse.data_ptr<float>(), delta.data_ptr<float>(),
                     (__hip_bfloat16*)dq.data_ptr(), BH, H, N, (float)scale, causal,
                     strides_of(q), strides_of(k), strides_of(v), strides_of(dout));
}

} // namespace dmlamd
