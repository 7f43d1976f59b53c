// Hand-written CDNA4 (gfx950) flash-attention FORWARD for MI355X.
//
// Causal, bf16, head_dim 128, arbitrary [B, H, S, D] strides (S % 64 == 0).
// One 4-wave workgroup owns one (batch, head, 64-query block); K/V tiles
// stream through LDS; Q fragments stay in registers; online softmax in
// fp32; P takes one per-wave LDS round trip to re-shape from the MFMA C
// layout to the A layout. Emits O and the logsumexp rows the aten flash
// backward consumes (torch.ops.aten._scaled_dot_product_flash_attention_
// backward), so training uses this forward + the library backward.
//
// MFMA: v_mfma_f32_16x16x32_bf16 per-wave tiles (layouts verified on
// silicon by mfma_probe in ops.hip / tests/test_ops_gpu.py):
//   A[16x32]: lane l -> A[l & 15][(l >> 4) * 8 + j]
//   B[32x16]: lane l -> B[(l >> 4) * 8 + j][l & 15]
//   C[16x16]: lane l, reg r -> C[(l >> 4) * 4 + r][l & 15]
//
// LDS layout (the v1 linear layouts measured 1113 us vs aotriton's 442 —
// every 256-B-stride row put a 16-lane read group on one bank; v2's
// transposed-V writes were 16-way write-conflicted, SQ_LDS_BANK_CONFLICT
// = 35% of wave cycles):
//   * K tile [64][128] with the guide's T2 XOR swizzle
//     (byte ^= (row & 15) << 4): contiguous uint4 staging writes and
//     conflict-free ds_read_b128 QK^T B-fragments.
//   * V transposed [d][kv] with a (d>>3)-keyed slot XOR (see vt_byte):
//     <=2-way scatter writes, single conflict-free b128 P@V B-fragment
//     reads (v4's scalar-read variant was VALU-bound on address math).
//   * P strip [16][64] with byte ^= (row & 7) << 4.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef unsigned short u16;
typedef unsigned int u32;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define ATTN_BM 64
#define ATTN_BN 64
#define ATTN_D 128
#define ATTN_WAVES 4
__device__ __forceinline__ u16 attn_f2bf(float f) {
  __hip_bfloat16_raw r = __float2bfloat16(f);
  return r.x;
}

// T2 swizzle for the K tile: 16-byte slot index XORed with row & 15.
__device__ __forceinline__ int k_byte(int row, int col_elem) {
  return (row * ATTN_D * 2 + col_elem * 2) ^ ((row & 15) << 4);
}

// Transposed V image [d][kv] with a (d>>3)-keyed slot XOR.
// Writes scatter one u16 per d (16 lanes share a kv row but have distinct
// c8 -> distinct (d>>3) -> distinct banks, <=2-way on the r halves);
// reads are one contiguous b128 of 8 kv for this lane's d (the XOR is
// constant across the 16-byte run and 36*low already spreads the lane
// group over distinct dword banks). Pitch 144 B keeps b128 16-B aligned;
// the XOR (<=240 B) may cross row ends, so the buffer carries 256 B of
// tail padding.
#define VT_PITCH_B 144
__device__ __forceinline__ int vt_byte(int d, int kv) {
  return (d * VT_PITCH_B + kv * 2) ^ (((d >> 3) & 15) << 4);
}

// P strip swizzle (row length 128 B = 8 slots): XOR with row & 7.
__device__ __forceinline__ int p_byte(int row, int col_elem) {
  return (row * ATTN_BN * 2 + col_elem * 2) ^ ((row & 7) << 4);
}

__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, u16* __restrict__ out,
    float* __restrict__ lse,
    long q_sb, long q_sh, long q_ss,
    long k_sb, long k_sh, long k_ss,
    long v_sb, long v_sh, long v_ss,
    int n_heads, int S, float scale) {
  const int qb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int low = lane & 15;   // A/C row | B/C col within a 16-tile
  const int kg = lane >> 4;    // lane group (k chunk | C row group)

  __shared__ u16 ldsK[ATTN_BN * ATTN_D];              // swizzled rows
  __shared__ u16 ldsV[ATTN_D * (VT_PITCH_B / 2) + 128];  // transposed image
  __shared__ u16 ldsP[ATTN_WAVES][16 * ATTN_BN];      // swizzled rows

  // ---- load this wave's Q fragments (rows wid*16 .. +15) ----
  const u16* qbase = q + (long)b * q_sb + (long)h * q_sh
                     + (long)(qb * ATTN_BM) * q_ss;
  union { bf16x8 v; uint4 u; } qfrag[4];
#pragma unroll
  for (int ks = 0; ks < 4; ++ks) {
    qfrag[ks].u = *reinterpret_cast<const uint4*>(
        qbase + (long)(wid * 16 + low) * q_ss + ks * 32 + kg * 8);
  }

  float m_run[4], l_run[4];
  float oacc[8][4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -INFINITY; l_run[r] = 0.f; }
#pragma unroll
  for (int ct = 0; ct < 8; ++ct)
#pragma unroll
    for (int r = 0; r < 4; ++r) oacc[ct][r] = 0.f;

  const u16* kbase = k + (long)b * k_sb + (long)h * k_sh;
  const u16* vbase = v + (long)b * v_sb + (long)h * v_sh;
  const int kv_end = (qb + 1) * ATTN_BM;  // causal upper bound (<= S)
  char* ldsKb = reinterpret_cast<char*>(ldsK);
  char* ldsVb = reinterpret_cast<char*>(ldsV);
  char* ldsPb = reinterpret_cast<char*>(ldsP[wid]);

  for (int kv0 = 0; kv0 < kv_end; kv0 += ATTN_BN) {
    // ---- stage K and V tiles (swizzled rows, contiguous uint4) ----
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const int idx = t * 256 + threadIdx.x;
      const int r = idx >> 4;            // kv row within tile
      const int c8 = (idx & 15) * 8;     // 8-elem column chunk
      *reinterpret_cast<uint4*>(&ldsKb[k_byte(r, c8)]) =
          *reinterpret_cast<const uint4*>(
              kbase + (long)(kv0 + r) * k_ss + c8);
      union { uint4 u; u16 h[8]; } vv;
      vv.u = *reinterpret_cast<const uint4*>(
          vbase + (long)(kv0 + r) * v_ss + c8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *reinterpret_cast<u16*>(&ldsVb[vt_byte(c8 + j, r)]) = vv.h[j];
    }
    __syncthreads();

    // ---- S = scale * (Q @ K^T), 16x64 strip per wave ----
    f32x4 sacc[4];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            &ldsKb[k_byte(ct * 16 + low, ks * 32 + kg * 8)]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks].v, bfrag,
                                                      acc, 0, 0, 0);
      }
      sacc[ct] = acc;
    }

    // ---- causal mask + online softmax (state per reg = per C row) ----
    const int qrow0 = qb * ATTN_BM + wid * 16 + kg * 4;  // + r
    float mx[4], alpha[4], psum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      mx[r] = -INFINITY;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        const int col = kv0 + ct * 16 + low;
        float s = sacc[ct][r] * scale;
        if (col > qrow0 + r) s = -INFINITY;
        sacc[ct][r] = s;
        mx[r] = fmaxf(mx[r], s);
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        mx[r] = fmaxf(mx[r], __shfl_xor(mx[r], off, 64));
      const float mnew = fmaxf(m_run[r], mx[r]);
      alpha[r] = (m_run[r] == -INFINITY) ? 0.f : __expf(m_run[r] - mnew);
      m_run[r] = mnew;
      psum[r] = 0.f;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        const float p = (sacc[ct][r] == -INFINITY)
                            ? 0.f : __expf(sacc[ct][r] - mnew);
        sacc[ct][r] = p;
        psum[r] += p;
        *reinterpret_cast<u16*>(
            &ldsPb[p_byte(kg * 4 + r, ct * 16 + low)]) = attn_f2bf(p);
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        psum[r] += __shfl_xor(psum[r], off, 64);
      l_run[r] = l_run[r] * alpha[r] + psum[r];
#pragma unroll
      for (int ct2 = 0; ct2 < 8; ++ct2) oacc[ct2][r] *= alpha[r];
    }
    __syncthreads();  // P strips visible; K/V reads done before restage

    // ---- O += P @ V ----
    // B-fragment: ONE contiguous b128 read of 8 kv for this lane's column
    // from the transposed V image. v4's 8 scalar reads per fragment cost
    // ~14 VALU per MFMA in address math (SQ_INSTS_VALU 118.8M) — the wide
    // read removes that. (ds_read_b64_tr_b16 was probed on silicon —
    // tests/test_ops_gpu.py tr_probe — and delivers only 16 distinct
    // values per 16-lane group, so it cannot feed this fragment shape.)
#pragma unroll
    for (int ct2 = 0; ct2 < 8; ++ct2) {
      f32x4 acc = {oacc[ct2][0], oacc[ct2][1], oacc[ct2][2], oacc[ct2][3]};
#pragma unroll
      for (int ks2 = 0; ks2 < 2; ++ks2) {
        bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
            &ldsPb[p_byte(low, ks2 * 32 + kg * 8)]);
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            &ldsVb[vt_byte(ct2 * 16 + low, ks2 * 32 + kg * 8)]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc,
                                                      0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[ct2][r] = acc[r];
    }
    __syncthreads();
  }

  // ---- epilogue: normalize, write O [B,H,S,D] contiguous + LSE ----
  u16* obase = out + (((long)b * n_heads + h) * S + qb * ATTN_BM
                      + wid * 16) * ATTN_D;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float inv_l = 1.0f / l_run[r];
#pragma unroll
    for (int ct2 = 0; ct2 < 8; ++ct2) {
      obase[(long)(kg * 4 + r) * ATTN_D + ct2 * 16 + low] =
          attn_f2bf(oacc[ct2][r] * inv_l);
    }
  }
  if (low == 0) {
    float* lbase = lse + ((long)b * n_heads + h) * S + qb * ATTN_BM
                   + wid * 16 + kg * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) lbase[r] = m_run[r] + __logf(l_run[r]);
  }
}

extern "C" int attn_fwd(void* stream, const void* q, const void* k,
                        const void* v, void* out, void* lse,
                        long q_sb, long q_sh, long q_ss,
                        long k_sb, long k_sh, long k_ss,
                        long v_sb, long v_sh, long v_ss,
                        int batch, int n_heads, int S, float scale) {
  if (S <= 0 || S % ATTN_BM != 0 || n_heads <= 0 || batch <= 0) return -1;
  dim3 grid(S / ATTN_BM, n_heads, batch), block(256);
  hipLaunchKernelGGL(attn_fwd_kernel, grid, block, 0,
                     reinterpret_cast<hipStream_t>(stream),
                     (const u16*)q, (const u16*)k, (const u16*)v, (u16*)out,
                     (float*)lse, q_sb, q_sh, q_ss, k_sb, k_sh, k_ss,
                     v_sb, v_sh, v_ss, n_heads, S, scale);
  return 0;
}
