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

__device__ __forceinline__ float bf2f_(u16 b) {
  union { u32 u; float f; } c;
  c.u = ((u32)b) << 16;
  return c.f;
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
// slot-XOR key k(m), m = d>>3: (m&7) | (((m>>1)&1)<<3).
// Constraints (brute-force verified over the full image):
//  * byte-map injective: rows 8m-1/8m share a 256-B block at odd m, so
//    adjacent key pairs must flip slot-bit-3 together (the 9m key violated
//    this and corrupted the image);
//  * v6 b128 reads (32-d spans, lane groups {0-3,12-15,20-27}): <=2-way
//    (the v5 key (d>>3)&15 was 3-way there);
//  * v5 reads <=2-way, write scatter <=4-way (k mod 8 = m mod 8 spread).
__device__ __forceinline__ int vt_byte(int d, int kv) {
  const int m = d >> 3;
  return (d * VT_PITCH_B + kv * 2) ^
         ((((m & 7) | (((m >> 1) & 1) << 3))) << 4);
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

// ===========================================================================
// v6: 8-wave swapped-operand forward (guide §B "8-warp 32×32 ladder").
//
// Structure: one 512-thread workgroup owns 256 q rows (32 per wave); K/V
// tiles of 64 keys double-buffered in LDS; Q rows live in registers.
// BOTH MFMA products are operand-swapped so the q index stays lane-local
// end to end:
//   S^T = K @ Q^T   via mfma_32x32x16(A=K-frag, B=Q-frag)  -> C[kv][q=lane&31]
//   O^T = V^T @ P   via mfma_32x32x16(A=Vt-frag, B=P-frag) -> C[d][q=lane&31]
// so the online-softmax state (m, l) is one scalar pair per lane (its q row,
// 16 kv per half-wave), the row reduce is an in-register tree plus ONE
// __shfl_xor(32) half-merge, and P never round-trips through LDS
// (v5 paid a pack + ds_write + barrier + ds_read per tile for that).
// Softmax runs in exp2 space (scale2 = scale*log2e folded into the S scale).
// Techniques: T12 (cvt_pk_bf16_f32 + permlane32_swap P repack), T13
// (defer-rescale THR=8), T14 (issue next tile's global loads before QK^T,
// LDS write before PV), causal tile skip + diagonal-only masking, reversed
// qb launch order (longest blocks first), native GQA (kv head = h / group).
// ===========================================================================

typedef __attribute__((ext_vector_type(16))) float f32x16;

#define V6_BM 256
#define V6_QBLK 32
#define V6_BN 64
// per-buffer LDS halves (u16 counts): K [64][128] swizzled rows, then the
// transposed V image (pitch 144 B + 256 B XOR tail)
#define V6_K_U16 (V6_BN * ATTN_D)                        // 8192
#define V6_V_U16 (ATTN_D * (VT_PITCH_B / 2) + 128)       // 9344
#define V6_BUF_U16 (V6_K_U16 + V6_V_U16)

__device__ __forceinline__ unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// raw v_exp_f32 (2^x): libm exp2f lowers to a ~5-instruction guarded
// sequence (ldexp + cndmask pairs); softmax arguments are <= 0 and
// underflow-to-0 is exactly what we want.
__device__ __forceinline__ float exp2_raw(float x) {
  float r;
  asm("v_exp_f32 %0, %1" : "=v"(r) : "v"(x));
  return r;
}

// 3-deep glds pipeline helpers (fwd v7 + the v7-style backward kernels):
// counted vmcnt only, raw barriers (no vmcnt(0) drain in any main loop)
__device__ __forceinline__ void v7_wait_vmcnt4() {
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
}
__device__ __forceinline__ void v7_wait_vmcnt0() {
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}
__device__ __forceinline__ void v7_barrier() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
}

#define V7_RING 3


template <int NW>   // waves per workgroup: 4 (BM=128, 2 WGs/CU) or 8 (BM=256)
__global__ __launch_bounds__(NW * 64) void attn_fwd_v6_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, u16* __restrict__ out,
    float* __restrict__ lse,
    long q_sb, long q_sh, long q_ss,
    long k_sb, long k_sh, long k_ss,
    long v_sb, long v_sh, long v_ss,
    int n_heads, int gqa_group, int S, float scale2) {
  constexpr int NT = NW * 64;          // threads
  constexpr int BM = NW * 32;          // q rows per workgroup
  constexpr int KP = 1024 / NT;        // K glds pieces per thread (16 B each)
  constexpr int VU = 512 / NT;         // V kv-pair units per thread
  const int qb = gridDim.x - 1 - blockIdx.x;   // longest-running blocks first
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / gqa_group;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int low = lane & 31;
  const int hi = lane >> 5;

  __shared__ __attribute__((aligned(16))) u16 lds[2][V6_BUF_U16];

  const int q0w = qb * BM + wid * V6_QBLK;
  const int qrow = q0w + low;                  // this lane's q row

  // Q row in registers: 8 k-chunks, lane holds Q[qrow][kc*16 + hi*8 + j]
  union F8 { bf16x8 v; uint4 u; u16 h[8]; };
  const u16* qptr = q + (long)b * q_sb + (long)h * q_sh + (long)qrow * q_ss;
  F8 qf[8];
#pragma unroll
  for (int kc = 0; kc < 8; ++kc)
    qf[kc].u = *reinterpret_cast<const uint4*>(qptr + kc * 16 + hi * 8);

  // O^T accumulators: 4 d-subtiles; lane holds O[d=crow(r,hi)+32*ds][qrow]
  f32x16 oacc[4];
#pragma unroll
  for (int ds = 0; ds < 4; ++ds)
#pragma unroll
    for (int r = 0; r < 16; ++r) oacc[ds][r] = 0.f;
  float m_run = -INFINITY, l_run = 0.f;

  const u16* kbase = k + (long)b * k_sb + (long)hkv * k_sh;
  const u16* vbase = v + (long)b * v_sb + (long)hkv * v_sh;
  const int n_tiles = (qb * BM + BM) / V6_BN;   // causal upper bound

  char* lds0 = reinterpret_cast<char*>(&lds[0][0]);
  char* lds1 = reinterpret_cast<char*>(&lds[1][0]);
  const int VOFF = V6_K_U16 * 2;               // V image byte offset in a buf

  // K staging goes by global->LDS DMA (glds): zero in-flight registers and
  // zero ds_write instructions; the T2 swizzle is inverted onto the SOURCE
  // address (rule 21: glds writes lane-linear), which keeps the LDS image
  // byte-identical to the ds_write version and costs nothing in coalescing
  // (the XOR permutes 16-B slots within one 256-B row = the same two 128-B
  // requests per 16 lanes). V cannot go by glds (transposed image) and
  // keeps register staging: kv-row pairs written as u32 after QK^T.
  const u16* kgp[KP];
  const u16* vgp[2 * VU];
#pragma unroll
  for (int i = 0; i < KP; ++i) {
    const int idx = i * NT + tid;
    const int L = idx * 16;                       // lane-linear LDS byte
    const int kr = L >> 8;
    const int kin = (((L & 255) ^ ((kr & 15) << 4)) >> 1);
    kgp[i] = kbase + (long)kr * k_ss + kin;
  }
#pragma unroll
  for (int u = 0; u < VU; ++u) {
    const int unit = u * NT + tid;                // (rp, c8) unit
    const int rp2 = (unit >> 4) * 2;
    const int c8 = (unit & 15) * 8;
#pragma unroll
    for (int i = 0; i < 2; ++i)
      vgp[u * 2 + i] = vbase + (long)(rp2 + i) * v_ss + c8;
  }
  const long kstep = (long)V6_BN * k_ss;
  const long vstep = (long)V6_BN * v_ss;
  const int wu64 = __builtin_amdgcn_readfirstlane(tid >> 6);

  uint4 vst[2 * VU];
#define V6_GLDS_K(BUFI)                                                     \
  _Pragma("unroll")                                                         \
  for (int i = 0; i < KP; ++i) {                                            \
    __builtin_amdgcn_global_load_lds(                                       \
        (const u32*)kgp[i], (u32*)&lds[BUFI][(i * NT + wu64 * 64) * 8],     \
        16, 0, 0);                                                          \
    kgp[i] += kstep;                                                        \
  }
#define V6_LOAD_V()                                                         \
  _Pragma("unroll")                                                         \
  for (int i = 0; i < 2 * VU; ++i) {                                        \
    vst[i] = *reinterpret_cast<const uint4*>(vgp[i]);                       \
    vgp[i] += vstep;                                                        \
  }
#define V6_WRITE_V(BUF)                                                     \
  _Pragma("unroll")                                                         \
  for (int u = 0; u < VU; ++u) {                                            \
    const int unit_ = u * NT + tid;                                         \
    const int rp2_ = (unit_ >> 4) * 2;                                      \
    const int c8_ = (unit_ & 15) * 8;                                       \
    union { uint4 u4; u16 h[8]; } va_, vb_;                                 \
    va_.u4 = vst[u * 2];                                                    \
    vb_.u4 = vst[u * 2 + 1];                                                \
    _Pragma("unroll")                                                       \
    for (int j = 0; j < 8; ++j) {                                           \
      const u32 pair_ = (u32)va_.h[j] | ((u32)vb_.h[j] << 16);              \
      *reinterpret_cast<u32*>(&(BUF)[VOFF + vt_byte(c8_ + j, rp2_)]) = pair_; \
    }                                                                       \
  }

  V6_GLDS_K(0)
  V6_LOAD_V()
  V6_WRITE_V(lds0)
  __syncthreads();

  // T5 static form: the second-dispatched half of the workgroup loses
  // VALU arbitration to the older half on every segment; one setprio(1)
  // for it (wave-uniform via readfirstlane) removes its start-of-segment
  // penalty (guide: -0.8..1.5% cycles, never negative)
  if (NW == 8 && __builtin_amdgcn_readfirstlane(threadIdx.x) >= NT / 2)
    __builtin_amdgcn_s_setprio(1);

  int cur = 0;
  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * V6_BN;
    const bool have_next = (t + 1) < n_tiles;
    if (have_next) { V6_GLDS_K(cur ^ 1) V6_LOAD_V() }   // issue early (T14)

    char* kb = cur ? lds1 : lds0;
    char* vbuf = kb + VOFF;
    const bool active = kv0 <= q0w + V6_QBLK - 1;
    const bool need_mask = kv0 + V6_BN > q0w;

    f32x16 s0, s1;
    union PF { bf16x8 v; unsigned u[4]; } pa[4];
    float pm = -INFINITY;
    if (active) {
      // ---- S^T = K @ Q^T over 2 kv-subtiles of 32 ----
#pragma unroll
      for (int r = 0; r < 16; ++r) { s0[r] = 0.f; s1[r] = 0.f; }
#pragma unroll
      for (int kc = 0; kc < 8; ++kc) {
        bf16x8 a0 = *reinterpret_cast<const bf16x8*>(
            &kb[k_byte(low, kc * 16 + hi * 8)]);
        s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, qf[kc].v, s0, 0, 0, 0);
        bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
            &kb[k_byte(32 + low, kc * 16 + hi * 8)]);
        s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, qf[kc].v, s1, 0, 0, 0);
      }
      // ---- scale into exp2 space + causal mask + tile max ----
      if (need_mask) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kvr = kv0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float x0 = s0[r] * scale2;
          if (kvr > qrow) x0 = -1e30f;
          s0[r] = x0;
          float x1 = s1[r] * scale2;
          if (kvr + 32 > qrow) x1 = -1e30f;
          s1[r] = x1;
          pm = fmaxf(pm, fmaxf(x0, x1));
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          s0[r] *= scale2;
          s1[r] *= scale2;
          pm = fmaxf(pm, fmaxf(s0[r], s1[r]));
        }
      }
      pm = fmaxf(pm, __shfl_xor(pm, 32, 64));   // merge q-row halves
    }
    // stage tile t+1 now: QK^T covered the load latency, and writing here
    // (instead of after the P repack) ends vst's live range before the
    // softmax (the 252-VGPR scratch parking came from carrying it there)
    if (have_next) {
      char* nb = cur ? lds0 : lds1;
      V6_WRITE_V(nb)
    }
    if (active) {
      // ---- defer-rescale (T13, THR=8 in exp2 space) ----
      if (!__all(pm - m_run <= 8.0f)) {
        const float mn = fmaxf(m_run, pm);
        const float alpha = (m_run == -INFINITY) ? 0.f : exp2_raw(m_run - mn);
        m_run = mn;
        l_run *= alpha;
#pragma unroll
        for (int ds = 0; ds < 4; ++ds)
#pragma unroll
          for (int r = 0; r < 16; ++r) oacc[ds][r] *= alpha;
      }
      // ---- P = exp2(s - m), row sum ----
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const float p0 = exp2_raw(s0[r] - m_run);
        const float p1 = exp2_raw(s1[r] - m_run);
        s0[r] = p0;
        s1[r] = p1;
        psum += p0 + p1;
      }
      psum += __shfl_xor(psum, 32, 64);
      l_run += psum;

      // ---- T12 repack: P f32 regs -> bf16 MFMA fragments, in-register ----
      // chunk c of 16 kv: lane holds P[kv = c*16 + hi*8 + j][qrow]
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
        for (int cc = 0; cc < 2; ++cc) {
          const int pb = cc * 8;
          unsigned a0, b0, a1, b1;
          if (sub == 0) {
            a0 = cvt_pk_bf16(s0[pb + 0], s0[pb + 1]);
            b0 = cvt_pk_bf16(s0[pb + 4], s0[pb + 5]);
            a1 = cvt_pk_bf16(s0[pb + 2], s0[pb + 3]);
            b1 = cvt_pk_bf16(s0[pb + 6], s0[pb + 7]);
          } else {
            a0 = cvt_pk_bf16(s1[pb + 0], s1[pb + 1]);
            b0 = cvt_pk_bf16(s1[pb + 4], s1[pb + 5]);
            a1 = cvt_pk_bf16(s1[pb + 2], s1[pb + 3]);
            b1 = cvt_pk_bf16(s1[pb + 6], s1[pb + 7]);
          }
          auto r02 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
          auto r13 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
          PF f;
          f.u[0] = r02[0];
          f.u[1] = r13[0];
          f.u[2] = r02[1];
          f.u[3] = r13[1];
          pa[sub * 2 + cc] = f;
        }
      }

      // ---- O^T += V^T @ P over 4 kv-chunks x 4 d-subtiles ----
#pragma unroll
      for (int c = 0; c < 4; ++c) {
#pragma unroll
        for (int ds = 0; ds < 4; ++ds) {
          bf16x8 vf = *reinterpret_cast<const bf16x8*>(
              &vbuf[vt_byte(ds * 32 + low, c * 16 + hi * 8)]);
          oacc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              vf, pa[c].v, oacc[ds], 0, 0, 0);
        }
      }
    }
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: O /= l, store bf16; LSE in natural-log space ----
  const float inv_l = 1.0f / l_run;
  u16* orow = out + (((long)b * n_heads + h) * S + qrow) * ATTN_D;
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int d0 = 8 * g + 4 * hi + 32 * ds;
      const unsigned w0 = cvt_pk_bf16(oacc[ds][4 * g + 0] * inv_l,
                                      oacc[ds][4 * g + 1] * inv_l);
      const unsigned w1 = cvt_pk_bf16(oacc[ds][4 * g + 2] * inv_l,
                                      oacc[ds][4 * g + 3] * inv_l);
      uint2 wv;
      wv.x = w0;
      wv.y = w1;
      *reinterpret_cast<uint2*>(orow + d0) = wv;
    }
  }
  if (hi == 0)
    lse[((long)b * n_heads + h) * S + qrow] =
        m_run * 0.6931471805599453f + __logf(l_run);
}

__global__ __launch_bounds__(512) void attn_fwd_v7_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, u16* __restrict__ out,
    float* __restrict__ lse,
    long q_sb, long q_sh, long q_ss,
    long k_sb, long k_sh, long k_ss,
    long v_sb, long v_sh, long v_ss,
    int n_heads, int gqa_group, int S, float scale2);

extern "C" int attn_fwd(void* stream, const void* q, const void* k,
                        const void* v, void* out, void* lse,
                        long q_sb, long q_sh, long q_ss,
                        long k_sb, long k_sh, long k_ss,
                        long v_sb, long v_sh, long v_ss,
                        int batch, int n_heads, int n_kv_heads, int S,
                        float scale) {
  if (S <= 0 || n_heads <= 0 || batch <= 0 || n_kv_heads <= 0 ||
      n_heads % n_kv_heads != 0)
    return -1;
  if (S % 128 == 0) {
    // BM=256 preferred: halves the K/V staging traffic vs BM=128 (each kv
    // tile is read by half as many workgroups); measured 344 vs 587 us at
    // B1H32S4096. BM=128 covers the S%128 shapes.
    const float scale2 = scale * 1.4426950408889634f;   // fold log2(e)
    if (S % 256 == 0) {
      // v7: 3-deep all-glds pipeline, single raw barrier per tile
      // (299 us vs v6's 346 at B1H32S4096; aotriton 425)
      dim3 grid(S / 256, n_heads, batch), block(512);
      hipLaunchKernelGGL(attn_fwd_v7_kernel, grid, block, 0,
                         reinterpret_cast<hipStream_t>(stream),
                         (const u16*)q, (const u16*)k, (const u16*)v,
                         (u16*)out, (float*)lse, q_sb, q_sh, q_ss,
                         k_sb, k_sh, k_ss, v_sb, v_sh, v_ss, n_heads,
                         n_heads / n_kv_heads, S, scale2);
      return 0;
    }
    dim3 grid(S / 128, n_heads, batch), block(256);
    hipLaunchKernelGGL((attn_fwd_v6_kernel<4>), grid, block, 0,
                       reinterpret_cast<hipStream_t>(stream),
                       (const u16*)q, (const u16*)k, (const u16*)v, (u16*)out,
                       (float*)lse, q_sb, q_sh, q_ss, k_sb, k_sh, k_ss,
                       v_sb, v_sh, v_ss, n_heads, n_heads / n_kv_heads, S,
                       scale2);
    return 0;
  }
  if (S % ATTN_BM != 0 || n_kv_heads != n_heads) return -1;  // v5: no GQA
  dim3 grid(S / ATTN_BM, n_heads, batch), block(256);
  hipLaunchKernelGGL(attn_fwd_kernel, grid, block, 0,
                     reinterpret_cast<hipStream_t>(stream),
                     (const u16*)q, (const u16*)k, (const u16*)v, (u16*)out,
                     (float*)lse, q_sb, q_sh, q_ss, k_sb, k_sh, k_ss,
                     v_sb, v_sh, v_ss, n_heads, S, scale);
  return 0;
}

// 8-wave instantiation kept callable for A/B benchmarking
extern "C" int attn_fwd_nw8(void* stream, const void* q, const void* k,
                            const void* v, void* out, void* lse,
                            long q_sb, long q_sh, long q_ss,
                            long k_sb, long k_sh, long k_ss,
                            long v_sb, long v_sh, long v_ss,
                            int batch, int n_heads, int n_kv_heads, int S,
                            float scale) {
  if (S <= 0 || S % 256 != 0 || n_heads % n_kv_heads != 0) return -1;
  dim3 grid(S / 256, n_heads, batch), block(512);
  const float scale2 = scale * 1.4426950408889634f;
  hipLaunchKernelGGL((attn_fwd_v6_kernel<8>), grid, block, 0,
                     reinterpret_cast<hipStream_t>(stream),
                     (const u16*)q, (const u16*)k, (const u16*)v, (u16*)out,
                     (float*)lse, q_sb, q_sh, q_ss, k_sb, k_sh, k_ss,
                     v_sb, v_sh, v_ss, n_heads, n_heads / n_kv_heads, S,
                     scale2);
  return 0;
}

// v5 kept callable for A/B benchmarking (scripts/attnbench.py v5)
extern "C" int attn_fwd_v5(void* stream, const void* q, const void* k,
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

// ===========================================================================
// v6 flash-attention BACKWARD (causal, bf16, D=128, GQA).
//
// Standard flash backward split into two walks plus a delta precompute:
//   delta[q]   = rowsum(dO[q] * O[q])                       (attn_delta)
//   P          = exp2(S*scale*log2e - lse*log2e)            (recomputed)
//   dV[kv]    += P^T @ dO        dK[kv] += scale * dS^T @ Q (attn_bwd_dkdv)
//   dS         = P o (dP - delta),  dP = dO @ V^T
//   dQ[q]     += scale * dS @ K                             (attn_bwd_dq)
//
// No online softmax state (lse is known), so the backward is simpler per
// tile than the forward. Both kernels reuse the v6 layout machinery:
// operand-swapped 32x32x16 MFMAs keep the OWNED axis lane-local (q rows in
// dq, kv rows in dkdv), P/dS repack stays in registers via
// cvt_pk_bf16_f32 + permlane32_swap, row-major images use the k_byte
// swizzle, transposed images the vt_byte image. GQA: dq walks q heads;
// dkdv owns a kv head and loops the group's q heads, accumulating dK/dV
// in registers across the loop.
// ===========================================================================

__global__ __launch_bounds__(256) void attn_delta_kernel(
    const u16* __restrict__ dout, const u16* __restrict__ o,
    float* __restrict__ delta, long n_rows) {
  // one 16-lane group per row; 8 elems per thread
  const long row = (long)blockIdx.x * 16 + (threadIdx.x >> 4);
  if (row >= n_rows) return;
  const int e8 = (threadIdx.x & 15) * 8;
  union { uint4 u; u16 h[8]; } a, b;
  a.u = *reinterpret_cast<const uint4*>(dout + row * ATTN_D + e8);
  b.u = *reinterpret_cast<const uint4*>(o + row * ATTN_D + e8);
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    acc += bf2f_(a.h[j]) * bf2f_(b.h[j]);
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) acc += __shfl_xor(acc, off, 64);
  if ((threadIdx.x & 15) == 0) delta[row] = acc;
}

// ---------------------------------------------------------------------------
// dQ kernel: one 512-thread WG owns 256 q rows (8 waves x 32, q lane-local).
// Per 64-kv tile: S^T = K@Q^T (16 MFMA), dP^T = V@dO^T (16), dS^T packed,
// dQ^T += K^T @ dS (16). LDS per buffer: K rows (k_byte) + K^T (vt image)
// + V rows (k_byte) = 50.4 KB, double-buffered.
// ---------------------------------------------------------------------------

#define BQ_K_U16 (V6_BN * ATTN_D)                      // K row image
#define BQ_KT_U16 (ATTN_D * (VT_PITCH_B / 2) + 128)    // K^T image
#define BQ_V_U16 (V6_BN * ATTN_D)                      // V row image
#define BQ_BUF_U16 (BQ_K_U16 + BQ_KT_U16 + BQ_V_U16)

__global__ __launch_bounds__(512) void attn_bwd_dq_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, const u16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    u16* __restrict__ dq,
    long q_sb, long q_sh, long q_ss,
    long k_sb, long k_sh, long k_ss,
    long v_sb, long v_sh, long v_ss,
    int n_heads, int gqa_group, int S, float scale) {
  // v7-style staging: K rows + V rows by glds into 3-slot rings, K^T
  // built LDS->LDS at tile end, ONE raw barrier per tile, counted vmcnt.
  constexpr int NT = 512;
  const int qb = gridDim.x - 1 - blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / gqa_group;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int low = lane & 31;
  const int hi = lane >> 5;
  const float scale2 = scale * 1.4426950408889634f;

  __shared__ __attribute__((aligned(16))) u16
      lds[V7_RING * 2 * V6_K_U16 + 2 * V6_V_U16];
#define DQ_K(SLOT) (&lds[(SLOT) * V6_K_U16])
#define DQ_VR(SLOT) (&lds[(V7_RING + (SLOT)) * V6_K_U16])
#define DQ_KT(P) (reinterpret_cast<char*>(&lds[2 * V7_RING * V6_K_U16]) \
                  + (P) * (V6_V_U16 * 2))

  const int q0w = qb * 256 + wid * 32;
  const int qrow = q0w + low;

  union F8 { bf16x8 v; uint4 u; u16 h[8]; };
  const u16* qptr = q + (long)b * q_sb + (long)h * q_sh + (long)qrow * q_ss;
  const u16* doptr = dout + (((long)b * n_heads + h) * S + qrow) * ATTN_D;
  F8 qf[8], dof[8];
#pragma unroll
  for (int kc = 0; kc < 8; ++kc) {
    qf[kc].u = *reinterpret_cast<const uint4*>(qptr + kc * 16 + hi * 8);
    dof[kc].u = *reinterpret_cast<const uint4*>(doptr + kc * 16 + hi * 8);
  }
  const long lrow = ((long)b * n_heads + h) * S + qrow;
  const float lse2 = lse[lrow] * 1.4426950408889634f;
  const float dlt = delta[lrow];

  f32x16 dqacc[4];
#pragma unroll
  for (int ds = 0; ds < 4; ++ds)
#pragma unroll
    for (int r = 0; r < 16; ++r) dqacc[ds][r] = 0.f;

  const u16* kbase = k + (long)b * k_sb + (long)hkv * k_sh;
  const u16* vbase = v + (long)b * v_sb + (long)hkv * v_sh;
  const int n_tiles = (qb + 1) * 4;

  const u16* kgp[2];
  const u16* vgp[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int idx = i * NT + tid;
    const int L = idx * 16;
    const int kr = L >> 8;
    const int kin = (((L & 255) ^ ((kr & 15) << 4)) >> 1);
    kgp[i] = kbase + (long)kr * k_ss + kin;
    vgp[i] = vbase + (long)kr * v_ss + kin;
  }
  const long kstep = (long)V6_BN * k_ss;
  const long vstep = (long)V6_BN * v_ss;
  const int lslot = (tid & ~63) * 8;

#define DQ_GLDS(SLOT)                                                       \
  _Pragma("unroll")                                                         \
  for (int i = 0; i < 2; ++i) {                                             \
    __builtin_amdgcn_global_load_lds(                                       \
        (const u32*)kgp[i], (u32*)&DQ_K(SLOT)[i * NT * 8 + lslot],          \
        16, 0, 0);                                                          \
    kgp[i] += kstep;                                                        \
    __builtin_amdgcn_global_load_lds(                                       \
        (const u32*)vgp[i], (u32*)&DQ_VR(SLOT)[i * NT * 8 + lslot],         \
        16, 0, 0);                                                          \
    vgp[i] += vstep;                                                        \
  }

  const int t_rp2 = (tid >> 4) * 2;
  const int t_c8 = (tid & 15) * 8;
#define DQ_TRANSPOSE(SRCSLOT, DST)                                          \
  {                                                                         \
    const char* srcb = reinterpret_cast<const char*>(DQ_K(SRCSLOT));        \
    union { uint4 u4; u16 h[8]; } va_, vb_;                                 \
    va_.u4 = *reinterpret_cast<const uint4*>(&srcb[k_byte(t_rp2, t_c8)]);   \
    vb_.u4 = *reinterpret_cast<const uint4*>(                               \
        &srcb[k_byte(t_rp2 + 1, t_c8)]);                                    \
    _Pragma("unroll")                                                       \
    for (int j = 0; j < 8; ++j) {                                           \
      const u32 pair_ = (u32)va_.h[j] | ((u32)vb_.h[j] << 16);              \
      *reinterpret_cast<u32*>(&(DST)[vt_byte(t_c8 + j, t_rp2)]) = pair_;    \
    }                                                                       \
  }

  DQ_GLDS(0)
  if (n_tiles > 1) DQ_GLDS(1)
  if (n_tiles > 1) v7_wait_vmcnt4(); else v7_wait_vmcnt0();
  v7_barrier();
  DQ_TRANSPOSE(0, DQ_KT(0))

  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * V6_BN;
    const bool have2 = (t + 2) < n_tiles;
    if (have2) DQ_GLDS((t + 2) % V7_RING)

    char* kb = reinterpret_cast<char*>(DQ_K(t % V7_RING));
    char* vb = reinterpret_cast<char*>(DQ_VR(t % V7_RING));
    char* ktb = DQ_KT(t & 1);
    const bool active = kv0 <= q0w + 31;
    const bool need_mask = kv0 + V6_BN > q0w;

    union PF { bf16x8 v; unsigned u[4]; } pds[4];
    if (active) {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        f32x16 sx, dp;
#pragma unroll
        for (int r = 0; r < 16; ++r) { sx[r] = 0.f; dp[r] = 0.f; }
#pragma unroll
        for (int kc = 0; kc < 8; ++kc) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &kb[k_byte(sub * 32 + low, kc * 16 + hi * 8)]);
          sx = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, qf[kc].v, sx,
                                                       0, 0, 0);
          bf16x8 a2 = *reinterpret_cast<const bf16x8*>(
              &vb[k_byte(sub * 32 + low, kc * 16 + hi * 8)]);
          dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a2, dof[kc].v, dp,
                                                       0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float x = sx[r] * scale2 - lse2;
          if (need_mask) {
            const int kvr = kv0 + sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
            if (kvr > qrow) x = -1e30f;
          }
          const float p = exp2_raw(x);
          sx[r] = p * (dp[r] - dlt) * scale;
        }
#pragma unroll
        for (int cc = 0; cc < 2; ++cc) {
          const int pb = cc * 8;
          unsigned a0 = cvt_pk_bf16(sx[pb + 0], sx[pb + 1]);
          unsigned b0 = cvt_pk_bf16(sx[pb + 4], sx[pb + 5]);
          unsigned a1 = cvt_pk_bf16(sx[pb + 2], sx[pb + 3]);
          unsigned b1 = cvt_pk_bf16(sx[pb + 6], sx[pb + 7]);
          auto r02 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
          auto r13 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
          PF f;
          f.u[0] = r02[0];
          f.u[1] = r13[0];
          f.u[2] = r02[1];
          f.u[3] = r13[1];
          pds[sub * 2 + cc] = f;
        }
      }
#pragma unroll
      for (int c = 0; c < 4; ++c) {
#pragma unroll
        for (int ds = 0; ds < 4; ++ds) {
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(
              &ktb[vt_byte(ds * 32 + low, c * 16 + hi * 8)]);
          dqacc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              kf, pds[c].v, dqacc[ds], 0, 0, 0);
        }
      }
    }
    if (t + 1 < n_tiles) {
      if (have2) v7_wait_vmcnt4(); else v7_wait_vmcnt0();
      DQ_TRANSPOSE((t + 1) % V7_RING, DQ_KT((t + 1) & 1))
    }
    v7_barrier();
  }

  u16* dqrow = dq + (((long)b * n_heads + h) * S + qrow) * ATTN_D;
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int d0 = 8 * g + 4 * hi + 32 * ds;
      const unsigned w0 = cvt_pk_bf16(dqacc[ds][4 * g + 0],
                                      dqacc[ds][4 * g + 1]);
      const unsigned w1 = cvt_pk_bf16(dqacc[ds][4 * g + 2],
                                      dqacc[ds][4 * g + 3]);
      uint2 wv;
      wv.x = w0;
      wv.y = w1;
      *reinterpret_cast<uint2*>(dqrow + d0) = wv;
    }
  }
}

// ---------------------------------------------------------------------------
// dV kernel: one 512-thread WG owns 256 kv rows (8 waves x 32, kv
// lane-local); loops the GQA group's q heads and their q tiles.
//   S^T tile: C[q32][kv32] = Q @ K^T   (K fragments persistent in registers)
//   P = 2^(s*scale2 - lse2[q]) (masked on the diagonal band)
//   dV^T += dO^T @ P: C[d][kv] via A = dO^T image, B = P-fragment repack
// LDS per buffer: Q row image + dO^T image + lse tile.
// ---------------------------------------------------------------------------

#define BV_Q_U16 (V6_BN * ATTN_D)
#define BV_DOT_U16 (ATTN_D * (VT_PITCH_B / 2) + 128)
#define BV_LSE_U16 128                      // 64 f32
#define BV_BUF_U16 (BV_Q_U16 + BV_DOT_U16 + BV_LSE_U16)

__global__ __launch_bounds__(512) void attn_bwd_dv_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ dout, const float* __restrict__ lse,
    u16* __restrict__ dv,
    long q_sb, long q_sh, long q_ss,
    long k_sb, long k_sh, long k_ss,
    int n_heads, int n_kv_heads, int S, float scale) {
  const int kvb = blockIdx.x;                // kv block of 256 (asc = heavy 1st)
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int gqa_group = n_heads / n_kv_heads;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int low = lane & 31;
  const int hi = lane >> 5;
  const float scale2 = scale * 1.4426950408889634f;

  __shared__ __attribute__((aligned(16))) u16 lds[2][BV_BUF_U16];
  char* lds0 = reinterpret_cast<char*>(&lds[0][0]);
  char* lds1 = reinterpret_cast<char*>(&lds[1][0]);
  const int DOTOFF = BV_Q_U16 * 2;
  const int LSEOFF = (BV_Q_U16 + BV_DOT_U16) * 2;

  const int kv0w = kvb * 256 + wid * 32;
  const int kvrow = kv0w + low;              // this lane's kv row

  // K row in registers (persistent across the whole WG lifetime)
  union F8 { bf16x8 v; uint4 u; u16 h[8]; };
  const u16* kptr = k + (long)b * k_sb + (long)hkv * k_sh + (long)kvrow * k_ss;
  F8 kf[8];
#pragma unroll
  for (int kc = 0; kc < 8; ++kc)
    kf[kc].u = *reinterpret_cast<const uint4*>(kptr + kc * 16 + hi * 8);

  f32x16 dvacc[4];
#pragma unroll
  for (int ds = 0; ds < 4; ++ds)
#pragma unroll
    for (int r = 0; r < 16; ++r) dvacc[ds][r] = 0.f;

  const int qt0 = kvb * 4;                   // first causal q tile
  const int qtn = S / 64;
  const int wu64 = __builtin_amdgcn_readfirstlane(tid >> 6);
  uint4 dtst[2];

  for (int g = 0; g < gqa_group; ++g) {
    const int h = hkv * gqa_group + g;
    const u16* qbase = q + (long)b * q_sb + (long)h * q_sh;
    const u16* dobase = dout + (((long)b * n_heads + h) * S) * ATTN_D;
    const float* lbase = lse + ((long)b * n_heads + h) * S;

    // per-thread staging pointers for this head
    const u16* qgp[2];
    const u16* dtp[2];
    {
      const int idx0 = tid, idx1 = 512 + tid;
      const int L0 = idx0 * 16, L1 = idx1 * 16;
      const int r0 = L0 >> 8, r1 = L1 >> 8;
      const int in0 = (((L0 & 255) ^ ((r0 & 15) << 4)) >> 1);
      const int in1 = (((L1 & 255) ^ ((r1 & 15) << 4)) >> 1);
      qgp[0] = qbase + (long)(qt0 * 64 + r0) * q_ss + in0;
      qgp[1] = qbase + (long)(qt0 * 64 + r1) * q_ss + in1;
      const int rp2 = (tid >> 4) * 2;
      const int c8 = (tid & 15) * 8;
      dtp[0] = dobase + (long)(qt0 * 64 + rp2) * ATTN_D + c8;
      dtp[1] = dobase + (long)(qt0 * 64 + rp2 + 1) * ATTN_D + c8;
    }
    const long qstep = (long)64 * q_ss;
    const long dostep = (long)64 * ATTN_D;

#define BV_STAGE_IN(BUFI, QT)                                               \
    {                                                                       \
      _Pragma("unroll")                                                     \
      for (int i = 0; i < 2; ++i) {                                         \
        __builtin_amdgcn_global_load_lds(                                   \
            (const u32*)qgp[i], (u32*)&lds[BUFI][(i * 512 + wu64 * 64) * 8],\
            16, 0, 0);                                                      \
        qgp[i] += qstep;                                                    \
      }                                                                     \
      dtst[0] = *reinterpret_cast<const uint4*>(dtp[0]);                    \
      dtst[1] = *reinterpret_cast<const uint4*>(dtp[1]);                    \
      dtp[0] += dostep;                                                     \
      dtp[1] += dostep;                                                     \
      if (tid < 64)                                                         \
        *reinterpret_cast<float*>(                                          \
            &lds[BUFI][LSEOFF / 2 + tid * 2]) =                             \
            lbase[(QT) * 64 + tid] * 1.4426950408889634f;                   \
    }
#define BV_WRITE_DOT(BUF)                                                   \
    {                                                                       \
      const int rp2_ = (tid >> 4) * 2;                                      \
      const int c8_ = (tid & 15) * 8;                                       \
      union { uint4 u4; u16 h[8]; } va_, vb_;                               \
      va_.u4 = dtst[0];                                                     \
      vb_.u4 = dtst[1];                                                     \
      _Pragma("unroll")                                                     \
      for (int j = 0; j < 8; ++j) {                                         \
        const u32 pair_ = (u32)va_.h[j] | ((u32)vb_.h[j] << 16);            \
        *reinterpret_cast<u32*>(&(BUF)[DOTOFF + vt_byte(c8_ + j, rp2_)]) =  \
            pair_;                                                          \
      }                                                                     \
    }

    BV_STAGE_IN(0, qt0)
    BV_WRITE_DOT(lds0)
    __syncthreads();

    int cur = 0;
    for (int qt = qt0; qt < qtn; ++qt) {
      const bool have_next = (qt + 1) < qtn;
      if (have_next) BV_STAGE_IN(cur ^ 1, qt + 1)

      char* qb_ = cur ? lds1 : lds0;
      char* dob = qb_ + DOTOFF;
      const float* lse2t = reinterpret_cast<const float*>(qb_ + LSEOFF);
      // this wave's kv rows need q >= kv0w; skip tiles fully in the past
      const bool active = qt * 64 + 63 >= kv0w;
      const bool need_mask = qt * 64 < kv0w + 32;

      union PF { bf16x8 v; unsigned u[4]; } pf[4];
      if (active) {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
          f32x16 s;
#pragma unroll
          for (int r = 0; r < 16; ++r) s[r] = 0.f;
#pragma unroll
          for (int kc = 0; kc < 8; ++kc) {
            bf16x8 a = *reinterpret_cast<const bf16x8*>(
                &qb_[k_byte(sub * 32 + low, kc * 16 + hi * 8)]);
            s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, kf[kc].v, s,
                                                        0, 0, 0);
          }
          // P = 2^(s*scale2 - lse2[q]); mask q < kv
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int qr = (r & 3) + 8 * (r >> 2) + 4 * hi;   // q within sub
            float x = s[r] * scale2 - lse2t[sub * 32 + qr];
            if (need_mask && qt * 64 + sub * 32 + qr < kvrow) x = -1e30f;
            s[r] = exp2_raw(x);
          }
          // repack P (C[q][kv]) -> B-fragments [k=q chunk][n=kv]
#pragma unroll
          for (int cc = 0; cc < 2; ++cc) {
            const int pb = cc * 8;
            unsigned a0 = cvt_pk_bf16(s[pb + 0], s[pb + 1]);
            unsigned b0 = cvt_pk_bf16(s[pb + 4], s[pb + 5]);
            unsigned a1 = cvt_pk_bf16(s[pb + 2], s[pb + 3]);
            unsigned b1 = cvt_pk_bf16(s[pb + 6], s[pb + 7]);
            auto r02 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
            auto r13 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
            PF f;
            f.u[0] = r02[0];
            f.u[1] = r13[0];
            f.u[2] = r02[1];
            f.u[3] = r13[1];
            pf[sub * 2 + cc] = f;
          }
        }
      }
      if (have_next) {
        char* nb = cur ? lds0 : lds1;
        BV_WRITE_DOT(nb)
      }
      if (active) {
        // dV^T += dO^T @ P over 4 q chunks x 4 d subtiles
#pragma unroll
        for (int c = 0; c < 4; ++c) {
#pragma unroll
          for (int ds = 0; ds < 4; ++ds) {
            bf16x8 af = *reinterpret_cast<const bf16x8*>(
                &dob[vt_byte(ds * 32 + low, c * 16 + hi * 8)]);
            dvacc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                af, pf[c].v, dvacc[ds], 0, 0, 0);
          }
        }
      }
      __syncthreads();
      cur ^= 1;
    }
  }

  // epilogue: dv [B,HKV,S,D] contiguous
  u16* dvrow = dv + (((long)b * n_kv_heads + hkv) * S + kvrow) * ATTN_D;
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int d0 = 8 * g + 4 * hi + 32 * ds;
      const unsigned w0 = cvt_pk_bf16(dvacc[ds][4 * g + 0],
                                      dvacc[ds][4 * g + 1]);
      const unsigned w1 = cvt_pk_bf16(dvacc[ds][4 * g + 2],
                                      dvacc[ds][4 * g + 3]);
      uint2 wv;
      wv.x = w0;
      wv.y = w1;
      *reinterpret_cast<uint2*>(dvrow + d0) = wv;
    }
  }
}

// ---------------------------------------------------------------------------
// dK kernel: same kv-block walk as dV.
//   S^T: C[q][kv] = Q @ K^T       (K fragments persistent)
//   dP:  C[q][kv] = dO @ V^T      (V fragments persistent)
//   dS = P o (dP - delta[q]) * scale
//   dK^T += Q^T @ dS: C[d][kv] via A = Q^T image, B = dS repack
// LDS per buffer: Q row image + dO row image + Q^T image + lse + delta.
// ---------------------------------------------------------------------------

#define BK_Q_U16 (V6_BN * ATTN_D)
#define BK_DO_U16 (V6_BN * ATTN_D)
#define BK_QT_U16 (ATTN_D * (VT_PITCH_B / 2) + 128)
#define BK_LSE_U16 128
#define BK_DLT_U16 128
#define BK_BUF_U16 (BK_Q_U16 + BK_DO_U16 + BK_QT_U16 + BK_LSE_U16 + BK_DLT_U16)

__global__ __launch_bounds__(512) void attn_bwd_dk_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, const u16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    u16* __restrict__ dk,
    long q_sb, long q_sh, long q_ss,
    long k_sb, long k_sh, long k_ss,
    long v_sb, long v_sh, long v_ss,
    int n_heads, int n_kv_heads, int S, float scale) {
  const int kvb = blockIdx.x;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int gqa_group = n_heads / n_kv_heads;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int low = lane & 31;
  const int hi = lane >> 5;
  const float scale2 = scale * 1.4426950408889634f;

  __shared__ __attribute__((aligned(16))) u16 lds[2][BK_BUF_U16];
  char* lds0 = reinterpret_cast<char*>(&lds[0][0]);
  char* lds1 = reinterpret_cast<char*>(&lds[1][0]);
  const int DOOFF = BK_Q_U16 * 2;
  const int QTOFF = (BK_Q_U16 + BK_DO_U16) * 2;
  const int LSEOFF = (BK_Q_U16 + BK_DO_U16 + BK_QT_U16) * 2;
  const int DLTOFF = LSEOFF + BK_LSE_U16 * 2;

  const int kv0w = kvb * 256 + wid * 32;
  const int kvrow = kv0w + low;

  union F8 { bf16x8 v; uint4 u; u16 h[8]; };
  const u16* kptr = k + (long)b * k_sb + (long)hkv * k_sh + (long)kvrow * k_ss;
  const u16* vptr = v + (long)b * v_sb + (long)hkv * v_sh + (long)kvrow * v_ss;
  F8 kf[8], vf[8];
#pragma unroll
  for (int kc = 0; kc < 8; ++kc) {
    kf[kc].u = *reinterpret_cast<const uint4*>(kptr + kc * 16 + hi * 8);
    vf[kc].u = *reinterpret_cast<const uint4*>(vptr + kc * 16 + hi * 8);
  }

  f32x16 dkacc[4];
#pragma unroll
  for (int ds = 0; ds < 4; ++ds)
#pragma unroll
    for (int r = 0; r < 16; ++r) dkacc[ds][r] = 0.f;

  const int qt0 = kvb * 4;
  const int qtn = S / 64;
  const int wu64 = __builtin_amdgcn_readfirstlane(tid >> 6);
  uint4 qtst[2];

  for (int g = 0; g < gqa_group; ++g) {
    const int h = hkv * gqa_group + g;
    const u16* qbase = q + (long)b * q_sb + (long)h * q_sh;
    const u16* dobase = dout + (((long)b * n_heads + h) * S) * ATTN_D;
    const float* lbase = lse + ((long)b * n_heads + h) * S;
    const float* dbase = delta + ((long)b * n_heads + h) * S;

    const u16* qgp[2];
    const u16* dgp[2];
    const u16* qtp[2];
    {
      const int idx0 = tid, idx1 = 512 + tid;
      const int L0 = idx0 * 16, L1 = idx1 * 16;
      const int r0 = L0 >> 8, r1 = L1 >> 8;
      const int in0 = (((L0 & 255) ^ ((r0 & 15) << 4)) >> 1);
      const int in1 = (((L1 & 255) ^ ((r1 & 15) << 4)) >> 1);
      qgp[0] = qbase + (long)(qt0 * 64 + r0) * q_ss + in0;
      qgp[1] = qbase + (long)(qt0 * 64 + r1) * q_ss + in1;
      dgp[0] = dobase + (long)(qt0 * 64 + r0) * ATTN_D + in0;
      dgp[1] = dobase + (long)(qt0 * 64 + r1) * ATTN_D + in1;
      const int rp2 = (tid >> 4) * 2;
      const int c8 = (tid & 15) * 8;
      qtp[0] = qbase + (long)(qt0 * 64 + rp2) * q_ss + c8;
      qtp[1] = qbase + (long)(qt0 * 64 + rp2 + 1) * q_ss + c8;
    }
    const long qstep = (long)64 * q_ss;
    const long dostep = (long)64 * ATTN_D;

#define BK_STAGE_IN(BUFI, QT)                                               \
    {                                                                       \
      _Pragma("unroll")                                                     \
      for (int i = 0; i < 2; ++i) {                                         \
        __builtin_amdgcn_global_load_lds(                                   \
            (const u32*)qgp[i], (u32*)&lds[BUFI][(i * 512 + wu64 * 64) * 8],\
            16, 0, 0);                                                      \
        qgp[i] += qstep;                                                    \
        __builtin_amdgcn_global_load_lds(                                   \
            (const u32*)dgp[i],                                             \
            (u32*)&lds[BUFI][DOOFF / 2 + (i * 512 + wu64 * 64) * 8],        \
            16, 0, 0);                                                      \
        dgp[i] += dostep;                                                   \
      }                                                                     \
      qtst[0] = *reinterpret_cast<const uint4*>(qtp[0]);                    \
      qtst[1] = *reinterpret_cast<const uint4*>(qtp[1]);                    \
      qtp[0] += qstep;                                                      \
      qtp[1] += qstep;                                                      \
      if (tid < 64)                                                         \
        *reinterpret_cast<float*>(&lds[BUFI][LSEOFF / 2 + tid * 2]) =       \
            lbase[(QT) * 64 + tid] * 1.4426950408889634f;                   \
      else if (tid < 128)                                                   \
        *reinterpret_cast<float*>(                                          \
            &lds[BUFI][DLTOFF / 2 + (tid - 64) * 2]) =                      \
            dbase[(QT) * 64 + tid - 64];                                    \
    }
#define BK_WRITE_QT(BUF)                                                    \
    {                                                                       \
      const int rp2_ = (tid >> 4) * 2;                                      \
      const int c8_ = (tid & 15) * 8;                                       \
      union { uint4 u4; u16 h[8]; } va_, vb_;                               \
      va_.u4 = qtst[0];                                                     \
      vb_.u4 = qtst[1];                                                     \
      _Pragma("unroll")                                                     \
      for (int j = 0; j < 8; ++j) {                                         \
        const u32 pair_ = (u32)va_.h[j] | ((u32)vb_.h[j] << 16);            \
        *reinterpret_cast<u32*>(&(BUF)[QTOFF + vt_byte(c8_ + j, rp2_)]) =   \
            pair_;                                                          \
      }                                                                     \
    }

    BK_STAGE_IN(0, qt0)
    BK_WRITE_QT(lds0)
    __syncthreads();

    int cur = 0;
    for (int qt = qt0; qt < qtn; ++qt) {
      const bool have_next = (qt + 1) < qtn;
      if (have_next) BK_STAGE_IN(cur ^ 1, qt + 1)

      char* qb_ = cur ? lds1 : lds0;
      char* dob = qb_ + DOOFF;
      char* qtb = qb_ + QTOFF;
      const float* lse2t = reinterpret_cast<const float*>(qb_ + LSEOFF);
      const float* dltt = reinterpret_cast<const float*>(qb_ + DLTOFF);
      const bool active = qt * 64 + 63 >= kv0w;
      const bool need_mask = qt * 64 < kv0w + 32;

      union PF { bf16x8 v; unsigned u[4]; } pf[4];
      if (active) {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
          f32x16 s, dp;
#pragma unroll
          for (int r = 0; r < 16; ++r) { s[r] = 0.f; dp[r] = 0.f; }
#pragma unroll
          for (int kc = 0; kc < 8; ++kc) {
            bf16x8 a = *reinterpret_cast<const bf16x8*>(
                &qb_[k_byte(sub * 32 + low, kc * 16 + hi * 8)]);
            s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, kf[kc].v, s,
                                                        0, 0, 0);
            bf16x8 a2 = *reinterpret_cast<const bf16x8*>(
                &dob[k_byte(sub * 32 + low, kc * 16 + hi * 8)]);
            dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a2, vf[kc].v, dp,
                                                         0, 0, 0);
          }
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int qr = (r & 3) + 8 * (r >> 2) + 4 * hi;
            float x = s[r] * scale2 - lse2t[sub * 32 + qr];
            if (need_mask && qt * 64 + sub * 32 + qr < kvrow) x = -1e30f;
            const float p = exp2_raw(x);
            s[r] = p * (dp[r] - dltt[sub * 32 + qr]) * scale;
          }
#pragma unroll
          for (int cc = 0; cc < 2; ++cc) {
            const int pb = cc * 8;
            unsigned a0 = cvt_pk_bf16(s[pb + 0], s[pb + 1]);
            unsigned b0 = cvt_pk_bf16(s[pb + 4], s[pb + 5]);
            unsigned a1 = cvt_pk_bf16(s[pb + 2], s[pb + 3]);
            unsigned b1 = cvt_pk_bf16(s[pb + 6], s[pb + 7]);
            auto r02 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
            auto r13 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
            PF f;
            f.u[0] = r02[0];
            f.u[1] = r13[0];
            f.u[2] = r02[1];
            f.u[3] = r13[1];
            pf[sub * 2 + cc] = f;
          }
        }
      }
      if (have_next) {
        char* nb = cur ? lds0 : lds1;
        BK_WRITE_QT(nb)
      }
      if (active) {
#pragma unroll
        for (int c = 0; c < 4; ++c) {
#pragma unroll
          for (int ds = 0; ds < 4; ++ds) {
            bf16x8 af = *reinterpret_cast<const bf16x8*>(
                &qtb[vt_byte(ds * 32 + low, c * 16 + hi * 8)]);
            dkacc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                af, pf[c].v, dkacc[ds], 0, 0, 0);
          }
        }
      }
      __syncthreads();
      cur ^= 1;
    }
  }

  u16* dkrow = dk + (((long)b * n_kv_heads + hkv) * S + kvrow) * ATTN_D;
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int d0 = 8 * g + 4 * hi + 32 * ds;
      const unsigned w0 = cvt_pk_bf16(dkacc[ds][4 * g + 0],
                                      dkacc[ds][4 * g + 1]);
      const unsigned w1 = cvt_pk_bf16(dkacc[ds][4 * g + 2],
                                      dkacc[ds][4 * g + 3]);
      uint2 wv;
      wv.x = w0;
      wv.y = w1;
      *reinterpret_cast<uint2*>(dkrow + d0) = wv;
    }
  }
}



// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

extern "C" int attn_bwd(void* stream, const void* q, const void* k,
                        const void* v, const void* out, const void* dout,
                        const void* lse, void* delta,
                        void* dq, void* dk, void* dv,
                        long q_sb, long q_sh, long q_ss,
                        long k_sb, long k_sh, long k_ss,
                        long v_sb, long v_sh, long v_ss,
                        int batch, int n_heads, int n_kv_heads, int S,
                        float scale) {
  if (S <= 0 || S % 256 != 0 || n_heads <= 0 || batch <= 0 ||
      n_kv_heads <= 0 || n_heads % n_kv_heads != 0)
    return -1;
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  const long n_rows = (long)batch * n_heads * S;
  hipLaunchKernelGGL(attn_delta_kernel,
                     dim3((unsigned)((n_rows + 15) / 16)), dim3(256), 0, st,
                     (const u16*)dout, (const u16*)out, (float*)delta,
                     n_rows);
  {
    dim3 grid(S / 256, n_heads, batch), block(512);
    hipLaunchKernelGGL(attn_bwd_dq_kernel, grid, block, 0, st,
                       (const u16*)q, (const u16*)k, (const u16*)v,
                       (const u16*)dout, (const float*)lse,
                       (const float*)delta, (u16*)dq,
                       q_sb, q_sh, q_ss, k_sb, k_sh, k_ss,
                       v_sb, v_sh, v_ss, n_heads, n_heads / n_kv_heads, S,
                       scale);
  }
  {
    dim3 grid(S / 256, n_kv_heads, batch), block(512);
    hipLaunchKernelGGL(attn_bwd_dv_kernel, grid, block, 0, st,
                       (const u16*)q, (const u16*)k, (const u16*)dout,
                       (const float*)lse, (u16*)dv,
                       q_sb, q_sh, q_ss, k_sb, k_sh, k_ss,
                       n_heads, n_kv_heads, S, scale);
    hipLaunchKernelGGL(attn_bwd_dk_kernel, grid, block, 0, st,
                       (const u16*)q, (const u16*)k, (const u16*)v,
                       (const u16*)dout, (const float*)lse,
                       (const float*)delta, (u16*)dk,
                       q_sb, q_sh, q_ss, k_sb, k_sh, k_ss,
                       v_sb, v_sh, v_ss, n_heads, n_kv_heads, S, scale);
  }
  return 0;
}

// ===========================================================================
// v7 forward: v6's math with a 3-deep ALL-glds staging pipeline.
//
// v6 parks ~49% of wave cycles at waits/barriers: its __syncthreads()
// carries a vmcnt(0) drain (glds in flight) and the V register staging
// serializes a global-load wait into the middle of every tile. v7:
//   * K AND row-major V arrive by glds into a 3-slot ring (issue runs two
//     tiles ahead; nothing in the main loop ever waits vmcnt(0));
//   * the transposed V image is built LDS->LDS (2 b128 reads + 8 paired
//     b32 writes per thread) from the landed row image — no global
//     staging registers at all;
//   * barriers are RAW s_barrier (guide: "3-buffer glds + raw barrier,
//     counted vmcnt only"), two per tile: one after the counted
//     vmcnt making slot t+1 visible for the transpose, one at tile end
//     guarding ring reuse.
// ===========================================================================

__global__ __launch_bounds__(512) void attn_fwd_v7_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, u16* __restrict__ out,
    float* __restrict__ lse,
    long q_sb, long q_sh, long q_ss,
    long k_sb, long k_sh, long k_ss,
    long v_sb, long v_sh, long v_ss,
    int n_heads, int gqa_group, int S, float scale2) {
  constexpr int NT = 512;
  const int qb = gridDim.x - 1 - blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / gqa_group;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int low = lane & 31;
  const int hi = lane >> 5;

  // ring: K rows (swizzled image) + V rows (swizzled image), both glds;
  // 2 transposed-V images built locally
  __shared__ __attribute__((aligned(16))) u16
      lds[V7_RING * 2 * V6_K_U16 + 2 * V6_V_U16];
  // slot addressing by arithmetic (pointer arrays spill under pressure)
#define LDSK(SLOT) (&lds[(SLOT) * V6_K_U16])
#define LDSVR(SLOT) (&lds[(V7_RING + (SLOT)) * V6_K_U16])
#define LDSVT(P) (reinterpret_cast<char*>(&lds[2 * V7_RING * V6_K_U16]) \
                  + (P) * (V6_V_U16 * 2))

  const int q0w = qb * 256 + wid * 32;
  const int qrow = q0w + low;

  union F8 { bf16x8 v; uint4 u; u16 h[8]; };
  const u16* qptr = q + (long)b * q_sb + (long)h * q_sh + (long)qrow * q_ss;
  F8 qf[8];
#pragma unroll
  for (int kc = 0; kc < 8; ++kc)
    qf[kc].u = *reinterpret_cast<const uint4*>(qptr + kc * 16 + hi * 8);

  f32x16 oacc[4];
#pragma unroll
  for (int ds = 0; ds < 4; ++ds)
#pragma unroll
    for (int r = 0; r < 16; ++r) oacc[ds][r] = 0.f;
  float m_run = -INFINITY, l_run = 0.f;

  const u16* kbase = k + (long)b * k_sb + (long)hkv * k_sh;
  const u16* vbase = v + (long)b * v_sb + (long)hkv * v_sh;
  const int n_tiles = (qb + 1) * 4;

  // glds source pointers (swizzle-inverted, advanced per issue)
  const u16* kgp[2];
  const u16* vgp[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int idx = i * NT + tid;
    const int L = idx * 16;
    const int kr = L >> 8;
    const int kin = (((L & 255) ^ ((kr & 15) << 4)) >> 1);
    kgp[i] = kbase + (long)kr * k_ss + kin;
    vgp[i] = vbase + (long)kr * v_ss + kin;
  }
  const long kstep = (long)V6_BN * k_ss;
  const long vstep = (long)V6_BN * v_ss;
  const int lslot = (tid & ~63) * 8;          // wave-uniform u16 base

#define V7_GLDS(SLOT)                                                       \
  _Pragma("unroll")                                                         \
  for (int i = 0; i < 2; ++i) {                                             \
    __builtin_amdgcn_global_load_lds(                                       \
        (const u32*)kgp[i], (u32*)&LDSK(SLOT)[i * NT * 8 + lslot],          \
        16, 0, 0);                                                          \
    kgp[i] += kstep;                                                        \
    __builtin_amdgcn_global_load_lds(                                       \
        (const u32*)vgp[i], (u32*)&LDSVR(SLOT)[i * NT * 8 + lslot],         \
        16, 0, 0);                                                          \
    vgp[i] += vstep;                                                        \
  }

  // LDS->LDS transpose of one landed V row-image into a vt image
  const int t_rp2 = (tid >> 4) * 2;
  const int t_c8 = (tid & 15) * 8;
#define V7_TRANSPOSE(SRCSLOT, DST)                                          \
  {                                                                         \
    const char* srcb = reinterpret_cast<const char*>(LDSVR(SRCSLOT));       \
    union { uint4 u4; u16 h[8]; } va_, vb_;                                 \
    va_.u4 = *reinterpret_cast<const uint4*>(&srcb[k_byte(t_rp2, t_c8)]);   \
    vb_.u4 = *reinterpret_cast<const uint4*>(                               \
        &srcb[k_byte(t_rp2 + 1, t_c8)]);                                    \
    _Pragma("unroll")                                                       \
    for (int j = 0; j < 8; ++j) {                                           \
      const u32 pair_ = (u32)va_.h[j] | ((u32)vb_.h[j] << 16);              \
      *reinterpret_cast<u32*>(&(DST)[vt_byte(t_c8 + j, t_rp2)]) = pair_;    \
    }                                                                       \
  }

  // prologue: issue tiles 0 and 1; build Vt[0] once tile 0 lands
  V7_GLDS(0)
  if (n_tiles > 1) V7_GLDS(1)
  if (n_tiles > 1) v7_wait_vmcnt4(); else v7_wait_vmcnt0();
  v7_barrier();
  V7_TRANSPOSE(0, LDSVT(0))
  // Vt[0] visible to every wave at the first in-loop barrier below

  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * V6_BN;
    const bool have2 = (t + 2) < n_tiles;
    if (have2) V7_GLDS((t + 2) % V7_RING)

    char* kb = reinterpret_cast<char*>(LDSK(t % V7_RING));
    char* vtb = LDSVT(t & 1);
    const bool active = kv0 <= q0w + 31;
    const bool need_mask = kv0 + V6_BN > q0w;

    f32x16 s0, s1;
    union PF { bf16x8 v; unsigned u[4]; } pa[4];
    float pm = -INFINITY;
    if (active) {
#pragma unroll
      for (int r = 0; r < 16; ++r) { s0[r] = 0.f; s1[r] = 0.f; }
#pragma unroll
      for (int kc = 0; kc < 8; ++kc) {
        bf16x8 a0 = *reinterpret_cast<const bf16x8*>(
            &kb[k_byte(low, kc * 16 + hi * 8)]);
        s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, qf[kc].v, s0, 0, 0, 0);
        bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
            &kb[k_byte(32 + low, kc * 16 + hi * 8)]);
        s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, qf[kc].v, s1, 0, 0, 0);
      }
      if (need_mask) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kvr = kv0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float x0 = s0[r] * scale2;
          if (kvr > qrow) x0 = -1e30f;
          s0[r] = x0;
          float x1 = s1[r] * scale2;
          if (kvr + 32 > qrow) x1 = -1e30f;
          s1[r] = x1;
          pm = fmaxf(pm, fmaxf(x0, x1));
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          s0[r] *= scale2;
          s1[r] *= scale2;
          pm = fmaxf(pm, fmaxf(s0[r], s1[r]));
        }
      }
      pm = fmaxf(pm, __shfl_xor(pm, 32, 64));

      if (!__all(pm - m_run <= 8.0f)) {
        const float mn = fmaxf(m_run, pm);
        const float alpha = (m_run == -INFINITY) ? 0.f : exp2_raw(m_run - mn);
        m_run = mn;
        l_run *= alpha;
#pragma unroll
        for (int ds = 0; ds < 4; ++ds)
#pragma unroll
          for (int r = 0; r < 16; ++r) oacc[ds][r] *= alpha;
      }
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const float p0 = exp2_raw(s0[r] - m_run);
        const float p1 = exp2_raw(s1[r] - m_run);
        s0[r] = p0;
        s1[r] = p1;
        psum += p0 + p1;
      }
      psum += __shfl_xor(psum, 32, 64);
      l_run += psum;

#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
        for (int cc = 0; cc < 2; ++cc) {
          const int pb = cc * 8;
          unsigned a0, b0, a1, b1;
          if (sub == 0) {
            a0 = cvt_pk_bf16(s0[pb + 0], s0[pb + 1]);
            b0 = cvt_pk_bf16(s0[pb + 4], s0[pb + 5]);
            a1 = cvt_pk_bf16(s0[pb + 2], s0[pb + 3]);
            b1 = cvt_pk_bf16(s0[pb + 6], s0[pb + 7]);
          } else {
            a0 = cvt_pk_bf16(s1[pb + 0], s1[pb + 1]);
            b0 = cvt_pk_bf16(s1[pb + 4], s1[pb + 5]);
            a1 = cvt_pk_bf16(s1[pb + 2], s1[pb + 3]);
            b1 = cvt_pk_bf16(s1[pb + 6], s1[pb + 7]);
          }
          auto r02 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
          auto r13 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
          PF f;
          f.u[0] = r02[0];
          f.u[1] = r13[0];
          f.u[2] = r02[1];
          f.u[3] = r13[1];
          pa[sub * 2 + cc] = f;
        }
      }

#pragma unroll
      for (int c = 0; c < 4; ++c) {
#pragma unroll
        for (int ds = 0; ds < 4; ++ds) {
          bf16x8 vf = *reinterpret_cast<const bf16x8*>(
              &vtb[vt_byte(ds * 32 + low, c * 16 + hi * 8)]);
          oacc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              vf, pa[c].v, oacc[ds], 0, 0, 0);
        }
      }
    }
    // end-of-tile: wait slot t+1 (counted — t+2 stays in flight), build
    // the NEXT transposed image (different Vt buffer than this tile's PV
    // reads, so it needs no pre-barrier), then ONE raw barrier guarding
    // ring reuse + Vt visibility for tile t+1
    if (t + 1 < n_tiles) {
      if (have2) v7_wait_vmcnt4(); else v7_wait_vmcnt0();
      V7_TRANSPOSE((t + 1) % V7_RING, LDSVT((t + 1) & 1))
    }
    v7_barrier();
  }

  const float inv_l = 1.0f / l_run;
  u16* orow = out + (((long)b * n_heads + h) * S + qrow) * ATTN_D;
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int d0 = 8 * g + 4 * hi + 32 * ds;
      const unsigned w0 = cvt_pk_bf16(oacc[ds][4 * g + 0] * inv_l,
                                      oacc[ds][4 * g + 1] * inv_l);
      const unsigned w1 = cvt_pk_bf16(oacc[ds][4 * g + 2] * inv_l,
                                      oacc[ds][4 * g + 3] * inv_l);
      uint2 wv;
      wv.x = w0;
      wv.y = w1;
      *reinterpret_cast<uint2*>(orow + d0) = wv;
    }
  }
  if (hi == 0)
    lse[((long)b * n_heads + h) * S + qrow] =
        m_run * 0.6931471805599453f + __logf(l_run);
}

extern "C" int attn_fwd_v7(void* stream, const void* q, const void* k,
                           const void* v, void* out, void* lse,
                           long q_sb, long q_sh, long q_ss,
                           long k_sb, long k_sh, long k_ss,
                           long v_sb, long v_sh, long v_ss,
                           int batch, int n_heads, int n_kv_heads, int S,
                           float scale) {
  if (S <= 0 || S % 256 != 0 || n_heads % n_kv_heads != 0) return -1;
  dim3 grid(S / 256, n_heads, batch), block(512);
  const float scale2 = scale * 1.4426950408889634f;
  hipLaunchKernelGGL(attn_fwd_v7_kernel, grid, block, 0,
                     reinterpret_cast<hipStream_t>(stream),
                     (const u16*)q, (const u16*)k, (const u16*)v, (u16*)out,
                     (float*)lse, q_sb, q_sh, q_ss, k_sb, k_sh, k_ss,
                     v_sb, v_sh, v_ss, n_heads, n_heads / n_kv_heads, S,
                     scale2);
  return 0;
}
