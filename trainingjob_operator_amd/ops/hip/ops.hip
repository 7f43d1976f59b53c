// MI355X (gfx950, CDNA4) training kernels for the Llama workload.
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  * wave64 everywhere: block = 256 threads = 4 waves; cross-lane reduce via
//    __shfl_xor over 64 lanes, cross-wave via a tiny LDS array.
//  * HBM3E-bound elementwise/norm ops use bf16x8 (16 B) vector loads — the
//    guide measures 2.35 -> 4.89 TB/s for RMSNorm from that change alone.
//  * bf16 <-> f32: bf16->f32 is an exact 16-bit shift; f32->bf16 uses the
//    RNE hardware conversion via __float2bfloat16.
//  * All kernels launch with >> 256 workgroups to fill 8 XCDs x 32 CUs.
//  * No CUDA compatibility paths: this file is HIP/CDNA4 only.
//
// These ops are the worker-side compute the reference operator never had
// (reference repo has zero kernels): the managed Llama-3-8B
// benchmark's hot non-GEMM ops, fused to one HBM round-trip each.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>
#include <cstdio>

#define WAVE 64
#define BLOCK 256
#define WAVES_PER_BLOCK (BLOCK / WAVE)

typedef unsigned short u16;
typedef unsigned int u32;

// ---------------------------------------------------------------------------
// bf16x8 helpers
// ---------------------------------------------------------------------------

union BF8 {
  uint4 v;       // 16 bytes
  u16 h[8];      // 8 bf16
};

__device__ __forceinline__ float bf2f(u16 b) {
  union { u32 u; float f; } c;
  c.u = ((u32)b) << 16;
  return c.f;
}

__device__ __forceinline__ u16 f2bf(float f) {
  __hip_bfloat16_raw r = __float2bfloat16(f);  // RNE hardware convert
  return r.x;
}

// ---------------------------------------------------------------------------
// block reduction (sum): wave shfl_xor tree + LDS cross-wave
// ---------------------------------------------------------------------------

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// lds must be a pointer into the block's single __shared__ float[>=WAVES+1]
__device__ __forceinline__ float block_reduce_sum(float v, float* lds) {
  v = wave_reduce_sum(v);
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < WAVES_PER_BLOCK; ++w) total += lds[w];
  __syncthreads();
  return total;
}

// ===========================================================================
// Fused residual-add + RMSNorm, forward.
//   res_out = x + res_in            (res_in optional)
//   y       = res_out * rsqrt(mean(res_out^2) + eps) * w
// One block per row; row cached in registers between the sumsq reduction and
// the normalize pass (H <= 8192 with bf16x8 per-thread chunks).
// ===========================================================================

template <int NITER>
__global__ void rmsnorm_fwd_kernel(
    const uint4* __restrict__ x, const uint4* __restrict__ res_in,
    const uint4* __restrict__ w, uint4* __restrict__ y,
    uint4* __restrict__ res_out, float* __restrict__ rrms_out,
    int H8, float eps) {
  __shared__ float lds[WAVES_PER_BLOCK];
  const long row = blockIdx.x;
  const uint4* xr = x + row * H8;
  const uint4* rr = res_in ? res_in + row * H8 : nullptr;
  uint4* yr = y + row * H8;
  uint4* ror = res_out ? res_out + row * H8 : nullptr;

  float vals[NITER][8];
  float sumsq = 0.f;
#pragma unroll
  for (int it = 0; it < NITER; ++it) {
    const int idx = it * BLOCK + threadIdx.x;
    BF8 a; a.v = xr[idx];
#pragma unroll
    for (int j = 0; j < 8; ++j) vals[it][j] = bf2f(a.h[j]);
    if (rr) {
      BF8 b; b.v = rr[idx];
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[it][j] += bf2f(b.h[j]);
      BF8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o.h[j] = f2bf(vals[it][j]);
      ror[idx] = o.v;
      // re-read the rounded residual so y is computed from exactly what the
      // next layer sees (keeps fwd bit-consistent with the unfused reference)
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[it][j] = bf2f(o.h[j]);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) sumsq += vals[it][j] * vals[it][j];
  }
  const int H = H8 * 8;
  const float total = block_reduce_sum(sumsq, lds);
  const float rrms = rsqrtf(total / (float)H + eps);
  if (threadIdx.x == 0 && rrms_out) rrms_out[row] = rrms;

#pragma unroll
  for (int it = 0; it < NITER; ++it) {
    const int idx = it * BLOCK + threadIdx.x;
    BF8 wv; wv.v = w[idx];
    BF8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o.h[j] = f2bf(vals[it][j] * rrms * bf2f(wv.h[j]));
    yr[idx] = o.v;
  }
}

// ===========================================================================
// RMSNorm backward.
//   xhat = res_out * rrms
//   dxhat = dy * w
//   dx = rrms * (dxhat - xhat * mean(dxhat * xhat)) + dres   (dres optional:
//        the gradient arriving through the residual stream, fused here to
//        save a full elementwise add pass)
//   dw_partial[block] = sum over the block's rows of dy * xhat
//        (plain fp32 stores — no atomics; a tiny reduce kernel finishes it.
//         The atomic version measured 433 us/call on MI355X, 17x the fwd.)
// Each block processes DW_ROWS rows serially, keeping its dw accumulator in
// registers across them.
// ===========================================================================

#define DW_ROWS 4

template <int NITER>
__global__ void rmsnorm_bwd_kernel(
    const uint4* __restrict__ dy, const uint4* __restrict__ res_out,
    const uint4* __restrict__ w, const float* __restrict__ rrms_in,
    const uint4* __restrict__ dres, uint4* __restrict__ dx,
    float* __restrict__ dw_partial, long n_rows, int H8) {
  __shared__ float lds[WAVES_PER_BLOCK];
  const int H = H8 * 8;
  float wv[NITER][8];
#pragma unroll
  for (int it = 0; it < NITER; ++it) {
    BF8 a; a.v = w[it * BLOCK + threadIdx.x];
#pragma unroll
    for (int j = 0; j < 8; ++j) wv[it][j] = bf2f(a.h[j]);
  }
  float dwacc[NITER][8] = {};

  const long row0 = (long)blockIdx.x * DW_ROWS;
  for (long row = row0; row < row0 + DW_ROWS && row < n_rows; ++row) {
    const uint4* dyr = dy + row * H8;
    const uint4* xr = res_out + row * H8;
    uint4* dxr = dx + row * H8;
    const float rrms = rrms_in[row];

    float xv[NITER][8], dxh[NITER][8], dyv[NITER][8];
    float dot = 0.f;
#pragma unroll
    for (int it = 0; it < NITER; ++it) {
      const int idx = it * BLOCK + threadIdx.x;
      BF8 a; a.v = xr[idx];
      BF8 d; d.v = dyr[idx];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        xv[it][j] = bf2f(a.h[j]);
        dyv[it][j] = bf2f(d.h[j]);
        dxh[it][j] = dyv[it][j] * wv[it][j];
        dot += dxh[it][j] * xv[it][j];
      }
    }
    const float rr = rrms;
    const float m = block_reduce_sum(dot, lds) * rr * rr / (float)H;
#pragma unroll
    for (int it = 0; it < NITER; ++it) {
      const int idx = it * BLOCK + threadIdx.x;
      BF8 o;
      if (dres) {
        BF8 dr; dr.v = dres[row * H8 + idx];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o.h[j] = f2bf(rr * (dxh[it][j] - xv[it][j] * m) + bf2f(dr.h[j]));
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o.h[j] = f2bf(rr * (dxh[it][j] - xv[it][j] * m));
      }
      dxr[idx] = o.v;
#pragma unroll
      for (int j = 0; j < 8; ++j) dwacc[it][j] += dyv[it][j] * xv[it][j] * rr;
    }
  }
  float* dwp = dw_partial + (long)blockIdx.x * H;
#pragma unroll
  for (int it = 0; it < NITER; ++it) {
    const int idx = it * BLOCK + threadIdx.x;
#pragma unroll
    for (int j = 0; j < 8; ++j) dwp[idx * 8 + j] = dwacc[it][j];
  }
}

// Generic-H variants (any H % 8 == 0): used for shapes the register-cached
// templated kernels cannot cover (H/8 not a multiple of 256, or H > 16384).
// Two streaming passes instead of a register row cache; one row per block.
// These keep small models (tests use H=64) on the native path instead of a
// silent fallback — the launchers return a status code and Python raises on
// anything unsupported (round-1 GPUTEST SIGFPE fix).

__global__ void rmsnorm_fwd_generic_kernel(
    const uint4* __restrict__ x, const uint4* __restrict__ res_in,
    const uint4* __restrict__ w, uint4* __restrict__ y,
    uint4* __restrict__ res_out, float* __restrict__ rrms_out,
    int H8, float eps) {
  __shared__ float lds[WAVES_PER_BLOCK];
  const long row = blockIdx.x;
  const uint4* xr = x + row * H8;
  const uint4* rr = res_in ? res_in + row * H8 : nullptr;
  uint4* yr = y + row * H8;
  uint4* ror = res_out ? res_out + row * H8 : nullptr;

  float sumsq = 0.f;
  for (int idx = threadIdx.x; idx < H8; idx += BLOCK) {
    BF8 a; a.v = xr[idx];
    float vals[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) vals[j] = bf2f(a.h[j]);
    if (rr) {
      BF8 b; b.v = rr[idx];
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[j] += bf2f(b.h[j]);
      BF8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o.h[j] = f2bf(vals[j]);
      ror[idx] = o.v;
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[j] = bf2f(o.h[j]);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) sumsq += vals[j] * vals[j];
  }
  const float total = block_reduce_sum(sumsq, lds);
  const float rrms = rsqrtf(total / (float)(H8 * 8) + eps);
  if (threadIdx.x == 0 && rrms_out) rrms_out[row] = rrms;

  // pass 2: each thread re-reads exactly the indices it wrote (same stride
  // walk), so reading the rounded residual back needs no extra sync.
  const uint4* src = rr ? (const uint4*)ror : xr;
  for (int idx = threadIdx.x; idx < H8; idx += BLOCK) {
    BF8 a; a.v = src[idx];
    BF8 wv; wv.v = w[idx];
    BF8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o.h[j] = f2bf(bf2f(a.h[j]) * rrms * bf2f(wv.h[j]));
    yr[idx] = o.v;
  }
}

__global__ void rmsnorm_bwd_generic_kernel(
    const uint4* __restrict__ dy, const uint4* __restrict__ res_out,
    const uint4* __restrict__ w, const float* __restrict__ rrms_in,
    const uint4* __restrict__ dres, uint4* __restrict__ dx,
    float* __restrict__ dw_partial, int H8) {
  __shared__ float lds[WAVES_PER_BLOCK];
  const long row = blockIdx.x;
  const int H = H8 * 8;
  const uint4* dyr = dy + row * H8;
  const uint4* xr = res_out + row * H8;
  uint4* dxr = dx + row * H8;
  const float rr = rrms_in[row];

  float dot = 0.f;
  for (int idx = threadIdx.x; idx < H8; idx += BLOCK) {
    BF8 a; a.v = xr[idx];
    BF8 d; d.v = dyr[idx];
    BF8 wv; wv.v = w[idx];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dot += bf2f(d.h[j]) * bf2f(wv.h[j]) * bf2f(a.h[j]);
  }
  const float m = block_reduce_sum(dot, lds) * rr * rr / (float)H;
  float* dwp = dw_partial + row * (long)H;
  for (int idx = threadIdx.x; idx < H8; idx += BLOCK) {
    BF8 a; a.v = xr[idx];
    BF8 d; d.v = dyr[idx];
    BF8 wv; wv.v = w[idx];
    BF8 o;
    if (dres) {
      BF8 dr; dr.v = dres[row * H8 + idx];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.h[j] = f2bf(rr * (bf2f(d.h[j]) * bf2f(wv.h[j])
                            - bf2f(a.h[j]) * m) + bf2f(dr.h[j]));
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.h[j] = f2bf(rr * (bf2f(d.h[j]) * bf2f(wv.h[j])
                            - bf2f(a.h[j]) * m));
    }
    dxr[idx] = o.v;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dwp[idx * 8 + j] = bf2f(d.h[j]) * bf2f(a.h[j]) * rr;
  }
}

__global__ void rmsnorm_dw_reduce_kernel(const float* __restrict__ dw_partial,
                                         int n_partials,
                                         float* __restrict__ dw_out, int H,
                                         int rows_per_block) {
  // 2D grid: x tiles the H columns, y tiles the partial rows. Each block
  // sums its row chunk for its 256 columns and atomically folds into
  // dw_out (zero-initialized). The 1D one-thread-per-column version had
  // only H/256 blocks and measured 66 GB/s / 242 us per call on MI355X.
  const int h = blockIdx.x * blockDim.x + threadIdx.x;
  if (h >= H) return;
  const int p0 = blockIdx.y * rows_per_block;
  int p1 = p0 + rows_per_block;
  if (p1 > n_partials) p1 = n_partials;
  float acc = 0.f;
  for (int p = p0; p < p1; ++p) acc += dw_partial[(long)p * H + h];
  atomicAdd(&dw_out[h], acc);
}

// ===========================================================================
// RoPE (neox/llama half-rotation), out-of-place, fwd and bwd in one kernel
// via `sign`: fwd sign=+1, bwd sign=-1 (inverse rotation).
// x: [T, n_heads, D] bf16, D = 2*HALF, pairs (i, i+HALF); pos = token % S.
// One bf16x8 vector per thread chunk; angles from inv_freq[HALF] f32.
// ===========================================================================

__global__ void rope_kernel(const uint4* __restrict__ x, uint4* __restrict__ out,
                            const float* __restrict__ inv_freq,
                            long total_vec, int vec_per_half, int n_heads,
                            int S, int D, float sign, int pos0,
                            const long* __restrict__ pos_dev) {
  const int HALF = D / 2;
  for (long g = blockIdx.x * (long)blockDim.x + threadIdx.x; g < total_vec;
       g += gridDim.x * (long)blockDim.x) {
    // g indexes vec8 chunks of the FIRST half of each head.
    const long per_head = vec_per_half;              // HALF/8 vec8 per head-half
    const long head_g = g / per_head;
    const int i0 = (int)(g % per_head) * 8;          // dim offset in [0, HALF)
    const long t = head_g / n_heads;                 // token index
    const int h = (int)(head_g % n_heads);
    // pos_dev: device-resident base position (hipGraph decode replays
    // shift position without host input); nullptr on the training path
    const int pos = pos0 + (int)(t % S) + (pos_dev ? (int)*pos_dev : 0);
    const long base = (t * n_heads + h) * (D / 8);   // vec8 index of head start
    BF8 a; a.v = x[base + i0 / 8];
    BF8 b; b.v = x[base + (HALF + i0) / 8];
    BF8 oa, ob;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float c, s;
      __sincosf((float)pos * inv_freq[i0 + j], &s, &c);
      s *= sign;
      const float x1 = bf2f(a.h[j]), x2 = bf2f(b.h[j]);
      oa.h[j] = f2bf(x1 * c - x2 * s);
      ob.h[j] = f2bf(x1 * s + x2 * c);
    }
    out[base + i0 / 8] = oa.v;
    out[base + (HALF + i0) / 8] = ob.v;
  }
}

// ===========================================================================
// SwiGLU: out = silu(g) * u, elementwise over [*, F].
// bwd recomputes silu from g (g, u are GEMM outputs already saved).
// ===========================================================================

__device__ __forceinline__ float sigmoidf_fast(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

__global__ void swiglu_fwd_kernel(const uint4* __restrict__ g,
                                  const uint4* __restrict__ u,
                                  uint4* __restrict__ out, long n8) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += gridDim.x * (long)blockDim.x) {
    BF8 gv, uv, ov;
    gv.v = g[i]; uv.v = u[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf2f(gv.h[j]);
      ov.h[j] = f2bf(gf * sigmoidf_fast(gf) * bf2f(uv.h[j]));
    }
    out[i] = ov.v;
  }
}

// Packed variant: gu rows are [gate(F) | up(F)] straight from the fused
// gate_up GEMM; out rows are [F]. Backward writes the packed dgu, which
// feeds the GEMM backward with no concat.
__global__ void swiglu_packed_fwd_kernel(const uint4* __restrict__ gu,
                                         uint4* __restrict__ out,
                                         long n_rows, int F8) {
  const long total = n_rows * F8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * (long)blockDim.x) {
    const long row = i / F8;
    const int col = (int)(i % F8);
    BF8 gv, uv, ov;
    gv.v = gu[row * 2 * F8 + col];
    uv.v = gu[row * 2 * F8 + F8 + col];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf2f(gv.h[j]);
      ov.h[j] = f2bf(gf * sigmoidf_fast(gf) * bf2f(uv.h[j]));
    }
    out[i] = ov.v;
  }
}

__global__ void swiglu_packed_bwd_kernel(const uint4* __restrict__ dout,
                                         const uint4* __restrict__ gu,
                                         uint4* __restrict__ dgu,
                                         long n_rows, int F8) {
  const long total = n_rows * F8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * (long)blockDim.x) {
    const long row = i / F8;
    const int col = (int)(i % F8);
    BF8 dov, gv, uv, dgv, duv;
    dov.v = dout[i];
    gv.v = gu[row * 2 * F8 + col];
    uv.v = gu[row * 2 * F8 + F8 + col];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float dof = bf2f(dov.h[j]);
      const float gf = bf2f(gv.h[j]);
      const float sg = sigmoidf_fast(gf);
      duv.h[j] = f2bf(dof * gf * sg);
      dgv.h[j] = f2bf(dof * bf2f(uv.h[j]) * sg * (1.f + gf * (1.f - sg)));
    }
    dgu[row * 2 * F8 + col] = dgv.v;
    dgu[row * 2 * F8 + F8 + col] = duv.v;
  }
}

__global__ void swiglu_bwd_kernel(const uint4* __restrict__ dout,
                                  const uint4* __restrict__ g,
                                  const uint4* __restrict__ u,
                                  uint4* __restrict__ dg,
                                  uint4* __restrict__ du, long n8) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += gridDim.x * (long)blockDim.x) {
    BF8 dov, gv, uv, dgv, duv;
    dov.v = dout[i]; gv.v = g[i]; uv.v = u[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float dof = bf2f(dov.h[j]);
      const float gf = bf2f(gv.h[j]);
      const float sg = sigmoidf_fast(gf);
      const float silu = gf * sg;
      duv.h[j] = f2bf(dof * silu);
      // d silu / dg = sg * (1 + gf * (1 - sg))
      dgv.h[j] = f2bf(dof * bf2f(uv.h[j]) * sg * (1.f + gf * (1.f - sg)));
    }
    dg[i] = dgv.v;
    du[i] = duv.v;
  }
}

// ===========================================================================
// Fused cross-entropy over bf16 logits [T, V].
// fwd: one block per row — online (max, sumexp) in one streaming read;
//      writes lse[row] f32 and loss[row] f32 (0 where target==ignore).
// bwd: dlogits = (softmax - onehot(target)) * gscale[row].
// ===========================================================================

__device__ __forceinline__ void online_combine(float& m, float& s, float m2, float s2) {
  const float mm = fmaxf(m, m2);
  if (mm == -INFINITY) { s = 0.f; m = mm; return; }  // both partials empty
  s = s * __expf(m - mm) + s2 * __expf(m2 - mm);
  m = mm;
}

__global__ void ce_fwd_kernel(const uint4* __restrict__ logits,
                              const int* __restrict__ targets,
                              float* __restrict__ lse_out,
                              float* __restrict__ loss_out,
                              int V, int ignore_index) {
  __shared__ float lds_m[WAVES_PER_BLOCK];
  __shared__ float lds_s[WAVES_PER_BLOCK];
  const long row = blockIdx.x;
  const int V8 = V / 8;
  const uint4* lr = logits + row * V8;

  float m = -INFINITY, s = 0.f;
  for (int i = threadIdx.x; i < V8; i += BLOCK) {
    BF8 a; a.v = lr[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf2f(a.h[j]);
      if (x > m) { s *= __expf(m - x); m = x; }
      s += __expf(x - m);
    }
  }
  // wave combine
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float m2 = __shfl_xor(m, off, WAVE);
    const float s2 = __shfl_xor(s, off, WAVE);
    online_combine(m, s, m2, s2);
  }
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  if (lane == 0) { lds_m[wid] = m; lds_s[wid] = s; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = lds_m[0], S = lds_s[0];
#pragma unroll
    for (int w = 1; w < WAVES_PER_BLOCK; ++w) online_combine(M, S, lds_m[w], lds_s[w]);
    const float lse = M + __logf(S);
    lse_out[row] = lse;
    const int t = targets[row];
    if (t == ignore_index) {
      loss_out[row] = 0.f;
    } else {
      const u16* lrow = reinterpret_cast<const u16*>(lr);
      loss_out[row] = lse - bf2f(lrow[t]);
    }
  }
}

__global__ void ce_bwd_kernel(const uint4* __restrict__ logits,
                              const int* __restrict__ targets,
                              const float* __restrict__ lse,
                              const float* __restrict__ gscale,  // per-row scale
                              uint4* __restrict__ dlogits,
                              int V, int ignore_index) {
  const long row = blockIdx.x;
  const int V8 = V / 8;
  const uint4* lr = logits + row * V8;
  uint4* dr = dlogits + row * V8;
  const int t = targets[row];
  const float l = lse[row];
  const float gs = (t == ignore_index) ? 0.f : gscale[row];
  for (int i = threadIdx.x; i < V8; i += BLOCK) {
    BF8 a; a.v = lr[i];
    BF8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int col = i * 8 + j;
      float p = __expf(bf2f(a.h[j]) - l);
      if (col == t) p -= 1.f;
      o.h[j] = f2bf(p * gs);
    }
    dr[i] = o.v;
  }
}

// ===========================================================================
// MoE token-dispatch row ops (bf16 rows, H % 8 == 0).
//   rows_gather:        out[i] = in[idx[i]]      (fwd of x[idx])
//   rows_scatter:       out[idx[i]] = in[i]      (bwd when idx is bijective)
//   rows_scatter_add:   accum_f32[idx[i]] += in[i]  (bwd with duplicate idx)
//   moe_combine:        y[t] = sum_j gates[t,j] * src[inv[t*K+j]]
// The torch index chain these replace launches one kernel per op with an
// index_put_ backward; fused they are single HBM-streaming passes.
// ===========================================================================

__global__ void rows_gather_kernel(const uint4* __restrict__ in,
                                   const long* __restrict__ idx,
                                   uint4* __restrict__ out,
                                   long n_out, int H8) {
  const long total = n_out * H8;
  for (long g = blockIdx.x * (long)blockDim.x + threadIdx.x; g < total;
       g += gridDim.x * (long)blockDim.x) {
    const long i = g / H8;
    const int h = (int)(g % H8);
    out[g] = in[idx[i] * H8 + h];
  }
}

__global__ void rows_scatter_kernel(const uint4* __restrict__ in,
                                    const long* __restrict__ idx,
                                    uint4* __restrict__ out,
                                    long n_in, int H8) {
  const long total = n_in * H8;
  for (long g = blockIdx.x * (long)blockDim.x + threadIdx.x; g < total;
       g += gridDim.x * (long)blockDim.x) {
    const long i = g / H8;
    const int h = (int)(g % H8);
    out[idx[i] * H8 + h] = in[g];
  }
}

__global__ void rows_scatter_add_f32_kernel(const uint4* __restrict__ in,
                                            const long* __restrict__ idx,
                                            float* __restrict__ accum,
                                            long n_in, int H8) {
  const long total = n_in * H8;
  for (long g = blockIdx.x * (long)blockDim.x + threadIdx.x; g < total;
       g += gridDim.x * (long)blockDim.x) {
    const long i = g / H8;
    const int h = (int)(g % H8);
    BF8 a; a.v = in[g];
    float* dst = accum + (idx[i] * (long)H8 + h) * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) atomicAdd(&dst[j], bf2f(a.h[j]));
  }
}

__global__ void moe_combine_kernel(const uint4* __restrict__ src,
                                   const long* __restrict__ inv,
                                   const float* __restrict__ gates,
                                   uint4* __restrict__ y,
                                   long T, int K, int H8) {
  const long total = T * H8;
  for (long g = blockIdx.x * (long)blockDim.x + threadIdx.x; g < total;
       g += gridDim.x * (long)blockDim.x) {
    const long t = g / H8;
    const int h = (int)(g % H8);
    float acc[8] = {};
    for (int j = 0; j < K; ++j) {
      const long row = inv[t * K + j];
      const float gj = gates ? gates[t * K + j] : 1.0f;
      BF8 a; a.v = src[row * H8 + h];
#pragma unroll
      for (int m = 0; m < 8; ++m) acc[m] += gj * bf2f(a.h[m]);
    }
    BF8 o;
#pragma unroll
    for (int m = 0; m < 8; ++m) o.h[m] = f2bf(acc[m]);
    y[g] = o.v;
  }
}

__global__ void moe_combine_bwd_dsrc_kernel(const uint4* __restrict__ dy,
                                            const long* __restrict__ inv,
                                            const float* __restrict__ gates,
                                            uint4* __restrict__ dsrc,
                                            long n_pairs, int K, int H8) {
  const long total = n_pairs * H8;
  for (long g = blockIdx.x * (long)blockDim.x + threadIdx.x; g < total;
       g += gridDim.x * (long)blockDim.x) {
    const long pair = g / H8;
    const int h = (int)(g % H8);
    const long row = inv[pair];
    const float gj = gates[pair];
    BF8 d; d.v = dy[(pair / K) * H8 + h];
    BF8 o;
#pragma unroll
    for (int m = 0; m < 8; ++m) o.h[m] = f2bf(gj * bf2f(d.h[m]));
    dsrc[row * H8 + h] = o.v;
  }
}

__global__ void moe_combine_bwd_dgate_kernel(const uint4* __restrict__ src,
                                             const uint4* __restrict__ dy,
                                             const long* __restrict__ inv,
                                             float* __restrict__ dgate,
                                             long n_pairs, int K, int H8) {
  // one 16-lane group per (t, j) pair
  const long pair = blockIdx.x * (long)(BLOCK / 16) + (threadIdx.x >> 4);
  if (pair >= n_pairs) return;
  const long row = inv[pair];
  const long t = pair / K;
  float acc = 0.f;
  for (int h = threadIdx.x & 15; h < H8; h += 16) {
    BF8 a; a.v = src[row * H8 + h];
    BF8 d; d.v = dy[t * H8 + h];
#pragma unroll
    for (int m = 0; m < 8; ++m) acc += bf2f(a.h[m]) * bf2f(d.h[m]);
  }
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) acc += __shfl_xor(acc, off, 64);
  if ((threadIdx.x & 15) == 0) dgate[pair] = acc;
}

// ===========================================================================
// Flat-buffer AdamW with optional fused global-norm gradient clipping.
//   normsq: device pointer to the summed grad L2^2 (or nullptr for no clip);
//   the kernel derives scale = clip / max(clip, sqrt(normsq)) — no host sync.
//   p32/m/v are fp32 flat; grad bf16 flat; p_bf16 written back for compute.
// ===========================================================================

__global__ void l2normsq_partial_kernel(const uint4* __restrict__ grad, long n8,
                                        float* __restrict__ partials) {
  __shared__ float lds[WAVES_PER_BLOCK];
  float acc = 0.f;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += gridDim.x * (long)blockDim.x) {
    BF8 a; a.v = grad[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf2f(a.h[j]);
      acc += x * x;
    }
  }
  acc = block_reduce_sum(acc, lds);
  if (threadIdx.x == 0) partials[blockIdx.x] = acc;
}

__global__ void reduce_partials_kernel(const float* __restrict__ partials, int n,
                                       float* __restrict__ out) {
  __shared__ float lds[WAVES_PER_BLOCK];
  float acc = 0.f;
  for (int i = threadIdx.x; i < n; i += BLOCK) acc += partials[i];
  acc = block_reduce_sum(acc, lds);
  if (threadIdx.x == 0) out[0] = acc;
}

__global__ void adamw_kernel(float* __restrict__ p32, float* __restrict__ m,
                             float* __restrict__ v, const uint2* __restrict__ grad,
                             uint2* __restrict__ p_bf16,
                             const float* __restrict__ normsq, long n4,
                             float lr, float beta1, float beta2, float eps,
                             float weight_decay, float bc1, float bc2,
                             float clip, float pre_scale,
                             const float* __restrict__ bc_dev) {
  // pre_scale folds the DDP 1/world_size average into the update (grads are
  // SUM-all-reduced); the clip compares against the POST-scale norm.
  // bc_dev (optional, [bc1, bc2] in device memory) overrides the host
  // bias-correction args so a hipGraph-captured step stays step-correct on
  // replay.
  if (bc_dev) { bc1 = bc_dev[0]; bc2 = bc_dev[1]; }
  float gscale = pre_scale;
  if (normsq) {
    const float norm = sqrtf(*normsq) * pre_scale;
    gscale = pre_scale * (clip / fmaxf(clip, norm));
  }
  float4* p4 = reinterpret_cast<float4*>(p32);
  float4* m4 = reinterpret_cast<float4*>(m);
  float4* v4 = reinterpret_cast<float4*>(v);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4;
       i += gridDim.x * (long)blockDim.x) {
    float4 p = p4[i], mm = m4[i], vv = v4[i];
    const uint2 gv = grad[i];
    const u16* gh = reinterpret_cast<const u16*>(&gv);
    float g[4] = {bf2f(gh[0]) * gscale, bf2f(gh[1]) * gscale,
                  bf2f(gh[2]) * gscale, bf2f(gh[3]) * gscale};
    float* pp = reinterpret_cast<float*>(&p);
    float* mp = reinterpret_cast<float*>(&mm);
    float* vp = reinterpret_cast<float*>(&vv);
    u16 outh[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      mp[j] = beta1 * mp[j] + (1.f - beta1) * g[j];
      vp[j] = beta2 * vp[j] + (1.f - beta2) * g[j] * g[j];
      const float mhat = mp[j] / bc1;
      const float vhat = vp[j] / bc2;
      pp[j] -= lr * (mhat / (sqrtf(vhat) + eps) + weight_decay * pp[j]);
      outh[j] = f2bf(pp[j]);
    }
    p4[i] = p; m4[i] = mm; v4[i] = vv;
    p_bf16[i] = *reinterpret_cast<uint2*>(outh);
  }
}

// bf16-moment variant: m/v stored bf16 (halves the optimizer's HBM state
// traffic: 28 -> 20 B/param/step and the checkpoint moment size). The
// update math still runs in fp32; moments round-trip through RNE bf16.
// Gated OFF by default behind TrainConfig.adamw_bf16_moments.
__global__ void adamw_bf16mom_kernel(
    float* __restrict__ p32, uint2* __restrict__ m, uint2* __restrict__ v,
    const uint2* __restrict__ grad, uint2* __restrict__ p_bf16,
    const float* __restrict__ normsq, long n4, float lr, float beta1,
    float beta2, float eps, float weight_decay, float bc1, float bc2,
    float clip, float pre_scale, const float* __restrict__ bc_dev) {
  if (bc_dev) { bc1 = bc_dev[0]; bc2 = bc_dev[1]; }
  float gscale = pre_scale;
  if (normsq) {
    const float norm = sqrtf(*normsq) * pre_scale;
    gscale = pre_scale * (clip / fmaxf(clip, norm));
  }
  float4* p4 = reinterpret_cast<float4*>(p32);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4;
       i += gridDim.x * (long)blockDim.x) {
    float4 p = p4[i];
    const uint2 gv = grad[i];
    const uint2 mv = m[i];
    const uint2 vv = v[i];
    const u16* gh = reinterpret_cast<const u16*>(&gv);
    const u16* mh = reinterpret_cast<const u16*>(&mv);
    const u16* vh = reinterpret_cast<const u16*>(&vv);
    float* pp = reinterpret_cast<float*>(&p);
    u16 outp[4], outm[4], outv[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float g = bf2f(gh[j]) * gscale;
      float mj = beta1 * bf2f(mh[j]) + (1.f - beta1) * g;
      float vj = beta2 * bf2f(vh[j]) + (1.f - beta2) * g * g;
      const float mhat = mj / bc1;
      const float vhat = vj / bc2;
      pp[j] -= lr * (mhat / (sqrtf(vhat) + eps) + weight_decay * pp[j]);
      outp[j] = f2bf(pp[j]);
      outm[j] = f2bf(mj);
      outv[j] = f2bf(vj);
    }
    p4[i] = p;
    m[i] = *reinterpret_cast<uint2*>(outm);
    v[i] = *reinterpret_cast<uint2*>(outv);
    p_bf16[i] = *reinterpret_cast<uint2*>(outp);
  }
}

// ===========================================================================
// MFMA layout probe (test-only): computes C = A @ B with one
// v_mfma_f32_16x16x32_bf16 per wave under the layout assumptions the
// attention kernel builds on:
//   A[16x32]: lane l holds A[l & 15][(l >> 4) * 8 + j], j = 0..7
//   B[32x16]: lane l holds B[(l >> 4) * 8 + j][l & 15]
//   C[16x16]: lane l reg r holds C[(l >> 4) * 4 + r][l & 15]
// Verified by tests/test_ops_gpu.py with asymmetric inputs (guide G9).
// ===========================================================================

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ void mfma_probe_kernel(const u16* __restrict__ A,
                                  const u16* __restrict__ B,
                                  float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int row = lane & 15;
  const int kg = lane >> 4;
  union { bf16x8 v; u16 h[8]; } a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a.h[j] = A[row * 32 + kg * 8 + j];        // A[row][k]
    b.h[j] = B[(kg * 8 + j) * 16 + row];      // B[k][col], col = lane&15
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) C[(kg * 4 + r) * 16 + row] = c[r];
}

// ===========================================================================
// 32x32x16 MFMA layout probe (test-only): one v_mfma_f32_32x32x16_bf16 under
// the layout assumptions the v6 attention kernel builds on:
//   A[32x16]: lane l holds A[l & 31][(l >> 5) * 8 + j], j = 0..7
//   B[16x32]: lane l holds B[(l >> 5) * 8 + j][l & 31]
//   C[32x32]: lane l reg r holds C[(r & 3) + 8 * (r >> 2) + 4 * (l >> 5)][l & 31]
// Verified on silicon with asymmetric inputs (guide G9).
// ===========================================================================

typedef __attribute__((ext_vector_type(16))) float f32x16;

__global__ void mfma_probe32_kernel(const u16* __restrict__ A,
                                    const u16* __restrict__ B,
                                    float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int low = lane & 31;
  const int hi = lane >> 5;
  union { bf16x8 v; u16 h[8]; } a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a.h[j] = A[low * 16 + hi * 8 + j];        // A[m=low][k]
    b.h[j] = B[(hi * 8 + j) * 32 + low];      // B[k][n=low]
  }
  f32x16 c;
#pragma unroll
  for (int r = 0; r < 16; ++r) c[r] = 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a.v, b.v, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    C[((r & 3) + 8 * (r >> 2) + 4 * hi) * 32 + low] = c[r];
}

// ===========================================================================
// ds_read_b64_tr_b16 semantics probe (test-only): LDS holds raw u16 == its
// element index; each lane issues one transpose-read at a mode-dependent
// address and dumps its 4 raw results, revealing the exact lane/elem ->
// LDS-offset mapping on silicon.
// ===========================================================================

__global__ void tr_probe_kernel(u16* __restrict__ out, int base_elems,
                                int mode) {
  __shared__ u16 lds[4096];
  for (int i = threadIdx.x; i < 4096; i += 64) lds[i] = (u16)i;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  int addr_elem = base_elems;
  if (mode == 1) addr_elem += lane;            // per-lane elem stride
  else if (mode == 2) addr_elem += (lane & 15);  // column-only offset
  else if (mode == 3) addr_elem += (lane & 15) + (lane >> 4) * 64;
  const __attribute__((address_space(3))) u16* p3 =
      (const __attribute__((address_space(3))) u16*)(&lds[addr_elem]);
  unsigned long long lo;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=&v"(lo) : "v"(p3) : "memory");
  union { unsigned long long q; u16 h[4]; } u;
  u.q = lo;
#pragma unroll
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = u.h[j];
}

// ===========================================================================
// extern "C" launchers (called from Python via ctypes with the torch stream)
// ===========================================================================

static inline int elementwise_grid(long nvec) {
  long g = (nvec + BLOCK - 1) / BLOCK;
  if (g > (1 << 18)) g = 1 << 18;   // grid-stride beyond this
  if (g < 1) g = 1;
  return (int)g;
}

#define STREAM reinterpret_cast<hipStream_t>(stream)


// ---------------------------------------------------------------------------
// Decode GEMV: y[n][M] = x[n][K] @ W[M][K]^T for n <= 8 tokens (bf16 in/
// out, fp32 accumulate). At decode batch sizes the projection GEMMs are
// pure weight streams (W is read once per token, x is KBs); hipBLASLt's
// skinny-GEMM kernels leave most of HBM3E idle on these shapes, so this
// is a bandwidth kernel: one wave per output row, lanes stride the row in
// 16-byte pieces (the whole wave reads 1 KB per iteration, coalesced), x
// re-reads hit L1, wave-reduce via shfl_xor, lane 0 writes. N is a
// template parameter so the per-token accumulators stay in registers.
// ---------------------------------------------------------------------------
template <int N>
__global__ __launch_bounds__(256) void gemv_bf16_kernel(
    const u16* __restrict__ w, const u16* __restrict__ x,
    u16* __restrict__ y, int M, int K) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int m = blockIdx.x * 4 + wid;
  if (m >= M) return;
  const u16* wrow = w + (long)m * K;
  float acc[N];
#pragma unroll
  for (int n = 0; n < N; ++n) acc[n] = 0.f;
  union V8 { uint4 u; u16 h[8]; };
#pragma unroll 4
  for (int c0 = lane * 8; c0 < K; c0 += 64 * 8) {
    V8 wv;
    wv.u = *reinterpret_cast<const uint4*>(wrow + c0);
#pragma unroll
    for (int n = 0; n < N; ++n) {
      V8 xv;
      xv.u = *reinterpret_cast<const uint4*>(x + (long)n * K + c0);
      float s = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) s += bf2f(wv.h[j]) * bf2f(xv.h[j]);
      acc[n] += s;
    }
  }
#pragma unroll
  for (int n = 0; n < N; ++n) {
#pragma unroll
    for (int off = 32; off; off >>= 1)
      acc[n] += __shfl_xor(acc[n], off, 64);
    if (lane == 0) y[(long)n * M + m] = f2bf(acc[n]);
  }
}


// ---------------------------------------------------------------------------
// Fused decode attention (flash-decoding): one token's GQA attention over
// the KV cache in TWO kernels instead of the ~10-kernel einsum/softmax/
// convert chain per layer. Stage 1 tiles the cache length into 128-row
// chunks — grid (n_kv, n_chunk, B), 128 threads — and emits per-chunk
// unnormalized partials (chunk max m, sum l, acc[D]); stage 2 merges the
// chunks with the standard log-sum-exp combine. The fill position is read
// from DEVICE memory (pos_dev) so the kernels replay inside a hipGraph
// with no host input; rows beyond *pos_dev are masked to -inf.
// Phase A: lane <-> cache row (q rows staged in LDS, broadcast reads);
// Phase B: lane <-> d column (v reads coalesce per row). fp32 everywhere
// between the bf16 loads and the bf16 store.
// ---------------------------------------------------------------------------

#define ADEC_CHUNK 64
#define ADEC_D 128

// grid (n_kv * G, n_chunk, B), ONE 64-thread wave per (kv head, q head,
// 64-row cache chunk): at decode batch sizes the whole chip is idle, so
// fill is everything — splitting the GQA heads apart and halving the
// chunk gives 8x the workgroups of the first version (which ran 40 WGs
// at b1 and was latency-bound at 18 us) for a trivial re-read of the
// tiny K rows. Phase A: lane <-> cache row; phase B: lane <-> d pair.
__global__ __launch_bounds__(64) void attn_decode_partial_kernel(
    const u16* __restrict__ q,       // [B, n_kv*G, D]
    const u16* __restrict__ kc,      // [B, n_kv, Lmax, D]
    const u16* __restrict__ vc,
    const long* __restrict__ pos_dev,
    float* __restrict__ partial,     // [B, n_kv, n_chunk, G, D+2]
    int n_kv, int G, int Lmax, int n_chunk, float scale) {
  const int hkv = blockIdx.x / G;
  const int g = blockIdx.x % G;
  const int chunk = blockIdx.y;
  const int b = blockIdx.z;
  const int tid = threadIdx.x;
  const long pos = *pos_dev;               // rows 0..pos are valid
  __shared__ float sp[ADEC_CHUNK];         // scores, then p
  __shared__ u16 qs[ADEC_D];
  reinterpret_cast<u32*>(qs)[tid] = reinterpret_cast<const u32*>(
      q + ((long)b * n_kv * G + hkv * G + g) * ADEC_D)[tid];
  __syncthreads();

  const long row = (long)chunk * ADEC_CHUNK + tid;
  if (row <= pos && row < Lmax) {
    const u16* krow = kc + (((long)b * n_kv + hkv) * Lmax + row) * ADEC_D;
    float sx = 0.f;
    union V8 { uint4 u; u16 h[8]; };
#pragma unroll 4
    for (int c = 0; c < ADEC_D; c += 8) {
      V8 kv8;
      kv8.u = *reinterpret_cast<const uint4*>(krow + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) sx += bf2f(qs[c + j]) * bf2f(kv8.h[j]);
    }
    sp[tid] = sx * scale;
  } else {
    sp[tid] = -1e30f;
  }
  __syncthreads();

  // chunk max (every lane scans the 64 LDS values: broadcast reads),
  // then p = exp(s - m) written back over s
  float m = -1e30f;
  for (int r = 0; r < ADEC_CHUNK; ++r) m = fmaxf(m, sp[r]);
  const float pv = (sp[tid] > -1e30f) ? __expf(sp[tid] - m) : 0.f;
  __syncthreads();
  sp[tid] = pv;
  __syncthreads();

  // phase B: lane <-> d pair; v reads coalesce per row (64 x u32)
  const long lim = pos + 1 - (long)chunk * ADEC_CHUNK;
  const int nrow = lim < 0 ? 0 : (lim < ADEC_CHUNK ? (int)lim : ADEC_CHUNK);
  const u16* vbase = vc
      + (((long)b * n_kv + hkv) * Lmax + (long)chunk * ADEC_CHUNK) * ADEC_D
      + tid * 2;
  float a0 = 0.f, a1 = 0.f;
  for (int r = 0; r < nrow; ++r) {
    const float p = sp[r];
    union { u32 w; u16 h[2]; } vv;
    vv.w = *reinterpret_cast<const u32*>(vbase + (long)r * ADEC_D);
    a0 += p * bf2f(vv.h[0]);
    a1 += p * bf2f(vv.h[1]);
  }

  float* out = partial
      + ((((long)b * n_kv + hkv) * n_chunk + chunk) * G + g)
      * (ADEC_D + 2);
  out[tid * 2] = a0;
  out[tid * 2 + 1] = a1;
  if (tid == 0) {
    float l = 0.f;
    for (int r = 0; r < ADEC_CHUNK; ++r) l += sp[r];
    out[ADEC_D] = m;
    out[ADEC_D + 1] = l;
  }
}

// grid (B * H), 128 threads: merge the chunks for one (b, head).
__global__ __launch_bounds__(128) void attn_decode_combine_kernel(
    const float* __restrict__ partial, u16* __restrict__ o,
    int n_kv, int G, int n_chunk) {
  const int head = blockIdx.x;          // b * (n_kv*G) + hkv*G + g
  const int H = n_kv * G;
  const int b = head / H;
  const int hkv = (head % H) / G;
  const int g = head % G;
  const int tid = threadIdx.x;          // <-> d
  const float* base = partial
      + (((long)b * n_kv + hkv) * n_chunk * G + g) * (ADEC_D + 2);
  float M = -1e30f;
  for (int c = 0; c < n_chunk; ++c)
    M = fmaxf(M, base[(long)c * G * (ADEC_D + 2) + ADEC_D]);
  float L = 0.f, od = 0.f;
  for (int c = 0; c < n_chunk; ++c) {
    const float* pc = base + (long)c * G * (ADEC_D + 2);
    const float w = __expf(pc[ADEC_D] - M);
    L += w * pc[ADEC_D + 1];
    od += w * pc[tid];
  }
  o[(long)head * ADEC_D + tid] = f2bf(od / fmaxf(L, 1e-30f));
}


// GEMV tuning variants (gemv_bf16_ab): UNROLL = in-flight W loads per
// wave, NT = nontemporal W loads (decode weights are a pure stream, so
// bypassing L2 avoids thrashing it), W1 = 64-thread workgroups (grid =
// one wave per row -> 4x more workgroups for small M).
template <int N, int UNROLL, bool NT, bool W1>
__global__ __launch_bounds__(W1 ? 64 : 256) void gemv_bf16_v_kernel(
    const u16* __restrict__ w, const u16* __restrict__ x,
    u16* __restrict__ y, int M, int K) {
  const int wid = W1 ? 0 : (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  const int m = W1 ? blockIdx.x : blockIdx.x * 4 + wid;
  if (m >= M) return;
  const u16* wrow = w + (long)m * K;
  float acc[N];
#pragma unroll
  for (int n = 0; n < N; ++n) acc[n] = 0.f;
  typedef __attribute__((ext_vector_type(4))) unsigned int u32x4_;
  union V8 { uint4 u; u32x4_ e; u16 h[8]; };
#pragma unroll UNROLL
  for (int c0 = lane * 8; c0 < K; c0 += 64 * 8) {
    V8 wv;
    if (NT)
      wv.e = __builtin_nontemporal_load(
          reinterpret_cast<const u32x4_*>(wrow + c0));
    else
      wv.u = *reinterpret_cast<const uint4*>(wrow + c0);
#pragma unroll
    for (int n = 0; n < N; ++n) {
      V8 xv;
      xv.u = *reinterpret_cast<const uint4*>(x + (long)n * K + c0);
      float s = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) s += bf2f(wv.h[j]) * bf2f(xv.h[j]);
      acc[n] += s;
    }
  }
#pragma unroll
  for (int n = 0; n < N; ++n) {
#pragma unroll
    for (int off = 32; off; off >>= 1)
      acc[n] += __shfl_xor(acc[n], off, 64);
    if (lane == 0) y[(long)n * M + m] = f2bf(acc[n]);
  }
}


// ---------------------------------------------------------------------------
// Fused decode rope + KV-cache write: consumes the packed qkv GEMV output
// [B, nh*D + 2*nkv*D] and in ONE launch ropes q into a contiguous
// [B, nh, D] buffer, ropes k into cache[pos], and copies v into cache
// [pos] — replacing 2 rope launches + 2 index_copy launches per layer in
// the captured decode step. pos comes from device memory (graph-safe).
// One 32-lane-...: one workgroup of D/2 threads per head; lane d holds
// the (d, d+D/2) Neox pair.
// ---------------------------------------------------------------------------
__global__ void rope_cache_kernel(
    const u16* __restrict__ qkv, u16* __restrict__ qout,
    u16* __restrict__ kcache, u16* __restrict__ vcache,
    const float* __restrict__ inv_freq, const long* __restrict__ pos_dev,
    int nh, int nkv, int Lmax, int D) {
  const int hx = blockIdx.x;             // 0..nh-1 q, ..+nkv k, ..+nkv v
  const int b = blockIdx.y;
  const int d = threadIdx.x;             // 0 .. D/2-1
  const long pos = *pos_dev;
  const int half = D / 2;
  const u16* src = qkv + ((long)b * (nh + 2 * nkv)) * D + (long)hx * D;
  const float x1 = bf2f(src[d]), x2 = bf2f(src[d + half]);
  u16* dst;
  if (hx < nh) {
    dst = qout + ((long)b * nh + hx) * D;
  } else if (hx < nh + nkv) {
    dst = kcache + (((long)b * nkv + (hx - nh)) * Lmax + pos) * D;
  } else {
    vcache[(((long)b * nkv + (hx - nh - nkv)) * Lmax + pos) * D + d] =
        src[d];
    vcache[(((long)b * nkv + (hx - nh - nkv)) * Lmax + pos) * D + d
           + half] = src[d + half];
    return;
  }
  float c, sn;
  __sincosf((float)pos * inv_freq[d], &sn, &c);
  dst[d] = f2bf(x1 * c - x2 * sn);
  dst[d + half] = f2bf(x1 * sn + x2 * c);
}


// ---------------------------------------------------------------------------
// Decode GEMV + SwiGLU: y[n][m] = silu(x @ Wg[m]) * (x @ Wu[m]) where the
// packed weight is [2F, K] = [gate | up] rows (the training layout fed to
// swiglu_packed). One wave computes BOTH dots for its m — same W bytes as
// the plain GEMV, one launch instead of two and no 2F-wide intermediate.
// ---------------------------------------------------------------------------
template <int N>
__global__ __launch_bounds__(256) void gemv_swiglu_kernel(
    const u16* __restrict__ w, const u16* __restrict__ x,
    u16* __restrict__ y, int F, int K) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int m = blockIdx.x * 4 + wid;
  if (m >= F) return;
  const u16* grow = w + (long)m * K;
  const u16* urow = w + (long)(m + F) * K;
  float ag[N], au[N];
#pragma unroll
  for (int n = 0; n < N; ++n) ag[n] = au[n] = 0.f;
  union V8 { uint4 u; u16 h[8]; };
#pragma unroll 4
  for (int c0 = lane * 8; c0 < K; c0 += 64 * 8) {
    V8 gv, uv;
    gv.u = *reinterpret_cast<const uint4*>(grow + c0);
    uv.u = *reinterpret_cast<const uint4*>(urow + c0);
#pragma unroll
    for (int n = 0; n < N; ++n) {
      V8 xv;
      xv.u = *reinterpret_cast<const uint4*>(x + (long)n * K + c0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xf = bf2f(xv.h[j]);
        ag[n] += bf2f(gv.h[j]) * xf;
        au[n] += bf2f(uv.h[j]) * xf;
      }
    }
  }
#pragma unroll
  for (int n = 0; n < N; ++n) {
#pragma unroll
    for (int off = 32; off; off >>= 1) {
      ag[n] += __shfl_xor(ag[n], off, 64);
      au[n] += __shfl_xor(au[n], off, 64);
    }
    if (lane == 0) {
      const float g = ag[n];
      const float sg = g / (1.f + __expf(-g));   // silu
      y[(long)n * F + m] = f2bf(sg * au[n]);
    }
  }
}


// ---------------------------------------------------------------------------
// Indirect (routed) decode GEMV for MoE: the expert id comes from a
// DEVICE tensor and the weight base from a device-resident pointer
// table, so b1 decode routing never synchronizes to the host (the
// grouped path costs one D2H per layer per token for the slice sizes).
// Each (token, k) routed pair is an independent grid column: pair n
// reads x row n/xdiv and expert table[idx[n]]'s weights.
//   gemv_moe_swiglu: y[n] = silu(Wg[e_n] @ x) * (Wu[e_n] @ x)
//   gemv_moe:        y[n] = W[e_n] @ x[n]
// Same wave-per-row layout as gemv_bf16; weight bytes scale with the
// pair count, so callers gate this to decode-sized T*k.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void gemv_moe_swiglu_kernel(
    const unsigned long long* __restrict__ table_g,
    const unsigned long long* __restrict__ table_u,
    const long* __restrict__ idx, const u16* __restrict__ x,
    u16* __restrict__ y, int F, int K, int xdiv) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int m = blockIdx.x * 4 + wid;
  const int n = blockIdx.y;
  if (m >= F) return;
  const long e = idx[n];
  const u16* grow = reinterpret_cast<const u16*>(table_g[e]) + (long)m * K;
  const u16* urow = reinterpret_cast<const u16*>(table_u[e]) + (long)m * K;
  const u16* xrow = x + (long)(n / xdiv) * K;
  float ag = 0.f, au = 0.f;
  union V8 { uint4 u; u16 h[8]; };
#pragma unroll 4
  for (int c0 = lane * 8; c0 < K; c0 += 64 * 8) {
    V8 gv, uv, xv;
    gv.u = *reinterpret_cast<const uint4*>(grow + c0);
    uv.u = *reinterpret_cast<const uint4*>(urow + c0);
    xv.u = *reinterpret_cast<const uint4*>(xrow + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float xf = bf2f(xv.h[j]);
      ag += bf2f(gv.h[j]) * xf;
      au += bf2f(uv.h[j]) * xf;
    }
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    ag += __shfl_xor(ag, off, 64);
    au += __shfl_xor(au, off, 64);
  }
  if (lane == 0) {
    const float sg = ag / (1.f + __expf(-ag));
    y[(long)n * F + m] = f2bf(sg * au);
  }
}

__global__ __launch_bounds__(256) void gemv_moe_kernel(
    const unsigned long long* __restrict__ table,
    const long* __restrict__ idx, const u16* __restrict__ x,
    u16* __restrict__ y, int M, int K, int xdiv) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int m = blockIdx.x * 4 + wid;
  const int n = blockIdx.y;
  if (m >= M) return;
  const long e = idx[n];
  const u16* wrow = reinterpret_cast<const u16*>(table[e]) + (long)m * K;
  const u16* xrow = x + (long)(n / xdiv) * K;
  float acc = 0.f;
  union V8 { uint4 u; u16 h[8]; };
#pragma unroll 4
  for (int c0 = lane * 8; c0 < K; c0 += 64 * 8) {
    V8 wv, xv;
    wv.u = *reinterpret_cast<const uint4*>(wrow + c0);
    xv.u = *reinterpret_cast<const uint4*>(xrow + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc += bf2f(wv.h[j]) * bf2f(xv.h[j]);
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) acc += __shfl_xor(acc, off, 64);
  if (lane == 0) y[(long)n * M + m] = f2bf(acc);
}

extern "C" {

int hipops_arch_check() {
  // host-side marker so Python can verify the lib loaded
  return 950;
}

void mfma_probe(void* stream, const void* A, const void* B, void* C) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, STREAM,
                     (const u16*)A, (const u16*)B, (float*)C);
}

void mfma_probe32(void* stream, const void* A, const void* B, void* C) {
  hipLaunchKernelGGL(mfma_probe32_kernel, dim3(1), dim3(64), 0, STREAM,
                     (const u16*)A, (const u16*)B, (float*)C);
}

void tr_probe(void* stream, void* out, int base_elems, int mode) {
  hipLaunchKernelGGL(tr_probe_kernel, dim3(1), dim3(64), 0, STREAM,
                     (u16*)out, base_elems, mode);
}

// Every launcher returns 0 on success, nonzero on an unsupported shape;
// the Python side (ops/functional.py) raises on nonzero — no silent
// fallback, no uninitialized output buffers (round-1 SIGFPE postmortem).

static inline bool rmsnorm_templated(int H) {
  const int H8 = H / 8;
  return H % 8 == 0 && H8 % BLOCK == 0 && H8 / BLOCK >= 1 && H8 / BLOCK <= 8;
}

int rmsnorm_fwd(void* stream, const void* x, const void* res_in, const void* w,
                void* y, void* res_out, void* rrms, long n_rows, int H,
                float eps) {
  if (H % 8 != 0 || H <= 0) return -1;
  if (n_rows <= 0) return 0;
  const int H8 = H / 8;
  dim3 grid((unsigned)n_rows), block(BLOCK);
  if (rmsnorm_templated(H)) {
    const int niter = H8 / BLOCK;
#define CASE(N) \
  case N: hipLaunchKernelGGL((rmsnorm_fwd_kernel<N>), grid, block, 0, STREAM, \
      (const uint4*)x, (const uint4*)res_in, (const uint4*)w, (uint4*)y, \
      (uint4*)res_out, (float*)rrms, H8, eps); return 0;
    switch (niter) {
      CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
    }
#undef CASE
  }
  hipLaunchKernelGGL(rmsnorm_fwd_generic_kernel, grid, block, 0, STREAM,
                     (const uint4*)x, (const uint4*)res_in, (const uint4*)w,
                     (uint4*)y, (uint4*)res_out, (float*)rrms, H8, eps);
  return 0;
}

long rmsnorm_bwd_partials(long n_rows, int H) {
  // templated path: DW_ROWS rows share one partial; generic: one per row
  if (rmsnorm_templated(H)) return (n_rows + DW_ROWS - 1) / DW_ROWS;
  return n_rows;
}

int rmsnorm_bwd(void* stream, const void* dy, const void* res_out,
                const void* w, const void* rrms, const void* dres, void* dx,
                void* dw_partial, long n_rows, int H) {
  if (H % 8 != 0 || H <= 0) return -1;
  if (n_rows <= 0) return 0;
  const int H8 = H / 8;
  dim3 block(BLOCK);
  if (rmsnorm_templated(H)) {
    const int niter = H8 / BLOCK;
    dim3 grid((unsigned)((n_rows + DW_ROWS - 1) / DW_ROWS));
#define CASE(N) \
  case N: hipLaunchKernelGGL((rmsnorm_bwd_kernel<N>), grid, block, 0, STREAM, \
      (const uint4*)dy, (const uint4*)res_out, (const uint4*)w, \
      (const float*)rrms, (const uint4*)dres, (uint4*)dx, (float*)dw_partial, \
      n_rows, H8); return 0;
    switch (niter) {
      CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
    }
#undef CASE
  }
  dim3 grid((unsigned)n_rows);
  hipLaunchKernelGGL(rmsnorm_bwd_generic_kernel, grid, block, 0, STREAM,
                     (const uint4*)dy, (const uint4*)res_out, (const uint4*)w,
                     (const float*)rrms, (const uint4*)dres, (uint4*)dx,
                     (float*)dw_partial, H8);
  return 0;
}

int rmsnorm_dw_reduce(void* stream, const void* dw_partial, long n_partials,
                      void* dw_f32, int H) {
  if (H <= 0 || n_partials <= 0) return -1;
  // aim for ~1024 blocks total to fill 256 CUs; bx >= 1 always (the round-1
  // form divided by H/BLOCK, which is 0 for H < 256 -> host SIGFPE)
  const int bx = (H + BLOCK - 1) / BLOCK;
  int ny = (1024 + bx - 1) / bx;
  if (ny < 1) ny = 1;
  if (ny > n_partials) ny = (int)n_partials;
  const int rows_per_block = (int)((n_partials + ny - 1) / ny);
  dim3 grid(bx, ny), block(BLOCK);
  hipLaunchKernelGGL(rmsnorm_dw_reduce_kernel, grid, block, 0, STREAM,
                     (const float*)dw_partial, (int)n_partials, (float*)dw_f32,
                     H, rows_per_block);
  return 0;
}

int rope(void* stream, const void* x, void* out, const void* inv_freq,
         long n_tokens, int n_heads, int S, int D, float sign) {
  if (D <= 0 || D % 16 != 0 || n_heads <= 0 || S <= 0) return -1;
  if (n_tokens <= 0) return 0;
  const int vec_per_half = (D / 2) / 8;
  const long total_vec = n_tokens * (long)n_heads * vec_per_half;
  dim3 grid(elementwise_grid(total_vec)), block(BLOCK);
  hipLaunchKernelGGL(rope_kernel, grid, block, 0, STREAM, (const uint4*)x,
                     (uint4*)out, (const float*)inv_freq, total_vec,
                     vec_per_half, n_heads, S, D, sign, 0, nullptr);
  return 0;
}

// decode variant: every token sits at position pos0 + (t % S) — a KV-cached
// step passes S = new-token count and the cache fill as pos0
int rope_at(void* stream, const void* x, void* out, const void* inv_freq,
            long n_tokens, int n_heads, int S, int D, float sign,
            int pos0) {
  if (D <= 0 || D % 16 != 0 || n_heads <= 0 || S <= 0 || pos0 < 0)
    return -1;
  if (n_tokens <= 0) return 0;
  const int vec_per_half = (D / 2) / 8;
  const long total_vec = n_tokens * (long)n_heads * vec_per_half;
  dim3 grid(elementwise_grid(total_vec)), block(BLOCK);
  hipLaunchKernelGGL(rope_kernel, grid, block, 0, STREAM, (const uint4*)x,
                     (uint4*)out, (const float*)inv_freq, total_vec,
                     vec_per_half, n_heads, S, D, sign, pos0, nullptr);
  return 0;
}

// rope_at with the base position read from DEVICE memory
int rope_at_dev(void* stream, const void* x, void* out, const void* inv_freq,
            long n_tokens, int n_heads, int S, int D, float sign,
            const void* pos_dev) {
  if (D <= 0 || D % 16 != 0 || n_heads <= 0 || S <= 0 || !pos_dev)
    return -1;
  if (n_tokens <= 0) return 0;
  const int vec_per_half = (D / 2) / 8;
  const long total_vec = n_tokens * (long)n_heads * vec_per_half;
  dim3 grid(elementwise_grid(total_vec)), block(BLOCK);
  hipLaunchKernelGGL(rope_kernel, grid, block, 0, STREAM, (const uint4*)x,
                     (uint4*)out, (const float*)inv_freq, total_vec,
                     vec_per_half, n_heads, S, D, sign, 0, (const long*)pos_dev);
  return 0;
}


int swiglu_fwd(void* stream, const void* g, const void* u, void* out, long n) {
  if (n % 8 != 0) return -1;
  if (n == 0) return 0;
  const long n8 = n / 8;
  dim3 grid(elementwise_grid(n8)), block(BLOCK);
  hipLaunchKernelGGL(swiglu_fwd_kernel, grid, block, 0, STREAM,
                     (const uint4*)g, (const uint4*)u, (uint4*)out, n8);
  return 0;
}

int swiglu_packed_fwd(void* stream, const void* gu, void* out, long n_rows,
                      int F) {
  if (F <= 0 || F % 8 != 0) return -1;
  if (n_rows <= 0) return 0;
  const int F8 = F / 8;
  dim3 grid(elementwise_grid(n_rows * (long)F8)), block(BLOCK);
  hipLaunchKernelGGL(swiglu_packed_fwd_kernel, grid, block, 0, STREAM,
                     (const uint4*)gu, (uint4*)out, n_rows, F8);
  return 0;
}

int swiglu_packed_bwd(void* stream, const void* dout, const void* gu,
                      void* dgu, long n_rows, int F) {
  if (F <= 0 || F % 8 != 0) return -1;
  if (n_rows <= 0) return 0;
  const int F8 = F / 8;
  dim3 grid(elementwise_grid(n_rows * (long)F8)), block(BLOCK);
  hipLaunchKernelGGL(swiglu_packed_bwd_kernel, grid, block, 0, STREAM,
                     (const uint4*)dout, (const uint4*)gu, (uint4*)dgu,
                     n_rows, F8);
  return 0;
}

int swiglu_bwd(void* stream, const void* dout, const void* g, const void* u,
               void* dg, void* du, long n) {
  if (n % 8 != 0) return -1;
  if (n == 0) return 0;
  const long n8 = n / 8;
  dim3 grid(elementwise_grid(n8)), block(BLOCK);
  hipLaunchKernelGGL(swiglu_bwd_kernel, grid, block, 0, STREAM,
                     (const uint4*)dout, (const uint4*)g, (const uint4*)u,
                     (uint4*)dg, (uint4*)du, n8);
  return 0;
}

int ce_fwd(void* stream, const void* logits, const void* targets, void* lse,
           void* loss, long n_rows, int V, int ignore_index) {
  if (V <= 0 || V % 8 != 0) return -1;
  if (n_rows <= 0) return 0;
  dim3 grid((unsigned)n_rows), block(BLOCK);
  hipLaunchKernelGGL(ce_fwd_kernel, grid, block, 0, STREAM,
                     (const uint4*)logits, (const int*)targets, (float*)lse,
                     (float*)loss, V, ignore_index);
  return 0;
}

int ce_bwd(void* stream, const void* logits, const void* targets,
           const void* lse, const void* gscale, void* dlogits, long n_rows,
           int V, int ignore_index) {
  if (V <= 0 || V % 8 != 0) return -1;
  if (n_rows <= 0) return 0;
  dim3 grid((unsigned)n_rows), block(BLOCK);
  hipLaunchKernelGGL(ce_bwd_kernel, grid, block, 0, STREAM,
                     (const uint4*)logits, (const int*)targets,
                     (const float*)lse, (const float*)gscale, (uint4*)dlogits,
                     V, ignore_index);
  return 0;
}

int rows_gather(void* stream, const void* in, const void* idx, void* out,
                long n_out, int H) {
  if (H <= 0 || H % 8 != 0) return -1;
  if (n_out <= 0) return 0;
  const int H8 = H / 8;
  dim3 grid(elementwise_grid(n_out * (long)H8)), block(BLOCK);
  hipLaunchKernelGGL(rows_gather_kernel, grid, block, 0, STREAM,
                     (const uint4*)in, (const long*)idx, (uint4*)out,
                     n_out, H8);
  return 0;
}

int rows_scatter(void* stream, const void* in, const void* idx, void* out,
                 long n_in, int H) {
  if (H <= 0 || H % 8 != 0) return -1;
  if (n_in <= 0) return 0;
  const int H8 = H / 8;
  dim3 grid(elementwise_grid(n_in * (long)H8)), block(BLOCK);
  hipLaunchKernelGGL(rows_scatter_kernel, grid, block, 0, STREAM,
                     (const uint4*)in, (const long*)idx, (uint4*)out,
                     n_in, H8);
  return 0;
}

int rows_scatter_add_f32(void* stream, const void* in, const void* idx,
                         void* accum, long n_in, int H) {
  if (H <= 0 || H % 8 != 0) return -1;
  if (n_in <= 0) return 0;
  const int H8 = H / 8;
  dim3 grid(elementwise_grid(n_in * (long)H8)), block(BLOCK);
  hipLaunchKernelGGL(rows_scatter_add_f32_kernel, grid, block, 0, STREAM,
                     (const uint4*)in, (const long*)idx, (float*)accum,
                     n_in, H8);
  return 0;
}

int moe_combine(void* stream, const void* src, const void* inv,
                const void* gates, void* y, long T, int K, int H) {
  if (H <= 0 || H % 8 != 0 || K <= 0) return -1;
  if (T <= 0) return 0;
  const int H8 = H / 8;
  dim3 grid(elementwise_grid(T * (long)H8)), block(BLOCK);
  hipLaunchKernelGGL(moe_combine_kernel, grid, block, 0, STREAM,
                     (const uint4*)src, (const long*)inv,
                     (const float*)gates, (uint4*)y, T, K, H8);
  return 0;
}

int moe_combine_bwd(void* stream, const void* src, const void* dy,
                    const void* inv, const void* gates, void* dsrc,
                    void* dgate, long n_pairs, int K, int H) {
  if (H <= 0 || H % 8 != 0 || K <= 0) return -1;
  if (n_pairs <= 0) return 0;
  const int H8 = H / 8;
  dim3 grid(elementwise_grid(n_pairs * (long)H8)), block(BLOCK);
  hipLaunchKernelGGL(moe_combine_bwd_dsrc_kernel, grid, block, 0, STREAM,
                     (const uint4*)dy, (const long*)inv, (const float*)gates,
                     (uint4*)dsrc, n_pairs, K, H8);
  long gblocks = (n_pairs + (BLOCK / 16) - 1) / (BLOCK / 16);
  if (gblocks > (1 << 20)) gblocks = 1 << 20;
  hipLaunchKernelGGL(moe_combine_bwd_dgate_kernel, dim3((unsigned)gblocks),
                     block, 0, STREAM, (const uint4*)src, (const uint4*)dy,
                     (const long*)inv, (float*)dgate, n_pairs, K, H8);
  return 0;
}

int adamw_step_bf16mom(void* stream, void* p32, void* m, void* v,
                       const void* grad, void* p_bf16, const void* normsq,
                       long n, float lr, float beta1, float beta2, float eps,
                       float weight_decay, float bc1, float bc2, float clip,
                       float pre_scale, const void* bc_dev) {
  if (n % 4 != 0) return -1;
  if (n == 0) return 0;
  const long n4 = n / 4;
  dim3 grid(elementwise_grid(n4)), block(BLOCK);
  hipLaunchKernelGGL(adamw_bf16mom_kernel, grid, block, 0, STREAM,
                     (float*)p32, (uint2*)m, (uint2*)v, (const uint2*)grad,
                     (uint2*)p_bf16, (const float*)normsq, n4, lr, beta1,
                     beta2, eps, weight_decay, bc1, bc2, clip, pre_scale,
                     (const float*)bc_dev);
  return 0;
}

int l2normsq(void* stream, const void* grad, long n, void* partials,
             int n_partials, void* out) {
  if (n % 8 != 0 || n <= 0 || n_partials <= 0) return -1;
  const long n8 = n / 8;
  int grid = elementwise_grid(n8);
  if (grid > n_partials) grid = n_partials;
  hipLaunchKernelGGL(l2normsq_partial_kernel, dim3(grid), dim3(BLOCK), 0,
                     STREAM, (const uint4*)grad, n8, (float*)partials);
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(1), dim3(BLOCK), 0, STREAM,
                     (const float*)partials, grid, (float*)out);
  return 0;
}

int adamw_step(void* stream, void* p32, void* m, void* v, const void* grad,
               void* p_bf16, const void* normsq, long n, float lr, float beta1,
               float beta2, float eps, float weight_decay, float bc1,
               float bc2, float clip, float pre_scale, const void* bc_dev) {
  if (n % 4 != 0) return -1;
  if (n == 0) return 0;
  const long n4 = n / 4;
  dim3 grid(elementwise_grid(n4)), block(BLOCK);
  hipLaunchKernelGGL(adamw_kernel, grid, block, 0, STREAM, (float*)p32,
                     (float*)m, (float*)v, (const uint2*)grad, (uint2*)p_bf16,
                     (const float*)normsq, n4, lr, beta1, beta2, eps,
                     weight_decay, bc1, bc2, clip, pre_scale,
                     (const float*)bc_dev);
  return 0;
}


// decode GEMV (gemv_bf16_kernel): rows of W must be contiguous; x/y are
// [N, K] / [N, M] row-major. K % 512 keeps every lane on whole 16-byte
// pieces; callers fall back to the library GEMM otherwise.
int gemv_bf16(void* stream, const void* w, const void* x, void* y,
              int M, int K, int N) {
  if (M <= 0 || K <= 0 || (K % 512) != 0 || N < 1 || N > 8) return -1;
  if (M >= 65536) {
    // lm_head-sized: one wave per WG + nontemporal W loads measured
    // 7.4 vs 6.4 TB/s (scripts/gemvbench.py) — the 1 GB matrix only
    // thrashes L2, and the bigger grid fills the 8 XCDs
    dim3 grid((unsigned)M), block(64);
    switch (N) {
#define GEMV_CASE(NN) \
      case NN: \
        hipLaunchKernelGGL((gemv_bf16_v_kernel<NN, 8, true, true>), grid, \
                           block, 0, STREAM, (const u16*)w, \
                           (const u16*)x, (u16*)y, M, K); \
        break;
      GEMV_CASE(1) GEMV_CASE(2) GEMV_CASE(3) GEMV_CASE(4)
      GEMV_CASE(5) GEMV_CASE(6) GEMV_CASE(7) GEMV_CASE(8)
#undef GEMV_CASE
    }
    return 0;
  }
  dim3 grid((unsigned)((M + 3) / 4)), block(256);
  switch (N) {
#define GEMV_CASE(NN) \
    case NN: \
      hipLaunchKernelGGL((gemv_bf16_kernel<NN>), grid, block, 0, STREAM, \
                         (const u16*)w, (const u16*)x, (u16*)y, M, K); \
      break;
    GEMV_CASE(1) GEMV_CASE(2) GEMV_CASE(3) GEMV_CASE(4)
    GEMV_CASE(5) GEMV_CASE(6) GEMV_CASE(7) GEMV_CASE(8)
#undef GEMV_CASE
  }
  return 0;
}


// fused decode attention: q [B, H, D=128] bf16, caches [B, n_kv, Lmax,
// 128] bf16 contiguous, pos_dev a device scalar (long) = last valid
// cache row, partial fp32 [B, n_kv, n_chunk, G, 130] scratch, o [B, H,
// 128] bf16. n_chunk must be (Lmax + 127) / 128.
int attn_decode(void* stream, const void* q, const void* kc,
                const void* vc, const void* pos_dev, void* partial,
                void* o, int B, int H, int n_kv, int Lmax, int n_chunk,
                float scale) {
  if (B <= 0 || n_kv <= 0 || H % n_kv != 0 ||
      n_chunk != (Lmax + ADEC_CHUNK - 1) / ADEC_CHUNK)
    return -1;
  const int G = H / n_kv;
  dim3 grid((unsigned)(n_kv * G), (unsigned)n_chunk, (unsigned)B);
  hipLaunchKernelGGL(attn_decode_partial_kernel, grid, dim3(64), 0,
                     STREAM, (const u16*)q, (const u16*)kc,
                     (const u16*)vc, (const long*)pos_dev,
                     (float*)partial, n_kv, G, Lmax, n_chunk, scale);
  hipLaunchKernelGGL(attn_decode_combine_kernel, dim3((unsigned)(B * H)),
                     dim3(128), 0, STREAM, (const float*)partial, (u16*)o,
                     n_kv, G, n_chunk);
  return 0;
}

// within-box A/B of GEMV variants (N=1 only): 0=unroll4 (shipped
// baseline layout), 1=unroll8, 2=unroll8+nontemporal, 3=wave-per-WG
// grid M + unroll8 + nontemporal.
int gemv_bf16_ab(void* stream, int which, const void* w, const void* x,
                 void* y, int M, int K) {
  if (K % 512 != 0) return -1;
  dim3 b256(256), b64(64);
  dim3 g4((unsigned)((M + 3) / 4)), g1((unsigned)M);
  switch (which) {
    case 0:
      hipLaunchKernelGGL((gemv_bf16_v_kernel<1, 4, false, false>), g4,
                         b256, 0, STREAM, (const u16*)w, (const u16*)x,
                         (u16*)y, M, K);
      break;
    case 1:
      hipLaunchKernelGGL((gemv_bf16_v_kernel<1, 8, false, false>), g4,
                         b256, 0, STREAM, (const u16*)w, (const u16*)x,
                         (u16*)y, M, K);
      break;
    case 2:
      hipLaunchKernelGGL((gemv_bf16_v_kernel<1, 8, true, false>), g4,
                         b256, 0, STREAM, (const u16*)w, (const u16*)x,
                         (u16*)y, M, K);
      break;
    case 3:
      hipLaunchKernelGGL((gemv_bf16_v_kernel<1, 8, true, true>), g1,
                         b64, 0, STREAM, (const u16*)w, (const u16*)x,
                         (u16*)y, M, K);
      break;
    default: return -1;
  }
  return 0;
}


// fused decode rope + cache write (rope_cache_kernel): qkv packed
// [B, (nh+2*nkv)*D] bf16 (q | k | v), caches [B, nkv, Lmax, D]
// contiguous, pos_dev device long scalar. D even, D/2 <= 1024.
int rope_cache(void* stream, const void* qkv, void* qout, void* kcache,
               void* vcache, const void* inv_freq, const void* pos_dev,
               int B, int nh, int nkv, int Lmax, int D) {
  if (D <= 0 || D % 16 != 0 || D / 2 > 1024 || nh <= 0 || nkv <= 0 ||
      B <= 0 || !pos_dev)
    return -1;
  dim3 grid((unsigned)(nh + 2 * nkv), (unsigned)B), block(D / 2);
  hipLaunchKernelGGL(rope_cache_kernel, grid, block, 0, STREAM,
                     (const u16*)qkv, (u16*)qout, (u16*)kcache,
                     (u16*)vcache, (const float*)inv_freq,
                     (const long*)pos_dev, nh, nkv, Lmax, D);
  return 0;
}


// decode GEMV+SwiGLU over the packed [2F, K] gate|up weight.
int gemv_swiglu(void* stream, const void* w, const void* x, void* y,
                int F, int K, int N) {
  if (F <= 0 || K <= 0 || (K % 512) != 0 || N < 1 || N > 8) return -1;
  dim3 grid((unsigned)((F + 3) / 4)), block(256);
  switch (N) {
#define GSW_CASE(NN) \
    case NN: \
      hipLaunchKernelGGL((gemv_swiglu_kernel<NN>), grid, block, 0, \
                         STREAM, (const u16*)w, (const u16*)x, (u16*)y, \
                         F, K); \
      break;
    GSW_CASE(1) GSW_CASE(2) GSW_CASE(3) GSW_CASE(4)
    GSW_CASE(5) GSW_CASE(6) GSW_CASE(7) GSW_CASE(8)
#undef GSW_CASE
  }
  return 0;
}


// routed decode GEMVs (tables = device u64 arrays of expert weight base
// pointers; idx = device long per routed pair). xdiv folds the
// pair->token mapping: the swiglu stage reads x[n / top_k], the down
// stage reads its own pair-major input (xdiv 1).
int gemv_moe_swiglu(void* stream, const void* table_g, const void* table_u,
                    const void* idx, const void* x, void* y,
                    int F, int K, int N, int xdiv) {
  if (F <= 0 || K <= 0 || (K % 512) != 0 || N < 1 || N > 64 || xdiv < 1)
    return -1;
  dim3 grid((unsigned)((F + 3) / 4), (unsigned)N), block(256);
  hipLaunchKernelGGL(gemv_moe_swiglu_kernel, grid, block, 0, STREAM,
                     (const unsigned long long*)table_g,
                     (const unsigned long long*)table_u,
                     (const long*)idx, (const u16*)x, (u16*)y, F, K, xdiv);
  return 0;
}

int gemv_moe(void* stream, const void* table, const void* idx,
             const void* x, void* y, int M, int K, int N, int xdiv) {
  if (M <= 0 || K <= 0 || (K % 512) != 0 || N < 1 || N > 64 || xdiv < 1)
    return -1;
  dim3 grid((unsigned)((M + 3) / 4), (unsigned)N), block(256);
  hipLaunchKernelGGL(gemv_moe_kernel, grid, block, 0, STREAM,
                     (const unsigned long long*)table,
                     (const long*)idx, (const u16*)x, (u16*)y, M, K, xdiv);
  return 0;
}

}  // extern "C"
