// BigCLAM CDNA4 (gfx950 / MI355X) HIP kernels — v2.
//
// Hand-written for the CDNA4 execution model: 64-wide wavefronts, float4
// HBM access sized for L1/L2 locality, LDS accumulators without atomics,
// wave/blocked shuffle reductions, fp64 LLH accumulation.  Kernel worklist
// per SURVEY.md §2-B:
//   K1 edge_grad_llh  — fused per-node gradient + local LLH
//                       (replaces codes/bigclamv3-7.scala:138-150)
//   K2 linesearch     — all 16 Armijo candidates per node in one edge pass
//                       (replaces the cartesian fan-out, scala:153-163)
//   K3 apply_step     — projected commit (scala:89-92, 171)
//   K3CS k3_colsum    — bf16 commit + fp32 column partial sums, one pass
//   K4 llh_only       — per-node local LLH (scala:106-120 / 177-200)
//   K5 k5_conductance — ego-net conductance seed ranking (scala:39-54)
//   KF kf_fused/kf_mfma — K1+K2 fused per node; the MFMA variants run
//                       the 16-candidate scoring as (deg x K)@(K x 16)
//                       GEMM tiles on v_mfma_f32_16x16x32_bf16 /
//                       v_mfma_f32_16x16x4_f32 (dispatch is K-aware,
//                       see launch_kf_mfma* and core/state.py)
//   KD/KW kd_dot/kw_grad — chunked large-K K1 (no K cap; x-buffer split)
//   KAF/KFS/K3S         — sparse-adaptive sweep on exact active column
//                       sets (support compaction, fused compact
//                       grad+Armijo, sparse commit + list rewrite)
//   K6 k6_seed_init     — device seed-init scatter (scala:60-87)
//   K7 k7_membership    — device community extraction (Bigclamv2:223-230)
//
// Design notes (measured rationale in profiles/r01_kernel_opt_log.md):
//  * K1 is BLOCK-per-node: per edge the workgroup does one cooperative dot
//    (block_allreduce) and an owned-k no-atomic accumulate into the LDS
//    gradient row; fv stays in REGISTERS between the dot and the axpy.
//  * K2 evaluates all 16 Armijo candidates DIRECTLY per edge element
//    (clamp-fma + acc-fma in registers, v_med3_f32 clamps) — candidate
//    rows are never materialized; fu/grad are LDS-staged per node; the 16
//    per-edge dot sums reduce in ONE 17-shuffle butterfly (wave_reduce16)
//    and the Armijo pick is a 16-lane ballot.
//  * LLH is accumulated in fp64 (the convergence test is a 1e-4 relative
//    change on a large-magnitude sum).
//
// Math contract (must match tests/oracle.py):
//   x    = Fu . Fv
//   p    = clamp(exp(-x), MIN_P, MAX_P)
//   llh += log1p(-p) + x           ;  grad_acc += Fv / (1-p)
//   grad = grad_acc - sumF + Fu    ;  llh += -Fu.sumF + Fu.Fu
// Line-search trial node term uses the identity
//   -Fu'.(sumF - Fu + Fu') + Fu'.Fu' == Fu'.(Fu - sumF).

#include <hip/hip_runtime.h>

#define WAVE 64
#define BLOCK 256
#define NWAVE (BLOCK / WAVE)
#define MAX_LS 16  // ladder length (ls_steps + 1); reference uses 16

typedef float v2f __attribute__((ext_vector_type(2)));  // v_pk_* pairs

// ---------------------------------------------------------------- reductions

__device__ __forceinline__ float wave_allreduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// Sum 16 per-lane values across the wave simultaneously.  Butterfly with
// value-halving: each xor step folds the lane count 2x while handing off
// half the surviving values to the other side, so the whole 16-value
// reduction costs 8+4+2+1+1+1 = 17 exchanges instead of 16 independent
// allreduces (96).  Result: value j lands on lane (j*4) in w[0]; the
// caller writes from whatever lane holds it.
__device__ __forceinline__ void wave_reduce16(float (&w)[MAX_LS], int lane) {
  // steps folding values: offsets 32,16,8,4 halve the value count
#pragma unroll
  for (int step = 0; step < 4; ++step) {
    const int off = 32 >> step;
    const int nv = 8 >> step;  // surviving values after this step
    const bool hi = (lane & off) != 0;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      if (i >= nv) break;
      const float send = hi ? w[i] : w[i + nv];
      const float recv = __shfl_xor(send, off, WAVE);
      w[i] = (hi ? w[i + nv] : w[i]) + recv;
    }
  }
  // lanes now hold ONE value each: lane group (l>>2)&15... value index is
  // bit-reversed-ish; the mapping resolved below.  Fold the last 4 lanes.
  w[0] += __shfl_xor(w[0], 2, WAVE);
  w[0] += __shfl_xor(w[0], 1, WAVE);
}

// After wave_reduce16, value j sits (replicated over a 4-lane group) on
// the lanes whose bits select it: lane l holds value
//   j(l) = ((l>>5)&1)<<3 | ((l>>4)&1)<<2 | ((l>>3)&1)<<1 | ((l>>2)&1)
// ("jmine" in the kernels; lanes with (l & 3) == 0 are the owners).

// block-wide sum; `red` is NWAVE floats of LDS; result on every thread.
__device__ __forceinline__ float block_allreduce_sum(float v, float* red) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  v = wave_allreduce_sum(v);
  if (lane == 0) red[wid] = v;
  __syncthreads();
  float r = (lane < NWAVE) ? red[lane] : 0.f;
#pragma unroll
  for (int off = NWAVE / 2; off > 0; off >>= 1) r += __shfl_xor(r, off, WAVE);
  r = __shfl(r, 0, WAVE);
  __syncthreads();
  return r;
}

// ------------------------------------------------------------------ helpers

__device__ __forceinline__ float clamp_p(float p, float min_p, float max_p) {
  return fminf(fmaxf(p, min_p), max_p);
}

__device__ __forceinline__ float dot4(float4 a, float4 b, float acc) {
  acc = fmaf(a.x, b.x, acc);
  acc = fmaf(a.y, b.y, acc);
  acc = fmaf(a.z, b.z, acc);
  return fmaf(a.w, b.w, acc);
}

__device__ __forceinline__ float4 ld4(const float* p) {
  return *reinterpret_cast<const float4*>(p);
}

// 2-wide fp32 vectors (v2f, defined with the reduction helpers) lower to
// CDNA packed-math (v_pk_fma_f32): one instruction per pair, doubling
// VALU throughput in the K2 inner loop.
__device__ __forceinline__ v2f pk_clamp_fma(v2f s, v2f g, v2f a, v2f lo,
                                            v2f hi) {
  // v_pk_fma_f32 + two v_med3_f32 (med3 = whole clamp in ONE VALU op;
  // no packed min/max f32 exists on CDNA4, med3 halves the clamp cost)
  const v2f t = __builtin_elementwise_fma(s, g, a);
  return v2f{__builtin_amdgcn_fmed3f(t.x, lo.x, hi.x),
             __builtin_amdgcn_fmed3f(t.y, lo.y, hi.y)};
}


// ------------------------------------------------------- bf16 storage path
//
// F stored as bf16 (K padded to a multiple of 8 -> 16B-aligned uint4 rows);
// all arithmetic in fp32, grad/sumF stay fp32.  bf16 = the high 16 bits of
// fp32, so unpacking is one shift/mask per element; packing uses RNE.

typedef unsigned int u32;

__device__ __forceinline__ float bf_lo(u32 v) {
  return __uint_as_float(v << 16);
}
__device__ __forceinline__ float bf_hi(u32 v) {
  return __uint_as_float(v & 0xffff0000u);
}

__device__ __forceinline__ v2f bf2(u32 v) {  // one packed bf16 pair -> v2f
  return v2f{bf_lo(v), bf_hi(v)};
}

struct f32x8 {
  float4 a, b;
};

__device__ __forceinline__ f32x8 ld8bf_u(uint4 u) {
  f32x8 r;
  r.a = float4{bf_lo(u.x), bf_hi(u.x), bf_lo(u.y), bf_hi(u.y)};
  r.b = float4{bf_lo(u.z), bf_hi(u.z), bf_lo(u.w), bf_hi(u.w)};
  return r;
}

__device__ __forceinline__ f32x8 ld8bf(const u32* p) {
  return ld8bf_u(*reinterpret_cast<const uint4*>(p));
}

__device__ __forceinline__ u32 pack_bf16_rne(float lo, float hi) {
  // round-to-nearest-even bf16 truncation of two fp32 values
  u32 l = __float_as_uint(lo);
  u32 h = __float_as_uint(hi);
  l += 0x7fffu + ((l >> 16) & 1u);
  h += 0x7fffu + ((h >> 16) & 1u);
  return (l >> 16) | (h & 0xffff0000u);
}

__device__ __forceinline__ float dot8(f32x8 a, f32x8 b, float acc) {
  return dot4(a.b, b.b, dot4(a.a, b.a, acc));
}

// ------------------------------------------------------------- MFMA types
//
// Fragment types for the CDNA4 matrix cores.  The line-search candidate
// scoring is GEMM-shaped per node: (deg x K neighbor rows) @ (K x 16
// clamped candidate rows)^T, mapped onto v_mfma_f32_16x16x32_bf16 /
// v_mfma_f32_16x16x4_f32 tiles of 16 edges x 16 candidates.
//
// A and B fragments are filled with the SAME lane->k mapping
// (k = kbase + (lane>>4)*8 + i for bf16, kbase + (lane>>4) for fp32), so
// the contraction pairs A[r][k] with B[k][c] correctly under ANY internal
// k-slot permutation; only the C/D register mapping is layout-sensitive:
//   col = lane & 15, row = (lane >> 4) * 4 + reg_idx   (ISA §10, probed
//   on-device by mfma_probe / tests/test_gpu.py).

typedef short bf16x8 __attribute__((ext_vector_type(8)));  // 8 bf16, 4 VGPR
typedef float f32x4v __attribute__((ext_vector_type(4)));  // MFMA acc

__device__ __forceinline__ bf16x8 as_bf16x8(uint4 v) {
  union {
    uint4 u;
    bf16x8 b;
  } c;
  c.u = v;
  return c.b;
}

// ------------------------------------------------------------------- K1
//
// One 256-thread block per LOCAL node (launch order = degree-descending so
// hub blocks start first).  LDS: gradient accumulator gacc[K] (owned-k,
// no atomics) + reduction scratch.  Per edge the block does a cooperative
// dot (block_allreduce) and an owned-k accumulate; fv's second read hits L1
// (the block touches one 4*K-byte row at a time).

template <int NSLOT>
__global__ void __launch_bounds__(BLOCK, 4) k1_grad_llh_t(
    const float* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, float* __restrict__ grad,
    double* __restrict__ llh, int n_local, int K, float min_p, float max_p) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* gacc = reinterpret_cast<float*>(smem);            // K floats
  float* red = reinterpret_cast<float*>(smem + (size_t)K * 4);  // NWAVE

  const float* __restrict__ fu = F + (size_t)u * K;
  // fu resident in registers: dot pass streams only fv, so the axpy
  // re-read of fv stays L1-hot (fu+fv would overflow the 32 KB L1)
  float4 fu4[NSLOT];
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 4 + sl * (BLOCK * 4);
    fu4[sl] = (k < K) ? ld4(fu + k) : float4{0.f, 0.f, 0.f, 0.f};
    if (k < K) *reinterpret_cast<float4*>(gacc + k) = float4{0.f, 0.f, 0.f, 0.f};
  }
  __syncthreads();

  double llh_acc = 0.0;  // accumulated on thread 0 only

#pragma clang loop unroll(disable)
  for (long long e = e0; e < e1; ++e) {
    const float* __restrict__ fv = F + (size_t)indices[e] * K;
    float4 b[NSLOT];  // fv stays in registers between dot and axpy
    float part = 0.f;
#pragma unroll
    for (int sl = 0; sl < NSLOT; ++sl) {
      const int k = tid * 4 + sl * (BLOCK * 4);
      b[sl] = (k < K) ? ld4(fv + k) : float4{0.f, 0.f, 0.f, 0.f};
      part = dot4(fu4[sl], b[sl], part);
    }
    const float x = block_allreduce_sum(part, red);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    const float w = 1.f / (1.f - p);
    if (tid == 0) llh_acc += (double)log1pf(-p) + (double)x;
#pragma unroll
    for (int sl = 0; sl < NSLOT; ++sl) {
      const int k = tid * 4 + sl * (BLOCK * 4);
      if (k < K) {
        float4 g = ld4(gacc + k);
        g.x = fmaf(w, b[sl].x, g.x);
        g.y = fmaf(w, b[sl].y, g.y);
        g.z = fmaf(w, b[sl].z, g.z);
        g.w = fmaf(w, b[sl].w, g.w);
        *reinterpret_cast<float4*>(gacc + k) = g;
      }
    }
    // no barrier needed: each thread owns its gacc elements; the next
    // edge's block_allreduce syncs before x is consumed.
  }

  // node terms: -Fu.sumF + Fu.Fu
  float p_fs = 0.f, p_ff = 0.f;
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 4 + sl * (BLOCK * 4);
    if (k < K) {
      const float4 s = ld4(sumF + k);
      p_fs = dot4(fu4[sl], s, p_fs);
      p_ff = dot4(fu4[sl], fu4[sl], p_ff);
    }
  }
  const float fs = block_allreduce_sum(p_fs, red);
  const float ff = block_allreduce_sum(p_ff, red);

  float* __restrict__ gout = grad + (size_t)u * K;
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 4 + sl * (BLOCK * 4);
    if (k < K) {
      const float4 g = ld4(gacc + k);
      const float4 sv = ld4(sumF + k);
      *reinterpret_cast<float4*>(gout + k) =
          float4{g.x - sv.x + fu4[sl].x, g.y - sv.y + fu4[sl].y,
                 g.z - sv.z + fu4[sl].z, g.w - sv.w + fu4[sl].w};
    }
  }
  if (tid == 0) llh[u] = llh_acc + (double)(-fs) + (double)ff;
}

// generic variant for K > 8192: fu streamed from memory (L2) per edge
extern "C" __global__ void __launch_bounds__(BLOCK) k1_grad_llh(
    const float* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, float* __restrict__ grad,
    double* __restrict__ llh, int n_local, int K, float min_p, float max_p) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* gacc = reinterpret_cast<float*>(smem);            // K floats
  float* red = reinterpret_cast<float*>(smem + (size_t)K * 4);  // NWAVE

  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    *reinterpret_cast<float4*>(gacc + k) = float4{0.f, 0.f, 0.f, 0.f};
  }
  __syncthreads();

  const float* __restrict__ fu = F + (size_t)u * K;
  double llh_acc = 0.0;  // accumulated on thread 0 only

  for (long long e = e0; e < e1; ++e) {
    const float* __restrict__ fv = F + (size_t)indices[e] * K;
    float part = 0.f;
    for (int k = tid * 4; k < K; k += BLOCK * 4) {
      part = dot4(ld4(fu + k), ld4(fv + k), part);
    }
    const float x = block_allreduce_sum(part, red);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    const float w = 1.f / (1.f - p);
    if (tid == 0) llh_acc += (double)log1pf(-p) + (double)x;
    for (int k = tid * 4; k < K; k += BLOCK * 4) {
      const float4 b = ld4(fv + k);  // L1-hot: just read by this block
      float4 g = ld4(gacc + k);
      g.x = fmaf(w, b.x, g.x);
      g.y = fmaf(w, b.y, g.y);
      g.z = fmaf(w, b.z, g.z);
      g.w = fmaf(w, b.w, g.w);
      *reinterpret_cast<float4*>(gacc + k) = g;
    }
    // no barrier needed: each thread owns its gacc elements; the next
    // edge's block_allreduce syncs before x is consumed.
  }

  // node terms: -Fu.sumF + Fu.Fu
  float p_fs = 0.f, p_ff = 0.f;
  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    const float4 a = ld4(fu + k);
    const float4 s = ld4(sumF + k);
    p_fs = dot4(a, s, p_fs);
    p_ff = dot4(a, a, p_ff);
  }
  const float fs = block_allreduce_sum(p_fs, red);
  const float ff = block_allreduce_sum(p_ff, red);

  float* __restrict__ gout = grad + (size_t)u * K;
  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    const float4 g = ld4(gacc + k);
    const float4 s = ld4(sumF + k);
    const float4 a = ld4(fu + k);
    *reinterpret_cast<float4*>(gout + k) =
        float4{g.x - s.x + a.x, g.y - s.y + a.y, g.z - s.z + a.z,
               g.w - s.w + a.w};
  }
  if (tid == 0) llh[u] = llh_acc + (double)(-fs) + (double)ff;
}

// ------------------------------------------------------------------- K4
//
// Wave-per-edge dots (validated at ~5.8 TB/s effective in the v1 profile).

extern "C" __global__ void __launch_bounds__(BLOCK) k4_llh_only(
    const float* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, double* __restrict__ llh, int n_local,
    int K, float min_p, float max_p) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  __shared__ __attribute__((aligned(16))) double wllh[NWAVE];
  __shared__ __attribute__((aligned(16))) float red[NWAVE];

  const float* __restrict__ fu = F + (size_t)u * K;
  double llh_w = 0.0;
  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const float* __restrict__ fv = F + (size_t)indices[e] * K;
    float part = 0.f;
    for (int k = lane * 4; k < K; k += WAVE * 4) {
      part = dot4(ld4(fu + k), ld4(fv + k), part);
    }
    const float x = wave_allreduce_sum(part);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    if (lane == 0) llh_w += (double)log1pf(-p) + (double)x;
  }
  if (lane == 0) wllh[wid] = llh_w;
  __syncthreads();

  float p_fs = 0.f, p_ff = 0.f;
  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    const float4 a = ld4(fu + k);
    const float4 s = ld4(sumF + k);
    p_fs = dot4(a, s, p_fs);
    p_ff = dot4(a, a, p_ff);
  }
  const float fs = block_allreduce_sum(p_fs, red);
  const float ff = block_allreduce_sum(p_ff, red);
  if (tid == 0) {
    double t = 0.0;
#pragma unroll
    for (int wv = 0; wv < NWAVE; ++wv) t += wllh[wv];
    llh[u] = t + (double)(-fs) + (double)ff;
  }
}

// ------------------------------------------------------------------- K2
//
// Block-per-node, wave-per-edge.  v3 design: NO candidate materialization.
// Per edge element k all 16 trial dots accumulate directly in registers:
//   acc_j += clamp(fu_k + s_j*g_k, MIN_F, MAX_F) * fv_k
// (4 VALU per element per candidate, v_pk-packable), so fv streams from
// HBM exactly once and fu/g are staged once per node in LDS.  The v2
// kernel rebuilt all 16 candidate rows in LDS per 32-edge tile — for
// mean-degree-6 graphs that build cost 3-10x the edge work itself
// (123 ms/sweep measured vs the 48 ms v1; this version targets ~8 ms:
// one 37 GB fv pass + ~600 G VALU ops at K=5000/com-Amazon).
// After each edge, 16 wave-allreduces hand candidate j's dot to lane j,
// which owns its transcendentals (exp/log1p parallel across 16 lanes).

template <bool STAGED>
__global__ void __launch_bounds__(BLOCK, 3) k2_ls_v3(
    const float* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const float* __restrict__ grad, const double* __restrict__ llh,
    const int* __restrict__ order, const float* __restrict__ ladder,
    float* __restrict__ best, int n_local, int K, int n_ladder, float alpha,
    float min_p, float max_p, float min_f, float max_f) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  __shared__ __attribute__((aligned(16))) float s_lad[MAX_LS];
  __shared__ __attribute__((aligned(16))) double acc_llh[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float acc_nt[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float red[NWAVE];
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* fu_s = reinterpret_cast<float*>(smem);  // K floats when STAGED
  float* g_s = fu_s + K;                         // K floats when STAGED

  if (tid < MAX_LS) s_lad[tid] = (tid < n_ladder) ? ladder[tid] : 0.f;

  const float* __restrict__ fu_g = F + (size_t)u * K;
  const float* __restrict__ gu_g = grad + (size_t)u * K;
  if (STAGED) {
    for (int k = tid * 4; k < K; k += BLOCK * 4) {
      *reinterpret_cast<float4*>(fu_s + k) = ld4(fu_g + k);
      *reinterpret_cast<float4*>(g_s + k) = ld4(gu_g + k);
    }
  }
  __syncthreads();
  const float* __restrict__ fu = STAGED ? fu_s : fu_g;
  const float* __restrict__ gu = STAGED ? g_s : gu_g;

  float s[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) s[j] = s_lad[j];

  // this lane owns candidate jmine after wave_reduce16 (lanes with
  // (lane & 3) == 0; 4-way replicated otherwise)
  const int jmine = (((lane >> 5) & 1) << 3) | (((lane >> 4) & 1) << 2) |
                    (((lane >> 3) & 1) << 1) | ((lane >> 2) & 1);
  double llh_mine = 0.0;  // candidate jmine edge terms (lane&3)==0 lanes
  const v2f lo2 = {min_f, min_f}, hi2 = {max_f, max_f};

  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const float* __restrict__ fv = F + (size_t)indices[e] * K;
    v2f acc2[MAX_LS];
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) acc2[j] = v2f{0.f, 0.f};
    constexpr int U = 4;
    constexpr int KSTR = WAVE * 4;
    for (int k = lane * 4; k < K; k += U * KSTR) {
      float4 b[U];  // grouped global loads keep 4 HBM reads in flight
#pragma unroll
      for (int t = 0; t < U; ++t) {
        const int kk = k + t * KSTR;
        b[t] = ld4(fv + (kk < K ? kk : (K - 4)));  // clamped: always valid
      }
#pragma unroll
      for (int t = 0; t < U; ++t) {
        if (k + t * KSTR >= K) break;
        const float4 a4 = ld4(fu + k + t * KSTR);
        const float4 g4 = ld4(gu + k + t * KSTR);
        const v2f b0 = {b[t].x, b[t].y}, b1 = {b[t].z, b[t].w};
        const v2f a0 = {a4.x, a4.y}, a1 = {a4.z, a4.w};
        const v2f g0 = {g4.x, g4.y}, g1 = {g4.z, g4.w};
#pragma unroll
        for (int j = 0; j < MAX_LS; ++j) {
          const v2f sj = {s[j], s[j]};
          v2f t2 = acc2[j];
          t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g0, a0, lo2, hi2),
                                         b0, t2);
          t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g1, a1, lo2, hi2),
                                         b1, t2);
          acc2[j] = t2;
        }
      }
    }
    float acc[MAX_LS];
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) acc[j] = acc2[j].x + acc2[j].y;
    wave_reduce16(acc, lane);  // 17 shfls for all 16 sums
    if ((lane & 3) == 0) {
      const float x = acc[0];
      const float p = clamp_p(__expf(-x), min_p, max_p);
      llh_mine += (double)log1pf(-p) + (double)x;
    }
  }

  // node terms cand_j.(Fu - sumF) + grad.grad, block-strided k, once
  v2f accn2[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) accn2[j] = v2f{0.f, 0.f};
  float p_gg = 0.f;
  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    const float4 a = ld4(fu + k);
    const float4 g = ld4(gu + k);
    const float4 sf = ld4(sumF + k);
    const v2f a0 = {a.x, a.y}, a1 = {a.z, a.w};
    const v2f g0 = {g.x, g.y}, g1 = {g.z, g.w};
    const v2f d0 = {a.x - sf.x, a.y - sf.y};
    const v2f d1 = {a.z - sf.z, a.w - sf.w};
    p_gg = dot4(g, g, p_gg);
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) {
      const v2f sj = {s[j], s[j]};
      v2f t = accn2[j];
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g0, a0, lo2, hi2), d0,
                                    t);
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g1, a1, lo2, hi2), d1,
                                    t);
      accn2[j] = t;
    }
  }
  float accn[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) accn[j] = accn2[j].x + accn2[j].y;
  wave_reduce16(accn, lane);
  if ((lane & 3) == 0) {
    acc_nt[wid][jmine] = accn[0];
    acc_llh[wid][jmine] = llh_mine;
  }
  const float gg = block_allreduce_sum(p_gg, red);  // ends in __syncthreads

  // parallel Armijo selection: 16 threads each evaluate one candidate,
  // ballot picks the FIRST accepted (ladder is descending -> max step).
  if (wid == 0) {
    bool ok = false;
    if (lane < MAX_LS) {
      double trial = 0.0;
#pragma unroll
      for (int wv = 0; wv < NWAVE; ++wv)
        trial += acc_llh[wv][lane] + (double)acc_nt[wv][lane];
      ok = (lane < n_ladder) &&
           (trial >= llh[u] + (double)(alpha * s_lad[lane] * gg));
    }
    const unsigned long long bal = __ballot(ok);
    if (lane == 0)
      best[u] = bal ? s_lad[__ffsll((unsigned long long)bal) - 1] : 0.f;
  }
}

// ------------------------------------------------------------------- K3
//
// Projected commit: F_u <- clamp(F_u + s_u*grad_u, MIN_F, MAX_F) for nodes
// with an accepted step.  One block per node; rows with s==0 return without
// touching memory.  sumF is recomputed afterwards (column sum + allreduce)
// which keeps the sumF == colsum(F) invariant exact.

extern "C" __global__ void __launch_bounds__(BLOCK) k3_apply_step(
    float* __restrict__ F, const float* __restrict__ grad,
    const float* __restrict__ steps, int n_local, int K, float min_f,
    float max_f) {
  const int u = blockIdx.x;
  const float s = steps[u];
  if (s <= 0.f) return;
  float* __restrict__ fu = F + (size_t)u * K;
  const float* __restrict__ gu = grad + (size_t)u * K;
  for (int k = threadIdx.x * 4; k < K; k += BLOCK * 4) {
    const float4 a = ld4(fu + k);
    const float4 g = ld4(gu + k);
    *reinterpret_cast<float4*>(fu + k) =
        float4{fminf(fmaxf(fmaf(s, g.x, a.x), min_f), max_f),
               fminf(fmaxf(fmaf(s, g.y, a.y), min_f), max_f),
               fminf(fmaxf(fmaf(s, g.z, a.z), min_f), max_f),
               fminf(fmaxf(fmaf(s, g.w, a.w), min_f), max_f)};
  }
}


// --------------------------------------------------------- bf16 kernels
// Same structures as the fp32 kernels; 8-element (16 B) load granularity.
// NSLOT covers K <= NSLOT*2048.

template <int NSLOT>
__global__ void __launch_bounds__(BLOCK) k1_grad_llh_bf16_t(
    const u32* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, float* __restrict__ grad,
    double* __restrict__ llh, int n_local, int K, float min_p, float max_p) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* gacc = reinterpret_cast<float*>(smem);  // K floats
  float* red = reinterpret_cast<float*>(smem + (size_t)K * 4);

  const u32* __restrict__ fu = F + (size_t)u * (K / 2);
  f32x8 fu8[NSLOT];
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 8 + sl * (BLOCK * 8);
    if (k < K) {
      fu8[sl] = ld8bf(fu + k / 2);
      *reinterpret_cast<float4*>(gacc + k) = float4{0.f, 0.f, 0.f, 0.f};
      *reinterpret_cast<float4*>(gacc + k + 4) = float4{0.f, 0.f, 0.f, 0.f};
    } else {
      fu8[sl].a = fu8[sl].b = float4{0.f, 0.f, 0.f, 0.f};
    }
  }
  __syncthreads();

  double llh_acc = 0.0;

#pragma clang loop unroll(disable)
  for (long long e = e0; e < e1; ++e) {
    const u32* __restrict__ fv = F + (size_t)indices[e] * (K / 2);
    uint4 braw[NSLOT];  // raw bf16 fv kept in registers between dot and axpy
    float part = 0.f;
#pragma unroll
    for (int sl = 0; sl < NSLOT; ++sl) {
      const int k = tid * 8 + sl * (BLOCK * 8);
      braw[sl] = (k < K) ? *reinterpret_cast<const uint4*>(fv + k / 2)
                         : uint4{0u, 0u, 0u, 0u};
      f32x8 b;
      b.a = float4{bf_lo(braw[sl].x), bf_hi(braw[sl].x), bf_lo(braw[sl].y),
                   bf_hi(braw[sl].y)};
      b.b = float4{bf_lo(braw[sl].z), bf_hi(braw[sl].z), bf_lo(braw[sl].w),
                   bf_hi(braw[sl].w)};
      part = dot8(fu8[sl], b, part);
    }
    const float x = block_allreduce_sum(part, red);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    const float w = 1.f / (1.f - p);
    if (tid == 0) llh_acc += (double)log1pf(-p) + (double)x;
#pragma unroll
    for (int sl = 0; sl < NSLOT; ++sl) {
      const int k = tid * 8 + sl * (BLOCK * 8);
      if (k < K) {
        f32x8 b;
        b.a = float4{bf_lo(braw[sl].x), bf_hi(braw[sl].x), bf_lo(braw[sl].y),
                     bf_hi(braw[sl].y)};
        b.b = float4{bf_lo(braw[sl].z), bf_hi(braw[sl].z), bf_lo(braw[sl].w),
                     bf_hi(braw[sl].w)};
        float4 g0 = ld4(gacc + k);
        float4 g1 = ld4(gacc + k + 4);
        g0.x = fmaf(w, b.a.x, g0.x);
        g0.y = fmaf(w, b.a.y, g0.y);
        g0.z = fmaf(w, b.a.z, g0.z);
        g0.w = fmaf(w, b.a.w, g0.w);
        g1.x = fmaf(w, b.b.x, g1.x);
        g1.y = fmaf(w, b.b.y, g1.y);
        g1.z = fmaf(w, b.b.z, g1.z);
        g1.w = fmaf(w, b.b.w, g1.w);
        *reinterpret_cast<float4*>(gacc + k) = g0;
        *reinterpret_cast<float4*>(gacc + k + 4) = g1;
      }
    }
  }

  float p_fs = 0.f, p_ff = 0.f;
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 8 + sl * (BLOCK * 8);
    if (k < K) {
      const float4 s0 = ld4(sumF + k);
      const float4 s1 = ld4(sumF + k + 4);
      p_fs = dot4(fu8[sl].a, s0, p_fs);
      p_fs = dot4(fu8[sl].b, s1, p_fs);
      p_ff = dot8(fu8[sl], fu8[sl], p_ff);
    }
  }
  const float fs = block_allreduce_sum(p_fs, red);
  const float ff = block_allreduce_sum(p_ff, red);

  float* __restrict__ gout = grad + (size_t)u * K;
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 8 + sl * (BLOCK * 8);
    if (k < K) {
      const float4 g0 = ld4(gacc + k);
      const float4 g1 = ld4(gacc + k + 4);
      const float4 s0 = ld4(sumF + k);
      const float4 s1 = ld4(sumF + k + 4);
      *reinterpret_cast<float4*>(gout + k) =
          float4{g0.x - s0.x + fu8[sl].a.x, g0.y - s0.y + fu8[sl].a.y,
                 g0.z - s0.z + fu8[sl].a.z, g0.w - s0.w + fu8[sl].a.w};
      *reinterpret_cast<float4*>(gout + k + 4) =
          float4{g1.x - s1.x + fu8[sl].b.x, g1.y - s1.y + fu8[sl].b.y,
                 g1.z - s1.z + fu8[sl].b.z, g1.w - s1.w + fu8[sl].b.w};
    }
  }
  if (tid == 0) llh[u] = llh_acc + (double)(-fs) + (double)ff;
}

extern "C" __global__ void __launch_bounds__(BLOCK) k4_llh_only_bf16(
    const u32* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, double* __restrict__ llh, int n_local,
    int K, float min_p, float max_p) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  __shared__ __attribute__((aligned(16))) double wllh[NWAVE];
  __shared__ __attribute__((aligned(16))) float red[NWAVE];

  const u32* __restrict__ fu = F + (size_t)u * (K / 2);
  double llh_w = 0.0;
  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const u32* __restrict__ fv = F + (size_t)indices[e] * (K / 2);
    float part = 0.f;
    for (int k = lane * 8; k < K; k += WAVE * 8) {
      part = dot8(ld8bf(fu + k / 2), ld8bf(fv + k / 2), part);
    }
    const float x = wave_allreduce_sum(part);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    if (lane == 0) llh_w += (double)log1pf(-p) + (double)x;
  }
  if (lane == 0) wllh[wid] = llh_w;
  __syncthreads();

  float p_fs = 0.f, p_ff = 0.f;
  for (int k = tid * 8; k < K; k += BLOCK * 8) {
    const f32x8 a = ld8bf(fu + k / 2);
    p_fs = dot4(a.a, ld4(sumF + k), p_fs);
    p_fs = dot4(a.b, ld4(sumF + k + 4), p_fs);
    p_ff = dot8(a, a, p_ff);
  }
  const float fs = block_allreduce_sum(p_fs, red);
  const float ff = block_allreduce_sum(p_ff, red);
  if (tid == 0) {
    double t = 0.0;
#pragma unroll
    for (int wv = 0; wv < NWAVE; ++wv) t += wllh[wv];
    llh[u] = t + (double)(-fs) + (double)ff;
  }
}

// ------------------------------------------------------------------- KF
//
// Fused K1+K2: one block per node runs the gradient pass (phase A, K1
// structure) and the 16-candidate line search (phase B, K2 structure)
// back-to-back.  grad_u is finalized IN LDS and consumed there by phase B
// — the separate-kernel flow wrote grad to HBM in K1 and re-staged it in
// K2 (2 x N x K x 4B), re-staged fu (N x K x 4B) and round-tripped llh.
// grad still goes to HBM once (K3 needs it); llh[u] likewise.
// Valid under Jacobi semantics because F is read-only for the whole
// kernel: node u's trial evaluations only need u's OWN gradient.
// LDS: gacc/grad K floats + fu_s K floats (+1 KB statics) -> K <= 8192
// fp32 via the NSLOT template (the K=5000 / K<=500 headline configs).

// Phase A as a shared device function (used by kf_fused_t and kf_mfma_t):
// on return gacc = the FINAL gradient row (also written to HBM), fu_s = a
// copy of F_u, *s_llh_u = llh[u] (also written to HBM).  Caller must
// __syncthreads() after phase A inside this function's tail (done here).
template <int NSLOT>
__device__ __forceinline__ void kf_phase_a(
    const float* __restrict__ F, const int* __restrict__ indices,
    const float* __restrict__ sumF, int u, long long e0, long long e1, int K,
    float min_p, float max_p, float* __restrict__ gacc,
    float* __restrict__ fu_s, float* red, double* s_llh_u,
    float* __restrict__ grad, double* __restrict__ llh) {
  const int tid = threadIdx.x;
  const float* __restrict__ fu_g = F + (size_t)u * K;
  float4 fu4[NSLOT];
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 4 + sl * (BLOCK * 4);
    fu4[sl] = (k < K) ? ld4(fu_g + k) : float4{0.f, 0.f, 0.f, 0.f};
    if (k < K) {
      *reinterpret_cast<float4*>(gacc + k) = float4{0.f, 0.f, 0.f, 0.f};
      *reinterpret_cast<float4*>(fu_s + k) = fu4[sl];
    }
  }
  __syncthreads();

  double llh_acc = 0.0;
#pragma clang loop unroll(disable)
  for (long long e = e0; e < e1; ++e) {
    const float* __restrict__ fv = F + (size_t)indices[e] * K;
    float4 b[NSLOT];
    float part = 0.f;
#pragma unroll
    for (int sl = 0; sl < NSLOT; ++sl) {
      const int k = tid * 4 + sl * (BLOCK * 4);
      b[sl] = (k < K) ? ld4(fv + k) : float4{0.f, 0.f, 0.f, 0.f};
      part = dot4(fu4[sl], b[sl], part);
    }
    const float x = block_allreduce_sum(part, red);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    const float w = 1.f / (1.f - p);
    if (tid == 0) llh_acc += (double)log1pf(-p) + (double)x;
#pragma unroll
    for (int sl = 0; sl < NSLOT; ++sl) {
      const int k = tid * 4 + sl * (BLOCK * 4);
      if (k < K) {
        float4 g = ld4(gacc + k);
        g.x = fmaf(w, b[sl].x, g.x);
        g.y = fmaf(w, b[sl].y, g.y);
        g.z = fmaf(w, b[sl].z, g.z);
        g.w = fmaf(w, b[sl].w, g.w);
        *reinterpret_cast<float4*>(gacc + k) = g;
      }
    }
  }

  float p_fs = 0.f, p_ff = 0.f;
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 4 + sl * (BLOCK * 4);
    if (k < K) {
      const float4 sv = ld4(sumF + k);
      p_fs = dot4(fu4[sl], sv, p_fs);
      p_ff = dot4(fu4[sl], fu4[sl], p_ff);
    }
  }
  const float fs = block_allreduce_sum(p_fs, red);
  const float ff = block_allreduce_sum(p_ff, red);

  // finalize grad in LDS + write to HBM (K3 needs it); publish llh_u
  float* __restrict__ gout = grad + (size_t)u * K;
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 4 + sl * (BLOCK * 4);
    if (k < K) {
      const float4 ga = ld4(gacc + k);
      const float4 sv = ld4(sumF + k);
      // same evaluation order as K1 ((gacc - sumF) + fu) for bitwise parity
      const float4 g = float4{ga.x - sv.x + fu4[sl].x, ga.y - sv.y + fu4[sl].y,
                              ga.z - sv.z + fu4[sl].z,
                              ga.w - sv.w + fu4[sl].w};
      *reinterpret_cast<float4*>(gacc + k) = g;
      *reinterpret_cast<float4*>(gout + k) = g;
    }
  }
  if (tid == 0) {
    const double l = llh_acc + (double)(-fs) + (double)ff;
    llh[u] = l;
    *s_llh_u = l;
  }
  __syncthreads();  // gacc = final grad, fu_s, s_llh_u visible
}

// Shared tail: per-candidate node terms, Armijo threshold and step pick.
// acc_llh[wid][j] must already hold each wave's candidate-j edge-term
// partial (the two phase-B variants fill it in different lane layouts but
// the SAME [wave][candidate] indexing).
__device__ __forceinline__ void kf_tail(
    const float* __restrict__ fu, const float* __restrict__ gu,
    const float* __restrict__ sumF, const float* s_lad,
    double (*acc_llh)[MAX_LS], float (*acc_nt)[MAX_LS], float* red,
    const double* s_llh_u, float* __restrict__ best, int u, int K,
    int n_ladder, float alpha, float min_f, float max_f) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  const int jmine = (((lane >> 5) & 1) << 3) | (((lane >> 4) & 1) << 2) |
                    (((lane >> 3) & 1) << 1) | ((lane >> 2) & 1);
  float s[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) s[j] = s_lad[j];
  const v2f lo2 = {min_f, min_f}, hi2 = {max_f, max_f};

  v2f accn2[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) accn2[j] = v2f{0.f, 0.f};
  float p_gg = 0.f;
  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    const float4 a = ld4(fu + k);
    const float4 g = ld4(gu + k);
    const float4 sf = ld4(sumF + k);
    const v2f a0 = {a.x, a.y}, a1 = {a.z, a.w};
    const v2f g0 = {g.x, g.y}, g1 = {g.z, g.w};
    const v2f d0 = {a.x - sf.x, a.y - sf.y};
    const v2f d1 = {a.z - sf.z, a.w - sf.w};
    p_gg = dot4(g, g, p_gg);
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) {
      const v2f sj = {s[j], s[j]};
      v2f t = accn2[j];
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g0, a0, lo2, hi2), d0,
                                    t);
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g1, a1, lo2, hi2), d1,
                                    t);
      accn2[j] = t;
    }
  }
  float accn[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) accn[j] = accn2[j].x + accn2[j].y;
  wave_reduce16(accn, lane);
  if ((lane & 3) == 0) acc_nt[wid][jmine] = accn[0];
  const float gg = block_allreduce_sum(p_gg, red);  // ends in __syncthreads

  if (wid == 0) {
    bool ok = false;
    if (lane < MAX_LS) {
      double trial = 0.0;
#pragma unroll
      for (int wv = 0; wv < NWAVE; ++wv)
        trial += acc_llh[wv][lane] + (double)acc_nt[wv][lane];
      ok = (lane < n_ladder) &&
           (trial >= *s_llh_u + (double)(alpha * s_lad[lane] * gg));
    }
    const unsigned long long bal = __ballot(ok);
    if (lane == 0)
      best[u] = bal ? s_lad[__ffsll((unsigned long long)bal) - 1] : 0.f;
  }
}

template <int NSLOT>
__global__ void __launch_bounds__(BLOCK, 3) kf_fused_t(
    const float* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, float* __restrict__ grad,
    double* __restrict__ llh, const float* __restrict__ ladder,
    float* __restrict__ best, int n_local, int K, int n_ladder, float alpha,
    float min_p, float max_p, float min_f, float max_f) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  __shared__ __attribute__((aligned(16))) float s_lad[MAX_LS];
  __shared__ __attribute__((aligned(16))) double acc_llh[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float acc_nt[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float red[NWAVE];
  __shared__ double s_llh_u;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* gacc = reinterpret_cast<float*>(smem);  // K floats: grad after A
  float* fu_s = gacc + K;                        // K floats: fu copy

  if (tid < MAX_LS) s_lad[tid] = (tid < n_ladder) ? ladder[tid] : 0.f;

  kf_phase_a<NSLOT>(F, indices, sumF, u, e0, e1, K, min_p, max_p, gacc, fu_s,
                    red, &s_llh_u, grad, llh);

  // ---------------- phase B: 16-candidate line search (K2 structure)
  const float* __restrict__ fu = fu_s;
  const float* __restrict__ gu = gacc;
  float s[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) s[j] = s_lad[j];
  const int jmine = (((lane >> 5) & 1) << 3) | (((lane >> 4) & 1) << 2) |
                    (((lane >> 3) & 1) << 1) | ((lane >> 2) & 1);
  double llh_mine = 0.0;
  const v2f lo2 = {min_f, min_f}, hi2 = {max_f, max_f};

  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const float* __restrict__ fv = F + (size_t)indices[e] * K;
    v2f acc2[MAX_LS];
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) acc2[j] = v2f{0.f, 0.f};
    constexpr int U = 4;
    constexpr int KSTR = WAVE * 4;
    for (int k = lane * 4; k < K; k += U * KSTR) {
      float4 b[U];
#pragma unroll
      for (int t = 0; t < U; ++t) {
        const int kk = k + t * KSTR;
        b[t] = ld4(fv + (kk < K ? kk : (K - 4)));
      }
#pragma unroll
      for (int t = 0; t < U; ++t) {
        if (k + t * KSTR >= K) break;
        const float4 a4 = ld4(fu + k + t * KSTR);
        const float4 g4 = ld4(gu + k + t * KSTR);
        const v2f b0 = {b[t].x, b[t].y}, b1 = {b[t].z, b[t].w};
        const v2f a0 = {a4.x, a4.y}, a1 = {a4.z, a4.w};
        const v2f g0 = {g4.x, g4.y}, g1 = {g4.z, g4.w};
#pragma unroll
        for (int j = 0; j < MAX_LS; ++j) {
          const v2f sj = {s[j], s[j]};
          v2f t2 = acc2[j];
          t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g0, a0, lo2, hi2),
                                         b0, t2);
          t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g1, a1, lo2, hi2),
                                         b1, t2);
          acc2[j] = t2;
        }
      }
    }
    float accf[MAX_LS];
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) accf[j] = acc2[j].x + acc2[j].y;
    wave_reduce16(accf, lane);
    if ((lane & 3) == 0) {
      const float x = accf[0];
      const float p = clamp_p(__expf(-x), min_p, max_p);
      llh_mine += (double)log1pf(-p) + (double)x;
    }
  }
  if ((lane & 3) == 0) acc_llh[wid][jmine] = llh_mine;

  kf_tail(fu, gu, sumF, s_lad, acc_llh, acc_nt, red, &s_llh_u, best, u, K,
          n_ladder, alpha, min_f, max_f);
}

// ------------------------------------------------------------ KF (MFMA)
//
// MFMA phase-B variant for HIGH-DEGREE nodes (the degree-descending launch
// order makes them a prefix; the Python side splits at deg >= 16).  The
// candidate scoring is run as a GEMM: per 16-edge tile, score[e][j] =
// sum_k fv[e][k] * clamp(fu[k] + s_j*g[k]) via v_mfma_f32_16x16x4_f32
// (exact fp32, ISA §12.10).  Candidate fragments are built once per
// 64-element k-chunk per wave and amortized over up to TCAP edge tiles,
// with accumulators resident in VGPRs across k-chunks.  vs the direct
// kernel this removes BOTH per-(edge,candidate,k) VALU FMAs and the
// per-edge 17-shuffle reductions — the measured issue-port bottleneck
// (profiles/r01_kernel_opt_log.md: VALUBusy 80% at the occupancy cap) —
// leaving fv streaming as the floor.
//
// Work split: the K dimension is quartered across the 4 waves (every wave
// computes partial scores for EVERY edge tile on its own k-quarter, then
// the quarters combine in a 4 KB LDS buffer per tile).  A tile-per-wave
// split would idle 3 of the 4 waves (= 3 of 4 SIMDs/CU) for every node
// with deg < 128 — most of the deg>=16 population on power-law graphs.
#define KF_TG 4  // edge tiles per accumulator group (4 x f32x4v VGPRs each)

template <int NSLOT>
__global__ void __launch_bounds__(BLOCK, 3) kf_mfma_t(
    const float* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, float* __restrict__ grad,
    double* __restrict__ llh, const float* __restrict__ ladder,
    float* __restrict__ best, int n_local, int K, int n_ladder, float alpha,
    float min_p, float max_p, float min_f, float max_f) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  __shared__ __attribute__((aligned(16))) float s_lad[MAX_LS];
  __shared__ __attribute__((aligned(16))) double acc_llh[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float acc_nt[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float red[NWAVE];
  __shared__ double s_llh_u;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* gacc = reinterpret_cast<float*>(smem);  // K floats: grad after A
  float* fu_s = gacc + K;                        // K floats: fu copy

  if (tid < MAX_LS) s_lad[tid] = (tid < n_ladder) ? ladder[tid] : 0.f;

  kf_phase_a<NSLOT>(F, indices, sumF, u, e0, e1, K, min_p, max_p, gacc, fu_s,
                    red, &s_llh_u, grad, llh);

  // ---------------- phase B: MFMA 16-edge x 16-candidate tiles
  __shared__ __attribute__((aligned(16))) float redt[NWAVE][WAVE][4];
  const int deg = (int)(e1 - e0);
  const int ntiles = (deg + 15) >> 4;
  const int arow = lane & 15;   // A row (edge-in-tile) and C column select
  const int kgrp = lane >> 4;   // k-group 0..3 (one float of 4 per step)
  const float s_j = s_lad[arow];  // B column j == lane & 15
  // K is partitioned into 16-element windows (one 4-step MFMA burst each,
  // fed from ONE float4 A-load per lane); each wave owns a contiguous
  // quarter of the windows.  K % 4 == 0, so a window's float4 at
  // w*16 + kgrp*4 is either fully inside [0,K) or fully outside.
  const int nwin = (K + 15) >> 4;
  const int wq = (nwin + NWAVE - 1) / NWAVE;
  const int ww0 = wid * wq;
  const int ww1 = min(nwin, ww0 + wq);
  constexpr int WC = 4;  // windows per candidate-fragment burst (64 k)
  double llh_j = 0.0;

  for (int tbase = 0; tbase < ntiles; tbase += KF_TG) {
    const int nt = min(KF_TG, ntiles - tbase);
    const float* aptr[KF_TG];
#pragma unroll
    for (int t = 0; t < KF_TG; ++t) {
      const long long e = e0 + (long long)(tbase + t) * 16 + arow;
      // padded rows get NO pointer: a clamped duplicate of the last edge
      // row cost ~3x extra load traffic at mean degree 5.5
      aptr[t] = (e < e1) ? F + (size_t)indices[e] * K : nullptr;
    }
    f32x4v acc[KF_TG];
#pragma unroll
    for (int t = 0; t < KF_TG; ++t) acc[t] = f32x4v{0.f, 0.f, 0.f, 0.f};

    // Per window w, MFMA step i=0..3 uses the k-assignment
    //   k(lane, i) = w*16 + (lane>>4)*4 + i
    // — a bijection onto the window shared by A and B, so ONE float4
    // A-load per lane feeds 4 MFMA steps (a scalar per-step gather was
    // 4x the load instructions and lost to the direct kernel).
    for (int wb = ww0; wb < ww1; wb += WC) {
      float bfrag[WC * 4];
#pragma unroll
      for (int c = 0; c < WC * 4; ++c) {
        const int k = (wb + (c >> 2)) * 16 + kgrp * 4 + (c & 3);
        bfrag[c] = (wb + (c >> 2) < ww1 && k < K)
                       ? __builtin_amdgcn_fmed3f(
                             fmaf(s_j, gacc[k], fu_s[k]), min_f, max_f)
                       : 0.f;
      }
#pragma unroll
      for (int t = 0; t < KF_TG; ++t) {
        if (t >= nt) break;
#pragma unroll
        for (int wi = 0; wi < WC; ++wi) {
          const int k4 = (wb + wi) * 16 + kgrp * 4;
          const float4 a4 = (wb + wi < ww1 && k4 + 3 < K && aptr[t])
                                ? ld4(aptr[t] + k4)
                                : float4{0.f, 0.f, 0.f, 0.f};
          acc[t] = __builtin_amdgcn_mfma_f32_16x16x4f32(a4.x, bfrag[wi * 4],
                                                        acc[t], 0, 0, 0);
          acc[t] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a4.y, bfrag[wi * 4 + 1], acc[t], 0, 0, 0);
          acc[t] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a4.z, bfrag[wi * 4 + 2], acc[t], 0, 0, 0);
          acc[t] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a4.w, bfrag[wi * 4 + 3], acc[t], 0, 0, 0);
        }
      }
    }
    // combine the four k-quarters per tile and run the LLH epilogue on a
    // rotating wave (spreads the exp/log work across SIMDs)
#pragma unroll
    for (int t = 0; t < KF_TG; ++t) {
      if (t >= nt) break;
      __syncthreads();  // redt free / previous tile consumed
      *reinterpret_cast<float4*>(&redt[wid][lane][0]) =
          float4{acc[t][0], acc[t][1], acc[t][2], acc[t][3]};
      __syncthreads();
      if (wid == ((tbase + t) & (NWAVE - 1))) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int erow = (tbase + t) * 16 + kgrp * 4 + r;
          if (erow < deg) {
            const float x = redt[0][lane][r] + redt[1][lane][r] +
                            redt[2][lane][r] + redt[3][lane][r];
            const float p = clamp_p(__expf(-x), min_p, max_p);
            llh_j += (double)log1pf(-p) + (double)x;
          }
        }
      }
    }
  }
  // fold the 4 row-groups: lanes l, l^16, l^32, l^48 share candidate l&15
  llh_j += __shfl_xor(llh_j, 16, WAVE);
  llh_j += __shfl_xor(llh_j, 32, WAVE);
  if (lane < MAX_LS) acc_llh[wid][lane] = llh_j;
  __syncthreads();  // all epilogue reads done before kf_tail reuses LDS

  kf_tail(fu_s, gacc, sumF, s_lad, acc_llh, acc_nt, red, &s_llh_u, best, u,
          K, n_ladder, alpha, min_f, max_f);
}

// bf16 fused K1+K2 (same structure as kf_fused_t): gacc fp32 K*4B +
// fu raw bf16 K*2B LDS -> covers K <= 16384 via the NSLOT template.
// bf16 phase A as a shared device function (kf_fused_bf16_t /
// kf_mfma_bf16_t): gacc = final fp32 grad, fu_s = raw bf16 F_u copy.
template <int NSLOT>
__device__ __forceinline__ void kf_phase_a_bf16(
    const u32* __restrict__ F, const int* __restrict__ indices,
    const float* __restrict__ sumF, int u, long long e0, long long e1, int K,
    float min_p, float max_p, float* __restrict__ gacc, u32* __restrict__ fu_s,
    float* red, double* s_llh_u, float* __restrict__ grad,
    double* __restrict__ llh) {
  const int tid = threadIdx.x;
  const u32* __restrict__ fu_g = F + (size_t)u * (K / 2);
  f32x8 fu8[NSLOT];
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 8 + sl * (BLOCK * 8);
    if (k < K) {
      const uint4 raw = *reinterpret_cast<const uint4*>(fu_g + k / 2);
      *reinterpret_cast<uint4*>(fu_s + k / 2) = raw;
      fu8[sl].a = float4{bf_lo(raw.x), bf_hi(raw.x), bf_lo(raw.y),
                         bf_hi(raw.y)};
      fu8[sl].b = float4{bf_lo(raw.z), bf_hi(raw.z), bf_lo(raw.w),
                         bf_hi(raw.w)};
      *reinterpret_cast<float4*>(gacc + k) = float4{0.f, 0.f, 0.f, 0.f};
      *reinterpret_cast<float4*>(gacc + k + 4) = float4{0.f, 0.f, 0.f, 0.f};
    } else {
      fu8[sl].a = fu8[sl].b = float4{0.f, 0.f, 0.f, 0.f};
    }
  }
  __syncthreads();

  double llh_acc = 0.0;
#pragma clang loop unroll(disable)
  for (long long e = e0; e < e1; ++e) {
    const u32* __restrict__ fv = F + (size_t)indices[e] * (K / 2);
    uint4 braw[NSLOT];
    float part = 0.f;
#pragma unroll
    for (int sl = 0; sl < NSLOT; ++sl) {
      const int k = tid * 8 + sl * (BLOCK * 8);
      braw[sl] = (k < K) ? *reinterpret_cast<const uint4*>(fv + k / 2)
                         : uint4{0u, 0u, 0u, 0u};
      f32x8 b;
      b.a = float4{bf_lo(braw[sl].x), bf_hi(braw[sl].x), bf_lo(braw[sl].y),
                   bf_hi(braw[sl].y)};
      b.b = float4{bf_lo(braw[sl].z), bf_hi(braw[sl].z), bf_lo(braw[sl].w),
                   bf_hi(braw[sl].w)};
      part = dot8(fu8[sl], b, part);
    }
    const float x = block_allreduce_sum(part, red);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    const float w = 1.f / (1.f - p);
    if (tid == 0) llh_acc += (double)log1pf(-p) + (double)x;
#pragma unroll
    for (int sl = 0; sl < NSLOT; ++sl) {
      const int k = tid * 8 + sl * (BLOCK * 8);
      if (k < K) {
        f32x8 b;
        b.a = float4{bf_lo(braw[sl].x), bf_hi(braw[sl].x), bf_lo(braw[sl].y),
                     bf_hi(braw[sl].y)};
        b.b = float4{bf_lo(braw[sl].z), bf_hi(braw[sl].z), bf_lo(braw[sl].w),
                     bf_hi(braw[sl].w)};
        float4 g0 = ld4(gacc + k);
        float4 g1 = ld4(gacc + k + 4);
        g0.x = fmaf(w, b.a.x, g0.x);
        g0.y = fmaf(w, b.a.y, g0.y);
        g0.z = fmaf(w, b.a.z, g0.z);
        g0.w = fmaf(w, b.a.w, g0.w);
        g1.x = fmaf(w, b.b.x, g1.x);
        g1.y = fmaf(w, b.b.y, g1.y);
        g1.z = fmaf(w, b.b.z, g1.z);
        g1.w = fmaf(w, b.b.w, g1.w);
        *reinterpret_cast<float4*>(gacc + k) = g0;
        *reinterpret_cast<float4*>(gacc + k + 4) = g1;
      }
    }
  }

  float p_fs = 0.f, p_ff = 0.f;
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 8 + sl * (BLOCK * 8);
    if (k < K) {
      const float4 s0 = ld4(sumF + k);
      const float4 s1 = ld4(sumF + k + 4);
      p_fs = dot4(fu8[sl].a, s0, p_fs);
      p_fs = dot4(fu8[sl].b, s1, p_fs);
      p_ff = dot8(fu8[sl], fu8[sl], p_ff);
    }
  }
  const float fs = block_allreduce_sum(p_fs, red);
  const float ff = block_allreduce_sum(p_ff, red);

  float* __restrict__ gout = grad + (size_t)u * K;
#pragma unroll
  for (int sl = 0; sl < NSLOT; ++sl) {
    const int k = tid * 8 + sl * (BLOCK * 8);
    if (k < K) {
      const float4 g0a = ld4(gacc + k);
      const float4 g1a = ld4(gacc + k + 4);
      const float4 s0 = ld4(sumF + k);
      const float4 s1 = ld4(sumF + k + 4);
      const float4 g0 =
          float4{g0a.x - s0.x + fu8[sl].a.x, g0a.y - s0.y + fu8[sl].a.y,
                 g0a.z - s0.z + fu8[sl].a.z, g0a.w - s0.w + fu8[sl].a.w};
      const float4 g1 =
          float4{g1a.x - s1.x + fu8[sl].b.x, g1a.y - s1.y + fu8[sl].b.y,
                 g1a.z - s1.z + fu8[sl].b.z, g1a.w - s1.w + fu8[sl].b.w};
      *reinterpret_cast<float4*>(gacc + k) = g0;
      *reinterpret_cast<float4*>(gacc + k + 4) = g1;
      *reinterpret_cast<float4*>(gout + k) = g0;
      *reinterpret_cast<float4*>(gout + k + 4) = g1;
    }
  }
  if (tid == 0) {
    const double l = llh_acc + (double)(-fs) + (double)ff;
    llh[u] = l;
    *s_llh_u = l;
  }
  __syncthreads();
}

// bf16 phase A without register-resident fu/fv (K beyond the NSLOT
// register budget, e.g. the com-Amazon K=25000 config): fu lives in LDS
// only and the weighted accumulate re-reads fv from global (L2-hot, the
// block just streamed it for the dot).  Covers K up to the LDS cap
// (gacc fp32 + fu bf16 = K*6 bytes <= 160 KB -> K <= 26000 padded).
__device__ __forceinline__ void kf_phase_a_bf16_lds(
    const u32* __restrict__ F, const int* __restrict__ indices,
    const float* __restrict__ sumF, int u, long long e0, long long e1, int K,
    float min_p, float max_p, float* __restrict__ gacc, u32* __restrict__ fu_s,
    float* red, double* s_llh_u, float* __restrict__ grad,
    double* __restrict__ llh) {
  const int tid = threadIdx.x;
  const u32* __restrict__ fu_g = F + (size_t)u * (K / 2);
  for (int k = tid * 8; k < K; k += BLOCK * 8) {
    *reinterpret_cast<uint4*>(fu_s + k / 2) =
        *reinterpret_cast<const uint4*>(fu_g + k / 2);
    *reinterpret_cast<float4*>(gacc + k) = float4{0.f, 0.f, 0.f, 0.f};
    *reinterpret_cast<float4*>(gacc + k + 4) = float4{0.f, 0.f, 0.f, 0.f};
  }
  __syncthreads();

  double llh_acc = 0.0;
#pragma clang loop unroll(disable)
  for (long long e = e0; e < e1; ++e) {
    const u32* __restrict__ fv = F + (size_t)indices[e] * (K / 2);
    float part = 0.f;
    for (int k = tid * 8; k < K; k += BLOCK * 8) {
      const f32x8 a = ld8bf(fu_s + k / 2);
      const f32x8 b = ld8bf(fv + k / 2);
      part = dot8(a, b, part);
    }
    const float x = block_allreduce_sum(part, red);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    const float w = 1.f / (1.f - p);
    if (tid == 0) llh_acc += (double)log1pf(-p) + (double)x;
    for (int k = tid * 8; k < K; k += BLOCK * 8) {
      const f32x8 b = ld8bf(fv + k / 2);  // L2 re-read
      float4 g0 = ld4(gacc + k);
      float4 g1 = ld4(gacc + k + 4);
      g0.x = fmaf(w, b.a.x, g0.x);
      g0.y = fmaf(w, b.a.y, g0.y);
      g0.z = fmaf(w, b.a.z, g0.z);
      g0.w = fmaf(w, b.a.w, g0.w);
      g1.x = fmaf(w, b.b.x, g1.x);
      g1.y = fmaf(w, b.b.y, g1.y);
      g1.z = fmaf(w, b.b.z, g1.z);
      g1.w = fmaf(w, b.b.w, g1.w);
      *reinterpret_cast<float4*>(gacc + k) = g0;
      *reinterpret_cast<float4*>(gacc + k + 4) = g1;
    }
  }

  float p_fs = 0.f, p_ff = 0.f;
  for (int k = tid * 8; k < K; k += BLOCK * 8) {
    const f32x8 a = ld8bf(fu_s + k / 2);
    const float4 s0 = ld4(sumF + k);
    const float4 s1 = ld4(sumF + k + 4);
    p_fs = dot4(a.a, s0, p_fs);
    p_fs = dot4(a.b, s1, p_fs);
    p_ff = dot8(a, a, p_ff);
  }
  const float fs = block_allreduce_sum(p_fs, red);
  const float ff = block_allreduce_sum(p_ff, red);

  float* __restrict__ gout = grad + (size_t)u * K;
  for (int k = tid * 8; k < K; k += BLOCK * 8) {
    const f32x8 a = ld8bf(fu_s + k / 2);
    const float4 g0a = ld4(gacc + k);
    const float4 g1a = ld4(gacc + k + 4);
    const float4 s0 = ld4(sumF + k);
    const float4 s1 = ld4(sumF + k + 4);
    const float4 g0 = float4{g0a.x - s0.x + a.a.x, g0a.y - s0.y + a.a.y,
                             g0a.z - s0.z + a.a.z, g0a.w - s0.w + a.a.w};
    const float4 g1 = float4{g1a.x - s1.x + a.b.x, g1a.y - s1.y + a.b.y,
                             g1a.z - s1.z + a.b.z, g1a.w - s1.w + a.b.w};
    *reinterpret_cast<float4*>(gacc + k) = g0;
    *reinterpret_cast<float4*>(gacc + k + 4) = g1;
    *reinterpret_cast<float4*>(gout + k) = g0;
    *reinterpret_cast<float4*>(gout + k + 4) = g1;
  }
  if (tid == 0) {
    const double l = llh_acc + (double)(-fs) + (double)ff;
    llh[u] = l;
    *s_llh_u = l;
  }
  __syncthreads();
}

// bf16 shared tail: node terms + Armijo pick (fu raw bf16 from LDS).
__device__ __forceinline__ void kf_tail_bf16(
    const u32* __restrict__ fu, const float* __restrict__ gu,
    const float* __restrict__ sumF, const float* s_lad,
    double (*acc_llh)[MAX_LS], float (*acc_nt)[MAX_LS], float* red,
    const double* s_llh_u, float* __restrict__ best, int u, int K,
    int n_ladder, float alpha, float min_f, float max_f) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  const int jmine = (((lane >> 5) & 1) << 3) | (((lane >> 4) & 1) << 2) |
                    (((lane >> 3) & 1) << 1) | ((lane >> 2) & 1);
  float s[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) s[j] = s_lad[j];
  const v2f lo2 = {min_f, min_f}, hi2 = {max_f, max_f};

  v2f accn2[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) accn2[j] = v2f{0.f, 0.f};
  float p_gg = 0.f;
  for (int k = tid * 8; k < K; k += BLOCK * 8) {
    const uint4 araw = *reinterpret_cast<const uint4*>(fu + k / 2);
    const v2f a0 = bf2(araw.x), a1 = bf2(araw.y), a2 = bf2(araw.z),
              a3 = bf2(araw.w);
    const float4 gA = ld4(gu + k);
    const float4 gB = ld4(gu + k + 4);
    const v2f g0 = {gA.x, gA.y}, g1 = {gA.z, gA.w};
    const v2f g2 = {gB.x, gB.y}, g3 = {gB.z, gB.w};
    const float4 sA = ld4(sumF + k);
    const float4 sB = ld4(sumF + k + 4);
    const v2f d0 = a0 - v2f{sA.x, sA.y}, d1 = a1 - v2f{sA.z, sA.w};
    const v2f d2 = a2 - v2f{sB.x, sB.y}, d3 = a3 - v2f{sB.z, sB.w};
    p_gg = dot4(gA, gA, dot4(gB, gB, p_gg));
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) {
      const v2f sj = {s[j], s[j]};
      v2f t = accn2[j];
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g0, a0, lo2, hi2), d0, t);
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g1, a1, lo2, hi2), d1, t);
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g2, a2, lo2, hi2), d2, t);
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g3, a3, lo2, hi2), d3, t);
      accn2[j] = t;
    }
  }
  float accn[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) accn[j] = accn2[j].x + accn2[j].y;
  wave_reduce16(accn, lane);
  if ((lane & 3) == 0) acc_nt[wid][jmine] = accn[0];
  const float gg = block_allreduce_sum(p_gg, red);  // ends in __syncthreads

  if (wid == 0) {
    bool ok = false;
    if (lane < MAX_LS) {
      double trial = 0.0;
#pragma unroll
      for (int wv = 0; wv < NWAVE; ++wv)
        trial += acc_llh[wv][lane] + (double)acc_nt[wv][lane];
      ok = (lane < n_ladder) &&
           (trial >= *s_llh_u + (double)(alpha * s_lad[lane] * gg));
    }
    const unsigned long long bal = __ballot(ok);
    if (lane == 0)
      best[u] = bal ? s_lad[__ffsll((unsigned long long)bal) - 1] : 0.f;
  }
}

template <int NSLOT>
__global__ void __launch_bounds__(BLOCK, 3) kf_fused_bf16_t(
    const u32* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, float* __restrict__ grad,
    double* __restrict__ llh, const float* __restrict__ ladder,
    float* __restrict__ best, int n_local, int K, int n_ladder, float alpha,
    float min_p, float max_p, float min_f, float max_f) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  __shared__ __attribute__((aligned(16))) float s_lad[MAX_LS];
  __shared__ __attribute__((aligned(16))) double acc_llh[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float acc_nt[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float red[NWAVE];
  __shared__ double s_llh_u;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* gacc = reinterpret_cast<float*>(smem);               // K floats
  u32* fu_s = reinterpret_cast<u32*>(smem + (size_t)K * 4);   // K/2 u32

  if (tid < MAX_LS) s_lad[tid] = (tid < n_ladder) ? ladder[tid] : 0.f;

  kf_phase_a_bf16<NSLOT>(F, indices, sumF, u, e0, e1, K, min_p, max_p, gacc,
                         fu_s, red, &s_llh_u, grad, llh);

  // ---------------- phase B (k2_ls_v3_bf16 structure; fu/g from LDS)
  const u32* __restrict__ fu = fu_s;
  const float* __restrict__ gu = gacc;
  float s[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) s[j] = s_lad[j];
  const int jmine = (((lane >> 5) & 1) << 3) | (((lane >> 4) & 1) << 2) |
                    (((lane >> 3) & 1) << 1) | ((lane >> 2) & 1);
  double llh_mine = 0.0;
  const v2f lo2 = {min_f, min_f}, hi2 = {max_f, max_f};

  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const u32* __restrict__ fv = F + (size_t)indices[e] * (K / 2);
    v2f acc[MAX_LS];
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) acc[j] = v2f{0.f, 0.f};
    constexpr int KSTR = WAVE * 8;
    for (int k = lane * 8; k < K; k += KSTR) {
      const uint4 braw = *reinterpret_cast<const uint4*>(fv + k / 2);
      const uint4 araw = *reinterpret_cast<const uint4*>(fu + k / 2);
      const float4 gA = ld4(gu + k);
      const float4 gB = ld4(gu + k + 4);
      const v2f b0 = bf2(braw.x), b1 = bf2(braw.y), b2 = bf2(braw.z),
                b3 = bf2(braw.w);
      const v2f a0 = bf2(araw.x), a1 = bf2(araw.y), a2 = bf2(araw.z),
                a3 = bf2(araw.w);
      const v2f g0 = {gA.x, gA.y}, g1 = {gA.z, gA.w};
      const v2f g2 = {gB.x, gB.y}, g3 = {gB.z, gB.w};
#pragma unroll
      for (int j = 0; j < MAX_LS; ++j) {
        const v2f sj = {s[j], s[j]};
        v2f t2 = acc[j];
        t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g0, a0, lo2, hi2),
                                       b0, t2);
        t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g1, a1, lo2, hi2),
                                       b1, t2);
        t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g2, a2, lo2, hi2),
                                       b2, t2);
        t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g3, a3, lo2, hi2),
                                       b3, t2);
        acc[j] = t2;
      }
    }
    float accf[MAX_LS];
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) accf[j] = acc[j].x + acc[j].y;
    wave_reduce16(accf, lane);
    if ((lane & 3) == 0) {
      const float x = accf[0];
      const float p = clamp_p(__expf(-x), min_p, max_p);
      llh_mine += (double)log1pf(-p) + (double)x;
    }
  }
  if ((lane & 3) == 0) acc_llh[wid][jmine] = llh_mine;

  kf_tail_bf16(fu, gu, sumF, s_lad, acc_llh, acc_nt, red, &s_llh_u, best, u,
               K, n_ladder, alpha, min_f, max_f);
}

// bf16 MFMA phase-B variant (see kf_mfma_t): v_mfma_f32_16x16x32_bf16,
// 8192 MACs/instruction — the A fragment is the raw bf16 fv row (16B
// uint4 per lane), the B fragment is the clamped candidate row built in
// fp32 from LDS fu/grad and packed to bf16 (RNE, same rounding as the K3
// bf16 commit).  Candidate fragments are built once per 256-element
// k-chunk per wave and reused across the accumulator tile group.
template <int NSLOT>
__global__ void __launch_bounds__(BLOCK, 3) kf_mfma_bf16_t(
    const u32* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, float* __restrict__ grad,
    double* __restrict__ llh, const float* __restrict__ ladder,
    float* __restrict__ best, int n_local, int K, int n_ladder, float alpha,
    float min_p, float max_p, float min_f, float max_f) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  __shared__ __attribute__((aligned(16))) float s_lad[MAX_LS];
  __shared__ __attribute__((aligned(16))) double acc_llh[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float acc_nt[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float red[NWAVE];
  __shared__ double s_llh_u;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* gacc = reinterpret_cast<float*>(smem);               // K floats
  u32* fu_s = reinterpret_cast<u32*>(smem + (size_t)K * 4);   // K/2 u32

  if (tid < MAX_LS) s_lad[tid] = (tid < n_ladder) ? ladder[tid] : 0.f;

  if constexpr (NSLOT == 0) {  // large-K path: fu in LDS only
    kf_phase_a_bf16_lds(F, indices, sumF, u, e0, e1, K, min_p, max_p, gacc,
                        fu_s, red, &s_llh_u, grad, llh);
  } else {
    kf_phase_a_bf16<NSLOT>(F, indices, sumF, u, e0, e1, K, min_p, max_p,
                           gacc, fu_s, red, &s_llh_u, grad, llh);
  }

  // ---------------- phase B: MFMA 16-edge x 16-candidate tiles
  // (k-quartered across waves; see kf_mfma_t)
  __shared__ __attribute__((aligned(16))) float redt[NWAVE][WAVE][4];
  const int deg = (int)(e1 - e0);
  const int ntiles = (deg + 15) >> 4;
  const int arow = lane & 15;   // A row (edge) / B col (candidate) select
  const int kgrp = lane >> 4;   // k-group 0..3 (8 contiguous bf16 each)
  const float s_j = s_lad[arow];
  const int nkstep = (K + 31) >> 5;  // 32-k MFMA steps (K % 8 == 0)
  const int sq = (nkstep + NWAVE - 1) / NWAVE;
  const int sw0 = wid * sq;
  const int sw1 = min(nkstep, sw0 + sq);
  // NSLOT=8 keeps 64 VGPRs of fu resident through phase B: halve the
  // candidate burst and accumulator group there to stay spill-free
  constexpr int KC = (NSLOT > 4) ? 4 : 8;  // candidate frags per burst
  constexpr int TG = (NSLOT > 4) ? 2 : KF_TG;  // acc tiles per group
  double llh_j = 0.0;

  for (int tbase = 0; tbase < ntiles; tbase += TG) {
    const int nt = min(TG, ntiles - tbase);
    const u32* aptr[TG];
#pragma unroll
    for (int t = 0; t < TG; ++t) {
      const long long e = e0 + (long long)(tbase + t) * 16 + arow;
      aptr[t] = (e < e1) ? F + (size_t)indices[e] * (K / 2) : nullptr;
    }
    f32x4v acc[TG];
#pragma unroll
    for (int t = 0; t < TG; ++t) acc[t] = f32x4v{0.f, 0.f, 0.f, 0.f};

    for (int sb = sw0; sb < sw1; sb += KC) {
      bf16x8 bfrag[KC];
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        const int k = (sb + c) * 32 + kgrp * 8;
        if (sb + c < sw1 && k < K) {
          const uint4 araw = *reinterpret_cast<const uint4*>(fu_s + k / 2);
          const float4 gA = ld4(gacc + k);
          const float4 gB = ld4(gacc + k + 4);
          const v2f a0 = bf2(araw.x), a1 = bf2(araw.y), a2 = bf2(araw.z),
                    a3 = bf2(araw.w);
          const v2f lo2 = {min_f, min_f}, hi2 = {max_f, max_f};
          const v2f sj = {s_j, s_j};
          const v2f c0 = pk_clamp_fma(sj, v2f{gA.x, gA.y}, a0, lo2, hi2);
          const v2f c1 = pk_clamp_fma(sj, v2f{gA.z, gA.w}, a1, lo2, hi2);
          const v2f c2 = pk_clamp_fma(sj, v2f{gB.x, gB.y}, a2, lo2, hi2);
          const v2f c3 = pk_clamp_fma(sj, v2f{gB.z, gB.w}, a3, lo2, hi2);
          bfrag[c] = as_bf16x8(
              uint4{pack_bf16_rne(c0.x, c0.y), pack_bf16_rne(c1.x, c1.y),
                    pack_bf16_rne(c2.x, c2.y), pack_bf16_rne(c3.x, c3.y)});
        } else {
          bfrag[c] = as_bf16x8(uint4{0u, 0u, 0u, 0u});
        }
      }
#pragma unroll
      for (int t = 0; t < TG; ++t) {
        if (t >= nt) break;
#pragma unroll
        for (int c = 0; c < KC; ++c) {
          const int k = (sb + c) * 32 + kgrp * 8;
          const bf16x8 afrag =
              (sb + c < sw1 && k < K && aptr[t])
                  ? as_bf16x8(*reinterpret_cast<const uint4*>(aptr[t] +
                                                              k / 2))
                  : as_bf16x8(uint4{0u, 0u, 0u, 0u});
          acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag[c],
                                                           acc[t], 0, 0, 0);
        }
      }
    }
#pragma unroll
    for (int t = 0; t < TG; ++t) {
      if (t >= nt) break;
      __syncthreads();  // redt free / previous tile consumed
      *reinterpret_cast<float4*>(&redt[wid][lane][0]) =
          float4{acc[t][0], acc[t][1], acc[t][2], acc[t][3]};
      __syncthreads();
      if (wid == ((tbase + t) & (NWAVE - 1))) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int erow = (tbase + t) * 16 + kgrp * 4 + r;
          if (erow < deg) {
            const float x = redt[0][lane][r] + redt[1][lane][r] +
                            redt[2][lane][r] + redt[3][lane][r];
            const float p = clamp_p(__expf(-x), min_p, max_p);
            llh_j += (double)log1pf(-p) + (double)x;
          }
        }
      }
    }
  }
  llh_j += __shfl_xor(llh_j, 16, WAVE);
  llh_j += __shfl_xor(llh_j, 32, WAVE);
  if (lane < MAX_LS) acc_llh[wid][lane] = llh_j;
  __syncthreads();  // all epilogue reads done before kf_tail runs

  kf_tail_bf16(fu_s, gacc, sumF, s_lad, acc_llh, acc_nt, red, &s_llh_u, best,
               u, K, n_ladder, alpha, min_f, max_f);
}

// K2 tiled variant for K too large to stage whole rows (fp32 K > ~20k,
// e.g. the com-Amazon K=25000 config): fu/grad are staged per 2048-element
// k-chunk (16 KB LDS) with the edge loop INSIDE the chunk loop; per-edge
// candidate dot partials accumulate across chunks in an LDS tile
// xpart[512 edges][16 candidates] (32 KB).  The inner body is identical to
// k2_ls_v3.  The unstaged fallback read fu/g from global per edge and
// L2-thrashed (200 KB/node working set): 247 ms/sweep at K=25000 vs the
// staged kernel's per-K-scaled ~100.

#define TK2_C 2048  // staged k-chunk elements
#define TK2_T 512   // edge-tile rows in xpart

extern "C" __global__ void __launch_bounds__(BLOCK, 3) k2_ls_tiled(
    const float* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const float* __restrict__ grad, const double* __restrict__ llh,
    const int* __restrict__ order, const float* __restrict__ ladder,
    float* __restrict__ best, int n_local, int K, int n_ladder, float alpha,
    float min_p, float max_p, float min_f, float max_f) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  __shared__ __attribute__((aligned(16))) float s_lad[MAX_LS];
  __shared__ __attribute__((aligned(16))) float acc_nt[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) double acc_llh2[MAX_LS][MAX_LS + 1];
  __shared__ __attribute__((aligned(16))) float red[NWAVE];
  __shared__ __attribute__((aligned(16))) float fu_c[TK2_C];
  __shared__ __attribute__((aligned(16))) float g_c[TK2_C];
  __shared__ __attribute__((aligned(16))) float xpart[TK2_T][MAX_LS];

  if (tid < MAX_LS) s_lad[tid] = (tid < n_ladder) ? ladder[tid] : 0.f;
  __syncthreads();
  float s[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) s[j] = s_lad[j];
  const int jmine = (((lane >> 5) & 1) << 3) | (((lane >> 4) & 1) << 2) |
                    (((lane >> 3) & 1) << 1) | ((lane >> 2) & 1);
  const v2f lo2 = {min_f, min_f}, hi2 = {max_f, max_f};
  const float* __restrict__ fu_g = F + (size_t)u * K;
  const float* __restrict__ gu_g = grad + (size_t)u * K;

  // thread's share of the edge-term LLH: candidate tid&15 over the edge
  // rows (tid>>4)-strided — folded across the 16 thread groups at the end
  double llh_t = 0.0;

  for (long long t0 = e0; t0 < e1; t0 += TK2_T) {
    const int tlen = (int)((e1 - t0) < TK2_T ? (e1 - t0) : (long long)TK2_T);
    for (int i = tid; i < tlen * MAX_LS; i += BLOCK)
      reinterpret_cast<float*>(xpart)[i] = 0.f;
    for (int c0 = 0; c0 < K; c0 += TK2_C) {
      const int clen = (K - c0) < TK2_C ? (K - c0) : TK2_C;
      __syncthreads();  // xpart zero / previous chunk fully consumed
      for (int k = tid * 4; k < clen; k += BLOCK * 4) {
        *reinterpret_cast<float4*>(fu_c + k) = ld4(fu_g + c0 + k);
        *reinterpret_cast<float4*>(g_c + k) = ld4(gu_g + c0 + k);
      }
      __syncthreads();
      for (long long e = t0 + wid; e < t0 + tlen; e += NWAVE) {
        const float* __restrict__ fv = F + (size_t)indices[e] * K + c0;
        v2f acc2[MAX_LS];
#pragma unroll
        for (int j = 0; j < MAX_LS; ++j) acc2[j] = v2f{0.f, 0.f};
        constexpr int U = 4;
        constexpr int KSTR = WAVE * 4;
        for (int k = lane * 4; k < clen; k += U * KSTR) {
          float4 b[U];
#pragma unroll
          for (int t = 0; t < U; ++t) {
            const int kk = k + t * KSTR;
            b[t] = ld4(fv + (kk < clen ? kk : (clen - 4)));
          }
#pragma unroll
          for (int t = 0; t < U; ++t) {
            if (k + t * KSTR >= clen) break;
            const float4 a4 = ld4(fu_c + k + t * KSTR);
            const float4 g4 = ld4(g_c + k + t * KSTR);
            const v2f b0 = {b[t].x, b[t].y}, b1 = {b[t].z, b[t].w};
            const v2f a0 = {a4.x, a4.y}, a1 = {a4.z, a4.w};
            const v2f g0 = {g4.x, g4.y}, g1 = {g4.z, g4.w};
#pragma unroll
            for (int j = 0; j < MAX_LS; ++j) {
              const v2f sj = {s[j], s[j]};
              v2f t2 = acc2[j];
              t2 = __builtin_elementwise_fma(
                  pk_clamp_fma(sj, g0, a0, lo2, hi2), b0, t2);
              t2 = __builtin_elementwise_fma(
                  pk_clamp_fma(sj, g1, a1, lo2, hi2), b1, t2);
              acc2[j] = t2;
            }
          }
        }
        float accf[MAX_LS];
#pragma unroll
        for (int j = 0; j < MAX_LS; ++j) accf[j] = acc2[j].x + acc2[j].y;
        wave_reduce16(accf, lane);
        if ((lane & 3) == 0) xpart[e - t0][jmine] += accf[0];
      }
    }
    __syncthreads();  // xpart complete for this tile
    {
      const int jt = tid & 15;
      for (int et = tid >> 4; et < tlen; et += MAX_LS) {
        const float x = xpart[et][jt];
        const float p = clamp_p(__expf(-x), min_p, max_p);
        llh_t += (double)log1pf(-p) + (double)x;
      }
    }
    __syncthreads();  // finalize done before the next tile re-zeroes xpart
  }

  acc_llh2[tid >> 4][tid & 15] = llh_t;

  // node terms cand_j.(Fu - sumF) + grad.grad over full K (global reads,
  // one pass per node)
  v2f accn2[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) accn2[j] = v2f{0.f, 0.f};
  float p_gg = 0.f;
  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    const float4 a = ld4(fu_g + k);
    const float4 g = ld4(gu_g + k);
    const float4 sf = ld4(sumF + k);
    const v2f a0 = {a.x, a.y}, a1 = {a.z, a.w};
    const v2f g0 = {g.x, g.y}, g1 = {g.z, g.w};
    const v2f d0 = {a.x - sf.x, a.y - sf.y};
    const v2f d1 = {a.z - sf.z, a.w - sf.w};
    p_gg = dot4(g, g, p_gg);
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) {
      const v2f sj = {s[j], s[j]};
      v2f t = accn2[j];
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g0, a0, lo2, hi2), d0,
                                    t);
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g1, a1, lo2, hi2), d1,
                                    t);
      accn2[j] = t;
    }
  }
  float accn[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) accn[j] = accn2[j].x + accn2[j].y;
  wave_reduce16(accn, lane);
  if ((lane & 3) == 0) acc_nt[wid][jmine] = accn[0];
  const float gg = block_allreduce_sum(p_gg, red);  // ends in __syncthreads

  if (wid == 0) {
    bool ok = false;
    if (lane < MAX_LS) {
      double trial = 0.0;
#pragma unroll
      for (int g16 = 0; g16 < MAX_LS; ++g16) trial += acc_llh2[g16][lane];
#pragma unroll
      for (int wv = 0; wv < NWAVE; ++wv) trial += (double)acc_nt[wv][lane];
      ok = (lane < n_ladder) &&
           (trial >= llh[u] + (double)(alpha * s_lad[lane] * gg));
    }
    const unsigned long long bal = __ballot(ok);
    if (lane == 0)
      best[u] = bal ? s_lad[__ffsll((unsigned long long)bal) - 1] : 0.f;
  }
}

// bf16 K2, same v3 structure as k2_ls_v3: fu staged raw bf16 (K*2B LDS),
// grad staged fp32 (K*4B LDS) — 6 B/element fits K<=25000 in 160 KB LDS.
template <bool STAGED>
__global__ void __launch_bounds__(BLOCK) k2_ls_v3_bf16(
    const u32* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const float* __restrict__ grad, const double* __restrict__ llh,
    const int* __restrict__ order, const float* __restrict__ ladder,
    float* __restrict__ best, int n_local, int K, int n_ladder, float alpha,
    float min_p, float max_p, float min_f, float max_f) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  __shared__ __attribute__((aligned(16))) float s_lad[MAX_LS];
  __shared__ __attribute__((aligned(16))) double acc_llh[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float acc_nt[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float red[NWAVE];
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* g_s = reinterpret_cast<float*>(smem);               // K floats
  u32* fu_s = reinterpret_cast<u32*>(smem + (size_t)K * 4);  // K/2 u32

  if (tid < MAX_LS) s_lad[tid] = (tid < n_ladder) ? ladder[tid] : 0.f;

  const u32* __restrict__ fu_g = F + (size_t)u * (K / 2);
  const float* __restrict__ gu_g = grad + (size_t)u * K;
  if (STAGED) {
    for (int k = tid * 8; k < K; k += BLOCK * 8) {
      *reinterpret_cast<uint4*>(fu_s + k / 2) =
          *reinterpret_cast<const uint4*>(fu_g + k / 2);
      *reinterpret_cast<float4*>(g_s + k) = ld4(gu_g + k);
      *reinterpret_cast<float4*>(g_s + k + 4) = ld4(gu_g + k + 4);
    }
  }
  __syncthreads();
  const u32* __restrict__ fu = STAGED ? fu_s : fu_g;
  const float* __restrict__ gu = STAGED ? g_s : gu_g;

  float s[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) s[j] = s_lad[j];

  const int jmine = (((lane >> 5) & 1) << 3) | (((lane >> 4) & 1) << 2) |
                    (((lane >> 3) & 1) << 1) | ((lane >> 2) & 1);
  double llh_mine = 0.0;
  const v2f lo2 = {min_f, min_f}, hi2 = {max_f, max_f};

  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const u32* __restrict__ fv = F + (size_t)indices[e] * (K / 2);
    v2f acc[MAX_LS];
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) acc[j] = v2f{0.f, 0.f};
    // U=1 (plain loop): measured best for bf16 — U=2 cost 15% (36.8 vs
    // 32.1 ms/sweep at K=5000), the unpack temps raise pressure enough
    // that grouping loses more to scheduling than it gains in MLP.
    constexpr int U = 1;
    constexpr int KSTR = WAVE * 8;
    for (int k = lane * 8; k < K; k += U * KSTR) {
      uint4 braw[U];  // only the global fv loads grouped (see fp32 variant)
#pragma unroll
      for (int t = 0; t < U; ++t) {
        const int kk = k + t * KSTR;
        const int ks = kk < K ? kk : (K - 8);  // clamped: load always valid
        braw[t] = *reinterpret_cast<const uint4*>(fv + ks / 2);
      }
#pragma unroll
      for (int t = 0; t < U; ++t) {
        if (k + t * KSTR >= K) break;
        const int kk = k + t * KSTR;
        const uint4 araw = *reinterpret_cast<const uint4*>(fu + kk / 2);
        const float4 gA = ld4(gu + kk);
        const float4 gB = ld4(gu + kk + 4);
        const v2f b0 = bf2(braw[t].x), b1 = bf2(braw[t].y),
                  b2 = bf2(braw[t].z), b3 = bf2(braw[t].w);
        const v2f a0 = bf2(araw.x), a1 = bf2(araw.y),
                  a2 = bf2(araw.z), a3 = bf2(araw.w);
        const v2f g0 = {gA.x, gA.y}, g1 = {gA.z, gA.w};
        const v2f g2 = {gB.x, gB.y}, g3 = {gB.z, gB.w};
#pragma unroll
        for (int j = 0; j < MAX_LS; ++j) {
          const v2f sj = {s[j], s[j]};
          v2f t2 = acc[j];
          t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g0, a0, lo2, hi2),
                                         b0, t2);
          t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g1, a1, lo2, hi2),
                                         b1, t2);
          t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g2, a2, lo2, hi2),
                                         b2, t2);
          t2 = __builtin_elementwise_fma(pk_clamp_fma(sj, g3, a3, lo2, hi2),
                                         b3, t2);
          acc[j] = t2;
        }
      }
    }
    float accf[MAX_LS];
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) accf[j] = acc[j].x + acc[j].y;
    wave_reduce16(accf, lane);
    if ((lane & 3) == 0) {
      const float x = accf[0];
      const float p = clamp_p(__expf(-x), min_p, max_p);
      llh_mine += (double)log1pf(-p) + (double)x;
    }
  }

  v2f accn[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) accn[j] = v2f{0.f, 0.f};
  float p_gg = 0.f;
  for (int k = tid * 8; k < K; k += BLOCK * 8) {
    const uint4 araw = *reinterpret_cast<const uint4*>(fu + k / 2);
    const v2f a0 = bf2(araw.x), a1 = bf2(araw.y), a2 = bf2(araw.z),
              a3 = bf2(araw.w);
    const float4 gA = ld4(gu + k);
    const float4 gB = ld4(gu + k + 4);
    const v2f g0 = {gA.x, gA.y}, g1 = {gA.z, gA.w};
    const v2f g2 = {gB.x, gB.y}, g3 = {gB.z, gB.w};
    const float4 sA = ld4(sumF + k);
    const float4 sB = ld4(sumF + k + 4);
    const v2f d0 = a0 - v2f{sA.x, sA.y}, d1 = a1 - v2f{sA.z, sA.w};
    const v2f d2 = a2 - v2f{sB.x, sB.y}, d3 = a3 - v2f{sB.z, sB.w};
    p_gg = dot4(gA, gA, dot4(gB, gB, p_gg));
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) {
      const v2f sj = {s[j], s[j]};
      v2f t = accn[j];
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g0, a0, lo2, hi2), d0, t);
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g1, a1, lo2, hi2), d1, t);
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g2, a2, lo2, hi2), d2, t);
      t = __builtin_elementwise_fma(pk_clamp_fma(sj, g3, a3, lo2, hi2), d3, t);
      accn[j] = t;
    }
  }
  float accn_f[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) accn_f[j] = accn[j].x + accn[j].y;
  wave_reduce16(accn_f, lane);
  if ((lane & 3) == 0) {
    acc_nt[wid][jmine] = accn_f[0];
    acc_llh[wid][jmine] = llh_mine;
  }
  const float gg = block_allreduce_sum(p_gg, red);  // ends in __syncthreads

  // parallel Armijo selection (see fp32 variant)
  if (wid == 0) {
    bool ok = false;
    if (lane < MAX_LS) {
      double trial = 0.0;
#pragma unroll
      for (int wv = 0; wv < NWAVE; ++wv)
        trial += acc_llh[wv][lane] + (double)acc_nt[wv][lane];
      ok = (lane < n_ladder) &&
           (trial >= llh[u] + (double)(alpha * s_lad[lane] * gg));
    }
    const unsigned long long bal = __ballot(ok);
    if (lane == 0)
      best[u] = bal ? s_lad[__ffsll((unsigned long long)bal) - 1] : 0.f;
  }
}

extern "C" __global__ void __launch_bounds__(BLOCK) k3_apply_step_bf16(
    u32* __restrict__ F, const float* __restrict__ grad,
    const float* __restrict__ steps, int n_local, int K, float min_f,
    float max_f) {
  const int u = blockIdx.x;
  const float s = steps[u];
  if (s <= 0.f) return;
  u32* __restrict__ fu = F + (size_t)u * (K / 2);
  const float* __restrict__ gu = grad + (size_t)u * K;
  for (int k = threadIdx.x * 8; k < K; k += BLOCK * 8) {
    const f32x8 a = ld8bf(fu + k / 2);
    const float4 g0 = ld4(gu + k);
    const float4 g1 = ld4(gu + k + 4);
    float v0 = fminf(fmaxf(fmaf(s, g0.x, a.a.x), min_f), max_f);
    float v1 = fminf(fmaxf(fmaf(s, g0.y, a.a.y), min_f), max_f);
    float v2 = fminf(fmaxf(fmaf(s, g0.z, a.a.z), min_f), max_f);
    float v3 = fminf(fmaxf(fmaf(s, g0.w, a.a.w), min_f), max_f);
    float v4 = fminf(fmaxf(fmaf(s, g1.x, a.b.x), min_f), max_f);
    float v5 = fminf(fmaxf(fmaf(s, g1.y, a.b.y), min_f), max_f);
    float v6 = fminf(fmaxf(fmaf(s, g1.z, a.b.z), min_f), max_f);
    float v7 = fminf(fmaxf(fmaf(s, g1.w, a.b.w), min_f), max_f);
    uint4 o;
    o.x = pack_bf16_rne(v0, v1);
    o.y = pack_bf16_rne(v2, v3);
    o.z = pack_bf16_rne(v4, v5);
    o.w = pack_bf16_rne(v6, v7);
    *reinterpret_cast<uint4*>(fu + k / 2) = o;
  }
}

// K3 + column-sum fused (bf16): projected commit AND the per-column fp32
// partial sums in ONE pass over F — the separate flow re-read F (plus a
// torch fp32 materialization) just to refresh sumF, ~3.5 ms/sweep at the
// com-Amazon K=5000 config.  Grid: (row stripes) x (2048-element
// k-chunks); each thread owns 8 columns of the chunk and accumulates the
// POST-ROUNDING bf16 values; per-stripe partials go to
// partials[stripe][K] and a deterministic stage-2 sum (torch, [S,K] ->
// [K]) produces sumF.  Rows with step 0 skip the grad read and the F
// write but still contribute to the column sums.
#define K3CS_ROWS 512   // rows per stripe
#define K3CS_KCH 2048   // k per chunk (256 threads x 8)

extern "C" __global__ void __launch_bounds__(BLOCK) k3_colsum_bf16(
    u32* __restrict__ F, const float* __restrict__ grad,
    const float* __restrict__ steps, float* __restrict__ partials,
    int n_local, int K, float min_f, float max_f) {
  const int r0 = blockIdx.x * K3CS_ROWS;
  const int r1 = min(n_local, r0 + K3CS_ROWS);
  const int k = blockIdx.y * K3CS_KCH + threadIdx.x * 8;
  if (k >= K) return;
  float c0 = 0.f, c1 = 0.f, c2 = 0.f, c3 = 0.f;
  float c4 = 0.f, c5 = 0.f, c6 = 0.f, c7 = 0.f;
  for (int u = r0; u < r1; ++u) {
    u32* __restrict__ fu = F + (size_t)u * (K / 2) + k / 2;
    const float s = steps[u];
    uint4 o = *reinterpret_cast<uint4*>(fu);
    if (s > 0.f) {
      const float4 g0 = ld4(grad + (size_t)u * K + k);
      const float4 g1 = ld4(grad + (size_t)u * K + k + 4);
      const v2f a0 = bf2(o.x), a1 = bf2(o.y), a2 = bf2(o.z), a3 = bf2(o.w);
      o.x = pack_bf16_rne(
          fminf(fmaxf(fmaf(s, g0.x, a0.x), min_f), max_f),
          fminf(fmaxf(fmaf(s, g0.y, a0.y), min_f), max_f));
      o.y = pack_bf16_rne(
          fminf(fmaxf(fmaf(s, g0.z, a1.x), min_f), max_f),
          fminf(fmaxf(fmaf(s, g0.w, a1.y), min_f), max_f));
      o.z = pack_bf16_rne(
          fminf(fmaxf(fmaf(s, g1.x, a2.x), min_f), max_f),
          fminf(fmaxf(fmaf(s, g1.y, a2.y), min_f), max_f));
      o.w = pack_bf16_rne(
          fminf(fmaxf(fmaf(s, g1.z, a3.x), min_f), max_f),
          fminf(fmaxf(fmaf(s, g1.w, a3.y), min_f), max_f));
      *reinterpret_cast<uint4*>(fu) = o;
    }
    const v2f b0 = bf2(o.x), b1 = bf2(o.y), b2 = bf2(o.z), b3 = bf2(o.w);
    c0 += b0.x; c1 += b0.y; c2 += b1.x; c3 += b1.y;
    c4 += b2.x; c5 += b2.y; c6 += b3.x; c7 += b3.y;
  }
  float* __restrict__ out = partials + (size_t)blockIdx.x * K + k;
  *reinterpret_cast<float4*>(out) = float4{c0, c1, c2, c3};
  *reinterpret_cast<float4*>(out + 4) = float4{c4, c5, c6, c7};
}

// ------------------------------------------------------------------- K5
//
// Ego-net conductance per node (replaces codes/bigclamv3-7.scala:39-54;
// formulas in SURVEY.md §2.4 / core/init.py).  Block per node u; waves
// split u's edge list; lanes split each neighbor v's adjacency.  The 2-hop
// endpoint w is "inside" the closed ego-net y = {u} ∪ N(u) iff w == u or w
// appears in u's SORTED adjacency row (binary search; canonical graphs
// store sorted rows — the row is L1-hot across the whole block).
//   cut   = #{2-hop endpoints outside y}
//   vol_S = |z| - cut,  |z| = deg(u) + Σ_{v∈N(u)} deg(v)
//   vol_T = Σdeg - vol_S - 2·cut
//   cond  = cut / max(min(vol_S, vol_T), 1)   (vol_S==0 -> 0, vol_T==0 -> 1)

extern "C" __global__ void __launch_bounds__(BLOCK) k5_conductance(
    const long long* __restrict__ indptr, const int* __restrict__ indices,
    double* __restrict__ cond, int n, double total_degree) {
  const int u = blockIdx.x;
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int deg = (int)(e1 - e0);
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  __shared__ unsigned long long s_cut, s_z;
  if (tid == 0) {
    s_cut = 0ull;
    s_z = 0ull;
  }
  __syncthreads();
  if (deg == 0) {
    if (tid == 0) cond[u] = 0.0;  // vol_S == 0 guard
    return;
  }

  const int* __restrict__ nu = indices + e0;  // sorted row of u
  unsigned long long cut_l = 0, z_l = 0;
  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const int v = indices[e];
    const long long f0 = indptr[v];
    const long long f1 = indptr[v + 1];
    if (lane == 0) z_l += (unsigned long long)(f1 - f0);
    for (long long f = f0 + lane; f < f1; f += WAVE) {
      const int w = indices[f];
      bool inside = (w == u);
      if (!inside) {  // binary search w in nu[0..deg)
        int lo = 0, hi = deg - 1;
        while (lo <= hi) {
          const int mid = (lo + hi) >> 1;
          const int x = nu[mid];
          if (x == w) {
            inside = true;
            break;
          }
          if (x < w)
            lo = mid + 1;
          else
            hi = mid - 1;
        }
      }
      if (!inside) ++cut_l;
    }
  }
  if (cut_l) atomicAdd(&s_cut, cut_l);
  if (z_l) atomicAdd(&s_z, z_l);
  __syncthreads();
  if (tid == 0) {
    const double zsz = (double)deg + (double)s_z;
    const double cut = (double)s_cut;
    const double vol_s = zsz - cut;
    const double vol_t = total_degree - vol_s - 2.0 * cut;
    double c;
    if (vol_s == 0.0)
      c = 0.0;
    else if (vol_t == 0.0)
      c = 1.0;
    else
      c = cut / fmax(fmin(vol_s, vol_t), 1.0);
    cond[u] = c;
  }
}

// ----------------------------------------------------------- chunked K1
//
// Large-K gradient path (fp32 K > 8192, bf16 K > 16384, and the only
// path above the fused kernels' LDS caps).  The one-pass K1 design needs
// gacc[K] (+ fu) resident in LDS — 100-150 KB at K=25000 caps occupancy
// at 1 block/CU and the sweep goes latency-bound (measured 235-247
// ms/sweep, profiles/r01_kernel_opt_log.md).  Chunking K restores
// occupancy but the per-edge dot x = Fu.Fv must complete before the
// weight w = 1/(1-exp(-x)) is known, so the pass splits in two around a
// tiny per-edge x buffer (nnz fp32 — 7.4 MB at the com-Amazon config):
//
//   KD kd_dot_t   per k-chunk: stage unpacked fu chunk in LDS, wave-per-
//                 edge partial dots accumulated into x[e] (lane-0 RMW,
//                 block-owned rows, no atomics).  LDS = ch*4 B.
//   KW kw_grad_t  per k-chunk: zero gacc chunk (owned-k), per 256-edge
//                 tile precompute w into LDS once, then barrier-free
//                 owned-k fma accumulate of w*fv chunks; chunk write-out
//                 folds -sumF + fu and accumulates the -Fu.sumF + Fu.Fu
//                 node terms across chunks.  llh (x-only) is summed
//                 edge-strided in the first chunk.  LDS = ch*4 + tile.
//
// Traffic = 2 edge passes over F chunks + one grad write + fu re-reads —
// ~2x the one-pass kernel's bytes but at 4-5 blocks/CU instead of 1.

#define KCHUNK 8192  // fp32 LDS chunk: 32 KB -> 4+ blocks/CU

template <bool BF16>
__global__ void __launch_bounds__(BLOCK) kd_dot_t(
    const void* __restrict__ Fvp, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const int* __restrict__ order,
    float* __restrict__ xbuf, int K, int ch) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* fus = reinterpret_cast<float*>(smem);  // ch floats, unpacked fp32
  const float* Ff = BF16 ? nullptr : reinterpret_cast<const float*>(Fvp);
  const u32* Fb = BF16 ? reinterpret_cast<const u32*>(Fvp) : nullptr;

  for (int k0 = 0; k0 < K; k0 += ch) {
    const int cw = min(ch, K - k0);
    if (BF16) {
      const u32* fu = Fb + (size_t)u * (K / 2) + k0 / 2;
      for (int k = tid * 8; k < cw; k += BLOCK * 8) {
        const f32x8 v = ld8bf(fu + k / 2);
        *reinterpret_cast<float4*>(fus + k) = v.a;
        *reinterpret_cast<float4*>(fus + k + 4) = v.b;
      }
    } else {
      const float* fu = Ff + (size_t)u * K + k0;
      for (int k = tid * 4; k < cw; k += BLOCK * 4)
        *reinterpret_cast<float4*>(fus + k) = ld4(fu + k);
    }
    __syncthreads();
    for (long long e = e0 + wid; e < e1; e += NWAVE) {
      float part = 0.f;
      if (BF16) {
        const u32* fv = Fb + (size_t)indices[e] * (K / 2) + k0 / 2;
        for (int k = lane * 8; k < cw; k += WAVE * 8) {
          const f32x8 b = ld8bf(fv + k / 2);
          part = dot4(*reinterpret_cast<const float4*>(fus + k), b.a, part);
          part =
              dot4(*reinterpret_cast<const float4*>(fus + k + 4), b.b, part);
        }
      } else {
        const float* fv = Ff + (size_t)indices[e] * K + k0;
        for (int k = lane * 4; k < cw; k += WAVE * 4)
          part = dot4(*reinterpret_cast<const float4*>(fus + k), ld4(fv + k),
                      part);
      }
      part = wave_allreduce_sum(part);
      if (lane == 0) xbuf[e] = k0 ? xbuf[e] + part : part;
    }
    __syncthreads();
  }
}

template <bool BF16>
__global__ void __launch_bounds__(BLOCK) kw_grad_t(
    const void* __restrict__ Fvp, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, const float* __restrict__ xbuf,
    float* __restrict__ grad, double* __restrict__ llh, int K, int ch,
    float min_p, float max_p) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* gacc = reinterpret_cast<float*>(smem);  // ch floats
  float* wtile = gacc + ch;                      // BLOCK floats
  const float* Ff = BF16 ? nullptr : reinterpret_cast<const float*>(Fvp);
  const u32* Fb = BF16 ? reinterpret_cast<const u32*>(Fvp) : nullptr;
  __shared__ double dred[NWAVE];

  double lacc = 0.0;  // edge llh terms, x-only; summed once (chunk 0)
  float p_fs = 0.f, p_ff = 0.f;  // node terms, accumulated across chunks

  for (int k0 = 0; k0 < K; k0 += ch) {
    const int cw = min(ch, K - k0);
    for (int k = tid * 4; k < cw; k += BLOCK * 4)
      *reinterpret_cast<float4*>(gacc + k) = float4{0.f, 0.f, 0.f, 0.f};
    // owned-k zeroing: no barrier needed before the owned-k accumulate,
    // but wtile below is cross-thread -> tiles carry their own barriers.
    for (long long et = e0; et < e1; et += BLOCK) {
      const int ne = (int)min((long long)BLOCK, e1 - et);
      __syncthreads();
      if (tid < ne) {
        const float x = xbuf[et + tid];
        const float p = clamp_p(__expf(-x), min_p, max_p);
        wtile[tid] = 1.f / (1.f - p);
        if (k0 == 0) lacc += (double)log1pf(-p) + (double)x;
      }
      __syncthreads();
      for (int i = 0; i < ne; ++i) {
        const float w = wtile[i];
        if (BF16) {
          const u32* fv = Fb + (size_t)indices[et + i] * (K / 2) + k0 / 2;
          for (int k = tid * 8; k < cw; k += BLOCK * 8) {
            const f32x8 b = ld8bf(fv + k / 2);
            float4 g = *reinterpret_cast<float4*>(gacc + k);
            g.x = fmaf(w, b.a.x, g.x);
            g.y = fmaf(w, b.a.y, g.y);
            g.z = fmaf(w, b.a.z, g.z);
            g.w = fmaf(w, b.a.w, g.w);
            *reinterpret_cast<float4*>(gacc + k) = g;
            float4 h = *reinterpret_cast<float4*>(gacc + k + 4);
            h.x = fmaf(w, b.b.x, h.x);
            h.y = fmaf(w, b.b.y, h.y);
            h.z = fmaf(w, b.b.z, h.z);
            h.w = fmaf(w, b.b.w, h.w);
            *reinterpret_cast<float4*>(gacc + k + 4) = h;
          }
        } else {
          const float* fv = Ff + (size_t)indices[et + i] * K + k0;
          for (int k = tid * 4; k < cw; k += BLOCK * 4) {
            const float4 b = ld4(fv + k);
            float4 g = *reinterpret_cast<float4*>(gacc + k);
            g.x = fmaf(w, b.x, g.x);
            g.y = fmaf(w, b.y, g.y);
            g.z = fmaf(w, b.z, g.z);
            g.w = fmaf(w, b.w, g.w);
            *reinterpret_cast<float4*>(gacc + k) = g;
          }
        }
      }
    }
    // chunk write-out (owned-k): grad = gacc - sumF + fu; node terms
    float* __restrict__ gout = grad + (size_t)u * K + k0;
    if (BF16) {
      const u32* fu = Fb + (size_t)u * (K / 2) + k0 / 2;
      for (int k = tid * 8; k < cw; k += BLOCK * 8) {
        const f32x8 a = ld8bf(fu + k / 2);
        const float4 sA = ld4(sumF + k0 + k);
        const float4 sB = ld4(sumF + k0 + k + 4);
        p_fs = dot4(a.a, sA, dot4(a.b, sB, p_fs));
        p_ff = dot4(a.a, a.a, dot4(a.b, a.b, p_ff));
        const float4 g = *reinterpret_cast<const float4*>(gacc + k);
        const float4 h = *reinterpret_cast<const float4*>(gacc + k + 4);
        *reinterpret_cast<float4*>(gout + k) =
            float4{g.x - sA.x + a.a.x, g.y - sA.y + a.a.y,
                   g.z - sA.z + a.a.z, g.w - sA.w + a.a.w};
        *reinterpret_cast<float4*>(gout + k + 4) =
            float4{h.x - sB.x + a.b.x, h.y - sB.y + a.b.y,
                   h.z - sB.z + a.b.z, h.w - sB.w + a.b.w};
      }
    } else {
      const float* fu = Ff + (size_t)u * K + k0;
      for (int k = tid * 4; k < cw; k += BLOCK * 4) {
        const float4 a = ld4(fu + k);
        const float4 s = ld4(sumF + k0 + k);
        p_fs = dot4(a, s, p_fs);
        p_ff = dot4(a, a, p_ff);
        const float4 g = *reinterpret_cast<const float4*>(gacc + k);
        *reinterpret_cast<float4*>(gout + k) =
            float4{g.x - s.x + a.x, g.y - s.y + a.y, g.z - s.z + a.z,
                   g.w - s.w + a.w};
      }
    }
    __syncthreads();  // gacc reused next chunk after all reads complete
  }

  // block-reduce the fp64 edge-llh partials and fp32 node terms
  double dv = lacc;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) dv += __shfl_xor(dv, off, WAVE);
  if (lane == 0) dred[wid] = dv;
  const float fs = block_allreduce_sum(p_fs, wtile);  // wtile as scratch
  const float ff = block_allreduce_sum(p_ff, wtile);
  __syncthreads();
  if (tid == 0) {
    double t = 0.0;
#pragma unroll
    for (int wv = 0; wv < NWAVE; ++wv) t += dred[wv];
    llh[u] = t + (double)(-fs) + (double)ff;
  }
}

// ------------------------------------------------- sparse-adaptive sweep
//
// Active-column realization of the sweep for the converged phase
// (docs/sparse_sweep_design.md; the reference's v3 sparse-F insight,
// codes/bigclamv3-7.scala:86).  All clamps produce EXACT zeros, so the
// sparsity is exact, not thresholded: measured converged density at the
// headline configs is 0.4-4% (profiles/r02_convergence.md), far below
// the 16-candidate dense streaming work.
//
//   KAF kaf_support_t  per-row support compaction of F: counts, then
//                      (idx, fp32 val) lists for rows with <= cap nnz.
//   K1S k1s_grad_t     per routed node: edge dots walk the NEIGHBOR's
//                      compact list (gathering fu elements from the
//                      L1-hot row), scatter w*val into a dense LDS
//                      gacc, then ONE dense K-scan emits the compact
//                      gradient (k, g) for the active set
//                      S_u = supp(gacc) ∪ supp(fu) plus llh_u and
//                      gg_u = Σ_S (g² − sumF²)  (full ‖g‖² = gg_u + GG,
//                      GG = Σ_K sumF², since g = −sumF off S).
//   K2S k2s_ls_t       16-candidate Armijo scoring entirely on the
//                      compact lists: candidates clamp(fu + s_j·g) are
//                      supported on S_u exactly, so trial dots walk the
//                      neighbor lists with a binary search into S_u.
//   K3S k3s_commit_t   sparse projected commit (writes confined to S_u).
//
// Routing is exact and per node: bound_u = s_u + Σ_{v∈N(u)} s_v ≥ |S_u|
// (host-side from KAF counts); nodes with bound ≤ cap take this path,
// the rest the dense kernels — the two launches cover disjoint nodes.

template <bool BF16>
__device__ __forceinline__ float f_elem(const void* __restrict__ Fp, int ldF,
                                        int row, int k) {
  if (BF16) {
    const unsigned short* p = reinterpret_cast<const unsigned short*>(Fp);
    return __uint_as_float((u32)p[(size_t)row * ldF + k] << 16);
  }
  return reinterpret_cast<const float*>(Fp)[(size_t)row * ldF + k];
}

__device__ __forceinline__ unsigned short pack1_bf16_rne(float v) {
  u32 l = __float_as_uint(v);
  l += 0x7fffu + ((l >> 16) & 1u);
  return (unsigned short)(l >> 16);
}

template <bool BF16, bool FILL>
__global__ void __launch_bounds__(BLOCK) kaf_support_t(
    const void* __restrict__ Fp, int n_rows, int K, int ldF, int cap,
    const long long* __restrict__ soffset, int* __restrict__ scount,
    int* __restrict__ sidx, float* __restrict__ sval,
    const unsigned char* __restrict__ dirty) {
  __shared__ int scan[BLOCK];
  const int r = blockIdx.x;
  if (r >= n_rows) return;
  // incremental mode: rows unchanged since the last commit keep their
  // persistent counts/lists (the commit kernels only touch accepted
  // rows; halo rows are always marked dirty at ws > 1)
  if (dirty != nullptr && !dirty[r]) return;
  // chunk rounded to 8 so the row reads vectorize (uint4 of 8 bf16 /
  // two float4); K is padded (bf16: %8, fp32: %4 -> guard the last 4)
  const int chunk = (((K + BLOCK - 1) / BLOCK) + 7) & ~7;
  const int c0 = min((int)threadIdx.x * chunk, K);
  const int c1 = min(c0 + chunk, K);
  int cnt = 0;
  if (BF16) {
    const u32* row =
        reinterpret_cast<const u32*>(Fp) + (size_t)r * (ldF / 2);
    for (int c = c0; c < c1; c += 8) {
      const f32x8 v = ld8bf(row + c / 2);
      cnt += (v.a.x != 0.f) + (v.a.y != 0.f) + (v.a.z != 0.f) +
             (v.a.w != 0.f) + (v.b.x != 0.f) + (v.b.y != 0.f) +
             (v.b.z != 0.f) + (v.b.w != 0.f);
    }
  } else {
    const float* row = reinterpret_cast<const float*>(Fp) + (size_t)r * ldF;
    for (int c = c0; c < c1; c += 4) {
      const float4 v = ld4(row + c);
      cnt += (v.x != 0.f) + (v.y != 0.f) + (v.z != 0.f) + (v.w != 0.f);
    }
  }
  scan[threadIdx.x] = cnt;
  __syncthreads();
#pragma unroll
  for (int off = 1; off < BLOCK; off <<= 1) {
    const int v = (threadIdx.x >= off) ? scan[threadIdx.x - off] : 0;
    __syncthreads();
    scan[threadIdx.x] += v;
    __syncthreads();
  }
  if (threadIdx.x == BLOCK - 1) scount[r] = scan[BLOCK - 1];
  if (!FILL || scan[BLOCK - 1] > cap) return;  // over-cap rows: count only
  long long w = soffset[r] + (scan[threadIdx.x] - cnt);
  if (BF16) {
    const u32* row =
        reinterpret_cast<const u32*>(Fp) + (size_t)r * (ldF / 2);
    for (int c = c0; c < c1; c += 8) {
      const f32x8 v = ld8bf(row + c / 2);
      const float el[8] = {v.a.x, v.a.y, v.a.z, v.a.w,
                           v.b.x, v.b.y, v.b.z, v.b.w};
#pragma unroll
      for (int t = 0; t < 8; ++t)
        if (el[t] != 0.f) {
          sidx[w] = c + t;
          sval[w] = el[t];
          ++w;
        }
    }
  } else {
    const float* row = reinterpret_cast<const float*>(Fp) + (size_t)r * ldF;
    for (int c = c0; c < c1; c += 4) {
      const float4 v = ld4(row + c);
      const float el[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
      for (int t = 0; t < 4; ++t)
        if (el[t] != 0.f) {
          sidx[w] = c + t;
          sval[w] = el[t];
          ++w;
        }
    }
  }
}

// KFS: the fused sparse sweep — ONE launch per sweep for the routed
// nodes.  Evolution is measured in profiles/r02_sparse_sweep.md; the
// current (v10) structure:
//   * ALL neighbor-list entries staged in LDS once (<= cap by routing;
//     per-edge bases come FREE from the host's epos prefix — the same
//     cumsum grad_ls_auto builds for the routing bounds);
//   * the active set S_u = supp(gacc) ∪ supp(fu) is built FIRST from an
//     LDS bitmap (phase 1 ORs bits while staging), so the accumulator
//     gacc is COMPACT (cap slots, not K): no O(K) LDS residency, no
//     O(K) zeroing, and staged entries transform to slot positions once
//     (binary search) — after that every phase runs on LDS positions
//     with zero global gathers;
//   * WAVE-per-edge dot + scatter via LDS atomicAdd — barrier-free edge
//     parallelism.  The atomics make gacc's fp32 summation order
//     run-dependent (unlike every dense kernel): the sparse path's
//     documented determinism trade (BIGCLAM_SPARSE=0 restores bitwise
//     reproducibility);
//   * the 16-candidate trial phase runs on an (edge-slot × candidate)
//     THREAD mapping — no wave-wide reductions;
//   * one packed 3-value block reduction (gg, fs, ff).
// LDS = bitmap K/32 + cap*(16|20) bytes: 21 KB at the headline K=5000
// bf16 (7 blocks/CU), ~78 KB at K=25000 with the occupancy-aware cap.

template <bool BF16>
__device__ __forceinline__ float vget(unsigned short v);
template <>
__device__ __forceinline__ float vget<true>(unsigned short v) {
  return __uint_as_float((u32)v << 16);
}

template <bool BF16>
__global__ void __launch_bounds__(BLOCK) kfs_sparse_t(
    const void* __restrict__ Fp, int K,
    const long long* __restrict__ indptr, const int* __restrict__ indices,
    const float* __restrict__ sumF, const int* __restrict__ order,
    const long long* __restrict__ soffset, const int* __restrict__ sidx,
    const float* __restrict__ sval, const int* __restrict__ scount,
    const long long* __restrict__ epos, const long long* __restrict__ goffset,
    int* __restrict__ gidx, float* __restrict__ gval,
    int* __restrict__ gcount, double* __restrict__ llh,
    const float* __restrict__ GGp, const float* __restrict__ ladder,
    float* __restrict__ best, int n_ladder, int cap, float alpha,
    float min_p, float max_p, float min_f, float max_f, int phases) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  const int nw = (K + 31) >> 5;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // layout: bmap[nw] u32 | gacc[cap] f32 | gS[cap] f32 | nidx[cap] u16 |
  //         kS[cap] u16 | nval[cap] u16|f32 | fuS[cap] u16|f32
  u32* bmap = reinterpret_cast<u32*>(smem);
  float* gacc = reinterpret_cast<float*>(bmap + nw);
  float* gS = gacc + cap;
  unsigned short* nidx = reinterpret_cast<unsigned short*>(gS + cap);
  unsigned short* kS = nidx + cap;
  char* pv = reinterpret_cast<char*>(kS + cap);
  float* nval_f = reinterpret_cast<float*>(pv);
  unsigned short* nval_h = reinterpret_cast<unsigned short*>(pv);
  pv += (size_t)cap * (BF16 ? 2 : 4);
  pv = (char*)(((size_t)pv + 3) & ~(size_t)3);
  float* fuS_f = reinterpret_cast<float*>(pv);
  unsigned short* fuS_h = reinterpret_cast<unsigned short*>(pv);
  __shared__ int scan[NWAVE];
  __shared__ float s_lad[MAX_LS];
  __shared__ double cllh[16][MAX_LS];
  __shared__ float cnt16[16][MAX_LS];
  __shared__ float red3[NWAVE][3];
  __shared__ double dred[NWAVE];
  __shared__ double sh_llh_base;
  __shared__ float sh_gg;

  if (tid < MAX_LS) s_lad[tid] = (tid < n_ladder) ? ladder[tid] : 0.f;
  for (int i = tid; i < nw; i += BLOCK) bmap[i] = 0u;
  __syncthreads();

  // phase 1: stage neighbor lists (wave-per-edge) + OR support bits
  const long long p0 = epos[e0];
  if (phases & 8)
  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const int v = indices[e];
    const long long off = soffset[v];
    const int sv = scount[v];
    const int base = (int)(epos[e] - p0);
    for (int j = lane; j < sv; j += WAVE) {
      const int k = sidx[off + j];
      nidx[base + j] = (unsigned short)k;
      if (BF16)
        nval_h[base + j] = pack1_bf16_rne(sval[off + j]);
      else
        nval_f[base + j] = sval[off + j];
      atomicOr(&bmap[k >> 5], 1u << (k & 31));
    }
  }
  {  // own support bits
    const long long offu = soffset[u];
    const int su = scount[u];
    for (int j = tid; j < su; j += BLOCK)
      atomicOr(&bmap[sidx[offu + j] >> 5], 1u << (sidx[offu + j] & 31));
  }
  __syncthreads();

  // phase 1.5: bitmap scan -> sorted compact kS + fuS gather + gacc
  // slot zeroing; then transform the staged entries to SLOT POSITIONS
  // (one binary search per entry) — no global gathers after this point
  const int wchunk = (nw + BLOCK - 1) / BLOCK;
  const int w0 = min(tid * wchunk, nw);
  const int w1 = min(w0 + wchunk, nw);
  int cnt = 0;
  if (phases & 2)
    for (int wv = w0; wv < w1; ++wv) cnt += __popc(bmap[wv]);
  int incl = cnt;
#pragma unroll
  for (int off = 1; off < WAVE; off <<= 1) {
    const int v = __shfl_up(incl, off, WAVE);
    if (lane >= off) incl += v;
  }
  if (lane == WAVE - 1) scan[wid] = incl;
  __syncthreads();
  int wbase = 0;
#pragma unroll
  for (int wv = 0; wv < NWAVE; ++wv)
    if (wv < wid) wbase += scan[wv];
  const int ns = scan[0] + scan[1] + scan[2] + scan[3];
  int pos = wbase + incl - cnt;
  for (int wv = w0; wv < w1; ++wv) {
    u32 bits = bmap[wv];
    while (bits) {
      const int b = __ffs(bits) - 1;
      bits &= bits - 1;
      const int k = (wv << 5) + b;
      kS[pos] = (unsigned short)k;
      const float f = f_elem<BF16>(Fp, K, u, k);
      if (BF16)
        fuS_h[pos] = pack1_bf16_rne(f);
      else
        fuS_f[pos] = f;
      gacc[pos] = 0.f;
      ++pos;
    }
  }
  __syncthreads();
  {  // staged column ids -> compact positions, in place
    const int total = (int)(epos[e1] - p0);
    for (int i = tid; i < total; i += BLOCK) {
      const unsigned short k = nidx[i];
      int lo = 0, hi = ns;
      while (lo < hi) {
        const int mid = (lo + hi) >> 1;
        if (kS[mid] < k)
          lo = mid + 1;
        else
          hi = mid;
      }
      nidx[i] = (unsigned short)lo;  // always found: k came from bmap
    }
  }
  __syncthreads();

  // phase 2: WAVE-per-edge dot (all operands in LDS) -> w-weighted LDS
  // atomicAdd scatter into the compact gacc; per-wave edge llh partials
  double llh_acc = 0.0;
  if (phases & 1)
  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const int base = (int)(epos[e] - p0);
    const int sv = (int)(epos[e + 1] - epos[e]);
    float part = 0.f;
    for (int j = lane; j < sv; j += WAVE) {
      const int p_ = (int)nidx[base + j];
      const float fv = BF16 ? vget<true>(nval_h[base + j]) : nval_f[base + j];
      const float fu = BF16 ? vget<true>(fuS_h[p_]) : fuS_f[p_];
      part += fu * fv;
    }
    const float x = wave_allreduce_sum(part);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    const float w = 1.f / (1.f - p);
    if (lane == 0) llh_acc += (double)log1pf(-p) + (double)x;
    for (int j = lane; j < sv; j += WAVE) {
      const float fv = BF16 ? vget<true>(nval_h[base + j]) : nval_f[base + j];
      atomicAdd(&gacc[nidx[base + j]], w * fv);
    }
  }
  __syncthreads();

  // phase 3: finalize g over the compact set + write the global pools;
  // node terms fs/ff and gg accumulate alongside
  const long long go = goffset[blockIdx.x];
  float gg_p = 0.f, fs_p = 0.f, ff_p = 0.f;
  for (int i = tid; i < ns; i += BLOCK) {
    const int k = (int)kS[i];
    const float f = BF16 ? vget<true>(fuS_h[i]) : fuS_f[i];
    const float sfk = sumF[k];
    const float g = gacc[i] - sfk + f;
    gS[i] = g;
    gidx[go + i] = k;
    gval[go + i] = g;
    gg_p += g * g - sfk * sfk;
    fs_p = fmaf(f, sfk, fs_p);
    ff_p = fmaf(f, f, ff_p);
  }
  // one packed reduction round: (gg, fs, ff) + the llh doubles
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    gg_p += __shfl_xor(gg_p, off, WAVE);
    fs_p += __shfl_xor(fs_p, off, WAVE);
    ff_p += __shfl_xor(ff_p, off, WAVE);
    llh_acc += __shfl_xor(llh_acc, off, WAVE);
  }
  if (lane == 0) {
    red3[wid][0] = gg_p;
    red3[wid][1] = fs_p;
    red3[wid][2] = ff_p;
    dred[wid] = llh_acc;
  }
  __syncthreads();
  if (tid == 0) {
    float ggt = 0.f, fst = 0.f, fft = 0.f;
    double lt = 0.0;
#pragma unroll
    for (int wv = 0; wv < NWAVE; ++wv) {
      ggt += red3[wv][0];
      fst += red3[wv][1];
      fft += red3[wv][2];
      lt += dred[wv];
    }
    gcount[blockIdx.x] = ns;
    sh_llh_base = lt + (double)(-fst) + (double)fft;
    sh_gg = ggt + GGp[0];
    llh[u] = sh_llh_base;
  }
  __syncthreads();

  // phase 4: 16-candidate trial scoring with an (edge-slot x candidate)
  // THREAD mapping: thread (e, j) serially accumulates one edge's
  // candidate-j trial dot over the staged entries
  if (phases & 4) {
    const int ej = tid >> 4;  // edge slot 0..15
    const int jc = tid & 15;  // candidate 0..15
    const float sj = s_lad[jc];
    double myllh = 0.0;
    for (long long et = e0; et < e1; et += 16) {
      const int ne = (int)min((long long)16, e1 - et);
      if (ej < ne && jc < n_ladder) {
        const long long e = et + ej;
        const int base = (int)(epos[e] - p0);
        const int sv = (int)(epos[e + 1] - epos[e]);
        float x = 0.f;
        for (int t = 0; t < sv; ++t) {
          const int p_ = (int)nidx[base + t];
          const float fv =
              BF16 ? vget<true>(nval_h[base + t]) : nval_f[base + t];
          const float fu = BF16 ? vget<true>(fuS_h[p_]) : fuS_f[p_];
          const float c = __builtin_amdgcn_fmed3f(fmaf(sj, gS[p_], fu),
                                                  min_f, max_f);
          x = fmaf(c, fv, x);
        }
        const float p = clamp_p(__expf(-x), min_p, max_p);
        myllh += (double)log1pf(-p) + (double)x;
      }
    }
    // node terms: thread (stripe=ej, j=jc) over strided elements
    float mynt = 0.f;
    if (jc < n_ladder) {
      for (int i = ej; i < ns; i += 16) {
        const float fu = BF16 ? vget<true>(fuS_h[i]) : fuS_f[i];
        const float c = __builtin_amdgcn_fmed3f(fmaf(sj, gS[i], fu), min_f,
                                                max_f);
        mynt = fmaf(c, fu - sumF[kS[i]], mynt);
      }
    }
    cllh[ej][jc] = myllh;
    cnt16[ej][jc] = mynt;
    __syncthreads();
    if (tid < MAX_LS) {
      double trial = 0.0;
      float nt = 0.f;
#pragma unroll
      for (int e = 0; e < 16; ++e) {
        trial += cllh[e][tid];
        nt += cnt16[e][tid];
      }
      const bool ok = (tid < n_ladder) &&
                      (trial + (double)nt >=
                       sh_llh_base + (double)(alpha * s_lad[tid] * sh_gg));
      const unsigned long long bal = __ballot(ok);
      if (tid == 0)
        best[u] = bal ? s_lad[__ffsll((unsigned long long)bal) - 1] : 0.f;
    }
  } else if (tid == 0) {
    best[u] = 0.f;
  }
}

// K3S also REWRITES the committed row's persistent support list
// (sidx/sval/scount) so the next sweep's KAF only rescans rows the
// DENSE path committed: the new row's support is a subset of S_u, and
// every new value is computed right here.  List values store the
// POST-ROUND value (bf16 pack->unpack), matching what KAF would read.
template <bool BF16>
__global__ void __launch_bounds__(BLOCK) k3s_commit_t(
    void* __restrict__ Fp, int ldF, const int* __restrict__ order,
    const long long* __restrict__ goffset, const int* __restrict__ gidx,
    const float* __restrict__ gval, const int* __restrict__ gcount,
    const float* __restrict__ best, const long long* __restrict__ soffset,
    int* __restrict__ sidx, float* __restrict__ sval,
    int* __restrict__ scount, int cap, float min_f, float max_f) {
  __shared__ int scan[NWAVE];
  const int u = order[blockIdx.x];
  const float s = best[u];
  if (s <= 0.f) return;
  const int ns = gcount[blockIdx.x];
  const long long go = goffset[blockIdx.x];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  // pass 1: commit + per-thread new-nnz count (strided i keeps the
  // ascending-k order within each thread's subsequence? NO — strided
  // threads interleave; use contiguous chunks so the scan emission
  // stays ascending)
  const int chunk = (ns + BLOCK - 1) / BLOCK;
  const int i0 = min(tid * chunk, ns);
  const int i1 = min(i0 + chunk, ns);
  int cnt = 0;
  for (int i = i0; i < i1; ++i) {
    const int k = gidx[go + i];
    const float f = f_elem<BF16>(Fp, ldF, u, k);
    float nf =
        __builtin_amdgcn_fmed3f(fmaf(s, gval[go + i], f), min_f, max_f);
    if (BF16) {
      const unsigned short h = pack1_bf16_rne(nf);
      reinterpret_cast<unsigned short*>(Fp)[(size_t)u * ldF + k] = h;
      nf = __uint_as_float((u32)h << 16);  // post-round value
    } else {
      reinterpret_cast<float*>(Fp)[(size_t)u * ldF + k] = nf;
    }
    cnt += nf != 0.f;
  }
  // 2-barrier block scan (wave shuffles + cross-wave offsets)
  int incl = cnt;
#pragma unroll
  for (int off = 1; off < WAVE; off <<= 1) {
    const int v = __shfl_up(incl, off, WAVE);
    if (lane >= off) incl += v;
  }
  if (lane == WAVE - 1) scan[wid] = incl;
  __syncthreads();
  int wbase = 0;
#pragma unroll
  for (int wv = 0; wv < NWAVE; ++wv)
    if (wv < wid) wbase += scan[wv];
  const int total = scan[0] + scan[1] + scan[2] + scan[3];
  if (tid == 0) scount[u] = total;
  if (total > cap) return;  // over-cap rows keep count only (KAF rule)
  long long w = soffset[u] + (wbase + incl - cnt);
  for (int i = i0; i < i1; ++i) {
    const int k = gidx[go + i];
    const float f = f_elem<BF16>(Fp, ldF, u, k);  // post-commit value
    if (f != 0.f) {
      sidx[w] = k;
      sval[w] = f;
      ++w;
    }
  }
}

// ------------------------------------------------- list-based column sum
//
// Sparse-path sumF refresh: column sums computed from the PERSISTENT
// support lists instead of a dense read of F (~60x fewer bytes at
// converged density).  Rows whose lists are not current — over-cap hubs
// and rows the DENSE path just committed (dirty) — read their F row
// directly, so the result is exact for the current F.  Per-stripe fp32
// partials land in the same [n_stripes, K] buffer as k3_colsum_bf16;
// stage 2 (partials.sum(0)) stays deterministic.  Within a block the
// list entries accumulate via LDS atomics — the same run-to-run fp-order
// trade the sparse gradient already makes (the dense path keeps the
// bitwise-deterministic k3_colsum).

#define KCS_ROWS 512
#define KCS_KCH 8192  // 32 KB LDS chunk

template <bool BF16>
__global__ void __launch_bounds__(BLOCK) kcs_lists_t(
    const void* __restrict__ Fp, int n_local, int K,
    const long long* __restrict__ soffset, const int* __restrict__ sidx,
    const float* __restrict__ sval, const int* __restrict__ scount,
    const unsigned char* __restrict__ dirty, int cap,
    float* __restrict__ partials) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* acc = reinterpret_cast<float*>(smem);  // KCS_KCH floats
  const int r0 = blockIdx.x * KCS_ROWS;
  const int r1 = min(n_local, r0 + KCS_ROWS);
  const int kc0 = blockIdx.y * KCS_KCH;
  const int kc1 = min(K, kc0 + KCS_KCH);
  const int cw = kc1 - kc0;
  const int tid = threadIdx.x;
  for (int k = tid * 4; k < cw; k += BLOCK * 4)
    *reinterpret_cast<float4*>(acc + k) = float4{0.f, 0.f, 0.f, 0.f};
  __syncthreads();
  for (int r = r0 + (tid >> 6); r < r1; r += NWAVE) {  // wave-per-row
    const int lane = tid & (WAVE - 1);
    const int c = scount[r];
    if (c <= cap && !dirty[r]) {
      // current list: add its entries that fall in this k-chunk
      const long long off = soffset[r];
      for (int j = lane; j < c; j += WAVE) {
        const int k = sidx[off + j];
        if (k >= kc0 && k < kc1) atomicAdd(&acc[k - kc0], sval[off + j]);
      }
    } else {
      // stale/over-cap row: dense read of the F row chunk
      for (int k = lane; k < cw; k += WAVE) {
        const float f = f_elem<BF16>(Fp, K, r, kc0 + k);
        if (f != 0.f) atomicAdd(&acc[k], f);
      }
    }
  }
  __syncthreads();
  float* __restrict__ out = partials + (size_t)blockIdx.x * K + kc0;
  for (int k = tid * 4; k < cw; k += BLOCK * 4)
    *reinterpret_cast<float4*>(out + k) =
        *reinterpret_cast<const float4*>(acc + k);
}

// ------------------------------------------------------------------- K6
//
// Seed-init F scatter (codes/bigclamv3-7.scala:60-87): community c's
// initial members are the neighbors of the c-th ranked seed —
// F[v, c] = 1 for v in N(seed_c) (plus the seed itself in the v2
// variant).  The caller uploads the seeds' COMPACT adjacency (k rows,
// global ids) — a few hundred KB — and the kernel scatters into the
// rank's F slice; rows outside [start, stop) are skipped, so the same
// launch works at any world size.  One block per seed/community.

template <bool BF16>
__global__ void __launch_bounds__(BLOCK) k6_seed_init_t(
    void* __restrict__ Fp, int K, const long long* __restrict__ sindptr,
    const long long* __restrict__ snbrs, const long long* __restrict__ seeds,
    int n_seeds, long long start, long long stop, int include_seed) {
  const int c = blockIdx.x;
  if (c >= n_seeds) return;
  const long long b0 = sindptr[c];
  const long long b1 = sindptr[c + 1];
  for (long long j = b0 + threadIdx.x; j < b1; j += BLOCK) {
    const long long v = snbrs[j];
    if (v < start || v >= stop) continue;
    const size_t at = (size_t)(v - start) * K + c;
    if (BF16)
      reinterpret_cast<unsigned short*>(Fp)[at] = pack1_bf16_rne(1.0f);
    else
      reinterpret_cast<float*>(Fp)[at] = 1.0f;
  }
  if (include_seed && threadIdx.x == 0) {
    const long long s = seeds[c];
    if (s >= start && s < stop) {
      const size_t at = (size_t)(s - start) * K + c;
      if (BF16)
        reinterpret_cast<unsigned short*>(Fp)[at] = pack1_bf16_rne(1.0f);
      else
        reinterpret_cast<float*>(Fp)[at] = 1.0f;
    }
  }
}

// ------------------------------------------------------------------- K7
//
// Community extraction (codes/Bigclamv2.scala:223-230): node u belongs to
// community c iff F[u,c] >= delta; a row whose max is below delta (but
// nonzero) falls back to its argmax columns (ties included); all-zero rows
// get no membership (documented deviation, engine/extract.py).  Runs as
// two deterministic passes with NO atomics: the COUNT pass writes per-row
// membership counts (host prefix-sums them into offsets), the FILL pass
// writes row u's community ids ascending at offsets[u].  Each thread owns
// a CONTIGUOUS column chunk, so the block-exclusive-scan write offsets
// preserve ascending-c order.  One block per row; HBM-bound (2 row reads).

template <bool BF16, bool FILL>
__global__ void __launch_bounds__(BLOCK) k7_membership(
    const void* __restrict__ Fv, int n, int K, int ldF, float delta,
    const long long* __restrict__ offsets, int* __restrict__ counts,
    int* __restrict__ comms) {
  __shared__ float redf[NWAVE];
  __shared__ int scan[BLOCK];
  const int u = blockIdx.x;
  if (u >= n) return;
  const float* Ff =
      BF16 ? nullptr : reinterpret_cast<const float*>(Fv) + (size_t)u * ldF;
  const u32* Fb = BF16
                      ? reinterpret_cast<const u32*>(Fv) + (size_t)u * (ldF / 2)
                      : nullptr;
  const int chunk = (K + BLOCK - 1) / BLOCK;
  const int c0 = min((int)threadIdx.x * chunk, K);
  const int c1 = min(c0 + chunk, K);
  float mymax = 0.f;
  float myabove = 0.f;
  for (int c = c0; c < c1; ++c) {
    const float f = BF16 ? ((c & 1) ? bf_hi(Fb[c >> 1]) : bf_lo(Fb[c >> 1]))
                         : Ff[c];
    mymax = fmaxf(mymax, f);
    myabove += (f >= delta) ? 1.f : 0.f;
  }
  // block max
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  float m = mymax;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    m = fmaxf(m, __shfl_xor(m, off, WAVE));
  if (lane == 0) redf[wid] = m;
  __syncthreads();
  float fmax = redf[0];
#pragma unroll
  for (int w = 1; w < NWAVE; ++w) fmax = fmaxf(fmax, redf[w]);
  __syncthreads();
  const float total_above = block_allreduce_sum(myabove, redf);
  // per-thread membership count under the row's mode
  int cnt = 0;
  if (total_above > 0.f) {
    cnt = (int)myabove;
  } else if (fmax > 0.f) {
    for (int c = c0; c < c1; ++c) {
      const float f = BF16 ? ((c & 1) ? bf_hi(Fb[c >> 1]) : bf_lo(Fb[c >> 1]))
                           : Ff[c];
      cnt += (f == fmax);
    }
  }
  // block inclusive scan of the 256 per-thread counts (Hillis-Steele)
  scan[threadIdx.x] = cnt;
  __syncthreads();
#pragma unroll
  for (int off = 1; off < BLOCK; off <<= 1) {
    const int v = (threadIdx.x >= off) ? scan[threadIdx.x - off] : 0;
    __syncthreads();
    scan[threadIdx.x] += v;
    __syncthreads();
  }
  if (!FILL) {
    if (threadIdx.x == BLOCK - 1) counts[u] = scan[BLOCK - 1];
    return;
  }
  long long w = offsets[u] + (scan[threadIdx.x] - cnt);  // exclusive offset
  if (total_above > 0.f) {
    for (int c = c0; c < c1; ++c) {
      const float f = BF16 ? ((c & 1) ? bf_hi(Fb[c >> 1]) : bf_lo(Fb[c >> 1]))
                           : Ff[c];
      if (f >= delta) comms[w++] = c;
    }
  } else if (fmax > 0.f) {
    for (int c = c0; c < c1; ++c) {
      const float f = BF16 ? ((c & 1) ? bf_hi(Fb[c >> 1]) : bf_lo(Fb[c >> 1]))
                           : Ff[c];
      if (f == fmax) comms[w++] = c;
    }
  }
}

// ----------------------------------------------------------- host launchers

#include <cstdlib>
#include <stdexcept>
#include <string>

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      throw std::runtime_error(std::string("HIP error: ") +                \
                               hipGetErrorString(_e) + " at " __FILE__);   \
    }                                                                      \
  } while (0)

static void allow_large_lds(const void* func, size_t bytes) {
  if (bytes > 65536) {
    HIP_CHECK(hipFuncSetAttribute(
        func, hipFuncAttributeMaxDynamicSharedMemorySize, (int)bytes));
  }
}

extern "C" void launch_k1_bf16(const void* F, const long long* indptr,
                               const int* indices, const float* sumF,
                               const int* order, float* grad, double* llh,
                               int n_local, int K, float min_p, float max_p,
                               hipStream_t stream) {
  if (n_local == 0) return;
  const size_t lds = (size_t)K * 4 + 4 * sizeof(float);
  const u32* Fb = reinterpret_cast<const u32*>(F);
#define K1B_CASE(NS)                                                        \
  do {                                                                      \
    allow_large_lds((const void*)&k1_grad_llh_bf16_t<NS>, lds);             \
    hipLaunchKernelGGL((k1_grad_llh_bf16_t<NS>), dim3(n_local), dim3(256),  \
                       lds, stream, Fb, indptr, indices, sumF, order, grad, \
                       llh, n_local, K, min_p, max_p);                      \
  } while (0)
  if (K <= 2048) {
    K1B_CASE(1);
  } else if (K <= 4096) {
    K1B_CASE(2);
  } else if (K <= 8192) {
    K1B_CASE(4);
  } else if (K <= 16384) {
    K1B_CASE(8);
  } else {
    throw std::runtime_error("bf16 K1: K > 16384 unsupported");
  }
#undef K1B_CASE
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k4_bf16(const void* F, const long long* indptr,
                               const int* indices, const float* sumF,
                               const int* order, double* llh, int n_local,
                               int K, float min_p, float max_p,
                               hipStream_t stream) {
  if (n_local == 0) return;
  hipLaunchKernelGGL(k4_llh_only_bf16, dim3(n_local), dim3(256), 0, stream,
                     reinterpret_cast<const u32*>(F), indptr, indices, sumF,
                     order, llh, n_local, K, min_p, max_p);
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k2_bf16(const void* F, const long long* indptr,
                               const int* indices, const float* sumF,
                               const float* grad, const double* llh,
                               const int* order, const float* ladder,
                               float* best, int n_local, int K, int n_ladder,
                               float alpha, float min_p, float max_p,
                               float min_f, float max_f, hipStream_t stream) {
  if (n_local == 0) return;
  if (n_ladder > 16) throw std::runtime_error("ladder length > 16 unsupported");
  const u32* Fb = reinterpret_cast<const u32*>(F);
  const bool nostage = getenv("BIGCLAM_K2_NOSTAGE") != nullptr;
  const size_t lds = (size_t)K * 6;  // g fp32 + fu raw bf16
  // staged only while >= 2 blocks/CU fit: at K=25000 the 150 KB staged
  // variant (1 block/CU) measured 205 ms/sweep vs 145 unstaged
  // (profiles/r02_largek_dispatch.md)
  if (!nostage && lds + 2048 <= 80 * 1024) {
    allow_large_lds((const void*)&k2_ls_v3_bf16<true>, lds);
    hipLaunchKernelGGL((k2_ls_v3_bf16<true>), dim3(n_local), dim3(256), lds,
                       stream, Fb, indptr, indices, sumF, grad, llh, order,
                       ladder, best, n_local, K, n_ladder, alpha, min_p,
                       max_p, min_f, max_f);
  } else {
    hipLaunchKernelGGL((k2_ls_v3_bf16<false>), dim3(n_local), dim3(256), 0,
                       stream, Fb, indptr, indices, sumF, grad, llh, order,
                       ladder, best, n_local, K, n_ladder, alpha, min_p,
                       max_p, min_f, max_f);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k3_bf16(void* F, const float* grad, const float* steps,
                               int n_local, int K, float min_f, float max_f,
                               hipStream_t stream) {
  if (n_local == 0) return;
  hipLaunchKernelGGL(k3_apply_step_bf16, dim3(n_local), dim3(256), 0, stream,
                     reinterpret_cast<u32*>(F), grad, steps, n_local, K,
                     min_f, max_f);
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k1(const float* F, const long long* indptr,
                          const int* indices, const float* sumF,
                          const int* order, float* grad, double* llh,
                          int n_local, int K, float min_p, float max_p,
                          hipStream_t stream) {
  if (n_local == 0) return;
  const size_t lds = (size_t)K * 4 + 4 * sizeof(float);
#define K1_CASE(NS)                                                         \
  do {                                                                      \
    allow_large_lds((const void*)&k1_grad_llh_t<NS>, lds);                  \
    hipLaunchKernelGGL((k1_grad_llh_t<NS>), dim3(n_local), dim3(256), lds,  \
                       stream, F, indptr, indices, sumF, order, grad, llh,  \
                       n_local, K, min_p, max_p);                           \
  } while (0)
  if (K <= 1024) {
    K1_CASE(1);
  } else if (K <= 2048) {
    K1_CASE(2);
  } else if (K <= 4096) {
    K1_CASE(4);
  } else if (K <= 8192) {
    K1_CASE(8);
  } else {
    allow_large_lds((const void*)k1_grad_llh, lds);
    hipLaunchKernelGGL(k1_grad_llh, dim3(n_local), dim3(256), lds, stream, F,
                       indptr, indices, sumF, order, grad, llh, n_local, K,
                       min_p, max_p);
  }
#undef K1_CASE
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k4(const float* F, const long long* indptr,
                          const int* indices, const float* sumF,
                          const int* order, double* llh, int n_local, int K,
                          float min_p, float max_p, hipStream_t stream) {
  if (n_local == 0) return;
  hipLaunchKernelGGL(k4_llh_only, dim3(n_local), dim3(256), 0, stream, F,
                     indptr, indices, sumF, order, llh, n_local, K, min_p,
                     max_p);
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k2(const float* F, const long long* indptr,
                          const int* indices, const float* sumF,
                          const float* grad, const double* llh,
                          const int* order, const float* ladder, float* best,
                          int n_local, int K, int n_ladder, float alpha,
                          float min_p, float max_p, float min_f, float max_f,
                          hipStream_t stream) {
  if (n_local == 0) return;
  if (n_ladder > 16) throw std::runtime_error("ladder length > 16 unsupported");
  // per-call getenv (cheap next to a multi-ms kernel) so tests can toggle
  const bool nostage = getenv("BIGCLAM_K2_NOSTAGE") != nullptr;
  const bool tiled = getenv("BIGCLAM_K2_TILED") != nullptr;
  const size_t lds = (size_t)K * 8;  // fu + g staged fp32
  if (tiled) {
    // chunk-staged tiled kernel: wins on high-mean-degree graphs where the
    // per-(node,chunk) staging barriers amortize; at mean degree ~6 the
    // unstaged kernel measured 14% faster (247 vs 286 ms, K=25000), so
    // tiled is opt-in pending a degree-based heuristic.
    hipLaunchKernelGGL(k2_ls_tiled, dim3(n_local), dim3(256), 0, stream, F,
                       indptr, indices, sumF, grad, llh, order, ladder, best,
                       n_local, K, n_ladder, alpha, min_p, max_p, min_f,
                       max_f);
  } else if (!nostage && lds + 2048 <= 80 * 1024) {
    // same 2-blocks/CU staging rule as the bf16 launcher (measured:
    // 1-block/CU staged loses to unstaged at K=25000)
    allow_large_lds((const void*)&k2_ls_v3<true>, lds);
    hipLaunchKernelGGL((k2_ls_v3<true>), dim3(n_local), dim3(256), lds,
                       stream, F, indptr, indices, sumF, grad, llh, order,
                       ladder, best, n_local, K, n_ladder, alpha, min_p,
                       max_p, min_f, max_f);
  } else {
    hipLaunchKernelGGL((k2_ls_v3<false>), dim3(n_local), dim3(256), 0, stream,
                       F, indptr, indices, sumF, grad, llh, order, ladder,
                       best, n_local, K, n_ladder, alpha, min_p, max_p, min_f,
                       max_f);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k3(float* F, const float* grad, const float* steps,
                          int n_local, int K, float min_f, float max_f,
                          hipStream_t stream) {
  if (n_local == 0) return;
  hipLaunchKernelGGL(k3_apply_step, dim3(n_local), dim3(256), 0, stream, F,
                     grad, steps, n_local, K, min_f, max_f);
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k5(const long long* indptr, const int* indices,
                          double* cond, int n, double total_degree,
                          hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL(k5_conductance, dim3(n), dim3(256), 0, stream, indptr,
                     indices, cond, n, total_degree);
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_kf(const float* F, const long long* indptr,
                          const int* indices, const float* sumF,
                          const int* order, float* grad, double* llh,
                          const float* ladder, float* best, int n_local,
                          int K, int n_ladder, float alpha, float min_p,
                          float max_p, float min_f, float max_f,
                          hipStream_t stream) {
  if (n_local == 0) return;
  if (n_ladder > 16) throw std::runtime_error("ladder length > 16 unsupported");
  if (K > 8192) throw std::runtime_error("fused kernel requires K <= 8192");
  const size_t lds = (size_t)K * 8;  // gacc + fu_s
#define KF_CASE(NS)                                                          \
  do {                                                                       \
    allow_large_lds((const void*)&kf_fused_t<NS>, lds);                      \
    hipLaunchKernelGGL((kf_fused_t<NS>), dim3(n_local), dim3(256), lds,      \
                       stream, F, indptr, indices, sumF, order, grad, llh,   \
                       ladder, best, n_local, K, n_ladder, alpha, min_p,     \
                       max_p, min_f, max_f);                                 \
  } while (0)
  if (K <= 1024) {
    KF_CASE(1);
  } else if (K <= 2048) {
    KF_CASE(2);
  } else if (K <= 4096) {
    KF_CASE(4);
  } else {
    KF_CASE(8);
  }
#undef KF_CASE
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_kf_bf16(const void* F, const long long* indptr,
                               const int* indices, const float* sumF,
                               const int* order, float* grad, double* llh,
                               const float* ladder, float* best, int n_local,
                               int K, int n_ladder, float alpha, float min_p,
                               float max_p, float min_f, float max_f,
                               hipStream_t stream) {
  if (n_local == 0) return;
  if (n_ladder > 16) throw std::runtime_error("ladder length > 16 unsupported");
  if (K > 16384) throw std::runtime_error("bf16 fused kernel requires K <= 16384");
  const size_t lds = (size_t)K * 6;  // gacc fp32 + fu raw bf16
  const u32* Fb = reinterpret_cast<const u32*>(F);
#define KFB_CASE(NS)                                                          \
  do {                                                                        \
    allow_large_lds((const void*)&kf_fused_bf16_t<NS>, lds);                  \
    hipLaunchKernelGGL((kf_fused_bf16_t<NS>), dim3(n_local), dim3(256), lds,  \
                       stream, Fb, indptr, indices, sumF, order, grad, llh,   \
                       ladder, best, n_local, K, n_ladder, alpha, min_p,      \
                       max_p, min_f, max_f);                                  \
  } while (0)
  if (K <= 2048) {
    KFB_CASE(1);
  } else if (K <= 4096) {
    KFB_CASE(2);
  } else if (K <= 8192) {
    KFB_CASE(4);
  } else {
    KFB_CASE(8);
  }
#undef KFB_CASE
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k3_colsum_bf16(void* F, const float* grad,
                                      const float* steps, float* partials,
                                      int n_local, int K, float min_f,
                                      float max_f, hipStream_t stream) {
  if (n_local == 0) return;
  const int ns = (n_local + K3CS_ROWS - 1) / K3CS_ROWS;
  const int nk = (K + K3CS_KCH - 1) / K3CS_KCH;
  hipLaunchKernelGGL(k3_colsum_bf16, dim3(ns, nk), dim3(256), 0, stream,
                     reinterpret_cast<u32*>(F), grad, steps, partials,
                     n_local, K, min_f, max_f);
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_kf_mfma(const float* F, const long long* indptr,
                               const int* indices, const float* sumF,
                               const int* order, float* grad, double* llh,
                               const float* ladder, float* best, int n_local,
                               int K, int n_ladder, float alpha, float min_p,
                               float max_p, float min_f, float max_f,
                               hipStream_t stream) {
  if (n_local == 0) return;
  if (n_ladder > 16) throw std::runtime_error("ladder length > 16 unsupported");
  if (K > 8192) throw std::runtime_error("fused kernel requires K <= 8192");
  const size_t lds = (size_t)K * 8;  // gacc + fu_s
#define KFM_CASE(NS)                                                         \
  do {                                                                       \
    allow_large_lds((const void*)&kf_mfma_t<NS>, lds);                       \
    hipLaunchKernelGGL((kf_mfma_t<NS>), dim3(n_local), dim3(256), lds,       \
                       stream, F, indptr, indices, sumF, order, grad, llh,   \
                       ladder, best, n_local, K, n_ladder, alpha, min_p,     \
                       max_p, min_f, max_f);                                 \
  } while (0)
  if (K <= 1024) {
    KFM_CASE(1);
  } else if (K <= 2048) {
    KFM_CASE(2);
  } else if (K <= 4096) {
    KFM_CASE(4);
  } else {
    KFM_CASE(8);
  }
#undef KFM_CASE
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_kf_mfma_bf16(const void* F, const long long* indptr,
                                    const int* indices, const float* sumF,
                                    const int* order, float* grad,
                                    double* llh, const float* ladder,
                                    float* best, int n_local, int K,
                                    int n_ladder, float alpha, float min_p,
                                    float max_p, float min_f, float max_f,
                                    hipStream_t stream) {
  if (n_local == 0) return;
  if (n_ladder > 16) throw std::runtime_error("ladder length > 16 unsupported");
  if (K > 26000)
    throw std::runtime_error("bf16 MFMA fused kernel requires K <= 26000");
  const size_t lds = (size_t)K * 6;  // gacc fp32 + fu raw bf16
  const u32* Fb = reinterpret_cast<const u32*>(F);
#define KFMB_CASE(NS)                                                        \
  do {                                                                       \
    allow_large_lds((const void*)&kf_mfma_bf16_t<NS>, lds);                  \
    hipLaunchKernelGGL((kf_mfma_bf16_t<NS>), dim3(n_local), dim3(256), lds,  \
                       stream, Fb, indptr, indices, sumF, order, grad, llh,  \
                       ladder, best, n_local, K, n_ladder, alpha, min_p,     \
                       max_p, min_f, max_f);                                 \
  } while (0)
  if (K <= 2048) {
    KFMB_CASE(1);
  } else if (K <= 4096) {
    KFMB_CASE(2);
  } else if (K <= 8192) {
    KFMB_CASE(4);
  } else if (K <= 16384) {
    KFMB_CASE(8);
  } else {
    KFMB_CASE(0);  // fu via LDS only (see kf_phase_a_bf16_lds)
  }
#undef KFMB_CASE
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k1_chunked(const void* F, int bf16,
                                  const long long* indptr, const int* indices,
                                  const float* sumF, const int* order,
                                  float* xbuf, float* grad, double* llh,
                                  int n_blocks, int K, float min_p,
                                  float max_p, hipStream_t stream) {
  if (n_blocks == 0) return;
  const int ch = (K < KCHUNK) ? K : KCHUNK;
  const size_t lds_d = (size_t)ch * 4;
  const size_t lds_w = (size_t)ch * 4 + BLOCK * 4;
  if (bf16) {
    hipLaunchKernelGGL((kd_dot_t<true>), dim3(n_blocks), dim3(BLOCK), lds_d,
                       stream, F, indptr, indices, order, xbuf, K, ch);
    hipLaunchKernelGGL((kw_grad_t<true>), dim3(n_blocks), dim3(BLOCK), lds_w,
                       stream, F, indptr, indices, sumF, order, xbuf, grad,
                       llh, K, ch, min_p, max_p);
  } else {
    hipLaunchKernelGGL((kd_dot_t<false>), dim3(n_blocks), dim3(BLOCK), lds_d,
                       stream, F, indptr, indices, order, xbuf, K, ch);
    hipLaunchKernelGGL((kw_grad_t<false>), dim3(n_blocks), dim3(BLOCK),
                       lds_w, stream, F, indptr, indices, sumF, order, xbuf,
                       grad, llh, K, ch, min_p, max_p);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_kaf(const void* F, int bf16, int n_rows, int K,
                           int cap, const long long* soffset, int* scount,
                           int* sidx, float* sval, int fill,
                           const unsigned char* dirty, hipStream_t stream) {
  if (n_rows == 0) return;
#define KAF_CASE(B, FI)                                                     \
  hipLaunchKernelGGL((kaf_support_t<B, FI>), dim3(n_rows), dim3(BLOCK), 0,  \
                     stream, F, n_rows, K, K, cap, soffset, scount, sidx,   \
                     sval, dirty)
  if (bf16) {
    if (fill)
      KAF_CASE(true, true);
    else
      KAF_CASE(true, false);
  } else {
    if (fill)
      KAF_CASE(false, true);
    else
      KAF_CASE(false, false);
  }
#undef KAF_CASE
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_kfs(const void* F, int bf16,
                           const long long* indptr, const int* indices,
                           const float* sumF, const int* order, int n_blocks,
                           const long long* soffset, const int* sidx,
                           const float* sval, const int* scount,
                           const long long* epos, const long long* goffset,
                           int* gidx, float* gval, int* gcount, double* llh,
                           const float* GGp, const float* ladder,
                           float* best, int n_ladder, int cap, int K,
                           float alpha, float min_p, float max_p,
                           float min_f, float max_f, hipStream_t stream) {
  if (n_blocks == 0) return;
  if (n_ladder > 16) throw std::runtime_error("ladder length > 16 unsupported");
  const size_t lds = ((K + 31) / 32) * 4 + (size_t)cap * (bf16 ? 16 : 20)
                     + 16;
  const char* ph = getenv("BIGCLAM_KFS_PHASES");  // measurement bisect only
  const int phases = ph ? atoi(ph) : 0xF;
  if (bf16) {
    allow_large_lds((const void*)&kfs_sparse_t<true>, lds);
    hipLaunchKernelGGL((kfs_sparse_t<true>), dim3(n_blocks), dim3(BLOCK),
                       lds, stream, F, K, indptr, indices, sumF, order,
                       soffset, sidx, sval, scount, epos, goffset, gidx,
                       gval, gcount, llh, GGp, ladder, best, n_ladder, cap,
                       alpha, min_p, max_p, min_f, max_f, phases);
  } else {
    allow_large_lds((const void*)&kfs_sparse_t<false>, lds);
    hipLaunchKernelGGL((kfs_sparse_t<false>), dim3(n_blocks), dim3(BLOCK),
                       lds, stream, F, K, indptr, indices, sumF, order,
                       soffset, sidx, sval, scount, epos, goffset, gidx,
                       gval, gcount, llh, GGp, ladder, best, n_ladder, cap,
                       alpha, min_p, max_p, min_f, max_f, phases);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k3s(void* F, int bf16, const int* order, int n_blocks,
                           const long long* goffset, const int* gidx,
                           const float* gval, const int* gcount,
                           const float* best, const long long* soffset,
                           int* sidx, float* sval, int* scount, int cap,
                           int K, float min_f, float max_f,
                           hipStream_t stream) {
  if (n_blocks == 0) return;
  if (bf16) {
    hipLaunchKernelGGL((k3s_commit_t<true>), dim3(n_blocks), dim3(BLOCK), 0,
                       stream, F, K, order, goffset, gidx, gval, gcount,
                       best, soffset, sidx, sval, scount, cap, min_f, max_f);
  } else {
    hipLaunchKernelGGL((k3s_commit_t<false>), dim3(n_blocks), dim3(BLOCK), 0,
                       stream, F, K, order, goffset, gidx, gval, gcount,
                       best, soffset, sidx, sval, scount, cap, min_f, max_f);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k6(void* F, int bf16, int K, const long long* sindptr,
                          const long long* snbrs, const long long* seeds,
                          int n_seeds, long long start, long long stop,
                          int include_seed, hipStream_t stream) {
  if (n_seeds == 0) return;
  if (bf16) {
    hipLaunchKernelGGL((k6_seed_init_t<true>), dim3(n_seeds), dim3(BLOCK), 0,
                       stream, F, K, sindptr, snbrs, seeds, n_seeds, start,
                       stop, include_seed);
  } else {
    hipLaunchKernelGGL((k6_seed_init_t<false>), dim3(n_seeds), dim3(BLOCK),
                       0, stream, F, K, sindptr, snbrs, seeds, n_seeds,
                       start, stop, include_seed);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_kcs_lists(const void* F, int bf16, int n_local,
                                 int K, const long long* soffset,
                                 const int* sidx, const float* sval,
                                 const int* scount,
                                 const unsigned char* dirty, int cap,
                                 float* partials, hipStream_t stream) {
  if (n_local == 0) return;
  const int ns = (n_local + KCS_ROWS - 1) / KCS_ROWS;
  const int nk = (K + KCS_KCH - 1) / KCS_KCH;
  const size_t lds = (size_t)(K < KCS_KCH ? ((K + 3) & ~3) : KCS_KCH) * 4;
  if (bf16) {
    hipLaunchKernelGGL((kcs_lists_t<true>), dim3(ns, nk), dim3(BLOCK), lds,
                       stream, F, n_local, K, soffset, sidx, sval, scount,
                       dirty, cap, partials);
  } else {
    hipLaunchKernelGGL((kcs_lists_t<false>), dim3(ns, nk), dim3(BLOCK), lds,
                       stream, F, n_local, K, soffset, sidx, sval, scount,
                       dirty, cap, partials);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k7_count(const void* F, int bf16, int n, int K,
                                int ldF, float delta, int* counts,
                                hipStream_t stream) {
  if (n == 0) return;
  if (bf16) {
    hipLaunchKernelGGL((k7_membership<true, false>), dim3(n), dim3(BLOCK), 0,
                       stream, F, n, K, ldF, delta, nullptr, counts, nullptr);
  } else {
    hipLaunchKernelGGL((k7_membership<false, false>), dim3(n), dim3(BLOCK), 0,
                       stream, F, n, K, ldF, delta, nullptr, counts, nullptr);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k7_fill(const void* F, int bf16, int n, int K, int ldF,
                               float delta, const long long* offsets,
                               int* comms, hipStream_t stream) {
  if (n == 0) return;
  if (bf16) {
    hipLaunchKernelGGL((k7_membership<true, true>), dim3(n), dim3(BLOCK), 0,
                       stream, F, n, K, ldF, delta, offsets, nullptr, comms);
  } else {
    hipLaunchKernelGGL((k7_membership<false, true>), dim3(n), dim3(BLOCK), 0,
                       stream, F, n, K, ldF, delta, offsets, nullptr, comms);
  }
  HIP_CHECK(hipGetLastError());
}

// ------------------------------------------------- MFMA layout probes
//
// On-device verification of the assumed C/D register mapping
// (col = lane&15, row = (lane>>4)*4 + reg): computes ONE 16x16 tile
// D = A @ B with B supplied column-major, written out per the assumed
// mapping; tests/test_gpu.py compares against a torch matmul with
// ASYMMETRIC inputs (a swapped mapping produces D^T and fails loudly).

extern "C" __global__ void mfma_probe_bf16_k(const u32* __restrict__ A,
                                             const u32* __restrict__ Bc,
                                             float* __restrict__ D) {
  // A: 16 rows x 32 k bf16 row-major (16 u32/row); Bc[c][k] = B[k][c].
  const int lane = threadIdx.x;
  const int r = lane & 15, kg = lane >> 4;
  const bf16x8 a =
      as_bf16x8(*reinterpret_cast<const uint4*>(A + r * 16 + kg * 4));
  const bf16x8 b =
      as_bf16x8(*reinterpret_cast<const uint4*>(Bc + r * 16 + kg * 4));
  f32x4v c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int i = 0; i < 4; ++i) D[(kg * 4 + i) * 16 + r] = c[i];
}

extern "C" __global__ void mfma_probe_f32_k(const float* __restrict__ A,
                                            const float* __restrict__ Bc,
                                            float* __restrict__ D) {
  // A: 16 rows x 4 k fp32 row-major; Bc[c][k] = B[k][c].
  const int lane = threadIdx.x;
  const int r = lane & 15, kg = lane >> 4;
  f32x4v c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x4f32(A[r * 4 + kg], Bc[r * 4 + kg], c,
                                           0, 0, 0);
#pragma unroll
  for (int i = 0; i < 4; ++i) D[(kg * 4 + i) * 16 + r] = c[i];
}

extern "C" void launch_mfma_probe_bf16(const void* A, const void* Bc,
                                       float* D, hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_bf16_k, dim3(1), dim3(64), 0, stream,
                     reinterpret_cast<const u32*>(A),
                     reinterpret_cast<const u32*>(Bc), D);
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_mfma_probe_f32(const float* A, const float* Bc,
                                      float* D, hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_f32_k, dim3(1), dim3(64), 0, stream, A, Bc,
                     D);
  HIP_CHECK(hipGetLastError());
}
