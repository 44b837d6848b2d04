// BigCLAM CDNA4 (gfx950 / MI355X) HIP kernels.
//
// Hand-written for the CDNA4 execution model: 64-wide wavefronts, LDS-staged
// per-node gradient accumulators, float4-vectorized HBM access, wave-shuffle
// reductions, fp64 LLH accumulation.  Kernel worklist per SURVEY.md §2-B:
//   K1 edge_grad_llh  — fused per-node gradient + local LLH
//                       (replaces codes/bigclamv3-7.scala:138-150)
//   K2 linesearch     — all 16 Armijo candidates per node in one edge pass
//                       (replaces the cartesian fan-out, scala:153-163)
//   K4 llh_only       — per-node local LLH (scala:106-120 / 177-200)
//
// Conventions:
//  * F is fp32 row-major [n_rows, K] with K padded to a multiple of 4 (rows
//    16B-aligned -> clean float4 coalescing).  Pad columns are identically
//    zero and stay zero through every op (grad_pad = -sumF_pad + F_pad = 0).
//  * one 256-thread workgroup (4 waves) per LOCAL node; the launch walks
//    nodes in degree-descending `order` so hub blocks start first.
//  * wave-per-edge: each wave owns one neighbor at a time -> no
//    __syncthreads in the edge loop; the K-dim dot is a lane-strided
//    float4 loop + shfl_xor butterfly (all 64 lanes end with the sum).
//  * per-node LLH is accumulated in fp64 (the convergence test is a 1e-4
//    relative change on a ~1e8-magnitude sum).
//
// NOTE: math must match tests/oracle.py bit-for-bit in structure:
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

// ---------------------------------------------------------------- reductions

__device__ __forceinline__ float wave_allreduce_sum(float v) {
  // butterfly: every lane ends with the full 64-lane sum
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ double wave_allreduce_sum(double v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// block-wide sum of one float per thread; `red` is 2*NWAVE floats of LDS.
// Result valid on every thread.  Costs 2 __syncthreads.
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

// ------------------------------------------------------------------- K1

extern "C" __global__ void __launch_bounds__(BLOCK) k1_grad_llh(
    const float* __restrict__ F, const long long* __restrict__ indptr,
    const int* __restrict__ indices, const float* __restrict__ sumF,
    const int* __restrict__ order, float* __restrict__ grad,
    double* __restrict__ llh, int n_local, int K, float min_p, float max_p) {
  const int u = order[blockIdx.x];
  const long long e0 = indptr[u];
  const long long e1 = indptr[u + 1];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* gacc = reinterpret_cast<float*>(smem);               // K floats
  double* wllh = reinterpret_cast<double*>(smem + (size_t)K * 4);  // NWAVE
  float* red = reinterpret_cast<float*>(wllh + NWAVE);        // 2*NWAVE

  for (int k = tid; k < K; k += BLOCK) gacc[k] = 0.f;
  __syncthreads();

  const float* __restrict__ fu = F + (size_t)u * K;
  double llh_w = 0.0;

  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const float* __restrict__ fv = F + (size_t)indices[e] * K;
    float part = 0.f;
    for (int k = lane * 4; k < K; k += WAVE * 4) {
      part = dot4(*reinterpret_cast<const float4*>(fu + k),
                  *reinterpret_cast<const float4*>(fv + k), part);
    }
    const float x = wave_allreduce_sum(part);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    const float w = 1.f / (1.f - p);
    if (lane == 0) llh_w += (double)log1pf(-p) + (double)x;
    // axpy into the shared accumulator (other waves write other neighbors
    // concurrently -> LDS float atomics; fv is L1-hot from the dot pass)
    for (int k = lane * 4; k < K; k += WAVE * 4) {
      const float4 b = *reinterpret_cast<const float4*>(fv + k);
      atomicAdd(&gacc[k + 0], w * b.x);
      atomicAdd(&gacc[k + 1], w * b.y);
      atomicAdd(&gacc[k + 2], w * b.z);
      atomicAdd(&gacc[k + 3], w * b.w);
    }
  }
  if (lane == 0) wllh[wid] = llh_w;
  __syncthreads();

  // node terms: -Fu.sumF + Fu.Fu (block-cooperative dots)
  float p_fs = 0.f, p_ff = 0.f;
  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    const float4 a = *reinterpret_cast<const float4*>(fu + k);
    const float4 s = *reinterpret_cast<const float4*>(sumF + k);
    p_fs = dot4(a, s, p_fs);
    p_ff = dot4(a, a, p_ff);
  }
  const float fs = block_allreduce_sum(p_fs, red);
  const float ff = block_allreduce_sum(p_ff, red);

  // write grad = gacc - sumF + fu
  float* __restrict__ gout = grad + (size_t)u * K;
  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    const float4 g = *reinterpret_cast<const float4*>(&gacc[k]);
    const float4 s = *reinterpret_cast<const float4*>(sumF + k);
    const float4 a = *reinterpret_cast<const float4*>(fu + k);
    float4 o;
    o.x = g.x - s.x + a.x;
    o.y = g.y - s.y + a.y;
    o.z = g.z - s.z + a.z;
    o.w = g.w - s.w + a.w;
    *reinterpret_cast<float4*>(gout + k) = o;
  }
  if (tid == 0) {
    double t = 0.0;
#pragma unroll
    for (int wv = 0; wv < NWAVE; ++wv) t += wllh[wv];
    llh[u] = t + (double)(-fs) + (double)ff;
  }
}

// ------------------------------------------------------------------- K4

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
  __shared__ __attribute__((aligned(16))) float red[2 * NWAVE];

  const float* __restrict__ fu = F + (size_t)u * K;
  double llh_w = 0.0;
  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const float* __restrict__ fv = F + (size_t)indices[e] * K;
    float part = 0.f;
    for (int k = lane * 4; k < K; k += WAVE * 4) {
      part = dot4(*reinterpret_cast<const float4*>(fu + k),
                  *reinterpret_cast<const float4*>(fv + k), part);
    }
    const float x = wave_allreduce_sum(part);
    const float p = clamp_p(__expf(-x), min_p, max_p);
    if (lane == 0) llh_w += (double)log1pf(-p) + (double)x;
  }
  if (lane == 0) wllh[wid] = llh_w;
  __syncthreads();

  float p_fs = 0.f, p_ff = 0.f;
  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    const float4 a = *reinterpret_cast<const float4*>(fu + k);
    const float4 s = *reinterpret_cast<const float4*>(sumF + k);
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
// All candidate steps evaluated in ONE pass over u's neighbors: each lane
// keeps MAX_LS partial dot accumulators; the candidate row
// cand_j = clamp(Fu + s_j*grad, MIN_F, MAX_F) is recomputed per float4 from
// the L1-hot Fu/grad rows (16 fma+clamp per loaded Fv float4), so Fv loads
// amortize over the whole ladder — the reference pays 16 full passes
// (SURVEY.md §2.8).

extern "C" __global__ void __launch_bounds__(BLOCK) k2_linesearch(
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

  __shared__ __attribute__((aligned(16))) double wacc[NWAVE][MAX_LS];
  __shared__ __attribute__((aligned(16))) float red[2 * NWAVE];
  __shared__ __attribute__((aligned(16))) float s_ladder[MAX_LS];

  if (tid < MAX_LS) s_ladder[tid] = (tid < n_ladder) ? ladder[tid] : 0.f;
  if (lane == 0) {
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) wacc[wid][j] = 0.0;
  }
  __syncthreads();

  const float* __restrict__ fu = F + (size_t)u * K;
  const float* __restrict__ gu = grad + (size_t)u * K;

  float s0, s1_, s2, s3, s4, s5, s6, s7, s8, s9, s10, s11, s12, s13, s14, s15;
  s0 = s_ladder[0]; s1_ = s_ladder[1]; s2 = s_ladder[2]; s3 = s_ladder[3];
  s4 = s_ladder[4]; s5 = s_ladder[5]; s6 = s_ladder[6]; s7 = s_ladder[7];
  s8 = s_ladder[8]; s9 = s_ladder[9]; s10 = s_ladder[10]; s11 = s_ladder[11];
  s12 = s_ladder[12]; s13 = s_ladder[13]; s14 = s_ladder[14]; s15 = s_ladder[15];

  double lacc[MAX_LS];
#pragma unroll
  for (int j = 0; j < MAX_LS; ++j) lacc[j] = 0.0;

  for (long long e = e0 + wid; e < e1; e += NWAVE) {
    const float* __restrict__ fv = F + (size_t)indices[e] * K;
    float acc[MAX_LS];
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) acc[j] = 0.f;
    for (int k = lane * 4; k < K; k += WAVE * 4) {
      const float4 a = *reinterpret_cast<const float4*>(fu + k);
      const float4 g = *reinterpret_cast<const float4*>(gu + k);
      const float4 b = *reinterpret_cast<const float4*>(fv + k);
#define LS_STEP(J, S)                                                       \
  {                                                                         \
    float4 c;                                                               \
    c.x = fminf(fmaxf(fmaf(S, g.x, a.x), min_f), max_f);                    \
    c.y = fminf(fmaxf(fmaf(S, g.y, a.y), min_f), max_f);                    \
    c.z = fminf(fmaxf(fmaf(S, g.z, a.z), min_f), max_f);                    \
    c.w = fminf(fmaxf(fmaf(S, g.w, a.w), min_f), max_f);                    \
    acc[J] = dot4(c, b, acc[J]);                                            \
  }
      LS_STEP(0, s0) LS_STEP(1, s1_) LS_STEP(2, s2) LS_STEP(3, s3)
      LS_STEP(4, s4) LS_STEP(5, s5) LS_STEP(6, s6) LS_STEP(7, s7)
      LS_STEP(8, s8) LS_STEP(9, s9) LS_STEP(10, s10) LS_STEP(11, s11)
      LS_STEP(12, s12) LS_STEP(13, s13) LS_STEP(14, s14) LS_STEP(15, s15)
#undef LS_STEP
    }
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) {
      const float x = wave_allreduce_sum(acc[j]);
      if (lane == 0) {
        const float p = clamp_p(__expf(-x), min_p, max_p);
        lacc[j] += (double)log1pf(-p) + (double)x;
      }
    }
  }
  if (lane == 0) {
#pragma unroll
    for (int j = 0; j < MAX_LS; ++j) wacc[wid][j] = lacc[j];
  }
  __syncthreads();

  // per-candidate node terms: cand_j.(Fu - sumF)   (identity: the
  // -cand.sumF' + cand.cand terms collapse; see header), plus gg = grad.grad.
  float p_gg = 0.f;
  for (int k = tid * 4; k < K; k += BLOCK * 4) {
    const float4 g = *reinterpret_cast<const float4*>(gu + k);
    p_gg = dot4(g, g, p_gg);
  }
  const float gg = block_allreduce_sum(p_gg, red);

  __shared__ double node_term[MAX_LS];
  for (int j = 0; j < n_ladder; ++j) {
    const float sj = s_ladder[j];
    float part = 0.f;
    for (int k = tid * 4; k < K; k += BLOCK * 4) {
      const float4 a = *reinterpret_cast<const float4*>(fu + k);
      const float4 g = *reinterpret_cast<const float4*>(gu + k);
      const float4 s = *reinterpret_cast<const float4*>(sumF + k);
      float4 c;
      c.x = fminf(fmaxf(fmaf(sj, g.x, a.x), min_f), max_f);
      c.y = fminf(fmaxf(fmaf(sj, g.y, a.y), min_f), max_f);
      c.z = fminf(fmaxf(fmaf(sj, g.z, a.z), min_f), max_f);
      c.w = fminf(fmaxf(fmaf(sj, g.w, a.w), min_f), max_f);
      part = fmaf(c.x, a.x - s.x, part);
      part = fmaf(c.y, a.y - s.y, part);
      part = fmaf(c.z, a.z - s.z, part);
      part = fmaf(c.w, a.w - s.w, part);
    }
    const float nt = block_allreduce_sum(part, red);
    if (tid == 0) node_term[j] = (double)nt;
  }
  __syncthreads();

  if (tid == 0) {
    const double llh_u = llh[u];
    float chosen = 0.f;
    for (int j = 0; j < n_ladder; ++j) {  // descending ladder: first accept
      double trial = node_term[j];
#pragma unroll
      for (int wv = 0; wv < NWAVE; ++wv) trial += wacc[wv][j];
      const float sj = s_ladder[j];
      if (trial >= llh_u + (double)(alpha * sj * gg)) {
        chosen = sj;
        break;
      }
    }
    best[u] = chosen;
  }
}
