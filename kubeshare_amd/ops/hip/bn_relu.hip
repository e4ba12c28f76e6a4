// Fused NHWC bf16 BatchNorm + ReLU (+ residual add) for gfx950.
//
// Motivation (profiles/resnet50_bs256_kernel_breakdown_untuned.txt):
// MIOpen's spatial BN kernels + separate ReLU/add elementwise passes
// are ~48% of the non-conv GPU time of a ResNet50 training step. The
// stock path makes 5 full passes over the activation in forward and 8
// in backward; these kernels make 3 and 7, and fold the residual add
// and its gradient for free.
//
// Design (cdna_hip_programming.md G13 vectorize, G7 ILP, G11 grid):
//  - channels-last bf16, C % 8 == 0: one lane owns 8 CONSECUTIVE
//    channels (one 16-byte load) — per-channel reductions never cross
//    lanes, and a wave's 64 lanes cover a 1 KiB contiguous span.
//  - UNROLL independent row-chunks per loop iteration so each lane
//    keeps >=4 16-B loads in flight (1 load/lane measured ~2.3 TB/s,
//    HBM-latency-bound at 32 waves/CU).
//  - block partials + a tree-reduce in the finalize kernel instead of
//    per-channel global atomics (2*C atomics x 2048 blocks measured as
//    ~0.4 ms of fixed cost per BN call).
//  - all stats/parameter math in f32; tensors read/written as bf16x8.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <vector>

typedef __attribute__((ext_vector_type(8))) unsigned short ushort8;
typedef __attribute__((ext_vector_type(8))) float float8;

__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
  union {
    unsigned int i;
    float f;
  } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

__device__ __forceinline__ unsigned short f32_to_bf16(float f) {
  union {
    float f;
    unsigned int i;
  } v;
  v.f = f;
  // round-to-nearest-even (matches PyTorch's float->bf16 cast)
  unsigned int lsb = (v.i >> 16) & 1;
  v.i += 0x7fffu + lsb;
  return (unsigned short)(v.i >> 16);
}

#define KS_BN_BLOCK 512
#define KS_BN_UNROLL 4
#define KS_BN_UNROLL_STATS 8

// ------------------------------------------------------------- fwd stats
// Block partials: sum_part[b*C + c], sq_part[(B + b)*C + c].
__global__ void bn_stats_kernel(const ushort8* __restrict__ x,
                                float* __restrict__ part,
                                long long M, int CG) {
  __shared__ float s_red[KS_BN_BLOCK * 8];
  const int tid = threadIdx.x;
  const int cg = tid % CG;
  const int roff = tid / CG;
  const int rows_per_blk = KS_BN_BLOCK / CG;
  // when CG does not divide the block, the tail threads would alias the
  // NEXT block's rows — they must idle (their LDS slots stay zero)
  const bool active = roff < rows_per_blk;
  const long long stride = (long long)gridDim.x * rows_per_blk;

  float8 acc = {0, 0, 0, 0, 0, 0, 0, 0};
  float8 acc2 = {0, 0, 0, 0, 0, 0, 0, 0};
  long long row = active ? (long long)blockIdx.x * rows_per_blk + roff : M;
  // unrolled main loop: KS_BN_UNROLL_STATS independent loads in flight
  for (; row + (KS_BN_UNROLL_STATS - 1) * stride < M;
       row += KS_BN_UNROLL_STATS * stride) {
    ushort8 v[KS_BN_UNROLL_STATS];
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL_STATS; u++)
      v[u] = x[(row + u * stride) * CG + cg];
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL_STATS; u++)
#pragma unroll
      for (int j = 0; j < 8; j++) {
        float f = bf16_to_f32(v[u][j]);
        acc[j] += f;
        acc2[j] += f * f;
      }
  }
  for (; row < M; row += stride) {
    ushort8 v = x[row * CG + cg];
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float f = bf16_to_f32(v[j]);
      acc[j] += f;
      acc2[j] += f * f;
    }
  }

  const int C = CG * 8;
  float* sum_part = part + (long long)blockIdx.x * C;
  float* sq_part = part + (long long)(gridDim.x + blockIdx.x) * C;
  // LDS tree over the lanes sharing a channel group, then plain stores
#pragma unroll
  for (int j = 0; j < 8; j++) s_red[tid * 8 + j] = acc[j];
  __syncthreads();
  if (tid < CG) {
    float8 t = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < rows_per_blk; r++)
#pragma unroll
      for (int j = 0; j < 8; j++) t[j] += s_red[(r * CG + tid) * 8 + j];
#pragma unroll
    for (int j = 0; j < 8; j++) sum_part[tid * 8 + j] = t[j];
  }
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 8; j++) s_red[tid * 8 + j] = acc2[j];
  __syncthreads();
  if (tid < CG) {
    float8 t = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < rows_per_blk; r++)
#pragma unroll
      for (int j = 0; j < 8; j++) t[j] += s_red[(r * CG + tid) * 8 + j];
#pragma unroll
    for (int j = 0; j < 8; j++) sq_part[tid * 8 + j] = t[j];
  }
}

// ------------------------------------------------- partial reduction
// part[b*C + c] (+ the sq/dscale half at offset nblocks*C) -> out[c].
// One 64-lane wave per 4 channels: lanes stride over blocks (all loads
// independent -> one latency, not nblocks of them), then shfl-reduce.
__global__ void bn_reduce_partials_kernel(const float* __restrict__ part,
                                          int nblocks, int C,
                                          float* __restrict__ out0,
                                          float* __restrict__ out1) {
  const int lane = threadIdx.x;        // 0..63 over blocks
  const int c = blockIdx.x * blockDim.y + threadIdx.y;
  if (c >= C) return;
  float a0 = 0.f, a1 = 0.f;
  for (int b = lane; b < nblocks; b += 64) {
    a0 += part[(long long)b * C + c];
    a1 += part[(long long)(nblocks + b) * C + c];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    a0 += __shfl_down(a0, off, 64);
    a1 += __shfl_down(a1, off, 64);
  }
  if (lane == 0) {
    out0[c] = a0;
    out1[c] = a1;
  }
}

// ------------------------------------------------------- fwd finalize
// Reduce the block partials, then mean/invstd, running-stat update
// (PyTorch semantics), and folded scale' = w*invstd, bias' = b - mean*s.
__global__ void bn_finalize_kernel(const float* __restrict__ sum,
                                   const float* __restrict__ sumsq,
                                   const float* __restrict__ weight,
                                   const float* __restrict__ bias,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float* __restrict__ save_mean,
                                   float* __restrict__ save_invstd,
                                   float* __restrict__ scale_out,
                                   float* __restrict__ bias_out,
                                   long long M, int C, float momentum,
                                   float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean = sum[c] / (float)M;
  float var = sumsq[c] / (float)M - mean * mean;
  var = var < 0.f ? 0.f : var;
  float invstd = rsqrtf(var + eps);
  save_mean[c] = mean;
  save_invstd[c] = invstd;
  if (running_mean) {
    float unbiased = M > 1 ? var * (float)M / (float)(M - 1) : var;
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
  float s = weight[c] * invstd;
  scale_out[c] = s;
  bias_out[c] = bias[c] - mean * s;
}

// --------------------------------------------------------- fwd apply
// y = [relu](x*scale' + bias' [+ res]); scale/bias in registers.
// RELU=false serves the reference ResNet's downsample-path BNs (no
// activation), which otherwise fall back to MIOpen's fp32 spatial BN.
template <bool WITH_RES, bool RELU = true>
__global__ void bn_apply_relu_kernel(const ushort8* __restrict__ x,
                                     const ushort8* __restrict__ res,
                                     ushort8* __restrict__ y,
                                     const float* __restrict__ scale,
                                     const float* __restrict__ biasf,
                                     long long M, int CG) {
  const int tid = threadIdx.x;
  const int cg = tid % CG;
  const int roff = tid / CG;
  const int rows_per_blk = KS_BN_BLOCK / CG;
  const bool active = roff < rows_per_blk;  // see bn_stats_kernel
  const long long stride = (long long)gridDim.x * rows_per_blk;
  float8 s, b;
#pragma unroll
  for (int j = 0; j < 8; j++) {
    s[j] = scale[cg * 8 + j];
    b[j] = biasf[cg * 8 + j];
  }
  long long row = active ? (long long)blockIdx.x * rows_per_blk + roff : M;
  for (; row + (KS_BN_UNROLL - 1) * stride < M;
       row += KS_BN_UNROLL * stride) {
    ushort8 v[KS_BN_UNROLL], r[KS_BN_UNROLL], o[KS_BN_UNROLL];
    long long k[KS_BN_UNROLL];
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
      k[u] = (row + u * stride) * CG + cg;
      v[u] = x[k[u]];
      if (WITH_RES) r[u] = res[k[u]];
    }
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
#pragma unroll
      for (int j = 0; j < 8; j++) {
        float f = fmaf(bf16_to_f32(v[u][j]), s[j], b[j]);
        if (WITH_RES) f += bf16_to_f32(r[u][j]);
        o[u][j] = f32_to_bf16(RELU && f < 0.f ? 0.f : f);
      }
      y[k[u]] = o[u];
    }
  }
  for (; row < M; row += stride) {
    const long long k = row * CG + cg;
    ushort8 v = x[k];
    ushort8 r;
    if (WITH_RES) r = res[k];
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float f = fmaf(bf16_to_f32(v[j]), s[j], b[j]);
      if (WITH_RES) f += bf16_to_f32(r[j]);
      o[j] = f32_to_bf16(RELU && f < 0.f ? 0.f : f);
    }
    y[k] = o;
  }
}

// --------------------------------------------------------- bwd stats
// dym = dy * (y > 0); dbias[c] = sum(dym); dscale[c] = sum(dym*xhat);
// optionally writes dym (= the residual branch's gradient).
template <bool WRITE_DYM>
__global__ void bn_bwd_stats_kernel(const ushort8* __restrict__ x,
                                    const ushort8* __restrict__ y,
                                    const ushort8* __restrict__ dy,
                                    ushort8* __restrict__ dym_out,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    float* __restrict__ part,
                                    long long M, int CG) {
  __shared__ float s_red[KS_BN_BLOCK * 8];
  const int tid = threadIdx.x;
  const int cg = tid % CG;
  const int roff = tid / CG;
  const int rows_per_blk = KS_BN_BLOCK / CG;
  const bool active = roff < rows_per_blk;  // see bn_stats_kernel
  const long long stride = (long long)gridDim.x * rows_per_blk;
  float8 mu, is;
#pragma unroll
  for (int j = 0; j < 8; j++) {
    mu[j] = mean[cg * 8 + j];
    is[j] = invstd[cg * 8 + j];
  }
  float8 adb = {0, 0, 0, 0, 0, 0, 0, 0};
  float8 ads = {0, 0, 0, 0, 0, 0, 0, 0};
  long long row = active ? (long long)blockIdx.x * rows_per_blk + roff : M;
  for (; row + (KS_BN_UNROLL - 1) * stride < M;
       row += KS_BN_UNROLL * stride) {
    ushort8 xv[KS_BN_UNROLL], yv[KS_BN_UNROLL], gv[KS_BN_UNROLL];
    long long k[KS_BN_UNROLL];
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
      k[u] = (row + u * stride) * CG + cg;
      xv[u] = x[k[u]];
      yv[u] = y[k[u]];
      gv[u] = dy[k[u]];
    }
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
      ushort8 dm;
#pragma unroll
      for (int j = 0; j < 8; j++) {
        float g = bf16_to_f32(yv[u][j]) > 0.f ? bf16_to_f32(gv[u][j]) : 0.f;
        float xhat = (bf16_to_f32(xv[u][j]) - mu[j]) * is[j];
        adb[j] += g;
        ads[j] += g * xhat;
        if (WRITE_DYM) dm[j] = f32_to_bf16(g);
      }
      if (WRITE_DYM) dym_out[k[u]] = dm;
    }
  }
  for (; row < M; row += stride) {
    const long long k = row * CG + cg;
    ushort8 xv = x[k], yv = y[k], gv = dy[k];
    ushort8 dm;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float g = bf16_to_f32(yv[j]) > 0.f ? bf16_to_f32(gv[j]) : 0.f;
      float xhat = (bf16_to_f32(xv[j]) - mu[j]) * is[j];
      adb[j] += g;
      ads[j] += g * xhat;
      if (WRITE_DYM) dm[j] = f32_to_bf16(g);
    }
    if (WRITE_DYM) dym_out[k] = dm;
  }

  const int C = CG * 8;
  float* db_part = part + (long long)blockIdx.x * C;
  float* ds_part = part + (long long)(gridDim.x + blockIdx.x) * C;
#pragma unroll
  for (int j = 0; j < 8; j++) s_red[tid * 8 + j] = adb[j];
  __syncthreads();
  if (tid < CG) {
    float8 t = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < rows_per_blk; r++)
#pragma unroll
      for (int j = 0; j < 8; j++) t[j] += s_red[(r * CG + tid) * 8 + j];
#pragma unroll
    for (int j = 0; j < 8; j++) db_part[tid * 8 + j] = t[j];
  }
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 8; j++) s_red[tid * 8 + j] = ads[j];
  __syncthreads();
  if (tid < CG) {
    float8 t = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < rows_per_blk; r++)
#pragma unroll
      for (int j = 0; j < 8; j++) t[j] += s_red[(r * CG + tid) * 8 + j];
#pragma unroll
    for (int j = 0; j < 8; j++) ds_part[tid * 8 + j] = t[j];
  }
}

// ----------------------------------------- bwd stats, no-y variant
// Non-residual BNs: the ReLU mask is recomputed from x and the folded
// per-channel constants (y>0 <=> bf16(relu(s*x+b)) != 0, replicating
// the forward's rounding exactly), so the backward never re-reads y:
// 7 tensor passes -> 5 for the non-residual blocks of ResNet50.
// RELU=false (downsample BNs): no mask, g = dy.
template <bool RELU = true>
__global__ void bn_bwd_stats_noy_kernel(const ushort8* __restrict__ x,
                                        const ushort8* __restrict__ dy,
                                        const float* __restrict__ mean,
                                        const float* __restrict__ invstd,
                                        const float* __restrict__ weight,
                                        const float* __restrict__ bias,
                                        float* __restrict__ part,
                                        long long M, int CG) {
  __shared__ float s_red[KS_BN_BLOCK * 8];
  const int tid = threadIdx.x;
  const int cg = tid % CG;
  const int roff = tid / CG;
  const int rows_per_blk = KS_BN_BLOCK / CG;
  const bool active = roff < rows_per_blk;  // see bn_stats_kernel
  const long long stride = (long long)gridDim.x * rows_per_blk;
  float8 mu, is, sc, bf;
#pragma unroll
  for (int j = 0; j < 8; j++) {
    const int c = cg * 8 + j;
    mu[j] = mean[c];
    is[j] = invstd[c];
    sc[j] = weight[c] * is[j];           // folded scale (as in fwd)
    bf[j] = bias[c] - mu[j] * sc[j];     // folded bias
  }
  float8 adb = {0, 0, 0, 0, 0, 0, 0, 0};
  float8 ads = {0, 0, 0, 0, 0, 0, 0, 0};
  long long row = active ? (long long)blockIdx.x * rows_per_blk + roff : M;
  for (; row + (KS_BN_UNROLL - 1) * stride < M;
       row += KS_BN_UNROLL * stride) {
    ushort8 xv[KS_BN_UNROLL], gv[KS_BN_UNROLL];
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
      const long long k = (row + u * stride) * CG + cg;
      xv[u] = x[k];
      gv[u] = dy[k];
    }
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++)
#pragma unroll
      for (int j = 0; j < 8; j++) {
        float xf = bf16_to_f32(xv[u][j]);
        float g = bf16_to_f32(gv[u][j]);
        if (RELU) {
          // exact fwd replication: y stored as bf16(relu(f))
          float f = fmaf(xf, sc[j], bf[j]);
          if (!f32_to_bf16(f > 0.f ? f : 0.f)) g = 0.f;
        }
        adb[j] += g;
        ads[j] += g * ((xf - mu[j]) * is[j]);
      }
  }
  for (; row < M; row += stride) {
    const long long k = row * CG + cg;
    ushort8 xv = x[k], gv = dy[k];
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float xf = bf16_to_f32(xv[j]);
      float g = bf16_to_f32(gv[j]);
      if (RELU) {
        float f = fmaf(xf, sc[j], bf[j]);
        if (!f32_to_bf16(f > 0.f ? f : 0.f)) g = 0.f;
      }
      adb[j] += g;
      ads[j] += g * ((xf - mu[j]) * is[j]);
    }
  }

  const int C = CG * 8;
  float* db_part = part + (long long)blockIdx.x * C;
  float* ds_part = part + (long long)(gridDim.x + blockIdx.x) * C;
#pragma unroll
  for (int j = 0; j < 8; j++) s_red[tid * 8 + j] = adb[j];
  __syncthreads();
  if (tid < CG) {
    float8 t = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < rows_per_blk; r++)
#pragma unroll
      for (int j = 0; j < 8; j++) t[j] += s_red[(r * CG + tid) * 8 + j];
#pragma unroll
    for (int j = 0; j < 8; j++) db_part[tid * 8 + j] = t[j];
  }
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 8; j++) s_red[tid * 8 + j] = ads[j];
  __syncthreads();
  if (tid < CG) {
    float8 t = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < rows_per_blk; r++)
#pragma unroll
      for (int j = 0; j < 8; j++) t[j] += s_red[(r * CG + tid) * 8 + j];
#pragma unroll
    for (int j = 0; j < 8; j++) ds_part[tid * 8 + j] = t[j];
  }
}

// ----------------------------------------- bwd apply, no-y variant
template <bool RELU = true>
__global__ void bn_bwd_apply_noy_kernel(const ushort8* __restrict__ x,
                                        const ushort8* __restrict__ dy,
                                        ushort8* __restrict__ dx,
                                        const float* __restrict__ mean,
                                        const float* __restrict__ invstd,
                                        const float* __restrict__ weight,
                                        const float* __restrict__ bias,
                                        const float* __restrict__ dbias,
                                        const float* __restrict__ dscale,
                                        long long M, int CG) {
  const int tid = threadIdx.x;
  const int cg = tid % CG;
  const int roff = tid / CG;
  const int rows_per_blk = KS_BN_BLOCK / CG;
  const bool active = roff < rows_per_blk;
  const long long stride = (long long)gridDim.x * rows_per_blk;
  const float invM = 1.f / (float)M;
  float8 mu, is, sc, bf, w, db, ds;
#pragma unroll
  for (int j = 0; j < 8; j++) {
    const int c = cg * 8 + j;
    mu[j] = mean[c];
    is[j] = invstd[c];
    sc[j] = weight[c] * is[j];
    bf[j] = bias[c] - mu[j] * sc[j];
    w[j] = sc[j];
    db[j] = dbias[c] * invM;
    ds[j] = dscale[c] * invM;
  }
  long long row = active ? (long long)blockIdx.x * rows_per_blk + roff : M;
  for (; row + (KS_BN_UNROLL - 1) * stride < M;
       row += KS_BN_UNROLL * stride) {
    ushort8 xv[KS_BN_UNROLL], gv[KS_BN_UNROLL];
    long long k[KS_BN_UNROLL];
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
      k[u] = (row + u * stride) * CG + cg;
      xv[u] = x[k[u]];
      gv[u] = dy[k[u]];
    }
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
      ushort8 o;
#pragma unroll
      for (int j = 0; j < 8; j++) {
        float xf = bf16_to_f32(xv[u][j]);
        float g = bf16_to_f32(gv[u][j]);
        if (RELU) {
          float f = fmaf(xf, sc[j], bf[j]);
          if (!f32_to_bf16(f > 0.f ? f : 0.f)) g = 0.f;
        }
        float xhat = (xf - mu[j]) * is[j];
        o[j] = f32_to_bf16(w[j] * (g - db[j] - xhat * ds[j]));
      }
      dx[k[u]] = o;
    }
  }
  for (; row < M; row += stride) {
    const long long k = row * CG + cg;
    ushort8 xv = x[k], gv = dy[k];
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float xf = bf16_to_f32(xv[j]);
      float g = bf16_to_f32(gv[j]);
      if (RELU) {
        float f = fmaf(xf, sc[j], bf[j]);
        if (!f32_to_bf16(f > 0.f ? f : 0.f)) g = 0.f;
      }
      float xhat = (xf - mu[j]) * is[j];
      o[j] = f32_to_bf16(w[j] * (g - db[j] - xhat * ds[j]));
    }
    dx[k] = o;
  }
}

// ----------------------------------------- bwd apply from dym
// Residual BNs: the stats pass already wrote dym (= dres); read it
// instead of y AND dy — 8 tensor passes -> 7.
__global__ void bn_bwd_apply_dym_kernel(const ushort8* __restrict__ x,
                                        const ushort8* __restrict__ dym,
                                        ushort8* __restrict__ dx,
                                        const float* __restrict__ mean,
                                        const float* __restrict__ invstd,
                                        const float* __restrict__ weight,
                                        const float* __restrict__ dbias,
                                        const float* __restrict__ dscale,
                                        long long M, int CG) {
  const int tid = threadIdx.x;
  const int cg = tid % CG;
  const int roff = tid / CG;
  const int rows_per_blk = KS_BN_BLOCK / CG;
  const bool active = roff < rows_per_blk;
  const long long stride = (long long)gridDim.x * rows_per_blk;
  const float invM = 1.f / (float)M;
  float8 mu, is, w, db, ds;
#pragma unroll
  for (int j = 0; j < 8; j++) {
    const int c = cg * 8 + j;
    mu[j] = mean[c];
    is[j] = invstd[c];
    w[j] = weight[c] * is[j];
    db[j] = dbias[c] * invM;
    ds[j] = dscale[c] * invM;
  }
  long long row = active ? (long long)blockIdx.x * rows_per_blk + roff : M;
  for (; row + (KS_BN_UNROLL - 1) * stride < M;
       row += KS_BN_UNROLL * stride) {
    ushort8 xv[KS_BN_UNROLL], gv[KS_BN_UNROLL];
    long long k[KS_BN_UNROLL];
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
      k[u] = (row + u * stride) * CG + cg;
      xv[u] = x[k[u]];
      gv[u] = dym[k[u]];
    }
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
      ushort8 o;
#pragma unroll
      for (int j = 0; j < 8; j++) {
        float g = bf16_to_f32(gv[u][j]);  // already ReLU-masked
        float xhat = (bf16_to_f32(xv[u][j]) - mu[j]) * is[j];
        o[j] = f32_to_bf16(w[j] * (g - db[j] - xhat * ds[j]));
      }
      dx[k[u]] = o;
    }
  }
  for (; row < M; row += stride) {
    const long long k = row * CG + cg;
    ushort8 xv = x[k], gv = dym[k];
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float g = bf16_to_f32(gv[j]);
      float xhat = (bf16_to_f32(xv[j]) - mu[j]) * is[j];
      o[j] = f32_to_bf16(w[j] * (g - db[j] - xhat * ds[j]));
    }
    dx[k] = o;
  }
}

// --------------------------------------------------------- bwd apply
// dx = w*invstd * (dym - dbias/M - xhat * dscale/M)
__global__ void bn_bwd_apply_kernel(const ushort8* __restrict__ x,
                                    const ushort8* __restrict__ y,
                                    const ushort8* __restrict__ dy,
                                    ushort8* __restrict__ dx,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ weight,
                                    const float* __restrict__ dbias,
                                    const float* __restrict__ dscale,
                                    long long M, int CG) {
  const int tid = threadIdx.x;
  const int cg = tid % CG;
  const int roff = tid / CG;
  const int rows_per_blk = KS_BN_BLOCK / CG;
  const bool active = roff < rows_per_blk;  // see bn_stats_kernel
  const long long stride = (long long)gridDim.x * rows_per_blk;
  const float invM = 1.f / (float)M;
  float8 mu, is, w, db, ds;
#pragma unroll
  for (int j = 0; j < 8; j++) {
    const int c = cg * 8 + j;
    mu[j] = mean[c];
    is[j] = invstd[c];
    w[j] = weight[c] * is[j];
    db[j] = dbias[c] * invM;
    ds[j] = dscale[c] * invM;
  }
  long long row = active ? (long long)blockIdx.x * rows_per_blk + roff : M;
  for (; row + (KS_BN_UNROLL - 1) * stride < M;
       row += KS_BN_UNROLL * stride) {
    ushort8 xv[KS_BN_UNROLL], yv[KS_BN_UNROLL], gv[KS_BN_UNROLL];
    long long k[KS_BN_UNROLL];
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
      k[u] = (row + u * stride) * CG + cg;
      xv[u] = x[k[u]];
      yv[u] = y[k[u]];
      gv[u] = dy[k[u]];
    }
#pragma unroll
    for (int u = 0; u < KS_BN_UNROLL; u++) {
      ushort8 o;
#pragma unroll
      for (int j = 0; j < 8; j++) {
        float g = bf16_to_f32(yv[u][j]) > 0.f ? bf16_to_f32(gv[u][j]) : 0.f;
        float xhat = (bf16_to_f32(xv[u][j]) - mu[j]) * is[j];
        o[j] = f32_to_bf16(w[j] * (g - db[j] - xhat * ds[j]));
      }
      dx[k[u]] = o;
    }
  }
  for (; row < M; row += stride) {
    const long long k = row * CG + cg;
    ushort8 xv = x[k], yv = y[k], gv = dy[k];
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float g = bf16_to_f32(yv[j]) > 0.f ? bf16_to_f32(gv[j]) : 0.f;
      float xhat = (bf16_to_f32(xv[j]) - mu[j]) * is[j];
      o[j] = f32_to_bf16(w[j] * (g - db[j] - xhat * ds[j]));
    }
    dx[k] = o;
  }
}

// ---------------------------------------------------------- eval apply
__global__ void bn_fold_eval_kernel(const float* __restrict__ weight,
                                    const float* __restrict__ bias,
                                    const float* __restrict__ rmean,
                                    const float* __restrict__ rvar,
                                    float* __restrict__ scale_out,
                                    float* __restrict__ bias_out, int C,
                                    float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s = weight[c] * rsqrtf(rvar[c] + eps);
  scale_out[c] = s;
  bias_out[c] = bias[c] - rmean[c] * s;
}

// ================================================================ host
namespace {

struct Geom {
  long long M;
  int C, CG, blocks, stat_blocks;
};

Geom geom_of(const torch::Tensor& x) {
  TORCH_CHECK(x.dim() == 4, "bn_relu: 4D NCHW tensor expected");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "bn_relu: bf16 only");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "bn_relu: channels-last only");
  Geom g;
  g.C = (int)x.size(1);
  TORCH_CHECK(g.C % 8 == 0 && g.C <= KS_BN_BLOCK * 8,
              "bn_relu: C must be a multiple of 8 and <= 8*KS_BN_BLOCK");
  g.M = x.numel() / g.C;
  g.CG = g.C / 8;
  long long work = g.M * g.CG;
  long long blocks = (work + KS_BN_BLOCK - 1) / KS_BN_BLOCK;
  // G11: cap + grid-stride (256 CUs want >>256 workgroups)
  g.blocks = (int)std::min<long long>(blocks, 2048);
  g.stat_blocks = (int)std::min<long long>(blocks, 1024);
  if (g.stat_blocks < 1) g.stat_blocks = 1;
  return g;
}

}  // namespace

std::vector<torch::Tensor> bn_relu_fwd_train(
    torch::Tensor x, torch::Tensor weight, torch::Tensor bias,
    torch::Tensor running_mean, torch::Tensor running_var, double momentum,
    double eps, c10::optional<torch::Tensor> res, bool relu) {
  Geom g = geom_of(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto f32 = x.options().dtype(at::kFloat);
  auto part = at::empty({(long long)2 * g.stat_blocks * g.C}, f32);
  auto y = at::empty_like(x);
  auto save_mean = at::empty({g.C}, f32);
  auto save_invstd = at::empty({g.C}, f32);
  auto scale = at::empty({g.C}, f32);
  auto biasf = at::empty({g.C}, f32);
  auto sums = at::empty({2 * g.C}, f32);

  hipLaunchKernelGGL(bn_stats_kernel, dim3(g.stat_blocks),
                     dim3(KS_BN_BLOCK), 0, stream.stream(),
                     reinterpret_cast<const ushort8*>(x.data_ptr()),
                     part.data_ptr<float>(), g.M, g.CG);
  hipLaunchKernelGGL(bn_reduce_partials_kernel,
                     dim3((g.C + 3) / 4), dim3(64, 4), 0, stream.stream(),
                     part.data_ptr<float>(), g.stat_blocks, g.C,
                     sums.data_ptr<float>(), sums.data_ptr<float>() + g.C);
  int fb = (g.C + 255) / 256;
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(fb), dim3(256), 0,
                     stream.stream(), sums.data_ptr<float>(),
                     sums.data_ptr<float>() + g.C,
                     weight.data_ptr<float>(), bias.data_ptr<float>(),
                     running_mean.defined()
                         ? running_mean.data_ptr<float>() : nullptr,
                     running_var.defined()
                         ? running_var.data_ptr<float>() : nullptr,
                     save_mean.data_ptr<float>(),
                     save_invstd.data_ptr<float>(), scale.data_ptr<float>(),
                     biasf.data_ptr<float>(), g.M, g.C, (float)momentum,
                     (float)eps);
  auto launch_apply = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(g.blocks), dim3(KS_BN_BLOCK), 0,
                       stream.stream(),
                       reinterpret_cast<const ushort8*>(x.data_ptr()),
                       res.has_value()
                           ? reinterpret_cast<const ushort8*>(res->data_ptr())
                           : nullptr,
                       reinterpret_cast<ushort8*>(y.data_ptr()),
                       scale.data_ptr<float>(), biasf.data_ptr<float>(), g.M,
                       g.CG);
  };
  if (res.has_value())
    relu ? launch_apply(bn_apply_relu_kernel<true, true>)
         : launch_apply(bn_apply_relu_kernel<true, false>);
  else
    relu ? launch_apply(bn_apply_relu_kernel<false, true>)
         : launch_apply(bn_apply_relu_kernel<false, false>);
  return {y, save_mean, save_invstd};
}

torch::Tensor bn_relu_fwd_eval(torch::Tensor x, torch::Tensor weight,
                               torch::Tensor bias, torch::Tensor rmean,
                               torch::Tensor rvar, double eps,
                               c10::optional<torch::Tensor> res, bool relu) {
  Geom g = geom_of(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto f32 = x.options().dtype(at::kFloat);
  auto scale = at::empty({g.C}, f32);
  auto biasf = at::empty({g.C}, f32);
  auto y = at::empty_like(x);
  int fb = (g.C + 255) / 256;
  hipLaunchKernelGGL(bn_fold_eval_kernel, dim3(fb), dim3(256), 0,
                     stream.stream(), weight.data_ptr<float>(),
                     bias.data_ptr<float>(), rmean.data_ptr<float>(),
                     rvar.data_ptr<float>(), scale.data_ptr<float>(),
                     biasf.data_ptr<float>(), g.C, (float)eps);
  auto launch_apply = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(g.blocks), dim3(KS_BN_BLOCK), 0,
                       stream.stream(),
                       reinterpret_cast<const ushort8*>(x.data_ptr()),
                       res.has_value()
                           ? reinterpret_cast<const ushort8*>(res->data_ptr())
                           : nullptr,
                       reinterpret_cast<ushort8*>(y.data_ptr()),
                       scale.data_ptr<float>(), biasf.data_ptr<float>(), g.M,
                       g.CG);
  };
  if (res.has_value())
    relu ? launch_apply(bn_apply_relu_kernel<true, true>)
         : launch_apply(bn_apply_relu_kernel<true, false>);
  else
    relu ? launch_apply(bn_apply_relu_kernel<false, true>)
         : launch_apply(bn_apply_relu_kernel<false, false>);
  return y;
}

std::vector<torch::Tensor> bn_relu_bwd(torch::Tensor x, torch::Tensor y,
                                       torch::Tensor dy, torch::Tensor weight,
                                       torch::Tensor bias, torch::Tensor mean,
                                       torch::Tensor invstd, bool need_dres,
                                       bool relu) {
  TORCH_CHECK(relu || !need_dres,
              "bn_relu_bwd: residual path requires the ReLU variant");
  Geom g = geom_of(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto f32 = x.options().dtype(at::kFloat);
  auto part = at::empty({(long long)2 * g.stat_blocks * g.C}, f32);
  auto red = at::empty({g.C * 2}, f32);
  float* dbias_p = red.data_ptr<float>();
  float* dscale_p = dbias_p + g.C;
  auto dx = at::empty_like(x);
  torch::Tensor dres;
  if (!dy.is_contiguous(at::MemoryFormat::ChannelsLast))
    dy = dy.contiguous(at::MemoryFormat::ChannelsLast);

  if (need_dres) {
    // residual: the stats pass writes dym (= dres), the apply pass
    // reads it back instead of y AND dy (8 tensor passes -> 7)
    dres = at::empty_like(x);
    hipLaunchKernelGGL(bn_bwd_stats_kernel<true>, dim3(g.stat_blocks),
                       dim3(KS_BN_BLOCK), 0, stream.stream(),
                       reinterpret_cast<const ushort8*>(x.data_ptr()),
                       reinterpret_cast<const ushort8*>(y.data_ptr()),
                       reinterpret_cast<const ushort8*>(dy.data_ptr()),
                       reinterpret_cast<ushort8*>(dres.data_ptr()),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       part.data_ptr<float>(), g.M, g.CG);
    hipLaunchKernelGGL(bn_reduce_partials_kernel,
                       dim3((g.C + 3) / 4), dim3(64, 4), 0, stream.stream(),
                       part.data_ptr<float>(), g.stat_blocks, g.C,
                       dbias_p, dscale_p);
    hipLaunchKernelGGL(bn_bwd_apply_dym_kernel, dim3(g.blocks),
                       dim3(KS_BN_BLOCK), 0, stream.stream(),
                       reinterpret_cast<const ushort8*>(x.data_ptr()),
                       reinterpret_cast<const ushort8*>(dres.data_ptr()),
                       reinterpret_cast<ushort8*>(dx.data_ptr()),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       weight.data_ptr<float>(), dbias_p, dscale_p,
                       g.M, g.CG);
    auto dbias = red.narrow(0, 0, g.C);
    auto dscale = red.narrow(0, g.C, g.C);
    return {dx, dscale, dbias, dres};
  }
  // non-residual: the ReLU mask is recomputed from x and the folded
  // scale/bias, so y is never read (7 tensor passes -> 5)
  auto launch_stats = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(g.stat_blocks), dim3(KS_BN_BLOCK), 0,
                       stream.stream(),
                       reinterpret_cast<const ushort8*>(x.data_ptr()),
                       reinterpret_cast<const ushort8*>(dy.data_ptr()),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       weight.data_ptr<float>(), bias.data_ptr<float>(),
                       part.data_ptr<float>(), g.M, g.CG);
  };
  relu ? launch_stats(bn_bwd_stats_noy_kernel<true>)
       : launch_stats(bn_bwd_stats_noy_kernel<false>);
  hipLaunchKernelGGL(bn_reduce_partials_kernel,
                     dim3((g.C + 3) / 4), dim3(64, 4), 0, stream.stream(),
                     part.data_ptr<float>(), g.stat_blocks, g.C,
                     dbias_p, dscale_p);
  auto launch_bapply = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(g.blocks), dim3(KS_BN_BLOCK), 0,
                       stream.stream(),
                       reinterpret_cast<const ushort8*>(x.data_ptr()),
                       reinterpret_cast<const ushort8*>(dy.data_ptr()),
                       reinterpret_cast<ushort8*>(dx.data_ptr()),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       weight.data_ptr<float>(), bias.data_ptr<float>(),
                       dbias_p, dscale_p, g.M, g.CG);
  };
  relu ? launch_bapply(bn_bwd_apply_noy_kernel<true>)
       : launch_bapply(bn_bwd_apply_noy_kernel<false>);
  auto dbias = red.narrow(0, 0, g.C);
  auto dscale = red.narrow(0, g.C, g.C);
  return {dx, dscale, dbias};
}
