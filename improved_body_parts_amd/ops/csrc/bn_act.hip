// Fused BatchNorm + LeakyReLU (+ residual add) over NHWC activations.
//
// Replaces the reference's nn.BatchNorm2d + nn.LeakyReLU pairs (cuDNN ops,
// reference models/layers_transposed.py:90-120) with single-pass CDNA4 kernels:
// the normalisation is folded to y = act(scale[c] * x + shift[c] (+ res)) with
// fp32 statistics, bf16 activations, 16-byte vector accesses per lane.
//
// Exposed ops (see bindings.cpp):
//   bn_stats        : per-channel sum / sum-of-squares of an [M][C] view
//   bn_act_fwd      : fused scale/shift + optional residual + optional leaky
//   bn_act_bwd      : dpre = dy * act'(y); per-channel Σdpre, Σdpre*xhat
//   bn_act_bwd_apply: dx from dpre + the reduced sums (training BN backward)
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace ibp {

// --------------------------------------------------------------------------
// per-channel statistics: sums[c] = Σ_m x[m][c], sumsq[c] = Σ_m x[m][c]^2
// grid.y tiles channels by blockDim.x; grid.x strides rows; fp32 atomics.
// --------------------------------------------------------------------------
template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x, float* __restrict__ sums,
                                float* __restrict__ sumsq, long long M, int C) {
  int c = blockIdx.y * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s = 0.f, q = 0.f;
  for (long long m = blockIdx.x; m < M; m += gridDim.x) {
    float v = ldf(x + m * C + c);
    s += v;
    q += v * v;
  }
  atomicAdd(&sums[c], s);
  atomicAdd(&sumsq[c], q);
}

// --------------------------------------------------------------------------
// y = act(scale[c]*x + shift[c] (+ res)); bf16 path moves 8 elems per lane.
// --------------------------------------------------------------------------
typedef unsigned short ushort8 __attribute__((ext_vector_type(8)));

// --------------------------------------------------------------------------
// vectorized bf16 stats: each lane owns 8 adjacent channels (one 16-B load
// per row), a block covers cpb channel-lanes x rpb row-groups. Partial sums
// cross the row-groups through LDS, then each block writes its cpb x 8
// channel partials to a [2][rows][C] workspace with PLAIN stores; a second
// tiny kernel reduces the rows. No atomics anywhere: global fp32 atomics
// serialise per address (~1024 colliding blocks cost a flat ~240 us floor
// on every call — measured, gpurun_out/bn_v8.txt of 2026-09-13).
// --------------------------------------------------------------------------
__global__ void bn_stats_bf16v8(const ushort8* __restrict__ x,
                                float* __restrict__ ws,  // [2][rows][C]
                                long long M, int C8, int cpb, int rpb) {
  __shared__ float red[256 * 8];
  const int tid = threadIdx.x;
  const int cl = tid % cpb;
  const int rs = tid / cpb;
  const int ch8 = blockIdx.y * cpb + cl;
  const bool active = ch8 < C8 && rs < rpb;
  float s[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float q[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  if (active) {
    const long long step = (long long)gridDim.x * rpb;
    long long m = (long long)blockIdx.x * rpb + rs;
    // 4 rows in flight per thread: a single serial load chain leaves the
    // kernel latency-bound at ~2 TB/s (measured)
    for (; m + 3 * step < M; m += 4 * step) {
      ushort8 x0 = x[m * C8 + ch8];
      ushort8 x1 = x[(m + step) * C8 + ch8];
      ushort8 x2 = x[(m + 2 * step) * C8 + ch8];
      ushort8 x3 = x[(m + 3 * step) * C8 + ch8];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v0 = us2f(x0[j]), v1 = us2f(x1[j]);
        float v2 = us2f(x2[j]), v3 = us2f(x3[j]);
        s[j] += (v0 + v1) + (v2 + v3);
        q[j] += (v0 * v0 + v1 * v1) + (v2 * v2 + v3 * v3);
      }
    }
    for (; m < M; m += step) {
      ushort8 xv = x[m * C8 + ch8];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = us2f(xv[j]);
        s[j] += v;
        q[j] += v * v;
      }
    }
  }
  const long long C = (long long)C8 * 8;
  const long long plane = (long long)gridDim.x * C;
  for (int pass = 0; pass < 2; ++pass) {
    const float* src = pass == 0 ? s : q;
    __syncthreads();
    #pragma unroll
    for (int j = 0; j < 8; ++j) red[tid * 8 + j] = active ? src[j] : 0.f;
    __syncthreads();
    if (rs == 0 && ch8 < C8) {
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float acc = 0.f;
        for (int r = 0; r < rpb; ++r) acc += red[(r * cpb + cl) * 8 + j];
        ws[pass * plane + (long long)blockIdx.x * C + ch8 * 8 + j] = acc;
      }
    }
  }
}

// stage 2: out[p][c] = sum_r ws[p][r][c]. 64 channel-lanes x 4 row-lanes per
// block, 8 independent accumulators per thread -> 32 loads in flight per
// lane-group (a single serial column walk costs rows x ~0.3us of latency).
__global__ __launch_bounds__(1024) void colsum_kernel(
    const float* __restrict__ ws, float* __restrict__ out, int rows, int C,
    int planes) {
  __shared__ float red[1024];
  const int total = C * planes;
  const int cl = threadIdx.x & 63;
  const int rl = threadIdx.x >> 6;  // 0..15
  const int i = blockIdx.x * 64 + cl;
  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  if (i < total) {
    const int p = i / C;
    const int c = i - p * C;
    const float* src = ws + (long long)p * rows * C + c;
    for (int r = rl; r < rows; r += 128) {
      #pragma unroll
      for (int u = 0; u < 8; ++u)
        if (r + u * 16 < rows) acc[u] += src[(long long)(r + u * 16) * C];
    }
  }
  float a = 0.f;
  #pragma unroll
  for (int u = 0; u < 8; ++u) a += acc[u];
  red[threadIdx.x] = a;
  __syncthreads();
  if (rl == 0 && i < total) {
    #pragma unroll
    for (int g = 1; g < 16; ++g) a += red[g * 64 + cl];
    out[i] = a;
  }
}

// column-sum + finalize fused: reduces the [2][rows][C] stats workspace AND
// computes mean/invstd/scale/shift (+ running stats) in one launch — the
// separate colsum was ~280 extra launches per training step (one per BN
// layer; 3.7% of the step in rocprof r2).
__global__ __launch_bounds__(1024) void colsum_finalize_kernel(
    const float* __restrict__ ws /*[2][rows][C]*/,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    long long* __restrict__ num_batches, float* __restrict__ out /*[4][C]*/,
    int rows, int C, long long M, float momentum, float eps) {
  __shared__ float red[2][1024];
  const int cl = threadIdx.x & 63;
  const int rl = threadIdx.x >> 6;  // 0..15
  const int c = blockIdx.x * 64 + cl;
  float a0 = 0.f, a1 = 0.f;
  if (c < C) {
    const float* s0 = ws + c;
    const float* s1 = ws + (long long)rows * C + c;
    for (int r = rl; r < rows; r += 16) {
      a0 += s0[(long long)r * C];
      a1 += s1[(long long)r * C];
    }
  }
  red[0][threadIdx.x] = a0;
  red[1][threadIdx.x] = a1;
  __syncthreads();
  if (rl == 0 && c < C) {
    #pragma unroll
    for (int g = 1; g < 16; ++g) {
      a0 += red[0][g * 64 + cl];
      a1 += red[1][g * 64 + cl];
    }
    const float invM = 1.0f / (float)M;
    float mean = a0 * invM;
    float var = fmaxf(a1 * invM - mean * mean, 0.f);
    float invstd = rsqrtf(var + eps);
    if (running_mean) {
      float unbiased = var * ((float)M / (float)(M > 1 ? M - 1 : 1));
      running_mean[c] = running_mean[c] * (1.f - momentum) + mean * momentum;
      running_var[c] = running_var[c] * (1.f - momentum) + unbiased * momentum;
    }
    float scale = gamma[c] * invstd;
    float shift = beta[c] - mean * scale;
    out[c] = mean;
    out[C + c] = invstd;
    out[2 * C + c] = scale;
    out[3 * C + c] = shift;
    if (c == 0 && num_batches) *num_batches += 1;
  }
}

// fold the per-channel BN statistics epilogue into ONE kernel: the Python
// mean/var/rsqrt/scale/shift chain was ~6 tiny fp32 launches per conv layer
// (~8% of the training step, rocprof). Updates running stats in place.
__global__ void bn_finalize_kernel(
    const float* __restrict__ sums, const float* __restrict__ sumsq,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    long long* __restrict__ num_batches, float* __restrict__ out /*[4][C]*/,
    long long M, int C, float momentum, float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float invM = 1.0f / (float)M;
  float mean = sums[c] * invM;
  float var = fmaxf(sumsq[c] * invM - mean * mean, 0.f);
  float invstd = rsqrtf(var + eps);
  if (running_mean) {
    float unbiased = var * ((float)M / (float)(M > 1 ? M - 1 : 1));
    running_mean[c] = running_mean[c] * (1.f - momentum) + mean * momentum;
    running_var[c] = running_var[c] * (1.f - momentum) + unbiased * momentum;
  }
  float scale = gamma[c] * invstd;
  float shift = beta[c] - mean * scale;
  out[c] = mean;
  out[C + c] = invstd;
  out[2 * C + c] = scale;
  out[3 * C + c] = shift;
  if (c == 0 && num_batches) *num_batches += 1;
}

typedef float float4w __attribute__((ext_vector_type(4)));

__global__ void bn_act_fwd_bf16v8(const unsigned short* __restrict__ x,
                                  const unsigned short* __restrict__ res,
                                  unsigned short* __restrict__ y,
                                  const float* __restrict__ scale,
                                  const float* __restrict__ shift,
                                  long long total8, int C8, float slope, int act) {
  const ushort8* xv = reinterpret_cast<const ushort8*>(x);
  const ushort8* rv = reinterpret_cast<const ushort8*>(res);
  ushort8* yv = reinterpret_cast<ushort8*>(y);
  const float4w* sc4 = reinterpret_cast<const float4w*>(scale);
  const float4w* sh4 = reinterpret_cast<const float4w*>(shift);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total8;
       i += (long long)gridDim.x * blockDim.x) {
    const int q8 = (int)(i % C8) * 2;
    float4w sc[2] = {sc4[q8], sc4[q8 + 1]};
    float4w sh[2] = {sh4[q8], sh4[q8 + 1]};
    ushort8 xi = xv[i];
    ushort8 out;
    if (res != nullptr) {
      ushort8 ri = rv[i];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = us2f(xi[j]) * sc[j >> 2][j & 3] + sh[j >> 2][j & 3]
                  + us2f(ri[j]);
        out[j] = f2us(act ? leaky(v, slope) : v);
      }
    } else {
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = us2f(xi[j]) * sc[j >> 2][j & 3] + sh[j >> 2][j & 3];
        out[j] = f2us(act ? leaky(v, slope) : v);
      }
    }
    yv[i] = out;
  }
}

// vectorized bf16 backward reduce: same block geometry as bn_stats_bf16v8;
// dpre is written back as one 16-B store per row.
template <bool NEED_XHAT>
__global__ void bn_act_bwd_reduce_bf16v8(
    const ushort8* __restrict__ dy, const ushort8* __restrict__ y,
    const ushort8* __restrict__ x, ushort8* __restrict__ dpre_out,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    float* __restrict__ sum_dpre /* ws [2][rows][C] */,
    long long M, int C8, int cpb, int rpb, float slope, int act) {
  __shared__ float red[256 * 8];
  const int tid = threadIdx.x;
  const int cl = tid % cpb;
  const int rs = tid / cpb;
  const int ch8 = blockIdx.y * cpb + cl;
  const bool active = ch8 < C8 && rs < rpb;
  float s[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float sx[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float mu[8], is[8];
  if (active && NEED_XHAT) {
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      mu[j] = mean[ch8 * 8 + j];
      is[j] = invstd[ch8 * 8 + j];
    }
  }
  if (active) {
    const long long step = (long long)gridDim.x * rpb;
    long long m = (long long)blockIdx.x * rpb + rs;
    // 2 rows in flight (3 input streams each — 6 loads outstanding)
    for (; m + step < M; m += 2 * step) {
      const long long i0 = m * C8 + ch8;
      const long long i1 = (m + step) * C8 + ch8;
      ushort8 gd0 = dy[i0], gd1 = dy[i1];
      ushort8 yy0, yy1;
      if (act) { yy0 = y[i0]; yy1 = y[i1]; }
      ushort8 xx0, xx1;
      if (NEED_XHAT) { xx0 = x[i0]; xx1 = x[i1]; }
      ushort8 o0, o1;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g0 = us2f(gd0[j]), g1 = us2f(gd1[j]);
        if (act) {
          g0 = us2f(yy0[j]) > 0.f ? g0 : g0 * slope;
          g1 = us2f(yy1[j]) > 0.f ? g1 : g1 * slope;
        }
        o0[j] = f2us(g0);
        o1[j] = f2us(g1);
        s[j] += g0 + g1;
        if (NEED_XHAT)
          sx[j] += (g0 * (us2f(xx0[j]) - mu[j]) + g1 * (us2f(xx1[j]) - mu[j]))
                   * is[j];
      }
      dpre_out[i0] = o0;
      dpre_out[i1] = o1;
    }
    for (; m < M; m += step) {
      const long long i = m * C8 + ch8;
      ushort8 gd = dy[i];
      ushort8 yy;
      if (act) yy = y[i];
      ushort8 xx;
      if (NEED_XHAT) xx = x[i];
      ushort8 out;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = us2f(gd[j]);
        if (act) g = us2f(yy[j]) > 0.f ? g : g * slope;
        out[j] = f2us(g);
        s[j] += g;
        if (NEED_XHAT) sx[j] += g * (us2f(xx[j]) - mu[j]) * is[j];
      }
      dpre_out[i] = out;
    }
  }
  // per-block partials to ws[2][rows][C] (plain stores; colsum finishes)
  const long long C = (long long)C8 * 8;
  const long long plane = (long long)gridDim.x * C;
  for (int pass = 0; pass < 2; ++pass) {
    const float* src = pass == 0 ? s : sx;
    __syncthreads();
    #pragma unroll
    for (int j = 0; j < 8; ++j) red[tid * 8 + j] = active ? src[j] : 0.f;
    __syncthreads();
    if (rs == 0 && ch8 < C8) {
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float acc = 0.f;
        for (int r = 0; r < rpb; ++r) acc += red[(r * cpb + cl) * 8 + j];
        sum_dpre[pass * plane + (long long)blockIdx.x * C + ch8 * 8 + j] = acc;
      }
    }
    if (!NEED_XHAT) break;  // sx plane not needed in eval mode
  }
}

typedef float float4v __attribute__((ext_vector_type(4)));

// vectorized bf16 BN backward apply: dx = P[c]*dpre + Q[c]*x + R[c].
// Per-channel coefficients are fetched as float4 pairs — 6 dwordx4 instead of
// 24 scalar dword loads per iteration (scalar form was vmem-issue bound).
__global__ void bn_act_bwd_apply_bf16v8(
    const ushort8* __restrict__ dpre, const ushort8* __restrict__ x,
    ushort8* __restrict__ dx, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ sum_dpre, const float* __restrict__ sum_dxhat,
    long long total8, int C8, int training, float invM) {
  const int C = C8 * 8;
  // PQR computed cooperatively into LDS — drops the separate per-layer
  // coefficient kernel launch (~280/step)
  extern __shared__ float pqr[];  // [3][C]
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    const float is = invstd[c];
    const float A = gamma[c] * is;
    float Q = 0.f, R = 0.f;
    if (training) {
      Q = -A * is * sum_dxhat[c] * invM;
      R = -A * sum_dpre[c] * invM - Q * mean[c];
    }
    pqr[c] = A;
    pqr[C + c] = Q;
    pqr[2 * C + c] = R;
  }
  __syncthreads();
  const float4v* pqrv = reinterpret_cast<const float4v*>(pqr);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total8;
       i += (long long)gridDim.x * blockDim.x) {
    const int q8 = (int)(i % C8) * 2;  // float4-group index of channel c8
    float4v P[2], Q[2], R[2];
    P[0] = pqrv[q8];
    P[1] = pqrv[q8 + 1];
    if (training) {
      Q[0] = pqrv[C / 4 + q8];
      Q[1] = pqrv[C / 4 + q8 + 1];
      R[0] = pqrv[C / 2 + q8];
      R[1] = pqrv[C / 2 + q8 + 1];
    }
    ushort8 g8 = dpre[i];
    ushort8 x8;
    if (training) x8 = x[i];
    ushort8 out;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = P[j >> 2][j & 3] * us2f(g8[j]);
      if (training) v += Q[j >> 2][j & 3] * us2f(x8[j]) + R[j >> 2][j & 3];
      out[j] = f2us(v);
    }
    dx[i] = out;
  }
}

template <typename T>
__global__ void bn_act_fwd_scalar(const T* __restrict__ x, const T* __restrict__ res,
                                  T* __restrict__ y, const float* __restrict__ scale,
                                  const float* __restrict__ shift, long long total,
                                  int C, float slope, int act) {
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    float v = ldf(x + i) * scale[c] + shift[c];
    if (res != nullptr) v += ldf(res + i);
    stf(y + i, act ? leaky(v, slope) : v);
  }
}

// --------------------------------------------------------------------------
// backward part 1: dpre = dy * (act ? (y > 0 ? 1 : slope) : 1)
//                  sum_dpre[c] += dpre ; sum_dxhat[c] += dpre * (x-mean)*invstd
// --------------------------------------------------------------------------
template <typename T, bool NEED_XHAT>
__global__ void bn_act_bwd_reduce_kernel(
    const T* __restrict__ dy, const T* __restrict__ y, const T* __restrict__ x,
    T* __restrict__ dpre_out, const float* __restrict__ mean,
    const float* __restrict__ invstd, float* __restrict__ sum_dpre,
    float* __restrict__ sum_dxhat, long long M, int C, float slope, int act) {
  int c = blockIdx.y * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s = 0.f, sx = 0.f;
  float mu = NEED_XHAT ? mean[c] : 0.f;
  float is = NEED_XHAT ? invstd[c] : 0.f;
  for (long long m = blockIdx.x; m < M; m += gridDim.x) {
    long long i = m * C + c;
    float g = ldf(dy + i);
    if (act) {
      float yy = ldf(y + i);
      g = yy > 0.f ? g : g * slope;
    }
    stf(dpre_out + i, g);
    s += g;
    if (NEED_XHAT) sx += g * (ldf(x + i) - mu) * is;
  }
  atomicAdd(&sum_dpre[c], s);
  if (NEED_XHAT) atomicAdd(&sum_dxhat[c], sx);
}

// --------------------------------------------------------------------------
// backward part 2 (training BN):
//   dx = gamma*invstd * (dpre - sum_dpre/M - xhat * sum_dxhat/M)
// eval BN: dx = gamma*invstd*dpre  (pass sums = nullptr)
// --------------------------------------------------------------------------
template <typename T>
__global__ void bn_act_bwd_apply_kernel(
    const T* __restrict__ dpre, const T* __restrict__ x, T* __restrict__ dx,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ sum_dpre,
    const float* __restrict__ sum_dxhat, long long M, int C) {
  long long total = M * C;
  float invM = 1.0f / (float)M;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    float g = ldf(dpre + i);
    float is = invstd[c];
    float w = gamma[c] * is;
    if (sum_dpre != nullptr) {
      float xhat = (ldf(x + i) - mean[c]) * is;
      g = g - sum_dpre[c] * invM - xhat * sum_dxhat[c] * invM;
    }
    stf(dx + i, w * g);
  }
}

}  // namespace ibp

// ===========================================================================
// host wrappers
// ===========================================================================
using torch::Tensor;

static inline bool dense_ok(const Tensor& t) {
  return t.is_contiguous() || t.is_contiguous(at::MemoryFormat::ChannelsLast);
}

static inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

static inline bool use_v8() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("IBP_BN_V8");
    v = (e == nullptr || e[0] != '0') ? 1 : 0;
  }
  return v == 1;
}

// pick the (channel-lanes, row-groups, row-blocks) geometry for the v8 kernels
static inline void v8_geometry(long long M, int C8, int& cpb, int& rpb,
                               int& rows) {
  cpb = std::min(C8, 256);
  rpb = 256 / cpb;
  // >=8 rows of work per thread; rows bounds the stage-2 column reduction
  rows = (int)std::min<long long>((M + (long long)rpb * 8 - 1) / (rpb * 8), 512);
  rows = std::max(rows, 1);
}

std::vector<Tensor> bn_stats(const Tensor& x_mc, int64_t C) {
  TORCH_CHECK(x_mc.is_cuda() && dense_ok(x_mc));
  long long M = x_mc.numel() / C;
  auto opts = x_mc.options().dtype(torch::kFloat32);
  dim3 block(256);
  if (use_v8() && x_mc.scalar_type() == at::ScalarType::BFloat16 && C % 8 == 0) {
    int C8 = (int)C / 8, cpb, rpb, rows;
    v8_geometry(M, C8, cpb, rpb, rows);
    Tensor ws = torch::empty({2LL * rows * C}, opts);
    Tensor both = torch::empty({2, C}, opts);
    dim3 grid(rows, (C8 + cpb - 1) / cpb);
    hipLaunchKernelGGL(ibp::bn_stats_bf16v8, grid, block, 0, cur_stream(),
                       reinterpret_cast<const ibp::ushort8*>(x_mc.data_ptr()),
                       ws.data_ptr<float>(), M, C8, cpb, rpb);
    dim3 cgrid((2 * (int)C + 63) / 64);  // colsum covers 64 channels/block
    hipLaunchKernelGGL(ibp::colsum_kernel, cgrid, dim3(1024), 0, cur_stream(),
                       ws.data_ptr<float>(), both.data_ptr<float>(),
                       rows, (int)C, 2);
    return {both[0], both[1]};
  }
  // one allocation + one async memset for both accumulators (a torch::zeros
  // pair costs two FillFunctor launches per conv layer — visible in rocprof)
  Tensor both = torch::empty({2, C}, opts);
  hipMemsetAsync(both.data_ptr(), 0, 2 * C * sizeof(float), cur_stream());
  Tensor sums = both[0];
  Tensor sumsq = both[1];
  int rows = (int)std::min<long long>((M + 63) / 64, 1024);
  dim3 grid(rows, (C + 255) / 256);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      x_mc.scalar_type(), "bn_stats", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(ibp::bn_stats_kernel<T>, grid, block, 0, cur_stream(),
                       reinterpret_cast<const T*>(x_mc.data_ptr()),
                       sums.data_ptr<float>(), sumsq.data_ptr<float>(), M, (int)C);
  });
  return {sums, sumsq};
}

// per-channel epilogue over given sums (SyncBN path all-reduces the sums
// between bn_stats and this); returns [mean, invstd, scale, shift].
std::vector<Tensor> bn_finalize(
    const Tensor& sums, const Tensor& sumsq, int64_t M, const Tensor& gamma,
    const Tensor& beta, const c10::optional<Tensor>& running_mean,
    const c10::optional<Tensor>& running_var,
    const c10::optional<Tensor>& num_batches, double momentum, double eps) {
  int64_t C = sums.numel();
  Tensor out = torch::empty({4, C}, sums.options().dtype(torch::kFloat32));
  dim3 block(256), grid(((int)C + 255) / 256);
  hipLaunchKernelGGL(ibp::bn_finalize_kernel, grid, block, 0, cur_stream(),
                     sums.data_ptr<float>(), sumsq.data_ptr<float>(),
                     gamma.data_ptr<float>(), beta.data_ptr<float>(),
                     running_mean ? running_mean->data_ptr<float>() : nullptr,
                     running_var ? running_var->data_ptr<float>() : nullptr,
                     num_batches ? reinterpret_cast<long long*>(
                         num_batches->data_ptr<int64_t>()) : nullptr,
                     out.data_ptr<float>(), M, (int)C, (float)momentum,
                     (float)eps);
  return {out[0], out[1], out[2], out[3]};
}

// stats + column-sum + per-channel epilogue in one call; returns
// [mean, invstd, scale, shift] (views of one [4][C] tensor). Training path:
// also updates running_mean/var (+ num_batches_tracked) in place.
std::vector<Tensor> bn_stats_finalize(
    const Tensor& x_mc, int64_t C, const Tensor& gamma, const Tensor& beta,
    const c10::optional<Tensor>& running_mean,
    const c10::optional<Tensor>& running_var,
    const c10::optional<Tensor>& num_batches, double momentum, double eps) {
  TORCH_CHECK(gamma.scalar_type() == at::ScalarType::Float &&
              beta.scalar_type() == at::ScalarType::Float,
              "bn_stats_finalize expects fp32 BN affine parameters");
  long long M = x_mc.numel() / C;
  if (use_v8() && x_mc.scalar_type() == at::ScalarType::BFloat16 && C % 8 == 0) {
    // fused path: stats workspace -> colsum+finalize in ONE kernel
    auto fopts = x_mc.options().dtype(torch::kFloat32);
    int C8 = (int)C / 8, cpb, rpb, rows;
    v8_geometry(M, C8, cpb, rpb, rows);
    Tensor ws = torch::empty({2LL * rows * C}, fopts);
    dim3 block(256), grid(rows, (C8 + cpb - 1) / cpb);
    hipLaunchKernelGGL(ibp::bn_stats_bf16v8, grid, block, 0, cur_stream(),
                       reinterpret_cast<const ibp::ushort8*>(x_mc.data_ptr()),
                       ws.data_ptr<float>(), M, C8, cpb, rpb);
    Tensor out = torch::empty({4, C}, fopts);
    dim3 cgrid(((int)C + 63) / 64);
    hipLaunchKernelGGL(ibp::colsum_finalize_kernel, cgrid, dim3(1024), 0,
                       cur_stream(), ws.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       running_mean ? running_mean->data_ptr<float>() : nullptr,
                       running_var ? running_var->data_ptr<float>() : nullptr,
                       num_batches ? reinterpret_cast<long long*>(
                           num_batches->data_ptr<int64_t>()) : nullptr,
                       out.data_ptr<float>(), rows, (int)C, M,
                       (float)momentum, (float)eps);
    return {out[0], out[1], out[2], out[3]};
  }
  auto sv = bn_stats(x_mc, C);
  Tensor out = torch::empty({4, C}, x_mc.options().dtype(torch::kFloat32));
  dim3 block(256), grid(((int)C + 255) / 256);
  hipLaunchKernelGGL(ibp::bn_finalize_kernel, grid, block, 0, cur_stream(),
                     sv[0].data_ptr<float>(), sv[1].data_ptr<float>(),
                     gamma.data_ptr<float>(), beta.data_ptr<float>(),
                     running_mean ? running_mean->data_ptr<float>() : nullptr,
                     running_var ? running_var->data_ptr<float>() : nullptr,
                     num_batches ? reinterpret_cast<long long*>(num_batches->data_ptr<int64_t>()) : nullptr,
                     out.data_ptr<float>(), M, (int)C, (float)momentum,
                     (float)eps);
  return {out[0], out[1], out[2], out[3]};
}

Tensor bn_act_fwd(const Tensor& x, const Tensor& scale, const Tensor& shift,
                  const c10::optional<Tensor>& residual, double slope, bool act) {
  TORCH_CHECK(x.is_cuda() && dense_ok(x));
  int C = (int)scale.numel();
  long long total = x.numel();
  Tensor y = torch::empty_like(x);
  const void* resp = nullptr;
  if (residual.has_value()) {
    TORCH_CHECK(dense_ok(*residual) && residual->numel() == total);
    resp = residual->data_ptr();
  }
  dim3 block(256);
  if (x.scalar_type() == at::ScalarType::BFloat16 && C % 8 == 0 &&
      total % 8 == 0) {
    long long total8 = total / 8;
    dim3 grid(ibp::grid_1d(total8, 256, 8192));
    hipLaunchKernelGGL(ibp::bn_act_fwd_bf16v8, grid, block, 0, cur_stream(),
                       reinterpret_cast<const unsigned short*>(x.data_ptr()),
                       reinterpret_cast<const unsigned short*>(resp),
                       reinterpret_cast<unsigned short*>(y.data_ptr()),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       total8, C / 8, (float)slope, act ? 1 : 0);
  } else {
    dim3 grid(ibp::grid_1d(total, 256, 8192));
    AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
        x.scalar_type(), "bn_act_fwd", [&] {
      using T = scalar_t;
      hipLaunchKernelGGL(ibp::bn_act_fwd_scalar<T>, grid, block, 0, cur_stream(),
                         reinterpret_cast<const T*>(x.data_ptr()),
                         reinterpret_cast<const T*>(resp),
                         reinterpret_cast<T*>(y.data_ptr()),
                         scale.data_ptr<float>(), shift.data_ptr<float>(),
                         total, C, (float)slope, act ? 1 : 0);
    });
  }
  return y;
}

std::vector<Tensor> bn_act_bwd(const Tensor& dy, const Tensor& y, const Tensor& x,
                               const c10::optional<Tensor>& mean,
                               const c10::optional<Tensor>& invstd,
                               double slope, bool act, bool need_xhat, int64_t C) {
  TORCH_CHECK(dy.is_cuda() && dense_ok(dy) && dense_ok(y));
  long long M = dy.numel() / C;
  auto fopts = dy.options().dtype(torch::kFloat32);
  Tensor dpre = torch::empty_like(dy);
  dim3 block(256);
  if (use_v8() && dy.scalar_type() == at::ScalarType::BFloat16 && C % 8 == 0) {
    int C8 = (int)C / 8, cpb, rpb, rows;
    v8_geometry(M, C8, cpb, rpb, rows);
    Tensor ws = torch::empty({2LL * rows * C}, fopts);
    Tensor both = torch::empty({2, C}, fopts);
    if (!need_xhat) {
      // only the sum_dpre plane is produced; second plane must read as zero
      hipMemsetAsync(ws.data_ptr<float>() + (long long)rows * C, 0,
                     (size_t)rows * C * sizeof(float), cur_stream());
    }
    dim3 gridv(rows, (C8 + cpb - 1) / cpb);
    auto launch = [&](auto need) {
      hipLaunchKernelGGL((ibp::bn_act_bwd_reduce_bf16v8<decltype(need)::value>),
                         gridv, block, 0, cur_stream(),
                         reinterpret_cast<const ibp::ushort8*>(dy.data_ptr()),
                         reinterpret_cast<const ibp::ushort8*>(y.data_ptr()),
                         reinterpret_cast<const ibp::ushort8*>(x.data_ptr()),
                         reinterpret_cast<ibp::ushort8*>(dpre.data_ptr()),
                         need.value ? mean->data_ptr<float>() : nullptr,
                         need.value ? invstd->data_ptr<float>() : nullptr,
                         ws.data_ptr<float>(),
                         M, C8, cpb, rpb, (float)slope, act ? 1 : 0);
    };
    if (need_xhat) launch(std::true_type{});
    else launch(std::false_type{});
    dim3 cgrid((2 * (int)C + 63) / 64);  // colsum covers 64 channels/block
    hipLaunchKernelGGL(ibp::colsum_kernel, cgrid, dim3(1024), 0, cur_stream(),
                       ws.data_ptr<float>(), both.data_ptr<float>(),
                       rows, (int)C, 2);
    return {dpre, both[0], both[1]};
  }
  Tensor both = torch::empty({2, C}, fopts);
  hipMemsetAsync(both.data_ptr(), 0, 2 * C * sizeof(float), cur_stream());
  Tensor sum_dpre = both[0];
  Tensor sum_dxhat = both[1];
  int rows = (int)std::min<long long>((M + 63) / 64, 1024);
  dim3 grid(rows, (C + 255) / 256);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      dy.scalar_type(), "bn_act_bwd", [&] {
    using T = scalar_t;
    if (need_xhat) {
      hipLaunchKernelGGL((ibp::bn_act_bwd_reduce_kernel<T, true>), grid, block, 0,
                         cur_stream(),
                         reinterpret_cast<const T*>(dy.data_ptr()),
                         reinterpret_cast<const T*>(y.data_ptr()),
                         reinterpret_cast<const T*>(x.data_ptr()),
                         reinterpret_cast<T*>(dpre.data_ptr()),
                         mean->data_ptr<float>(), invstd->data_ptr<float>(),
                         sum_dpre.data_ptr<float>(), sum_dxhat.data_ptr<float>(),
                         M, (int)C, (float)slope, act ? 1 : 0);
    } else {
      hipLaunchKernelGGL((ibp::bn_act_bwd_reduce_kernel<T, false>), grid, block, 0,
                         cur_stream(),
                         reinterpret_cast<const T*>(dy.data_ptr()),
                         reinterpret_cast<const T*>(y.data_ptr()),
                         reinterpret_cast<const T*>(x.data_ptr()),
                         reinterpret_cast<T*>(dpre.data_ptr()),
                         nullptr, nullptr,
                         sum_dpre.data_ptr<float>(), sum_dxhat.data_ptr<float>(),
                         M, (int)C, (float)slope, act ? 1 : 0);
    }
  });
  return {dpre, sum_dpre, sum_dxhat};
}

Tensor bn_act_bwd_apply(const Tensor& dpre, const Tensor& x, const Tensor& mean,
                        const Tensor& invstd, const Tensor& gamma,
                        const c10::optional<Tensor>& sum_dpre,
                        const c10::optional<Tensor>& sum_dxhat, int64_t C) {
  long long M = dpre.numel() / C;
  Tensor dx = torch::empty_like(dpre);
  dim3 block(256);
  if (use_v8() && dpre.scalar_type() == at::ScalarType::BFloat16 && C % 8 == 0) {
    long long total8 = dpre.numel() / 8;
    bool training = sum_dpre.has_value();
    dim3 gridv(ibp::grid_1d(total8, 256, 8192));
    const size_t lds = 3 * (size_t)C * sizeof(float);
    hipLaunchKernelGGL(ibp::bn_act_bwd_apply_bf16v8, gridv, block, lds,
                       cur_stream(),
                       reinterpret_cast<const ibp::ushort8*>(dpre.data_ptr()),
                       reinterpret_cast<const ibp::ushort8*>(x.data_ptr()),
                       reinterpret_cast<ibp::ushort8*>(dx.data_ptr()),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(),
                       training ? sum_dpre->data_ptr<float>() : nullptr,
                       training ? sum_dxhat->data_ptr<float>() : nullptr,
                       total8, (int)C / 8, training ? 1 : 0,
                       1.0f / (float)M);
    return dx;
  }
  dim3 grid(ibp::grid_1d(dpre.numel(), 256, 8192));
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      dpre.scalar_type(), "bn_act_bwd_apply", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(ibp::bn_act_bwd_apply_kernel<T>, grid, block, 0, cur_stream(),
                       reinterpret_cast<const T*>(dpre.data_ptr()),
                       reinterpret_cast<const T*>(x.data_ptr()),
                       reinterpret_cast<T*>(dx.data_ptr()),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(),
                       sum_dpre.has_value() ? sum_dpre->data_ptr<float>() : nullptr,
                       sum_dxhat.has_value() ? sum_dxhat->data_ptr<float>() : nullptr,
                       M, (int)C);
  });
  return dx;
}
