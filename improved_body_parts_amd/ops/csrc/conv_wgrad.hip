// MFMA implicit-GEMM convolution WEIGHT gradient for CDNA4 (gfx950), NHWC bf16.
//
// Completes first-party ownership of the conv triple (fwd / dgrad / wgrad —
// the reference delegates all three to cuDNN; round-1 left
// wgrad on MIOpen igemm_wrw):
//
//   dW[k = (f, cin)][cout] = sum_m  x[pix(m) + tap(f)][cin] * dy[m][cout]
//     m enumerates output pixels (N*Ho*Wo), f = kh*KW + kw.
//
// The hard part (and why round 1 punted): the contraction index m is the
// STRIDED dimension of both NHWC operands, while MFMA wants each lane to hold
// 8 m-contiguous values. Solution: stage [64 m][16 ch] subtiles in LDS with a
// PERMUTED row order and read fragments with gfx950's hardware transpose read
// `ds_read_b64_tr_b16` (guide T10) — LDS writes stay vectorized b128 and the
// transpose itself is free.
//
//   * tr read semantics (verified by tr16_probe): 64 lanes, per-lane address
//     base + l*8B, delivers to lane l elem j the element at
//     (l&15) + j*16 + (l>>4)*64  — i.e. column (l&15) of the [4][16] block
//     fetched by its 16-lane group.
//   * row permutation when staging pixel m (within a 32-row block):
//     imgrow = 16*((m>>2)&1) + 4*(m>>3) + (m&3), so that the two tr reads of a
//     fragment (base, base+512B) deliver exactly m-slices g*8+0..3 / g*8+4..7.
//   * block tile: 64 dW-rows x 64 couts, 4 waves as 2x2 of 32x32,
//     double-buffered 64-m stages.
//   * split-M determinism: each m-chunk writes its own fp32 workspace slice
//     with plain stores; a combine kernel reduces slices in fixed order and
//     scatters into the [Cout][Cin][KH][KW] weight-grad layout.
//
// Supported: any KHxKW / stride / dilation with Cin % 16 == 0, plus arbitrary
// Cin when KH==KW==1 (50-ch Merge convs) or via the elementwise gather tail
// (7x7 Cin=3 stem). Used by ops/conv_kernels.py: conv_wgrad.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace ibp {

typedef short short4v __attribute__((ext_vector_type(4)));
typedef short short8 __attribute__((ext_vector_type(8)));
typedef float floatx4 __attribute__((ext_vector_type(4)));
typedef unsigned short ushortv8 __attribute__((ext_vector_type(8)));

// LDS geometry: 8 subtiles (4 A-cin + 4 B-cout), each [64 rows][16 ch] plus
// a 16-B pad so consecutive subtiles start on different write-bank groups.
constexpr int SUB_SHORTS = 64 * 16 + 8;   // 1032 shorts = 2064 B (8-B aligned)
constexpr int BUF_SHORTS = 8 * SUB_SHORTS;

struct WgradParams {
  const unsigned short* x;    // NHWC bf16
  const unsigned short* dy;   // N,Ho,Wo,Cout bf16
  float* ws;                  // [chunks][KD][Cout] fp32 workspace
  int N, H, W, Cin, Cout, KH, KW;
  int stride, pad_h, pad_w, dil_h, dil_w, Ho, Wo;
  long long M;                // N*Ho*Wo
  int KD;                     // KH*KW*Cin (dW rows)
  int kd_tiles, co_tiles, chunks;
  long long chunk_len;        // multiple of 64
};

__device__ __forceinline__ int wg_imgrow(int m) {
  // permuted row for pixel m (0..63) inside a [64][16] subtile image
  int pm = m & 31;
  return (m >> 5) * 32 + 16 * ((pm >> 2) & 1) + 4 * (pm >> 3) + (pm & 3);
}

__device__ __forceinline__ short4v tr16_read(const unsigned short* p) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) short4v*)p);
}

__global__ __launch_bounds__(256, 4) void conv_wgrad_kernel(WgradParams p) {
  __shared__ unsigned short lds[2][BUF_SHORTS];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // tile coords: blockIdx = ((chunk * kd_tiles) + kdt) * co_tiles + cot
  int bid = blockIdx.x;
  const int cot = bid % p.co_tiles; bid /= p.co_tiles;
  const int kdt = bid % p.kd_tiles;
  const int chunk = bid / p.kd_tiles;
  const int kd0 = kdt * 64;
  const int co0 = cot * 64;
  const long long m_begin = (long long)chunk * p.chunk_len;
  const long long m_end = min(m_begin + p.chunk_len, p.M);
  if (m_begin >= m_end) return;

  // wave sub-tile: 2x2 waves of 32x32
  const int wm = wave >> 1;       // dW-row half
  const int wn = wave & 1;        // cout half

  floatx4 acc[2][2];
  #pragma unroll
  for (int i = 0; i < 2; ++i)
    #pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

  // ---- staging geometry: thread t handles pixel m0+(t>>2), subtile q=t&3
  const int st_m = tid >> 2;
  const int st_q = tid & 3;
  const int st_row = wg_imgrow(st_m);

  // A-piece dW rows [kd0+16q, +16): tap + cin base (may cross taps only when
  // the elementwise path is active)
  const int a_k0 = kd0 + 16 * st_q;
  const int a_f = p.Cin > 0 ? a_k0 / p.Cin : 0;
  const int a_ci = a_k0 - a_f * p.Cin;
  const int a_kh = p.KW > 0 ? a_f / p.KW : 0;
  const int a_kw = a_f - a_kh * p.KW;
  // piece crosses a tap boundary (or runs past KD) -> per-element re-derive
  const bool a_elementwise = (a_ci + 16 > p.Cin) || (a_k0 + 16 > p.KD);

  unsigned short a_reg[16];
  unsigned short b_reg[16];

  auto load_stage = [&](long long m0) {
    const long long m = m0 + st_m;
    int n = 0, ho = 0, wo = 0;
    bool m_ok = m < p.M;
    if (m_ok) {
      long long t = m;
      wo = (int)(t % p.Wo); t /= p.Wo;
      ho = (int)(t % p.Ho);
      n = (int)(t / p.Ho);
    }
    // ---- A: x[n, ho*s - pad + kh*dil, wo*s - pad + kw*dil, ci0..ci0+16)
    if (!a_elementwise) {
      const int hi = ho * p.stride - p.pad_h + a_kh * p.dil_h;
      const int wi = wo * p.stride - p.pad_w + a_kw * p.dil_w;
      const bool inside = m_ok && hi >= 0 && hi < p.H && wi >= 0 && wi < p.W;
      const unsigned short* src =
          p.x + (((long long)n * p.H + hi) * p.W + wi) * p.Cin + a_ci;
      const bool aligned = ((reinterpret_cast<uintptr_t>(src)) & 15u) == 0;
      if (inside && aligned) {
        #pragma unroll
        for (int v = 0; v < 2; ++v)
          *reinterpret_cast<ushortv8*>(&a_reg[v * 8]) =
              *reinterpret_cast<const ushortv8*>(src + v * 8);
      } else if (inside) {
        #pragma unroll
        for (int e = 0; e < 16; ++e) a_reg[e] = src[e];
      } else {
        #pragma unroll
        for (int e = 0; e < 16; ++e) a_reg[e] = 0;
      }
    } else {
      #pragma unroll
      for (int e = 0; e < 16; ++e) {
        const int k = a_k0 + e;
        unsigned short v = 0;
        if (m_ok && k < p.KD) {
          const int f = k / p.Cin;
          const int ci = k - f * p.Cin;
          const int kh = f / p.KW, kw = f - kh * p.KW;
          const int hi = ho * p.stride - p.pad_h + kh * p.dil_h;
          const int wi = wo * p.stride - p.pad_w + kw * p.dil_w;
          if (hi >= 0 && hi < p.H && wi >= 0 && wi < p.W)
            v = p.x[(((long long)n * p.H + hi) * p.W + wi) * p.Cin + ci];
        }
        a_reg[e] = v;
      }
    }
    // ---- B: dy[m, co0 + 16q .. +16)
    {
      const int co = co0 + 16 * st_q;
      const unsigned short* src = p.dy + m * p.Cout + co;
      const bool full = m_ok && co + 16 <= p.Cout;
      const bool aligned = ((reinterpret_cast<uintptr_t>(src)) & 15u) == 0;
      if (full && aligned) {
        #pragma unroll
        for (int v = 0; v < 2; ++v)
          *reinterpret_cast<ushortv8*>(&b_reg[v * 8]) =
              *reinterpret_cast<const ushortv8*>(src + v * 8);
      } else {
        #pragma unroll
        for (int e = 0; e < 16; ++e)
          b_reg[e] = (m_ok && co + e < p.Cout) ? src[e] : 0;
      }
    }
  };

  auto write_stage = [&](int buf) {
    unsigned short* a_dst = &lds[buf][st_q * SUB_SHORTS + st_row * 16];
    unsigned short* b_dst = &lds[buf][(4 + st_q) * SUB_SHORTS + st_row * 16];
    #pragma unroll
    for (int v = 0; v < 2; ++v)
      *reinterpret_cast<ushortv8*>(a_dst + v * 8) =
          *reinterpret_cast<const ushortv8*>(&a_reg[v * 8]);
    #pragma unroll
    for (int v = 0; v < 2; ++v)
      *reinterpret_cast<ushortv8*>(b_dst + v * 8) =
          *reinterpret_cast<const ushortv8*>(&b_reg[v * 8]);
  };

  // fragment read: subtile s, contraction slice base ks (0/32), via 2 tr reads
  auto frag = [&](int buf, int sub, int ks) -> short8 {
    const unsigned short* base = &lds[buf][sub * SUB_SHORTS + ks * 16 + lane * 4];
    short4v lo = tr16_read(base);
    short4v hi = tr16_read(base + 256);
    short8 r;
    r[0] = lo[0]; r[1] = lo[1]; r[2] = lo[2]; r[3] = lo[3];
    r[4] = hi[0]; r[5] = hi[1]; r[6] = hi[2]; r[7] = hi[3];
    return r;
  };

  auto compute = [&](int buf) {
    #pragma unroll
    for (int ks = 0; ks < 64; ks += 32) {
      short8 af[2], bf[2];
      #pragma unroll
      for (int i = 0; i < 2; ++i) af[i] = frag(buf, wm * 2 + i, ks);
      #pragma unroll
      for (int j = 0; j < 2; ++j) bf[j] = frag(buf, 4 + wn * 2 + j, ks);
      #pragma unroll
      for (int i = 0; i < 2; ++i)
        #pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
  };

  // ---- main loop over the chunk's m range, register-staged double buffer
  const long long n_stages = (m_end - m_begin + 63) >> 6;
  load_stage(m_begin);
  write_stage(0);
  __syncthreads();
  for (long long t = 0; t < n_stages; ++t) {
    if (t + 1 < n_stages) load_stage(m_begin + (t + 1) * 64);
    compute((int)(t & 1));
    if (t + 1 < n_stages) write_stage((int)((t + 1) & 1));
    __syncthreads();
  }

  // ---- epilogue: D lane map col = lane&15, row = (lane>>4)*4 + r
  const int ecol = lane & 15;
  const int erow4 = (lane >> 4) * 4;
  float* out = p.ws + (long long)chunk * p.KD * p.Cout;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    #pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int co = co0 + wn * 32 + j * 16 + ecol;
      if (co >= p.Cout) continue;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kd = kd0 + wm * 32 + i * 16 + erow4 + r;
        if (kd >= p.KD) continue;
        out[(long long)kd * p.Cout + co] = acc[i][j][r];
      }
    }
  }
}

// combine: dW[cout][cin][kh][kw] (bf16) = sum over chunks of ws[c][f*Cin+ci][cout]
__global__ void wgrad_combine_kernel(const float* __restrict__ ws,
                                     unsigned short* __restrict__ dw,
                                     long long total, int Cin, int KHW,
                                     int Cout, int KD, int chunks) {
  for (long long e = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       e < total; e += (long long)gridDim.x * blockDim.x) {
    // e indexes [Cout][Cin][KHW]
    const int f = (int)(e % KHW);
    const long long t = e / KHW;
    const int ci = (int)(t % Cin);
    const int co = (int)(t / Cin);
    const long long src = (long long)(f * Cin + ci) * Cout + co;
    float v = 0.f;
    for (int c = 0; c < chunks; ++c)
      v += ws[(long long)c * KD * Cout + src];
    dw[e] = f2us(v);
  }
}

// ---- tr16 probe: empirically pins the ds_read_b64_tr_b16 delivery map -----
__global__ void tr16_probe_kernel(short* out) {
  __shared__ unsigned short img[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x)
    img[i] = (unsigned short)i;
  __syncthreads();
  if (threadIdx.x < 64) {
    short4v v = tr16_read(&img[threadIdx.x * 4]);
    #pragma unroll
    for (int j = 0; j < 4; ++j) out[threadIdx.x * 4 + j] = v[j];
  }
}

}  // namespace ibp

// ===========================================================================
using torch::Tensor;

Tensor conv_mfma_wgrad(const Tensor& x, const Tensor& dy, int64_t N, int64_t H,
                       int64_t W, int64_t Cin, int64_t Cout, int64_t KH,
                       int64_t KW, int64_t stride, int64_t pad_h, int64_t pad_w,
                       int64_t dil_h, int64_t dil_w, int64_t Ho, int64_t Wo) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(dy.scalar_type() == at::ScalarType::BFloat16);
  ibp::WgradParams p;
  p.x = reinterpret_cast<const unsigned short*>(x.data_ptr());
  p.dy = reinterpret_cast<const unsigned short*>(dy.data_ptr());
  p.N = (int)N; p.H = (int)H; p.W = (int)W; p.Cin = (int)Cin;
  p.Cout = (int)Cout; p.KH = (int)KH; p.KW = (int)KW;
  p.stride = (int)stride; p.pad_h = (int)pad_h; p.pad_w = (int)pad_w;
  p.dil_h = (int)dil_h; p.dil_w = (int)dil_w; p.Ho = (int)Ho; p.Wo = (int)Wo;
  p.M = (long long)N * Ho * Wo;
  p.KD = (int)(KH * KW * Cin);
  p.kd_tiles = (p.KD + 63) / 64;
  p.co_tiles = (int)((Cout + 63) / 64);
  // split M into chunks for parallelism: aim ~768 blocks, chunk length a
  // multiple of 64; every chunk slice is fully written (plain stores) before
  // the combine reduces them in fixed order -> deterministic
  const int tiles = p.kd_tiles * p.co_tiles;
  long long max_chunks = (p.M + 63) / 64;
  long long want = (768 + tiles - 1) / tiles;
  long long chunks = std::min<long long>(std::max<long long>(want, 1), max_chunks);
  p.chunk_len = (((p.M + chunks - 1) / chunks + 63) / 64) * 64;
  p.chunks = (int)((p.M + p.chunk_len - 1) / p.chunk_len);

  auto stream = at::hip::getCurrentHIPStream().stream();
  Tensor ws = torch::empty({(long long)p.chunks * p.KD * Cout},
                           x.options().dtype(torch::kFloat32));
  p.ws = ws.data_ptr<float>();
  // zero only the tiles' dead rows? every in-range (kd, cout) is written by
  // exactly one block per chunk; out-of-range rows are never read back.
  dim3 grid(p.chunks * tiles), block(256);
  hipLaunchKernelGGL(ibp::conv_wgrad_kernel, grid, block, 0, stream, p);

  Tensor dw = torch::empty({Cout, Cin, KH, KW}, x.options());
  long long total = Cout * Cin * KH * KW;
  dim3 cgrid(ibp::grid_1d(total, 256, 4096)), cblock(256);
  hipLaunchKernelGGL(ibp::wgrad_combine_kernel, cgrid, cblock, 0, stream,
                     p.ws, reinterpret_cast<unsigned short*>(dw.data_ptr()),
                     total, (int)Cin, (int)(KH * KW), (int)Cout, p.KD,
                     p.chunks);
  return dw;
}

Tensor tr16_probe() {
  Tensor out = torch::empty({64, 4}, torch::TensorOptions()
                                         .dtype(torch::kInt16)
                                         .device(torch::kCUDA));
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(ibp::tr16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     out.data_ptr<short>());
  return out;
}
