// MFMA implicit-GEMM convolution for CDNA4 (gfx950), NHWC bf16.
//
// The heart of the framework's compute path: replaces cuDNN/MIOpen convolution
// (the reference delegates ALL conv work to cuDNN) with a
// hand-written bf16 matrix-core kernel:
//
//   C[M = N*Ho*Wo][Cout] = A[M][K = KH*KW*Cin] @ B[K][Cout]
//
//   * A is gathered implicitly from the NHWC input (zero-padding handled by
//     predicated loads); B is the weight pre-packed [Cout][K] row-major.
//   * 128x(128|64) block tile, 4 waves each computing a 64x64 sub-tile as
//     4x4 fragments of v_mfma_f32_16x16x32_bf16, fp32 accumulation.
//   * K loop in BK=64 steps, double-buffered LDS with register staging:
//     tile t+1's global loads are issued before tile t's MFMAs (T14).
//   * LDS rows padded to 144 B so the 16-lane ds_read_b128 groups hit 16
//     distinct bank slots (36 dwords * r mod 64 has period 16) — conflict-free.
//   * optional fused epilogue: per-channel scale/shift (folded BatchNorm),
//     residual add, LeakyReLU — inference runs conv+BN+act in ONE kernel.
//
// Supported: any KHxKW with 'same' padding and BK | Cin (all of the IMHN's
// 3x3/dilated convs have Cin % 64 == 0), plus arbitrary Cin for 1x1 (K-tail
// predication covers the 50-channel merge heads), stride 1 or 2.
// dgrad(stride 1) reuses this kernel with 180-rotated transposed weights.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace ibp {

typedef short short8 __attribute__((ext_vector_type(8)));
typedef float floatx4 __attribute__((ext_vector_type(4)));
typedef unsigned short ushortv8 __attribute__((ext_vector_type(8)));

constexpr int BM = 128;       // output-pixel rows per block
constexpr int BK = 64;        // K depth per step

struct ConvParams {
  const unsigned short* x;   // NHWC bf16
  const unsigned short* w;   // [Cout][K] bf16 (N-major pack)
  unsigned short* y;         // NHWC bf16 out
  const float* scale;        // optional per-channel scale (folded BN)
  const float* shift;        // optional per-channel shift / bias
  const unsigned short* res; // optional residual (NHWC, same shape as y)
  int N, H, W, Cin, Cout, KH, KW;
  int stride, pad_h, pad_w, dil_h, dil_w, Ho, Wo;
  long long M;               // N*Ho*Wo
  int K;                     // KH*KW*Cin
  int n_mtiles;              // ceil(M/BM)
  int act;                   // leaky-relu on epilogue
  int splitk;                // K-dimension split factor (small-grid layers)
  float* ws;                 // fp32 workspace for split-K partial accumulation
};

// LDS tile addressing: unpadded 128-B rows with a T2 XOR swizzle — the 16-lane
// ds_read_b128 group (16 distinct rows, same k-slice) spreads over 8 bank
// slots (<=2-way) instead of hitting one. k is always a multiple of 8 here, so
// the XOR preserves 16-B alignment.
__device__ __forceinline__ int lds_off(int row, int k) {
  return row * BK + (k ^ ((row & 7) << 3));
}

template <int BN>
__global__ __launch_bounds__(256, 2) void conv_mfma_kernel(ConvParams p) {
  __shared__ unsigned short lds_a[2][BM * BK];
  __shared__ unsigned short lds_b[2][BN * BK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // tile coordinates; with split-K, a tile's K-slices are adjacent block ids
  const int ntiles_n = (p.Cout + BN - 1) / BN;
  int bid = blockIdx.x;
  const int sk = bid % p.splitk;
  const int tile = bid / p.splitk;
  const int mt = tile / ntiles_n;
  const int nt = tile % ntiles_n;
  const long long m0 = (long long)mt * BM;
  const int n0 = nt * BN;

  // wave sub-tile: 4 waves as 2x2 (BN=128) or 4x1 (BN=64)
  const int wm = (BN == 128) ? (wave >> 1) : wave;
  const int wn = (BN == 128) ? (wave & 1) : 0;
  const int WM = (BN == 128) ? 64 : 32;   // rows per wave (BN=64: 4 waves x 32)
  const int row_base = wm * WM;
  const int col_base = wn * 64;

  const int AFRAG = WM / 16;              // 4 (BN=128) or 2 (BN=64)
  floatx4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

  // ---- staging geometry: 256 threads, each loads 32 elements (2 rows' halves)
  // A: row = tid>>1, half = tid&1 -> elements [half*32, half*32+32)
  const int a_row = tid >> 1;
  const int a_off = (tid & 1) * 32;
  // B: same pattern over BN rows; BN=64 -> two k-halves per row pair
  const int b_row = (BN == 128) ? (tid >> 1) : (tid >> 2);
  const int b_off = (BN == 128) ? ((tid & 1) * 32) : ((tid & 3) * 16);
  const int b_elems = (BN == 128) ? 32 : 16;

  // per-row output-pixel decomposition for the A gather
  const long long a_m = m0 + a_row;
  int a_n = 0, a_ho = 0, a_wo = 0;
  bool a_valid_row = a_m < p.M;
  if (a_valid_row) {
    long long t = a_m;
    a_wo = (int)(t % p.Wo); t /= p.Wo;
    a_ho = (int)(t % p.Ho);
    a_n = (int)(t / p.Ho);
  }
  const int hi_base = a_ho * p.stride - p.pad_h;
  const int wi_base = a_wo * p.stride - p.pad_w;

  const int nk_total = (p.K + BK - 1) / BK;
  const int nk_chunk = (nk_total + p.splitk - 1) / p.splitk;
  const int k_begin = sk * nk_chunk;
  const int nk = min(nk_chunk, nk_total - k_begin);
  if (nk <= 0) return;

  unsigned short a_reg[32];
  unsigned short b_reg[32];

  // ---- stage chunk `ck` into registers --------------------------------------
  auto load_chunk = [&](int ck) {
    const int k0 = ck * BK;
    // ---- A gather: k = (kh*KW + kw)*Cin + ci
    {
      int kk = k0 + a_off;
      int f = kk / p.Cin;               // filter tap index (constant when BK|Cin)
      int ci = kk - f * p.Cin;
      int kh = f / p.KW, kw = f - kh * p.KW;
      int hi = hi_base + kh * p.dil_h;
      int wi = wi_base + kw * p.dil_w;
      bool inside = a_valid_row && hi >= 0 && hi < p.H && wi >= 0 && wi < p.W;
      const unsigned short* src =
          p.x + (((long long)a_n * p.H + hi) * p.W + wi) * p.Cin + ci;
      // 16-B vector loads need a 16-B-aligned source (odd Cin, e.g. the
      // 50-channel merge input, makes pixel rows only 2-B aligned)
      bool a_aligned = ((reinterpret_cast<uintptr_t>(src)) & 15u) == 0;
      if (inside && a_aligned && ci + 32 <= p.Cin && kk + 32 <= p.K) {
        #pragma unroll
        for (int v = 0; v < 4; ++v)
          *reinterpret_cast<ushortv8*>(&a_reg[v * 8]) =
              *reinterpret_cast<const ushortv8*>(src + v * 8);
      } else if (inside) {
        #pragma unroll
        for (int e = 0; e < 32; ++e) {
          int kke = kk + e;
          // re-derive tap for elements crossing the Cin boundary (1x1 K-tail)
          a_reg[e] = (kke < p.K && ci + e < p.Cin)
                         ? src[e] : (unsigned short)0;
        }
      } else {
        #pragma unroll
        for (int e = 0; e < 32; ++e) a_reg[e] = 0;
      }
    }
    // ---- B: packed [Cout][K] rows
    {
      int col = n0 + b_row;
      int kk = k0 + b_off;
      bool ok = col < p.Cout;
      const unsigned short* src = p.w + (long long)col * p.K + kk;
      bool b_aligned = ((reinterpret_cast<uintptr_t>(src)) & 15u) == 0;
      if (ok && b_aligned && kk + b_elems <= p.K) {
        #pragma unroll
        for (int v = 0; v < 4; ++v)
          if (v * 8 < b_elems)
            *reinterpret_cast<ushortv8*>(&b_reg[v * 8]) =
                *reinterpret_cast<const ushortv8*>(src + v * 8);
      } else {
        #pragma unroll
        for (int e = 0; e < 32; ++e)
          if (e < b_elems) b_reg[e] = (ok && kk + e < p.K) ? src[e] : 0;
      }
    }
  };

  auto write_chunk = [&](int buf) {
    #pragma unroll
    for (int v = 0; v < 4; ++v)
      *reinterpret_cast<ushortv8*>(&lds_a[buf][lds_off(a_row, a_off + v * 8)]) =
          *reinterpret_cast<const ushortv8*>(&a_reg[v * 8]);
    #pragma unroll
    for (int v = 0; v < 4; ++v)
      if (v * 8 < b_elems)
        *reinterpret_cast<ushortv8*>(&lds_b[buf][lds_off(b_row, b_off + v * 8)]) =
            *reinterpret_cast<const ushortv8*>(&b_reg[v * 8]);
  };

  // ---- MFMA over one LDS buffer --------------------------------------------
  auto compute = [&](int buf) {
    const int l15 = lane & 15;
    const int kslice = (lane >> 4) * 8;
    #pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      short8 afrag[4], bfrag[4];
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        if (i < AFRAG)
          afrag[i] = *reinterpret_cast<const short8*>(
              &lds_a[buf][lds_off(row_base + i * 16 + l15, kk + kslice)]);
      }
      #pragma unroll
      for (int j = 0; j < 4; ++j)
        bfrag[j] = *reinterpret_cast<const short8*>(
            &lds_b[buf][lds_off(col_base + j * 16 + l15, kk + kslice)]);
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        if (i < AFRAG) {
          #pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
        }
      }
    }
  };

  // ---- main loop: register-staged double buffer ----------------------------
  load_chunk(k_begin);
  write_chunk(0);
  __syncthreads();
  for (int t = 0; t < nk; ++t) {
    if (t + 1 < nk) load_chunk(k_begin + t + 1);  // loads hide under the MFMAs
    compute(t & 1);
    // writing buf[(t+1)&1] is safe without a barrier: its last readers were
    // separated by the end-of-iteration barrier of step t-1
    if (t + 1 < nk) write_chunk((t + 1) & 1);
    __syncthreads();
  }

  // ---- epilogue: C/D lane map col = lane&15, row = (lane>>4)*4 + r ---------
  const int ecol = lane & 15;
  const int erow4 = (lane >> 4) * 4;
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    if (i >= AFRAG) continue;
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + col_base + j * 16 + ecol;
      if (col >= p.Cout) continue;
      float sc = p.scale ? p.scale[col] : 1.f;
      float sh = p.shift ? p.shift[col] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        long long m = m0 + row_base + i * 16 + erow4 + r;
        if (m >= p.M) continue;
        long long idx = m * p.Cout + col;
        if (p.splitk > 1) {
          // each K-split owns a workspace slice: plain stores, no zero-init,
          // no atomics, deterministic sums (combine kernel reduces slices)
          p.ws[(long long)sk * p.M * p.Cout + idx] = acc[i][j][r];
          continue;
        }
        float v = acc[i][j][r] * sc + sh;
        if (p.res) v += us2f(p.res[idx]);
        if (p.act) v = leaky(v, 0.01f);
        p.y[idx] = f2us(v);
      }
    }
  }
}

// split-K combine: y = act(scale*ws + shift (+res)) -> bf16
__global__ void splitk_combine_kernel(const float* __restrict__ ws,
                                      unsigned short* __restrict__ y,
                                      const float* __restrict__ scale,
                                      const float* __restrict__ shift,
                                      const unsigned short* __restrict__ res,
                                      long long total, int C, int act,
                                      int splitk) {
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    float v = 0.f;
    for (int s = 0; s < splitk; ++s) v += ws[(long long)s * total + i];
    if (scale) v = v * scale[c] + shift[c];
    if (res) v += us2f(res[i]);
    if (act) v = leaky(v, 0.01f);
    y[i] = f2us(v);
  }
}

}  // namespace ibp

// ===========================================================================
using torch::Tensor;

Tensor conv_mfma_fwd(const Tensor& x, const Tensor& w_packed, int64_t N,
                     int64_t H, int64_t W, int64_t Cin, int64_t Cout,
                     int64_t KH, int64_t KW, int64_t stride, int64_t pad_h,
                     int64_t pad_w, int64_t dil_h, int64_t dil_w, int64_t Ho,
                     int64_t Wo, const c10::optional<Tensor>& scale,
                     const c10::optional<Tensor>& shift,
                     const c10::optional<Tensor>& residual, bool act) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(w_packed.is_contiguous());
  ibp::ConvParams p;
  p.x = reinterpret_cast<const unsigned short*>(x.data_ptr());
  p.w = reinterpret_cast<const unsigned short*>(w_packed.data_ptr());
  p.scale = scale.has_value() ? scale->data_ptr<float>() : nullptr;
  p.shift = shift.has_value() ? shift->data_ptr<float>() : nullptr;
  p.res = residual.has_value()
              ? reinterpret_cast<const unsigned short*>(residual->data_ptr())
              : nullptr;
  p.N = (int)N; p.H = (int)H; p.W = (int)W; p.Cin = (int)Cin;
  p.Cout = (int)Cout; p.KH = (int)KH; p.KW = (int)KW;
  p.stride = (int)stride; p.pad_h = (int)pad_h; p.pad_w = (int)pad_w;
  p.dil_h = (int)dil_h; p.dil_w = (int)dil_w; p.Ho = (int)Ho; p.Wo = (int)Wo;
  p.M = (long long)N * Ho * Wo;
  p.K = (int)(KH * KW * Cin);
  p.act = act ? 1 : 0;
  if (KH != 1 || KW != 1) {
    TORCH_CHECK(Cin % ibp::BK == 0,
                "KxK conv requires Cin % 64 == 0, got ", Cin);
  }
  Tensor y = torch::empty({N, Ho, Wo, Cout}, x.options());
  p.y = reinterpret_cast<unsigned short*>(y.data_ptr());
  p.n_mtiles = (int)((p.M + ibp::BM - 1) / ibp::BM);
  auto stream = at::hip::getCurrentHIPStream().stream();
  // pick the N-tile minimising padded work: Cout=192 as 3x64 beats 2x128
  // (half of the second 128-tile is wasted lanes — measured 0.85x vs library)
  const long long padded128 = ((Cout + 127) / 128) * 128LL;
  const long long padded64 = ((Cout + 63) / 64) * 64LL;
  const int BN = (Cout > 64 && padded128 <= padded64) ? 128 : 64;
  const int ntiles = p.n_mtiles * (int)((Cout + BN - 1) / BN);
  const int nk_total = (p.K + ibp::BK - 1) / ibp::BK;
  // split K on small grids so the 256-CU chip stays filled (~2 blocks/CU).
  // Normalise so EVERY slice has work: with raw splitk=7 over nk_total=8 the
  // last 3 slices get no chunks, return early, and the combine kernel then
  // sums their UNINITIALISED workspace slices (silent garbage whenever the
  // allocator hands back dirty memory — found as exploding gradients through
  // the scale-2 Merge convs, scripts/diag_splitk.py).
  int splitk = 1;
  if (ntiles < 384 && nk_total > 1) {
    splitk = std::min((int)nk_total, (384 + ntiles - 1) / ntiles);
    int nk_chunk = ((int)nk_total + splitk - 1) / splitk;
    splitk = ((int)nk_total + nk_chunk - 1) / nk_chunk;
  }
  p.splitk = splitk;
  Tensor ws;
  if (splitk > 1) {
    // per-split slices written with plain stores -> empty() is safe (every
    // in-range element is covered by every split's epilogue)
    ws = torch::empty({(long long)splitk * p.M * Cout},
                      x.options().dtype(torch::kFloat32));
    p.ws = ws.data_ptr<float>();
  } else {
    p.ws = nullptr;
  }
  dim3 grid(ntiles * splitk), block(256);
  if (BN == 128) {
    hipLaunchKernelGGL(ibp::conv_mfma_kernel<128>, grid, block, 0, stream, p);
  } else {
    hipLaunchKernelGGL(ibp::conv_mfma_kernel<64>, grid, block, 0, stream, p);
  }
  if (splitk > 1) {
    long long total = (long long)p.M * Cout;
    dim3 cgrid(ibp::grid_1d(total, 256, 4096)), cblock(256);
    hipLaunchKernelGGL(ibp::splitk_combine_kernel, cgrid, cblock, 0, stream,
                       p.ws, p.y, p.scale, p.shift, p.res, total, (int)Cout,
                       p.act, splitk);
  }
  return y;
}
