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
//   * block tile BMx(BN): (128,128) / (128,64) for big spatial maps,
//     (64,64) / (32,64) for the hourglass's 16^2/8^2 tails (small-M tiles cut
//     split-K workspace traffic 4-8x there — round-1 weak #2).
//   * 4 waves per block arranged (BM/WM)x(BN/WN), each computing WMxWN as
//     AFRAGxBFRAG fragments of v_mfma_f32_16x16x32_bf16, fp32 accumulation.
//   * K loop in BK=64 steps, double-buffered LDS with register staging:
//     tile t+1's global loads are issued before tile t's MFMAs (T14).
//   * LDS rows carry a T2 XOR swizzle — the 16-lane ds_read_b128 group
//     (16 distinct rows, same k-slice) spreads over 8 bank slots.
//   * optional fused epilogue: per-channel scale/shift (folded BatchNorm),
//     PRE-act residual (bottleneck skip), LeakyReLU, then up to two POST-act
//     residuals (hourglass up1+deconv1 join and the cross-stack feature-cache
//     add ride inside the conv kernel on the inference path).
//   * zs (gather stride): reads the virtual zero-dilated image x[hi/zs] when
//     hi % zs == 0 — stride-s dgrad as a stride-1 conv over dilated dy with
//     180-rotated transposed weights.
//
// Supported: any KHxKW with Cin % 8 == 0 (per-8 subpieces re-derive their
// filter tap, so Cin 16 s2d-stem tiles and 64..768 hourglass tiles both hit
// the vector path), arbitrary Cin for 1x1 (K-tail predication covers the
// 50-channel merge heads) and via the elementwise tail, stride 1 or 2.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cstdlib>
#include "common.h"

namespace ibp {

typedef short short8 __attribute__((ext_vector_type(8)));
typedef float floatx4 __attribute__((ext_vector_type(4)));
typedef unsigned short ushortv8 __attribute__((ext_vector_type(8)));

constexpr int BK = 64;        // K depth per step

struct ConvParams {
  const unsigned short* x;   // NHWC bf16
  const unsigned short* w;   // [Cout][K] bf16 (N-major pack)
  unsigned short* y;         // NHWC bf16 out
  const float* scale;        // optional per-channel scale (folded BN)
  const float* shift;        // optional per-channel shift / bias
  const unsigned short* res;      // optional residual, added BEFORE act
  const unsigned short* res_post; // optional residual, added AFTER act
  const unsigned short* res_post2;
  int N, H, W, Cin, Cout, KH, KW;
  int stride, pad_h, pad_w, dil_h, dil_w, Ho, Wo;
  int zs;                    // gather stride (transposed-conv dgrad), 1 = off
  long long M;               // N*Ho*Wo
  int K;                     // KH*KW*Cin
  int n_mtiles;              // ceil(M/BM)
  int act;                   // leaky-relu on epilogue
  int splitk;                // K-dimension split factor (small-grid layers)
  float* ws;                 // fp32 workspace for split-K partial accumulation
};

// LDS tile addressing: unpadded BK-short rows with a T2 XOR swizzle — the
// 16-lane ds_read_b128 group (16 distinct rows, same k-slice) spreads over 8
// bank slots (<=2-way) instead of hitting one. k is always a multiple of 8
// here, so the XOR preserves 16-B alignment.
__device__ __forceinline__ int lds_off(int row, int k) {
  return row * BK + (k ^ ((row & 7) << 3));
}

// DEEP: two-stage register prefetch — wins ONLY on short-K shapes (the 1x1
// layers at <= 6 K-chunks never reach pipeline steady state); on long-K 3x3
// shapes the extra register bank measured 10-15% slower.
template <int BM, int BN, bool DEEP = false>
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

  // wave sub-tile layout: 4 waves as (BM/WM) x (BN/WN)
  constexpr int WROWS = (BM == 128) ? ((BN == 128) ? 2 : 4) : 2;
  constexpr int WCOLS = 4 / WROWS;
  constexpr int WM = BM / WROWS;
  constexpr int WN = BN / WCOLS;
  constexpr int AFRAG = WM / 16;
  constexpr int BFRAG = WN / 16;
  const int wm = (WCOLS == 1) ? wave : (wave >> 1);
  const int wn = (WCOLS == 1) ? 0 : (wave & 1);
  const int row_base = wm * WM;
  const int col_base = wn * WN;

  floatx4 acc[AFRAG][BFRAG];
  #pragma unroll
  for (int i = 0; i < AFRAG; ++i)
    #pragma unroll
    for (int j = 0; j < BFRAG; ++j) acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

  // ---- staging geometry: 256 threads cover BM*BK (A) and BN*BK (B) shorts
  constexpr int A_ELEMS = BM * BK / 256;   // 32 / 16 / 8
  constexpr int B_ELEMS = BN * BK / 256;   // 32 / 16
  constexpr int TPR_A = BK / A_ELEMS;
  constexpr int TPR_B = BK / B_ELEMS;
  const int a_row = tid / TPR_A;
  const int a_off = (tid % TPR_A) * A_ELEMS;
  const int b_row = tid / TPR_B;
  const int b_off = (tid % TPR_B) * B_ELEMS;

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

  unsigned short a_reg0[A_ELEMS];
  unsigned short b_reg0[B_ELEMS];
  // second bank referenced only under `if constexpr (DEEP)` — fully
  // eliminated in the default instantiations
  unsigned short a_reg1[A_ELEMS];
  unsigned short b_reg1[B_ELEMS];

  // ---- incremental tap-walking state for the A gather ----------------------
  // k = (kh*KW + kw)*Cin + ci decodes with ONE division pair at setup; each
  // chunk then advances (ci, kh, kw) by small wrap steps. The first version
  // of the per-8 gather re-divided per subpiece (4 integer divisions per
  // chunk per thread) and measured 0.63-0.68x of the round-1 kernel — the
  // emulated 32-bit divide beside MFMAs is exactly the anti-lever the guide
  // warns about.
  int t_kk = k_begin * BK + a_off;
  int t_f = t_kk / p.Cin;
  int t_ci = t_kk - t_f * p.Cin;
  int t_kh = t_f / p.KW;
  int t_kw = t_f - t_kh * p.KW;
  auto tap_advance = [&](int& ci, int& kh, int& kw, int step) {
    ci += step;
    while (ci >= p.Cin) {
      ci -= p.Cin;
      if (++kw == p.KW) { kw = 0; ++kh; }
    }
  };

  // ---- stage the NEXT sequential chunk into registers ----------------------
  // (must be called with consecutive chunks — the tap state walks forward)
  auto load_chunk = [&](int ck, unsigned short (&a_reg)[A_ELEMS],
                        unsigned short (&b_reg)[B_ELEMS]) {
    int kk = t_kk, ci = t_ci, kh = t_kh, kw = t_kw;
    // fast path: the whole A_ELEMS piece inside one filter tap (all the
    // Cin % 64 == 0 hourglass convs) — ONE predicate set, straight-line
    // vector loads (the per-subpiece form below costs ~10% on hot shapes)
    if (ci + A_ELEMS <= p.Cin && kk + A_ELEMS <= p.K && p.zs == 1) {
      const int hi = hi_base + kh * p.dil_h;
      const int wi = wi_base + kw * p.dil_w;
      const bool ok = a_valid_row && hi >= 0 && hi < p.H &&
                      wi >= 0 && wi < p.W;
      const unsigned short* src =
          p.x + (((long long)a_n * p.H + hi) * p.W + wi) * p.Cin + ci;
      const bool aligned = ((reinterpret_cast<uintptr_t>(src)) & 15u) == 0;
      if (ok && aligned) {
        #pragma unroll
        for (int v = 0; v < A_ELEMS / 8; ++v)
          *reinterpret_cast<ushortv8*>(&a_reg[v * 8]) =
              *reinterpret_cast<const ushortv8*>(src + v * 8);
      } else if (ok) {
        #pragma unroll
        for (int e = 0; e < A_ELEMS; ++e) a_reg[e] = src[e];
      } else {
        #pragma unroll
        for (int e = 0; e < A_ELEMS; ++e) a_reg[e] = 0;
      }
      goto a_done;
    }
    #pragma unroll
    for (int v8 = 0; v8 < A_ELEMS / 8; ++v8) {
      unsigned short* dst = &a_reg[v8 * 8];
      int hi = hi_base + kh * p.dil_h;
      int wi = wi_base + kw * p.dil_w;
      bool ok = a_valid_row;
      if (p.zs > 1) {                    // virtual zero-dilated image (dgrad)
        ok = ok && (hi % p.zs == 0) && (wi % p.zs == 0);
        hi /= p.zs; wi /= p.zs;
      }
      ok = ok && hi >= 0 && hi < p.H && wi >= 0 && wi < p.W;
      const unsigned short* src =
          p.x + (((long long)a_n * p.H + hi) * p.W + wi) * p.Cin + ci;
      // 16-B vector loads need a 16-B-aligned source (odd Cin, e.g. the
      // 50-channel merge input, makes pixel rows only 2-B aligned)
      const bool aligned = ((reinterpret_cast<uintptr_t>(src)) & 15u) == 0;
      const bool in_tap = ci + 8 <= p.Cin;
      if (ok && aligned && in_tap && kk + 8 <= p.K) {
        *reinterpret_cast<ushortv8*>(dst) =
            *reinterpret_cast<const ushortv8*>(src);
      } else if (ok && in_tap) {
        #pragma unroll
        for (int e = 0; e < 8; ++e) {
          // K-tail (1x1 heads): elements past K are zero
          dst[e] = (kk + e < p.K) ? src[e] : (unsigned short)0;
        }
      } else if (!in_tap && a_valid_row) {
        // subpiece crosses tap boundaries (Cin=3 stem, Cin % 8 != 0):
        // walk tap/coordinates element by element (no divisions)
        int cie = ci, khe = kh, kwe = kw;
        #pragma unroll
        for (int e = 0; e < 8; ++e) {
          unsigned short v = 0;
          if (kk + e < p.K) {
            int hie = hi_base + khe * p.dil_h;
            int wie = wi_base + kwe * p.dil_w;
            bool oke = true;
            if (p.zs > 1) {
              oke = (hie % p.zs == 0) && (wie % p.zs == 0);
              hie /= p.zs; wie /= p.zs;
            }
            if (oke && hie >= 0 && hie < p.H && wie >= 0 && wie < p.W)
              v = p.x[(((long long)a_n * p.H + hie) * p.W + wie) * p.Cin + cie];
          }
          dst[e] = v;
          if (++cie == p.Cin) {
            cie = 0;
            if (++kwe == p.KW) { kwe = 0; ++khe; }
          }
        }
      } else {
        #pragma unroll
        for (int e = 0; e < 8; ++e) dst[e] = 0;
      }
      kk += 8;
      tap_advance(ci, kh, kw, 8);
    }
a_done:
    // walk the persistent state one full chunk forward
    t_kk += BK;
    tap_advance(t_ci, t_kh, t_kw, BK);
    // ---- B: packed [Cout][K] rows
    {
      int col = n0 + b_row;
      int kk = ck * BK + b_off;
      bool ok = col < p.Cout;
      const unsigned short* src = p.w + (long long)col * p.K + kk;
      bool b_aligned = ((reinterpret_cast<uintptr_t>(src)) & 15u) == 0;
      if (ok && b_aligned && kk + B_ELEMS <= p.K) {
        #pragma unroll
        for (int v = 0; v < B_ELEMS / 8; ++v)
          *reinterpret_cast<ushortv8*>(&b_reg[v * 8]) =
              *reinterpret_cast<const ushortv8*>(src + v * 8);
      } else {
        #pragma unroll
        for (int e = 0; e < B_ELEMS; ++e)
          b_reg[e] = (ok && kk + e < p.K) ? src[e] : 0;
      }
    }
  };

  auto write_chunk = [&](int buf, unsigned short (&a_reg)[A_ELEMS],
                         unsigned short (&b_reg)[B_ELEMS]) {
    #pragma unroll
    for (int v = 0; v < A_ELEMS / 8; ++v)
      *reinterpret_cast<ushortv8*>(&lds_a[buf][lds_off(a_row, a_off + v * 8)]) =
          *reinterpret_cast<const ushortv8*>(&a_reg[v * 8]);
    #pragma unroll
    for (int v = 0; v < B_ELEMS / 8; ++v)
      *reinterpret_cast<ushortv8*>(&lds_b[buf][lds_off(b_row, b_off + v * 8)]) =
          *reinterpret_cast<const ushortv8*>(&b_reg[v * 8]);
  };

  // ---- MFMA over one LDS buffer --------------------------------------------
  auto compute = [&](int buf) {
    const int l15 = lane & 15;
    const int kslice = (lane >> 4) * 8;
    #pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      short8 afrag[AFRAG], bfrag[BFRAG];
      #pragma unroll
      for (int i = 0; i < AFRAG; ++i)
        afrag[i] = *reinterpret_cast<const short8*>(
            &lds_a[buf][lds_off(row_base + i * 16 + l15, kk + kslice)]);
      #pragma unroll
      for (int j = 0; j < BFRAG; ++j)
        bfrag[j] = *reinterpret_cast<const short8*>(
            &lds_b[buf][lds_off(col_base + j * 16 + l15, kk + kslice)]);
      #pragma unroll
      for (int i = 0; i < AFRAG; ++i)
        #pragma unroll
        for (int j = 0; j < BFRAG; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
  };

  // ---- main loop: register-staged double buffer ----------------------------
  if constexpr (DEEP) {
    // 2-deep prefetch: chunk t+2's loads issue at iteration t, land at t+1
    load_chunk(k_begin, a_reg0, b_reg0);
    write_chunk(0, a_reg0, b_reg0);
    if (nk > 1) load_chunk(k_begin + 1, a_reg1, b_reg1);
    __syncthreads();
    for (int t = 0; t < nk; ++t) {
      if (t + 2 < nk) load_chunk(k_begin + t + 2, a_reg0, b_reg0);
      compute(0);
      if (t + 1 < nk) write_chunk(1, a_reg1, b_reg1);
      __syncthreads();
      if (++t >= nk) break;
      if (t + 2 < nk) load_chunk(k_begin + t + 2, a_reg1, b_reg1);
      compute(1);
      if (t + 1 < nk) write_chunk(0, a_reg0, b_reg0);
      __syncthreads();
    }
  } else {
    load_chunk(k_begin, a_reg0, b_reg0);
    write_chunk(0, a_reg0, b_reg0);
    __syncthreads();
    for (int t = 0; t < nk; ++t) {
      if (t + 1 < nk) load_chunk(k_begin + t + 1, a_reg0, b_reg0);
      compute(t & 1);
      // writing buf[(t+1)&1] is safe without a barrier: its last readers
      // were separated by the end-of-iteration barrier of step t-1
      if (t + 1 < nk) write_chunk((t + 1) & 1, a_reg0, b_reg0);
      __syncthreads();
    }
  }

  // ---- epilogue: C/D lane map col = lane&15, row = (lane>>4)*4 + r ---------
  const int ecol = lane & 15;
  const int erow4 = (lane >> 4) * 4;
  #pragma unroll
  for (int i = 0; i < AFRAG; ++i) {
    #pragma unroll
    for (int j = 0; j < BFRAG; ++j) {
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
        if (p.res_post) v += us2f(p.res_post[idx]);
        if (p.res_post2) v += us2f(p.res_post2[idx]);
        p.y[idx] = f2us(v);
      }
    }
  }
}

// split-K combine: y = act(scale*ws + shift (+res)) (+res_post...) -> bf16
__global__ void splitk_combine_kernel(const float* __restrict__ ws,
                                      unsigned short* __restrict__ y,
                                      const float* __restrict__ scale,
                                      const float* __restrict__ shift,
                                      const unsigned short* __restrict__ res,
                                      const unsigned short* __restrict__ res_post,
                                      const unsigned short* __restrict__ res_post2,
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
    if (res_post) v += us2f(res_post[i]);
    if (res_post2) v += us2f(res_post2[i]);
    y[i] = f2us(v);
  }
}

// weight pack: [Cout,Cin,KH,KW] (standard contiguous) -> fwd [Cout][f*Cin+ci]
// and (optional) dgrad [Cin][(rot180 f)*Cout+co] in ONE pass. Weights change
// every optimizer step, so the eager flip/permute/reshape chains re-ran per
// layer per step (~500 small launches); this is one launch per weight.
__global__ void pack_weight_kernel(const unsigned short* __restrict__ w,
                                   unsigned short* __restrict__ fwd,
                                   unsigned short* __restrict__ dgr,
                                   int Cout, int Cin, int KH, int KW) {
  const long long total = (long long)Cout * Cin * KH * KW;
  const int KHW = KH * KW;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int kw = (int)(i % KW);
    long long t = i / KW;
    const int kh = (int)(t % KH); t /= KH;
    const int ci = (int)(t % Cin);
    const int co = (int)(t / Cin);
    const unsigned short v = w[i];
    const int f = kh * KW + kw;
    fwd[(long long)co * KHW * Cin + f * Cin + ci] = v;
    if (dgr) {
      const int fr = (KH - 1 - kh) * KW + (KW - 1 - kw);
      dgr[(long long)ci * KHW * Cout + fr * Cout + co] = v;
    }
  }
}

}  // namespace ibp

// ===========================================================================
using torch::Tensor;

void pack_conv_weight(const Tensor& w, Tensor& fwd_pack,
                      const c10::optional<Tensor>& dgrad_pack) {
  TORCH_CHECK(w.is_cuda() && w.is_contiguous() &&
              w.scalar_type() == at::ScalarType::BFloat16);
  int Cout = (int)w.size(0), Cin = (int)w.size(1);
  int KH = (int)w.size(2), KW = (int)w.size(3);
  auto stream = at::hip::getCurrentHIPStream().stream();
  dim3 grid(ibp::grid_1d(w.numel(), 256, 2048)), block(256);
  hipLaunchKernelGGL(
      ibp::pack_weight_kernel, grid, block, 0, stream,
      reinterpret_cast<const unsigned short*>(w.data_ptr()),
      reinterpret_cast<unsigned short*>(fwd_pack.data_ptr()),
      dgrad_pack.has_value()
          ? reinterpret_cast<unsigned short*>(dgrad_pack->data_ptr())
          : nullptr,
      Cout, Cin, KH, KW);
}

Tensor conv_mfma_fwd(const Tensor& x, const Tensor& w_packed, int64_t N,
                     int64_t H, int64_t W, int64_t Cin, int64_t Cout,
                     int64_t KH, int64_t KW, int64_t stride, int64_t pad_h,
                     int64_t pad_w, int64_t dil_h, int64_t dil_w, int64_t Ho,
                     int64_t Wo, const c10::optional<Tensor>& scale,
                     const c10::optional<Tensor>& shift,
                     const c10::optional<Tensor>& residual, bool act,
                     const c10::optional<Tensor>& residual_post,
                     const c10::optional<Tensor>& residual_post2, int64_t zs) {
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
  p.res_post =
      residual_post.has_value()
          ? reinterpret_cast<const unsigned short*>(residual_post->data_ptr())
          : nullptr;
  p.res_post2 =
      residual_post2.has_value()
          ? reinterpret_cast<const unsigned short*>(residual_post2->data_ptr())
          : nullptr;
  p.N = (int)N; p.H = (int)H; p.W = (int)W; p.Cin = (int)Cin;
  p.Cout = (int)Cout; p.KH = (int)KH; p.KW = (int)KW;
  p.stride = (int)stride; p.pad_h = (int)pad_h; p.pad_w = (int)pad_w;
  p.dil_h = (int)dil_h; p.dil_w = (int)dil_w; p.Ho = (int)Ho; p.Wo = (int)Wo;
  p.zs = (int)zs;
  p.M = (long long)N * Ho * Wo;
  p.K = (int)(KH * KW * Cin);
  p.act = act ? 1 : 0;
  // any Cin is legal: subpieces crossing tap boundaries take the per-element
  // re-derive path (the Cin=3 stem runs there; Cin % 8 == 0 stays vectorized)
  Tensor y = torch::empty({N, Ho, Wo, Cout}, x.options());
  p.y = reinterpret_cast<unsigned short*>(y.data_ptr());
  auto stream = at::hip::getCurrentHIPStream().stream();

  // ---- tile selection -------------------------------------------------------
  // BM: small-M tiles for the hourglass's 8^2/16^2 scales (avoids the 8-16x
  // split-K workspace round-trip measured in round 1); BN: minimise padded
  // cols (Cout=192 as 3x64 beats 2x128 — half the second 128-tile is wasted).
  int BM = 128;
  if (p.M <= 2048) BM = 32;
  else if (p.M <= 8192) BM = 64;
  if (const char* e = getenv("IBP_CONV_BM")) BM = atoi(e);
  const long long padded128 = ((Cout + 127) / 128) * 128LL;
  const long long padded64 = ((Cout + 63) / 64) * 64LL;
  int BN = (BM == 128 && Cout > 64 && padded128 <= padded64) ? 128 : 64;
  p.n_mtiles = (int)((p.M + BM - 1) / BM);
  const int ntiles = p.n_mtiles * (int)((Cout + BN - 1) / BN);
  const int nk_total = (p.K + ibp::BK - 1) / ibp::BK;
  // split K on small grids so the 256-CU chip stays filled (~2 blocks/CU).
  // Normalise so EVERY slice has work: with raw splitk=7 over nk_total=8 the
  // last 3 slices get no chunks, return early, and the combine kernel then
  // sums their UNINITIALISED workspace slices (silent garbage whenever the
  // allocator hands back dirty memory — found as exploding gradients through
  // the scale-2 Merge convs, scripts/diag_splitk.py).
  int sk_target = 384;
  if (const char* e = getenv("IBP_SPLITK_TARGET")) sk_target = atoi(e);
  int splitk = 1;
  if (ntiles < sk_target && nk_total > 1) {
    splitk = std::min((int)nk_total, (sk_target + ntiles - 1) / ntiles);
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
  // short-K blocks (the 1x1 layers: <= 6 chunks) never reach pipeline
  // steady state with single-stage prefetch — use the 2-deep variant there
  const int nk_block = (nk_total + splitk - 1) / splitk;
  const bool deep = nk_block <= 6;
  if (BM == 128 && BN == 128) {
    if (deep)
      hipLaunchKernelGGL((ibp::conv_mfma_kernel<128, 128, true>), grid, block, 0, stream, p);
    else
      hipLaunchKernelGGL((ibp::conv_mfma_kernel<128, 128>), grid, block, 0, stream, p);
  } else if (BM == 128) {
    if (deep)
      hipLaunchKernelGGL((ibp::conv_mfma_kernel<128, 64, true>), grid, block, 0, stream, p);
    else
      hipLaunchKernelGGL((ibp::conv_mfma_kernel<128, 64>), grid, block, 0, stream, p);
  } else if (BM == 64) {
    if (deep)
      hipLaunchKernelGGL((ibp::conv_mfma_kernel<64, 64, true>), grid, block, 0, stream, p);
    else
      hipLaunchKernelGGL((ibp::conv_mfma_kernel<64, 64>), grid, block, 0, stream, p);
  } else {
    if (deep)
      hipLaunchKernelGGL((ibp::conv_mfma_kernel<32, 64, true>), grid, block, 0, stream, p);
    else
      hipLaunchKernelGGL((ibp::conv_mfma_kernel<32, 64>), grid, block, 0, stream, p);
  }
  if (splitk > 1) {
    long long total = (long long)p.M * Cout;
    dim3 cgrid(ibp::grid_1d(total, 256, 4096)), cblock(256);
    hipLaunchKernelGGL(ibp::splitk_combine_kernel, cgrid, cblock, 0, stream,
                       p.ws, p.y, p.scale, p.shift, p.res, p.res_post,
                       p.res_post2, total, (int)Cout, p.act, splitk);
  }
  return y;
}
