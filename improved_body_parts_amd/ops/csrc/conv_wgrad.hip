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
//   * block tile TKDxTCO x NWAVES: (64,64)x4 for heads/merges/small layers
//     (occ 4), (64,128)x4 for wide 1x1s, and the (128,128)x8 "mid" tile for
//     the LDS-read-bound wide-spatial KxK layers (same waves/SIMD as 64x64,
//     25% fewer tr-read bytes per FLOP); double-buffered 64-m stages.
//   * the per-stage pixel decomposition m -> (n, ho, wo) advances by +64 with
//     carry steps (no 64-bit divisions in the loop).
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

// LDS geometry: (TKD+TCO)/16 subtiles, each [64 rows][16 ch] plus a 16-B pad
// so consecutive subtiles start on different write-bank groups. The stride
// MUST stay 16-B aligned: an 8-B pad (tried for a conflict-free write
// stagger) misaligns every b128 staging write on odd subtiles — measured
// 0.31-0.46x, far worse than the 2-way conflicts it removed. A fully
// conflict-free stagger is impossible here: {q*s} mod 32 is arithmetic, the
// needed residue set is not.
constexpr int SUB_SHORTS = 64 * 16 + 8;   // 1032 shorts = 2064 B (16-B aligned)

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
  int step_dho, step_dwo;     // 64 / Wo, 64 % Wo (m-walk carry steps)
  int kd_cb;                  // Cin/TKD when Cin % TKD == 0 (tile reorder), else 0
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

// ELEM compiles in the tap-crossing per-element path (stem, Cin % 16 != 0,
// partial last kd-tile) — its precomputed pe_pack walk costs ~16 VGPRs, so
// the clean-tiled hot layers instantiate ELEM=false and keep occupancy 4.
// NWAVES=8 (512 threads) is the 128x128 "mid" tile: same waves/SIMD as the
// 64x64 tile but 25% fewer LDS tr-read bytes per FLOP (the 64-tile at occ 4
// demands ~4.5x the LDS read rate its MFMA shadow covers — wgrad's binding
// resource on the wide 3x3 layers).
template <int TKD, int TCO, bool ELEM, int NWAVES = 4>
__global__
__launch_bounds__(NWAVES * 64,
                  (TKD == 64 && TCO == 64) ? 3
                                           : (NWAVES == 8 ? 4 : 2))
    void conv_wgrad_kernel(
    WgradParams p) {
  constexpr int ASUB = TKD / 16;
  constexpr int BSUB = TCO / 16;
  constexpr int NSUB = ASUB + BSUB;
  constexpr int NPIECE = NSUB / NWAVES;   // 16-ch pieces per thread per stage
  constexpr int WCOLS = NWAVES / 2;       // waves as 2 x WCOLS
  __shared__ unsigned short lds[2][NSUB * SUB_SHORTS];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // tile coords: blockIdx = ((chunk * kd_tiles) + kdt) * co_tiles + cot
  int bid = blockIdx.x;
  const int cot = bid % p.co_tiles; bid /= p.co_tiles;
  int kdt = bid % p.kd_tiles;
  const int chunk = bid / p.kd_tiles;
  if (p.kd_cb > 0) {
    // launch the taps of one cin-block adjacently: the 9(taps) x co_tiles
    // blocks re-reading the same x window then run concurrently and share L2
    const int ntap = p.kd_tiles / p.kd_cb;
    kdt = (kdt % ntap) * p.kd_cb + kdt / ntap;
  }
  const int kd0 = kdt * TKD;
  const int co0 = cot * TCO;
  const long long m_begin = (long long)chunk * p.chunk_len;
  const long long m_end = min(m_begin + p.chunk_len, p.M);
  if (m_begin >= m_end) return;

  // wave sub-tile: 2 x WCOLS waves of (TKD/2) x (TCO/WCOLS)
  constexpr int AFRAG = (TKD / 2) / 16;
  constexpr int BFRAG = (TCO / WCOLS) / 16;
  const int wm = wave / WCOLS;    // dW-row half
  const int wn = wave % WCOLS;    // cout column group

  floatx4 acc[AFRAG][BFRAG];
  #pragma unroll
  for (int i = 0; i < AFRAG; ++i)
    #pragma unroll
    for (int j = 0; j < BFRAG; ++j) acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

  // ---- staging geometry: thread t handles pixel m0 + t/NWAVES and NPIECE
  // subtiles q, q+NWAVES, ... (q = t%NWAVES); first ASUB subtiles are A
  const int st_m = tid / NWAVES;
  const int st_q = tid % NWAVES;
  const int st_row = wg_imgrow(st_m);

  // per-piece dW-row bases (tap + cin) — divisions happen ONCE here; the
  // steady-state stage loop is division-free (emulated integer divide beside
  // MFMAs is the anti-lever the guide warns about)
  int pc_ci[NPIECE], pc_kh[NPIECE], pc_kw[NPIECE];
  bool pc_elem[NPIECE];     // piece crosses a tap boundary -> per-element path
  bool pc_isA[NPIECE];
  int pc_co[NPIECE];
  // tap-crossing pieces (stem Cin=3, Cin % 16 != 0 tails): the per-element
  // (ci, kh, kw) walk is STAGE-INVARIANT — precompute it once, packed as
  // ci | kh<<10 | kw<<20 | dead<<30, and decode with shifts in the loop
  // (round-5 measurement: a non-unrolled runtime walk per stage cost 2-3x;
  // an unrolled division chain spilled registers)
  int pe_pack[ELEM ? 16 : 1];
  if (ELEM) {
    #pragma unroll
    for (int e = 0; e < (ELEM ? 16 : 1); ++e) pe_pack[e] = 1 << 30;
  }
  #pragma unroll
  for (int pi = 0; pi < NPIECE; ++pi) {
    const int sub = st_q + NWAVES * pi;
    if (sub < ASUB) {
      const int k0 = kd0 + 16 * sub;
      const int f = p.Cin > 0 ? k0 / p.Cin : 0;
      pc_isA[pi] = true;
      pc_ci[pi] = k0 - f * p.Cin;
      pc_kh[pi] = f / p.KW;
      pc_kw[pi] = f - pc_kh[pi] * p.KW;
      pc_elem[pi] = ELEM &&
                    ((pc_ci[pi] + 16 > p.Cin) || (k0 + 16 > p.KD));
      pc_co[pi] = 0;
      if (ELEM && pc_elem[pi]) {
        // at most one A piece per thread (ASUB == 4, st_q in 0..3; the
        // 128-tile has two A pieces per thread but is only selected for
        // clean-tiled shapes where pc_elem is impossible)
        int cie = pc_ci[pi], khe = pc_kh[pi], kwe = pc_kw[pi];
        #pragma unroll 1
        for (int e = 0; e < 16; ++e) {
          pe_pack[ELEM ? e : 0] =
              (k0 + e < p.KD) ? (cie | (khe << 10) | (kwe << 20)) : (1 << 30);
          if (++cie == p.Cin) {
            cie = 0;
            if (++kwe == p.KW) { kwe = 0; ++khe; }
          }
        }
      }
    } else {
      pc_isA[pi] = false;
      pc_co[pi] = co0 + 16 * (sub - ASUB);
      pc_ci[pi] = pc_kh[pi] = pc_kw[pi] = 0;
      pc_elem[pi] = false;
    }
  }

  // two register banks -> loads are issued TWO stages ahead of their LDS
  // write (PMC r2: 58.7% wait at one-stage prefetch — global latency was not
  // covered by a single compute phase). Separate named arrays + reference
  // parameters keep them in registers (a regs[bank] runtime index would
  // spill the array to scratch).
  unsigned short regs0[NPIECE][16], regs1[NPIECE][16];

  // ---- incremental pixel decomposition m -> (n, ho, wo): one 64-bit division
  // pair at setup, then +64 carry steps per stage
  long long mm = m_begin + st_m;
  int st_n = 0, st_ho = 0, st_wo = 0;
  {
    long long t = mm < p.M ? mm : 0;
    st_wo = (int)(t % p.Wo); t /= p.Wo;
    st_ho = (int)(t % p.Ho);
    st_n = (int)(t / p.Ho);
  }

  auto load_stage = [&](unsigned short (&regs)[NPIECE][16]) {
    const bool m_ok = mm < p.M;
    #pragma unroll
    for (int pi = 0; pi < NPIECE; ++pi) {
      unsigned short* dst = regs[pi];
      if (pc_isA[pi]) {
        if (!pc_elem[pi]) {
          const int hi = st_ho * p.stride - p.pad_h + pc_kh[pi] * p.dil_h;
          const int wi = st_wo * p.stride - p.pad_w + pc_kw[pi] * p.dil_w;
          const bool inside =
              m_ok && hi >= 0 && hi < p.H && wi >= 0 && wi < p.W;
          const unsigned short* src =
              p.x + (((long long)st_n * p.H + hi) * p.W + wi) * p.Cin +
              pc_ci[pi];
          const bool aligned =
              ((reinterpret_cast<uintptr_t>(src)) & 15u) == 0;
          if (inside && aligned) {
            #pragma unroll
            for (int v = 0; v < 2; ++v)
              *reinterpret_cast<ushortv8*>(&dst[v * 8]) =
                  *reinterpret_cast<const ushortv8*>(src + v * 8);
          } else if (inside) {
            #pragma unroll
            for (int e = 0; e < 16; ++e) dst[e] = src[e];
          } else {
            #pragma unroll
            for (int e = 0; e < 16; ++e) dst[e] = 0;
          }
        } else if (ELEM) {
          // tap-crossing piece: decode the precomputed packed walk — shifts
          // and adds only, fully unrolled for ILP
          const int hb = st_ho * p.stride - p.pad_h;
          const int wb = st_wo * p.stride - p.pad_w;
          const long long nbase = (long long)st_n * p.H;
          #pragma unroll
          for (int e = 0; e < 16; ++e) {
            const int pk = pe_pack[ELEM ? e : 0];
            unsigned short v = 0;
            if (m_ok && pk < (1 << 30)) {
              const int cie = pk & 1023;
              const int khe = (pk >> 10) & 1023;
              const int kwe = pk >> 20;
              const int hi = hb + khe * p.dil_h;
              const int wi = wb + kwe * p.dil_w;
              if (hi >= 0 && hi < p.H && wi >= 0 && wi < p.W)
                v = p.x[((nbase + hi) * p.W + wi) * p.Cin + cie];
            }
            dst[e] = v;
          }
        }
      } else {
        const int co = pc_co[pi];
        const unsigned short* src = p.dy + mm * p.Cout + co;
        const bool full = m_ok && co + 16 <= p.Cout;
        const bool aligned = ((reinterpret_cast<uintptr_t>(src)) & 15u) == 0;
        if (full && aligned) {
          #pragma unroll
          for (int v = 0; v < 2; ++v)
            *reinterpret_cast<ushortv8*>(&dst[v * 8]) =
                *reinterpret_cast<const ushortv8*>(src + v * 8);
        } else {
          #pragma unroll
          for (int e = 0; e < 16; ++e)
            dst[e] = (m_ok && co + e < p.Cout) ? src[e] : 0;
        }
      }
    }
    // advance the pixel walker by one stage (+64 pixels)
    mm += 64;
    st_wo += p.step_dwo;
    st_ho += p.step_dho + (st_wo >= p.Wo ? 1 : 0);
    if (st_wo >= p.Wo) st_wo -= p.Wo;
    while (st_ho >= p.Ho) { st_ho -= p.Ho; ++st_n; }
  };

  auto write_stage = [&](int buf, unsigned short (&regs)[NPIECE][16]) {
    #pragma unroll
    for (int pi = 0; pi < NPIECE; ++pi) {
      const int sub = st_q + NWAVES * pi;
      unsigned short* dst = &lds[buf][sub * SUB_SHORTS + st_row * 16];
      #pragma unroll
      for (int v = 0; v < 2; ++v)
        *reinterpret_cast<ushortv8*>(dst + v * 8) =
            *reinterpret_cast<const ushortv8*>(&regs[pi][v * 8]);
    }
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
      short8 af[AFRAG], bf[BFRAG];
      #pragma unroll
      for (int i = 0; i < AFRAG; ++i)
        af[i] = frag(buf, wm * AFRAG + i, ks);
      #pragma unroll
      for (int j = 0; j < BFRAG; ++j)
        bf[j] = frag(buf, ASUB + wn * BFRAG + j, ks);
      #pragma unroll
      for (int i = 0; i < AFRAG; ++i)
        #pragma unroll
        for (int j = 0; j < BFRAG; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
  };

  // ---- main loop: double-buffered LDS, 2-deep register prefetch ----------
  // stage t+2's loads are issued at iteration t, written to LDS at t+1 —
  // each load has two full compute phases to land
  const long long n_stages = (m_end - m_begin + 63) >> 6;
  if constexpr (NWAVES == 8) {
    // single-bank prefetch: the 8-wave block hides latency by wave count;
    // a second register bank pushed the 128-VGPR budget into spills
    load_stage(regs0);
    write_stage(0, regs0);
    __syncthreads();
    for (long long t = 0; t < n_stages; ++t) {
      if (t + 1 < n_stages) load_stage(regs0);
      compute((int)(t & 1));
      if (t + 1 < n_stages) write_stage((int)((t + 1) & 1), regs0);
      __syncthreads();
    }
  } else {
  load_stage(regs0);                           // stage 0
  write_stage(0, regs0);
  if (n_stages > 1) load_stage(regs1);         // stage 1
  __syncthreads();
  for (long long t = 0; t < n_stages; ++t) {   // 2x unrolled over reg banks
    if (t + 2 < n_stages) load_stage(regs0);   // stage t+2
    compute(0);
    if (t + 1 < n_stages) write_stage(1, regs1);
    __syncthreads();
    if (++t >= n_stages) break;
    if (t + 2 < n_stages) load_stage(regs1);
    compute(1);
    if (t + 1 < n_stages) write_stage(0, regs0);
    __syncthreads();
  }
  }

  // ---- epilogue: D lane map col = lane&15, row = (lane>>4)*4 + r
  const int ecol = lane & 15;
  const int erow4 = (lane >> 4) * 4;
  float* out = p.ws + (long long)chunk * p.KD * p.Cout;
  #pragma unroll
  for (int i = 0; i < AFRAG; ++i) {
    #pragma unroll
    for (int j = 0; j < BFRAG; ++j) {
      const int co = co0 + wn * (TCO / WCOLS) + j * 16 + ecol;
      if (co >= p.Cout) continue;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kd = kd0 + wm * (TKD / 2) + i * 16 + erow4 + r;
        if (kd >= p.KD) continue;
        out[(long long)kd * p.Cout + co] = acc[i][j][r];
      }
    }
  }
}

// tree stage: partial[g][e] = sum of ws[g*G .. g*G+G)[e] — full element x
// group parallelism (a flat chunk loop was latency-bound on the 1x1 layers,
// whose tiny tile grids need ~100-200 M-chunks: 6% of the train step)
__global__ void wgrad_reduce_stage_kernel(const float* __restrict__ ws,
                                          float* __restrict__ out,
                                          long long per_chunk, int chunks,
                                          int G, int groups) {
  const long long total = per_chunk * groups;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int g = (int)(i / per_chunk);
    const long long e = i - (long long)g * per_chunk;
    const int c0 = g * G;
    const int c1 = min(c0 + G, chunks);
    float v = 0.f;
    for (int c = c0; c < c1; ++c) v += ws[(long long)c * per_chunk + e];
    out[i] = v;
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
  p.step_dho = 64 / p.Wo;
  p.step_dwo = 64 % p.Wo;
  p.kd_cb = 0;  // set below once TKD is chosen
  // Tile selection. The wide-spatial 3x3 layers are L3-BANDWIDTH-bound on
  // re-reads: with a TCOxTKD grid each x byte is read (taps * Cout/TCO)
  // times and each dy byte (KD/TKD) times per chunk (~9.7 GB for the
  // 256ch@128^2 layer at 64x64 — about the measured runtime at L3 rate).
  // TCO=128 halves both factors; TKD stays 64 (128x128 at 2 blocks/CU
  // measured slower — latency). Small-M layers keep 64x64 (fit in L2, and
  // the 12-subtile LDS drops occupancy 4 -> 3).
  const char* bt = getenv("IBP_WGRAD_BIG");
  const bool big = bt && bt[0] == '1' && p.KD >= 128 && Cout >= 128;
  // wide tile measured: 1x1 @128^2 0.78 -> 0.87x MIOpen, but 3x3 0.60 ->
  // 0.37x (the occupancy-3->2 latency cost beats the halved re-reads when
  // the tap factor, not Cout/TCO, dominates traffic) -> 1x1 only
  const bool wide = !big && KH == 1 && KW == 1 && Cout >= 128 &&
                    p.M >= 65536;
  // mid tile (128x128 @ 8 waves): for the LDS-read-bound wide-spatial KxK
  // layers — same waves/SIMD as 64x64 but 25% fewer tr-read bytes per FLOP
  const bool mid = !big && !wide && KH * KW > 1 && p.KD >= 1024 &&
                   Cout >= 128 && p.M >= 65536;
  const int TKD = (big || mid) ? 128 : 64;
  const int TCO = (big || mid || wide) ? 128 : 64;
  p.kd_tiles = (p.KD + TKD - 1) / TKD;
  p.co_tiles = (int)((Cout + TCO - 1) / TCO);
  if (KH * KW > 1 && Cin % TKD == 0) p.kd_cb = (int)(Cin / TKD);
  if (const char* e = getenv("IBP_WGRAD_NOREORDER"))
    if (e[0] == '1') p.kd_cb = 0;
  // split M into chunks for parallelism: aim ~768 blocks, chunk length a
  // multiple of 64; every chunk slice is fully written (plain stores) before
  // the combine reduces them in fixed order -> deterministic
  const int tiles = p.kd_tiles * p.co_tiles;
  // r14 sweep on (256ch, 3x3, 128^2): 768 blocks -> 0.758 ms, 3072 -> 0.712,
  // 6144 -> 0.691 — more chunks win there (load balance). But a chunk must
  // still run >= ~32 pipeline stages or the block is pure prologue (r15: the
  // unconditional 6144 target dropped 1-stage chunks to 0.28-0.48x) — cap
  // chunks so every block keeps >= 32 x 64-m stages.
  long long max_chunks = std::max<long long>(p.M / (64 * 32), 1);
  int target_blocks = 6144;
  if (const char* e = getenv("IBP_WGRAD_BLOCKS")) target_blocks = atoi(e);
  long long want = (target_blocks + tiles - 1) / tiles;
  long long chunks = std::min<long long>(std::max<long long>(want, 1), max_chunks);
  p.chunk_len = (((p.M + chunks - 1) / chunks + 63) / 64) * 64;
  p.chunks = (int)((p.M + p.chunk_len - 1) / p.chunk_len);

  auto stream = at::hip::getCurrentHIPStream().stream();
  Tensor ws = torch::empty({(long long)p.chunks * p.KD * Cout},
                           x.options().dtype(torch::kFloat32));
  p.ws = ws.data_ptr<float>();
  // every in-range (kd, cout) is written by exactly one block per chunk;
  // out-of-range rows are never read back -> no zero-init needed
  dim3 grid(p.chunks * tiles), block(256);
  // tap-crossing pieces exist iff Cin is not 16-aligned or the last kd-tile
  // is partial — only then compile in the per-element path (costs VGPRs)
  const bool elem = (Cin % 16 != 0) || (p.KD % TKD != 0);
  if (mid && elem) {
    hipLaunchKernelGGL((ibp::conv_wgrad_kernel<128, 128, true, 8>), grid,
                       dim3(512), 0, stream, p);
  } else if (mid) {
    hipLaunchKernelGGL((ibp::conv_wgrad_kernel<128, 128, false, 8>), grid,
                       dim3(512), 0, stream, p);
  } else if (big && elem) {
    hipLaunchKernelGGL((ibp::conv_wgrad_kernel<128, 128, true>), grid, block,
                       0, stream, p);
  } else if (big) {
    hipLaunchKernelGGL((ibp::conv_wgrad_kernel<128, 128, false>), grid, block,
                       0, stream, p);
  } else if (wide && elem) {
    hipLaunchKernelGGL((ibp::conv_wgrad_kernel<64, 128, true>), grid, block, 0,
                       stream, p);
  } else if (wide) {
    hipLaunchKernelGGL((ibp::conv_wgrad_kernel<64, 128, false>), grid, block,
                       0, stream, p);
  } else if (elem) {
    hipLaunchKernelGGL((ibp::conv_wgrad_kernel<64, 64, true>), grid, block, 0,
                       stream, p);
  } else {
    hipLaunchKernelGGL((ibp::conv_wgrad_kernel<64, 64, false>), grid, block, 0,
                       stream, p);
  }

  Tensor dw = torch::empty({Cout, Cin, KH, KW}, x.options());
  long long total = Cout * Cin * KH * KW;
  const float* ws_ptr = p.ws;
  int chunks_left = p.chunks;
  Tensor partial;
  while (chunks_left > 16) {
    // tree stages until the final scatter loops over <= 16 slices
    const int G = 8;
    const int groups = (chunks_left + G - 1) / G;
    const long long per_chunk = (long long)p.KD * Cout;
    Tensor next = torch::empty({(long long)groups * per_chunk},
                               x.options().dtype(torch::kFloat32));
    dim3 rgrid(ibp::grid_1d(per_chunk * groups, 256, 8192)), rblock(256);
    hipLaunchKernelGGL(ibp::wgrad_reduce_stage_kernel, rgrid, rblock, 0,
                       stream, ws_ptr, next.data_ptr<float>(), per_chunk,
                       chunks_left, G, groups);
    partial = next;  // keep alive until the stream consumes it
    ws_ptr = partial.data_ptr<float>();
    chunks_left = groups;
  }
  dim3 cgrid(ibp::grid_1d(total, 256, 4096)), cblock(256);
  hipLaunchKernelGGL(ibp::wgrad_combine_kernel, cgrid, cblock, 0, stream,
                     ws_ptr, reinterpret_cast<unsigned short*>(dw.data_ptr()),
                     total, (int)Cin, (int)(KH * KW), (int)Cout, p.KD,
                     chunks_left);
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
