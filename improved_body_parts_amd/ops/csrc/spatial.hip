// NHWC spatial kernels: 2x2 max-pool, x2 nearest upsample, squeeze-excitation.
//
// Replace the reference's nn.MaxPool2d(2,2) / nn.Upsample(nearest) / SELayer
// (reference models/layers_transposed.py:209-210, :285-306). All kernels are
// memory-bound; bf16 moves 8 channels per lane where the layout allows.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace ibp {

// ---- maxpool 2x2 stride 2 -------------------------------------------------
// out[n][ho][wo][c] = max over the 2x2 window; argmax stored as 2 bits for bwd.
template <typename T>
__global__ void maxpool2x2_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                      unsigned char* __restrict__ arg,
                                      int N, int H, int W, int C) {
  int Ho = H >> 1, Wo = W >> 1;
  long long total = (long long)N * Ho * Wo * C;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    long long p = i / C;
    int wo = (int)(p % Wo);
    long long q = p / Wo;
    int ho = (int)(q % Ho);
    int n = (int)(q / Ho);
    const T* base = x + (((long long)n * H + 2 * ho) * W + 2 * wo) * C + c;
    float v00 = ldf(base);
    float v01 = ldf(base + C);
    float v10 = ldf(base + (long long)W * C);
    float v11 = ldf(base + (long long)W * C + C);
    float m = v00;
    int a = 0;
    if (v01 > m) { m = v01; a = 1; }
    if (v10 > m) { m = v10; a = 2; }
    if (v11 > m) { m = v11; a = 3; }
    stf(y + i, m);
    arg[i] = (unsigned char)a;
  }
}

template <typename T>
__global__ void maxpool2x2_bwd_kernel(const T* __restrict__ dy,
                                      const unsigned char* __restrict__ arg,
                                      T* __restrict__ dx,
                                      int N, int H, int W, int C) {
  // iterate over INPUT positions: each belongs to exactly one window
  int Ho = H >> 1, Wo = W >> 1;
  long long total = (long long)N * H * W * C;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    long long p = i / C;
    int w = (int)(p % W);
    long long q = p / W;
    int h = (int)(q % H);
    int n = (int)(q / H);
    if (h >= 2 * Ho || w >= 2 * Wo) { stf(dx + i, 0.f); continue; }
    int ho = h >> 1, wo = w >> 1;
    int pos = ((h & 1) << 1) | (w & 1);
    long long o = (((long long)n * Ho + ho) * Wo + wo) * C + c;
    stf(dx + i, arg[o] == pos ? ldf(dy + o) : 0.f);
  }
}

// ---- nearest x2 upsample ---------------------------------------------------
template <typename T>
__global__ void upsample2x_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                      int N, int H, int W, int C) {
  int Ho = H * 2, Wo = W * 2;
  long long total = (long long)N * Ho * Wo * C;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    long long p = i / C;
    int wo = (int)(p % Wo);
    long long q = p / Wo;
    int ho = (int)(q % Ho);
    int n = (int)(q / Ho);
    long long src = (((long long)n * H + (ho >> 1)) * W + (wo >> 1)) * C + c;
    stf(y + i, ldf(x + src));
  }
}

template <typename T>
__global__ void upsample2x_bwd_kernel(const T* __restrict__ dy, T* __restrict__ dx,
                                      int N, int H, int W, int C) {
  // dx[h][w] = sum of the 4 dys that sampled it (H, W are INPUT dims)
  int Ho = H * 2, Wo = W * 2;
  long long total = (long long)N * H * W * C;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    long long p = i / C;
    int w = (int)(p % W);
    long long q = p / W;
    int h = (int)(q % H);
    int n = (int)(q / H);
    const T* base = dy + (((long long)n * Ho + 2 * h) * Wo + 2 * w) * C + c;
    float s = ldf(base) + ldf(base + C) + ldf(base + (long long)Wo * C)
            + ldf(base + (long long)Wo * C + C);
    stf(dx + i, s);
  }
}

// ---- squeeze-excitation helpers -------------------------------------------
// channel reduction r[n][c] = sum_hw a[n][h][w][c] * (b ? b[n][h][w][c] : 1)
// Partials go to a [N][rows][C] workspace with plain stores; a column-sum
// stage reduces rows (the previous atomicAdd form serialised rows-way per
// (n,c) address and needed a zero-init launch).
template <typename T, bool PROD>
__global__ void se_reduce_kernel(const T* __restrict__ a, const T* __restrict__ b,
                                 float* __restrict__ ws, int N, long long HW, int C) {
  int c = blockIdx.y * blockDim.x + threadIdx.x;
  if (c >= C) return;
  int n = blockIdx.z;
  const T* pa = a + (long long)n * HW * C;
  const T* pb = PROD ? b + (long long)n * HW * C : nullptr;
  float s = 0.f;
  for (long long m = blockIdx.x; m < HW; m += gridDim.x) {
    float v = ldf(pa + m * C + c);
    if (PROD) v *= ldf(pb + m * C + c);
    s += v;
  }
  ws[((long long)n * gridDim.x + blockIdx.x) * C + c] = s;
}

// out[p][c] = sum_r ws[p][r][c] (same shape as bn_act.hip's colsum; kept
// local — device code cannot cross TUs under -fno-gpu-rdc)
__global__ __launch_bounds__(1024) void se_colsum_kernel(
    const float* __restrict__ ws, float* __restrict__ out, int rows, int C,
    int planes) {
  __shared__ float red[1024];
  const int total = C * planes;
  const int cl = threadIdx.x & 63;
  const int rl = threadIdx.x >> 6;
  const int i = blockIdx.x * 64 + cl;
  float a = 0.f;
  if (i < total) {
    const int p = i / C;
    const int c = i - p * C;
    const float* src = ws + (long long)p * rows * C + c;
    for (int r = rl; r < rows; r += 16) a += src[(long long)r * C];
  }
  red[threadIdx.x] = a;
  __syncthreads();
  if (rl == 0 && i < total) {
    #pragma unroll
    for (int g = 1; g < 16; ++g) a += red[g * 64 + cl];
    out[i] = a;
  }
}

// y = x * s[n][c] (+ optional add[n][c] broadcast)

// fused SE gate: s = sigmoid(W2 @ leaky(W1 @ pooled + b1) + b2) — replaces
// two library GEMM launches + bias/activation elementwise on the inference
// path (the FCs are 256x16: far below any GEMM library's useful size).
template <typename T>
__global__ void se_gate_kernel(const float* __restrict__ pooled,
                               const T* __restrict__ w1, const T* __restrict__ b1,
                               const T* __restrict__ w2, const T* __restrict__ b2,
                               float* __restrict__ s, int C, int CH,
                               float slope, float pool_scale) {
  __shared__ float pl[768 + 64];
  const int n = blockIdx.x;
  const int tid = threadIdx.x;
  // pool_scale folds the GAP division (sums -> mean) into this kernel
  for (int c = tid; c < C; c += blockDim.x)
    pl[c] = pooled[(long long)n * C + c] * pool_scale;
  __syncthreads();
  if (tid < CH) {
    float acc = ldf(b1 + tid);
    const T* row = w1 + (long long)tid * C;
    for (int c = 0; c < C; ++c) acc += ldf(row + c) * pl[c];
    pl[C + tid] = leaky(acc, slope);
  }
  __syncthreads();
  for (int c = tid; c < C; c += blockDim.x) {
    float acc = ldf(b2 + c);
    const T* row = w2 + (long long)c * CH;
    for (int j = 0; j < CH; ++j) acc += ldf(row + j) * pl[C + j];
    s[(long long)n * C + c] = 1.f / (1.f + __expf(-acc));
  }
}

template <typename T>
__global__ void se_scale_kernel(const T* __restrict__ x, const float* __restrict__ s,
                                const float* __restrict__ addc, T* __restrict__ y,
                                int N, long long HW, int C) {
  long long total = (long long)N * HW * C;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int n = (int)(i / (HW * C));
    float v = ldf(x + i) * s[(long long)n * C + c];
    if (addc != nullptr) v += addc[(long long)n * C + c];
    stf(y + i, v);
  }
}

// ---- vectorized bf16 variants (C % 8 == 0): 16-B accesses, one pixel
// decomposition per 8 channels — the scalar forms were per-element
// div/mod + 2-B-access bound (~4x off bandwidth in the r2 profile) --------
typedef unsigned short ushort8v __attribute__((ext_vector_type(8)));

__global__ void maxpool2x2_fwd_bf16v8(const ushort8v* __restrict__ x,
                                      ushort8v* __restrict__ y,
                                      unsigned char* __restrict__ arg,
                                      int N, int H, int W, int C8) {
  const int Ho = H >> 1, Wo = W >> 1;
  const long long total = (long long)N * Ho * Wo * C8;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int c8 = (int)(i % C8);
    long long p = i / C8;
    const int wo = (int)(p % Wo);
    long long q = p / Wo;
    const int ho = (int)(q % Ho);
    const int n = (int)(q / Ho);
    const ushort8v* base =
        x + (((long long)n * H + 2 * ho) * W + 2 * wo) * C8 + c8;
    ushort8v v00 = base[0];
    ushort8v v01 = base[C8];
    ushort8v v10 = base[(long long)W * C8];
    ushort8v v11 = base[(long long)W * C8 + C8];
    ushort8v out;
    unsigned char* ab = arg + i * 8;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float m = us2f(v00[j]);
      int a = 0;
      float f01 = us2f(v01[j]), f10 = us2f(v10[j]), f11 = us2f(v11[j]);
      if (f01 > m) { m = f01; a = 1; }
      if (f10 > m) { m = f10; a = 2; }
      if (f11 > m) { m = f11; a = 3; }
      out[j] = f2us(m);
      ab[j] = (unsigned char)a;
    }
    y[i] = out;
  }
}

__global__ void maxpool2x2_bwd_bf16v8(const ushort8v* __restrict__ dy,
                                      const unsigned char* __restrict__ arg,
                                      ushort8v* __restrict__ dx,
                                      int N, int H, int W, int C8) {
  const int Ho = H >> 1, Wo = W >> 1;
  const long long total = (long long)N * H * W * C8;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int c8 = (int)(i % C8);
    long long p = i / C8;
    const int w = (int)(p % W);
    long long q = p / W;
    const int h = (int)(q % H);
    const int n = (int)(q / H);
    ushort8v out = {0, 0, 0, 0, 0, 0, 0, 0};
    if (h < 2 * Ho && w < 2 * Wo) {
      const long long o =
          (((long long)n * Ho + (h >> 1)) * Wo + (w >> 1)) * C8 + c8;
      const int pos = ((h & 1) << 1) | (w & 1);
      const ushort8v g = dy[o];
      const unsigned char* ab = arg + o * 8;
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        if (ab[j] == pos) out[j] = g[j];
    }
    dx[i] = out;
  }
}

__global__ void upsample2x_fwd_bf16v8(const ushort8v* __restrict__ x,
                                      ushort8v* __restrict__ y,
                                      int N, int H, int W, int C8) {
  const int Ho = H * 2, Wo = W * 2;
  const long long total = (long long)N * Ho * Wo * C8;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int c8 = (int)(i % C8);
    long long p = i / C8;
    const int wo = (int)(p % Wo);
    long long q = p / Wo;
    const int ho = (int)(q % Ho);
    const int n = (int)(q / Ho);
    y[i] = x[(((long long)n * H + (ho >> 1)) * W + (wo >> 1)) * C8 + c8];
  }
}

__global__ void upsample2x_bwd_bf16v8(const ushort8v* __restrict__ dy,
                                      ushort8v* __restrict__ dx,
                                      int N, int H, int W, int C8) {
  const int Ho = H * 2, Wo = W * 2;
  const long long total = (long long)N * H * W * C8;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int c8 = (int)(i % C8);
    long long p = i / C8;
    const int w = (int)(p % W);
    long long q = p / W;
    const int h = (int)(q % H);
    const int n = (int)(q / H);
    const ushort8v* base =
        dy + (((long long)n * Ho + 2 * h) * Wo + 2 * w) * C8 + c8;
    ushort8v d0 = base[0], d1 = base[C8];
    ushort8v d2 = base[(long long)Wo * C8], d3 = base[(long long)Wo * C8 + C8];
    ushort8v out;
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      out[j] = f2us((us2f(d0[j]) + us2f(d1[j])) +
                    (us2f(d2[j]) + us2f(d3[j])));
    dx[i] = out;
  }
}

__global__ void se_scale_bf16v8(const ushort8v* __restrict__ x,
                                const float* __restrict__ s,
                                const float* __restrict__ addc,
                                ushort8v* __restrict__ y,
                                int N, long long HW, int C8) {
  const long long total = (long long)N * HW * C8;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int c8 = (int)(i % C8);
    const int n = (int)(i / (HW * C8));
    const float* sc = s + (long long)n * C8 * 8 + c8 * 8;
    const float* ad = addc ? addc + (long long)n * C8 * 8 + c8 * 8 : nullptr;
    ushort8v xv = x[i];
    ushort8v out;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = us2f(xv[j]) * sc[j];
      if (ad) v += ad[j];
      out[j] = f2us(v);
    }
    y[i] = out;
  }
}

}  // namespace ibp

using torch::Tensor;
static inline hipStream_t cur_stream2() {
  return at::hip::getCurrentHIPStream().stream();
}

static inline bool v8_ok(const Tensor& t, int64_t C) {
  return t.scalar_type() == at::ScalarType::BFloat16 && C % 8 == 0;
}

std::vector<Tensor> maxpool2x2_fwd(const Tensor& x, int64_t N, int64_t H, int64_t W,
                                   int64_t C) {
  Tensor y = torch::empty({N, H / 2, W / 2, C}, x.options());
  Tensor arg = torch::empty({N, H / 2, W / 2, C},
                            x.options().dtype(torch::kUInt8));
  dim3 block(256), grid(ibp::grid_1d(y.numel(), 256, 8192));
  if (v8_ok(x, C)) {
    dim3 g8(ibp::grid_1d(y.numel() / 8, 256, 8192));
    hipLaunchKernelGGL(ibp::maxpool2x2_fwd_bf16v8, g8, block, 0, cur_stream2(),
                       reinterpret_cast<const ibp::ushort8v*>(x.data_ptr()),
                       reinterpret_cast<ibp::ushort8v*>(y.data_ptr()),
                       arg.data_ptr<unsigned char>(), (int)N, (int)H, (int)W,
                       (int)(C / 8));
    return {y, arg};
  }
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      x.scalar_type(), "maxpool2x2_fwd", [&] {
    hipLaunchKernelGGL(ibp::maxpool2x2_fwd_kernel<scalar_t>, grid, block, 0,
                       cur_stream2(), reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<scalar_t*>(y.data_ptr()),
                       arg.data_ptr<unsigned char>(), (int)N, (int)H, (int)W, (int)C);
  });
  return {y, arg};
}

Tensor maxpool2x2_bwd(const Tensor& dy, const Tensor& arg, int64_t N, int64_t H,
                      int64_t W, int64_t C) {
  Tensor dx = torch::empty({N, H, W, C}, dy.options());
  dim3 block(256), grid(ibp::grid_1d(dx.numel(), 256, 8192));
  if (v8_ok(dy, C)) {
    dim3 g8(ibp::grid_1d(dx.numel() / 8, 256, 8192));
    hipLaunchKernelGGL(ibp::maxpool2x2_bwd_bf16v8, g8, block, 0, cur_stream2(),
                       reinterpret_cast<const ibp::ushort8v*>(dy.data_ptr()),
                       arg.data_ptr<unsigned char>(),
                       reinterpret_cast<ibp::ushort8v*>(dx.data_ptr()),
                       (int)N, (int)H, (int)W, (int)(C / 8));
    return dx;
  }
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      dy.scalar_type(), "maxpool2x2_bwd", [&] {
    hipLaunchKernelGGL(ibp::maxpool2x2_bwd_kernel<scalar_t>, grid, block, 0,
                       cur_stream2(), reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                       arg.data_ptr<unsigned char>(),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()),
                       (int)N, (int)H, (int)W, (int)C);
  });
  return dx;
}

Tensor upsample2x_fwd(const Tensor& x, int64_t N, int64_t H, int64_t W, int64_t C) {
  Tensor y = torch::empty({N, H * 2, W * 2, C}, x.options());
  dim3 block(256), grid(ibp::grid_1d(y.numel(), 256, 8192));
  if (v8_ok(x, C)) {
    dim3 g8(ibp::grid_1d(y.numel() / 8, 256, 8192));
    hipLaunchKernelGGL(ibp::upsample2x_fwd_bf16v8, g8, block, 0, cur_stream2(),
                       reinterpret_cast<const ibp::ushort8v*>(x.data_ptr()),
                       reinterpret_cast<ibp::ushort8v*>(y.data_ptr()),
                       (int)N, (int)H, (int)W, (int)(C / 8));
    return y;
  }
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      x.scalar_type(), "upsample2x_fwd", [&] {
    hipLaunchKernelGGL(ibp::upsample2x_fwd_kernel<scalar_t>, grid, block, 0,
                       cur_stream2(), reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<scalar_t*>(y.data_ptr()),
                       (int)N, (int)H, (int)W, (int)C);
  });
  return y;
}

Tensor upsample2x_bwd(const Tensor& dy, int64_t N, int64_t H, int64_t W, int64_t C) {
  Tensor dx = torch::empty({N, H, W, C}, dy.options());
  dim3 block(256), grid(ibp::grid_1d(dx.numel(), 256, 8192));
  if (v8_ok(dy, C)) {
    dim3 g8(ibp::grid_1d(dx.numel() / 8, 256, 8192));
    hipLaunchKernelGGL(ibp::upsample2x_bwd_bf16v8, g8, block, 0, cur_stream2(),
                       reinterpret_cast<const ibp::ushort8v*>(dy.data_ptr()),
                       reinterpret_cast<ibp::ushort8v*>(dx.data_ptr()),
                       (int)N, (int)H, (int)W, (int)(C / 8));
    return dx;
  }
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      dy.scalar_type(), "upsample2x_bwd", [&] {
    hipLaunchKernelGGL(ibp::upsample2x_bwd_kernel<scalar_t>, grid, block, 0,
                       cur_stream2(), reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()),
                       (int)N, (int)H, (int)W, (int)C);
  });
  return dx;
}

// r[n][c] = sum_hw a (*b); used for SE global-avg-pool (b absent) and
// SE backward ds (b = the other operand)
Tensor se_reduce(const Tensor& a, const c10::optional<Tensor>& b, int64_t N,
                 int64_t HW, int64_t C) {
  Tensor out = torch::empty({N, C}, a.options().dtype(torch::kFloat32));
  dim3 block(256);
  int rows = (int)std::min<long long>((HW + 63) / 64, 512);
  dim3 grid(rows, (C + 255) / 256, N);
  Tensor ws = torch::empty({N * (int64_t)rows * C},
                           a.options().dtype(torch::kFloat32));
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      a.scalar_type(), "se_reduce", [&] {
    using T = scalar_t;
    if (b.has_value()) {
      hipLaunchKernelGGL((ibp::se_reduce_kernel<T, true>), grid, block, 0,
                         cur_stream2(), reinterpret_cast<const T*>(a.data_ptr()),
                         reinterpret_cast<const T*>(b->data_ptr()),
                         ws.data_ptr<float>(), (int)N, HW, (int)C);
    } else {
      hipLaunchKernelGGL((ibp::se_reduce_kernel<T, false>), grid, block, 0,
                         cur_stream2(), reinterpret_cast<const T*>(a.data_ptr()),
                         nullptr, ws.data_ptr<float>(), (int)N, HW, (int)C);
    }
  });
  dim3 cgrid(((int)(N * C) + 63) / 64);
  hipLaunchKernelGGL(ibp::se_colsum_kernel, cgrid, dim3(1024), 0, cur_stream2(),
                     ws.data_ptr<float>(), out.data_ptr<float>(), rows, (int)C,
                     (int)N);
  return out;
}

Tensor se_gate(const Tensor& pooled, const Tensor& w1, const Tensor& b1,
               const Tensor& w2, const Tensor& b2, double slope,
               double pool_scale) {
  TORCH_CHECK(pooled.is_cuda() && pooled.dim() == 2 &&
              pooled.scalar_type() == at::ScalarType::Float);
  int64_t N = pooled.size(0), C = pooled.size(1);
  int64_t CH = w1.size(0);
  TORCH_CHECK(C <= 768 && CH <= 64, "se_gate LDS layout limit");
  TORCH_CHECK(w1.is_contiguous() && w2.is_contiguous() &&
              b1.is_contiguous() && b2.is_contiguous());
  // all four params are reinterpreted as w1's dtype below — a mixed-dtype
  // SE module would silently read garbage without these checks
  TORCH_CHECK(b1.scalar_type() == w1.scalar_type() &&
              w2.scalar_type() == w1.scalar_type() &&
              b2.scalar_type() == w1.scalar_type(),
              "se_gate: w1/b1/w2/b2 dtypes must match");
  TORCH_CHECK(w1.dim() == 2 && w1.size(1) == C && b1.numel() == CH &&
              w2.dim() == 2 && w2.size(0) == C && w2.size(1) == CH &&
              b2.numel() == C,
              "se_gate: expected w1 [CH,C], b1 [CH], w2 [C,CH], b2 [C]");
  Tensor s = torch::empty({N, C}, pooled.options());
  dim3 block(256), grid((unsigned)N);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      w1.scalar_type(), "se_gate", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(ibp::se_gate_kernel<T>, grid, block, 0, cur_stream2(),
                       pooled.data_ptr<float>(),
                       reinterpret_cast<const T*>(w1.data_ptr()),
                       reinterpret_cast<const T*>(b1.data_ptr()),
                       reinterpret_cast<const T*>(w2.data_ptr()),
                       reinterpret_cast<const T*>(b2.data_ptr()),
                       s.data_ptr<float>(), (int)C, (int)CH, (float)slope,
                       (float)pool_scale);
  });
  return s;
}

Tensor se_scale(const Tensor& x, const Tensor& s, const c10::optional<Tensor>& addc,
                int64_t N, int64_t HW, int64_t C) {
  Tensor y = torch::empty_like(x);
  dim3 block(256), grid(ibp::grid_1d(x.numel(), 256, 8192));
  if (v8_ok(x, C)) {
    dim3 g8(ibp::grid_1d(x.numel() / 8, 256, 8192));
    hipLaunchKernelGGL(ibp::se_scale_bf16v8, g8, block, 0, cur_stream2(),
                       reinterpret_cast<const ibp::ushort8v*>(x.data_ptr()),
                       s.data_ptr<float>(),
                       addc.has_value() ? addc->data_ptr<float>() : nullptr,
                       reinterpret_cast<ibp::ushort8v*>(y.data_ptr()),
                       (int)N, HW, (int)(C / 8));
    return y;
  }
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      x.scalar_type(), "se_scale", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(ibp::se_scale_kernel<T>, grid, block, 0, cur_stream2(),
                       reinterpret_cast<const T*>(x.data_ptr()),
                       s.data_ptr<float>(),
                       addc.has_value() ? addc->data_ptr<float>() : nullptr,
                       reinterpret_cast<T*>(y.data_ptr()), (int)N, HW, (int)C);
  });
  return y;
}
