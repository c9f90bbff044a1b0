// Hand-written Linear (fully-connected) fwd/bwd kernels.
//
// SURVEY §2a row "Linear fwd+bwd ... up to 4096x512 tiny".  Every fc layer
// in the model zoo is LATENCY-bound, not math-bound (largest: Net1 fc1
// [128,1600]x[1600,512] = 105 MFLOP-class; ResNet18 head [128,512]x[512,10]),
// so the right design is one fused kernel per pass — fp32 accumulation,
// 16-B vector loads along the contiguous K axis, bias fused into the
// forward — rather than a rocBLAS GEMM + separate bias/epilogue launches.
// Weights stay in torch's nn.Linear layout ([Nout, K] row-major), so both
// the forward (x rows . W rows) and the backward-weight (columns of gy
// against x rows) read K-contiguous vectors.

#include "fedkit_common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;

template <typename T>
__device__ __forceinline__ float dot8(const T* a, const T* b) {
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) s += to_f32(a[j]) * to_f32(b[j]);
  return s;
}

// y[m][n] = x[m][:] . W[n][:] + b[n]  — one WAVE per output element,
// lanes split the K axis (8 elements each), shuffle-reduce.
template <typename T>
__global__ __launch_bounds__(256)
void linear_fwd_kernel(const T* __restrict__ x,   // [M][K]
                       const T* __restrict__ w,   // [N][K]
                       const float* __restrict__ b,  // [N] or null
                       T* __restrict__ y,         // [M][N]
                       long long M, int N, int K) {
  const int lane = threadIdx.x & 63;
  const long long wstride =
      ((long long)gridDim.x * blockDim.x) >> 6;   // waves in the grid
  for (long long wid = ((long long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
       wid < M * N; wid += wstride) {
    const int n = (int)(wid % N);
    const long long m = wid / N;
    const T* xr = x + m * K;
    const T* wr = w + (long long)n * K;
    float acc = 0.f;
    for (int k = lane * 8; k + 8 <= K; k += 64 * 8)
      acc += dot8(xr + k, wr + k);
    // K tail (K % 8 != 0): lane 0 picks it up serially
    if (lane == 0)
      for (int k = (K / 8) * 8; k < K; ++k)
        acc += to_f32(xr[k]) * to_f32(wr[k]);
    for (int off = 32; off > 0; off >>= 1)
      acc += __shfl_xor(acc, off, 64);
    if (lane == 0) {
      float out = acc;
      if (b) out += b[n];
      from_f32(out, y[m * N + n]);
    }
  }
}

// gx[m][k8] = sum_n gy[m][n] * W[n][k8..] — thread per (m, k-chunk).
template <typename T>
__global__ __launch_bounds__(256)
void linear_bwd_data_kernel(const T* __restrict__ gy,  // [M][N]
                            const T* __restrict__ w,   // [N][K]
                            T* __restrict__ gx,        // [M][K]
                            long long M, int N, int K) {
  const int kc8 = (K + 7) / 8;
  for (long long e = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       e < M * kc8; e += (long long)gridDim.x * blockDim.x) {
    const long long m = e / kc8;
    const int k0 = (int)(e % kc8) * 8;
    const int kw = min(8, K - k0);
    float acc[8] = {};
    const T* gr = gy + m * N;
    for (int n = 0; n < N; ++n) {
      float g = to_f32(gr[n]);
      const T* wr = w + (long long)n * K + k0;
      for (int j = 0; j < kw; ++j) acc[j] += g * to_f32(wr[j]);
    }
    for (int j = 0; j < kw; ++j) from_f32(acc[j], gx[m * K + k0 + j]);
  }
}

// gw[n][k8] = sum_m gy[m][n] * x[m][k8..]; k-chunk 0 waves also emit
// gb[n] = sum_m gy[m][n].  One WAVE per (n, k-chunk), lanes split the M
// (reduction) axis, xor-shuffle combine — grids of N*kc8 THREADS left the
// chip idle (the [128,512]x[512,10] head made a 640-thread launch that
// measured 126 us; waves over M give 64x the parallelism).
template <typename T>
__global__ __launch_bounds__(256)
void linear_bwd_weight_kernel(const T* __restrict__ gy,  // [M][N]
                              const T* __restrict__ x,   // [M][K]
                              float* __restrict__ gw,    // [N][K] fp32
                              float* __restrict__ gb,    // [N] fp32 or null
                              long long M, int N, int K) {
  const int kc8 = (K + 7) / 8;
  const int lane = threadIdx.x & 63;
  const long long wstride = ((long long)gridDim.x * blockDim.x) >> 6;
  for (long long wid = ((long long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
       wid < (long long)N * kc8; wid += wstride) {
    const int n = (int)(wid / kc8);
    const int k0 = (int)(wid % kc8) * 8;
    const int kw = min(8, K - k0);
    float acc[8] = {};
    float bacc = 0.f;
    for (long long m = lane; m < M; m += 64) {
      float g = to_f32(gy[m * N + n]);
      const T* xr = x + m * K + k0;
      for (int j = 0; j < kw; ++j) acc[j] += g * to_f32(xr[j]);
      bacc += g;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j)
      for (int off = 32; off > 0; off >>= 1)
        acc[j] += __shfl_xor(acc[j], off, 64);
    for (int off = 32; off > 0; off >>= 1)
      bacc += __shfl_xor(bacc, off, 64);
    if (lane == 0) {
      for (int j = 0; j < kw; ++j) gw[(long long)n * K + k0 + j] = acc[j];
      if (gb && k0 == 0) gb[n] = bacc;
    }
  }
}

}  // namespace

at::Tensor fedkit_linear_fwd(const at::Tensor& x, const at::Tensor& w,
                             const c10::optional<at::Tensor>& bias) {
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1),
              "linear_fwd expects x [M,K], w [N,K]");
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  long long M = x.size(0);
  int N = (int)w.size(0), K = (int)w.size(1);
  auto y = at::empty({M, (long long)N}, x.options());
  at::Tensor bf;
  const float* bp = nullptr;
  if (bias.has_value()) {
    bf = bias->to(at::kFloat).contiguous();
    bp = bf.data_ptr<float>();
  }
  long long waves = M * N;
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(xc, "linear_fwd", {
    hipLaunchKernelGGL((linear_fwd_kernel<scalar_t>),
                       dim3(grid_1d(waves * 64, 256, 1u << 20)), dim3(256),
                       0, stream, (const scalar_t*)xc.data_ptr(),
                       (const scalar_t*)wc.data_ptr(), bp,
                       (scalar_t*)y.data_ptr(), M, N, K);
  });
  return y;
}

at::Tensor fedkit_linear_bwd_data(const at::Tensor& gy, const at::Tensor& w) {
  auto gc = gy.contiguous();
  auto wc = w.contiguous();
  long long M = gy.size(0);
  int N = (int)w.size(0), K = (int)w.size(1);
  auto gx = at::empty({M, (long long)K}, gy.options());
  const int kc8 = (K + 7) / 8;
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(gc, "linear_bwd_data", {
    hipLaunchKernelGGL((linear_bwd_data_kernel<scalar_t>),
                       dim3(grid_1d(M * kc8, 256, 1u << 20)), dim3(256), 0,
                       stream, (const scalar_t*)gc.data_ptr(),
                       (const scalar_t*)wc.data_ptr(),
                       (scalar_t*)gx.data_ptr(), M, N, K);
  });
  return gx;
}

std::vector<at::Tensor> fedkit_linear_bwd_weight(const at::Tensor& gy,
                                                 const at::Tensor& x,
                                                 bool want_bias) {
  auto gc = gy.contiguous();
  auto xc = x.contiguous();
  long long M = gy.size(0);
  int N = (int)gy.size(1), K = (int)x.size(1);
  auto opts = x.options().dtype(at::kFloat);
  auto gw = at::empty({(long long)N, (long long)K}, opts);
  at::Tensor gb;
  float* gbp = nullptr;
  if (want_bias) {
    gb = at::empty({N}, opts);
    gbp = gb.data_ptr<float>();
  }
  const int kc8 = (K + 7) / 8;
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(gc, "linear_bwd_weight", {
    hipLaunchKernelGGL((linear_bwd_weight_kernel<scalar_t>),
                       dim3(grid_1d((long long)N * kc8 * 64, 256, 1u << 20)),
                       dim3(256), 0, stream,
                       (const scalar_t*)gc.data_ptr(),
                       (const scalar_t*)xc.data_ptr(), gw.data_ptr<float>(),
                       gbp, M, N, K);
  });
  if (want_bias) return {gw, gb};
  return {gw};
}
