// NHWC pooling for MI355X (SURVEY.md §2a "MaxPool2d 2x2, AvgPool2d 4x4/2x2").
// channels_last puts C innermost, so a pooling window is KH*KW vector loads
// of the SAME 16-B channel chunk — fully coalesced, no shuffles.
//   max_pool2d fwd: emits y and a per-output uint8 argmax index (window
//     position 0..KH*KW-1 per channel) so the backward is one scatter pass
//     with no recomparison;
//   avg_pool2d fwd/bwd: plain window mean / uniform spread.
// Non-overlapping windows only (stride == kernel, the only form the
// reference's models use: simple_models.py:13, 49-50, 89-92, 213, 464).

#include "fedkit_common.h"

namespace {

typedef __hip_bfloat16 bf16;

template <typename T, int VEC>
struct alignas(sizeof(T) * VEC) PVec { T v[VEC]; };

template <typename T, int VEC>
__global__ void max_pool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                    unsigned char* __restrict__ idx,
                                    int C /* vec units */, int H, int W,
                                    int P, int Q, int k, long long total) {
  using V = PVec<T, VEC>;
  const V* xv = reinterpret_cast<const V*>(x);
  V* yv = reinterpret_cast<V*>(y);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    long long t = i / C;
    int q = (int)(t % Q);
    t /= Q;
    int p = (int)(t % P);
    int n = (int)(t / P);
    const V* win = xv + (((long long)n * H + p * k) * W + q * k) * C + c;
    V best = win[0];
    unsigned char arg[VEC] = {};
    for (int u = 1; u < k * k; ++u) {
      V cand = win[(u / k) * (long long)W * C + (u % k) * C];
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        if (to_f32(cand.v[j]) > to_f32(best.v[j])) {
          best.v[j] = cand.v[j];
          arg[j] = (unsigned char)u;
        }
    }
    yv[i] = best;
#pragma unroll
    for (int j = 0; j < VEC; ++j) idx[i * VEC + j] = arg[j];
  }
}

template <typename T, int VEC>
__global__ void max_pool_bwd_kernel(const T* __restrict__ gy,
                                    const unsigned char* __restrict__ idx,
                                    T* __restrict__ gx, int C, int H, int W,
                                    int P, int Q, int k, long long total) {
  using V = PVec<T, VEC>;
  const V* gv = reinterpret_cast<const V*>(gy);
  V* ov = reinterpret_cast<V*>(gx);
  // gx pre-zeroed; windows are disjoint so no atomics needed
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    long long t = i / C;
    int q = (int)(t % Q);
    t /= Q;
    int p = (int)(t % P);
    int n = (int)(t / P);
    V g = gv[i];
    V* win = ov + (((long long)n * H + p * k) * W + q * k) * C + c;
    // scatter each lane's element to its argmax position; window cells
    // not hit stay zero.  VEC elements may go to different cells, so
    // read-modify-write per cell via scalar stores.
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      int u = idx[i * VEC + j];
      T* cell = reinterpret_cast<T*>(
          win + (u / k) * (long long)W * C + (u % k) * C);
      cell[j] = g.v[j];
    }
  }
}

template <typename T, int VEC>
__global__ void avg_pool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                    int C, int H, int W, int P, int Q, int k,
                                    long long total) {
  using V = PVec<T, VEC>;
  const V* xv = reinterpret_cast<const V*>(x);
  V* yv = reinterpret_cast<V*>(y);
  float inv = 1.0f / (k * k);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    long long t = i / C;
    int q = (int)(t % Q);
    t /= Q;
    int p = (int)(t % P);
    int n = (int)(t / P);
    const V* win = xv + (((long long)n * H + p * k) * W + q * k) * C + c;
    float acc[VEC] = {};
    for (int u = 0; u < k * k; ++u) {
      V cand = win[(u / k) * (long long)W * C + (u % k) * C];
#pragma unroll
      for (int j = 0; j < VEC; ++j) acc[j] += to_f32(cand.v[j]);
    }
    V out;
#pragma unroll
    for (int j = 0; j < VEC; ++j) from_f32(acc[j] * inv, out.v[j]);
    yv[i] = out;
  }
}

template <typename T, int VEC>
__global__ void avg_pool_bwd_kernel(const T* __restrict__ gy,
                                    T* __restrict__ gx, int C, int H, int W,
                                    int P, int Q, int k, long long total) {
  using V = PVec<T, VEC>;
  const V* gv = reinterpret_cast<const V*>(gy);
  V* ov = reinterpret_cast<V*>(gx);
  float inv = 1.0f / (k * k);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    long long t = i / C;
    int q = (int)(t % Q);
    t /= Q;
    int p = (int)(t % P);
    int n = (int)(t / P);
    V g = gv[i];
    V out;
#pragma unroll
    for (int j = 0; j < VEC; ++j) from_f32(to_f32(g.v[j]) * inv, out.v[j]);
    V* win = ov + (((long long)n * H + p * k) * W + q * k) * C + c;
    for (int u = 0; u < k * k; ++u)
      win[(u / k) * (long long)W * C + (u % k) * C] = out;
  }
}

void check_pool(const at::Tensor& x, long k) {
  TORCH_CHECK(x.dim() == 4 &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "pool expects channels_last NHWC");
  TORCH_CHECK(x.size(2) % k == 0 && x.size(3) % k == 0,
              "pool: H/W must divide the window (non-overlapping form)");
}

}  // namespace

#define POOL_DISPATCH(x, NAME, ...)                                        \
  DISPATCH_F32_BF16(x, NAME, {                                             \
    if (C % 8 == 0 && sizeof(scalar_t) == 2) {                             \
      constexpr int VEC = 8;                                               \
      using T = scalar_t;                                                  \
      __VA_ARGS__;                                                         \
    } else if (C % 4 == 0 && sizeof(scalar_t) == 4) {                      \
      constexpr int VEC = 4;                                               \
      using T = scalar_t;                                                  \
      __VA_ARGS__;                                                         \
    } else {                                                               \
      constexpr int VEC = 1;                                               \
      using T = scalar_t;                                                  \
      __VA_ARGS__;                                                         \
    }                                                                      \
  })

std::vector<at::Tensor> fedkit_max_pool2d_fwd(const at::Tensor& x, long k) {
  check_pool(x, k);
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int P = H / (int)k, Q = W / (int)k;
  auto y = at::empty({N, C, P, Q},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto idx = at::empty({N, P, Q, C}, x.options().dtype(at::kByte));
  auto stream = fedkit_stream();
  POOL_DISPATCH(x, "max_pool_fwd", {
    long long total = (long long)N * P * Q * C / VEC;
    hipLaunchKernelGGL((max_pool_fwd_kernel<T, VEC>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, stream,
                       (const T*)x.data_ptr(), (T*)y.data_ptr(),
                       idx.data_ptr<unsigned char>(), C / VEC, H, W, P, Q,
                       (int)k, total);
  });
  return {y, idx};
}

at::Tensor fedkit_max_pool2d_bwd(const at::Tensor& gy, const at::Tensor& idx,
                                 long k, long H, long W) {
  check_pool(gy, 1);
  int N = gy.size(0), C = gy.size(1), P = gy.size(2), Q = gy.size(3);
  // at::zeros IGNORES memory_format in options (empty honors it) —
  // empty + zero_ keeps the channels_last layout the kernel indexes
  auto gx = at::empty({N, C, (long)H, (long)W},
                      gy.options().memory_format(at::MemoryFormat::ChannelsLast));
  gx.zero_();
  auto stream = fedkit_stream();
  POOL_DISPATCH(gy, "max_pool_bwd", {
    long long total = (long long)N * P * Q * C / VEC;
    hipLaunchKernelGGL((max_pool_bwd_kernel<T, VEC>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, stream,
                       (const T*)gy.data_ptr(),
                       idx.data_ptr<unsigned char>(), (T*)gx.data_ptr(),
                       C / VEC, (int)H, (int)W, P, Q, (int)k, total);
  });
  return gx;
}

at::Tensor fedkit_avg_pool2d_fwd(const at::Tensor& x, long k) {
  check_pool(x, k);
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int P = H / (int)k, Q = W / (int)k;
  auto y = at::empty({N, C, P, Q},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto stream = fedkit_stream();
  POOL_DISPATCH(x, "avg_pool_fwd", {
    long long total = (long long)N * P * Q * C / VEC;
    hipLaunchKernelGGL((avg_pool_fwd_kernel<T, VEC>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, stream,
                       (const T*)x.data_ptr(), (T*)y.data_ptr(), C / VEC, H,
                       W, P, Q, (int)k, total);
  });
  return y;
}

at::Tensor fedkit_avg_pool2d_bwd(const at::Tensor& gy, long k, long H,
                                 long W) {
  check_pool(gy, 1);
  int N = gy.size(0), C = gy.size(1), P = gy.size(2), Q = gy.size(3);
  auto gx = at::empty({N, C, (long)H, (long)W},
                      gy.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto stream = fedkit_stream();
  POOL_DISPATCH(gy, "avg_pool_bwd", {
    long long total = (long long)N * P * Q * C / VEC;
    hipLaunchKernelGGL((avg_pool_bwd_kernel<T, VEC>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, stream,
                       (const T*)gy.data_ptr(), (T*)gx.data_ptr(), C / VEC,
                       (int)H, (int)W, P, Q, (int)k, total);
  });
  return gx;
}
