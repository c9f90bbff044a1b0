#include "hip/hip_runtime.h"
// Vectorized elementwise kernels: ELU fwd/bwd, fused residual-add + ELU.
// Replaces the reference's F.elu dispatches (simple_models.py:20-24,150-153;
// SURVEY.md §2a "ELU fwd+bwd").  Pure HBM-bandwidth ops on MI355X: 16 B/lane
// vector access (8 bf16 / 4 fp32), grid-stride, grid capped at 2048 blocks.

#include "fedkit_common.h"

namespace {

template <typename T, int VEC>
struct alignas(sizeof(T) * VEC) VecT { T v[VEC]; };

__device__ __forceinline__ float elu_f(float x) {
  return x > 0.f ? x : __expf(x) - 1.f;
}
// backward from the saved OUTPUT: y>0 -> 1 else y+1 (= exp(x))
__device__ __forceinline__ float elu_bwd_f(float gy, float y) {
  return y > 0.f ? gy : gy * (y + 1.f);
}

template <typename T, int VEC>
__global__ void elu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                               long long nvec) {
  using V = VecT<T, VEC>;
  const V* xv = reinterpret_cast<const V*>(x);
  V* yv = reinterpret_cast<V*>(y);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < nvec; i += (long long)gridDim.x * blockDim.x) {
    V a = xv[i];
    V r;
#pragma unroll
    for (int j = 0; j < VEC; ++j) from_f32(elu_f(to_f32(a.v[j])), r.v[j]);
    yv[i] = r;
  }
}

template <typename T, int VEC>
__global__ void elu_bwd_kernel(const T* __restrict__ gy, const T* __restrict__ y,
                               T* __restrict__ gx, long long nvec) {
  using V = VecT<T, VEC>;
  const V* gv = reinterpret_cast<const V*>(gy);
  const V* yv = reinterpret_cast<const V*>(y);
  V* ov = reinterpret_cast<V*>(gx);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < nvec; i += (long long)gridDim.x * blockDim.x) {
    V g = gv[i], a = yv[i];
    V r;
#pragma unroll
    for (int j = 0; j < VEC; ++j)
      from_f32(elu_bwd_f(to_f32(g.v[j]), to_f32(a.v[j])), r.v[j]);
    ov[i] = r;
  }
}

template <typename T>
__global__ void elu_fwd_tail(const T* __restrict__ x, T* __restrict__ y,
                             long long start, long long n) {
  long long i = start + blockIdx.x * (long long)blockDim.x + threadIdx.x;
  if (i < n) from_f32(elu_f(to_f32(x[i])), y[i]);
}

template <typename T>
__global__ void elu_bwd_tail(const T* __restrict__ gy, const T* __restrict__ y,
                             T* __restrict__ gx, long long start, long long n) {
  long long i = start + blockIdx.x * (long long)blockDim.x + threadIdx.x;
  if (i < n) from_f32(elu_bwd_f(to_f32(gy[i]), to_f32(y[i])), gx[i]);
}

template <typename T, int VEC>
void launch_elu_fwd(const at::Tensor& x, at::Tensor& y) {
  long long n = x.numel();
  long long nvec = n / VEC;
  auto stream = fedkit_stream();
  if (nvec > 0)
    hipLaunchKernelGGL((elu_fwd_kernel<T, VEC>), dim3(grid_1d(nvec, 256)),
                       dim3(256), 0, stream,
                       (const T*)x.data_ptr(), (T*)y.data_ptr(), nvec);
  long long tail = n - nvec * VEC;
  if (tail > 0)
    hipLaunchKernelGGL((elu_fwd_tail<T>), dim3(1), dim3(256), 0, stream,
                       (const T*)x.data_ptr(), (T*)y.data_ptr(), nvec * VEC, n);
}

template <typename T, int VEC>
void launch_elu_bwd(const at::Tensor& gy, const at::Tensor& y, at::Tensor& gx) {
  long long n = y.numel();
  long long nvec = n / VEC;
  auto stream = fedkit_stream();
  if (nvec > 0)
    hipLaunchKernelGGL((elu_bwd_kernel<T, VEC>), dim3(grid_1d(nvec, 256)),
                       dim3(256), 0, stream, (const T*)gy.data_ptr(),
                       (const T*)y.data_ptr(), (T*)gx.data_ptr(), nvec);
  long long tail = n - nvec * VEC;
  if (tail > 0)
    hipLaunchKernelGGL((elu_bwd_tail<T>), dim3(1), dim3(256), 0, stream,
                       (const T*)gy.data_ptr(), (const T*)y.data_ptr(),
                       (T*)gx.data_ptr(), nvec * VEC, n);
}

}  // namespace

// Accept either NCHW-contiguous or channels_last tensors untouched (the
// math is layout-independent); only truly strided inputs get copied.
static at::Tensor densify(const at::Tensor& t) {
  if (t.is_contiguous()) return t;
  if (t.dim() == 4 && t.is_contiguous(at::MemoryFormat::ChannelsLast)) return t;
  return t.contiguous();
}

static at::Tensor densify_like(const at::Tensor& t, const at::Tensor& ref) {
  if (ref.dim() == 4 && ref.is_contiguous(at::MemoryFormat::ChannelsLast))
    return t.contiguous(at::MemoryFormat::ChannelsLast);
  return t.contiguous();
}

at::Tensor fedkit_elu_fwd(const at::Tensor& x_in) {
  auto x = densify(x_in);
  auto y = at::empty_like(x);
  if (x.numel() == 0) return y;
  DISPATCH_F32_BF16(x, "elu_fwd", {
    constexpr int VEC = 16 / sizeof(scalar_t);
    launch_elu_fwd<scalar_t, VEC>(x, y);
  });
  return y;
}

at::Tensor fedkit_elu_bwd(const at::Tensor& gy_in, const at::Tensor& y_in) {
  auto y = densify(y_in);
  auto gy = densify_like(gy_in, y);
  auto gx = at::empty_like(y);
  if (y.numel() == 0) return gx;
  TORCH_CHECK(gy.scalar_type() == y.scalar_type(), "elu_bwd dtype mismatch");
  DISPATCH_F32_BF16(y, "elu_bwd", {
    constexpr int VEC = 16 / sizeof(scalar_t);
    launch_elu_bwd<scalar_t, VEC>(gy, y, gx);
  });
  return gx;
}
