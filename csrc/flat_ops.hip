// Multi-tensor flat pack/unpack/axpy — the federation/optimizer data plane.
// The reference does these as Python loops of per-tensor copies
// (simple_utils.py:47-77, lbfgsnew.py:81-121); here ONE kernel covers up to
// 48 tensors per launch via a descriptor table passed by kernel argument
// (fits the 4 KB kernarg budget), grid-stride over the total element count.
// fp32 only (parameters, gradients and federation vectors are fp32 master).

#include "fedkit_common.h"

namespace {

constexpr int kMaxTensors = 48;

struct Desc {
  const float* src[kMaxTensors];
  float* dst[kMaxTensors];
  long long offset[kMaxTensors + 1];  // prefix offsets into the flat vector
  int n;
};

__device__ __forceinline__ int find_tensor(const Desc& d, long long i) {
  // offsets are sorted; ~48 entries -> short scalar loop, mostly uniform
  int lo = 0, hi = d.n - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (i >= d.offset[mid]) lo = mid; else hi = mid - 1;
  }
  return lo;
}

// flat[off_t + j] = tensor_t[j]
__global__ void pack_kernel(Desc d, float* __restrict__ flat, long long total) {
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int t = find_tensor(d, i);
    flat[i] = d.src[t][i - d.offset[t]];
  }
}

// tensor_t[j] = flat[off_t + j]
__global__ void unpack_kernel(Desc d, const float* __restrict__ flat,
                              long long total) {
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int t = find_tensor(d, i);
    d.dst[t][i - d.offset[t]] = flat[i];
  }
}

// tensor_t[j] += alpha * flat[off_t + j]   (LBFGS _add_grad / axpy)
__global__ void axpy_kernel(Desc d, const float* __restrict__ flat, float alpha,
                            long long total) {
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int t = find_tensor(d, i);
    d.dst[t][i - d.offset[t]] += alpha * flat[i];
  }
}

enum class Op { Pack, Unpack, Axpy };

void run_chunks(const std::vector<at::Tensor>& tensors, const at::Tensor& flat,
                Op op, float alpha) {
  TORCH_CHECK(flat.is_contiguous() && flat.scalar_type() == at::kFloat,
              "flat vector must be contiguous fp32");
  long long flat_off = 0;
  auto stream = fedkit_stream();
  size_t idx = 0;
  while (idx < tensors.size()) {
    Desc d;
    d.n = 0;
    long long base = flat_off;
    while (idx < tensors.size() && d.n < kMaxTensors) {
      const at::Tensor& t = tensors[idx];
      TORCH_CHECK(t.scalar_type() == at::kFloat, "flat ops are fp32-only");
      TORCH_CHECK(t.is_non_overlapping_and_dense(),
                  "flat ops need dense tensors");
      d.offset[d.n] = flat_off - base;
      d.src[d.n] = (const float*)t.data_ptr();
      d.dst[d.n] = (float*)t.data_ptr();
      flat_off += t.numel();
      ++d.n;
      ++idx;
    }
    d.offset[d.n] = flat_off - base;
    long long total = flat_off - base;
    if (total == 0) continue;
    const float* fsrc = (const float*)flat.data_ptr() + base;
    float* fdst = (float*)flat.data_ptr() + base;
    int grid = grid_1d(total, 256);
    switch (op) {
      case Op::Pack:
        hipLaunchKernelGGL(pack_kernel, dim3(grid), dim3(256), 0, stream,
                           d, fdst, total);
        break;
      case Op::Unpack:
        hipLaunchKernelGGL(unpack_kernel, dim3(grid), dim3(256), 0, stream,
                           d, fsrc, total);
        break;
      case Op::Axpy:
        hipLaunchKernelGGL(axpy_kernel, dim3(grid), dim3(256), 0, stream,
                           d, fsrc, alpha, total);
        break;
    }
  }
  TORCH_CHECK(flat_off == flat.numel(),
              "flat vector size mismatch: ", flat.numel(), " vs ", flat_off);
}

}  // namespace

void fedkit_pack_params(std::vector<at::Tensor> tensors, at::Tensor flat) {
  run_chunks(tensors, flat, Op::Pack, 0.f);
}

void fedkit_unpack_params(at::Tensor flat, std::vector<at::Tensor> tensors) {
  run_chunks(tensors, flat, Op::Unpack, 0.f);
}

void fedkit_add_flat_params(std::vector<at::Tensor> tensors, at::Tensor flat,
                            double alpha) {
  run_chunks(tensors, flat, Op::Axpy, (float)alpha);
}
