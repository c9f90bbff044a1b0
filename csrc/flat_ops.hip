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

// ------------------------------------------------- L-BFGS fused multi-ops
// The two-loop recursion's 3n+1 separate torch dots each cost a launch AND
// a device->host sync (lbfgsnew.py:588-659 equivalent).  fedkit instead
// keeps the small Gram matrices (S^T Y, Y^T Y) on the host and refreshes
// them with ONE fused pass per history update: multi_dot computes x . v_i
// for up to 24 history vectors in a single read of x (partials slab +
// wave-per-column finalize, same shape as the BN stage 2), and lincomb
// materializes d = c_g * g + sum c_i v_i in one pass.  One sync per
// step() iteration instead of ~21.

namespace {

constexpr int kMaxVecs = 24;

struct VecPack {
  const float* p[kMaxVecs];
  int n;
};

struct CoefPack {
  const float* p[kMaxVecs];
  float c[kMaxVecs];
  int n;
};

// partials[b][v] = block b's partial dot of x . p[v]
__global__ void multi_dot_kernel(VecPack vp, const float* __restrict__ x,
                                 long long total,
                                 float* __restrict__ part /* [nb][nvec] */) {
  __shared__ float red[256];
  float acc[kMaxVecs] = {};
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += stride) {
    float xv = x[i];
    for (int v = 0; v < vp.n; ++v) acc[v] += xv * vp.p[v][i];
  }
  for (int v = 0; v < vp.n; ++v) {
    __syncthreads();
    red[threadIdx.x] = acc[v];
    __syncthreads();
    for (int off = 128; off; off >>= 1) {
      if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
      __syncthreads();
    }
    if (threadIdx.x == 0) part[(long long)blockIdx.x * vp.n + v] = red[0];
  }
}

// out[v] = sum_b part[b][v] — one wave per column (nb <= 640 rows)
__global__ void multi_dot_finalize_kernel(const float* __restrict__ part,
                                          int nb, int nvec,
                                          float* __restrict__ out) {
  int v = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (v >= nvec) return;
  float s = 0.f;
  for (int b = lane; b < nb; b += 64) s += part[(long long)b * nvec + v];
#pragma unroll
  for (int off = 32; off; off >>= 1) s += __shfl_down(s, off, 64);
  if (lane == 0) out[v] = s;
}

// out[i] = cg * g[i] + sum_v c[v] * p[v][i]
__global__ void lincomb_kernel(CoefPack cp, const float* __restrict__ g,
                               float cg, float* __restrict__ out,
                               long long total) {
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += stride) {
    float s = cg * g[i];
    for (int v = 0; v < cp.n; ++v) s += cp.c[v] * cp.p[v][i];
    out[i] = s;
  }
}

}  // namespace

// returns a DEVICE fp32 tensor [len(vecs)] of dots x . vecs[i]; the caller
// syncs once with .cpu() when it needs the scalars
at::Tensor fedkit_multi_dot(std::vector<at::Tensor> vecs,
                            const at::Tensor& x) {
  TORCH_CHECK(!vecs.empty() && (int)vecs.size() <= kMaxVecs,
              "multi_dot supports 1..", kMaxVecs, " vectors");
  TORCH_CHECK(x.scalar_type() == at::kFloat && x.is_contiguous(),
              "multi_dot: x must be contiguous fp32");
  VecPack vp;
  vp.n = (int)vecs.size();
  long long total = x.numel();
  for (int i = 0; i < vp.n; ++i) {
    TORCH_CHECK(vecs[i].scalar_type() == at::kFloat &&
                vecs[i].is_contiguous() && vecs[i].numel() == total,
                "multi_dot: vecs must be contiguous fp32 of x's length");
    vp.p[i] = vecs[i].data_ptr<float>();
  }
  auto stream = fedkit_stream();
  int nb = grid_1d(total, 256, 512);
  auto part = at::empty({nb, vp.n}, x.options());
  auto out = at::empty({vp.n}, x.options());
  hipLaunchKernelGGL(multi_dot_kernel, dim3(nb), dim3(256), 0, stream, vp,
                     x.data_ptr<float>(), total, part.data_ptr<float>());
  hipLaunchKernelGGL(multi_dot_finalize_kernel,
                     dim3((vp.n + 3) / 4), dim3(256), 0, stream,
                     part.data_ptr<float>(), nb, vp.n, out.data_ptr<float>());
  return out;
}

// out = cg * g + sum coeffs[i] * vecs[i]   (fused direction build)
at::Tensor fedkit_lincomb(const at::Tensor& g, double cg,
                          std::vector<at::Tensor> vecs,
                          std::vector<double> coeffs) {
  TORCH_CHECK(vecs.size() == coeffs.size() && (int)vecs.size() <= kMaxVecs,
              "lincomb: vecs/coeffs mismatch or too many");
  TORCH_CHECK(g.scalar_type() == at::kFloat && g.is_contiguous(),
              "lincomb: g must be contiguous fp32");
  CoefPack cp;
  cp.n = (int)vecs.size();
  long long total = g.numel();
  for (int i = 0; i < cp.n; ++i) {
    TORCH_CHECK(vecs[i].scalar_type() == at::kFloat &&
                vecs[i].is_contiguous() && vecs[i].numel() == total,
                "lincomb: vecs must be contiguous fp32 of g's length");
    cp.p[i] = vecs[i].data_ptr<float>();
    cp.c[i] = (float)coeffs[i];
  }
  auto out = at::empty_like(g);
  hipLaunchKernelGGL(lincomb_kernel, dim3(grid_1d(total, 256)), dim3(256), 0,
                     fedkit_stream(), cp, g.data_ptr<float>(), (float)cg,
                     out.data_ptr<float>(), total);
  return out;
}

// --------------------------------------------------- batched weight casts
// In a full-model training step every FedConv2d pays one fp32->bf16 cast
// kernel forward and one bf16->fp32 grad cast backward (~40 launches,
// ~200 us/step on ResNet18).  One descriptor-table kernel covers all of
// them per direction; layouts are preserved because dsts are allocated
// *_like their srcs and both sides are traversed flat.

namespace {

struct CastDesc {
  const void* src[kMaxTensors];
  void* dst[kMaxTensors];
  long long offset[kMaxTensors + 1];
  int n;
};

__device__ __forceinline__ int cast_find(const CastDesc& d, long long i) {
  int lo = 0, hi = d.n - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (i >= d.offset[mid]) lo = mid; else hi = mid - 1;
  }
  return lo;
}

// 8 elements per loop iteration (guide G13: 16-B/lane traffic): ONE
// boundary search per unit, vector loads/stores on the in-tensor fast
// path, scalar fallback only for units straddling a tensor boundary
// (<= n units total).  The scalar original measured 68 us for the
// ResNet18 grad-cast sweep — pure search+scalar-store overhead.
__global__ void cast_f32_to_bf16_kernel(CastDesc d, long long total) {
  typedef __attribute__((ext_vector_type(4))) float f4;
  const long long units = (total + 7) >> 3;
  for (long long u = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       u < units; u += (long long)gridDim.x * blockDim.x) {
    long long i0 = u << 3;
    int t = cast_find(d, i0);
    long long j = i0 - d.offset[t];
    if (i0 + 8 <= d.offset[t + 1]) {
      const float* s = (const float*)d.src[t] + j;
      __hip_bfloat16* p = (__hip_bfloat16*)d.dst[t] + j;
      f4 a, b;
      __builtin_memcpy(&a, s, 16);
      __builtin_memcpy(&b, s + 4, 16);
      __hip_bfloat16 o[8];
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        o[k] = __float2bfloat16(a[k]);
        o[4 + k] = __float2bfloat16(b[k]);
      }
      __builtin_memcpy(p, o, 16);
    } else {
      for (long long i = i0; i < i0 + 8 && i < total; ++i) {
        int tt = cast_find(d, i);
        long long jj = i - d.offset[tt];
        ((__hip_bfloat16*)d.dst[tt])[jj] =
            __float2bfloat16(((const float*)d.src[tt])[jj]);
      }
    }
  }
}

__global__ void cast_bf16_to_f32_kernel(CastDesc d, long long total) {
  typedef __attribute__((ext_vector_type(4))) float f4;
  const long long units = (total + 7) >> 3;
  for (long long u = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       u < units; u += (long long)gridDim.x * blockDim.x) {
    long long i0 = u << 3;
    int t = cast_find(d, i0);
    long long j = i0 - d.offset[t];
    if (i0 + 8 <= d.offset[t + 1]) {
      const __hip_bfloat16* s = (const __hip_bfloat16*)d.src[t] + j;
      float* p = (float*)d.dst[t] + j;
      __hip_bfloat16 in[8];
      __builtin_memcpy(in, s, 16);
      f4 a, b;
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        a[k] = __bfloat162float(in[k]);
        b[k] = __bfloat162float(in[4 + k]);
      }
      __builtin_memcpy(p, &a, 16);
      __builtin_memcpy(p + 4, &b, 16);
    } else {
      for (long long i = i0; i < i0 + 8 && i < total; ++i) {
        int tt = cast_find(d, i);
        long long jj = i - d.offset[tt];
        ((float*)d.dst[tt])[jj] =
            __bfloat162float(((const __hip_bfloat16*)d.src[tt])[jj]);
      }
    }
  }
}

CastDesc build_cast_desc(const std::vector<at::Tensor>& srcs,
                         const std::vector<at::Tensor>& dsts,
                         long long* total_out) {
  TORCH_CHECK(srcs.size() == dsts.size() &&
              (int)srcs.size() <= kMaxTensors,
              "cast: 1..", kMaxTensors, " tensor pairs");
  CastDesc d;
  d.n = (int)srcs.size();
  long long off = 0;
  for (int i = 0; i < d.n; ++i) {
    TORCH_CHECK(srcs[i].numel() == dsts[i].numel(), "cast: numel mismatch");
    d.src[i] = srcs[i].data_ptr();
    d.dst[i] = dsts[i].data_ptr();
    d.offset[i] = off;
    off += srcs[i].numel();
  }
  d.offset[d.n] = off;
  *total_out = off;
  return d;
}

}  // namespace

void fedkit_cast_f32_to_bf16(std::vector<at::Tensor> srcs,
                             std::vector<at::Tensor> dsts) {
  long long total;
  auto d = build_cast_desc(srcs, dsts, &total);
  hipLaunchKernelGGL(cast_f32_to_bf16_kernel, dim3(grid_1d(total, 256)),
                     dim3(256), 0, fedkit_stream(), d, total);
}

void fedkit_cast_bf16_to_f32(std::vector<at::Tensor> srcs,
                             std::vector<at::Tensor> dsts) {
  long long total;
  auto d = build_cast_desc(srcs, dsts, &total);
  hipLaunchKernelGGL(cast_bf16_to_f32_kernel, dim3(grid_1d(total, 256)),
                     dim3(256), 0, fedkit_stream(), d, total);
}

// ------------------------------------------------- fused Welford update
// LBFGS batch mode, new-minibatch statistics (SURVEY §2a "Online grad
// mean/variance (Welford) ... fused into the flat-grad kernel"; reference
// lbfgsnew.py:600-615).  One kernel replaces 2 clones + 2 axpys +
// 1 addcmul + 1 sum (~6 launches + 2 allocations):
//   d_old = g - avg;  avg += d_old/n;  d_new = g - avg;
//   avg_sq += d_new * d_old;  out = sum(avg_sq)
namespace {
__global__ void welford_update_kernel(const float* __restrict__ g,
                                      float* __restrict__ avg,
                                      float* __restrict__ avg_sq,
                                      float inv_n,
                                      float* __restrict__ out,
                                      long long n) {
  float acc = 0.f;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x) {
    float gi = g[i];
    float d_old = gi - avg[i];
    float a_new = avg[i] + d_old * inv_n;
    avg[i] = a_new;
    float sq = avg_sq[i] + (gi - a_new) * d_old;
    avg_sq[i] = sq;
    acc += sq;
  }
  for (int off = 32; off > 0; off >>= 1)
    acc += __shfl_xor(acc, off, 64);
  if ((threadIdx.x & 63) == 0) atomicAdd(out, acc);
}
}  // namespace

at::Tensor fedkit_welford_update(const at::Tensor& g, at::Tensor avg,
                                 at::Tensor avg_sq, double inv_n) {
  TORCH_CHECK(g.scalar_type() == at::kFloat && avg.scalar_type() == at::kFloat
              && avg_sq.scalar_type() == at::kFloat,
              "welford_update is fp32");
  long long n = g.numel();
  auto out = at::zeros({}, g.options());
  hipLaunchKernelGGL(welford_update_kernel, dim3(grid_1d(n, 256)), dim3(256),
                     0, fedkit_stream(), g.contiguous().data_ptr<float>(),
                     avg.data_ptr<float>(), avg_sq.data_ptr<float>(),
                     (float)inv_n, out.data_ptr<float>(), n);
  return out;
}
