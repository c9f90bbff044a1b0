#include "hip/hip_runtime.h"
// Fused cross-entropy (log-softmax + NLL, mean reduction) fwd/bwd.
// Replaces nn.CrossEntropyLoss dispatch (federated_multi.py:130-132;
// SURVEY.md §2a "CrossEntropyLoss").  CIFAR shapes: [batch<=1024, 10] — one
// thread per row, serial loop over the small class dim, fp32 math for any
// input dtype, one atomicAdd per row into the scalar loss.

#include "fedkit_common.h"

namespace {

template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const long* __restrict__ labels,
                              float* __restrict__ loss_sum,
                              float* __restrict__ lse_out,
                              int B, int C) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  const T* row = logits + (long long)i * C;
  float m = -INFINITY;
  for (int c = 0; c < C; ++c) m = fmaxf(m, to_f32(row[c]));
  float s = 0.f;
  for (int c = 0; c < C; ++c) s += __expf(to_f32(row[c]) - m);
  float lse = m + __logf(s);
  lse_out[i] = lse;
  float li = lse - to_f32(row[labels[i]]);
  atomicAdd(loss_sum, li / B);
}

template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const long* __restrict__ labels,
                              const float* __restrict__ lse,
                              const float* __restrict__ gloss,
                              T* __restrict__ gx, int B, int C) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  const T* row = logits + (long long)i * C;
  T* grow = gx + (long long)i * C;
  float g = gloss[0] / B;
  float l = lse[i];
  long lab = labels[i];
  for (int c = 0; c < C; ++c) {
    float p = __expf(to_f32(row[c]) - l);
    from_f32(g * (p - (c == lab ? 1.f : 0.f)), grow[c]);
  }
}

}  // namespace

std::vector<at::Tensor> fedkit_cross_entropy_fwd(const at::Tensor& logits_in,
                                                 const at::Tensor& labels) {
  auto logits = logits_in.contiguous();
  TORCH_CHECK(logits.dim() == 2, "ce expects [B, C] logits");
  TORCH_CHECK(labels.scalar_type() == at::kLong, "ce expects int64 labels");
  int B = logits.size(0), C = logits.size(1);
  auto opts = logits.options().dtype(at::kFloat);
  auto loss = at::zeros({}, opts);
  auto lse = at::empty({B}, opts);
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(logits, "ce_fwd", {
    hipLaunchKernelGGL((ce_fwd_kernel<scalar_t>),
                       dim3((B + 255) / 256), dim3(256), 0, stream,
                       (const scalar_t*)logits.data_ptr(),
                       labels.contiguous().data_ptr<long>(),
                       loss.data_ptr<float>(), lse.data_ptr<float>(), B, C);
  });
  return {loss, lse};
}

at::Tensor fedkit_cross_entropy_bwd(const at::Tensor& logits_in,
                                    const at::Tensor& labels,
                                    const at::Tensor& lse,
                                    const at::Tensor& gloss) {
  auto logits = logits_in.contiguous();
  int B = logits.size(0), C = logits.size(1);
  auto gx = at::empty_like(logits);
  auto g = gloss.to(at::kFloat).contiguous();
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(logits, "ce_bwd", {
    hipLaunchKernelGGL((ce_bwd_kernel<scalar_t>),
                       dim3((B + 255) / 256), dim3(256), 0, stream,
                       (const scalar_t*)logits.data_ptr(),
                       labels.contiguous().data_ptr<long>(),
                       lse.data_ptr<float>(), g.data_ptr<float>(),
                       (scalar_t*)gx.data_ptr(), B, C);
  });
  return gx;
}
