// Fused VAE / VAE-CL / InfoNCE loss kernels (SURVEY.md §2a rows "KLD /
// ELBO terms, cost1/2/21/3" and "InfoNCE similarity matrix").
//
// The reference computes these with Python loops over the batch
// (federated_vae_cl.py:101-140) and an O(p^4) torch.dot loop
// (federated_cpc.py:165-178).  Round 1 vectorized them into torch
// reductions; this file replaces those with single HIP kernels:
//
//   vae_elbo_fwd/bwd:     MSE(sum) + analytic KLD in ONE reduction pass
//                         (federated_vae.py:97-108).
//   vaecl_terms_fwd/bwd:  per-(cluster, sample) reductions
//                           R1[ci,b] = sum_d (x-mu_th)^2/(2 s_th)
//                                      + 0.5 log(2 pi s_th)      (cost1)
//                           R3[ci,b] = sum_d s_q/s_p - log(s_q/s_p)
//                                      + (mu_p-mu_q)^2/s_p - 1   (2*cost3)
//                         one workgroup per (ci,b) row; the tiny pk
//                         weighting/cost2/cost21 stay in torch (they carry
//                         the ekhat autograd).
//   info_nce_fwd/bwd:     column-normalize Z/Zhat [D,P], zz = Zn^T Zhatn,
//                         row softmax, -sum log(diag + 1e-6) — one
//                         workgroup (P <= 64, D arbitrary); backward is one
//                         elementwise pass using the identity
//                         t_i = sum_j dzz[i,j] zz[i,j] for the norm chain.
//
// All math fp32 regardless of input dtype (bf16 activations upcast on
// load), matching the fp32 torch reference within roundoff.

#include "fedkit_common.h"

namespace {

// ---------------------------------------------------------------- reductions

__device__ __forceinline__ float wg_reduce_sum(float v, float* red) {
  // wave shuffle then cross-wave LDS combine; 256 threads = 4 waves.
  // Trailing barrier lets the caller reuse `red` for a second reduction.
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, 64);
  int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) red[wave] = v;
  __syncthreads();
  float s = red[0] + red[1] + red[2] + red[3];
  __syncthreads();
  return s;
}

// --------------------------------------------------------------- plain VAE

template <typename T>
__global__ void vae_elbo_fwd_kernel(const T* __restrict__ recon,
                                    const T* __restrict__ x,
                                    const T* __restrict__ mu,
                                    const T* __restrict__ logvar,
                                    float* __restrict__ out,
                                    long long nrec, long long nlat) {
  __shared__ float red[4];
  float acc = 0.f;
  for (long long i = blockIdx.x * blockDim.x + threadIdx.x; i < nrec;
       i += (long long)gridDim.x * blockDim.x) {
    float d = to_f32(recon[i]) - to_f32(x[i]);
    acc += d * d;
  }
  // KLD part: -0.5 sum(1 + logvar - mu^2 - exp(logvar))
  for (long long i = blockIdx.x * blockDim.x + threadIdx.x; i < nlat;
       i += (long long)gridDim.x * blockDim.x) {
    float m = to_f32(mu[i]), lv = to_f32(logvar[i]);
    acc += -0.5f * (1.f + lv - m * m - __expf(lv));
  }
  float s = wg_reduce_sum(acc, red);
  if (threadIdx.x == 0) atomicAdd(out, s);
}

template <typename T>
__global__ void vae_elbo_bwd_kernel(const T* __restrict__ recon,
                                    const T* __restrict__ x,
                                    const T* __restrict__ mu,
                                    const T* __restrict__ logvar,
                                    const float* __restrict__ gloss,
                                    T* __restrict__ grecon,
                                    T* __restrict__ gmu,
                                    T* __restrict__ glogvar,
                                    long long nrec, long long nlat) {
  float g = gloss[0];
  for (long long i = blockIdx.x * blockDim.x + threadIdx.x; i < nrec;
       i += (long long)gridDim.x * blockDim.x)
    from_f32(g * 2.f * (to_f32(recon[i]) - to_f32(x[i])), grecon[i]);
  for (long long i = blockIdx.x * blockDim.x + threadIdx.x; i < nlat;
       i += (long long)gridDim.x * blockDim.x) {
    float m = to_f32(mu[i]), lv = to_f32(logvar[i]);
    from_f32(g * m, gmu[i]);
    from_f32(g * (-0.5f) * (1.f - __expf(lv)), glogvar[i]);
  }
}

// ------------------------------------------------------------ VAE-CL terms

template <typename T>
__global__ void vaecl_terms_fwd_kernel(
    const T* __restrict__ x,          // [B, D1]
    const T* __restrict__ mu_th,      // [KcB, D1]
    const T* __restrict__ s_th,       // [KcB, D1]
    const T* __restrict__ mu_q,       // [KcB, D2]
    const T* __restrict__ s_q,        // [KcB, D2]
    const T* __restrict__ mu_p,       // [KcB, D2]
    const T* __restrict__ s_p,        // [KcB, D2]
    float* __restrict__ R1,           // [KcB]
    float* __restrict__ R3,           // [KcB]
    int B, long long D1, long long D2) {
  __shared__ float red[4];
  const long long row = blockIdx.x;          // ci*B + b
  const long long b = row % B;
  constexpr float kLog2Pi = 1.8378770664093453f;  // log(2*pi)

  float a1 = 0.f;
  const T* mt = mu_th + row * D1;
  const T* st = s_th + row * D1;
  const T* xr = x + b * D1;
  for (long long d = threadIdx.x; d < D1; d += blockDim.x) {
    float s2 = to_f32(st[d]);
    float df = to_f32(xr[d]) - to_f32(mt[d]);
    a1 += df * df / (2.f * s2) + 0.5f * (__logf(s2) + kLog2Pi);
  }
  float r1 = wg_reduce_sum(a1, red);
  if (threadIdx.x == 0) R1[row] = r1;
  __syncthreads();

  float a3 = 0.f;
  const T* mq = mu_q + row * D2;
  const T* sq = s_q + row * D2;
  const T* mp = mu_p + row * D2;
  const T* sp = s_p + row * D2;
  for (long long d = threadIdx.x; d < D2; d += blockDim.x) {
    float q = to_f32(sq[d]), p = to_f32(sp[d]);
    float ratio = q / p;
    float md = to_f32(mp[d]) - to_f32(mq[d]);
    a3 += ratio - __logf(ratio) + md * md / p - 1.f;
  }
  float r3 = wg_reduce_sum(a3, red);
  if (threadIdx.x == 0) R3[row] = r3;
}

template <typename T>
__global__ void vaecl_terms_bwd_kernel(
    const T* __restrict__ x, const T* __restrict__ mu_th,
    const T* __restrict__ s_th, const T* __restrict__ mu_q,
    const T* __restrict__ s_q, const T* __restrict__ mu_p,
    const T* __restrict__ s_p,
    const float* __restrict__ gR1,     // [KcB]
    const float* __restrict__ gR3,     // [KcB]
    T* __restrict__ gmu_th, T* __restrict__ gs_th,
    T* __restrict__ gmu_q, T* __restrict__ gs_q,
    T* __restrict__ gmu_p, T* __restrict__ gs_p,
    int B, long long D1, long long D2) {
  const long long row = blockIdx.x;
  const long long b = row % B;
  const float g1 = gR1[row];
  const float g3 = gR3[row];
  const T* mt = mu_th + row * D1;
  const T* st = s_th + row * D1;
  const T* xr = x + b * D1;
  for (long long d = threadIdx.x; d < D1; d += blockDim.x) {
    float s2 = to_f32(st[d]);
    float df = to_f32(xr[d]) - to_f32(mt[d]);   // x - mu
    from_f32(g1 * (-df / s2), gmu_th[row * D1 + d]);
    from_f32(g1 * (-df * df / (2.f * s2 * s2) + 0.5f / s2),
             gs_th[row * D1 + d]);
  }
  const T* mq = mu_q + row * D2;
  const T* sq = s_q + row * D2;
  const T* mp = mu_p + row * D2;
  const T* sp = s_p + row * D2;
  for (long long d = threadIdx.x; d < D2; d += blockDim.x) {
    float q = to_f32(sq[d]), p = to_f32(sp[d]);
    float md = to_f32(mp[d]) - to_f32(mq[d]);   // mu_p - mu_q
    from_f32(g3 * (-2.f * md / p), gmu_q[row * D2 + d]);
    from_f32(g3 * (1.f / p - 1.f / q), gs_q[row * D2 + d]);
    from_f32(g3 * (2.f * md / p), gmu_p[row * D2 + d]);
    from_f32(g3 * (-q / (p * p) + 1.f / p - md * md / (p * p)),
             gs_p[row * D2 + d]);
  }
}

// ----------------------------------------------------------------- InfoNCE

constexpr int kMaxP = 64;

template <typename T>
__global__ void info_nce_fwd_kernel(const T* __restrict__ Z,     // [D, P]
                                    const T* __restrict__ Zhat,  // [D, P]
                                    float* __restrict__ loss,    // scalar
                                    float* __restrict__ zz_out,  // [P, P]
                                    float* __restrict__ soft,    // [P, P]
                                    float* __restrict__ norms,   // [2, P]
                                    long long D, int P) {
  // single workgroup: P <= 64, D arbitrary.  Phase 1: column norms.
  __shared__ float nz[kMaxP], nzh[kMaxP];
  __shared__ float zz[kMaxP * kMaxP];
  const int tid = threadIdx.x;
  for (int i = tid; i < P; i += blockDim.x) {
    float s1 = 0.f, s2 = 0.f;
    for (long long d = 0; d < D; ++d) {
      float a = to_f32(Z[d * P + i]);
      float b = to_f32(Zhat[d * P + i]);
      s1 += a * a;
      s2 += b * b;
    }
    nz[i] = sqrtf(s1);
    nzh[i] = sqrtf(s2);
    norms[i] = nz[i];
    norms[P + i] = nzh[i];
  }
  __syncthreads();
  // Phase 2: zz[i][j] = (Z[:,i] . Zhat[:,j]) / (nz[i] nzh[j]).
  // Each thread owns (i,j) pairs; lanes of a wave share i and read
  // consecutive j -> coalesced Zhat rows.
  for (int ij = tid; ij < P * P; ij += blockDim.x) {
    int i = ij / P, j = ij % P;
    float s = 0.f;
    for (long long d = 0; d < D; ++d)
      s += to_f32(Z[d * P + i]) * to_f32(Zhat[d * P + j]);
    float v = s / (nz[i] * nzh[j]);
    zz[ij] = v;
    zz_out[ij] = v;
  }
  __syncthreads();
  // Phase 3: row softmax, store S, add -log(diag + 1e-6) into loss.
  for (int i = tid; i < P; i += blockDim.x) {
    float m = -INFINITY;
    for (int j = 0; j < P; ++j) m = fmaxf(m, zz[i * P + j]);
    float s = 0.f;
    for (int j = 0; j < P; ++j) s += __expf(zz[i * P + j] - m);
    float diag = 0.f;
    for (int j = 0; j < P; ++j) {
      float p = __expf(zz[i * P + j] - m) / s;
      soft[i * P + j] = p;
      if (j == i) diag = p;
    }
    atomicAdd(loss, -__logf(diag + 1e-6f));
  }
}

template <typename T>
__global__ void info_nce_bwd_kernel(const T* __restrict__ Z,
                                    const T* __restrict__ Zhat,
                                    const float* __restrict__ dzz,  // [P,P]
                                    const float* __restrict__ tu,   // [2,P]
                                    const float* __restrict__ norms,  // [2,P]
                                    T* __restrict__ gZ, T* __restrict__ gZhat,
                                    long long D, int P) {
  __shared__ float sdzz[kMaxP * kMaxP];
  __shared__ float st[kMaxP], su[kMaxP], snz[kMaxP], snzh[kMaxP];
  for (int k = threadIdx.x; k < P * P; k += blockDim.x) sdzz[k] = dzz[k];
  for (int k = threadIdx.x; k < P; k += blockDim.x) {
    st[k] = tu[k];
    su[k] = tu[P + k];
    snz[k] = norms[k];
    snzh[k] = norms[P + k];
  }
  __syncthreads();
  // dZ[d,i]   = ( sum_j Zhatn[d,j] dzz[i,j]  - Zn[d,i]  t_i ) / nz[i]
  // dZhat[d,j]= ( sum_i Zn[d,i]   dzz[i,j]  - Zhatn[d,j] u_j ) / nzh[j]
  for (long long e = blockIdx.x * blockDim.x + threadIdx.x; e < D * P;
       e += (long long)gridDim.x * blockDim.x) {
    long long d = e / P;
    int i = (int)(e % P);
    float acc1 = 0.f, acc2 = 0.f;
    for (int j = 0; j < P; ++j) {
      acc1 += (to_f32(Zhat[d * P + j]) / snzh[j]) * sdzz[i * P + j];
      acc2 += (to_f32(Z[d * P + j]) / snz[j]) * sdzz[j * P + i];
    }
    float zn = to_f32(Z[d * P + i]) / snz[i];
    float zhn = to_f32(Zhat[d * P + i]) / snzh[i];
    from_f32((acc1 - zn * st[i]) / snz[i], gZ[d * P + i]);
    from_f32((acc2 - zhn * su[i]) / snzh[i], gZhat[d * P + i]);
  }
}

}  // namespace

// ------------------------------------------------------------- host wrappers

at::Tensor fedkit_vae_elbo_fwd(const at::Tensor& recon, const at::Tensor& x,
                               const at::Tensor& mu, const at::Tensor& logvar) {
  TORCH_CHECK(recon.is_contiguous() && x.is_contiguous() &&
              mu.is_contiguous() && logvar.is_contiguous(),
              "vae_elbo expects contiguous tensors");
  auto out = at::zeros({}, recon.options().dtype(at::kFloat));
  long long nrec = recon.numel(), nlat = mu.numel();
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(recon, "vae_elbo_fwd", {
    hipLaunchKernelGGL((vae_elbo_fwd_kernel<scalar_t>),
                       dim3(grid_1d(nrec, 256, 512)), dim3(256), 0, stream,
                       (const scalar_t*)recon.data_ptr(),
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)mu.data_ptr(),
                       (const scalar_t*)logvar.data_ptr(),
                       out.data_ptr<float>(), nrec, nlat);
  });
  return out;
}

std::vector<at::Tensor> fedkit_vae_elbo_bwd(const at::Tensor& recon,
                                            const at::Tensor& x,
                                            const at::Tensor& mu,
                                            const at::Tensor& logvar,
                                            const at::Tensor& gloss) {
  auto grecon = at::empty_like(recon);
  auto gmu = at::empty_like(mu);
  auto glogvar = at::empty_like(logvar);
  long long nrec = recon.numel(), nlat = mu.numel();
  auto g = gloss.to(at::kFloat).contiguous();
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(recon, "vae_elbo_bwd", {
    hipLaunchKernelGGL((vae_elbo_bwd_kernel<scalar_t>),
                       dim3(grid_1d(nrec, 256, 512)), dim3(256), 0, stream,
                       (const scalar_t*)recon.data_ptr(),
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)mu.data_ptr(),
                       (const scalar_t*)logvar.data_ptr(),
                       g.data_ptr<float>(),
                       (scalar_t*)grecon.data_ptr(),
                       (scalar_t*)gmu.data_ptr(),
                       (scalar_t*)glogvar.data_ptr(), nrec, nlat);
  });
  return {grecon, gmu, glogvar};
}

std::vector<at::Tensor> fedkit_vaecl_terms_fwd(
    const at::Tensor& x, const at::Tensor& mu_th, const at::Tensor& s_th,
    const at::Tensor& mu_q, const at::Tensor& s_q, const at::Tensor& mu_p,
    const at::Tensor& s_p, long B) {
  TORCH_CHECK(mu_th.is_contiguous() && s_th.is_contiguous() &&
              mu_q.is_contiguous() && s_q.is_contiguous() &&
              mu_p.is_contiguous() && s_p.is_contiguous() &&
              x.is_contiguous(), "vaecl_terms expects contiguous tensors");
  long long KcB = mu_th.size(0);
  long long D1 = mu_th.numel() / KcB;
  long long D2 = mu_q.numel() / KcB;
  TORCH_CHECK(x.numel() == B * D1, "x/mu_th shape mismatch");
  auto R1 = at::empty({KcB}, x.options().dtype(at::kFloat));
  auto R3 = at::empty({KcB}, x.options().dtype(at::kFloat));
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(mu_th, "vaecl_terms_fwd", {
    hipLaunchKernelGGL((vaecl_terms_fwd_kernel<scalar_t>),
                       dim3((unsigned)KcB), dim3(256), 0, stream,
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)mu_th.data_ptr(),
                       (const scalar_t*)s_th.data_ptr(),
                       (const scalar_t*)mu_q.data_ptr(),
                       (const scalar_t*)s_q.data_ptr(),
                       (const scalar_t*)mu_p.data_ptr(),
                       (const scalar_t*)s_p.data_ptr(),
                       R1.data_ptr<float>(), R3.data_ptr<float>(),
                       (int)B, D1, D2);
  });
  return {R1, R3};
}

std::vector<at::Tensor> fedkit_vaecl_terms_bwd(
    const at::Tensor& x, const at::Tensor& mu_th, const at::Tensor& s_th,
    const at::Tensor& mu_q, const at::Tensor& s_q, const at::Tensor& mu_p,
    const at::Tensor& s_p, const at::Tensor& gR1, const at::Tensor& gR3,
    long B) {
  long long KcB = mu_th.size(0);
  long long D1 = mu_th.numel() / KcB;
  long long D2 = mu_q.numel() / KcB;
  auto gmu_th = at::empty_like(mu_th);
  auto gs_th = at::empty_like(s_th);
  auto gmu_q = at::empty_like(mu_q);
  auto gs_q = at::empty_like(s_q);
  auto gmu_p = at::empty_like(mu_p);
  auto gs_p = at::empty_like(s_p);
  auto g1 = gR1.to(at::kFloat).contiguous();
  auto g3 = gR3.to(at::kFloat).contiguous();
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(mu_th, "vaecl_terms_bwd", {
    hipLaunchKernelGGL((vaecl_terms_bwd_kernel<scalar_t>),
                       dim3((unsigned)KcB), dim3(256), 0, stream,
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)mu_th.data_ptr(),
                       (const scalar_t*)s_th.data_ptr(),
                       (const scalar_t*)mu_q.data_ptr(),
                       (const scalar_t*)s_q.data_ptr(),
                       (const scalar_t*)mu_p.data_ptr(),
                       (const scalar_t*)s_p.data_ptr(),
                       g1.data_ptr<float>(), g3.data_ptr<float>(),
                       (scalar_t*)gmu_th.data_ptr(),
                       (scalar_t*)gs_th.data_ptr(),
                       (scalar_t*)gmu_q.data_ptr(),
                       (scalar_t*)gs_q.data_ptr(),
                       (scalar_t*)gmu_p.data_ptr(),
                       (scalar_t*)gs_p.data_ptr(), (int)B, D1, D2);
  });
  return {gmu_th, gs_th, gmu_q, gs_q, gmu_p, gs_p};
}

std::vector<at::Tensor> fedkit_info_nce_fwd(const at::Tensor& Z,
                                            const at::Tensor& Zhat) {
  TORCH_CHECK(Z.dim() == 2 && Z.sizes() == Zhat.sizes(),
              "info_nce expects matching [D, P]");
  long long D = Z.size(0);
  int P = (int)Z.size(1);
  TORCH_CHECK(P <= kMaxP, "info_nce kernel supports P <= ", kMaxP);
  auto Zc = Z.contiguous();
  auto Zhc = Zhat.contiguous();
  auto opts = Z.options().dtype(at::kFloat);
  auto loss = at::zeros({}, opts);
  auto zz = at::empty({P, P}, opts);
  auto soft = at::empty({P, P}, opts);
  auto norms = at::empty({2, P}, opts);
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(Zc, "info_nce_fwd", {
    hipLaunchKernelGGL((info_nce_fwd_kernel<scalar_t>), dim3(1), dim3(256),
                       0, stream, (const scalar_t*)Zc.data_ptr(),
                       (const scalar_t*)Zhc.data_ptr(),
                       loss.data_ptr<float>(), zz.data_ptr<float>(),
                       soft.data_ptr<float>(), norms.data_ptr<float>(),
                       D, P);
  });
  return {loss, zz, soft, norms};
}

std::vector<at::Tensor> fedkit_info_nce_bwd(const at::Tensor& Z,
                                            const at::Tensor& Zhat,
                                            const at::Tensor& dzz,
                                            const at::Tensor& tu,
                                            const at::Tensor& norms) {
  long long D = Z.size(0);
  int P = (int)Z.size(1);
  auto Zc = Z.contiguous();
  auto Zhc = Zhat.contiguous();
  auto gZ = at::empty_like(Zc);
  auto gZh = at::empty_like(Zhc);
  auto stream = fedkit_stream();
  DISPATCH_F32_BF16(Zc, "info_nce_bwd", {
    hipLaunchKernelGGL((info_nce_bwd_kernel<scalar_t>),
                       dim3(grid_1d(D * P, 256, 1024)), dim3(256), 0, stream,
                       (const scalar_t*)Zc.data_ptr(),
                       (const scalar_t*)Zhc.data_ptr(),
                       dzz.contiguous().data_ptr<float>(),
                       tu.contiguous().data_ptr<float>(),
                       norms.contiguous().data_ptr<float>(),
                       (scalar_t*)gZ.data_ptr(), (scalar_t*)gZh.data_ptr(),
                       D, P);
  });
  return {gZ, gZh};
}
