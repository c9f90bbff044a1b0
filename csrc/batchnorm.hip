// NHWC BatchNorm2d fwd/bwd for MI355X (SURVEY.md §2a "BatchNorm2d").
// channels_last layout puts C innermost, so per-channel reductions read
// coalesced 16 B/lane vectors.  Structure:
//   fwd (training): partial-sums kernel (per-block rows, no atomics)
//     -> fused colsum+finalize (mean/invstd + running-stat update) -> apply
//     (normalize + scale/shift + optional fused residual-add / ELU).
//   fwd (eval): stats-from-running -> apply.
//   bwd: partial sums of (g, g*xhat) -> colsum -> apply
//     dx = gamma*invstd*(g - mean(g) - xhat*mean(g*xhat)), where
//     g = dy, or dy*elu'(y) fused in-kernel when the forward fused ELU
//     (kills the standalone elu_bwd launch and the g HBM round-trip; the
//     residual-branch gradient streams out of apply only when needed).
// Measured dead ends, kept out on purpose: a last-block column sum of the
// partials slab (rows land in other XCDs' L2s -> HBM latency on C lanes,
// 10-20x slower) and cross-block fp32 atomics into a shared 2C accumulator
// (same-address atomics from 8 XCDs serialize at coherence-point latency,
// ~190 us flat).  The separate wave-per-column colsum launch is ~6 us and
// is the latency floor here.
// All statistics fp32 regardless of activation dtype (bf16 data paths keep
// fp32 BN stats — SURVEY.md §7 hard part 4).  C must divide 256 or be a
// multiple of 256 (ResNet: 64/128/256/512) so each thread owns ONE channel
// across its grid-stride walk; asserted on the host side.

#include "fedkit_common.h"

namespace {

template <typename T, int VEC>
struct alignas(sizeof(T) * VEC) VecT { T v[VEC]; };

// Vectorized (16 B/lane) per-channel reduction, stage 1: each thread owns
// VEC consecutive channels fixed across its grid-stride walk (C/VEC divides
// the 256-thread block), accumulates in registers (4x unrolled so four 16-B
// loads are in flight per lane), reduces across the block via LDS, and each
// block writes its private partial row [2][C] — NO atomics; stage 2 (the
// finalize / bwd-reduce kernels) sums the <=1024 partial rows.
template <typename T, int VEC>
__global__ void bn_partials_kernel(const T* __restrict__ x, long long M,
                                   int Cv /* C/VEC */,
                                   float* __restrict__ part /* [nb][2][C] */) {
  using V = VecT<T, VEC>;
  const int C = Cv * VEC;
  const V* xv = reinterpret_cast<const V*>(x);
  __shared__ float red[256 * VEC];
  long long total = M * Cv;
  long long stride = (long long)gridDim.x * blockDim.x;
  long long i0 = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  int tid = threadIdx.x;
  int c0 = (int)(i0 % Cv) * VEC;
  float s[VEC] = {}, sq[VEC] = {};
  long long i = i0;
  for (; i + 3 * stride < total; i += 4 * stride) {
    V a0 = xv[i], a1 = xv[i + stride], a2 = xv[i + 2 * stride],
      a3 = xv[i + 3 * stride];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float v0 = to_f32(a0.v[j]), v1 = to_f32(a1.v[j]);
      float v2 = to_f32(a2.v[j]), v3 = to_f32(a3.v[j]);
      s[j] += (v0 + v1) + (v2 + v3);
      sq[j] += (v0 * v0 + v1 * v1) + (v2 * v2 + v3 * v3);
    }
  }
  for (; i < total; i += stride) {
    V a = xv[i];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float v = to_f32(a.v[j]);
      s[j] += v;
      sq[j] += v * v;
    }
  }
  const int members = 256 / Cv;
  const int g = tid % Cv;
  float* out = part + (long long)blockIdx.x * 2 * C;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    float* src = pass == 0 ? s : sq;
    __syncthreads();
#pragma unroll
    for (int j = 0; j < VEC; ++j) red[tid * VEC + j] = src[j];
    __syncthreads();
    if (tid < Cv) {
      float acc[VEC] = {};
      for (int k = 0; k < members; ++k)
#pragma unroll
        for (int j = 0; j < VEC; ++j) acc[j] += red[(g + k * Cv) * VEC + j];
#pragma unroll
      for (int j = 0; j < VEC; ++j) out[pass * C + c0 + j] = acc[j];
    }
  }
}

// stage 2: sum the per-block partial rows, then mean/invstd + running update
__global__ void bn_finalize_kernel(const float* __restrict__ part, int nb,
                                   int C, long long count, float eps,
                                   float momentum, bool training, bool track,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float* __restrict__ save_mean,
                                   float* __restrict__ save_invstd) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean, var;
  if (training) {
    float s = 0.f, sq = 0.f;
    for (int b = 0; b < nb; ++b) {
      s += part[(long long)b * 2 * C + c];
      sq += part[(long long)b * 2 * C + C + c];
    }
    mean = s / count;
    var = fmaxf(sq / count - mean * mean, 0.f);  // biased
    if (track) {
      float unbiased = count > 1 ? var * count / (count - 1) : var;
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
  } else {
    mean = running_mean[c];
    var = running_var[c];
  }
  save_mean[c] = mean;
  save_invstd[c] = rsqrtf(var + eps);
}

// stage 2: column sums of the [nb][C2] partials matrix into ws[C2].
// One WAVE per column: lane l sums rows l, l+64, ... (64 independent load
// chains), then a wave shuffle tree.  The previous single-workgroup serial
// loop over nb<=1024 rows was latency-bound at ~210 us and dominated the
// whole training step (65% of GPU time in rocprof); this form is ~5 us.
__global__ void bn_colsum_kernel(const float* __restrict__ part, int nb,
                                 int C2 /* 2*C */, float* __restrict__ ws) {
  int col = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (col >= C2) return;
  float s = 0.f;
  for (int b = lane; b < nb; b += 64) s += part[(long long)b * C2 + col];
#pragma unroll
  for (int off = 32; off; off >>= 1) s += __shfl_down(s, off, 64);
  if (lane == 0) ws[col] = s;
}

// fused stage 2 + statistics for the TRAINING forward: one wave per channel
// sums both partial columns (sum, sumsq) and lane 0 derives mean/invstd and
// updates the running stats — one launch instead of colsum + finalize.
__global__ void bn_colsum_finalize_kernel(
    const float* __restrict__ part, int nb, int C, long long count, float eps,
    float momentum, bool track, float* __restrict__ running_mean,
    float* __restrict__ running_var, float* __restrict__ save_mean,
    float* __restrict__ save_invstd) {
  int c = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (c >= C) return;
  float s = 0.f, sq = 0.f;
  for (int b = lane; b < nb; b += 64) {
    const float* row = part + (long long)b * 2 * C;
    s += row[c];
    sq += row[C + c];
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    s += __shfl_down(s, off, 64);
    sq += __shfl_down(sq, off, 64);
  }
  if (lane == 0) {
    float mean = s / count;
    float var = fmaxf(sq / count - mean * mean, 0.f);  // biased
    if (track) {
      float unbiased = count > 1 ? var * count / (count - 1) : var;
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
    save_mean[c] = mean;
    save_invstd[c] = rsqrtf(var + eps);
  }
}

// finalize from CONV-EPILOGUE partials ([Kout/64][mtiles][2][64], written
// by conv_fwd_kernel<BNPART=true> while y was still in registers): one
// wave per channel, same math as bn_colsum_finalize_kernel — BN's own
// stage-1 pass over y never runs for conv-fed layers.
__global__ void bn_conv_colsum_finalize_kernel(
    const float* __restrict__ part, int mtiles, int C, long long count,
    float eps, float momentum, bool track, float* __restrict__ running_mean,
    float* __restrict__ running_var, float* __restrict__ save_mean,
    float* __restrict__ save_invstd) {
  int c = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (c >= C) return;
  const float* base = part + (long long)(c >> 6) * mtiles * 128 + (c & 63);
  float s = 0.f, sq = 0.f;
  for (int b = lane; b < mtiles; b += 64) {
    const float* row = base + (long long)b * 128;
    s += row[0];
    sq += row[64];
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    s += __shfl_down(s, off, 64);
    sq += __shfl_down(sq, off, 64);
  }
  if (lane == 0) {
    float mean = s / count;
    float var = fmaxf(sq / count - mean * mean, 0.f);  // biased
    if (track) {
      float unbiased = count > 1 ? var * count / (count - 1) : var;
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
    save_mean[c] = mean;
    save_invstd[c] = rsqrtf(var + eps);
  }
}

// normalize + scale/shift, with optional fused residual add (RES) and ELU
// epilogue (the reference's `elu(bn(conv) [+ shortcut])` patterns,
// simple_models.py:150-153) — one pass instead of bn/add/elu separate
// kernels (each an extra HBM read+write of the activation).
// pad geometry for the apply-into-pad fusion (pad == 0 -> identity).
// BN's normalized output almost always feeds a padded conv; writing the
// PADDED image directly from the apply kernel removes the separate pad
// launch and a full read+write pass of y (round 2).
struct PadGeom {
  int H, W, Hp, Wp, pad;
};

// vector index over the unpadded domain -> index into the padded image
__device__ __forceinline__ long long pad_vec_idx(const PadGeom& pg,
                                                 long long i, int Cv) {
  long long m = i / Cv;
  int cv = (int)(i - m * Cv);
  int w = (int)(m % pg.W);
  long long t = m / pg.W;
  int h = (int)(t % pg.H);
  long long n = t / pg.H;
  return (((n * pg.Hp) + h + pg.pad) * pg.Wp + (w + pg.pad)) * Cv + cv;
}

template <typename T, int VEC, bool ELU, bool RES>
__global__ void bn_apply_kernel(const T* __restrict__ x,
                                const T* __restrict__ res, T* __restrict__ y,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                long long nvec /* PADDED count if pad>0 */,
                                int Cv /* C / VEC */, PadGeom pg,
                                PadGeom rpg /* residual geometry */) {
  using V = VecT<T, VEC>;
  const V* xv = reinterpret_cast<const V*>(x);
  const V* rv = reinterpret_cast<const V*>(res);
  V* yv = reinterpret_cast<V*>(y);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < nvec; i += (long long)gridDim.x * blockDim.x) {
    long long ix = i;
    bool border = false;
    if (pg.pad) {
      // i walks the PADDED image; borders get zeros, interior reads x
      long long mp = i / Cv;
      int cv = (int)(i - mp * Cv);
      int wp = (int)(mp % pg.Wp);
      long long t = mp / pg.Wp;
      int hp = (int)(t % pg.Hp);
      long long n = t / pg.Hp;
      int h = hp - pg.pad, w = wp - pg.pad;
      border = h < 0 || w < 0 || h >= pg.H || w >= pg.W;
      if (!border)
        ix = (((n * pg.H) + h) * pg.W + w) * Cv + cv;
    }
    V r;
    if (border) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) from_f32(0.f, r.v[j]);
      yv[i] = r;
      continue;
    }
    int c0 = (int)(ix % Cv) * VEC;
    V a = xv[ix];
    V rr;
    if (RES) rr = rv[rpg.pad ? pad_vec_idx(rpg, ix, Cv) : ix];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      int c = c0 + j;
      float v = (to_f32(a.v[j]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
      if (RES) v += to_f32(rr.v[j]);
      if (ELU) v = fedkit_elu_f(v);
      from_f32(v, r.v[j]);
    }
    yv[i] = r;
  }
}

// elu'(y) from the SAVED OUTPUT y = elu(z): dy/dz = y > 0 ? 1 : y + 1
__device__ __forceinline__ float elu_bwd_f(float y) {
  return y > 0.f ? 1.f : y + 1.f;
}

template <typename T, int VEC, bool ELU>
__global__ void bn_bwd_partials_kernel(const T* __restrict__ x,
                                       const T* __restrict__ gy,
                                       const T* __restrict__ yout, long long M,
                                       int Cv, const float* __restrict__ mean,
                                       const float* __restrict__ invstd,
                                       float* __restrict__ part /* [nb][2][C] */,
                                       PadGeom pg) {
  using V = VecT<T, VEC>;
  const int C = Cv * VEC;
  const V* xv = reinterpret_cast<const V*>(x);
  const V* gv = reinterpret_cast<const V*>(gy);
  const V* yv = reinterpret_cast<const V*>(yout);
  __shared__ float red[256 * VEC];
  long long total = M * Cv;
  long long stride = (long long)gridDim.x * blockDim.x;
  long long i0 = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  int tid = threadIdx.x;
  int c0 = (int)(i0 % Cv) * VEC;
  float m[VEC], is[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    m[j] = mean[c0 + j];
    is[j] = invstd[c0 + j];
  }
  // the SAVED y may live in a PADDED image (apply-into-pad forward);
  // gy and every gradient stay in the unpadded domain
  auto yidx = [&](long long i) {
    return pg.pad ? pad_vec_idx(pg, i, Cv) : i;
  };
  float sdy[VEC] = {}, sdyx[VEC] = {};
  long long i = i0;
  for (; i + stride < total; i += 2 * stride) {
    V a0 = xv[i], g0 = gv[i], a1 = xv[i + stride], g1 = gv[i + stride];
    V y0, y1;
    if (ELU) {
      y0 = yv[yidx(i)];
      y1 = yv[yidx(i + stride)];
    }
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float gg0 = to_f32(g0.v[j]), gg1 = to_f32(g1.v[j]);
      if (ELU) {
        gg0 *= elu_bwd_f(to_f32(y0.v[j]));
        gg1 *= elu_bwd_f(to_f32(y1.v[j]));
      }
      sdy[j] += gg0 + gg1;
      sdyx[j] += gg0 * (to_f32(a0.v[j]) - m[j]) * is[j]
               + gg1 * (to_f32(a1.v[j]) - m[j]) * is[j];
    }
  }
  for (; i < total; i += stride) {
    V a = xv[i], g = gv[i];
    V yy;
    if (ELU) yy = yv[yidx(i)];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float gg = to_f32(g.v[j]);
      if (ELU) gg *= elu_bwd_f(to_f32(yy.v[j]));
      sdy[j] += gg;
      sdyx[j] += gg * (to_f32(a.v[j]) - m[j]) * is[j];
    }
  }
  const int members = 256 / Cv;
  const int g8 = tid % Cv;
  float* out = part + (long long)blockIdx.x * 2 * C;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    float* src = pass == 0 ? sdy : sdyx;
    __syncthreads();
#pragma unroll
    for (int j = 0; j < VEC; ++j) red[tid * VEC + j] = src[j];
    __syncthreads();
    if (tid < Cv) {
      float acc[VEC] = {};
      for (int k = 0; k < members; ++k)
#pragma unroll
        for (int j = 0; j < VEC; ++j) acc[j] += red[(g8 + k * Cv) * VEC + j];
#pragma unroll
      for (int j = 0; j < VEC; ++j) out[pass * C + c0 + j] = acc[j];
    }
  }
}

// dx = gamma*invstd*(g - mean(g) - xhat*mean(g*xhat)); recomputes
// g = dy * elu'(y) when fused and optionally streams g out (the residual
// branch gradient) — WANTG costs one extra write but no extra launch.
template <typename T, int VEC, bool ELU, bool WANTG>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ x,
                                    const T* __restrict__ gy,
                                    const T* __restrict__ yout,
                                    T* __restrict__ gx, T* __restrict__ gout,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ ws,
                                    long long M, long long nvec, int Cv,
                                    PadGeom pg /* saved-y geometry */) {
  using V = VecT<T, VEC>;
  const V* xv = reinterpret_cast<const V*>(x);
  const V* gv = reinterpret_cast<const V*>(gy);
  const V* yv = reinterpret_cast<const V*>(yout);
  V* ov = reinterpret_cast<V*>(gx);
  V* gov = reinterpret_cast<V*>(gout);
  int C = Cv * VEC;
  float inv_count = 1.f / (float)M;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < nvec; i += (long long)gridDim.x * blockDim.x) {
    int c0 = (int)(i % Cv) * VEC;
    V a = xv[i], g = gv[i], r, go;
    V yy;
    if (ELU) yy = yv[pg.pad ? pad_vec_idx(pg, i, Cv) : i];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      int c = c0 + j;
      float is = invstd[c];
      float xhat = (to_f32(a.v[j]) - mean[c]) * is;
      float dy = to_f32(g.v[j]);
      if (ELU) dy *= elu_bwd_f(to_f32(yy.v[j]));
      if (WANTG) from_f32(dy, go.v[j]);
      float val = gamma[c] * is *
          (dy - ws[c] * inv_count - xhat * ws[C + c] * inv_count);
      from_f32(val, r.v[j]);
    }
    ov[i] = r;
    if (WANTG) gov[i] = go;
  }
}

void check_nhwc(const at::Tensor& x) {
  TORCH_CHECK(x.dim() == 4, "bn expects 4D input");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "bn expects channels_last input");
}

}  // namespace

std::vector<at::Tensor> fedkit_bn_fwd(const at::Tensor& x,
                                      const at::Tensor& gamma,
                                      const at::Tensor& beta,
                                      at::Tensor running_mean,
                                      at::Tensor running_var, bool training,
                                      double momentum, double eps,
                                      c10::optional<at::Tensor> residual,
                                      bool elu,
                                      c10::optional<at::Tensor> conv_part,
                                      long pad_out, long res_pad) {
  check_nhwc(x);
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  long long M = (long long)N * H * W;
  auto fopts = x.options().dtype(at::kFloat);
  auto save_mean = at::empty({C}, fopts);
  auto save_invstd = at::empty({C}, fopts);
  // pad_out > 0: the apply writes the PADDED image the next conv
  // consumes (border zeros included) — no separate pad launch/pass
  const int pad = (int)pad_out;
  at::Tensor y = pad == 0
      ? at::empty_like(x)
      : at::empty({N, C, H + 2 * pad, W + 2 * pad},
                  x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto stream = fedkit_stream();
  auto gamma_f = gamma.contiguous();
  auto beta_f = beta.contiguous();
  TORCH_CHECK(gamma_f.scalar_type() == at::kFloat, "bn gamma must be fp32");

  if (training && conv_part.has_value()) {
    // stage 1 already happened inside the producing conv's epilogue
    auto& cp = *conv_part;
    TORCH_CHECK(cp.scalar_type() == at::kFloat && cp.dim() == 4 &&
                cp.size(0) * 64 == C && cp.size(2) == 2 && cp.size(3) == 64,
                "bad conv_part shape for C=", C);
    bool track = running_mean.defined();
    int nwaves = 4;
    hipLaunchKernelGGL(bn_conv_colsum_finalize_kernel,
                       dim3((C + nwaves - 1) / nwaves), dim3(256), 0, stream,
                       cp.data_ptr<float>(), (int)cp.size(1), C, M,
                       (float)eps, (float)momentum, track,
                       track ? running_mean.data_ptr<float>() : nullptr,
                       track ? running_var.data_ptr<float>() : nullptr,
                       save_mean.data_ptr<float>(),
                       save_invstd.data_ptr<float>());
  } else if (training) {
    DISPATCH_F32_BF16(x, "bn_partials", {
      constexpr int VEC = 16 / sizeof(scalar_t);
      TORCH_CHECK(C % VEC == 0 && 256 % (C / VEC) == 0,
                  "bn kernel needs C % ", VEC, " == 0 and (C/", VEC,
                  ") | 256, got C=", C);
      int nb = grid_1d(M * C / VEC, 256, 512);
      auto part = at::empty({nb, 2, C}, fopts);
      hipLaunchKernelGGL((bn_partials_kernel<scalar_t, VEC>),
                         dim3(nb), dim3(256), 0,
                         stream, (const scalar_t*)x.data_ptr(), M, C / VEC,
                         part.data_ptr<float>());
      hipLaunchKernelGGL(bn_colsum_finalize_kernel, dim3((C + 3) / 4),
                         dim3(256), 0, stream, part.data_ptr<float>(), nb, C,
                         M, (float)eps, (float)momentum,
                         running_mean.defined(),
                         running_mean.defined() ? running_mean.data_ptr<float>() : nullptr,
                         running_var.defined() ? running_var.data_ptr<float>() : nullptr,
                         save_mean.data_ptr<float>(),
                         save_invstd.data_ptr<float>());
    });
  } else {
    hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + 255) / 256), dim3(256),
                       0, stream, (const float*)nullptr, 0, C, M, (float)eps,
                       (float)momentum, false, false,
                       running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(),
                       save_mean.data_ptr<float>(),
                       save_invstd.data_ptr<float>());
  }
  const bool has_res = residual.has_value();
  const void* res_ptr = nullptr;
  const int rpad = (int)res_pad;
  if (has_res) {
    check_nhwc(*residual);
    TORCH_CHECK(residual->scalar_type() == x.scalar_type(),
                "bn residual must match input dtype");
    TORCH_CHECK(residual->size(2) == H + 2 * rpad &&
                residual->size(3) == W + 2 * rpad,
                "bn residual geometry mismatch for res_pad=", rpad);
    res_ptr = residual->data_ptr();
  }
  DISPATCH_F32_BF16(x, "bn_apply", {
    constexpr int VEC = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % VEC == 0, "bn needs C % ", VEC, " == 0");
    PadGeom pg = {H, W, H + 2 * pad, W + 2 * pad, pad};
    PadGeom rpg = {H, W, H + 2 * rpad, W + 2 * rpad, rpad};
    long long nvec = (pad == 0 ? M
                      : (long long)N * pg.Hp * pg.Wp) * C / VEC;
    auto launch = [&](auto ekind, auto rkind) {
      hipLaunchKernelGGL(
          (bn_apply_kernel<scalar_t, VEC, decltype(ekind)::value,
                           decltype(rkind)::value>),
          dim3(grid_1d(nvec, 256)), dim3(256), 0, stream,
          (const scalar_t*)x.data_ptr(), (const scalar_t*)res_ptr,
          (scalar_t*)y.data_ptr(), save_mean.data_ptr<float>(),
          save_invstd.data_ptr<float>(), gamma_f.data_ptr<float>(),
          beta_f.data_ptr<float>(), nvec, C / VEC, pg, rpg);
    };
    using T0 = std::integral_constant<bool, false>;
    using T1 = std::integral_constant<bool, true>;
    if (elu && has_res)       launch(T1{}, T1{});
    else if (elu)             launch(T1{}, T0{});
    else if (has_res)         launch(T0{}, T1{});
    else                      launch(T0{}, T0{});
  });
  return {y, save_mean, save_invstd};
}

// elu_y: the saved post-ELU output when the forward fused ELU (elu' is
// recomputed from it inside both backward kernels); want_g additionally
// returns g = dy*elu'(y) (the residual-branch gradient) as a 4th output.
std::vector<at::Tensor> fedkit_bn_bwd(const at::Tensor& gy, const at::Tensor& x,
                                      const at::Tensor& gamma,
                                      const at::Tensor& save_mean,
                                      const at::Tensor& save_invstd,
                                      c10::optional<at::Tensor> elu_y,
                                      bool want_g, long pad_in) {
  check_nhwc(x);
  check_nhwc(gy);
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  long long M = (long long)N * H * W;
  // pad_in: the SAVED y (elu_y) lives in a padded image (apply-into-pad
  // forward); gy and every gradient stay in the UNPADDED domain
  const int pad = (int)pad_in;
  PadGeom pg = {H, W, H + 2 * pad, W + 2 * pad, pad};
  TORCH_CHECK(gy.size(2) == H && gy.size(3) == W,
              "bn_bwd: gy must be unpadded");
  TORCH_CHECK(!pad || (elu_y.has_value() && elu_y->size(2) == pg.Hp),
              "bn_bwd: elu_y geometry mismatch for pad_in=", pad);
  auto fopts = x.options().dtype(at::kFloat);
  auto ws = at::empty({2, C}, fopts);
  auto gx = at::empty_like(x);
  auto stream = fedkit_stream();
  auto gamma_f = gamma.contiguous();
  const bool elu = elu_y.has_value();
  const void* y_ptr = elu ? elu_y->data_ptr() : nullptr;
  TORCH_CHECK(!want_g || elu, "want_g requires the fused-ELU backward");
  at::Tensor gout;
  if (want_g) gout = at::empty_like(x);   // residual grad is UNPADDED
  using T0 = std::integral_constant<bool, false>;
  using T1 = std::integral_constant<bool, true>;
  DISPATCH_F32_BF16(x, "bn_bwd_partials", {
    constexpr int VEC = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % VEC == 0 && 256 % (C / VEC) == 0,
                "bn kernel needs C % ", VEC, " == 0 and (C/", VEC,
                ") | 256, got C=", C);
    int nb = grid_1d(M * C / VEC, 256, 512);
    auto part = at::empty({nb, 2, C}, fopts);
    auto launch1 = [&](auto ekind) {
      hipLaunchKernelGGL((bn_bwd_partials_kernel<scalar_t, VEC,
                                                 decltype(ekind)::value>),
                         dim3(nb), dim3(256), 0,
                         stream, (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)gy.data_ptr(),
                         (const scalar_t*)y_ptr, M, C / VEC,
                         save_mean.data_ptr<float>(),
                         save_invstd.data_ptr<float>(), part.data_ptr<float>(),
                         pg);
    };
    if (elu) launch1(T1{}); else launch1(T0{});
    hipLaunchKernelGGL(bn_colsum_kernel, dim3((2 * C + 3) / 4), dim3(256),
                       0, stream, part.data_ptr<float>(), nb, 2 * C,
                       ws.data_ptr<float>());
  });
  DISPATCH_F32_BF16(x, "bn_bwd_apply", {
    constexpr int VEC = 16 / sizeof(scalar_t);
    long long nvec = M * C / VEC;
    auto launch2 = [&](auto ekind, auto wkind) {
      hipLaunchKernelGGL((bn_bwd_apply_kernel<scalar_t, VEC,
                                              decltype(ekind)::value,
                                              decltype(wkind)::value>),
                         dim3(grid_1d(nvec, 256)), dim3(256), 0, stream,
                         (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)gy.data_ptr(),
                         (const scalar_t*)y_ptr, (scalar_t*)gx.data_ptr(),
                         want_g ? (scalar_t*)gout.data_ptr() : nullptr,
                         save_mean.data_ptr<float>(),
                         save_invstd.data_ptr<float>(),
                         gamma_f.data_ptr<float>(), ws.data_ptr<float>(), M,
                         nvec, C / VEC, pg);
    };
    if (elu && want_g)  launch2(T1{}, T1{});
    else if (elu)       launch2(T1{}, T0{});
    else                launch2(T0{}, T0{});
  });
  // gw = sum(g * xhat), gb = sum(g): zero-copy views into ws
  std::vector<at::Tensor> out = {gx, ws.select(0, 1), ws.select(0, 0)};
  if (want_g) out.push_back(gout);
  return out;
}
