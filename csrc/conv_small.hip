// Direct NHWC conv for tiny input-channel counts (ResNet conv1: C=3,
// simple_models.py:191).
//
// Two paths:
//  * conv_c3_mfma_kernel — the flagship conv1 shape (C=3, 3x3, stride 1,
//    pad 1, Kout % 64 == 0).  The scalar direct kernel measured 102 us on
//    [128,3,32,32]->64 (instruction-bound: 27 MACs + 54 cvt per output);
//    this one puts the work on MFMA: input is pre-padded to FOUR channels
//    ([N][Hp][Wp][4], so every (r,s) filter tap is one aligned 8-B load),
//    weights are pre-packed to a zero-padded [Kout][64] k-major tile
//    (k = (r*3+s)*4 + c, taps 9*4 = 36 rounded up to the MFMA K-tile 64),
//    and each block computes a [128 m][Kout<=64] output tile with a single
//    staged K-tile — no k-loop, no pipeline: zero LDS, stage taps + glds
//    weights, barrier, 2 mfma k-steps, store.
//  * conv_small_kernel — generic fallback (any C/K/R/S/stride), scalar
//    per-output-pixel with weights in LDS.

#include "fedkit_common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __hip_bfloat16 bf16;

#define GLDS16S(gptr, lptr) \
  __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(gptr), \
                                   (__attribute__((address_space(3))) void*)(lptr), 16, 0, 0)

__global__ __launch_bounds__(256)
void conv_small_kernel(const bf16* __restrict__ x,  // [N][H][W][C]
                       const bf16* __restrict__ w,  // [K][R][S][C]
                       bf16* __restrict__ y,        // [N][P][Q][K]
                       int N, int H, int W, int C, int K, int R, int S,
                       int P, int Q, int stride, int pad) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* wl = (bf16*)smem;
  int wtot = K * R * S * C;
  for (int i = threadIdx.x; i < wtot; i += blockDim.x) wl[i] = w[i];
  __syncthreads();

  int pixels_per_block = blockDim.x / K;      // K <= 256
  int k = threadIdx.x % K;
  int mloc = threadIdx.x / K;
  long long M = (long long)N * P * Q;
  for (long long m = (long long)blockIdx.x * pixels_per_block + mloc; m < M;
       m += (long long)gridDim.x * pixels_per_block) {
    int q = (int)(m % Q);
    int p = (int)((m / Q) % P);
    int n = (int)(m / ((long long)P * Q));
    float acc = 0.f;
    for (int r = 0; r < R; ++r) {
      int h = p * stride + r - pad;
      if (h < 0 || h >= H) continue;
      for (int s = 0; s < S; ++s) {
        int wcol = q * stride + s - pad;
        if (wcol < 0 || wcol >= W) continue;
        const bf16* xr = x + (((long long)n * H + h) * W + wcol) * C;
        const bf16* wr = wl + ((k * R + r) * S + s) * C;
        for (int c = 0; c < C; ++c)
          acc += __bfloat162float(xr[c]) * __bfloat162float(wr[c]);
      }
    }
    y[m * K + k] = __float2bfloat16(acc);
  }
}

// ---------------------------------------------------------------- C=3 MFMA

// [N][H][W][3] -> zero-padded [N][H+2p][W+2p][4] (channel 3 = 0)
__global__ void pad_c3to4_kernel(const bf16* __restrict__ x,
                                 bf16* __restrict__ xp, int N, int H, int W,
                                 int Hp, int Wp, int pad) {
  typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
  long long total = (long long)N * Hp * Wp;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int wp = (int)(i % Wp);
    long long t = i / Wp;
    int hp = (int)(t % Hp);
    int n = (int)(t / Hp);
    int h = hp - pad, w = wp - pad;
    bf16x4 v = {};
    if (h >= 0 && h < H && w >= 0 && w < W) {
      const __bf16* src =
          reinterpret_cast<const __bf16*>(x) + (((long long)n * H + h) * W + w) * 3;
      v[0] = src[0];
      v[1] = src[1];
      v[2] = src[2];
    }
    *(bf16x4*)(reinterpret_cast<__bf16*>(xp) + i * 4) = v;
  }
}

// [K][3][3][3] channels_last weights -> [K][64] k-major, k = (r*3+s)*4 + c,
// zeros at c == 3 and k >= 36
__global__ void pack_w_c3_kernel(const bf16* __restrict__ w,
                                 bf16* __restrict__ wp, int K) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= K * 64) return;
  int k = i & 63;
  int kout = i >> 6;
  int tap = k >> 2, c = k & 3;
  float v = 0.f;
  if (tap < 9 && c < 3)
    v = __bfloat162float(w[kout * 27 + tap * 3 + c]);
  wp[i] = __float2bfloat16(v);
}

// LDS byte offset of element (row, k) of a [rows][64] bf16 tile with the
// 16-B chunk XOR swizzle (same scheme as the conv2d_mfma kernels)
__device__ __forceinline__ int c3_off(int row, int k) {
  int chunk = (k >> 3) ^ (row & 7);
  return row * 128 + chunk * 16 + (k & 7) * 2;
}

__global__ __launch_bounds__(256)
void conv_c3_mfma_kernel(const bf16* __restrict__ xp,  // [N][Hp][Wp][4]
                         const bf16* __restrict__ wp,  // [K][64]
                         bf16* __restrict__ y,         // [M][K]
                         int N, int Hp, int Wp, int K, int P, int Q) {
  constexpr int BM = 128;
  __shared__ char smem[BM * 128 + 64 * 128];   // A [128][64] + B [64][64]
  char* smA = smem;
  char* smB = smem + BM * 128;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const long long M = (long long)N * P * Q;
  const int bn = blockIdx.y;                   // kout tile (K/64)

  // zero the A tile (taps only cover logical k 0..35), stage B via glds
  for (int i = tid; i < BM * 64 / 8; i += 256)
    *(bf16x8*)(smA + i * 16) = bf16x8{};
  {
    int d = tid;                               // 2 B slots: d, d+256
#pragma unroll
    for (int i = 0; i < 2; ++i, d += 256) {
      int row = d >> 3;
      int k8 = (d & 7) ^ (row & 7);            // swizzled source chunk
      GLDS16S(wp + ((long long)bn * 64 + row) * 64 + k8 * 8,
              smB + (i * 4 + wave) * 1024);
    }
  }
  __syncthreads();                             // zeros visible before taps

  // stage A: 9 taps x BM rows of 8-B (4-channel) pieces
  const long long m0 = (long long)blockIdx.x * BM;
#pragma unroll
  for (int pass = 0; pass < (BM * 9 + 255) / 256; ++pass) {
    int d = pass * 256 + tid;
    if (d < BM * 9) {
      int row = d / 9, tap = d % 9;
      long long m = m0 + row;
      if (m >= M) m = M - 1;                   // store side is m-guarded
      int q = (int)(m % Q);
      int p = (int)((m / Q) % P);
      int n = (int)(m / ((long long)P * Q));
      int r = tap / 3, s = tap % 3;
      const bf16* src =
          xp + (((long long)n * Hp + p + r) * Wp + q + s) * 4;
      // tap t = logical k chunk t>>1, half (t&1); swizzle by row
      int chunk = (tap >> 1) ^ (row & 7);
      *(uint64_t*)(smA + row * 128 + chunk * 16 + (tap & 1) * 8) =
          *(const uint64_t*)src;
    }
  }
  // drain the B-tile DMA (vmcnt) AND this wave's A-tap ds_writes
  // (lgkmcnt) before the raw barrier publishes the tiles
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  constexpr int MR = 4;                        // (BM/2)/16 frags per wave
  const int frag_row = lane & 15;
  const int frag_k = (lane >> 4) * 8;
  const int wm = (wave >> 1) * (BM / 2);
  const int wn = (wave & 1) * 32;
  f32x4 acc[MR][2] = {};
#pragma unroll
  for (int kk = 0; kk < 64; kk += 32) {
    bf16x8 a[MR], b[2];
#pragma unroll
    for (int f = 0; f < MR; ++f)
      a[f] = *(const bf16x8*)(smA + c3_off(wm + f * 16 + frag_row,
                                           kk + frag_k));
#pragma unroll
    for (int f = 0; f < 2; ++f)
      b[f] = *(const bf16x8*)(smB + c3_off(wn + f * 16 + frag_row,
                                           kk + frag_k));
#pragma unroll
    for (int fa = 0; fa < MR; ++fa)
#pragma unroll
      for (int fb = 0; fb < 2; ++fb)
        acc[fa][fb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[fa], b[fb], acc[fa][fb], 0, 0, 0);
  }

  const int col = bn * 64 + wn + frag_row;
#pragma unroll
  for (int fa = 0; fa < MR; ++fa) {
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      long long m = m0 + wm + fa * 16 + (lane >> 4) * 4 + v;
      if (m < M) {
#pragma unroll
        for (int fb = 0; fb < 2; ++fb)
          y[m * K + col + fb * 16] = __float2bfloat16(acc[fa][fb][v]);
      }
    }
  }
}

}  // namespace

at::Tensor fedkit_conv_small_fwd(const at::Tensor& x, const at::Tensor& w,
                                 long stride, long padding) {
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int K = w.size(0), R = w.size(2), S = w.size(3);
  int P = (H + 2 * (int)padding - R) / (int)stride + 1;
  int Q = (W + 2 * (int)padding - S) / (int)stride + 1;
  auto y = at::empty({N, K, P, Q},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  long long M = (long long)N * P * Q;
  auto stream = fedkit_stream();

  if (C == 3 && R == 3 && S == 3 && stride == 1 && padding == 1 &&
      K % 64 == 0) {
    int Hp = H + 2, Wp = W + 2;
    auto xp = at::empty({(long long)N * Hp * Wp * 4},
                        x.options());
    auto wpk = at::empty({K, 64}, x.options());
    hipLaunchKernelGGL(pad_c3to4_kernel,
                       dim3(grid_1d((long long)N * Hp * Wp, 256)), dim3(256),
                       0, stream, (const bf16*)x.data_ptr(),
                       (bf16*)xp.data_ptr(), N, H, W, Hp, Wp, 1);
    hipLaunchKernelGGL(pack_w_c3_kernel, dim3((K * 64 + 255) / 256),
                       dim3(256), 0, stream, (const bf16*)w.data_ptr(),
                       (bf16*)wpk.data_ptr(), K);
    dim3 grid((unsigned)((M + 127) / 128), K / 64);
    hipLaunchKernelGGL(conv_c3_mfma_kernel, grid, dim3(256), 0, stream,
                       (const bf16*)xp.data_ptr(), (const bf16*)wpk.data_ptr(),
                       (bf16*)y.data_ptr(), N, Hp, Wp, K, P, Q);
    return y;
  }

  TORCH_CHECK(K <= 256 && 256 % K == 0, "conv_small needs Kout | 256");
  int ppb = 256 / K;
  int grid = grid_1d((M + ppb - 1) / ppb, 1, 2048);
  int smem = K * R * S * C * 2;
  hipLaunchKernelGGL(conv_small_kernel, dim3(grid), dim3(256), smem, stream,
                     (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
                     (bf16*)y.data_ptr(), N, H, W, C, K, R, S, P, Q,
                     (int)stride, (int)padding);
  return y;
}
