// EXPERIMENTAL -- correct but NOT routed (ops/gemm.py keeps hipBLASLt).
// Graph-timed on MI355X: these measure 10-17 us vs hipBLASLt's tuned 5-9 us
// at the bench shapes. The batched wgrad kernel's sustained ~230 TFLOP/s
// puts a realistic custom single-GEMM at ~7 us for these sizes -- the
// wgrad win came from cross-problem batching, which the fwd/dgrad chain's
// sequential dependencies cannot use. Kept with numerics tests as the
// starting point for a future deep-pipeline attempt.
//
// Small-GEMM MFMA Linear kernels (gfx950): the model's fwd / dgrad GEMMs
// at the reference micro-batch (R ~= 1024 token rows, N/K in {512..4096})
// leave hipBLASLt's tile sizes underfilling the 256-CU chip (96-128
// workgroups) at ~10 us per GEMM vs a ~2 us roofline. These kernels use
// 64x64-output tiles, ONE wave64 per tile (blockDim 64 -> 128..512 blocks
// per launch), fragments loaded straight from global (both operands are
// contraction-major for the forward pass; L2 feeds the re-reads), and
// split-K over the contraction dim when it is >= 1024 so per-wave serial
// work stays ~32 MFMA k-steps.
//
//   fwd   y[R,N] = x[R,K] @ W[N,K]^T (+bias)    contraction K
//   dgrad dx[R,K] = dy[R,N] @ W[N,K]            contraction N; W^T tiles are
//                                               transpose-staged through LDS
//
// MFMA fragment maps and the XOR-swizzled LDS idioms follow wgrad_mfma.hip;
// fp32 accumulation, bf16 RNE output (or fp32 split-K partials + a combine
// kernel). Replaces tf's XLA-fused matmuls the MI355X way (reference
// run_classifier.py model_fn -> tf.layers.dense).

#include <hip/hip_runtime.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned short ush4w;

static inline __device__ int lswz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}

static inline __device__ unsigned short pk_bf16(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  unsigned int x = c.i;
  if ((x & 0x7fffffffu) > 0x7f800000u) return (unsigned short)((x >> 16) | 0x0040u);
  return (unsigned short)((x + (((x >> 16) & 1u) + 0x7fffu)) >> 16);
}

static inline __device__ int xcd_remap(int nwg, int orig) {
  const int q = nwg / 8, r = nwg % 8, xcd = orig % 8;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
}

// ---------------------------------------------------------------------------
// forward: one wave per 64x64 y-tile, frags straight from global.
// kbeg/kend give this launch's K-range (split-K): F32OUT writes a fp32
// partial slab (no bias), else bf16 y (+bias) via an LDS bounce.
// ---------------------------------------------------------------------------
template <bool F32OUT>
static __device__ __forceinline__ void lin_fwd_body(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ w,
    const unsigned short* __restrict__ bias, void* __restrict__ out,
    int R, int N, int K, int kbeg, int kend) {
  const int id = xcd_remap(gridDim.x, blockIdx.x);
  const int tj = id % (N / 64), ti = id / (N / 64);
  const int i0 = ti * 64, j0 = tj * 64;
  const int lane = threadIdx.x;
  const int lo31 = lane & 31;
  const int hi = (lane >> 5) & 1;

  // Both panels are staged K-major (no transpose) through swizzled LDS:
  // direct-from-global fragments would make every lane touch a different
  // row (64 cache lines per 1 KB fragment); the stage loads are 16 B
  // row-runs instead, and fragments then read LDS. Single-buffered with
  // register prefetch of the next chunk (wgrad_mfma.hip 256 variant).
  __shared__ __attribute__((aligned(16))) unsigned short lx[64 * 64];
  __shared__ __attribute__((aligned(16))) unsigned short lw[64 * 64];

  // stage sub-blocks: 2 iters x (4 rows x 8 cols) per lane
  const int sr0a = ((lane + 0) / 8) * 4, sca = ((lane + 0) % 8) * 8;
  const int sr0b = ((lane + 64) / 8) * 4, scb = ((lane + 64) % 8) * 8;
  bf16x8 px[2][4], pw[2][4];

#define FW_ISSUE(kc)                                                           \
  {                                                                            \
    const unsigned short* xp = x + (long long)i0 * K + (kc);                   \
    const unsigned short* wp = w + (long long)j0 * K + (kc);                   \
    _Pragma("unroll") for (int t = 0; t < 4; ++t) {                            \
      px[0][t] = *(const bf16x8*)(xp + (long long)(sr0a + t) * K + sca);       \
      px[1][t] = *(const bf16x8*)(xp + (long long)(sr0b + t) * K + scb);       \
      pw[0][t] = *(const bf16x8*)(wp + (long long)(sr0a + t) * K + sca);       \
      pw[1][t] = *(const bf16x8*)(wp + (long long)(sr0b + t) * K + scb);       \
    }                                                                          \
  }

#define FW_WRITE()                                                             \
  {                                                                            \
    _Pragma("unroll") for (int it = 0; it < 2; ++it) {                         \
      const int r0 = it == 0 ? sr0a : sr0b;                                    \
      const int cc = it == 0 ? sca : scb;                                      \
      _Pragma("unroll") for (int t = 0; t < 4; ++t) {                          \
        *(bf16x8*)((char*)lx + lswz(r0 + t, cc * 2)) = px[it][t];              \
        *(bf16x8*)((char*)lw + lswz(r0 + t, cc * 2)) = pw[it][t];              \
      }                                                                        \
    }                                                                          \
  }

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.f);

  FW_ISSUE(kbeg);
  for (int kc = kbeg; kc < kend; kc += 64) {
    __syncthreads();
    FW_WRITE();
    if (kc + 64 < kend) FW_ISSUE(kc + 64);
    __syncthreads();
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      bf16x8 a0 = *(const bf16x8*)((char*)lx + lswz(lo31, s * 32 + hi * 16));
      bf16x8 a1 = *(const bf16x8*)((char*)lx + lswz(32 + lo31, s * 32 + hi * 16));
      bf16x8 b0 = *(const bf16x8*)((char*)lw + lswz(lo31, s * 32 + hi * 16));
      bf16x8 b1 = *(const bf16x8*)((char*)lw + lswz(32 + lo31, s * 32 + hi * 16));
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
  }
#undef FW_ISSUE
#undef FW_WRITE

  if (F32OUT) {
    float* po = (float*)out;
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int col = j0 + j * 32 + lo31;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int row = i0 + i * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          po[(long long)row * N + col] = acc[i][j][r];
        }
      }
    return;
  }

  // bf16 epilogue: bias add + RNE pack, bounced through swizzled LDS so
  // global stores are full 128 B rows.
  unsigned short* ytile = lx;
  float bv[2] = {0.f, 0.f};
  if (bias) {
    bv[0] = (float)*(const __bf16*)(bias + j0 + lo31);
    bv[1] = (float)*(const __bf16*)(bias + j0 + 32 + lo31);
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int jj = j * 32 + lo31;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int ii = i * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        *(unsigned short*)((char*)ytile + lswz(ii, jj * 2)) =
            pk_bf16(acc[i][j][r] + bv[j]);
      }
    }
  __syncthreads();
  {
    unsigned short* yrow =
        (unsigned short*)out + (long long)(i0 + lane) * N + j0;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      uint4 v = *(const uint4*)((char*)ytile + lswz(lane, c * 16));
      *(uint4*)(yrow + c * 8) = v;
    }
  }
}

extern "C" __global__ __launch_bounds__(64) void k_lin_fwd_bf16(
    const unsigned short* x, const unsigned short* w, const unsigned short* bias,
    unsigned short* y, int R, int N, int K) {
  lin_fwd_body<false>(x, w, bias, y, R, N, K, 0, K);
}

extern "C" __global__ __launch_bounds__(64) void k_lin_fwd_f32(
    const unsigned short* x, const unsigned short* w, float* part,
    int R, int N, int K, int S) {
  // split s = blockIdx.y owns K-range [s*K/S, (s+1)*K/S) and slab part + s*R*N
  const int s = blockIdx.y;
  const int klen = K / S;
  lin_fwd_body<true>(x, w, nullptr, part + (long long)s * R * N, R, N, K,
                     s * klen, (s + 1) * klen);
}

// ---------------------------------------------------------------------------
// dgrad: dx[R,K] = dy[R,N] @ W[N,K]. dy frags straight from global (N-major
// rows); W^T tiles ([64 n][64 k] -> LDS [64 k][64 n]) double-buffer-staged
// with the 4-row-pack transpose. One wave per 64x64 dx-tile.
// ---------------------------------------------------------------------------
template <bool F32OUT>
static __device__ __forceinline__ void lin_dgrad_body(
    const unsigned short* __restrict__ dy, const unsigned short* __restrict__ w,
    void* __restrict__ out, int R, int N, int K, int nbeg, int nend) {
  const int id = xcd_remap(gridDim.x, blockIdx.x);
  const int tk = id % (K / 64), ti = id / (K / 64);
  const int i0 = ti * 64, k0 = tk * 64;
  const int lane = threadIdx.x;
  const int lo31 = lane & 31;
  const int hi = (lane >> 5) & 1;

  // dy panel staged straight ([64 i][64 n]); W panel transpose-staged
  // ([64 n][64 k] -> [64 k][64 n]) with the 4-row-pack.
  __shared__ __attribute__((aligned(16))) unsigned short ldy[64 * 64];
  __shared__ __attribute__((aligned(16))) unsigned short lwt[64 * 64];

  const int sr0a = ((lane + 0) / 8) * 4, sca = ((lane + 0) % 8) * 8;
  const int sr0b = ((lane + 64) / 8) * 4, scb = ((lane + 64) % 8) * 8;
  bf16x8 pdy[2][4], pw[2][4];

#define DG_ISSUE(nb)                                                           \
  {                                                                            \
    const unsigned short* dp = dy + (long long)i0 * N + (nb);                  \
    const unsigned short* wp = w + (long long)(nb) * K + k0;                   \
    _Pragma("unroll") for (int t = 0; t < 4; ++t) {                            \
      pdy[0][t] = *(const bf16x8*)(dp + (long long)(sr0a + t) * N + sca);      \
      pdy[1][t] = *(const bf16x8*)(dp + (long long)(sr0b + t) * N + scb);      \
      pw[0][t] = *(const bf16x8*)(wp + (long long)(sr0a + t) * K + sca);       \
      pw[1][t] = *(const bf16x8*)(wp + (long long)(sr0b + t) * K + scb);       \
    }                                                                          \
  }

#define DG_WRITE()                                                             \
  {                                                                            \
    _Pragma("unroll") for (int it = 0; it < 2; ++it) {                         \
      const int r0 = it == 0 ? sr0a : sr0b;                                    \
      const int cc = it == 0 ? sca : scb;                                      \
      _Pragma("unroll") for (int t = 0; t < 4; ++t)                            \
        *(bf16x8*)((char*)ldy + lswz(r0 + t, cc * 2)) = pdy[it][t];            \
      const unsigned short* u0 = (const unsigned short*)&pw[it][0];            \
      const unsigned short* u1 = (const unsigned short*)&pw[it][1];            \
      const unsigned short* u2 = (const unsigned short*)&pw[it][2];            \
      const unsigned short* u3 = (const unsigned short*)&pw[it][3];            \
      _Pragma("unroll") for (int c = 0; c < 8; ++c) {                          \
        ush4w pack = {u0[c], u1[c], u2[c], u3[c]};                             \
        *(ush4w*)((char*)lwt + lswz(cc + c, r0 * 2)) = pack;                   \
      }                                                                        \
    }                                                                          \
  }

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.f);

  DG_ISSUE(nbeg);
  for (int nb = nbeg; nb < nend; nb += 64) {
    __syncthreads();
    DG_WRITE();
    if (nb + 64 < nend) DG_ISSUE(nb + 64);
    __syncthreads();
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      bf16x8 a0 = *(const bf16x8*)((char*)ldy + lswz(lo31, s * 32 + hi * 16));
      bf16x8 a1 = *(const bf16x8*)((char*)ldy + lswz(32 + lo31, s * 32 + hi * 16));
      bf16x8 b0 = *(const bf16x8*)((char*)lwt + lswz(lo31, s * 32 + hi * 16));
      bf16x8 b1 = *(const bf16x8*)((char*)lwt + lswz(32 + lo31, s * 32 + hi * 16));
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
  }
#undef DG_ISSUE
#undef DG_WRITE

  if (F32OUT) {
    float* po = (float*)out;
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int col = k0 + j * 32 + lo31;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int row = i0 + i * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          po[(long long)row * K + col] = acc[i][j][r];
        }
      }
    return;
  }

  unsigned short* xtile = ldy;
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int jj = j * 32 + lo31;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int ii = i * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        *(unsigned short*)((char*)xtile + lswz(ii, jj * 2)) =
            pk_bf16(acc[i][j][r]);
      }
    }
  __syncthreads();
  {
    unsigned short* orow =
        (unsigned short*)out + (long long)(i0 + lane) * K + k0;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      uint4 v = *(const uint4*)((char*)xtile + lswz(lane, c * 16));
      *(uint4*)(orow + c * 8) = v;
    }
  }
}

extern "C" __global__ __launch_bounds__(64) void k_lin_dgrad_bf16(
    const unsigned short* dy, const unsigned short* w, unsigned short* dx,
    int R, int N, int K) {
  lin_dgrad_body<false>(dy, w, dx, R, N, K, 0, N);
}

extern "C" __global__ __launch_bounds__(64) void k_lin_dgrad_f32(
    const unsigned short* dy, const unsigned short* w, float* part,
    int R, int N, int K, int S) {
  const int s = blockIdx.y;
  const int nlen = N / S;
  lin_dgrad_body<true>(dy, w, part + (long long)s * R * K, R, N, K,
                       s * nlen, (s + 1) * nlen);
}

// ---------------------------------------------------------------------------
// split-K combine: y[t] = bf16( sum_s part[s][t] (+ bias[t % C]) )
// ---------------------------------------------------------------------------
extern "C" __global__ void k_splitk_combine(
    const float* __restrict__ part, int S, long long total, int C,
    const unsigned short* __restrict__ bias, unsigned short* __restrict__ y) {
  const long long stride = (long long)gridDim.x * blockDim.x * 4;
  for (long long t = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       t < total; t += stride) {
    float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
    for (int s = 0; s < S; ++s) {
      const float4 v = *(const float4*)(part + (long long)s * total + t);
      s0 += v.x; s1 += v.y; s2 += v.z; s3 += v.w;
    }
    if (bias) {
      const int c = (int)(t % C);
      const ush4w b = *(const ush4w*)(bias + c);
      union { unsigned int i; float f; } cv;
      cv.i = (unsigned int)b[0] << 16; s0 += cv.f;
      cv.i = (unsigned int)b[1] << 16; s1 += cv.f;
      cv.i = (unsigned int)b[2] << 16; s2 += cv.f;
      cv.i = (unsigned int)b[3] << 16; s3 += cv.f;
    }
    ush4w o = {pk_bf16(s0), pk_bf16(s1), pk_bf16(s2), pk_bf16(s3)};
    *(ush4w*)(y + t) = o;
  }
}
