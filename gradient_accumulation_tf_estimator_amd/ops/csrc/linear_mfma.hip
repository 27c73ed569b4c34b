// Custom MFMA Linear forward / dgrad kernels (gfx950) for the encoder's
// token-rows-GEMM shapes (R ~= 1024 rows, N/K in {512..4096}).
//
//   fwd:   y[R,N] (bf16) = x[R,K] @ W[N,K]^T (+ bias[N])
//   dgrad: dx[R,K] (bf16) = dy[R,N] @ W[N,K]
//
// Why: these shapes launch 20-60 workgroups in hipBLASLt (one block-wave on
// a 256-CU chip), so each GEMM's cost is block LATENCY, not throughput.
// A 128x128-tile kernel with the 2-phase register pipeline (issue chunk
// t+1's loads before chunk t's MFMAs -- cdna_hip_programming.md T3/T14)
// keeps that latency to a few microseconds. No split-K: underfilled
// one-wave grids are latency-bound, and partial-accumulate traffic would
// cost more than it saves at these sizes.
//
// fwd reads BOTH operands natural-row (contraction K is the contiguous dim
// of x rows and W rows); dgrad transpose-stages W chunks (contraction N is
// W's row dim) with the pack-4 transpose from wgrad_mfma.hip. Outputs
// bounce through LDS so global writes are coalesced bf16 rows.
// Fragment maps as in attn.hip.

#include <hip/hip_runtime.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned short ush4l;

static inline __device__ int lswz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}
static inline __device__ unsigned short lf2bf(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  return (unsigned short)((c.i + (((c.i >> 16) & 1u) + 0x7fffu)) >> 16);
}

// natural [128 rows][64 cols] chunk: issue/write split (256 threads)
struct LinStage {
  bf16x8 r[4];
};

static __device__ __forceinline__ void lin_issue(const unsigned short* g,
                                                 long long ld, int c0,
                                                 LinStage& st) {
  // thread -> 4 rows x 8 cols? no: 128 rows x 8 chunks of 16B = 1024 chunks;
  // 256 threads x 4 chunks each, chunk = (row, 16 B slice)
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int c = threadIdx.x + i * 256;
    st.r[i] = *(const bf16x8*)(g + (long long)(c >> 3) * ld + c0 + ((c & 7) << 3));
  }
}

static __device__ __forceinline__ void lin_write(unsigned short* lds,
                                                 const LinStage& st) {
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int c = threadIdx.x + i * 256;
    *(bf16x8*)((char*)lds + lswz(c >> 3, (c & 7) << 4)) = st.r[i];
  }
}

// transposed [64 rows r][128 cols] chunk -> LDS [128][64] (dgrad's W)
struct LinStageT {
  bf16x8 r[2][4];
};

static __device__ __forceinline__ void lin_issue_T(const unsigned short* g,
                                                   long long ld, int c0,
                                                   LinStageT& st) {
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int blk = threadIdx.x + i * 256;  // 4r x 8c sub-blocks (512 total)
    const int r0 = (blk / 16) * 4;
    const int cc = (blk % 16) * 8;
#pragma unroll
    for (int t = 0; t < 4; ++t)
      st.r[i][t] = *(const bf16x8*)(g + (long long)(r0 + t) * ld + c0 + cc);
  }
}

static __device__ __forceinline__ void lin_write_T(unsigned short* lds,
                                                   const LinStageT& st) {
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int blk = threadIdx.x + i * 256;
    const int r0 = (blk / 16) * 4;
    const int cc = (blk % 16) * 8;
    const unsigned short* u0 = (const unsigned short*)&st.r[i][0];
    const unsigned short* u1 = (const unsigned short*)&st.r[i][1];
    const unsigned short* u2 = (const unsigned short*)&st.r[i][2];
    const unsigned short* u3 = (const unsigned short*)&st.r[i][3];
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      ush4l pack = {u0[c], u1[c], u2[c], u3[c]};
      *(ush4l*)((char*)lds + lswz(cc + c, r0 * 2)) = pack;
    }
  }
}

// shared epilogue: 128x128 fp32 tile regs -> LDS bounce -> coalesced bf16
// rows of out[r][cols] (+ optional fp32 bias[c0 + n])
static __device__ __forceinline__ void lin_epilogue(
    f32x16 (&acc)[2][2], char* smem, unsigned short* out, long long ld,
    int r0g, int c0g, const unsigned short* bias, int wn, int wk, int lo31,
    int hi) {
  float* ftile = (float*)smem;  // [128 c][128 r] fp32 rows 512 B swizzled
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int c = wk * 64 + j * 32 + lo31;  // tile-local out col
#pragma unroll
      for (int rq = 0; rq < 4; ++rq) {
        const int rb = wn * 64 + i * 32 + 8 * rq + 4 * hi;
        float4 pk = make_float4(acc[i][j][rq * 4 + 0], acc[i][j][rq * 4 + 1],
                                acc[i][j][rq * 4 + 2], acc[i][j][rq * 4 + 3]);
        *(float4*)((char*)ftile + (long long)c * 512 +
                   ((rb * 4) ^ ((c & 7) << 4))) = pk;
      }
    }
  __syncthreads();
  const int r = threadIdx.x >> 1;
  const int ch = (threadIdx.x & 1) * 64;
  unsigned short* grow = out + (long long)(r0g + r) * ld + c0g + ch;
#pragma unroll
  for (int c8 = 0; c8 < 8; ++c8) {
    bf16x8 v;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int c = ch + c8 * 8 + e;
      float f = *(const float*)((char*)ftile + (long long)c * 512 +
                                ((r * 4) ^ ((c & 7) << 4)));
      if (bias) {
        union { unsigned int i; float fb; } bb;
        bb.i = (unsigned int)bias[c0g + c] << 16;
        f += bb.fb;
      }
      v[e] = (__bf16)f;
    }
    *(bf16x8*)(grow + c8 * 8) = v;
  }
}

// ---------------- forward: y = x @ W^T (+bias) ----------------
extern "C" __global__ __launch_bounds__(256) void k_linear_fwd(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ W,
    const unsigned short* __restrict__ bias, unsigned short* __restrict__ y,
    int R, int N, int K) {
  const int tiles_n = N / 128;
  const int tr = blockIdx.x / tiles_n, tn = blockIdx.x % tiles_n;
  const int r0 = tr * 128, n0 = tn * 128;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo31 = lane & 31;
  const int hi = (lane >> 5) & 1;
  const int wn = wave >> 1, wk = wave & 1;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned short* xl[2];
  unsigned short* wl[2];
  xl[0] = (unsigned short*)smem;
  wl[0] = (unsigned short*)(smem + 16384);
  xl[1] = (unsigned short*)(smem + 32768);
  wl[1] = (unsigned short*)(smem + 49152);

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.f);

  LinStage sx, sw;
  lin_issue(x + (long long)r0 * K, K, 0, sx);
  lin_issue(W + (long long)n0 * K, K, 0, sw);
  lin_write(xl[0], sx);
  lin_write(wl[0], sw);
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += 64) {
    if (k0 + 64 < K) {
      lin_issue(x + (long long)r0 * K + k0 + 64, K, 0, sx);
      lin_issue(W + (long long)n0 * K + k0 + 64, K, 0, sw);
    }
    __syncthreads();
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      // A = x rows (i = r), B = W rows (j = n); contraction k
      bf16x8 a0 = *(const bf16x8*)((char*)xl[cur] + lswz(wn * 64 + lo31, s * 32 + hi * 16));
      bf16x8 a1 = *(const bf16x8*)((char*)xl[cur] + lswz(wn * 64 + 32 + lo31, s * 32 + hi * 16));
      bf16x8 b0 = *(const bf16x8*)((char*)wl[cur] + lswz(wk * 64 + lo31, s * 32 + hi * 16));
      bf16x8 b1 = *(const bf16x8*)((char*)wl[cur] + lswz(wk * 64 + 32 + lo31, s * 32 + hi * 16));
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    if (k0 + 64 < K) {
      lin_write(xl[cur ^ 1], sx);
      lin_write(wl[cur ^ 1], sw);
    }
    __syncthreads();
    cur ^= 1;
  }
  // acc[i][j]: D rows = r (A side), cols = n (B side)
  lin_epilogue(acc, smem, y, N, r0, n0, bias, wn, wk, lo31, hi);
}

// ---------------- dgrad: dx = dy @ W ----------------
extern "C" __global__ __launch_bounds__(256) void k_linear_dgrad(
    const unsigned short* __restrict__ dy, const unsigned short* __restrict__ W,
    unsigned short* __restrict__ dx, int R, int N, int K) {
  const int tiles_k = K / 128;
  const int tr = blockIdx.x / tiles_k, tk = blockIdx.x % tiles_k;
  const int r0 = tr * 128, k0 = tk * 128;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo31 = lane & 31;
  const int hi = (lane >> 5) & 1;
  const int wn = wave >> 1, wk = wave & 1;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned short* dyl[2];
  unsigned short* wl[2];
  dyl[0] = (unsigned short*)smem;
  wl[0] = (unsigned short*)(smem + 16384);
  dyl[1] = (unsigned short*)(smem + 32768);
  wl[1] = (unsigned short*)(smem + 49152);

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.f);

  LinStage sdy;
  LinStageT sw;
  lin_issue(dy + (long long)r0 * N, N, 0, sdy);
  lin_issue_T(W, K, k0, sw);  // [64 n][128 k] panel -> LDS [128 k][64 n]
  lin_write(dyl[0], sdy);
  lin_write_T(wl[0], sw);
  int cur = 0;
  for (int nn0 = 0; nn0 < N; nn0 += 64) {
    if (nn0 + 64 < N) {
      lin_issue(dy + (long long)r0 * N + nn0 + 64, N, 0, sdy);
      lin_issue_T(W + (long long)(nn0 + 64) * K, K, k0, sw);
    }
    __syncthreads();
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      // A = dy rows (i = r), B = W^T rows (j = k); contraction n
      bf16x8 a0 = *(const bf16x8*)((char*)dyl[cur] + lswz(wn * 64 + lo31, s * 32 + hi * 16));
      bf16x8 a1 = *(const bf16x8*)((char*)dyl[cur] + lswz(wn * 64 + 32 + lo31, s * 32 + hi * 16));
      bf16x8 b0 = *(const bf16x8*)((char*)wl[cur] + lswz(wk * 64 + lo31, s * 32 + hi * 16));
      bf16x8 b1 = *(const bf16x8*)((char*)wl[cur] + lswz(wk * 64 + 32 + lo31, s * 32 + hi * 16));
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    if (nn0 + 64 < N) {
      lin_write(dyl[cur ^ 1], sdy);
      lin_write_T(wl[cur ^ 1], sw);
    }
    __syncthreads();
    cur ^= 1;
  }
  lin_epilogue(acc, smem, dx, K, r0, k0, nullptr, wn, wk, lo31, hi);
}
