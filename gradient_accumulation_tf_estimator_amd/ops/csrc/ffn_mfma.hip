// Fused-epilogue FFN GEMMs for the encoder MLP (gfx950).
//
//   k_ffn_fwd        : Y[R,I]    = gelu_tanh(X[R,H] @ W[I,H]^T + b[I]),
//                      AUX[R,I]  = the bf16 pre-activation (for backward)
//   k_ffn_dgrad_dgelu: dPRE[R,I] = gelu'(AUX) o (dY[R,H] @ Wo[H,I])
//
// This build's hipBLASLt exposes no GELU_AUX / DGELU epilogue solutions
// (ops/fused.py FusedFFN note), so the production FFN runs the activation
// as standalone elementwise kernels (k_biasgelu_fwd ~42 us/window,
// k_biasgelu_bwd_ew ~49 us/window at the bert-small fused-window shape)
// plus an extra HBM round-trip of the [R,I] intermediate each way.  These
// kernels put the activation where it belongs -- in the GEMM epilogue --
// using the same MFMA/LDS idioms as wgrad_mfma.hip: 256x128 output tiles,
// 8 wave64s per workgroup (4 r-quadrants x 2 i-halves, 64x64 per wave),
// v_mfma_f32_32x32x16_bf16, XOR-swizzled 128 B LDS rows, and an LDS-bounced
// epilogue so global writes are coalesced full rows.
//
// Both kernels contract over H (512..1024), chunked by 64:
//  * X / dY are [R,H] row-major -> a [256 r][64 h] chunk stages DIRECTLY
//    (contraction already minor; no transpose pass needed, unlike wgrad).
//  * W (fwd) is [I,H] row-major -> [128 i][64 h] chunks also stage directly.
//  * Wo (bwd) is [H,I] row-major -> contraction-major, so its [64 h][128 i]
//    chunks stage TRANSPOSED with wgrad's 4-row-pack transpose.
// Tile count (R/256)*(I/128) = 256 at the bert-small fused window
// (R=4096, I=2048) -- exactly one workgroup per CU.
//
// Replaces (reference parity): the tf.nn.gelu inside the BERT encoder the
// reference trains through modeling.py via optimization.py:25's train_op;
// here the activation is an epilogue, not an op.

#include <hip/hip_runtime.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned short ush4f;

// XOR swizzle for [rows][64] bf16 tiles (128 B rows): bits 4-6.
static inline __device__ int fswz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}

// ---- gelu (identical formulas to fused_ln_gelu.hip) ----
#define FGELU_C0 0.7978845608028654f
#define FGELU_C1 0.044715f

static inline __device__ float ffast_tanh(float u) {
  u = fminf(fmaxf(u, -9.f), 9.f);
  const float e = __expf(2.f * u);
  return (e - 1.f) / (e + 1.f);
}

static inline __device__ float fgelu_fwd1(float h) {
  const float u = FGELU_C0 * fmaf(FGELU_C1 * h * h, h, h);
  return 0.5f * h * (1.f + ffast_tanh(u));
}

static inline __device__ float fgelu_bwd1(float h, float dy) {
  const float u = FGELU_C0 * fmaf(FGELU_C1 * h * h, h, h);
  const float t = ffast_tanh(u);
  const float du = FGELU_C0 * fmaf(3.f * FGELU_C1 * h, h, 1.f);
  return dy * (0.5f * (1.f + t) + 0.5f * h * (1.f - t * t) * du);
}

// ---- direct staging: [ROWS r][64 c] global chunk -> LDS [r][64 c] ----
// (contraction minor in global memory; rows keep their layout, 16 B per
// thread-iter, swizzled 128 B LDS rows). 512 threads.
template <int ROWS>
struct DStage {
  static constexpr int ITER = ROWS * 8 / 512;  // vec8 slots per thread
  bf16x8 r[ITER];
};

template <int ROWS>
static __device__ __forceinline__ void dstage_issue(const unsigned short* g,
                                                    long long ld,
                                                    DStage<ROWS>& s) {
#pragma unroll
  for (int i = 0; i < DStage<ROWS>::ITER; ++i) {
    const int blk = threadIdx.x + i * 512;
    const int row = blk >> 3;
    const int c8 = (blk & 7) * 8;
    s.r[i] = *(const bf16x8*)(g + (long long)row * ld + c8);
  }
}

template <int ROWS>
static __device__ __forceinline__ void dstage_write(unsigned short* lds,
                                                    const DStage<ROWS>& s) {
#pragma unroll
  for (int i = 0; i < DStage<ROWS>::ITER; ++i) {
    const int blk = threadIdx.x + i * 512;
    const int row = blk >> 3;
    const int c8 = (blk & 7) * 8;
    *(bf16x8*)((char*)lds + fswz(row, c8 * 2)) = s.r[i];
  }
}

// ---- transpose staging (wgrad idiom): [64 h][128 c] global -> LDS
// [128 c][64 h]; 4-row x 4-col packs (64/4 x 128/4 = 512 sub-blocks, one
// per thread) so the LDS writes are 8 B 4-row packs as in wgrad_mfma.hip
// (a 2-row/4 B variant measured 2.5 LdsBankConflict per dispatch). ----
typedef __attribute__((ext_vector_type(2))) unsigned short ush2f;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

struct TStage {
  bf16x4 r[4];  // 4 h-rows x 4 c per thread
};

static __device__ __forceinline__ void tstage_issue(const unsigned short* g,
                                                    long long ld, int c0,
                                                    TStage& s) {
  const int blk = threadIdx.x;
  const int r0 = (blk >> 5) * 4;
  const int cc = (blk & 31) * 4;
#pragma unroll
  for (int t = 0; t < 4; ++t)
    s.r[t] = *(const bf16x4*)(g + (long long)(r0 + t) * ld + c0 + cc);
}

static __device__ __forceinline__ void tstage_write(unsigned short* lds,
                                                    const TStage& s) {
  const int blk = threadIdx.x;
  const int r0 = (blk >> 5) * 4;
  const int cc = (blk & 31) * 4;
  const unsigned short* u0 = (const unsigned short*)&s.r[0];
  const unsigned short* u1 = (const unsigned short*)&s.r[1];
  const unsigned short* u2 = (const unsigned short*)&s.r[2];
  const unsigned short* u3 = (const unsigned short*)&s.r[3];
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    ush4f pack = {u0[c], u1[c], u2[c], u3[c]};
    *(ush4f*)((char*)lds + fswz(cc + c, r0 * 2)) = pack;
  }
}

// ---- shared compute core: 256r x 128i tile over H, acc[2][2] per wave ----
// Template the epilogue via MODE: 0 = bias+gelu (writes y + aux),
// 1 = dgelu(aux) scale (writes dpre).
template <int MODE>
static __device__ __forceinline__ void ffn_tile(
    const unsigned short* a,   // [R,H] row-major, r0 applied by caller
    const unsigned short* b,   // fwd: W [I,H] (i0 applied); bwd: Wo [H,I]
    const unsigned short* bias,      // fwd only, [I] bf16 (i0 applied)
    const unsigned short* aux_in,    // bwd only, [R,I] at (r0, i0)
    unsigned short* out0,            // y / dpre at (r0, i0), ld = I
    unsigned short* out1,            // aux (fwd only)
    int H, long long ldI) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo31 = lane & 31;
  const int hi = (lane >> 5) & 1;
  const int wr = wave >> 1;  // 4 x 64-row quadrant
  const int wi = wave & 1;   // 2 x 64-col half

  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned short* As = (unsigned short*)smem;           // [256 r][64 h]
  unsigned short* Bs = (unsigned short*)(smem + 32768); // [128 i][64 h]

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.f);

  DStage<256> sa;
  DStage<128> sbf;  // fwd B
  TStage sbt;       // bwd B
  dstage_issue<256>(a, H, sa);
  if (MODE == 0)
    dstage_issue<128>(b, H, sbf);
  else
    tstage_issue(b, ldI, 0, sbt);
  for (int h0 = 0; h0 < H; h0 += 64) {
    __syncthreads();  // previous chunk's MFMA readers done
    dstage_write<256>(As, sa);
    if (MODE == 0)
      dstage_write<128>(Bs, sbf);
    else
      tstage_write(Bs, sbt);
    if (h0 + 64 < H) {
      dstage_issue<256>(a + h0 + 64, H, sa);
      if (MODE == 0)
        dstage_issue<128>(b + h0 + 64, H, sbf);
      else
        tstage_issue(b + (long long)(h0 + 64) * ldI, ldI, 0, sbt);
    }
    __syncthreads();
#pragma unroll
    for (int s = 0; s < 4; ++s) {  // 16-deep h sub-steps
      bf16x8 a0 = *(const bf16x8*)((char*)As + fswz(wr * 64 + lo31, s * 32 + hi * 16));
      bf16x8 a1 = *(const bf16x8*)((char*)As + fswz(wr * 64 + 32 + lo31, s * 32 + hi * 16));
      bf16x8 b0 = *(const bf16x8*)((char*)Bs + fswz(wi * 64 + lo31, s * 32 + hi * 16));
      bf16x8 b1 = *(const bf16x8*)((char*)Bs + fswz(wi * 64 + 32 + lo31, s * 32 + hi * 16));
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
  }

  // Epilogue in 4 passes of 64 r-rows, bounced through a [64 r][128 i]
  // fp32 LDS tile (32 KB, reusing the staging space): the MFMA fragment
  // holds 4-consecutive-r columns per lane (scattered in i), the drain
  // reads full i-rows so both bf16 global writes are 32 B per thread.
  float* ftile = (float*)smem;
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    __syncthreads();
    if (wr == p) {
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          const int col = wi * 64 + j * 32 + lo31;
#pragma unroll
          for (int rq = 0; rq < 4; ++rq) {
            const int rb = i * 32 + 8 * rq + 4 * hi;  // r within pass
#pragma unroll
            for (int e = 0; e < 4; ++e)
              *(float*)((char*)ftile + (long long)(rb + e) * 512 +
                        ((col * 4) ^ (((rb + e) & 7) << 4))) =
                  acc[i][j][rq * 4 + e];
          }
        }
    }
    __syncthreads();
    {
      // 512 threads x 16 i-elems = one 64x128 pass; 32 B bf16 writes
      const int r = threadIdx.x >> 3;
      const int c16 = (threadIdx.x & 7) * 16;
      const long long gr = (long long)(p * 64 + r) * ldI + c16;
      unsigned short ybuf[16];
      unsigned short xbuf[16];
      bf16x8 auxv[2];
      if (MODE == 1) {
        auxv[0] = *(const bf16x8*)(aux_in + gr);
        auxv[1] = *(const bf16x8*)(aux_in + gr + 8);
      }
#pragma unroll
      for (int e = 0; e < 16; ++e) {
        const int col = c16 + e;
        float v = *(const float*)((char*)ftile + (long long)r * 512 +
                                  ((col * 4) ^ ((r & 7) << 4)));
        if (MODE == 0) {
          __bf16 bb = ((const __bf16*)bias)[col];
          const float pre = v + (float)bb;
          const __bf16 preb = (__bf16)pre;
          xbuf[e] = *(const unsigned short*)&preb;
          const __bf16 yb = (__bf16)fgelu_fwd1(pre);
          ybuf[e] = *(const unsigned short*)&yb;
        } else {
          const float h = (float)((const __bf16*)&auxv[e >> 3])[e & 7];
          const __bf16 db = (__bf16)fgelu_bwd1(h, v);
          ybuf[e] = *(const unsigned short*)&db;
        }
      }
      *(bf16x8*)(out0 + gr) = *(const bf16x8*)&ybuf[0];
      *(bf16x8*)(out0 + gr + 8) = *(const bf16x8*)&ybuf[8];
      if (MODE == 0) {
        *(bf16x8*)(out1 + gr) = *(const bf16x8*)&xbuf[0];
        *(bf16x8*)(out1 + gr + 8) = *(const bf16x8*)&xbuf[8];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 256x256-tile variant (GA_FFN_TILE=256): halves the W-panel re-reads
// (16 -> 8 r-tiles share each staged i-slice) at the cost of half the
// workgroups (128 at the bert-small fused window vs 256 CUs). acc[2][4]
// per wave as in wgrad_mfma.hip's 256 kernel; 64 KB LDS (two [256][64]
// panels), epilogue bounced through a [64 r][256 i] fp32 tile.
// ---------------------------------------------------------------------------
struct TStage256 {
  bf16x8 r[2][2];  // 2 iters x 2 h-rows x 8 c
};

static __device__ __forceinline__ void tstage256_issue(const unsigned short* g,
                                                       long long ld,
                                                       TStage256& s) {
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int blk = threadIdx.x + it * 512;
    const int r0 = (blk >> 5) * 2;
    const int cc = (blk & 31) * 8;
#pragma unroll
    for (int t = 0; t < 2; ++t)
      s.r[it][t] = *(const bf16x8*)(g + (long long)(r0 + t) * ld + cc);
  }
}

static __device__ __forceinline__ void tstage256_write(unsigned short* lds,
                                                       const TStage256& s) {
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int blk = threadIdx.x + it * 512;
    const int r0 = (blk >> 5) * 2;
    const int cc = (blk & 31) * 8;
    const unsigned short* u0 = (const unsigned short*)&s.r[it][0];
    const unsigned short* u1 = (const unsigned short*)&s.r[it][1];
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      ush2f pack = {u0[c], u1[c]};
      *(ush2f*)((char*)lds + fswz(cc + c, r0 * 2)) = pack;
    }
  }
}

template <int MODE>
static __device__ __forceinline__ void ffn_tile256(
    const unsigned short* a, const unsigned short* b,
    const unsigned short* bias, const unsigned short* aux_in,
    unsigned short* out0, unsigned short* out1, int H, long long ldI) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo31 = lane & 31;
  const int hi = (lane >> 5) & 1;
  const int wr = wave >> 1;  // 4 x 64-row quadrant
  const int wk = wave & 1;   // 2 x 128-col half

  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned short* As = (unsigned short*)smem;           // [256 r][64 h]
  unsigned short* Bs = (unsigned short*)(smem + 32768); // [256 i][64 h]

  f32x16 acc[2][4];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x16)(0.f);

  DStage<256> sa, sbf;
  TStage256 sbt;
  dstage_issue<256>(a, H, sa);
  if (MODE == 0)
    dstage_issue<256>(b, H, sbf);
  else
    tstage256_issue(b, ldI, sbt);
  for (int h0 = 0; h0 < H; h0 += 64) {
    __syncthreads();
    dstage_write<256>(As, sa);
    if (MODE == 0)
      dstage_write<256>(Bs, sbf);
    else
      tstage256_write(Bs, sbt);
    if (h0 + 64 < H) {
      dstage_issue<256>(a + h0 + 64, H, sa);
      if (MODE == 0)
        dstage_issue<256>(b + h0 + 64, H, sbf);
      else
        tstage256_issue(b + (long long)(h0 + 64) * ldI, ldI, sbt);
    }
    __syncthreads();
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      bf16x8 a0 = *(const bf16x8*)((char*)As + fswz(wr * 64 + lo31, s * 32 + hi * 16));
      bf16x8 a1 = *(const bf16x8*)((char*)As + fswz(wr * 64 + 32 + lo31, s * 32 + hi * 16));
      bf16x8 b0 = *(const bf16x8*)((char*)Bs + fswz(wk * 128 + lo31, s * 32 + hi * 16));
      bf16x8 b1 = *(const bf16x8*)((char*)Bs + fswz(wk * 128 + 32 + lo31, s * 32 + hi * 16));
      bf16x8 b2 = *(const bf16x8*)((char*)Bs + fswz(wk * 128 + 64 + lo31, s * 32 + hi * 16));
      bf16x8 b3 = *(const bf16x8*)((char*)Bs + fswz(wk * 128 + 96 + lo31, s * 32 + hi * 16));
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[0][2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b2, acc[0][2], 0, 0, 0);
      acc[0][3] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b3, acc[0][3], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
      acc[1][2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b2, acc[1][2], 0, 0, 0);
      acc[1][3] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b3, acc[1][3], 0, 0, 0);
    }
  }

  // 4 passes over the 64-r quadrants; ftile [64 r][256 i] fp32 = 64 KB
  float* ftile = (float*)smem;
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    __syncthreads();
    if (wr == p) {
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int col = wk * 128 + j * 32 + lo31;
#pragma unroll
          for (int rq = 0; rq < 4; ++rq) {
            const int rb = i * 32 + 8 * rq + 4 * hi;
#pragma unroll
            for (int e = 0; e < 4; ++e)
              *(float*)((char*)ftile + (long long)(rb + e) * 1024 +
                        ((col * 4) ^ (((rb + e) & 7) << 4))) =
                  acc[i][j][rq * 4 + e];
          }
        }
    }
    __syncthreads();
    {
      const int r = threadIdx.x >> 3;
      const int c32 = (threadIdx.x & 7) * 32;
      const long long gr = (long long)(p * 64 + r) * ldI + c32;
      unsigned short ybuf[32];
      unsigned short xbuf[32];
      bf16x8 auxv[4];
      if (MODE == 1) {
#pragma unroll
        for (int q = 0; q < 4; ++q) auxv[q] = *(const bf16x8*)(aux_in + gr + q * 8);
      }
#pragma unroll
      for (int e = 0; e < 32; ++e) {
        const int col = c32 + e;
        float v = *(const float*)((char*)ftile + (long long)r * 1024 +
                                  ((col * 4) ^ ((r & 7) << 4)));
        if (MODE == 0) {
          __bf16 bb = ((const __bf16*)bias)[col];
          const float pre = v + (float)bb;
          const __bf16 preb = (__bf16)pre;
          xbuf[e] = *(const unsigned short*)&preb;
          const __bf16 yb = (__bf16)fgelu_fwd1(pre);
          ybuf[e] = *(const unsigned short*)&yb;
        } else {
          const float h = (float)((const __bf16*)&auxv[e >> 3])[e & 7];
          const __bf16 db = (__bf16)fgelu_bwd1(h, v);
          ybuf[e] = *(const unsigned short*)&db;
        }
      }
#pragma unroll
      for (int q = 0; q < 4; ++q)
        *(bf16x8*)(out0 + gr + q * 8) = *(const bf16x8*)&ybuf[q * 8];
      if (MODE == 0) {
#pragma unroll
        for (int q = 0; q < 4; ++q)
          *(bf16x8*)(out1 + gr + q * 8) = *(const bf16x8*)&xbuf[q * 8];
      }
    }
  }
}

// XCD-aware bijective remap (wgrad idiom): consecutive remapped ids share
// the X panel (same r-stripe), so its re-reads stay in one XCD's L2.
static inline __device__ int ffn_remap() {
  const int nwg = gridDim.x, orig = blockIdx.x;
  const int q = nwg / 8, r = nwg % 8, xcd = orig % 8;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
}

extern "C" __global__ __launch_bounds__(512) void k_ffn_fwd(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ w,
    const unsigned short* __restrict__ bias, unsigned short* __restrict__ y,
    unsigned short* __restrict__ aux, int H, int I) {
  const int id = ffn_remap();
  const int ti = id % (I / 128), tr = id / (I / 128);
  const long long r0 = (long long)tr * 256, i0 = (long long)ti * 128;
  ffn_tile<0>(x + r0 * H, w + i0 * H, bias + i0, nullptr,
              y + r0 * I + i0, aux + r0 * I + i0, H, I);
}

extern "C" __global__ __launch_bounds__(512) void k_ffn_dgrad_dgelu(
    const unsigned short* __restrict__ dy, const unsigned short* __restrict__ wo,
    const unsigned short* __restrict__ auxp, unsigned short* __restrict__ dpre,
    int H, int I) {
  const int id = ffn_remap();
  const int ti = id % (I / 128), tr = id / (I / 128);
  const long long r0 = (long long)tr * 256, i0 = (long long)ti * 128;
  ffn_tile<1>(dy + r0 * H, wo + i0, nullptr, auxp + r0 * I + i0,
              dpre + r0 * I + i0, nullptr, H, I);
}

extern "C" __global__ __launch_bounds__(512) void k_ffn_fwd_t256(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ w,
    const unsigned short* __restrict__ bias, unsigned short* __restrict__ y,
    unsigned short* __restrict__ aux, int H, int I) {
  const int id = ffn_remap();
  const int ti = id % (I / 256), tr = id / (I / 256);
  const long long r0 = (long long)tr * 256, i0 = (long long)ti * 256;
  ffn_tile256<0>(x + r0 * H, w + i0 * H, bias + i0, nullptr,
                 y + r0 * I + i0, aux + r0 * I + i0, H, I);
}

extern "C" __global__ __launch_bounds__(512) void k_ffn_dgrad_dgelu_t256(
    const unsigned short* __restrict__ dy, const unsigned short* __restrict__ wo,
    const unsigned short* __restrict__ auxp, unsigned short* __restrict__ dpre,
    int H, int I) {
  const int id = ffn_remap();
  const int ti = id % (I / 256), tr = id / (I / 256);
  const long long r0 = (long long)tr * 256, i0 = (long long)ti * 256;
  ffn_tile256<1>(dy + r0 * H, wo + i0, nullptr, auxp + r0 * I + i0,
                 dpre + r0 * I + i0, nullptr, H, I);
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------
namespace {

void check_ffn(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
                  t.scalar_type() == at::kBFloat16,
              "ffn_mfma: ", name, " must be contiguous CUDA bf16");
}

// GA_FFN_TILE=256 selects the wide-tile variant (halved W re-reads, half
// the workgroups); default 128. Read once.
bool use_t256() {
  static const bool v = [] {
    const char* e = getenv("GA_FFN_TILE");
    return e && atoi(e) == 256;
  }();
  return v;
}

std::vector<at::Tensor> ffn_fwd(at::Tensor x2d, at::Tensor w, at::Tensor bias) {
  check_ffn(x2d, "x");
  check_ffn(w, "w");
  check_ffn(bias, "bias");
  const int64_t H = x2d.size(1), R = x2d.size(0), I = w.size(0);
  TORCH_CHECK(w.size(1) == H && bias.size(0) == I, "ffn_fwd shape mismatch");
  TORCH_CHECK(R % 256 == 0 && I % 128 == 0 && H % 64 == 0,
              "ffn_fwd needs R%256==0, I%128==0, H%64==0, got ", R, "x", I,
              "x", H);
  auto y = at::empty({R, I}, x2d.options());
  auto aux = at::empty({R, I}, x2d.options());
  auto stream = at::cuda::getCurrentHIPStream().stream();
  if (use_t256() && I % 256 == 0)
    hipLaunchKernelGGL(k_ffn_fwd_t256, dim3((R / 256) * (I / 256)), dim3(512),
                       65536, stream, (const unsigned short*)x2d.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       (const unsigned short*)bias.data_ptr(),
                       (unsigned short*)y.data_ptr(),
                       (unsigned short*)aux.data_ptr(), (int)H, (int)I);
  else
    hipLaunchKernelGGL(k_ffn_fwd, dim3((R / 256) * (I / 128)), dim3(512), 49152,
                       stream, (const unsigned short*)x2d.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       (const unsigned short*)bias.data_ptr(),
                       (unsigned short*)y.data_ptr(),
                       (unsigned short*)aux.data_ptr(), (int)H, (int)I);
  return {y, aux};
}

at::Tensor ffn_dgrad_dgelu(at::Tensor dy2d, at::Tensor wo, at::Tensor aux) {
  check_ffn(dy2d, "dy");
  check_ffn(wo, "wo");
  check_ffn(aux, "aux");
  const int64_t H = dy2d.size(1), R = dy2d.size(0), I = wo.size(1);
  TORCH_CHECK(wo.size(0) == H && aux.size(0) == R && aux.size(1) == I,
              "ffn_dgrad shape mismatch");
  TORCH_CHECK(R % 256 == 0 && I % 128 == 0 && H % 64 == 0,
              "ffn_dgrad needs R%256==0, I%128==0, H%64==0, got ", R, "x", I,
              "x", H);
  auto dpre = at::empty({R, I}, dy2d.options());
  auto stream = at::cuda::getCurrentHIPStream().stream();
  if (use_t256() && I % 256 == 0)
    hipLaunchKernelGGL(k_ffn_dgrad_dgelu_t256, dim3((R / 256) * (I / 256)),
                       dim3(512), 65536, stream,
                       (const unsigned short*)dy2d.data_ptr(),
                       (const unsigned short*)wo.data_ptr(),
                       (const unsigned short*)aux.data_ptr(),
                       (unsigned short*)dpre.data_ptr(), (int)H, (int)I);
  else
    hipLaunchKernelGGL(k_ffn_dgrad_dgelu, dim3((R / 256) * (I / 128)),
                       dim3(512), 49152, stream,
                       (const unsigned short*)dy2d.data_ptr(),
                       (const unsigned short*)wo.data_ptr(),
                       (const unsigned short*)aux.data_ptr(),
                       (unsigned short*)dpre.data_ptr(), (int)H, (int)I);
  return dpre;
}

}  // namespace

void register_ffn_mfma(pybind11::module_& mod) {
  mod.def("ffn_fwd", &ffn_fwd,
          "y, aux(pre) = gelu(x @ W^T + b), MFMA with fused epilogue");
  mod.def("ffn_dgrad_dgelu", &ffn_dgrad_dgelu,
          "dpre = dgelu(aux) o (dy @ Wo), MFMA with fused epilogue");
}
