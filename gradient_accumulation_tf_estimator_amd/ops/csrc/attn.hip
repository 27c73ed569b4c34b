// Hand-written MFMA attention for the reference bench shapes (gfx950).
//
// Scope: seq_len S <= 128 (S % 32 == 0), head_dim 64, no mask, no dropout,
// bf16 in/out -- the reference's BERT configs all use head_dim 64 and the
// headline bench is seq128. Larger shapes fall back to torch SDPA
// (ops/fused.py gating).
//
// Why hand-written: at micro-batch 8 the torch flash path costs far more in
// layout copies (packed-QKV permutes), dq/dk/dv zero-fills and kernel count
// than in math. This kernel reads the fused-QKV projection's packed
// [B,S,3,H] output DIRECTLY and the backward writes the packed gradient
// buffer completely (no fills, no permutes, no .contiguous()).
//
// Structure (forward), one 4-wave block per (batch, head):
//   * K and V^T staged in LDS (XOR-swizzled rows for conflict-free
//     ds_read_b128 fragment reads; V transposed at stage time so the PV
//     B-operand reads are row-contiguous).
//   * "Swapped" QK^T: mfma(A=K, B=Q) gives S^T[k][q] so each lane holds one
//     q-column -> row softmax needs only one __shfl_xor(32) lane-pair
//     combine (cdna_hip_programming.md App. B attention recipe).
//   * P^T fp32 -> PV A-fragments fully in registers via v_cvt_pk_bf16_f32
//     pairs + __builtin_amdgcn_permlane32_swap (T12/T21 primitives).
//   * lse (base-2) saved for the backward's P recompute.
//
// MFMA fragment maps used throughout (v_mfma_f32_32x32x16_bf16):
//   A[i][k]: lane l holds i = l&31, k = (l>>5)*8 + c, c = 0..7
//   B[k][j]: lane l holds j = l&31, k = (l>>5)*8 + c
//   C/D   : lane l holds col j = l&31, row i = (r&3) + 8*(r>>2) + 4*(l>>5)

#include <hip/hip_runtime.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) unsigned short ush4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

#define ATTN_D 64
#define LOG2E 1.44269504088896340736f

static inline __device__ float attn_bf2f(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = (unsigned int)u << 16;
  return c.f;
}

// XOR-swizzled byte offset inside a [rows][64] bf16 tile (128 B rows):
// spread each 16-lane ds_read_b128 group over slots (G4 recipe).
static inline __device__ int swz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}

// [rows][128] bf16 tiles (256 B rows), same idea.
static inline __device__ int swz256(int row, int byte_in_row) {
  return row * 256 + (byte_in_row ^ ((row & 7) << 4));
}

// Stage a [S][64] bf16 matrix from global (row stride row_stride_elems) into
// LDS with swz layout. 256 threads; each handles S*64/ (256*8) rows of 8.
static __device__ void stage_64(const unsigned short* g, int row_stride,
                                unsigned short* lds, int S) {
  const int chunks = S * 8;  // 16 B chunks (8 bf16)
  for (int c = threadIdx.x; c < chunks; c += blockDim.x) {
    const int row = c >> 3, off = (c & 7) << 4;
    const unsigned short* src = g + (long long)row * row_stride + ((c & 7) << 3);
    // XOR mask touches bits 4-6 only, so the 16 B chunk stays one block
    *(bf16x8*)((char*)lds + swz(row, off)) = *(const bf16x8*)src;
  }
}

// Stage V transposed: LDS V^T[64][S] (S*2-byte rows padded to 256 B) with
// swz256 on rows of 128 bf16. Narrow writes, amortized once per block.
static __device__ void stage_64_T(const unsigned short* g, int row_stride,
                                  unsigned short* lds, int S) {
  // element (s, d) -> lds row d, col s
  const int total = S * ATTN_D;
  for (int e = threadIdx.x * 4; e < total; e += blockDim.x * 4) {
    const int s = e / ATTN_D, d0 = e % ATTN_D;
    const unsigned short* src = g + (long long)s * row_stride + d0;
    // 4 consecutive d of one s -> 4 different LDS rows, same col
    for (int t = 0; t < 4; ++t) {
      *(unsigned short*)((char*)lds + swz256(d0 + t, s * 2)) = src[t];
    }
  }
}

extern "C" __global__ __launch_bounds__(256) void k_attn_fwd(
    const unsigned short* __restrict__ qkv,  // [B,S,3,H]
    unsigned short* __restrict__ out,        // [B,S,H]
    float* __restrict__ lse_out,             // [B,nh,S] base-2 lse
    int B, int S, int nh) {
  const int H = nh * ATTN_D;
  const int b = blockIdx.x / nh, h = blockIdx.x % nh;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo31 = lane & 31;
  const int hi = lane >> 5;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned short* Klds = (unsigned short*)smem;            // [S][64] swz
  unsigned short* Vtlds = (unsigned short*)(smem + 16384); // [64][128] swz256

  const long long base = ((long long)b * S) * (3LL * H) + (long long)h * ATTN_D;
  stage_64(qkv + base + H, 3 * H, Klds, S);        // K
  stage_64_T(qkv + base + 2 * H, 3 * H, Vtlds, S); // V -> V^T
  __syncthreads();

  const int q0 = wave * 32;
  const int NT = S / 32;  // k-tiles
  if (q0 >= S) return;    // safe: no further barriers

  // Q B-fragments from global: lane -> q row lo31, d chunk hi*8 + 16*kk
  bf16x8 qf[4];
  {
    const unsigned short* qrow = qkv + base + (long long)(q0 + lo31) * 3 * H;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      qf[kk] = *(const bf16x8*)(qrow + kk * 16 + hi * 8);
  }

  // ---- S^T = K Q^T : per k-tile 32x32 acc ----
  f32x16 acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) acc[t] = (f32x16)(0.f);
  for (int t = 0; t < NT; ++t) {
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      // A = K[k=32t+lo31][d = hi*8 + 16kk ..]: swizzled b128 read
      bf16x8 a = *(const bf16x8*)((char*)Klds + swz(32 * t + lo31, kk * 32 + hi * 16));
      acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, qf[kk], acc[t], 0, 0, 0);
    }
  }

  // ---- softmax over k (lane pair l <-> l^32 holds one q column) ----
  const float scale2 = 0.125f * LOG2E;  // 1/sqrt(64) folded into exp2
  float m2 = -1e30f;
#pragma unroll
  for (int t = 0; t < 4; ++t)
    if (t < NT)
#pragma unroll
      for (int r = 0; r < 16; ++r) m2 = fmaxf(m2, acc[t][r]);
  m2 = fmaxf(m2, __shfl_xor(m2, 32, 64)) * scale2;
  float sum = 0.f;
#pragma unroll
  for (int t = 0; t < 4; ++t)
    if (t < NT)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        acc[t][r] = __builtin_amdgcn_exp2f(acc[t][r] * scale2 - m2);
        sum += acc[t][r];
      }
  sum += __shfl_xor(sum, 32, 64);
  const float inv_sum = 1.f / sum;
  const float lse2 = m2 + log2f(sum);
  if (hi == 0 && lo31 < 32) lse_out[((long long)b * nh + h) * S + q0 + lo31] = lse2;

  // normalize now so O needs no epilogue divide
#pragma unroll
  for (int t = 0; t < 4; ++t)
    if (t < NT)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[t][r] *= inv_sum;

  // ---- P^T fp32 -> PV A-fragments in registers (cvt_pk + permlane) ----
  // For k-tile t: frag(t,0) covers k' 32t+0..15, frag(t,1) k' 32t+16..31.
  bf16x8 pf[4][2];
#pragma unroll
  for (int t = 0; t < 4; ++t) {
    if (t >= NT) break;
#pragma unroll
    for (int halfk = 0; halfk < 2; ++halfk) {
      const int rb = halfk * 8;
      unsigned int u0, u1, v0, v1;
      // s_nop 1 inside the string: asm writes are invisible to the hazard
      // recognizer and v_permlane reads them within 2 states (T21 hazard)
      asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
          : "=v"(u0) : "v"(acc[t][rb + 0]), "v"(acc[t][rb + 1]));
      asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
          : "=v"(u1) : "v"(acc[t][rb + 2]), "v"(acc[t][rb + 3]));
      asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
          : "=v"(v0) : "v"(acc[t][rb + 4]), "v"(acc[t][rb + 5]));
      asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
          : "=v"(v1) : "v"(acc[t][rb + 6]), "v"(acc[t][rb + 7]));
      auto r0 = __builtin_amdgcn_permlane32_swap(u0, v0, false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap(u1, v1, false, false);
      unsigned int* pw = (unsigned int*)&pf[t][halfk];
      pw[0] = r0[0];
      pw[1] = r1[0];
      pw[2] = r0[1];
      pw[3] = r1[1];
    }
  }

  // ---- O = P V : two 32-d output tiles ----
#pragma unroll
  for (int dt = 0; dt < 2; ++dt) {
    f32x16 oc = (f32x16)(0.f);
    for (int t = 0; t < NT; ++t) {
#pragma unroll
      for (int halfk = 0; halfk < 2; ++halfk) {
        // B = V^T[d = dt*32 + lo31][k' = 32t + 16*halfk + hi*8 ..]
        bf16x8 bv = *(const bf16x8*)((char*)Vtlds +
                                     swz256(dt * 32 + lo31,
                                            (32 * t + 16 * halfk + hi * 8) * 2));
        bf16x8 pa = pf[t][halfk];
        oc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, bv, oc, 0, 0, 0);
      }
    }
    // D: col d = dt*32 + lo31, row q = (r&3) + 8*(r>>2) + 4*hi
    unsigned short* obase = out + ((long long)b * S + q0) * H + h * ATTN_D + dt * 32 + lo31;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int q = (r & 3) + 8 * (r >> 2) + 4 * hi;
      union { float f; unsigned int i; } c;
      c.f = oc[r];
      unsigned int x = c.i;
      unsigned short bf =
          (unsigned short)((x + (((x >> 16) & 1u) + 0x7fffu)) >> 16);
      obase[(long long)q * H] = bf;
    }
  }
}

// ---------------------------------------------------------------------------
// Backward: one 4-wave block per (b,h); wave w owns q-tile w for S^T/dS/dQ
// and k-tile w for dV/dK. P / dS^T staged in LDS (bf16) for the operand
// transposes; everything reads/writes the packed [B,S,3,H] buffers directly.
//
//   D_q   = rowsum(dO o O)
//   P^T   = exp2(scale2 * S^T - lse2)          (normalized, recomputed)
//   dP^T  = V dO^T                              mfma(A=V, B=dO-frags)
//   dS^T  = scale * P^T o (dP^T - D_q)
//   dV    = P^T dO                              mfma(A=P-lds, B=dO^T-lds)
//   dQ    = dS K                                mfma(A=dS-frags, B=K^T-lds)
//   dK    = dS^T Q                              mfma(A=dS-lds, B=Q^T-lds)

static __device__ void write_tile_bf16(unsigned short* base, long long row_stride,
                                       int lo31, int hi, const f32x16& acc) {
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    union { float f; unsigned int i; } c;
    c.f = acc[r];
    unsigned int x = c.i;
    base[row * row_stride] =
        (unsigned short)((x + (((x >> 16) & 1u) + 0x7fffu)) >> 16);
  }
}

extern "C" __global__ __launch_bounds__(256) void k_attn_bwd(
    const unsigned short* __restrict__ qkv,  // [B,S,3,H]
    const unsigned short* __restrict__ out,  // [B,S,H] (forward O)
    const unsigned short* __restrict__ dout, // [B,S,H]
    const float* __restrict__ lse_in,        // [B,nh,S] base-2
    unsigned short* __restrict__ dqkv,       // [B,S,3,H]
    int B, int S, int nh) {
  const int H = nh * ATTN_D;
  const int b = blockIdx.x / nh, h = blockIdx.x % nh;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo31 = lane & 31;
  const int hi = lane >> 5;
  const int NT = S / 32;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned short* Klds = (unsigned short*)smem;             // [S][64]   swz
  unsigned short* Ktlds = (unsigned short*)(smem + 16384);  // [64][128] swz256
  unsigned short* Vlds = (unsigned short*)(smem + 32768);   // [S][64]   swz
  unsigned short* dOtlds = (unsigned short*)(smem + 49152); // [64][128] swz256
  unsigned short* Qtlds = (unsigned short*)(smem + 65536);  // [64][128] swz256
  unsigned short* Plds = (unsigned short*)(smem + 81920);   // [S][128]  swz256

  const long long base = ((long long)b * S) * (3LL * H) + (long long)h * ATTN_D;
  const long long obase = ((long long)b * S) * H + (long long)h * ATTN_D;
  stage_64(qkv + base + H, 3 * H, Klds, S);
  stage_64_T(qkv + base + H, 3 * H, Ktlds, S);
  stage_64(qkv + base + 2 * H, 3 * H, Vlds, S);
  stage_64_T(dout + obase, H, dOtlds, S);
  stage_64_T(qkv + base, 3 * H, Qtlds, S);
  __syncthreads();

  const int q0 = wave * 32;
  const int active = q0 < S;  // inactive waves still hit barriers below

  // ---- per-q D = rowsum(dO o O), lse ----
  float D_q = 0.f, lse2 = 0.f;
  if (active) {
    const unsigned short* dor = dout + obase + (long long)(q0 + lo31) * H + hi * 32;
    const unsigned short* orow = out + obase + (long long)(q0 + lo31) * H + hi * 32;
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 dv = *(const bf16x8*)(dor + c * 8);
      bf16x8 ov = *(const bf16x8*)(orow + c * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) s += (float)dv[e] * (float)ov[e];
    }
    D_q = s + __shfl_xor(s, 32, 64);
    lse2 = lse_in[((long long)b * nh + h) * S + q0 + lo31];
  }

  // ---- recompute S^T and P^T; dP^T ----
  f32x16 acc[4], dacc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) {
    acc[t] = (f32x16)(0.f);
    dacc[t] = (f32x16)(0.f);
  }
  bf16x8 qf[4], dof[4];
  if (active) {
    const unsigned short* qrow = qkv + base + (long long)(q0 + lo31) * 3 * H;
    const unsigned short* drow = dout + obase + (long long)(q0 + lo31) * H;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      qf[kk] = *(const bf16x8*)(qrow + kk * 16 + hi * 8);
      dof[kk] = *(const bf16x8*)(drow + kk * 16 + hi * 8);
    }
    for (int t = 0; t < NT; ++t) {
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        bf16x8 ak = *(const bf16x8*)((char*)Klds + swz(32 * t + lo31, kk * 32 + hi * 16));
        bf16x8 av = *(const bf16x8*)((char*)Vlds + swz(32 * t + lo31, kk * 32 + hi * 16));
        acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ak, qf[kk], acc[t], 0, 0, 0);
        dacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(av, dof[kk], dacc[t], 0, 0, 0);
      }
    }
    const float scale2 = 0.125f * LOG2E;
    const float scale = 0.125f;
#pragma unroll
    for (int t = 0; t < 4; ++t)
      if (t < NT)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float p = __builtin_amdgcn_exp2f(acc[t][r] * scale2 - lse2);
          acc[t][r] = p;                                  // P^T
          dacc[t][r] = scale * p * (dacc[t][r] - D_q);    // dS^T (scaled)
        }
  }

  // ---- stage P^T -> Plds[k][q] (bf16, narrow writes) ----
  if (active) {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      if (t >= NT) break;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int k = 32 * t + (r & 3) + 8 * (r >> 2) + 4 * hi;
        union { float f; unsigned int i; } c;
        c.f = acc[t][r];
        unsigned int x = c.i;
        *(unsigned short*)((char*)Plds + swz256(k, (q0 + lo31) * 2)) =
            (unsigned short)((x + (((x >> 16) & 1u) + 0x7fffu)) >> 16);
      }
    }
  }
  __syncthreads();

  // ---- dV tile (k-rows 32*wave..+32): mfma over q ----
  const int k0 = wave * 32;
  if (k0 < S) {
#pragma unroll
    for (int dt = 0; dt < 2; ++dt) {
      f32x16 a = (f32x16)(0.f);
      for (int qs = 0; qs < S / 16; ++qs) {
        bf16x8 ap = *(const bf16x8*)((char*)Plds +
                                     swz256(k0 + lo31, (qs * 16 + hi * 8) * 2));
        bf16x8 bd = *(const bf16x8*)((char*)dOtlds +
                                     swz256(dt * 32 + lo31, (qs * 16 + hi * 8) * 2));
        a = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ap, bd, a, 0, 0, 0);
      }
      write_tile_bf16(dqkv + base + 2 * H + (long long)k0 * 3 * H + dt * 32 + lo31,
                      3 * H, lo31, hi, a);
    }
  }
  __syncthreads();

  // ---- stage dS^T -> Plds[k][q] (overwrite) ----
  if (active) {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      if (t >= NT) break;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int k = 32 * t + (r & 3) + 8 * (r >> 2) + 4 * hi;
        union { float f; unsigned int i; } c;
        c.f = dacc[t][r];
        unsigned int x = c.i;
        *(unsigned short*)((char*)Plds + swz256(k, (q0 + lo31) * 2)) =
            (unsigned short)((x + (((x >> 16) & 1u) + 0x7fffu)) >> 16);
      }
    }
  }

  // ---- dQ (own q-tile): A = dS frags from regs (cvt_pk + permlane) ----
  if (active) {
    bf16x8 sf[4][2];
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      if (t >= NT) break;
#pragma unroll
      for (int halfk = 0; halfk < 2; ++halfk) {
        const int rb = halfk * 8;
        unsigned int u0, u1, v0, v1;
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
            : "=v"(u0) : "v"(dacc[t][rb + 0]), "v"(dacc[t][rb + 1]));
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
            : "=v"(u1) : "v"(dacc[t][rb + 2]), "v"(dacc[t][rb + 3]));
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
            : "=v"(v0) : "v"(dacc[t][rb + 4]), "v"(dacc[t][rb + 5]));
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
            : "=v"(v1) : "v"(dacc[t][rb + 6]), "v"(dacc[t][rb + 7]));
        auto r0 = __builtin_amdgcn_permlane32_swap(u0, v0, false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(u1, v1, false, false);
        unsigned int* pw = (unsigned int*)&sf[t][halfk];
        pw[0] = r0[0];
        pw[1] = r1[0];
        pw[2] = r0[1];
        pw[3] = r1[1];
      }
    }
#pragma unroll
    for (int dt = 0; dt < 2; ++dt) {
      f32x16 a = (f32x16)(0.f);
      for (int t = 0; t < NT; ++t) {
#pragma unroll
        for (int halfk = 0; halfk < 2; ++halfk) {
          bf16x8 bk = *(const bf16x8*)((char*)Ktlds +
                                       swz256(dt * 32 + lo31,
                                              (32 * t + 16 * halfk + hi * 8) * 2));
          a = __builtin_amdgcn_mfma_f32_32x32x16_bf16(sf[t][halfk], bk, a, 0, 0, 0);
        }
      }
      write_tile_bf16(dqkv + base + (long long)q0 * 3 * H + dt * 32 + lo31,
                      3 * H, lo31, hi, a);
    }
  }
  __syncthreads();

  // ---- dK tile (k-rows 32*wave..+32): A = dS-lds, B = Q^T-lds ----
  if (k0 < S) {
#pragma unroll
    for (int dt = 0; dt < 2; ++dt) {
      f32x16 a = (f32x16)(0.f);
      for (int qs = 0; qs < S / 16; ++qs) {
        bf16x8 as = *(const bf16x8*)((char*)Plds +
                                     swz256(k0 + lo31, (qs * 16 + hi * 8) * 2));
        bf16x8 bq = *(const bf16x8*)((char*)Qtlds +
                                     swz256(dt * 32 + lo31, (qs * 16 + hi * 8) * 2));
        a = __builtin_amdgcn_mfma_f32_32x32x16_bf16(as, bq, a, 0, 0, 0);
      }
      write_tile_bf16(dqkv + base + H + (long long)k0 * 3 * H + dt * 32 + lo31,
                      3 * H, lo31, hi, a);
    }
  }
}
