// MI355X (gfx950 / CDNA4) kernels for the gradient-accumulation engine.
//
// Replaces the implicit native op surface of the TF1 reference
// (SURVEY.md section 2.3): per-variable AssignAdd / RealDiv /
// clip_by_global_norm / Adam elementwise chains become three grid-stride
// kernels over ONE flat, 256-B-aligned buffer set:
//
//   k_accum_*   : accum += grad (fp32 upcast); grad = 0      [every micro-step]
//   k_sqnorm    : sum(accum^2) -> device scalar (wave64 shuffle reduce + LDS
//                 tree + one atomic per block)               [apply step]
//   k_apply_*   : g = accum/K * clip_coef; AdamWeightDecay (no bias
//                 correction, eps outside sqrt, decoupled wd below the decay
//                 boundary); p -= lr*u; optional bf16 write-back; accum = 0
//                                                            [apply step]
//
// All kernels are HBM-bandwidth-bound: 16 B/lane vectorized access
// (float4 / ushort4-as-bf16x4), grid-stride with a ~2048-workgroup cap
// (256 CUs x 8 XCDs need >>256 workgroups; cdna_hip_programming.md G11).
// lr and the squared norm travel through device scalars so the apply step
// is hipGraph-capturable with a schedule-updated lr (no recapture).
//
// Every flat length is a multiple of 64 elements (engine/flat.py ALIGN), so
// the float4 loops have no scalar tail and the decay boundary is float4-
// uniform.

#include <hip/hip_runtime.h>

#define GA_THREADS 256
#define GA_MAX_BLOCKS 2048

static inline __device__ float bf16_to_f32(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = (unsigned int)u << 16;
  return c.f;
}

// round-to-nearest-even f32 -> bf16, matching PyTorch's cast
static inline __device__ unsigned short f32_to_bf16(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  unsigned int x = c.i;
  if ((x & 0x7fffffffu) > 0x7f800000u) return (unsigned short)((x >> 16) | 0x0040u); // NaN
  unsigned int round_bias = ((x >> 16) & 1u) + 0x7fffu;
  return (unsigned short)((x + round_bias) >> 16);
}

extern "C" __global__ void k_accum_f32(float4* __restrict__ accum,
                                       float4* __restrict__ grad,
                                       long long n4) {
  long long stride = (long long)gridDim.x * blockDim.x;
  const float4 z = make_float4(0.f, 0.f, 0.f, 0.f);
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    float4 a = accum[i];
    float4 g = grad[i];
    a.x += g.x; a.y += g.y; a.z += g.z; a.w += g.w;
    accum[i] = a;
    grad[i] = z;
  }
}

extern "C" __global__ void k_accum_bf16(float4* __restrict__ accum,
                                        ushort4* __restrict__ grad,
                                        long long n4) {
  long long stride = (long long)gridDim.x * blockDim.x;
  const ushort4 z = make_ushort4(0, 0, 0, 0);
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    float4 a = accum[i];
    ushort4 g = grad[i];
    a.x += bf16_to_f32(g.x); a.y += bf16_to_f32(g.y);
    a.z += bf16_to_f32(g.z); a.w += bf16_to_f32(g.w);
    accum[i] = a;
    grad[i] = z;
  }
}

// Global squared norm of the flat accum buffer into out[0].
// out must be zeroed before launch (engine does hipMemsetAsync-equivalent).
extern "C" __global__ void k_sqnorm(const float4* __restrict__ accum,
                                    long long n4,
                                    float* __restrict__ out) {
  // four independent per-component accumulator chains + 4 loads in flight
  // per iteration (2-load version measured 2.95 TB/s = well under the HBM
  // read peak; quadrupling the outstanding loads covers the miss latency)
  long long stride = (long long)gridDim.x * blockDim.x * 4;
  float sx = 0.f, sy = 0.f, sz = 0.f, sw = 0.f;
  long long i = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  for (; i + 3 < n4; i += stride) {
    float4 a = accum[i];
    float4 b = accum[i + 1];
    float4 c = accum[i + 2];
    float4 d = accum[i + 3];
    sx = fmaf(a.x, a.x, sx); sy = fmaf(a.y, a.y, sy);
    sz = fmaf(a.z, a.z, sz); sw = fmaf(a.w, a.w, sw);
    sx = fmaf(b.x, b.x, sx); sy = fmaf(b.y, b.y, sy);
    sz = fmaf(b.z, b.z, sz); sw = fmaf(b.w, b.w, sw);
    sx = fmaf(c.x, c.x, sx); sy = fmaf(c.y, c.y, sy);
    sz = fmaf(c.z, c.z, sz); sw = fmaf(c.w, c.w, sw);
    sx = fmaf(d.x, d.x, sx); sy = fmaf(d.y, d.y, sy);
    sz = fmaf(d.z, d.z, sz); sw = fmaf(d.w, d.w, sw);
  }
  // tail: only this thread's own (partial) 4-group -- unreachable today
  // (flat totals are 64-element aligned so n4 % 4 == 0)
  for (long long j = i; j < n4 && j < i + 4; ++j) {
    float4 a = accum[j];
    sx = fmaf(a.x, a.x, sx); sy = fmaf(a.y, a.y, sy);
    sz = fmaf(a.z, a.z, sz); sw = fmaf(a.w, a.w, sw);
  }
  float s = (sx + sy) + (sz + sw);
  // wave64 shuffle reduction
  for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off, 64);
  __shared__ float ws[GA_THREADS / 64];
  int lane = threadIdx.x & 63;
  int wid = threadIdx.x >> 6;
  if (lane == 0) ws[wid] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
    for (int w = 0; w < GA_THREADS / 64; ++w) t += ws[w];
    atomicAdd(out, t);  // device-scope, one per block
  }
}

// Fused normalize + clip + AdamWeightDecay + (optional bf16 write-back) + zero.
// Math contract: /root/reference/optimization.py:80-88 (apply branch order)
// and :150-171 (AdamWeightDecay update), cited in SURVEY.md section 2.2.
template <bool HAS_MODEL>
static __device__ void apply_body(float4* __restrict__ accum,
                                  float4* __restrict__ m,
                                  float4* __restrict__ v,
                                  float4* __restrict__ p,
                                  ushort4* __restrict__ model,
                                  const float* __restrict__ lr_p,
                                  const float* __restrict__ sq_p,
                                  long long n4, long long boundary4,
                                  float inv_k, float clip, float wd,
                                  float b1, float b2, float eps) {
  const float lr = lr_p[0];
  float coef = 1.f;
  if (clip > 0.f) {
    float norm = sqrtf(sq_p[0]) * inv_k;  // norm of accum/K (clip AFTER normalize)
    coef = clip / fmaxf(norm, clip);      // tf.clip_by_global_norm scale
  }
  const float s = inv_k * coef;
  const float omb1 = 1.f - b1, omb2 = 1.f - b2;
  const float4 z = make_float4(0.f, 0.f, 0.f, 0.f);
  // four float4 groups per iteration -- MEASURED SLOWER than the 2-group
  // variant (19.6-19.8k vs 20.8-21.1k samples/s same-box interleaved):
  // the extra in-flight registers cost occupancy on this 4-stream RMW.
  // Kept behind GA_APPLY_G4 for re-measurement; apply_body_g2 below is
  // the default.
  long long stride = (long long)gridDim.x * blockDim.x * 4;
  long long i = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  for (; i < n4; i += stride) {
    float4 a[4], mm[4], vv[4], pp[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      a[j] = accum[i + j];
      mm[j] = m[i + j];
      vv[j] = v[i + j];
      pp[j] = p[i + j];
    }
#define GA_C(av, mv, vvv, pv, c, dww)                  \
    {                                                  \
      float g = av.c * s;                              \
      mv.c = fmaf(b1, mv.c, omb1 * g);                 \
      vvv.c = fmaf(b2, vvv.c, omb2 * g * g);           \
      float u = mv.c / (sqrtf(vvv.c) + eps);           \
      u = fmaf(dww, pv.c, u);                          \
      pv.c = fmaf(-lr, u, pv.c);                       \
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float dw = ((i + j) < boundary4) ? wd : 0.f;
      GA_C(a[j], mm[j], vv[j], pp[j], x, dw)
      GA_C(a[j], mm[j], vv[j], pp[j], y, dw)
      GA_C(a[j], mm[j], vv[j], pp[j], z, dw)
      GA_C(a[j], mm[j], vv[j], pp[j], w, dw)
    }
#undef GA_C
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      m[i + j] = mm[j];
      v[i + j] = vv[j];
      p[i + j] = pp[j];
      accum[i + j] = z;
    }
    if (HAS_MODEL) {
#pragma unroll
      for (int pr = 0; pr < 2; ++pr) {
        union { ushort4 u4[2]; uint4 u16; } pk;
        pk.u4[0] = make_ushort4(
            f32_to_bf16(pp[2 * pr].x), f32_to_bf16(pp[2 * pr].y),
            f32_to_bf16(pp[2 * pr].z), f32_to_bf16(pp[2 * pr].w));
        pk.u4[1] = make_ushort4(
            f32_to_bf16(pp[2 * pr + 1].x), f32_to_bf16(pp[2 * pr + 1].y),
            f32_to_bf16(pp[2 * pr + 1].z), f32_to_bf16(pp[2 * pr + 1].w));
        *(uint4*)(model + i + 2 * pr) = pk.u16;
      }
    }
  }
}

template <bool HAS_MODEL>
static __device__ void apply_body_g2(float4* __restrict__ accum,
                                  float4* __restrict__ m,
                                  float4* __restrict__ v,
                                  float4* __restrict__ p,
                                  ushort4* __restrict__ model,
                                  const float* __restrict__ lr_p,
                                  const float* __restrict__ sq_p,
                                  long long n4, long long boundary4,
                                  float inv_k, float clip, float wd,
                                  float b1, float b2, float eps) {
  const float lr = lr_p[0];
  float coef = 1.f;
  if (clip > 0.f) {
    float norm = sqrtf(sq_p[0]) * inv_k;  // norm of accum/K (clip AFTER normalize)
    coef = clip / fmaxf(norm, clip);      // tf.clip_by_global_norm scale
  }
  const float s = inv_k * coef;
  const float omb1 = 1.f - b1, omb2 = 1.f - b2;
  const float4 z = make_float4(0.f, 0.f, 0.f, 0.f);
  // two float4 groups per iteration: independent mm/vv/u chains double the
  // in-flight loads and hide the sqrt+div latency; the bf16 model write
  // becomes one 16-byte store per pair
  long long stride = (long long)gridDim.x * blockDim.x * 2;
  long long i = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 2;
  for (; i < n4; i += stride) {
    const bool two = (i + 1) < n4;
    float4 a = accum[i], mm = m[i], vv = v[i], pp = p[i];
    float4 a2, mm2, vv2, pp2;
    if (two) { a2 = accum[i + 1]; mm2 = m[i + 1]; vv2 = v[i + 1]; pp2 = p[i + 1]; }
    const float dw = (i < boundary4) ? wd : 0.f;
    const float dw2 = ((i + 1) < boundary4) ? wd : 0.f;
#define GA_C2(av, mv, vvv, pv, c, dww)                  \
    {                                                  \
      float g = av.c * s;                              \
      mv.c = fmaf(b1, mv.c, omb1 * g);                 \
      vvv.c = fmaf(b2, vvv.c, omb2 * g * g);           \
      float u = mv.c / (sqrtf(vvv.c) + eps);           \
      u = fmaf(dww, pv.c, u);                          \
      pv.c = fmaf(-lr, u, pv.c);                       \
    }
    GA_C2(a, mm, vv, pp, x, dw) GA_C2(a, mm, vv, pp, y, dw)
    GA_C2(a, mm, vv, pp, z, dw) GA_C2(a, mm, vv, pp, w, dw)
    if (two) {
      GA_C2(a2, mm2, vv2, pp2, x, dw2) GA_C2(a2, mm2, vv2, pp2, y, dw2)
      GA_C2(a2, mm2, vv2, pp2, z, dw2) GA_C2(a2, mm2, vv2, pp2, w, dw2)
    }
#undef GA_C2
    m[i] = mm; v[i] = vv; p[i] = pp; accum[i] = z;
    if (two) { m[i + 1] = mm2; v[i + 1] = vv2; p[i + 1] = pp2; accum[i + 1] = z; }
    if (HAS_MODEL) {
      if (two) {
        union { ushort4 u4[2]; uint4 u16; } pk;
        pk.u4[0] = make_ushort4(f32_to_bf16(pp.x), f32_to_bf16(pp.y),
                                f32_to_bf16(pp.z), f32_to_bf16(pp.w));
        pk.u4[1] = make_ushort4(f32_to_bf16(pp2.x), f32_to_bf16(pp2.y),
                                f32_to_bf16(pp2.z), f32_to_bf16(pp2.w));
        *(uint4*)(model + i) = pk.u16;
      } else {
        model[i] = make_ushort4(f32_to_bf16(pp.x), f32_to_bf16(pp.y),
                                f32_to_bf16(pp.z), f32_to_bf16(pp.w));
      }
    }
  }
}

extern "C" __global__ void k_apply_f32_g2(float4* accum, float4* m, float4* v, float4* p,
                                          const float* lr_p, const float* sq_p,
                                          long long n4, long long boundary4,
                                          float inv_k, float clip, float wd,
                                          float b1, float b2, float eps) {
  apply_body_g2<false>(accum, m, v, p, nullptr, lr_p, sq_p, n4, boundary4,
                       inv_k, clip, wd, b1, b2, eps);
}
extern "C" __global__ void k_apply_bf16_g2(float4* accum, float4* m, float4* v, float4* p,
                                           ushort4* model,
                                           const float* lr_p, const float* sq_p,
                                           long long n4, long long boundary4,
                                           float inv_k, float clip, float wd,
                                           float b1, float b2, float eps) {
  apply_body_g2<true>(accum, m, v, p, model, lr_p, sq_p, n4, boundary4,
                      inv_k, clip, wd, b1, b2, eps);
}

extern "C" __global__ void k_apply_f32(float4* accum, float4* m, float4* v, float4* p,
                                       const float* lr_p, const float* sq_p,
                                       long long n4, long long boundary4,
                                       float inv_k, float clip, float wd,
                                       float b1, float b2, float eps) {
  apply_body<false>(accum, m, v, p, nullptr, lr_p, sq_p, n4, boundary4,
                    inv_k, clip, wd, b1, b2, eps);
}

extern "C" __global__ void k_apply_bf16(float4* accum, float4* m, float4* v, float4* p,
                                        ushort4* model,
                                        const float* lr_p, const float* sq_p,
                                        long long n4, long long boundary4,
                                        float inv_k, float clip, float wd,
                                        float b1, float b2, float eps) {
  apply_body<true>(accum, m, v, p, model, lr_p, sq_p, n4, boundary4,
                   inv_k, clip, wd, b1, b2, eps);
}
