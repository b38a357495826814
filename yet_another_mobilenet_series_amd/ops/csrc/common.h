// Common device helpers for the gfx950 (CDNA4) kernel engine.
//
// Conventions (see ops/hip_ops.py for the host-side contract):
//  - activations: NHWC rows, C fastest, bf16 storage, fp32 math
//  - per-channel statistics / parameter tensors: fp32
//  - wave size 64 (CDNA), block sizes multiples of 64
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>
#include <stdexcept>
#include <string>

// Hard-error for launcher dispatch fallthrough: an unsupported
// (kernel, stride, lop, act, ...) combination must raise — propagated by
// pybind as RuntimeError — never silently return with the output tensor
// uninitialized (no silent corruption).  Host-only.
[[noreturn]] inline void yams_launch_fail(const char* fn, long a, long b,
                                          long c, long d) {
  throw std::runtime_error(
      std::string("yams: no kernel instantiation for ") + fn + "(" +
      std::to_string(a) + ", " + std::to_string(b) + ", " +
      std::to_string(c) + ", " + std::to_string(d) + ")");
}

#define DEV_INLINE __device__ __forceinline__

typedef unsigned short ushort_t;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;  // v_pk_fma_f32 pair
typedef __attribute__((ext_vector_type(8))) short bf16x8;   // MFMA A/B frag (4 VGPRs)
typedef __attribute__((ext_vector_type(4))) short short4_t; // 8-byte vector load
typedef __attribute__((ext_vector_type(8))) short short8_t; // 16-byte vector load
typedef __attribute__((ext_vector_type(4))) float float4_t;

// ---- bf16 <-> f32 (bf16 = upper 16 bits of f32; RNE on the way down) ----
DEV_INLINE float bf2f(ushort_t u) {
  union { uint32_t i; float f; } v;
  v.i = uint32_t(u) << 16;
  return v.f;
}

#ifdef __HIP_DEVICE_COMPILE__
// native convert: hipcc lowers the __bf16 cast to v_cvt_pk_bf16_f32
// (RNE, NaN-quieting) — ONE instruction per f32 PAIR vs the ~6-VALU
// manual round below.  PMC showed several kernels VALU-issue-bound with
// the store/EPI converts a large slice (profiles/r02/pmc_summary).
DEV_INLINE ushort_t f2bf(float f) {
  __bf16 b = static_cast<__bf16>(f);
  union { __bf16 b; ushort_t u; } v;
  v.b = b;
  return v.u;
}
#else
DEV_INLINE ushort_t f2bf(float f) {
  union { float f; uint32_t i; } v;
  v.f = f;
  uint32_t i = v.i;
  // round-to-nearest-even; NaN quietens
  if ((i & 0x7fffffffu) > 0x7f800000u) return ushort_t((i >> 16) | 0x40);
  i += 0x7fffu + ((i >> 16) & 1u);
  return ushort_t(i >> 16);
}
#endif

// ---- activations (codes shared with ops/hip_ops.py) ----
#define ACT_IDENTITY 0
#define ACT_RELU 1
#define ACT_RELU6 2
#define ACT_HSWISH 3
#define ACT_HSIGMOID 4

template <int ACT>
DEV_INLINE float act_fwd(float v) {
  if (ACT == ACT_RELU) return fmaxf(v, 0.f);
  if (ACT == ACT_RELU6) return fminf(fmaxf(v, 0.f), 6.f);
  if (ACT == ACT_HSWISH) return v * fminf(fmaxf(v + 3.f, 0.f), 6.f) * (1.f / 6.f);
  if (ACT == ACT_HSIGMOID) return fminf(fmaxf(v + 3.f, 0.f), 6.f) * (1.f / 6.f);
  return v;
}

// derivative w.r.t. the pre-activation value v
template <int ACT>
DEV_INLINE float act_bwd(float v) {
  if (ACT == ACT_RELU) return v > 0.f ? 1.f : 0.f;
  if (ACT == ACT_RELU6) return (v > 0.f && v < 6.f) ? 1.f : 0.f;
  if (ACT == ACT_HSWISH) {
    if (v <= -3.f) return 0.f;
    if (v >= 3.f) return 1.f;
    return (2.f * v + 3.f) * (1.f / 6.f);
  }
  if (ACT == ACT_HSIGMOID) return (v > -3.f && v < 3.f) ? (1.f / 6.f) : 0.f;
  return 1.f;
}

// runtime-act variants (wave-uniform switch: used by load-op paths where
// templating every act would explode the kernel instantiation count)
DEV_INLINE float act_fwd_rt(float v, int act) {
  switch (act) {
    case ACT_RELU: return fmaxf(v, 0.f);
    case ACT_RELU6: return fminf(fmaxf(v, 0.f), 6.f);
    case ACT_HSWISH:
      return v * fminf(fmaxf(v + 3.f, 0.f), 6.f) * (1.f / 6.f);
    case ACT_HSIGMOID: return fminf(fmaxf(v + 3.f, 0.f), 6.f) * (1.f / 6.f);
    default: return v;
  }
}

DEV_INLINE float act_bwd_rt(float v, int act) {
  switch (act) {
    case ACT_RELU: return v > 0.f ? 1.f : 0.f;
    case ACT_RELU6: return (v > 0.f && v < 6.f) ? 1.f : 0.f;
    case ACT_HSWISH:
      return v <= -3.f ? 0.f : (v >= 3.f ? 1.f : (2.f * v + 3.f) * (1.f / 6.f));
    case ACT_HSIGMOID: return (v > -3.f && v < 3.f) ? (1.f / 6.f) : 0.f;
    default: return 1.f;
  }
}

// Load-op: per-element transform applied while STAGING an operand stream
// (the fusion mechanism: BN-normalize+activation on forward loads,
// BN-backward dz computation on gradient loads — SURVEY.md §7 stage 4).
#define LOP_NONE 0
#define LOP_BNACT 1   // v' = act(a[c]*v + b[c]);           coef = ab[2C]
#define LOP_BNBWD 2   // v' = dz(dy=v, z=zv);               coef = [4C] {a,b,K0,K2}

struct Lop {
  int kind = LOP_NONE;
  int act = ACT_IDENTITY;
  const float* coef = nullptr;   // [2C] or [4C], SoA
  const ushort_t* z = nullptr;   // paired z stream for LOP_BNBWD
  int C = 0;
};

DEV_INLINE float lop_apply(const Lop& L, float v, float zv, int c) {
  if (L.kind == LOP_BNACT)
    return act_fwd_rt(fmaf(v, L.coef[c], L.coef[L.C + c]), L.act);
  if (L.kind == LOP_BNBWD) {
    const float a = L.coef[c];
    const float da = v * act_bwd_rt(fmaf(zv, a, L.coef[L.C + c]), L.act);
    return fmaf(a, da, L.coef[2 * L.C + c]) - L.coef[3 * L.C + c] * zv;
  }
  return v;
}

// compile-time-kind variants (kernels template on the lop KIND so the
// no-op path keeps its register budget; act stays a runtime switch)
template <int LOPK>
DEV_INLINE float lop_applyT(const Lop& L, float v, float zv, int c) {
  if (LOPK == LOP_BNACT)
    return act_fwd_rt(fmaf(v, L.coef[c], L.coef[L.C + c]), L.act);
  if (LOPK == LOP_BNBWD) {
    const float a = L.coef[c];
    const float da = v * act_bwd_rt(fmaf(zv, a, L.coef[L.C + c]), L.act);
    return fmaf(a, da, L.coef[2 * L.C + c]) - L.coef[3 * L.C + c] * zv;
  }
  return v;
}

template <int LOPK>
DEV_INLINE float lop_loadT(const Lop& L, const ushort_t* p, int64_t off,
                           int c) {
  const float v = bf2f(p[off]);
  if (LOPK == LOP_NONE) return v;
  const float zv = LOPK == LOP_BNBWD ? bf2f(L.z[off]) : 0.f;
  return lop_applyT<LOPK>(L, v, zv, c);
}

// Register-resident per-thread coefficient copy for loops where the
// channel range is FIXED per thread (dw kernels, per-image reductions):
// the coef pointer loads would otherwise be re-issued per element (the
// compiler cannot hoist them past the output stores).
template <int LOPK, int V>
struct LopReg {
  float a[V], b[V], k0[V], k2[V];
  int act;
  DEV_INLINE void init(const Lop& L, int c) {
    act = L.act;
    if (LOPK == LOP_NONE) return;
#pragma unroll
    for (int e = 0; e < V; ++e) {
      a[e] = L.coef[c + e];
      b[e] = L.coef[L.C + c + e];
    }
    if (LOPK == LOP_BNBWD) {
#pragma unroll
      for (int e = 0; e < V; ++e) {
        k0[e] = L.coef[2 * L.C + c + e];
        k2[e] = L.coef[3 * L.C + c + e];
      }
    }
  }
  DEV_INLINE float apply(float v, float zv, int e) const {
    if (LOPK == LOP_BNACT) return act_fwd_rt(fmaf(v, a[e], b[e]), act);
    if (LOPK == LOP_BNBWD) {
      const float da = v * act_bwd_rt(fmaf(zv, a[e], b[e]), act);
      return fmaf(a[e], da, k0[e]) - k2[e] * zv;
    }
    return v;
  }
};

// compile-time (kind, act) register-resident variant
template <int LOPK, int ACT, int V>
struct LopRegT {
  float a[V], b[V], k0[V], k2[V];
  DEV_INLINE void init(const Lop& L, int c) {
    if (LOPK == LOP_NONE) return;
#pragma unroll
    for (int e = 0; e < V; ++e) {
      a[e] = L.coef[c + e];
      b[e] = L.coef[L.C + c + e];
    }
    if (LOPK == LOP_BNBWD) {
#pragma unroll
      for (int e = 0; e < V; ++e) {
        k0[e] = L.coef[2 * L.C + c + e];
        k2[e] = L.coef[3 * L.C + c + e];
      }
    }
  }
  DEV_INLINE float apply(float v, float zv, int e) const {
    if (LOPK == LOP_BNACT) return act_fwd<ACT>(fmaf(v, a[e], b[e]));
    if (LOPK == LOP_BNBWD) {
      const float da = v * act_bwd<ACT>(fmaf(zv, a[e], b[e]));
      return fmaf(a[e], da, k0[e]) - k2[e] * zv;
    }
    return v;
  }
};

// 8-wide vectorized lop for STREAMING channel ranges (GEMM staging):
// coefs fetched as float4 pairs (requires c % 4 == 0, 8 in-bounds elems);
// output written back as bf16.
template <int LOPK>
DEV_INLINE void lop_vec8(const Lop& L, const ushort_t* p, int64_t off,
                         int c, ushort_t* dst) {
  short8_t v = *reinterpret_cast<const short8_t*>(&p[off]);
  if (LOPK == LOP_NONE) {
    *reinterpret_cast<short8_t*>(dst) = v;
    return;
  }
  float a[8], b[8];
  *reinterpret_cast<float4_t*>(a) =
      *reinterpret_cast<const float4_t*>(&L.coef[c]);
  *reinterpret_cast<float4_t*>(a + 4) =
      *reinterpret_cast<const float4_t*>(&L.coef[c + 4]);
  *reinterpret_cast<float4_t*>(b) =
      *reinterpret_cast<const float4_t*>(&L.coef[L.C + c]);
  *reinterpret_cast<float4_t*>(b + 4) =
      *reinterpret_cast<const float4_t*>(&L.coef[L.C + c + 4]);
  short8_t out;
  if (LOPK == LOP_BNACT) {
#pragma unroll
    for (int e = 0; e < 8; ++e)
      out[e] = short(f2bf(act_fwd_rt(
          fmaf(bf2f(ushort_t(v[e])), a[e], b[e]), L.act)));
  } else {
    float k0[8], k2[8];
    *reinterpret_cast<float4_t*>(k0) =
        *reinterpret_cast<const float4_t*>(&L.coef[2 * L.C + c]);
    *reinterpret_cast<float4_t*>(k0 + 4) =
        *reinterpret_cast<const float4_t*>(&L.coef[2 * L.C + c + 4]);
    *reinterpret_cast<float4_t*>(k2) =
        *reinterpret_cast<const float4_t*>(&L.coef[3 * L.C + c]);
    *reinterpret_cast<float4_t*>(k2 + 4) =
        *reinterpret_cast<const float4_t*>(&L.coef[3 * L.C + c + 4]);
    short8_t z = *reinterpret_cast<const short8_t*>(&L.z[off]);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float zv = bf2f(ushort_t(z[e]));
      const float da = bf2f(ushort_t(v[e])) *
                       act_bwd_rt(fmaf(zv, a[e], b[e]), L.act);
      out[e] = short(f2bf(fmaf(a[e], da, k0[e]) - k2[e] * zv));
    }
  }
  *reinterpret_cast<short8_t*>(dst) = out;
}

// bf16 load + lop for one element of a stream at flat offset `off`,
// channel `c` (zv fetched only when the lop needs it)
DEV_INLINE float lop_load(const Lop& L, const ushort_t* p, int64_t off,
                          int c) {
  const float v = bf2f(p[off]);
  if (L.kind == LOP_NONE) return v;
  const float zv = L.kind == LOP_BNBWD ? bf2f(L.z[off]) : 0.f;
  return lop_apply(L, v, zv, c);
}

// dispatch an `auto fn = [&](auto act_tag){...}` over a runtime act code
#define ACT_SWITCH(ACT_CODE, ...)                                      \
  switch (ACT_CODE) {                                                  \
    case ACT_IDENTITY: { constexpr int kAct = ACT_IDENTITY; __VA_ARGS__; break; } \
    case ACT_RELU:     { constexpr int kAct = ACT_RELU;     __VA_ARGS__; break; } \
    case ACT_RELU6:    { constexpr int kAct = ACT_RELU6;    __VA_ARGS__; break; } \
    case ACT_HSWISH:   { constexpr int kAct = ACT_HSWISH;   __VA_ARGS__; break; } \
    case ACT_HSIGMOID: { constexpr int kAct = ACT_HSIGMOID; __VA_ARGS__; break; } \
    default: yams_launch_fail("act_switch", (ACT_CODE), 0, 0, 0);        \
  }

DEV_INLINE int64_t idx4(int n, int h, int w, int c, int H, int W, int C) {
  return ((int64_t(n) * H + h) * W + w) * C + c;
}

constexpr int ceil_div(int64_t a, int64_t b) { return int((a + b - 1) / b); }

// Wave-level f32 sum over all 64 lanes (butterfly shuffles).
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

#define HIP_KERNEL_ASSERT_OK(expr)                                   \
  do {                                                               \
    hipError_t _e = (expr);                                          \
    if (_e != hipSuccess) {                                          \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e),       \
             __FILE__, __LINE__);                                    \
    }                                                                \
  } while (0)
