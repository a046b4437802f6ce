// Fused LayerNorm for CDNA4 (gfx950) — forward + single-pass backward.
//
// Motivation (rocprof, GPT-2 small bf16 step): torch's LayerNorm stack is
// ~8% of step time and its backward reads (dy, x) twice (GradInput +
// PartGradGammaBeta). Here one wave64 owns one row: the row lives in
// registers between the statistics and normalize passes (COLS is a
// template parameter so all register indexing is static), cross-lane sums
// use shfl_xor, and the backward fuses dx with per-block dgamma/dbeta
// partials (single read of dy/x) accumulated in registers — each lane owns
// a fixed column set — and flushed once per wave; a tiny second kernel
// folds the per-wave partials.
//
// bf16/f16/f32 activations, fp32 statistics; instantiated for the common
// transformer widths (128..4096, divisible by the 16-byte vector width).

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>

#include "common.h"

#include <stdexcept>
#include <string>

namespace adapcc {

#define DEV_INLINE __device__ __forceinline__

namespace {

template <typename T> struct LnIo;
template <> struct LnIo<float> {
  using Vec = float4;
  static constexpr int kPerVec = 4;
  DEV_INLINE static void unpack(const Vec& v, float* o) {
    o[0] = v.x; o[1] = v.y; o[2] = v.z; o[3] = v.w;
  }
  DEV_INLINE static Vec pack(const float* i) {
    return make_float4(i[0], i[1], i[2], i[3]);
  }
};
template <> struct LnIo<__hip_bfloat16> {
  struct Vec { uint4 raw; };
  static constexpr int kPerVec = 8;
  DEV_INLINE static void unpack(const Vec& v, float* o) {
    const unsigned* w = reinterpret_cast<const unsigned*>(&v.raw);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      __hip_bfloat162 h = *reinterpret_cast<const __hip_bfloat162*>(&w[i]);
      o[2 * i] = __bfloat162float(h.x);
      o[2 * i + 1] = __bfloat162float(h.y);
    }
  }
  DEV_INLINE static Vec pack(const float* i) {
    Vec v; unsigned* w = reinterpret_cast<unsigned*>(&v.raw);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      __hip_bfloat162 h{__float2bfloat16(i[2 * k]),
                        __float2bfloat16(i[2 * k + 1])};
      w[k] = *reinterpret_cast<const unsigned*>(&h);
    }
    return v;
  }
};
template <> struct LnIo<__half> {
  struct Vec { uint4 raw; };
  static constexpr int kPerVec = 8;
  DEV_INLINE static void unpack(const Vec& v, float* o) {
    const unsigned* w = reinterpret_cast<const unsigned*>(&v.raw);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      __half2 h = *reinterpret_cast<const __half2*>(&w[i]);
      o[2 * i] = __half2float(__low2half(h));
      o[2 * i + 1] = __half2float(__high2half(h));
    }
  }
  DEV_INLINE static Vec pack(const float* i) {
    Vec v; unsigned* w = reinterpret_cast<unsigned*>(&v.raw);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      __half2 h = __floats2half2_rn(i[2 * k], i[2 * k + 1]);
      w[k] = *reinterpret_cast<const unsigned*>(&h);
    }
    return v;
  }
};

DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

}  // namespace

// Per-lane vector chunks for a COLS-wide row: ceil(COLS / (64*PV)) chunks;
// in chunk s, lanes with s*64 + lane < COLS/PV are active.
template <typename T, int COLS>
struct RowShape {
  using IO = LnIo<T>;
  static constexpr int PV = IO::kPerVec;
  static_assert(COLS % PV == 0, "COLS must be vector-divisible");
  static constexpr int kVecs = COLS / PV;             // 16B vectors per row
  static constexpr int kChunks = (kVecs + 63) / 64;   // per-lane chunks
  static constexpr int kElems = kChunks * PV;         // per-lane fp32 regs
};

template <typename T, int COLS>
__global__ void __launch_bounds__(256) ln_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w, const T* __restrict__ b,
    T* __restrict__ y, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, long rows, float eps) {
  using RS = RowShape<T, COLS>;
  using IO = typename RS::IO;
  using Vec = typename IO::Vec;
  constexpr int PV = RS::PV;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long row = (long)blockIdx.x * 4 + wave;
  if (row >= rows) return;

  const T* xr = x + row * COLS;
  float vals[RS::kElems];
  float sum = 0.f;
#pragma unroll
  for (int s = 0; s < RS::kChunks; ++s) {
    const int vi = s * 64 + lane;
    if (vi < RS::kVecs) {
      Vec v = reinterpret_cast<const Vec*>(xr)[vi];
      IO::unpack(v, &vals[s * PV]);
#pragma unroll
      for (int k = 0; k < PV; ++k) sum += vals[s * PV + k];
    } else {
#pragma unroll
      for (int k = 0; k < PV; ++k) vals[s * PV + k] = 0.f;
    }
  }
  sum = wave_sum(sum);
  const float mean = sum / COLS;
  float var = 0.f;
#pragma unroll
  for (int s = 0; s < RS::kChunks; ++s) {
    if (s * 64 + lane < RS::kVecs) {
#pragma unroll
      for (int k = 0; k < PV; ++k) {
        const float d = vals[s * PV + k] - mean;
        var += d * d;
      }
    }
  }
  var = wave_sum(var);
  const float rstd = rsqrtf(var / COLS + eps);
  if (lane == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }

  T* yr = y + row * COLS;
#pragma unroll
  for (int s = 0; s < RS::kChunks; ++s) {
    const int vi = s * 64 + lane;
    if (vi < RS::kVecs) {
      float wv[PV], bv[PV], out[PV];
      Vec a = reinterpret_cast<const Vec*>(w)[vi];
      Vec c = reinterpret_cast<const Vec*>(b)[vi];
      IO::unpack(a, wv);
      IO::unpack(c, bv);
#pragma unroll
      for (int k = 0; k < PV; ++k)
        out[k] = (vals[s * PV + k] - mean) * rstd * wv[k] + bv[k];
      reinterpret_cast<Vec*>(yr)[vi] = IO::pack(out);
    }
  }
}

template <typename T, int COLS>
__global__ void __launch_bounds__(256) ln_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ w, const float* __restrict__ mean_in,
    const float* __restrict__ rstd_in, T* __restrict__ dx,
    float* __restrict__ ws_gamma, float* __restrict__ ws_beta, long rows) {
  using RS = RowShape<T, COLS>;
  using IO = typename RS::IO;
  using Vec = typename IO::Vec;
  constexpr int PV = RS::PV;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long wave_stride = (long)gridDim.x * 4;

  // each lane owns the same column set for every row it touches ->
  // dgamma/dbeta partials accumulate in registers, flushed once per wave
  float dg[RS::kElems];
  float db[RS::kElems];
#pragma unroll
  for (int i = 0; i < RS::kElems; ++i) dg[i] = db[i] = 0.f;

  // weight is row-invariant: load it once
  float wv[RS::kElems];
#pragma unroll
  for (int s = 0; s < RS::kChunks; ++s) {
    const int vi = s * 64 + lane;
    if (vi < RS::kVecs) {
      Vec v = reinterpret_cast<const Vec*>(w)[vi];
      IO::unpack(v, &wv[s * PV]);
    }
  }

  for (long row = (long)blockIdx.x * 4 + wave; row < rows; row += wave_stride) {
    const T* dyr = dy + row * COLS;
    const T* xr = x + row * COLS;
    const float mean = mean_in[row];
    const float rstd = rstd_in[row];

    float dyv[RS::kElems], xh[RS::kElems];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int s = 0; s < RS::kChunks; ++s) {
      const int vi = s * 64 + lane;
      if (vi < RS::kVecs) {
        float d[PV], xx[PV];
        Vec vd = reinterpret_cast<const Vec*>(dyr)[vi];
        Vec vx = reinterpret_cast<const Vec*>(xr)[vi];
        IO::unpack(vd, d);
        IO::unpack(vx, xx);
#pragma unroll
        for (int k = 0; k < PV; ++k) {
          const float xhat = (xx[k] - mean) * rstd;
          const float a = d[k] * wv[s * PV + k];
          dyv[s * PV + k] = d[k];
          xh[s * PV + k] = xhat;
          s1 += a;
          s2 += a * xhat;
        }
      }
    }
    s1 = wave_sum(s1) / COLS;
    s2 = wave_sum(s2) / COLS;

    T* dxr = dx + row * COLS;
#pragma unroll
    for (int s = 0; s < RS::kChunks; ++s) {
      const int vi = s * 64 + lane;
      if (vi < RS::kVecs) {
        float out[PV];
#pragma unroll
        for (int k = 0; k < PV; ++k) {
          const float a = dyv[s * PV + k] * wv[s * PV + k];
          out[k] = (a - s1 - xh[s * PV + k] * s2) * rstd;
          dg[s * PV + k] += dyv[s * PV + k] * xh[s * PV + k];
          db[s * PV + k] += dyv[s * PV + k];
        }
        reinterpret_cast<Vec*>(dxr)[vi] = IO::pack(out);
      }
    }
  }

  // flush per-wave partials (one slot per wave, no atomics); each lane's
  // PV floats are contiguous -> vectorized float4 stores
  const long slot = (long)blockIdx.x * 4 + wave;
  float* wg = ws_gamma + slot * COLS;
  float* wb = ws_beta + slot * COLS;
#pragma unroll
  for (int s = 0; s < RS::kChunks; ++s) {
    const int vi = s * 64 + lane;
    if (vi < RS::kVecs) {
#pragma unroll
      for (int q = 0; q < PV / 4; ++q) {
        reinterpret_cast<float4*>(wg + vi * PV)[q] =
            make_float4(dg[s * PV + 4 * q], dg[s * PV + 4 * q + 1],
                        dg[s * PV + 4 * q + 2], dg[s * PV + 4 * q + 3]);
        reinterpret_cast<float4*>(wb + vi * PV)[q] =
            make_float4(db[s * PV + 4 * q], db[s * PV + 4 * q + 1],
                        db[s * PV + 4 * q + 2], db[s * PV + 4 * q + 3]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// dispatch
// ---------------------------------------------------------------------------

#define LN_COLS_LIST(X) \
  X(128) X(256) X(384) X(512) X(640) X(768) X(1024) X(1280) X(1536) \
  X(2048) X(3072) X(4096)

bool ln_supported_api(long cols, int dtype) {
  const int pv = dtype_size((Dtype)dtype) == 4 ? 4 : 8;
  if (cols % pv) return false;
#define LN_CHECK(C) if (cols == C) return true;
  LN_COLS_LIST(LN_CHECK)
#undef LN_CHECK
  return false;
}

template <typename T>
static void ln_fwd_dispatch(const void* x, const void* w, const void* b,
                            void* y, float* mean, float* rstd, long rows,
                            long cols, float eps, hipStream_t s) {
  const dim3 block(256);
  const dim3 grid((rows + 3) / 4);
#define LN_FWD_CASE(C)                                                      \
  if (cols == C) {                                                          \
    hipLaunchKernelGGL((ln_fwd_kernel<T, C>), grid, block, 0, s,            \
                       (const T*)x, (const T*)w, (const T*)b, (T*)y, mean,  \
                       rstd, rows, eps);                                    \
    return;                                                                 \
  }
  LN_COLS_LIST(LN_FWD_CASE)
#undef LN_FWD_CASE
  throw std::runtime_error("ln_forward: unsupported cols=" +
                           std::to_string(cols) +
                           " (check ln_supported first)");
}

template <typename T>
static void ln_bwd_dispatch(const void* dy, const void* x, const void* w,
                            const float* mean, const float* rstd, void* dx,
                            float* ws_gamma, float* ws_beta, long rows,
                            long cols, int nblocks, hipStream_t s) {
  const dim3 block(256);
#define LN_BWD_CASE(C)                                                      \
  if (cols == C) {                                                          \
    hipLaunchKernelGGL((ln_bwd_kernel<T, C>), dim3(nblocks), block, 0, s,   \
                       (const T*)dy, (const T*)x, (const T*)w, mean, rstd,  \
                       (T*)dx, ws_gamma, ws_beta, rows);                    \
    return;                                                                 \
  }
  LN_COLS_LIST(LN_BWD_CASE)
#undef LN_BWD_CASE
  throw std::runtime_error("ln_backward: unsupported cols=" +
                           std::to_string(cols) +
                           " (check ln_supported first)");
}

void ln_forward(int dtype, const void* x, const void* w, const void* b,
                void* y, float* mean, float* rstd, long rows, long cols,
                float eps, hipStream_t stream) {
  switch ((Dtype)dtype) {
    case Dtype::F32:
      ln_fwd_dispatch<float>(x, w, b, y, mean, rstd, rows, cols, eps, stream);
      break;
    case Dtype::BF16:
      ln_fwd_dispatch<__hip_bfloat16>(x, w, b, y, mean, rstd, rows, cols,
                                      eps, stream);
      break;
    case Dtype::F16:
      ln_fwd_dispatch<__half>(x, w, b, y, mean, rstd, rows, cols, eps,
                              stream);
      break;
  }
}

void ln_backward(int dtype, const void* dy, const void* x, const void* w,
                 const float* mean, const float* rstd, void* dx,
                 float* ws_gamma, float* ws_beta, long rows, long cols,
                 int nblocks, hipStream_t stream) {
  switch ((Dtype)dtype) {
    case Dtype::F32:
      ln_bwd_dispatch<float>(dy, x, w, mean, rstd, dx, ws_gamma, ws_beta,
                             rows, cols, nblocks, stream);
      break;
    case Dtype::BF16:
      ln_bwd_dispatch<__hip_bfloat16>(dy, x, w, mean, rstd, dx, ws_gamma,
                                      ws_beta, rows, cols, nblocks, stream);
      break;
    case Dtype::F16:
      ln_bwd_dispatch<__half>(dy, x, w, mean, rstd, dx, ws_gamma, ws_beta,
                              rows, cols, nblocks, stream);
      break;
  }
}

}  // namespace adapcc
