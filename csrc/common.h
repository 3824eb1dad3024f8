// sodkit CDNA4 (gfx950 / MI355X) kernel common helpers.
// Hand-written HIP for MI355X only — no CUDA compatibility, no hipify.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define SOD_DEVINL __device__ __forceinline__

// Wave width is 64 on CDNA (hard-coded per the CDNA4 programming guide).
constexpr int WAVE = 64;

// ---------------------------------------------------------------------------
// Vector types
// ---------------------------------------------------------------------------
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x2 __attribute__((ext_vector_type(2)));
typedef short s16x8 __attribute__((ext_vector_type(8)));   // 8 x bf16/fp16 bits
typedef short s16x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));

// ---------------------------------------------------------------------------
// Element traits: bf16 / fp16 / fp32 <-> float conversions
// ---------------------------------------------------------------------------
template <typename T> struct ElemTraits;

template <> struct ElemTraits<__hip_bfloat16> {
    SOD_DEVINL static float to_f32(__hip_bfloat16 v) { return __bfloat162float(v); }
    SOD_DEVINL static __hip_bfloat16 from_f32(float v) { return __float2bfloat16(v); }
};
template <> struct ElemTraits<__half> {
    SOD_DEVINL static float to_f32(__half v) { return __half2float(v); }
    SOD_DEVINL static __half from_f32(float v) {
        // saturate to the fp16 range (NaN-preserving: a NaN fails both
        // comparisons).  Early-training eval passes run activations through
        // unconverged BN running stats; letting them become inf poisons the
        // whole map, saturation keeps eval finite (bf16 needs none).
        if (v > 65504.f) v = 65504.f;
        else if (v < -65504.f) v = -65504.f;
        return __float2half(v);
    }
};
template <> struct ElemTraits<float> {
    SOD_DEVINL static float to_f32(float v) { return v; }
    SOD_DEVINL static float from_f32(float v) { return v; }
};

// MFMA op selection for 16x16x32 (bf16/f16 inputs, f32 accum).
template <typename T> struct MfmaOp;
template <> struct MfmaOp<__hip_bfloat16> {
    SOD_DEVINL static f32x4 mma(s16x8 a, s16x8 b, f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            (bf16x8)a, (bf16x8)b, c, 0, 0, 0);
    }
};
template <> struct MfmaOp<__half> {
    SOD_DEVINL static f32x4 mma(s16x8 a, s16x8 b, f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_f16(
            (f16x8)a, (f16x8)b, c, 0, 0, 0);
    }
};

// Reinterpret one 16-bit lane of an s16 vector as T and convert to f32.
template <typename T>
SOD_DEVINL float s16_as_f32(short bits) {
    T t;
    *reinterpret_cast<short *>(&t) = bits;
    return ElemTraits<T>::to_f32(t);
}

// Pack 4 f32 accumulator values into 4 consecutive 16-bit elems.
template <typename T>
SOD_DEVINL s16x4 pack4(const f32x4 &v) {
    s16x4 r;
    T tmp;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        tmp = ElemTraits<T>::from_f32(v[i]);
        r[i] = *reinterpret_cast<const short *>(&tmp);
    }
    return r;
}

// ---------------------------------------------------------------------------
// Wave / block reductions
// ---------------------------------------------------------------------------
SOD_DEVINL float wave_reduce_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, 64);
    return v;  // valid in lane 0
}

SOD_DEVINL float wave_allreduce_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_xor(v, off, 64);
    return v;
}

// Block-level sum into flat-thread 0 (needs 16 floats of LDS scratch).
// Uses the FLAT thread id so 2D/3D blocks work (a threadIdx.x-only wave id
// is wrong for a 16x16 block: every thread would claim wave 0 and the
// scratch writes race).
template <int NWAVES>
SOD_DEVINL float block_reduce_sum(float v, float *lds_scratch) {
    int ft = (threadIdx.z * blockDim.y + threadIdx.y) * blockDim.x + threadIdx.x;
    int lane = ft & 63;
    int w = ft >> 6;
    v = wave_reduce_sum(v);
    if (lane == 0) lds_scratch[w] = v;
    __syncthreads();
    float r = 0.f;
    if (w == 0) {
        r = (lane < NWAVES) ? lds_scratch[lane] : 0.f;
#pragma unroll
        for (int off = NWAVES >> 1; off > 0; off >>= 1)
            r += __shfl_down(r, off, 64);
    }
    return r;  // valid in flat-thread 0
}

SOD_DEVINL int ceil_div(int a, int b) { return (a + b - 1) / b; }
