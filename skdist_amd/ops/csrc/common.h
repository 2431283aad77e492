// Shared helpers for the skdist_amd CDNA4 (gfx950) kernels.
//
// Conventions (see /opt/skills/guides/cdna_hip_programming.md):
//   * wave = 64 lanes; blocks are multiples of 64 threads
//   * MFMA bf16 16x16x32: per-lane fragments of 8 bf16 (A, B) and 4 f32
//     (C/D).  C/D mapping: col = lane&15, row = (lane>>4)*4 + reg.
//     A mapping: row = lane&15, k = (lane>>4)*8 + j  (j = 0..7)
//     B mapping: col = lane&15, k = (lane>>4)*8 + j
//     -> both operands are loaded K-contiguous, so every GEMM here takes
//        A as [M][K] row-major and B as [N][K] row-major ("B transposed"),
//        and fragment loads are single ds_read_b128s.
//   * out-of-range staging loads go through bounded buffer descriptors
//     (raw_buffer_load returns 0 out of bounds) so tiles never need edge
//     guards; only final global stores are guarded.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) int int4v;
typedef __attribute__((ext_vector_type(4))) short short4v;

#define MFMA_BF16_16x16x32(a, b, c) \
    __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

// Bounded buffer descriptor: OOB reads return 0, OOB writes drop.
// 0x00020000 = DFMT/NFMT config for raw buffers on gfx9+ (see guide T8).
static __device__ __forceinline__ __amdgpu_buffer_rsrc_t
make_rsrc(const void* ptr, unsigned long long nbytes) {
    return __builtin_amdgcn_make_buffer_rsrc(
        const_cast<void*>(ptr), /*stride*/ (short)0, nbytes, 0x00020000);
}

static __device__ __forceinline__ int4v
buf_load_dwordx4(__amdgpu_buffer_rsrc_t rsrc, int byte_off) {
    return __builtin_amdgcn_raw_buffer_load_b128(rsrc, byte_off, 0, 0);
}

static __device__ __forceinline__ float
buf_load_f32(__amdgpu_buffer_rsrc_t rsrc, int byte_off) {
    int v = __builtin_amdgcn_raw_buffer_load_b32(rsrc, byte_off, 0, 0);
    return __builtin_bit_cast(float, v);
}

static __device__ __forceinline__ int
buf_load_i32(__amdgpu_buffer_rsrc_t rsrc, int byte_off) {
    return __builtin_amdgcn_raw_buffer_load_b32(rsrc, byte_off, 0, 0);
}

static __device__ __forceinline__ float bf16_to_f32(__bf16 x) {
    return (float)x;
}

static __device__ __forceinline__ __bf16 f32_to_bf16(float x) {
    return (__bf16)x;  // v_cvt (RNE on gfx950)
}

// loss ids — keep in sync with skdist_amd/models/_sgd.py
#define LOSS_LOG 0
#define LOSS_HINGE 1
#define LOSS_SQUARED 2

static __device__ __forceinline__ float
dloss(int loss_id, float z, float t) {
    if (loss_id == LOSS_LOG) {
        // sigmoid(z) - t, numerically clamped
        float zc = fminf(fmaxf(z, -30.f), 30.f);
        return 1.f / (1.f + __expf(-zc)) - t;
    }
    if (loss_id == LOSS_HINGE) {
        float s = 2.f * t - 1.f;
        return (s * z < 1.f) ? -s : 0.f;
    }
    return z - t;  // squared
}

#define HIP_CHECK(expr)                                                   \
    do {                                                                  \
        hipError_t _e = (expr);                                           \
        if (_e != hipSuccess) return _e;                                  \
    } while (0)
