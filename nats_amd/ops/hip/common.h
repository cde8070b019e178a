// Common device helpers for the nats_amd CDNA4 (gfx950) kernels.
//
// Conventions:
//  * wavefront = 64 lanes; blocks are multiples of 64 threads,
//  * bf16 storage / fp32 accumulate via MFMA f32_16x16x32_bf16,
//  * fragment lane maps (verified on-device by mfma_selftest):
//      A (M16xK32): row = lane&15, k = (lane>>4)*8 + i   (i = 0..7)
//      B (K32xN16): col = lane&15, k = (lane>>4)*8 + i
//      C/D (16x16): col = lane&15, row = (lane>>4)*4 + i (i = 0..3)
//    With B stored TRANSPOSED (row-major N x K, "Bt"), both A and B
//    fragments are 16-byte contiguous loads — the layout every packed
//    weight buffer in this extension uses.
#pragma once

#include <hip/hip_runtime.h>

#define NATS_WAVE 64

typedef __bf16 bf16_t;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float nats_sigmoid(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

// Load an 8-element bf16 fragment (16B) from a row-major [rows][ld] matrix:
// element (r, k0..k0+7). Caller guarantees 16B alignment of (base + r*ld + k).
__device__ __forceinline__ bf16x8 load_frag8(const bf16_t* base, int r,
                                             long ld, int k) {
  return *(const bf16x8*)(base + (long)r * ld + k);
}

// A-fragment for mfma_f32_16x16x32_bf16 from row-major [M][ld] (M>=16 rows
// starting at row0): lane-determined (row, k).
__device__ __forceinline__ bf16x8 frag_a_rowmajor(const bf16_t* base,
                                                  int row0, long ld, int k0) {
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  return load_frag8(base, row0 + (lane & 15), ld, k0 + (lane >> 4) * 8);
}

// B-fragment from a TRANSPOSED row-major [N][ld] buffer (row = output col).
__device__ __forceinline__ bf16x8 frag_bt_rowmajor(const bf16_t* base,
                                                   int col0, long ld, int k0) {
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  return load_frag8(base, col0 + (lane & 15), ld, k0 + (lane >> 4) * 8);
}

// Scatter a C/D fragment into a row-major [M][ld] float buffer at (row0, col0).
__device__ __forceinline__ void store_cd_rowmajor(float* base, const f32x4& d,
                                                  int row0, long ld,
                                                  int col0) {
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  const int col = col0 + (lane & 15);
  const int rbase = row0 + (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) base[(long)(rbase + i) * ld + col] = d[i];
}

// 4-deep software-pipelined MFMA K-loop: four iterations' fragment loads
// (both operands) stay in flight ahead of each MFMA so the L2 latency of
// the 16 scattered 16B lines per fragment amortises over 4 iterations.
// (The plain loop measured ~13.9us/step on gfx950 — latency-bound; the
// 2-deep variant was still ~30us on the K=3k decoder GEMMs.)
#define NATS_MFMA_KLOOP(ACC, APTR, AROW, ALD, BPTR, BROW, BLD, KBEG, KEND)   \
  do {                                                                       \
    const int _ke = (KEND);                                                  \
    int _k = (KBEG);                                                         \
    if (_k + 128 <= _ke) {                                                   \
      bf16x8 _a0 = frag_a_rowmajor((APTR), (AROW), (ALD), _k);               \
      bf16x8 _b0 = frag_bt_rowmajor((BPTR), (BROW), (BLD), _k);              \
      bf16x8 _a1 = frag_a_rowmajor((APTR), (AROW), (ALD), _k + 32);          \
      bf16x8 _b1 = frag_bt_rowmajor((BPTR), (BROW), (BLD), _k + 32);         \
      bf16x8 _a2 = frag_a_rowmajor((APTR), (AROW), (ALD), _k + 64);          \
      bf16x8 _b2 = frag_bt_rowmajor((BPTR), (BROW), (BLD), _k + 64);         \
      bf16x8 _a3 = frag_a_rowmajor((APTR), (AROW), (ALD), _k + 96);          \
      bf16x8 _b3 = frag_bt_rowmajor((BPTR), (BROW), (BLD), _k + 96);         \
      for (_k += 128; _k + 32 <= _ke; _k += 32) {                            \
        bf16x8 _an = frag_a_rowmajor((APTR), (AROW), (ALD), _k);             \
        bf16x8 _bn = frag_bt_rowmajor((BPTR), (BROW), (BLD), _k);            \
        ACC = __builtin_amdgcn_mfma_f32_16x16x32_bf16(_a0, _b0, ACC, 0, 0, 0);\
        _a0 = _a1; _b0 = _b1; _a1 = _a2; _b1 = _b2;                          \
        _a2 = _a3; _b2 = _b3; _a3 = _an; _b3 = _bn;                          \
      }                                                                      \
      ACC = __builtin_amdgcn_mfma_f32_16x16x32_bf16(_a0, _b0, ACC, 0, 0, 0); \
      ACC = __builtin_amdgcn_mfma_f32_16x16x32_bf16(_a1, _b1, ACC, 0, 0, 0); \
      ACC = __builtin_amdgcn_mfma_f32_16x16x32_bf16(_a2, _b2, ACC, 0, 0, 0); \
      ACC = __builtin_amdgcn_mfma_f32_16x16x32_bf16(_a3, _b3, ACC, 0, 0, 0); \
    } else {                                                                 \
      for (; _k < _ke; _k += 32) {                                           \
        bf16x8 _a0 = frag_a_rowmajor((APTR), (AROW), (ALD), _k);             \
        bf16x8 _b0 = frag_bt_rowmajor((BPTR), (BROW), (BLD), _k);            \
        ACC = __builtin_amdgcn_mfma_f32_16x16x32_bf16(_a0, _b0, ACC, 0, 0, 0);\
      }                                                                      \
    }                                                                        \
  } while (0)

#define HIP_CHECK(expr)                                              \
  do {                                                               \
    hipError_t _e = (expr);                                          \
    if (_e != hipSuccess) {                                          \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",\
                  __FILE__, ":", __LINE__);                          \
    }                                                                \
  } while (0)
