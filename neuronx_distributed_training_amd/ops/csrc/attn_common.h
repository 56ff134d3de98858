// Shared MFMA fragment helpers for the CDNA4 attention kernels.
//
// mfma_f32_16x16x32_bf16 operand maps (verified on gfx950 by
// tests/test_gpu_kernels.py::test_mfma_layout_probe):
//   A (16×32): lane holds row = lane&15, k = (lane>>4)*8 + j   (j = 0..7)
//   B (32×16): lane holds col = lane&15, k = (lane>>4)*8 + j
//   C/D (16×16): lane holds col = lane&15, row = (lane>>4)*4 + r (r = 0..3)
#pragma once
#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define MFMA_16x16x32(A, B, C) \
  __builtin_amdgcn_mfma_f32_16x16x32_bf16((A), (B), (C), 0, 0, 0)

// Load an A-fragment from a row-major LDS/global tile with row stride
// `stride` (in elements): rows = 16 consecutive, k-window of 32 at `k0`.
DEVINL bf16x8_t load_frag_a(const __bf16* base, int stride, int k0, int lane) {
  const __bf16* p = base + (lane & 15) * stride + k0 + (lane >> 4) * 8;
  bf16x8_t f;
  *(int4*)&f = *(const int4*)p;  // 8 contiguous bf16 = 16 B
  return f;
}

// Load a B-fragment where the tile is stored TRANSPOSED as [N][K] row-major
// (i.e. B[k][j] lives at tile[j*stride + k]): col j = lane&15 picks the row,
// the 8 k's are contiguous. This covers K^T reads from k_lds[key][d] etc.
DEVINL bf16x8_t load_frag_b_rowmajorT(const __bf16* base, int stride, int k0,
                                      int lane) {
  const __bf16* p = base + (lane & 15) * stride + k0 + (lane >> 4) * 8;
  bf16x8_t f;
  *(int4*)&f = *(const int4*)p;
  return f;
}

// --- XOR slot swizzle for TRANSPOSED images (V^T / Q^T / dO^T / K^T) ---
// The transpose WRITER's 32-lane groups step the image row by 8, i.e.
// 8·stride·2 B ≈ 0 (mod 128 B) for any 16B-aligned stride → every lane of
// a group lands on ONE bank (16-way ds_write conflict; PMC measured
// 12-15% of wave cycles). XORing byte bits 4-5 with image-row bits 3-4
// spreads a group over eight 16 B slots; the same XOR is applied on the
// b128 reads (both-sides-or-neither, guide §5.4 rule 21).
DEVINL uint tr_swz(uint byte_off, int img_row) {
  return byte_off ^ (((uint)(img_row >> 3) & 7u) << 4);
}

// XCD-affinity remap: workgroups dispatch round-robin over the 8 XCDs
// (xcd ~ wgid % 8, each with its own 4 MB L2). The natural (qblock, bh)
// grid spreads one head's q-blocks across all XCDs, so every XCD streams
// MANY K/V (or Q/dO) sequences and the L2 thrashes. This bijection gives
// each XCD whole (b, h) sequences at a time, keeping the per-head stream
// L2-resident (guide: +10-12% at seq 8k on HBM-heavy kernels). Falls
// back to the identity when gridDim.y % 8 != 0.
DEVINL void xcd_remap(int& qb, int& bh) {
  const int gx = gridDim.x, gy = gridDim.y;
  if ((gy & 7) != 0) {
    qb = blockIdx.x;
    bh = blockIdx.y;
    return;
  }
  const long wgid = (long)blockIdx.y * gx + blockIdx.x;
  const int xcd = (int)(wgid & 7);
  const long idx = wgid >> 3;
  bh = xcd + 8 * (int)(idx / gx);
  qb = (int)(idx % gx);
}

// swizzled 16 B read of a transposed image: abs image row r0 + (lane&15),
// k-window k0 (+8 per upper lane group).
DEVINL bf16x8_t load_frag_b_trT_swz(const __bf16* img, int stride, int r0,
                                    int k0, int lane) {
  const int r = r0 + (lane & 15);
  const uint byte = tr_swz((uint)(r * stride + k0 + (lane >> 4) * 8) * 2, r);
  bf16x8_t f;
  *(int4*)&f = *(const int4*)((const char*)img + byte);
  return f;
}

// ---- 32x32x16 bf16 MFMA fragments (maps probe-verified on gfx950 by
// tests/test_gpu_kernels.py::test_mfma32_layout_probe):
//   A (32x16): lane row = lane&31, k = (lane>>5)*8 + j
//   B (16x32): lane col = lane&31, k = (lane>>5)*8 + j
//   C (32x32): col = lane&31, row = (r&3) + 8*(r>>2) + 4*(lane>>5)
typedef float f32x16_t __attribute__((ext_vector_type(16)));
#define MFMA_32x32x16(A, B, C) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16((A), (B), (C), 0, 0, 0)

// A-fragment from a row-major [32 rows][stride] tile, k-window k0
DEVINL bf16x8_t load_frag_a32(const __bf16* base, int stride, int k0,
                              int lane) {
  const __bf16* p = base + (lane & 31) * stride + k0 + (lane >> 5) * 8;
  bf16x8_t f;
  *(int4*)&f = *(const int4*)p;
  return f;
}

// B-fragment from a tr_swz image [cols][stride] (col-major source):
// col = r0 + lane&31 picks the image row, 8 contiguous k's.
DEVINL bf16x8_t load_frag_b32_trT_swz(const __bf16* img, int stride, int r0,
                                      int k0, int lane) {
  const int r = r0 + (lane & 31);
  const uint byte = tr_swz((uint)(r * stride + k0 + (lane >> 5) * 8) * 2, r);
  bf16x8_t f;
  *(int4*)&f = *(const int4*)((const char*)img + byte);
  return f;
}
