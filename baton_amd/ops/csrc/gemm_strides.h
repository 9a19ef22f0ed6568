// Host-safe POD shared by kernels (gemm.hip), launchers and bindings.
//
// Strided-batch GEMM extension: leading dims + a two-level batch decode
// (z -> outer, head) so attention operands can be consumed as VIEWS of
// [B, S, h, dh] / [B, S, 3, h, dh] tensors without permute copies. All
// fields 0/1 = classic behavior (lda/ldb/ldc derived from the layout,
// linear batch offsets).
#pragma once

struct GemmStrides {
  long long lda, ldb, ldc;    // 0 = derive (TA ? M : K / TB ? K : N / N)
  int heads;                  // >1: z = outer*heads + head
  long long a2, b2, c2;       // inner (head) strides
};
