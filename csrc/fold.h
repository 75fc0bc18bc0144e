// Host-side helper for the deterministic column fold used by the fused
// elementwise kernels' bias gradients: partials (nb, C) fp32 -> out (C).
//
// For large nb a coalesced chunk fold (stage A, common.h) first reduces to
// 64 rows; the per-column final kernel then only gathers 64 x C values.
// Chunking is fixed by (nb, C) alone, so results are run-to-run
// deterministic either way.
#pragma once

#include <ATen/ATen.h>

#include "common.h"

static inline void unicore_fold_columns(const float* partials, float* out,
                                        int nb, int C,
                                        const at::TensorOptions& fopt,
                                        hipStream_t stream) {
  if (nb > 256) {
    constexpr int kRows = 64;
    auto tmp = at::empty({kRows, (int64_t)C}, fopt);
    const int chunk = (nb + kRows - 1) / kRows;
    unicore_col_fold_stage_kernel<<<dim3((C + 255) / 256, kRows), 256, 0,
                                    stream>>>(partials, tmp.data_ptr<float>(),
                                              nb, chunk, C);
    unicore_col_fold_kernel<<<C, 256, 0, stream>>>(tmp.data_ptr<float>(), out,
                                                   kRows, C);
  } else {
    unicore_col_fold_kernel<<<C, 256, 0, stream>>>(partials, out, nb, C);
  }
}
