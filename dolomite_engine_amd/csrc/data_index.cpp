// dolomite_hip — host-side dataset index builders (CPU, C-ABI).
//
// Restatement of the reference's only native component
// (data/megatron/utils/helpers.cpp):
//   - build_sample_idx (helpers.cpp:74-148 int32 / 151-225 int64): pack the
//     epoch-replicated document stream into seq_length+1 token windows,
//     producing [num_samples+1, 2] rows of (doc_idx position, token offset);
//   - build_blending_indices (helpers.cpp:17-69): greedy max-error
//     interleave of datasets toward target weights.
//
// Exposed through the same C-ABI .so as the GPU kernels; the caller
// (dolomite_engine_amd/megatron.py) owns all buffers.

#include <stdint.h>

#include "../../include/dolomite_hip.h"

template <typename IdxT>
static int build_sample_idx_impl(const int32_t* sizes, const int32_t* doc_idx,
                                 int32_t seq_length, int32_t num_epochs,
                                 int64_t tokens_per_epoch, IdxT* sample_idx,
                                 int64_t num_samples) {
    if (seq_length <= 1 || num_epochs <= 0 || tokens_per_epoch <= 1) return 9101;

    int64_t row = 0;
    int64_t stream_pos = 0;
    int32_t tok_off = 0;
    sample_idx[2 * row] = (IdxT)stream_pos;
    sample_idx[2 * row + 1] = (IdxT)tok_off;
    ++row;

    while (row <= num_samples) {
        int32_t window_left = seq_length + 1;
        while (window_left != 0) {
            int32_t doc = doc_idx[stream_pos];
            int32_t doc_left = sizes[doc] - tok_off;
            window_left -= doc_left;
            if (window_left <= 0) {
                // the window ends inside this document; the next window
                // starts at the last token consumed (windows overlap by one
                // token, matching the seq_length+1 sampling)
                tok_off += (window_left + doc_left - 1);
                window_left = 0;
            } else {
                ++stream_pos;
                tok_off = 0;
            }
        }
        sample_idx[2 * row] = (IdxT)stream_pos;
        sample_idx[2 * row + 1] = (IdxT)tok_off;
        ++row;
    }
    return 0;
}

extern "C" int dolomite_build_sample_idx_i32(const int32_t* sizes, const int32_t* doc_idx,
                                             int32_t seq_length, int32_t num_epochs,
                                             int64_t tokens_per_epoch, int32_t* sample_idx,
                                             int64_t num_samples) {
    return build_sample_idx_impl<int32_t>(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch,
                                          sample_idx, num_samples);
}

extern "C" int dolomite_build_sample_idx_i64(const int32_t* sizes, const int32_t* doc_idx,
                                             int32_t seq_length, int32_t num_epochs,
                                             int64_t tokens_per_epoch, int64_t* sample_idx,
                                             int64_t num_samples) {
    return build_sample_idx_impl<int64_t>(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch,
                                          sample_idx, num_samples);
}

extern "C" int dolomite_build_blending_indices(int16_t* dataset_index, int64_t* dataset_row,
                                               const double* weights, int32_t num_datasets,
                                               int64_t size) {
    if (num_datasets <= 0 || num_datasets > 32767) return 9102;
    int64_t* drawn = new int64_t[num_datasets];
    for (int32_t i = 0; i < num_datasets; ++i) drawn[i] = 0;

    for (int64_t sample_idx = 0; sample_idx < size; ++sample_idx) {
        double t = sample_idx > 1 ? (double)sample_idx : 1.0;
        int32_t pick = 0;
        double max_deficit = weights[0] * t - (double)drawn[0];
        for (int32_t d = 1; d < num_datasets; ++d) {
            double deficit = weights[d] * t - (double)drawn[d];
            if (deficit > max_deficit) {
                max_deficit = deficit;
                pick = d;
            }
        }
        dataset_index[sample_idx] = (int16_t)pick;
        dataset_row[sample_idx] = drawn[pick];
        drawn[pick] += 1;
    }
    delete[] drawn;
    return 0;
}
