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

    int64_t sample_index = 0;
    int64_t doc_idx_index = 0;
    int32_t doc_offset = 0;
    sample_idx[2 * sample_index] = (IdxT)doc_idx_index;
    sample_idx[2 * sample_index + 1] = (IdxT)doc_offset;
    ++sample_index;

    while (sample_index <= num_samples) {
        int32_t remaining_seq_length = seq_length + 1;
        while (remaining_seq_length != 0) {
            int32_t doc_id = doc_idx[doc_idx_index];
            int32_t doc_length = sizes[doc_id] - doc_offset;
            remaining_seq_length -= doc_length;
            if (remaining_seq_length <= 0) {
                // the window ends inside this document; the next window
                // starts at the last token consumed (windows overlap by one
                // token, matching the seq_length+1 sampling)
                doc_offset += (remaining_seq_length + doc_length - 1);
                remaining_seq_length = 0;
            } else {
                ++doc_idx_index;
                doc_offset = 0;
            }
        }
        sample_idx[2 * sample_index] = (IdxT)doc_idx_index;
        sample_idx[2 * sample_index + 1] = (IdxT)doc_offset;
        ++sample_index;
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

extern "C" int dolomite_build_blending_indices(int16_t* dataset_index, int64_t* dataset_sample_index,
                                               const double* weights, int32_t num_datasets,
                                               int64_t size) {
    if (num_datasets <= 0 || num_datasets > 32767) return 9102;
    int64_t* current_samples = new int64_t[num_datasets];
    for (int32_t i = 0; i < num_datasets; ++i) current_samples[i] = 0;

    for (int64_t sample_idx = 0; sample_idx < size; ++sample_idx) {
        double sample_idx_double = sample_idx > 1 ? (double)sample_idx : 1.0;
        int32_t max_error_index = 0;
        double max_error = weights[0] * sample_idx_double - (double)current_samples[0];
        for (int32_t d = 1; d < num_datasets; ++d) {
            double error = weights[d] * sample_idx_double - (double)current_samples[d];
            if (error > max_error) {
                max_error = error;
                max_error_index = d;
            }
        }
        dataset_index[sample_idx] = (int16_t)max_error_index;
        dataset_sample_index[sample_idx] = current_samples[max_error_index];
        current_samples[max_error_index] += 1;
    }
    delete[] current_samples;
    return 0;
}
