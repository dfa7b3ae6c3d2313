// Launch-side declarations for the gfx950 kernels (reduce.hip / pack.hip).
#pragma once

#include <hip/hip_runtime.h>

#include "common.h"

namespace tma {

void launch_reduce(void* dst, const void* src, size_t count, DType d, RedOp op,
                   hipStream_t stream);
void launch_scale(void* data, size_t count, DType d, double alpha, hipStream_t stream);

void launch_pack(void* bucket, DType bucket_dtype, const void* metas_dev, int ntensors,
                 size_t total_elems, DType tensor_dtype, hipStream_t stream);
void launch_unpack(const void* bucket, DType bucket_dtype, const void* metas_dev,
                   int ntensors, size_t total_elems, DType tensor_dtype, double alpha,
                   hipStream_t stream);

}  // namespace tma
