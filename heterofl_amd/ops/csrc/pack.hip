// K13 of SURVEY §2b: device-side slice pack for Federation.distribute
// (reference hot site: src/fed.py:161-178, host deepcopies per client).
// Every dense-prefix slice is a (rows, cols) strided 2-D copy out of the
// global master tensor; ONE kernel launch copies every client's every
// slice from a descriptor table instead of ~600 narrow+clone launches per
// round.  Descriptors and destination buffers are cached per rate
// assignment (pointers are stable: global params update in place).
#include "common.h"

struct PackSpan {
    const float* src;
    float* dst;
    int rows;
    int cols;
    int src_stride;
    int pad;
};

__global__ void __launch_bounds__(256)
pack_slices_kernel(const PackSpan* __restrict__ spans, int n,
                   int rows_per_block) {
    const int d = blockIdx.x;
    if (d >= n) return;
    const PackSpan sp = spans[d];
    const int r0 = blockIdx.y * rows_per_block;
    const int r1 = min(sp.rows, r0 + rows_per_block);
    for (int r = r0; r < r1; ++r) {
        const float* s = sp.src + (long)r * sp.src_stride;
        float* t = sp.dst + (long)r * sp.cols;
        for (int c = threadIdx.x; c < sp.cols; c += blockDim.x)
            t[c] = s[c];
    }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

void pack_slices(at::Tensor blob, int64_t n, int64_t max_rows) {
    TORCH_CHECK(blob.is_cuda() && blob.is_contiguous());
    const int rpb = 16;
    const int ny = (int)((max_rows + rpb - 1) / rpb);
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(pack_slices_kernel, dim3((unsigned)n, ny), dim3(256),
                       0, stream, (const PackSpan*)blob.data_ptr(), (int)n,
                       rpb);
}
