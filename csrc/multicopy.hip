// sodkit — batched multi-segment device copy (deferred-grad gather).
//
// Motivation (round-2 trace, profiles/r2/kernel_summary_close.txt): with
// param .grad pre-assigned as bucket views, autograd's AccumulateGrad runs
// one tiny add_ kernel PER PARAMETER per step (~280 launches of <= 512
// threads at the ~4.5 us launch floor — ~3.5% of the single-GPU step, pure
// launch tax).  The deferred-grad mode (sodkit/parallel/reducer.py) instead
// lets autograd ASSIGN fresh grad tensors (zero kernels) and lands them in
// the flat buckets with THIS single kernel: one launch copies every
// (grad -> bucket-slot) segment, ~2x model bytes of traffic (~25 us at
// HBM3E rates) instead of ~300 launch floors.
//
// Segments are static in size/destination (bucket layout) but the source
// pointers change per step (autograd allocations), so the block->segment
// chunk map is built once on the host and only src[] is refreshed.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {
namespace mc {

constexpr int MC_CHUNK = 16384;  // bytes per workgroup

__global__ void __launch_bounds__(256)
multicopy_kernel(const long *__restrict__ src, const long *__restrict__ dst,
                 const long *__restrict__ nbytes,
                 const int *__restrict__ segmap,
                 const int *__restrict__ chunkmap) {
    int seg = segmap[blockIdx.x];
    long off = (long)chunkmap[blockIdx.x] * MC_CHUNK;
    long n = nbytes[seg] - off;
    if (n > MC_CHUNK) n = MC_CHUNK;
    const char *s = reinterpret_cast<const char *>(src[seg]) + off;
    char *d = reinterpret_cast<char *>(dst[seg]) + off;
    if (((reinterpret_cast<unsigned long>(s) |
          reinterpret_cast<unsigned long>(d)) & 15) == 0) {
        long n16 = n & ~15L;
        const uint4 *sv = reinterpret_cast<const uint4 *>(s);
        uint4 *dv = reinterpret_cast<uint4 *>(d);
        for (long i = threadIdx.x; i < (n16 >> 4); i += blockDim.x)
            dv[i] = sv[i];
        // element tail (all param dtypes are 2- or 4-byte => n is even)
        for (long i = n16 + 2 * threadIdx.x; i < n; i += 2 * (long)blockDim.x)
            *reinterpret_cast<short *>(d + i) =
                *reinterpret_cast<const short *>(s + i);
    } else {
        for (long i = 2 * threadIdx.x; i < n; i += 2 * (long)blockDim.x)
            *reinterpret_cast<short *>(d + i) =
                *reinterpret_cast<const short *>(s + i);
    }
}

}  // namespace mc
}  // namespace sodkit

int64_t multicopy_chunk_bytes() { return sodkit::mc::MC_CHUNK; }

// src/dst/nbytes: int64 [nseg] device; segmap/chunkmap: int32 [nblocks]
// device (block i copies chunk chunkmap[i] of segment segmap[i]).
void multicopy(torch::Tensor src, torch::Tensor dst, torch::Tensor nbytes,
               torch::Tensor segmap, torch::Tensor chunkmap) {
    TORCH_CHECK(src.is_cuda() && dst.is_cuda() && segmap.is_cuda());
    TORCH_CHECK(src.scalar_type() == at::kLong &&
                dst.scalar_type() == at::kLong &&
                nbytes.scalar_type() == at::kLong &&
                segmap.scalar_type() == at::kInt &&
                chunkmap.scalar_type() == at::kInt);
    int blocks = (int)segmap.numel();
    if (blocks == 0) return;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(sodkit::mc::multicopy_kernel, dim3(blocks), dim3(256),
                       0, stream.stream(),
                       src.data_ptr<long>(), dst.data_ptr<long>(),
                       nbytes.data_ptr<long>(), segmap.data_ptr<int>(),
                       chunkmap.data_ptr<int>());
}

// ---------------------------------------------------------------------------
// multipermute: batched KRSC -> CRSK weight transpose (dgrad w_t cache).
// The dgrad A-operand is w.permute(3,1,2,0).contiguous(); materializing it
// per conv per step was ~110 ATen copy launches (~0.64 ms).  Weights only
// change in the fused SGD step, so the cache layer (sodkit/ops/conv.py
// _WtCache) refreshes every cached w_t with ONE launch of this kernel
// right after the update.  Per block: one 64x64 (k, c) tile of one (r,s)
// slice, staged through LDS (int slots: column reads stay bank-free).
// ---------------------------------------------------------------------------

namespace sodkit {
namespace mp {

__global__ void __launch_bounds__(256)
multipermute_kernel(const long *__restrict__ src, const long *__restrict__ dst,
                    const int *__restrict__ Kd, const int *__restrict__ Cd,
                    const int *__restrict__ RSd, const int *__restrict__ seg,
                    const int *__restrict__ rsb, const int *__restrict__ ktb,
                    const int *__restrict__ ctb) {
    __shared__ int tile[64][65];
    int b = blockIdx.x;
    int s = seg[b], rs = rsb[b];
    int k0 = ktb[b] * 64, c0 = ctb[b] * 64;
    int K = Kd[s], C = Cd[s], RS = RSd[s];
    const unsigned short *sp = reinterpret_cast<const unsigned short *>(src[s]);
    unsigned short *dp = reinterpret_cast<unsigned short *>(dst[s]);
    int tx = threadIdx.x & 63, ty = threadIdx.x >> 6;
#pragma unroll
    for (int i = 0; i < 16; ++i) {
        int k = k0 + ty + i * 4;
        unsigned short v = 0;
        if (k < K && c0 + tx < C)
            v = sp[((long)k * RS + rs) * C + c0 + tx];
        tile[ty + i * 4][tx] = v;
    }
    __syncthreads();
#pragma unroll
    for (int i = 0; i < 16; ++i) {
        int c = c0 + ty + i * 4;
        if (c < C && k0 + tx < K)
            dp[((long)c * RS + rs) * K + k0 + tx] =
                (unsigned short)tile[tx][ty + i * 4];
    }
}

}  // namespace mp
}  // namespace sodkit

void multipermute(torch::Tensor src, torch::Tensor dst, torch::Tensor K,
                  torch::Tensor C, torch::Tensor RS, torch::Tensor seg,
                  torch::Tensor rsb, torch::Tensor ktb, torch::Tensor ctb) {
    TORCH_CHECK(src.is_cuda() && seg.is_cuda());
    TORCH_CHECK(src.scalar_type() == at::kLong &&
                seg.scalar_type() == at::kInt);
    int blocks = (int)seg.numel();
    if (blocks == 0) return;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(sodkit::mp::multipermute_kernel, dim3(blocks),
                       dim3(256), 0, stream.stream(),
                       src.data_ptr<long>(), dst.data_ptr<long>(),
                       K.data_ptr<int>(), C.data_ptr<int>(),
                       RS.data_ptr<int>(), seg.data_ptr<int>(),
                       rsb.data_ptr<int>(), ktb.data_ptr<int>(),
                       ctb.data_ptr<int>());
}
