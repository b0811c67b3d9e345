// Launch interface for the CDNA4 reduction kernels (csrc/hip/kernels.hip).
// Host-compilable header: no device code.
//
// Parity with the reference kernel set (gloo/cuda.cu:283-409:
// sum/product/min/max elementwise over {i8,u8,i32,i64,u64,f16,f32,f64}
// plus torch bf16) — re-designed for gfx950: 16-byte-per-lane vectorized
// grid-stride loops sized for 256 CUs (see kernels.hip).
#pragma once

#include <cstddef>
#include <cstdint>

#include "collectives/reduce_fns.h"

struct ihipStream_t;
typedef struct ihipStream_t* hipStream_t;

namespace glooamd {
namespace hip {

// dst[i] = op(a[i], b[i]) for n elements, on stream (async).
void launchReduce2(
    void* dst,
    const void* a,
    const void* b,
    size_t n,
    DType dtype,
    ReduceOp op,
    hipStream_t stream);

// dst[i] = op(srcs[0][i], ..., srcs[k-1][i]); k <= 8. Single fused pass
// (used for local multi-pointer reduction; reference equivalent does k-1
// separate kernel sweeps).
void launchReduceN(
    void* dst,
    const void* const* srcs,
    int k,
    size_t n,
    DType dtype,
    ReduceOp op,
    hipStream_t stream);

// In-place k-way allreduce: every pointer receives the reduced result
// in one fused pass (k <= 8).
void launchReduceNAll(
    void* const* ptrs,
    int k,
    size_t n,
    DType dtype,
    ReduceOp op,
    hipStream_t stream);

// Stream-ordered doorbell ops over system-scope atomics (fine-grained
// flag memory, peer-writable over xGMI). One 64-thread workgroup.
void launchWriteFlag(uint64_t* addr, uint64_t val, hipStream_t stream);
// Blocks the stream until *addr >= val (acquire, system scope: also
// makes peer SDMA-written data visible to later kernels on the stream).
void launchWaitFlagGte(const uint64_t* addr, uint64_t val, hipStream_t stream);

// Relative doorbells for hipGraph replay: target = *basePtr + offset,
// evaluated at execution time (offset may be negative; non-positive
// targets are no-ops). Always kernel-based (stream-op packets cannot be
// captured into graphs).
void launchWriteFlagRel(
    uint64_t* addr,
    const uint64_t* basePtr,
    int64_t offset,
    hipStream_t stream);
void launchWaitFlagGteRel(
    const uint64_t* addr,
    const uint64_t* basePtr,
    int64_t offset,
    hipStream_t stream);

// Test-fixture fill: ptr[i] = (i % 47) * stride + val (reference parity:
// gloo/cuda_private.cu:38-61 initializeMemory).
void launchFillPattern(
    void* ptr,
    size_t n,
    DType dtype,
    double val,
    double stride,
    hipStream_t stream);

} // namespace hip
} // namespace glooamd
