// Shared helpers for the DI-engine MI355X HIP kernels (gfx950 only).
#pragma once
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

// current PyTorch stream for this device (hipStream_t)
static inline hipStream_t ding_current_stream() {
    return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_INPUT(x)                                                                   \
    TORCH_CHECK(x.is_cuda(), #x " must be a HIP tensor");                                \
    TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

constexpr int kWave = 64;  // CDNA4 wavefront width

static inline int cdiv(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }

#define HIP_CHECK_LAST()                                                                 \
    do {                                                                                 \
        hipError_t e = hipGetLastError();                                                \
        TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ", hipGetErrorString(e)); \
    } while (0)
