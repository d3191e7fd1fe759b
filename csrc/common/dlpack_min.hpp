// Minimal DLPack ABI structs (public spec, dmlc/dlpack v0.8) — vendored so
// the extension can hand zero-copy device-tensor views to PyTorch without a
// build-time dependency. Only the fields the exchange protocol needs.
#pragma once

#include <cstdint>

extern "C" {

typedef enum {
    kDLCPU = 1,
    kDLCUDA = 2,
    kDLROCM = 10,
} DLDeviceType;

typedef struct {
    int32_t device_type;
    int32_t device_id;
} DLDevice;

typedef enum {
    kDLInt = 0,
    kDLUInt = 1,
    kDLFloat = 2,
    kDLBfloat = 4,
    kDLComplex = 5,
} DLDataTypeCode;

typedef struct {
    uint8_t code;
    uint8_t bits;
    uint16_t lanes;
} DLDataType;

typedef struct {
    void* data;
    DLDevice device;
    int32_t ndim;
    DLDataType dtype;
    int64_t* shape;
    int64_t* strides;
    uint64_t byte_offset;
} DLTensor;

typedef struct DLManagedTensor {
    DLTensor dl_tensor;
    void* manager_ctx;
    void (*deleter)(struct DLManagedTensor* self);
} DLManagedTensor;

} // extern "C"
