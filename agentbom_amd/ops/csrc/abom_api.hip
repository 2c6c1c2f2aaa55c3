// Library-level utilities for the agent-bom MI355X engine.
#include "abom_common.h"

extern "C" int abom_abi_version() { return 1; }

extern "C" int abom_device_count() {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}

extern "C" int abom_synchronize(void* stream) {
    return (int)hipStreamSynchronize((hipStream_t)stream);
}

extern "C" const char* abom_error_string(int code) { return hipGetErrorString((hipError_t)code); }
