// Shared internals of the chunkflow_amd extension (cfx.hip + cc.hip are
// compiled into one shared object).
#ifndef CFX_INTERNAL_H
#define CFX_INTERNAL_H

#include <hip/hip_runtime.h>

#include <string>
#include <vector>

#include "../../include/chunkflow_amd.h"

struct ProfEntry {
    hipEvent_t e0, e1;
    int kid;
    double bytes;
};

struct cfx_ctx {
    int device = 0;
    hipStream_t stream = nullptr;  // legacy default stream unless adopted
    bool profile = false;
    std::vector<ProfEntry> pending;
    unsigned long long prof_count[CFX_K_COUNT] = {};
    double prof_ms[CFX_K_COUNT] = {};
    double prof_bytes[CFX_K_COUNT] = {};
    unsigned int* dev_max = nullptr;   // scratch for cfx_max
    unsigned int* cc_counts = nullptr;  // per-chunk root counts (cc.hip)
    int cc_counts_cap = 0;
};

extern thread_local std::string g_err;

#define CFX_CHECK(expr)                                                      \
    do {                                                                     \
        hipError_t _e = (expr);                                              \
        if (_e != hipSuccess) {                                              \
            g_err = std::string(#expr) + ": " + hipGetErrorString(_e);       \
            return -1;                                                       \
        }                                                                    \
    } while (0)

int prof_begin(cfx_ctx* ctx, hipEvent_t* e0);
int prof_end(cfx_ctx* ctx, hipEvent_t e0, int kid, double bytes);

#endif  // CFX_INTERNAL_H
