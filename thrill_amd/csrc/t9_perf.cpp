/* t9_perf.cpp — optional per-kernel-class HIP event timing, used by
 * bench.py's roofline leg (achieved = algorithmic bytes per launch /
 * measured launch duration). Off by default; zero cost when disabled. */

#include "t9_common.h"

#include <mutex>
#include <cstring>
#include <vector>

namespace {
struct Rec {
    hipEvent_t a, b;
    const char* cls;
};
bool g_on = false;
std::vector<Rec> g_recs;
std::mutex g_mu;
} // namespace

bool t9perf_on() { return g_on; }

void* t9perf_begin(hipStream_t s, const char* cls) {
    Rec* r = new Rec;
    r->cls = cls;
    hipEventCreate(&r->a);
    hipEventCreate(&r->b);
    hipEventRecord(r->a, s);
    return r;
}

void t9perf_end(void* tok, hipStream_t s) {
    Rec* r = (Rec*)tok;
    hipEventRecord(r->b, s);
    std::lock_guard<std::mutex> lk(g_mu);
    g_recs.push_back(*r);
    delete r;
}

extern "C" {

int t9_perf_enable(int on) {
    g_on = on != 0;
    return T9_OK;
}

/* total milliseconds and launch count for one kernel class */
int t9_perf_read(const char* cls, double* total_ms, u64* launches) {
    if (!cls || !total_ms || !launches) return T9_EINVAL;
    std::lock_guard<std::mutex> lk(g_mu);
    double ms = 0;
    u64 n = 0;
    for (auto& r : g_recs) {
        if (strcmp(r.cls, cls) != 0) continue;
        HIP_TRY(hipEventSynchronize(r.b));
        float f = 0;
        HIP_TRY(hipEventElapsedTime(&f, r.a, r.b));
        ms += f;
        ++n;
    }
    *total_ms = ms;
    *launches = n;
    return T9_OK;
}

int t9_perf_reset(void) {
    std::lock_guard<std::mutex> lk(g_mu);
    for (auto& r : g_recs) {
        hipEventDestroy(r.a);
        hipEventDestroy(r.b);
    }
    g_recs.clear();
    return T9_OK;
}

} /* extern "C" */
