// Tracked raw device allocations — the snapshot-visible hipMalloc surface.
//
// The reference checkpoints a PID's ENTIRE GPU memory with NVIDIA's external
// cuda-checkpoint binary (reference gpu_memory_snapshot.py:158-300). ROCm has
// no such tool, so raw device memory a process wants snapshotted must come
// from this tracked allocator: every allocation carries a stable string key,
// the snapshot machinery enumerates {key -> (ptr, size)} and pages each D2H,
// and a restored process re-allocates by key and pages back (pointers are
// process-local; keys are the durable identity).
//
// ma_hip_live_bytes() additionally reports the device's total used memory so
// the snapshot layer can detect UNTRACKED allocations (fidelity check: used
// minus torch-reserved minus tracked ≈ 0, else the snapshot is degraded).

#include <hip/hip_runtime.h>

#include <cstring>
#include <map>
#include <mutex>
#include <string>

namespace {

struct Tracked {
    void* ptr;
    unsigned long long size;
};

std::mutex g_mu;
std::map<std::string, Tracked> g_tracked;

}  // namespace

extern "C" int ma_tracked_alloc(const char* key, unsigned long long size, void** out) {
    std::lock_guard<std::mutex> lock(g_mu);
    auto it = g_tracked.find(key);
    if (it != g_tracked.end()) return hipErrorAlreadyAcquired;
    void* ptr = nullptr;
    hipError_t err = hipMalloc(&ptr, size);
    if (err != hipSuccess) return (int)err;
    g_tracked[key] = Tracked{ptr, size};
    if (out) *out = ptr;
    return 0;
}

extern "C" int ma_tracked_free(const char* key) {
    std::lock_guard<std::mutex> lock(g_mu);
    auto it = g_tracked.find(key);
    if (it == g_tracked.end()) return hipErrorInvalidValue;
    hipError_t err = hipFree(it->second.ptr);
    g_tracked.erase(it);
    return (int)err;
}

extern "C" int ma_tracked_lookup(const char* key, void** ptr, unsigned long long* size) {
    std::lock_guard<std::mutex> lock(g_mu);
    auto it = g_tracked.find(key);
    if (it == g_tracked.end()) return hipErrorInvalidValue;
    if (ptr) *ptr = it->second.ptr;
    if (size) *size = it->second.size;
    return 0;
}

extern "C" int ma_tracked_count() {
    std::lock_guard<std::mutex> lock(g_mu);
    return (int)g_tracked.size();
}

// enumerate key i into a caller buffer (keys are <= 255 bytes)
extern "C" int ma_tracked_key(int index, char* out, int out_len) {
    std::lock_guard<std::mutex> lock(g_mu);
    int i = 0;
    for (auto& kv : g_tracked) {
        if (i++ == index) {
            std::strncpy(out, kv.first.c_str(), out_len - 1);
            out[out_len - 1] = '\0';
            return 0;
        }
    }
    return hipErrorInvalidValue;
}

extern "C" int ma_tracked_read(const char* key, void* host) {
    std::lock_guard<std::mutex> lock(g_mu);
    auto it = g_tracked.find(key);
    if (it == g_tracked.end()) return hipErrorInvalidValue;
    return (int)hipMemcpy(host, it->second.ptr, it->second.size, hipMemcpyDeviceToHost);
}

extern "C" int ma_tracked_write(const char* key, const void* host) {
    std::lock_guard<std::mutex> lock(g_mu);
    auto it = g_tracked.find(key);
    if (it == g_tracked.end()) return hipErrorInvalidValue;
    return (int)hipMemcpy(it->second.ptr, host, it->second.size, hipMemcpyHostToDevice);
}

// free device memory but KEEP the registry row (page-out: the key survives,
// restore re-allocates under the same key)
extern "C" int ma_tracked_release(const char* key) {
    std::lock_guard<std::mutex> lock(g_mu);
    auto it = g_tracked.find(key);
    if (it == g_tracked.end()) return hipErrorInvalidValue;
    if (it->second.ptr) {
        hipError_t err = hipFree(it->second.ptr);
        if (err != hipSuccess) return (int)err;
        it->second.ptr = nullptr;
    }
    return 0;
}

extern "C" int ma_tracked_reacquire(const char* key, void** out) {
    std::lock_guard<std::mutex> lock(g_mu);
    auto it = g_tracked.find(key);
    if (it == g_tracked.end()) return hipErrorInvalidValue;
    if (it->second.ptr == nullptr) {
        hipError_t err = hipMalloc(&it->second.ptr, it->second.size);
        if (err != hipSuccess) return (int)err;
    }
    if (out) *out = it->second.ptr;
    return 0;
}

extern "C" int ma_hip_live_bytes(unsigned long long* used, unsigned long long* total) {
    size_t free_b = 0, total_b = 0;
    hipError_t err = hipMemGetInfo(&free_b, &total_b);
    if (err != hipSuccess) return (int)err;
    if (used) *used = (unsigned long long)(total_b - free_b);
    if (total) *total = (unsigned long long)total_b;
    return 0;
}

extern "C" unsigned long long ma_tracked_total_bytes() {
    std::lock_guard<std::mutex> lock(g_mu);
    unsigned long long t = 0;
    for (auto& kv : g_tracked)
        if (kv.second.ptr) t += kv.second.size;
    return t;
}
