// libgpushare_memguard.so — per-container VRAM budget enforcement.
//
// The reference plugin's isolation is advisory: NVIDIA_VISIBLE_DEVICES plus
// Aliyun's closed-source cGPU kernel module when real limits are wanted
// (node label cgpu.disable.isolation toggles it).  ROCm has no
// cGPU equivalent, so this preload library provides the enforcement layer:
// injected by Allocate() (Mount + LD_PRELOAD + GPUSHARE_MEM_LIMIT_BYTES
// env), it interposes the HIP allocation entry points and fails requests
// that would push the container past its gpu-mem share with
// hipErrorOutOfMemory — exactly what a well-behaved framework (PyTorch's
// caching allocator included) already handles as a normal OOM.
//
// Deliberately linked against NOTHING but libdl/libc: the real HIP symbols
// are resolved lazily with dlsym(RTLD_NEXT, ...) from whatever libamdhip64
// the application loads, so preloading into non-GPU processes (shells,
// sidecars) is a no-op.  hipMemGetInfo is clamped to the budget so
// frameworks that size pools from "free VRAM" stay inside their share.
//
// Thread-safe; per-pointer sizes tracked for exact release accounting.

#include <atomic>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <dlfcn.h>
#include <mutex>
#include <unordered_map>

namespace {

constexpr int HIP_SUCCESS = 0;
constexpr int HIP_ERROR_OOM = 2;            // hipErrorOutOfMemory
constexpr int HIP_ERROR_INVALID_VALUE = 1;  // hipErrorInvalidValue

using MallocFn = int (*)(void**, size_t);
using MallocFlagsFn = int (*)(void**, size_t, unsigned int);
using MallocAsyncFn = int (*)(void**, size_t, void*);
using FreeFn = int (*)(void*);
using FreeAsyncFn = int (*)(void*, void*);
using MemGetInfoFn = int (*)(size_t*, size_t*);

std::atomic<int64_t> g_used{0};
int64_t g_limit = -1;  // -1: unlimited (env absent) — pure passthrough

std::mutex g_sizes_mu;
std::unordered_map<void*, size_t>& sizes() {
    static std::unordered_map<void*, size_t> m;
    return m;
}

void init_limit() {
    static std::once_flag once;
    std::call_once(once, [] {
        const char* env = std::getenv("GPUSHARE_MEM_LIMIT_BYTES");
        if (env && *env) {
            char* end = nullptr;
            long long v = std::strtoll(env, &end, 10);
            if (end != env && v > 0) g_limit = v;
        }
    });
}

void* hip_handle() {
    // RTLD_NEXT only scans the GLOBAL scope after this library.  PyTorch
    // loads libamdhip64 as a dependency of an RTLD_LOCAL extension module,
    // so the runtime is absent from the global scope — grab a handle to
    // the already-loaded copy by soname instead (RTLD_NOLOAD never maps a
    // second copy; the final plain dlopen covers a not-yet-loaded runtime
    // and resolves to the same file the app will get).
    static void* h = [] {
        for (const char* name :
             {"libamdhip64.so.7", "libamdhip64.so.6", "libamdhip64.so"}) {
            void* p = dlopen(name, RTLD_LAZY | RTLD_NOLOAD);
            if (p) return p;
        }
        return dlopen("libamdhip64.so", RTLD_LAZY);
    }();
    return h;
}

template <typename Fn>
Fn real(const char* name) {
    void* sym = dlsym(RTLD_NEXT, name);
    if (sym == nullptr && hip_handle() != nullptr)
        sym = dlsym(hip_handle(), name);
    return reinterpret_cast<Fn>(sym);
}

bool reserve(size_t size) {
    init_limit();
    if (g_limit < 0) return true;
    int64_t prev = g_used.fetch_add((int64_t)size);
    if (prev + (int64_t)size > g_limit) {
        g_used.fetch_sub((int64_t)size);
        return false;
    }
    return true;
}

void track(void* ptr, size_t size) {
    if (g_limit < 0 || ptr == nullptr) return;
    std::lock_guard<std::mutex> lk(g_sizes_mu);
    sizes()[ptr] = size;
}

void untrack(void* ptr) {
    if (g_limit < 0 || ptr == nullptr) return;
    size_t size = 0;
    {
        std::lock_guard<std::mutex> lk(g_sizes_mu);
        auto it = sizes().find(ptr);
        if (it == sizes().end()) return;  // not ours (pre-preload alloc)
        size = it->second;
        sizes().erase(it);
    }
    g_used.fetch_sub((int64_t)size);
}

int guarded_alloc(MallocFn fn, void** ptr, size_t size) {
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    if (!reserve(size)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size);
    if (rc == HIP_SUCCESS) {
        track(*ptr, size);
    } else if (g_limit >= 0) {
        g_used.fetch_sub((int64_t)size);
    }
    return rc;
}

}  // namespace

extern "C" {

int hipMalloc(void** ptr, size_t size) {
    static MallocFn fn = real<MallocFn>("hipMalloc");
    return guarded_alloc(fn, ptr, size);
}

int hipMallocManaged(void** ptr, size_t size, unsigned int flags) {
    static MallocFlagsFn fn = real<MallocFlagsFn>("hipMallocManaged");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    if (!reserve(size)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size, flags);
    if (rc == HIP_SUCCESS) track(*ptr, size);
    else if (g_limit >= 0) g_used.fetch_sub((int64_t)size);
    return rc;
}

int hipExtMallocWithFlags(void** ptr, size_t size, unsigned int flags) {
    static MallocFlagsFn fn = real<MallocFlagsFn>("hipExtMallocWithFlags");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    if (!reserve(size)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size, flags);
    if (rc == HIP_SUCCESS) track(*ptr, size);
    else if (g_limit >= 0) g_used.fetch_sub((int64_t)size);
    return rc;
}

int hipMallocAsync(void** ptr, size_t size, void* stream) {
    static MallocAsyncFn fn = real<MallocAsyncFn>("hipMallocAsync");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    if (!reserve(size)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size, stream);
    if (rc == HIP_SUCCESS) track(*ptr, size);
    else if (g_limit >= 0) g_used.fetch_sub((int64_t)size);
    return rc;
}

int hipFree(void* ptr) {
    static FreeFn fn = real<FreeFn>("hipFree");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    int rc = fn(ptr);
    if (rc == HIP_SUCCESS) untrack(ptr);
    return rc;
}

int hipFreeAsync(void* ptr, void* stream) {
    static FreeAsyncFn fn = real<FreeAsyncFn>("hipFreeAsync");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    int rc = fn(ptr, stream);
    if (rc == HIP_SUCCESS) untrack(ptr);
    return rc;
}

int hipMemGetInfo(size_t* free_out, size_t* total_out) {
    static MemGetInfoFn fn = real<MemGetInfoFn>("hipMemGetInfo");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    int rc = fn(free_out, total_out);
    init_limit();
    if (rc == HIP_SUCCESS && g_limit >= 0) {
        // clamp to the container's budget so pool-sizing frameworks
        // (PyTorch "expandable_segments", fraction-of-free heuristics)
        // stay inside their share
        int64_t used = g_used.load();
        int64_t budget_free = g_limit > used ? g_limit - used : 0;
        if (total_out && (uint64_t)g_limit < (uint64_t)*total_out)
            *total_out = (size_t)g_limit;
        if (free_out && (uint64_t)budget_free < (uint64_t)*free_out)
            *free_out = (size_t)budget_free;
    }
    return rc;
}

// introspection for tests / debugging
int64_t gpushare_memguard_used() { return g_used.load(); }
int64_t gpushare_memguard_limit() {
    init_limit();
    return g_limit;
}

}  // extern "C"
