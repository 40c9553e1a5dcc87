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
// Accounting is POD-SCOPED and CRASH-SAFE: every process owns one slot
// {pid, used} in a /dev/shm-backed table (per-pod tmpfs in k8s, so tenants
// cannot collide), and the budget check sums the slots of processes that
// are still alive.  A worker killed with SIGTERM/SIGKILL repays its
// reservation implicitly — the kernel frees its GPU memory, and its slot
// is reclaimed by the next process that finds the pid dead.  torchrun
// fleets and dataloader children therefore share ONE budget.
//
// Deliberately linked against NOTHING but libdl/libc: HIP symbols resolve
// lazily from the app's own runtime (dlsym RTLD_NEXT, then a dlopen handle
// of the already-loaded libamdhip64 — PyTorch loads it RTLD_LOCAL), so
// preloading into non-GPU processes is a no-op.  hipMemGetInfo is clamped
// to the budget so frameworks that size pools from "free VRAM" stay inside
// their share.

#include <atomic>
#include <cerrno>
#include <csignal>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <dlfcn.h>
#include <fcntl.h>
#include <mutex>
#include <pthread.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>
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

struct Slot {
    std::atomic<int32_t> pid;
    std::atomic<int64_t> used;
};
constexpr int NSLOTS = 512;  // 8 KiB table

int64_t g_limit = -1;        // -1: unlimited (env absent) — pure passthrough
Slot* g_table = nullptr;     // shm table; null ⇒ process-local fallback
Slot g_local{{0}, {0}};
Slot* g_my = nullptr;        // this process's slot
int32_t g_my_pid = 0;

std::mutex g_sizes_mu;
std::unordered_map<void*, size_t>& sizes() {
    static std::unordered_map<void*, size_t> m;
    return m;
}

bool pid_alive(int32_t pid) {
    return kill(pid, 0) == 0 || errno != ESRCH;
}

void map_table() {
    const char* uid = std::getenv("GPUSHARE_POD_UID");
    char path[256];
    std::snprintf(path, sizeof(path), "/dev/shm/gpushare.memguard.%s",
                  (uid && *uid) ? uid : "pod");
    int fd = open(path, O_CREAT | O_RDWR | O_CLOEXEC, 0600);
    if (fd < 0) return;
    if (ftruncate(fd, sizeof(Slot) * NSLOTS) == 0) {
        void* p = mmap(nullptr, sizeof(Slot) * NSLOTS,
                       PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
        if (p != MAP_FAILED) g_table = reinterpret_cast<Slot*>(p);
    }
    close(fd);
}

Slot* claim_slot() {
    int32_t me = (int32_t)getpid();
    // pass 1: free slots; pass 2: reclaim dead owners
    for (int pass = 0; pass < 2; ++pass) {
        for (int i = 0; i < NSLOTS; ++i) {
            int32_t cur = g_table[i].pid.load(std::memory_order_relaxed);
            if (cur == me) {
                // our pid already owns a slot: we exec()'d (old image's GPU
                // memory is gone) or inherited a recycled pid — either way
                // the old reservation is dead
                g_table[i].used.store(0);
                return &g_table[i];
            }
            bool takeable =
                (pass == 0) ? cur == 0 : (cur != 0 && !pid_alive(cur));
            if (takeable &&
                g_table[i].pid.compare_exchange_strong(cur, me)) {
                g_table[i].used.store(0);
                return &g_table[i];
            }
        }
    }
    return &g_local;  // table full: enforce process-locally
}

void bind_slot() {
    int32_t me = (int32_t)getpid();
    if (g_my != nullptr && g_my_pid == me) return;
    // first call, or first call after fork(): take our own slot so a
    // child never charges (or repays) its parent's reservation
    g_my = (g_table != nullptr) ? claim_slot() : &g_local;
    if (g_my == &g_local) {
        g_local.pid.store(me);
        g_local.used.store(0);
    }
    g_my_pid = me;
}

void init_limit() {
    static std::once_flag once;
    std::call_once(once, [] {
        const char* env = std::getenv("GPUSHARE_MEM_LIMIT_BYTES");
        if (!env || !*env) return;
        char* end = nullptr;
        long long v = std::strtoll(env, &end, 10);
        if (end == env || v <= 0) return;
        g_limit = v;
        map_table();
    });
}

int64_t total_used(bool sweep_dead) {
    if (g_table == nullptr) return g_my->used.load();
    int64_t sum = 0;
    for (int i = 0; i < NSLOTS; ++i) {
        int32_t pid = g_table[i].pid.load(std::memory_order_relaxed);
        if (pid == 0) continue;
        if (sweep_dead && pid != g_my_pid && !pid_alive(pid)) {
            // dead owner: its GPU memory was freed by the kernel driver —
            // reclaim the reservation
            if (g_table[i].pid.compare_exchange_strong(pid, 0))
                g_table[i].used.store(0);
            continue;
        }
        sum += g_table[i].used.load(std::memory_order_relaxed);
    }
    return sum;
}

// serializes the check-then-commit against sibling processes well enough:
// each process reserves in ITS slot first, then validates the global sum,
// so concurrent racers can transiently overshoot by at most the in-flight
// requests — never lose a reservation
bool reserve(size_t size) {
    init_limit();
    if (g_limit < 0) return true;
    bind_slot();
    g_my->used.fetch_add((int64_t)size);
    if (total_used(false) > g_limit) {
        if (total_used(true) > g_limit) {  // sweep dead owners, re-check
            g_my->used.fetch_sub((int64_t)size);
            return false;
        }
    }
    return true;
}

void unreserve(size_t size) { g_my->used.fetch_sub((int64_t)size); }

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
    unreserve(size);
}

// clean exit: release the slot immediately (crash/SIGKILL exits are
// reclaimed lazily by pid_alive sweeps instead)
__attribute__((destructor)) void memguard_release_slot() {
    if (g_my != nullptr && g_my != &g_local && g_my_pid == (int32_t)getpid()) {
        g_my->used.store(0);
        g_my->pid.store(0);
    }
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

int guarded_alloc(MallocFn fn, void** ptr, size_t size) {
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    if (!reserve(size)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size);
    if (rc == HIP_SUCCESS) {
        track(*ptr, size);
    } else if (g_limit >= 0) {
        unreserve(size);
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
    else if (g_limit >= 0) unreserve(size);
    return rc;
}

int hipExtMallocWithFlags(void** ptr, size_t size, unsigned int flags) {
    static MallocFlagsFn fn = real<MallocFlagsFn>("hipExtMallocWithFlags");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    if (!reserve(size)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size, flags);
    if (rc == HIP_SUCCESS) track(*ptr, size);
    else if (g_limit >= 0) unreserve(size);
    return rc;
}

int hipMallocAsync(void** ptr, size_t size, void* stream) {
    static MallocAsyncFn fn = real<MallocAsyncFn>("hipMallocAsync");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    if (!reserve(size)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size, stream);
    if (rc == HIP_SUCCESS) track(*ptr, size);
    else if (g_limit >= 0) unreserve(size);
    return rc;
}

int hipMallocPitch(void** ptr, size_t* pitch, size_t width, size_t height) {
    using PitchFn = int (*)(void**, size_t*, size_t, size_t);
    static PitchFn fn = real<PitchFn>("hipMallocPitch");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    // the true footprint (pitch × height) is only known after the call:
    // allocate first, account after, undo if over budget
    int rc = fn(ptr, pitch, width, height);
    if (rc != HIP_SUCCESS) return rc;
    init_limit();
    if (g_limit < 0) return rc;
    size_t size = (*pitch) * height;
    if (!reserve(size)) {
        static FreeFn freer = real<FreeFn>("hipFree");
        if (freer != nullptr) freer(*ptr);
        return HIP_ERROR_OOM;
    }
    track(*ptr, size);
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
        // clamp to the pod's budget so pool-sizing frameworks (PyTorch
        // "expandable_segments", fraction-of-free heuristics) stay inside
        // their share
        bind_slot();
        int64_t used = total_used(false);
        int64_t budget_free = g_limit > used ? g_limit - used : 0;
        if (total_out && (uint64_t)g_limit < (uint64_t)*total_out)
            *total_out = (size_t)g_limit;
        if (free_out && (uint64_t)budget_free < (uint64_t)*free_out)
            *free_out = (size_t)budget_free;
    }
    return rc;
}

// introspection for tests / debugging
int64_t gpushare_memguard_used() {
    init_limit();
    if (g_limit < 0) return 0;
    bind_slot();
    return total_used(false);
}

int64_t gpushare_memguard_limit() {
    init_limit();
    return g_limit;
}

}  // extern "C"
