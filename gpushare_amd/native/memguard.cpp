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
// Accounting is CONTAINER-SCOPED and CRASH-SAFE: every process owns one
// slot {pid, used, used_per_device[]} in a /dev/shm-backed table and the
// budget check sums the slots of processes that are still alive.  The
// table path carries both the pod UID and a per-container token
// (GPUSHARE_CONTAINER_TOKEN): containers of one pod share /dev/shm but NOT
// a PID namespace, so a pod-wide table would let one container's liveness
// sweep (kill(pid,0)) wrongly reclaim a sibling container's live
// reservations.  Scoping the table per container makes every pid in a
// table resolvable, and matches the k8s resource model — `aliyun.com/
// gpu-mem` limits are per-container, and GPUSHARE_MEM_LIMIT_BYTES is
// already this container's share.  A worker killed with SIGTERM/SIGKILL
// repays its reservation implicitly — the kernel frees its GPU memory, and
// its slot is reclaimed by the next process that finds the pid dead.
// torchrun fleets and dataloader children therefore share ONE budget.
//
// Multi-GPU placements additionally carry per-device sub-budgets
// (GPUSHARE_MEM_LIMIT_BYTES_PER_DEVICE, comma-separated bytes aligned with
// the injected HIP_VISIBLE_DEVICES ordinals): the scheduler extender's
// xGMI split reserves different amounts on different physical GPUs, and a
// tenant that concentrated its whole pod budget on one split member would
// starve co-tenants binpacked there.  Each allocation is charged to the
// calling thread's current HIP device ordinal and checked against that
// ordinal's sub-budget as well as the container total.
//
// Deliberately linked against NOTHING but libdl/libc: HIP symbols resolve
// lazily from the app's own runtime (dlsym RTLD_NEXT, then a dlopen handle
// of the already-loaded libamdhip64 — PyTorch loads it RTLD_LOCAL), so
// preloading into non-GPU processes is a no-op.  hipMemGetInfo is clamped
// to the budget so frameworks that size pools from "free VRAM" stay inside
// their share.
//
// TRUST MODEL (see docs/operations.md "memguard trust model"): this is a
// cooperative-runtime boundary, not a security boundary.  Covered entry
// points: hipMalloc / hipMallocManaged / hipExtMallocWithFlags /
// hipMallocAsync / hipMallocFromPoolAsync / hipMallocPitch / the VMM
// family (hipMemCreate, hipMemRelease) and the corresponding frees.
// Out of scope by policy: hipHostMalloc (GTT/host RAM, not VRAM — it does
// not consume the shared HBM the budget protects), statically-linked HIP
// runtimes, and direct KFD ioctls.

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
using MallocPoolAsyncFn = int (*)(void**, size_t, void*, void*);
using FreeFn = int (*)(void*);
using FreeAsyncFn = int (*)(void*, void*);
using MemGetInfoFn = int (*)(size_t*, size_t*);
using GetDeviceFn = int (*)(int*);
using MemCreateFn = int (*)(void**, size_t, const void*, unsigned long long);
using MemReleaseFn = int (*)(void*);

// one physical GPU-sharing pod never spans more than one node's GPUs
constexpr int MAX_DEV = 8;

struct Slot {
    std::atomic<int32_t> pid;
    std::atomic<int32_t> pad_;
    std::atomic<int64_t> used;
    std::atomic<int64_t> used_dev[MAX_DEV];
};
constexpr int NSLOTS = 512;  // 40 KiB table

int64_t g_limit = -1;        // -1: unlimited (env absent) — pure passthrough
int64_t g_dev_limit[MAX_DEV];  // -1: no per-device cap for that ordinal
int g_ndev_limits = 0;
Slot* g_table = nullptr;     // shm table; null ⇒ process-local fallback
Slot g_local;
Slot* g_my = nullptr;        // this process's slot
int32_t g_my_pid = 0;

struct Alloc {
    size_t size;
    int dev;
};
std::mutex g_sizes_mu;
std::unordered_map<void*, Alloc>& sizes() {
    static std::unordered_map<void*, Alloc> m;
    return m;
}

bool pid_alive(int32_t pid) {
    return kill(pid, 0) == 0 || errno != ESRCH;
}

void map_table() {
    const char* uid = std::getenv("GPUSHARE_POD_UID");
    const char* token = std::getenv("GPUSHARE_CONTAINER_TOKEN");
    char path[256];
    std::snprintf(path, sizeof(path), "/dev/shm/gpushare.memguard.%s.%s",
                  (uid && *uid) ? uid : "pod",
                  (token && *token) ? token : "c");
    int fd = open(path, O_CREAT | O_RDWR | O_CLOEXEC, 0600);
    if (fd < 0) return;
    if (ftruncate(fd, sizeof(Slot) * NSLOTS) == 0) {
        void* p = mmap(nullptr, sizeof(Slot) * NSLOTS,
                       PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
        if (p != MAP_FAILED) g_table = reinterpret_cast<Slot*>(p);
    }
    close(fd);
}

void clear_slot(Slot* s) {
    s->used.store(0);
    for (int d = 0; d < MAX_DEV; ++d) s->used_dev[d].store(0);
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
                clear_slot(&g_table[i]);
                return &g_table[i];
            }
            bool takeable =
                (pass == 0) ? cur == 0 : (cur != 0 && !pid_alive(cur));
            if (takeable &&
                g_table[i].pid.compare_exchange_strong(cur, me)) {
                clear_slot(&g_table[i]);
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
        clear_slot(&g_local);
    }
    g_my_pid = me;
}

void init_limit() {
    static std::once_flag once;
    std::call_once(once, [] {
        for (int d = 0; d < MAX_DEV; ++d) g_dev_limit[d] = -1;
        const char* env = std::getenv("GPUSHARE_MEM_LIMIT_BYTES");
        if (!env || !*env) return;
        char* end = nullptr;
        long long v = std::strtoll(env, &end, 10);
        if (end == env || v <= 0) return;
        g_limit = v;
        const char* per = std::getenv("GPUSHARE_MEM_LIMIT_BYTES_PER_DEVICE");
        if (per && *per) {
            const char* p = per;
            for (int d = 0; d < MAX_DEV && *p; ++d) {
                char* stop = nullptr;
                long long dv = std::strtoll(p, &stop, 10);
                if (stop == p) break;
                if (dv >= 0) {
                    g_dev_limit[d] = dv;
                    g_ndev_limits = d + 1;
                }
                p = (*stop == ',') ? stop + 1 : stop;
            }
        }
        map_table();
    });
}

int64_t total_used(bool sweep_dead, int64_t* dev_sum = nullptr) {
    if (g_table == nullptr) {
        if (dev_sum != nullptr)
            for (int d = 0; d < MAX_DEV; ++d)
                dev_sum[d] = g_my->used_dev[d].load();
        return g_my->used.load();
    }
    int64_t sum = 0;
    if (dev_sum != nullptr)
        for (int d = 0; d < MAX_DEV; ++d) dev_sum[d] = 0;
    for (int i = 0; i < NSLOTS; ++i) {
        int32_t pid = g_table[i].pid.load(std::memory_order_relaxed);
        if (pid == 0) continue;
        if (sweep_dead && pid != g_my_pid && !pid_alive(pid)) {
            // dead owner (same PID namespace — the table is
            // container-scoped): its GPU memory was freed by the kernel
            // driver, reclaim the reservation
            if (g_table[i].pid.compare_exchange_strong(pid, 0))
                clear_slot(&g_table[i]);
            continue;
        }
        sum += g_table[i].used.load(std::memory_order_relaxed);
        if (dev_sum != nullptr)
            for (int d = 0; d < MAX_DEV; ++d)
                dev_sum[d] +=
                    g_table[i].used_dev[d].load(std::memory_order_relaxed);
    }
    return sum;
}

int current_device();  // fwd: defined after real<> (needs hip_handle)

bool over_dev_budget(int dev, const int64_t* dev_sum) {
    return dev < g_ndev_limits && g_dev_limit[dev] >= 0 &&
           dev_sum[dev] > g_dev_limit[dev];
}

// serializes the check-then-commit against sibling processes well enough:
// each process reserves in ITS slot first, then validates the global sum,
// so concurrent racers can transiently overshoot by at most the in-flight
// requests — never lose a reservation
bool reserve(size_t size, int dev) {
    init_limit();
    if (g_limit < 0) return true;
    bind_slot();
    g_my->used.fetch_add((int64_t)size);
    g_my->used_dev[dev].fetch_add((int64_t)size);
    int64_t dev_sum[MAX_DEV];
    if (total_used(false, dev_sum) > g_limit ||
        over_dev_budget(dev, dev_sum)) {
        // sweep dead owners, re-check
        if (total_used(true, dev_sum) > g_limit ||
            over_dev_budget(dev, dev_sum)) {
            g_my->used.fetch_sub((int64_t)size);
            g_my->used_dev[dev].fetch_sub((int64_t)size);
            return false;
        }
    }
    return true;
}

void unreserve(size_t size, int dev) {
    g_my->used.fetch_sub((int64_t)size);
    g_my->used_dev[dev].fetch_sub((int64_t)size);
}

void track(void* ptr, size_t size, int dev) {
    if (g_limit < 0 || ptr == nullptr) return;
    std::lock_guard<std::mutex> lk(g_sizes_mu);
    sizes()[ptr] = Alloc{size, dev};
}

void untrack(void* ptr) {
    if (g_limit < 0 || ptr == nullptr) return;
    Alloc a{0, 0};
    {
        std::lock_guard<std::mutex> lk(g_sizes_mu);
        auto it = sizes().find(ptr);
        if (it == sizes().end()) return;  // not ours (pre-preload alloc)
        a = it->second;
        sizes().erase(it);
    }
    unreserve(a.size, a.dev);
}

// fork hygiene: the child must not inherit the parent's pointer→size
// tracking (freeing an inherited pointer would drain the CHILD's fresh
// slot into negative territory), and g_sizes_mu must never be forked in
// a locked state (a parent thread mid-track would deadlock every child
// allocation forever).
__attribute__((constructor)) void memguard_fork_handlers() {
    pthread_atfork(
        [] { g_sizes_mu.lock(); },
        [] { g_sizes_mu.unlock(); },
        [] {
            g_sizes_mu.unlock();
            sizes().clear();  // child's own allocs start fresh; bind_slot
                              // already gives it a zeroed slot
        });
}

// clean exit: release the slot immediately (crash/SIGKILL exits are
// reclaimed lazily by pid_alive sweeps instead)
__attribute__((destructor)) void memguard_release_slot() {
    if (g_my != nullptr && g_my != &g_local && g_my_pid == (int32_t)getpid()) {
        clear_slot(g_my);
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

int current_device() {
    // same resolution path as the interposed entry points: PyTorch loads
    // libamdhip64 RTLD_LOCAL, so RTLD_NEXT alone never sees hipGetDevice
    // and per-device charging would silently pin to ordinal 0
    static GetDeviceFn fn = real<GetDeviceFn>("hipGetDevice");
    int dev = 0;
    if (fn != nullptr && fn(&dev) == HIP_SUCCESS && dev >= 0 &&
        dev < MAX_DEV)
        return dev;
    return 0;
}

int guarded_alloc(MallocFn fn, void** ptr, size_t size) {
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    int dev = current_device();
    if (!reserve(size, dev)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size);
    if (rc == HIP_SUCCESS) {
        track(*ptr, size, dev);
    } else if (g_limit >= 0) {
        unreserve(size, dev);
    }
    return rc;
}

// hipMemAllocationProp prefix, mirrored to avoid a HIP-header dependency
// (this library must keep linking against libdl/libc only).  Layout per
// hip_runtime_api.h: type(4) requestedHandleType(4) location{type(4),
// id(4)} — only read when the struct pointer is non-null and the location
// type says "device" (1).
struct MemLocationPrefix {
    int32_t type;
    int32_t requested_handle_type;
    int32_t location_type;
    int32_t location_id;
};

}  // namespace

extern "C" {

int hipMalloc(void** ptr, size_t size) {
    static MallocFn fn = real<MallocFn>("hipMalloc");
    return guarded_alloc(fn, ptr, size);
}

int hipMallocManaged(void** ptr, size_t size, unsigned int flags) {
    static MallocFlagsFn fn = real<MallocFlagsFn>("hipMallocManaged");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    int dev = current_device();
    if (!reserve(size, dev)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size, flags);
    if (rc == HIP_SUCCESS) track(*ptr, size, dev);
    else if (g_limit >= 0) unreserve(size, dev);
    return rc;
}

int hipExtMallocWithFlags(void** ptr, size_t size, unsigned int flags) {
    static MallocFlagsFn fn = real<MallocFlagsFn>("hipExtMallocWithFlags");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    int dev = current_device();
    if (!reserve(size, dev)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size, flags);
    if (rc == HIP_SUCCESS) track(*ptr, size, dev);
    else if (g_limit >= 0) unreserve(size, dev);
    return rc;
}

int hipMallocAsync(void** ptr, size_t size, void* stream) {
    static MallocAsyncFn fn = real<MallocAsyncFn>("hipMallocAsync");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    // stream-ordered allocs land on the stream's device; PyTorch (and HIP
    // semantics generally) set the thread's current device before the
    // call, so the current ordinal is the right charge target
    int dev = current_device();
    if (!reserve(size, dev)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size, stream);
    if (rc == HIP_SUCCESS) track(*ptr, size, dev);
    else if (g_limit >= 0) unreserve(size, dev);
    return rc;
}

int hipMallocFromPoolAsync(void** ptr, size_t size, void* pool, void* stream) {
    static MallocPoolAsyncFn fn =
        real<MallocPoolAsyncFn>("hipMallocFromPoolAsync");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    int dev = current_device();
    if (!reserve(size, dev)) return HIP_ERROR_OOM;
    int rc = fn(ptr, size, pool, stream);
    if (rc == HIP_SUCCESS) track(*ptr, size, dev);
    else if (g_limit >= 0) unreserve(size, dev);
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
    int dev = current_device();
    if (!reserve(size, dev)) {
        static FreeFn freer = real<FreeFn>("hipFree");
        if (freer != nullptr) freer(*ptr);
        return HIP_ERROR_OOM;
    }
    track(*ptr, size, dev);
    return rc;
}

// VMM family: hipMemCreate reserves physical VRAM (the expensive part);
// hipMemMap/hipMemUnmap only wire virtual ranges and are left alone.
// PyTorch's expandable_segments allocator is built on exactly this path.
int hipMemCreate(void** handle, size_t size, const void* prop,
                 unsigned long long flags) {
    static MemCreateFn fn = real<MemCreateFn>("hipMemCreate");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    int dev = current_device();
    if (prop != nullptr) {
        const auto* p = reinterpret_cast<const MemLocationPrefix*>(prop);
        // hipMemLocationTypeDevice == 1; id is a visible-device ordinal
        if (p->location_type == 1 && p->location_id >= 0 &&
            p->location_id < MAX_DEV)
            dev = p->location_id;
    }
    if (!reserve(size, dev)) return HIP_ERROR_OOM;
    int rc = fn(handle, size, prop, flags);
    if (rc == HIP_SUCCESS) track(*handle, size, dev);
    else if (g_limit >= 0) unreserve(size, dev);
    return rc;
}

int hipMemRelease(void* handle) {
    static MemReleaseFn fn = real<MemReleaseFn>("hipMemRelease");
    if (fn == nullptr) return HIP_ERROR_INVALID_VALUE;
    int rc = fn(handle);
    if (rc == HIP_SUCCESS) untrack(handle);
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
        // stay inside their share; with a per-device split, clamp to the
        // current ordinal's sub-budget
        bind_slot();
        int64_t dev_sum[MAX_DEV];
        int64_t used = total_used(false, dev_sum);
        int64_t budget_free = g_limit > used ? g_limit - used : 0;
        int64_t budget_total = g_limit;
        int dev = current_device();
        if (dev < g_ndev_limits && g_dev_limit[dev] >= 0) {
            budget_total = g_dev_limit[dev];
            int64_t dev_free = g_dev_limit[dev] > dev_sum[dev]
                                   ? g_dev_limit[dev] - dev_sum[dev]
                                   : 0;
            if (dev_free < budget_free) budget_free = dev_free;
        }
        if (total_out && (uint64_t)budget_total < (uint64_t)*total_out)
            *total_out = (size_t)budget_total;
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

int64_t gpushare_memguard_used_dev(int dev) {
    init_limit();
    if (g_limit < 0 || dev < 0 || dev >= MAX_DEV) return 0;
    bind_slot();
    int64_t dev_sum[MAX_DEV];
    total_used(false, dev_sum);
    return dev_sum[dev];
}

int64_t gpushare_memguard_limit() {
    init_limit();
    return g_limit;
}

int64_t gpushare_memguard_dev_limit(int dev) {
    init_limit();
    if (dev < 0 || dev >= MAX_DEV) return -1;
    return g_dev_limit[dev];
}

}  // extern "C"
