// Torch pluggable allocator backed by hipMallocAsync with the Spark OOM
// state machine wrapped around every allocation.
//
// The reference installs its SparkResourceAdaptor as RMM's current device
// resource so every libcudf allocation runs the retry protocol
// (SparkResourceAdaptorJni.cpp:2113-2141). The MI355X equivalent: torch's
// CUDAPluggableAllocator points at srj_torch_malloc/srj_torch_free in this
// file; the adaptor (host-side, _host.so) is reached through function
// pointers installed at runtime, and an optional byte cap turns the device
// into a deterministic small pool for OOM-protocol tests.
//
// Exceptions: the state machine's throw decisions surface as
// std::runtime_error with a Gpu*OOM marker prefix; torch re-raises them as
// RuntimeError and spark_rapids_jni_amd.memory maps them back to the typed
// exception hierarchy.
#include <hip/hip_runtime.h>
#include <pthread.h>

#include <atomic>
#include <cstdint>
#include <stdexcept>

namespace {

typedef int (*pre_fn_t)(long);
typedef int (*fail_fn_t)(long);
typedef void (*succ_fn_t)(long, long);
typedef void (*dealloc_fn_t)(long, long);

std::atomic<pre_fn_t> g_pre{nullptr};
std::atomic<fail_fn_t> g_fail{nullptr};
std::atomic<succ_fn_t> g_succ{nullptr};
std::atomic<dealloc_fn_t> g_dealloc{nullptr};
std::atomic<long long> g_limit{-1};
std::atomic<long long> g_used{0};

// same codes as src/host/resource_adaptor.cpp AllocResult
[[noreturn]] void throw_code(int code) {
  switch (code) {
    case 1: throw std::runtime_error(
        "GpuRetryOOM: device allocation failed, rollback and retry");
    case 2: throw std::runtime_error(
        "GpuSplitAndRetryOOM: device allocation failed, split and retry");
    case 3: throw std::runtime_error("CpuRetryOOM: host allocation failed");
    case 4: throw std::runtime_error(
        "CpuSplitAndRetryOOM: host allocation failed");
    case 5: throw std::runtime_error("ThreadRemoved: task shutting down");
    default: throw std::runtime_error(
        "srj device pool out of memory (no retry)");
  }
}

inline long cur_tid() {
  // must match Python's threading.get_ident() (pthread_self on CPython)
  return (long)pthread_self();
}

}  // namespace

extern "C" {

void srj_install_ra_hooks(uintptr_t pre, uintptr_t fail, uintptr_t succ,
                          uintptr_t dealloc) {
  g_pre.store((pre_fn_t)pre);
  g_fail.store((fail_fn_t)fail);
  g_succ.store((succ_fn_t)succ);
  g_dealloc.store((dealloc_fn_t)dealloc);
}

void srj_clear_ra_hooks() {
  g_pre.store(nullptr);
  g_fail.store(nullptr);
  g_succ.store(nullptr);
  g_dealloc.store(nullptr);
}

void srj_set_device_pool_limit(long long bytes) { g_limit.store(bytes); }

long long srj_device_pool_used() { return g_used.load(); }

void* srj_torch_malloc(size_t size, int device, hipStream_t stream) {
  if (size == 0) return nullptr;
  long tid = cur_tid();
  (void)hipSetDevice(device);
  for (;;) {
    if (auto pre = g_pre.load()) {
      int code = pre(tid);
      if (code != 0) throw_code(code);
    }
    void* ptr = nullptr;
    long long lim = g_limit.load();
    bool capped = lim >= 0 && g_used.load() + (long long)size > lim;
    if (!capped) {
      hipError_t e = hipMallocAsync(&ptr, size, stream);
      if (e != hipSuccess) {
        (void)hipGetLastError();
        ptr = nullptr;
      }
    }
    if (ptr != nullptr) {
      g_used.fetch_add((long long)size);
      if (auto succ = g_succ.load()) succ(tid, (long)size);
      return ptr;
    }
    auto fail = g_fail.load();
    if (!fail) throw std::runtime_error(
        "srj device pool out of memory (no adaptor installed)");
    int code = fail(tid);  // may block until another thread frees
    if (code == 0) continue;  // retry the allocation
    throw_code(code);
  }
}

void srj_torch_free(void* ptr, size_t size, int device, hipStream_t stream) {
  if (ptr != nullptr) {
    (void)hipSetDevice(device);
    (void)hipFreeAsync(ptr, stream);
    g_used.fetch_sub((long long)size);
  }
  if (auto d = g_dealloc.load()) d(cur_tid(), (long)size);
}

}  // extern "C"
