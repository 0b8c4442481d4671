// HIP fault-injection preload shim.
//
// Reference parity: faultinj/faultinj.cu + faultinj/README.md — the reference
// loads a CUPTI callback library via CUDA_INJECTION64_PATH and substitutes
// CUDA return codes per a JSON config. HIP has no injection-path hook, so
// (as SURVEY.md §5.3 notes) the honest MI355X port is an LD_PRELOAD shim over
// the HIP runtime: interpose hipMalloc / hipMemcpyAsync / hipLaunchKernel /
// hipStreamSynchronize and return configured error codes with
// percent/count/seed matching; hot-reload when dynamic=true.
//
// Config (env SRJ_FAULT_INJECT_CONFIG = path to JSON, same shape as the
// reference's src/test/cpp/faultinj/test_faultinj.json):
//   {"seed": 42, "dynamic": false,
//    "faults": [{"name": "hipMalloc", "code": 2, "percent": 50, "count": -1}]}
// count >= 0 limits how many times the fault fires.
//
// Build: g++ -shared -fPIC src/tools/faultinj.cpp -o libsrjfaultinj.so -ldl
#define __HIP_PLATFORM_AMD__ 1
#include <dlfcn.h>
#include <hip/hip_runtime_api.h>
#include <sys/stat.h>

#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <mutex>
#include <random>
#include <string>
#include <vector>

namespace {

struct Fault {
  std::string name;
  int code = 1;        // hipError_t to substitute
  int percent = 100;   // probability of firing
  long count = -1;     // remaining fires (-1 = unlimited)
};

struct Config {
  std::vector<Fault> faults;
  unsigned seed = 12345;
  bool dynamic = false;
  time_t mtime = 0;
  std::string path;
};

std::mutex g_mu;
Config g_cfg;
std::mt19937* g_rng = nullptr;
bool g_loaded = false;

// minimal JSON scanning (flat schema only; avoids external deps)
long find_num(const std::string& s, size_t from, const char* key, long dflt) {
  size_t k = s.find(std::string("\"") + key + "\"", from);
  if (k == std::string::npos) return dflt;
  k = s.find(':', k);
  if (k == std::string::npos) return dflt;
  return strtol(s.c_str() + k + 1, nullptr, 10);
}

std::string find_str(const std::string& s, size_t from, const char* key) {
  size_t k = s.find(std::string("\"") + key + "\"", from);
  if (k == std::string::npos) return "";
  k = s.find(':', k);
  k = s.find('"', k);
  size_t e = s.find('"', k + 1);
  return s.substr(k + 1, e - k - 1);
}

void load_config_locked() {
  const char* p = getenv("SRJ_FAULT_INJECT_CONFIG");
  if (!p) {
    g_loaded = true;
    return;
  }
  struct stat st{};
  if (stat(p, &st) != 0) {
    g_loaded = true;
    return;
  }
  if (g_loaded && !g_cfg.dynamic) return;
  if (g_loaded && g_cfg.dynamic && st.st_mtime == g_cfg.mtime) return;
  FILE* f = fopen(p, "rb");
  if (!f) return;
  std::string s;
  char buf[4096];
  size_t n;
  while ((n = fread(buf, 1, sizeof buf, f)) > 0) s.append(buf, n);
  fclose(f);
  Config c;
  c.path = p;
  c.mtime = st.st_mtime;
  c.seed = (unsigned)find_num(s, 0, "seed", 12345);
  size_t d = s.find("\"dynamic\"");
  c.dynamic = d != std::string::npos && s.find("true", d) != std::string::npos;
  size_t pos = s.find("\"faults\"");
  while (pos != std::string::npos) {
    pos = s.find('{', pos + 1);
    if (pos == std::string::npos) break;
    size_t end = s.find('}', pos);
    if (end == std::string::npos) break;
    std::string obj = s.substr(pos, end - pos + 1);
    Fault ft;
    ft.name = find_str(obj, 0, "name");
    ft.code = (int)find_num(obj, 0, "code", 1);
    ft.percent = (int)find_num(obj, 0, "percent", 100);
    ft.count = find_num(obj, 0, "count", -1);
    if (!ft.name.empty()) c.faults.push_back(ft);
    pos = end + 1;
    if (s.find('{', pos) == std::string::npos) break;
  }
  g_cfg = c;
  delete g_rng;
  g_rng = new std::mt19937(c.seed);
  g_loaded = true;
  fprintf(stderr, "[srj-faultinj] loaded %zu fault(s) from %s\n",
          c.faults.size(), p);
}

bool should_fault(const char* api, int* code) {
  std::lock_guard<std::mutex> lk(g_mu);
  load_config_locked();
  for (auto& f : g_cfg.faults) {
    if (f.name != api) continue;
    if (f.count == 0) continue;
    int roll = (int)((*g_rng)() % 100);
    if (roll < f.percent) {
      if (f.count > 0) --f.count;
      *code = f.code;
      fprintf(stderr, "[srj-faultinj] injecting error %d into %s\n", f.code,
              api);
      return true;
    }
  }
  return false;
}

template <typename T>
T real(const char* name) {
  static_assert(sizeof(T) == sizeof(void*), "fn ptr");
  void* p = dlsym(RTLD_NEXT, name);
  if (!p) {
    // torch dlopens libamdhip64 after the preload (RTLD_LOCAL), so
    // RTLD_NEXT may not see it — resolve through an explicit handle.
    static void* h = [] {
      void* x = dlopen("libamdhip64.so", RTLD_LAZY | RTLD_NOLOAD);
      if (!x) x = dlopen("libamdhip64.so.7", RTLD_LAZY | RTLD_NOLOAD);
      if (!x) x = dlopen("libamdhip64.so", RTLD_LAZY);
      return x;
    }();
    if (h) p = dlsym(h, name);
  }
  if (!p) {
    fprintf(stderr, "[srj-faultinj] FATAL: cannot resolve %s\n", name);
    abort();
  }
  return reinterpret_cast<T>(p);
}

}  // namespace

extern "C" {

hipError_t hipMalloc(void** ptr, size_t size) {
  int code;
  if (should_fault("hipMalloc", &code)) return (hipError_t)code;
  static auto fn = real<hipError_t (*)(void**, size_t)>("hipMalloc");
  return fn(ptr, size);
}

hipError_t hipMemcpyAsync(void* dst, const void* src, size_t n, hipMemcpyKind k,
                          hipStream_t s) {
  int code;
  if (should_fault("hipMemcpyAsync", &code)) return (hipError_t)code;
  static auto fn = real<hipError_t (*)(void*, const void*, size_t, hipMemcpyKind,
                                       hipStream_t)>("hipMemcpyAsync");
  return fn(dst, src, n, k, s);
}

hipError_t hipLaunchKernel(const void* f, dim3 grid, dim3 block, void** args,
                           size_t shared, hipStream_t stream) {
  int code;
  if (should_fault("hipLaunchKernel", &code)) return (hipError_t)code;
  static auto fn =
      real<hipError_t (*)(const void*, dim3, dim3, void**, size_t, hipStream_t)>(
          "hipLaunchKernel");
  return fn(f, grid, block, args, shared, stream);
}

hipError_t hipStreamSynchronize(hipStream_t s) {
  int code;
  if (should_fault("hipStreamSynchronize", &code)) return (hipError_t)code;
  static auto fn = real<hipError_t (*)(hipStream_t)>("hipStreamSynchronize");
  return fn(s);
}

hipError_t hipDeviceSynchronize(void) {
  int code;
  if (should_fault("hipDeviceSynchronize", &code)) return (hipError_t)code;
  static auto fn = real<hipError_t (*)(void)>("hipDeviceSynchronize");
  return fn();
}

}  // extern "C"
