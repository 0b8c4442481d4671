// Kernel-level activity capture via roctracer (the MI355X analog of the
// reference's CUPTI activity API: ProfilerJni.cpp:267-292 registers CUPTI
// activity buffers for KERNEL/MEMCPY records; here the HIP_OPS activity
// domain delivers the same dispatch begin/end timestamps + kernel names).
//
// Exposed as a small C API consumed by spark_rapids_jni_amd/tools/
// profiler.py over ctypes:
//   srj_ktrace_start()                 enable HIP_OPS activity capture
//   srj_ktrace_stop()                  flush + disable
//   srj_ktrace_count()                 number of buffered records
//   srj_ktrace_get(i, name_buf, n, out6[])
//       -> copies record i: kind, begin_ns, end_ns, correlation, device,
//          queue into out6 and the kernel/op name into name_buf
//   srj_ktrace_clear()
#include <roctracer/roctracer.h>
#include <roctracer/roctracer_hip.h>

#include <cstdint>
#include <cstring>
#include <mutex>
#include <string>
#include <vector>

namespace {

struct KRecord {
  std::string name;
  uint32_t op;
  uint64_t begin_ns;
  uint64_t end_ns;
  uint64_t correlation;
  int device;
  uint64_t queue;
};

std::mutex g_mu;
std::vector<KRecord> g_records;
bool g_open = false;

void activity_cb(const char* begin, const char* end, void* /*arg*/) {
  const roctracer_record_t* record =
      reinterpret_cast<const roctracer_record_t*>(begin);
  const roctracer_record_t* end_record =
      reinterpret_cast<const roctracer_record_t*>(end);
  std::lock_guard<std::mutex> lk(g_mu);
  while (record < end_record) {
    if (record->domain == ACTIVITY_DOMAIN_HIP_OPS) {
      KRecord r;
      // the record union carries kernel_name ONLY for dispatch ops; copy
      // and barrier records overlay `bytes` there (reading it as a
      // pointer faults)
      bool const is_dispatch = record->op == HIP_OP_ID_DISPATCH;
      const char* opname =
          roctracer_op_string(record->domain, record->op, record->kind);
      if (is_dispatch && record->kernel_name != nullptr) {
        r.name = record->kernel_name;
      } else {
        r.name = opname ? opname : "hip_op";
      }
      r.op = record->op;
      r.begin_ns = record->begin_ns;
      r.end_ns = record->end_ns;
      r.correlation = record->correlation_id;
      r.device = record->device_id;
      r.queue = record->queue_id;
      g_records.push_back(std::move(r));
    }
    roctracer_next_record(record, &record);
  }
}

}  // namespace

extern "C" {

int srj_ktrace_start() {
  if (!g_open) {
    roctracer_properties_t props{};
    props.buffer_size = 1 << 20;
    props.buffer_callback_fun = activity_cb;
    if (roctracer_open_pool(&props) != ROCTRACER_STATUS_SUCCESS) return -1;
    g_open = true;
  }
  if (roctracer_enable_domain_activity(ACTIVITY_DOMAIN_HIP_OPS) !=
      ROCTRACER_STATUS_SUCCESS)
    return -2;
  return 0;
}

int srj_ktrace_stop() {
  roctracer_disable_domain_activity(ACTIVITY_DOMAIN_HIP_OPS);
  roctracer_flush_activity();
  return 0;
}

long srj_ktrace_count() {
  std::lock_guard<std::mutex> lk(g_mu);
  return (long)g_records.size();
}

int srj_ktrace_get(long i, char* name_buf, int name_cap, uint64_t* out6) {
  std::lock_guard<std::mutex> lk(g_mu);
  if (i < 0 || i >= (long)g_records.size()) return -1;
  const KRecord& r = g_records[i];
  std::snprintf(name_buf, name_cap, "%s", r.name.c_str());
  out6[0] = r.op;
  out6[1] = r.begin_ns;
  out6[2] = r.end_ns;
  out6[3] = r.correlation;
  out6[4] = (uint64_t)r.device;
  out6[5] = r.queue;
  return 0;
}

void srj_ktrace_clear() {
  std::lock_guard<std::mutex> lk(g_mu);
  g_records.clear();
}

}  // extern "C"
