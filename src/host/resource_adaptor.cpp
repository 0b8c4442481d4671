// Spark OOM-retry resource adaptor: the L1 memory/resource state machine.
//
// Re-implementation (fresh, MI355X build) of the semantics of the reference's
// SparkResourceAdaptorJni.cpp (spark-rapids-jni, 2,903 LoC) and
// docs/memory_management.md: every allocation by a registered task thread runs
// through a 9-state machine; on OOM the thread blocks until memory frees; if
// every active task is blocked, the lowest-priority thread is rolled back
// (GpuRetryOOM -> spill + retry); if every thread is already rolled back
// (BUFN), the lowest-priority one splits its input (GpuSplitAndRetryOOM).
//
// Differences from the reference by design:
//  * the pool being adapted is torch's ROCm caching allocator (288 GB HBM3E)
//    rather than RMM; this core also supports a simulated pool so the whole
//    machine is testable on CPU (the reference tests against a small real
//    pool; same coverage intent, see RmmSparkTest.java / RmmSparkMonteCarlo).
//  * Python threads stand in for JVM threads; blocking calls release the GIL.
//
// State enum mirrors docs/memory_management.md (reference
// SparkResourceAdaptorJni.cpp:91-105).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstdint>
#include <map>
#include <mutex>
#include <set>
#include <string>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

namespace srj_host {

enum class ThreadState : int {
  UNKNOWN = -1,
  RUNNING = 0,
  ALLOC = 1,
  ALLOC_FREE = 2,
  BLOCKED = 3,
  BUFN_THROW = 4,
  BUFN_WAIT = 5,
  BUFN = 6,
  SPLIT_THROW = 7,
  REMOVE_THROW = 8,
};

const char* state_name(ThreadState s) {
  switch (s) {
    case ThreadState::RUNNING: return "THREAD_RUNNING";
    case ThreadState::ALLOC: return "THREAD_ALLOC";
    case ThreadState::ALLOC_FREE: return "THREAD_ALLOC_FREE";
    case ThreadState::BLOCKED: return "THREAD_BLOCKED";
    case ThreadState::BUFN_THROW: return "THREAD_BUFN_THROW";
    case ThreadState::BUFN_WAIT: return "THREAD_BUFN_WAIT";
    case ThreadState::BUFN: return "THREAD_BUFN";
    case ThreadState::SPLIT_THROW: return "THREAD_SPLIT_THROW";
    case ThreadState::REMOVE_THROW: return "THREAD_REMOVE_THROW";
    default: return "UNKNOWN";
  }
}

// Result codes surfaced to Python (raised as the typed OOM exceptions that
// mirror the reference's GpuRetryOOM / GpuSplitAndRetryOOM / ... hierarchy).
enum AllocResult : int {
  ALLOC_OK = 0,
  THROW_GPU_RETRY = 1,
  THROW_GPU_SPLIT = 2,
  THROW_CPU_RETRY = 3,
  THROW_CPU_SPLIT = 4,
  THROW_REMOVED = 5,
  ALLOC_FAILED_NO_RETRY = 6,  // unregistered thread, plain failure
};

struct TaskMetrics {
  int64_t num_retry = 0;
  int64_t num_split_retry = 0;
  int64_t block_time_ns = 0;
  int64_t lost_compute_ns = 0;
  int64_t max_memory = 0;      // high-water of task footprint
  int64_t current_memory = 0;  // live footprint
};

struct ThreadInfo {
  long tid;
  ThreadState state = ThreadState::RUNNING;
  std::vector<long> task_ids;  // dedicated: exactly 1; pool: 0..n
  bool is_dedicated = false;
  bool is_shuffle = false;
  long priority = 0;    // higher value = LOWER priority (picked first)
  int retry_oom_injected = 0;
  int split_oom_injected = 0;
  int cpu_retry_oom_injected = 0;
  bool in_spill_range = false;
  int64_t blocked_since_ns = 0;
  std::condition_variable cv;
};

static int64_t now_ns() {
  return std::chrono::duration_cast<std::chrono::nanoseconds>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

class SparkResourceAdaptor {
 public:
  explicit SparkResourceAdaptor(int64_t pool_limit = -1,
                                int64_t host_limit = -1)
      : pool_limit_(pool_limit), host_limit_(host_limit) {}

  // ---- thread/task association (RmmSpark.java:176-260 semantics) ----------
  void start_dedicated_task_thread(long tid, long task_id) {
    std::unique_lock<std::mutex> lk(mu_);
    auto& t = threads_[tid];
    t.tid = tid;
    t.is_dedicated = true;
    if (std::find(t.task_ids.begin(), t.task_ids.end(), task_id) ==
        t.task_ids.end())
      t.task_ids.push_back(task_id);
    t.priority = task_priority(task_id);
    active_tasks_.insert(task_id);
    log_state(lk, tid, "start_dedicated_task_thread");
  }

  void pool_thread_working_on_tasks(long tid, const std::vector<long>& tasks,
                                    bool is_shuffle) {
    std::unique_lock<std::mutex> lk(mu_);
    auto& t = threads_[tid];
    t.tid = tid;
    t.is_dedicated = false;
    t.is_shuffle = is_shuffle;
    for (long task : tasks) {
      if (std::find(t.task_ids.begin(), t.task_ids.end(), task) ==
          t.task_ids.end())
        t.task_ids.push_back(task);
      active_tasks_.insert(task);
    }
    t.priority = is_shuffle ? -1 : lowest_task_priority(t.task_ids);
    log_state(lk, tid, "pool_thread_working_on_tasks");
  }

  void pool_thread_finished_for_tasks(long tid, const std::vector<long>& tasks) {
    std::unique_lock<std::mutex> lk(mu_);
    auto it = threads_.find(tid);
    if (it == threads_.end()) return;
    for (long task : tasks) {
      auto& v = it->second.task_ids;
      v.erase(std::remove(v.begin(), v.end(), task), v.end());
    }
    if (it->second.task_ids.empty()) remove_thread(lk, tid);
  }

  void remove_thread_association(long tid) {
    std::unique_lock<std::mutex> lk(mu_);
    remove_thread(lk, tid);
  }

  void task_done(long task_id) {
    std::unique_lock<std::mutex> lk(mu_);
    active_tasks_.erase(task_id);
    std::vector<long> to_remove;
    for (auto& [tid, t] : threads_) {
      auto& v = t.task_ids;
      v.erase(std::remove(v.begin(), v.end(), task_id), v.end());
      if (v.empty()) to_remove.push_back(tid);
    }
    for (long tid : to_remove) remove_thread(lk, tid);
    // progress was made: wake every BUFN thread to retry
    for (auto& [tid, t] : threads_) {
      if (t.state == ThreadState::BUFN) {
        set_state(lk, t, ThreadState::RUNNING, "task_done wakes BUFN");
        t.cv.notify_all();
      }
    }
    wake_blocked(lk, "task_done");
  }

  // ---- the allocation protocol -------------------------------------------
  // Simulated-pool allocate: runs the full pre_alloc/alloc/post_alloc loop of
  // the reference's allocate() (SparkResourceAdaptorJni.cpp:2113-2141).
  int alloc_sim(long tid, int64_t nbytes) {
    std::unique_lock<std::mutex> lk(mu_);
    auto it = threads_.find(tid);
    if (it == threads_.end()) {
      // untracked thread: plain pool semantics
      if (pool_limit_ >= 0 && pool_used_ + nbytes > pool_limit_)
        return ALLOC_FAILED_NO_RETRY;
      pool_used_ += nbytes;
      return ALLOC_OK;
    }
    while (true) {
      ThreadInfo& t = threads_.at(tid);
      int pre = pre_alloc(lk, t, /*is_gpu=*/true);
      if (pre != ALLOC_OK) return pre;
      bool fits = pool_limit_ < 0 || pool_used_ + nbytes <= pool_limit_;
      bool inject = t.retry_oom_injected > 0 || t.split_oom_injected > 0;
      if (fits && !inject) {
        pool_used_ += nbytes;
        post_alloc_success(lk, t, nbytes);
        return ALLOC_OK;
      }
      int post = post_alloc_failed(lk, t, /*was_oom=*/true, /*is_gpu=*/true);
      if (post != ALLOC_OK) return post;
      // post == OK means "retry the allocation now"
    }
  }

  void dealloc_sim(long tid, int64_t nbytes) {
    std::unique_lock<std::mutex> lk(mu_);
    pool_used_ -= nbytes;
    if (pool_used_ < 0) pool_used_ = 0;
    track_dealloc(lk, tid, nbytes);
    wake_blocked(lk, "dealloc");
  }

  // External-pool hooks (torch allocator integration): the caller performs
  // the real hipMalloc-backed allocation between pre/post.
  int pre_alloc_external(long tid) {
    std::unique_lock<std::mutex> lk(mu_);
    auto it = threads_.find(tid);
    if (it == threads_.end()) return ALLOC_OK;
    return pre_alloc(lk, it->second, true);
  }

  void post_alloc_success_external(long tid, int64_t nbytes) {
    std::unique_lock<std::mutex> lk(mu_);
    auto it = threads_.find(tid);
    if (it == threads_.end()) return;
    post_alloc_success(lk, it->second, nbytes);
  }

  // returns ALLOC_OK if the caller should retry the allocation
  int post_alloc_failed_external(long tid, bool was_oom) {
    std::unique_lock<std::mutex> lk(mu_);
    auto it = threads_.find(tid);
    if (it == threads_.end()) return ALLOC_FAILED_NO_RETRY;
    return post_alloc_failed(lk, it->second, was_oom, true);
  }

  void dealloc_external(long tid, int64_t nbytes) {
    std::unique_lock<std::mutex> lk(mu_);
    track_dealloc(lk, tid, nbytes);
    wake_blocked(lk, "dealloc_external");
  }

  // block after catching a retry OOM (RmmSpark.blockThreadUntilReady)
  int block_thread_until_ready(long tid) {
    std::unique_lock<std::mutex> lk(mu_);
    auto it = threads_.find(tid);
    if (it == threads_.end()) return ALLOC_OK;
    ThreadInfo& t = it->second;
    if (t.state == ThreadState::BUFN_WAIT) {
      set_state(lk, t, ThreadState::BUFN, "block_thread_until_ready");
      check_and_break_deadlocks(lk);
      int64_t start = now_ns();
      t.blocked_since_ns = start;
      while (t.state == ThreadState::BUFN) t.cv.wait(lk);
      add_block_time(lk, t, now_ns() - start);
      if (t.state == ThreadState::SPLIT_THROW) {
        // the split victim is the designated progress-maker: it halves its
        // input and retries immediately (doc step 3) — do not re-block it
        set_state(lk, t, ThreadState::RUNNING, "split thrown");
        bump_task_metric(lk, t, &TaskMetrics::num_split_retry);
        return THROW_GPU_SPLIT;
      }
      if (t.state == ThreadState::REMOVE_THROW) {
        remove_thread(lk, tid);
        return THROW_REMOVED;
      }
    }
    return ALLOC_OK;
  }

  // ---- OOM injection (RmmSpark.forceRetryOOM / forceSplitAndRetryOOM) -----
  void force_retry_oom(long tid, int count) {
    std::unique_lock<std::mutex> lk(mu_);
    threads_.at(tid).retry_oom_injected = count;
  }
  void force_split_and_retry_oom(long tid, int count) {
    std::unique_lock<std::mutex> lk(mu_);
    threads_.at(tid).split_oom_injected = count;
  }

  // ---- spill ranges (RmmSpark.spillRangeStart/Done) -----------------------
  void spill_range_start(long tid) {
    std::unique_lock<std::mutex> lk(mu_);
    auto it = threads_.find(tid);
    if (it != threads_.end()) it->second.in_spill_range = true;
  }
  void spill_range_done(long tid) {
    std::unique_lock<std::mutex> lk(mu_);
    auto it = threads_.find(tid);
    if (it != threads_.end()) it->second.in_spill_range = false;
  }

  // ---- introspection / metrics -------------------------------------------
  std::string get_state_of(long tid) {
    std::unique_lock<std::mutex> lk(mu_);
    auto it = threads_.find(tid);
    return it == threads_.end() ? "UNKNOWN" : state_name(it->second.state);
  }

  int64_t pool_used() {
    std::unique_lock<std::mutex> lk(mu_);
    return pool_used_;
  }

  py::dict get_and_reset_metrics(long task_id) {
    std::unique_lock<std::mutex> lk(mu_);
    TaskMetrics m = metrics_[task_id];
    metrics_[task_id] = TaskMetrics{};
    metrics_[task_id].current_memory = m.current_memory;
    metrics_[task_id].max_memory = m.current_memory;
    py::dict d;
    d["num_retry"] = m.num_retry;
    d["num_split_retry"] = m.num_split_retry;
    d["block_time_ns"] = m.block_time_ns;
    d["lost_compute_ns"] = m.lost_compute_ns;
    d["max_memory"] = m.max_memory;
    d["current_memory"] = m.current_memory;
    return d;
  }

  std::vector<std::string> state_log() {
    std::unique_lock<std::mutex> lk(mu_);
    return log_;
  }
  void enable_logging(bool on) {
    std::unique_lock<std::mutex> lk(mu_);
    logging_ = on;
  }

 private:
  // priority: monotonic per task (reference task_priority.hpp:26 semantics;
  // later tasks get numerically larger values = lower priority; shuffle
  // threads use -1 = highest and are never picked as deadlock victims).
  long task_priority(long task_id) {
    auto it = task_prio_.find(task_id);
    if (it != task_prio_.end()) return it->second;
    long p = next_prio_++;
    task_prio_[task_id] = p;
    return p;
  }
  long lowest_task_priority(const std::vector<long>& tasks) {
    long best = 0;
    for (long t : tasks) best = std::max(best, task_priority(t));
    return best;
  }

  void set_state(std::unique_lock<std::mutex>& lk, ThreadInfo& t,
                 ThreadState s, const char* why) {
    if (logging_)
      log_.push_back(std::to_string(t.tid) + "," + state_name(t.state) + "," +
                     state_name(s) + "," + why);
    t.state = s;
  }

  void log_state(std::unique_lock<std::mutex>&, long tid, const char* what) {
    if (logging_) log_.push_back(std::to_string(tid) + ",-," + "-" + "," + what);
  }

  void remove_thread(std::unique_lock<std::mutex>& lk, long tid) {
    auto it = threads_.find(tid);
    if (it == threads_.end()) return;
    ThreadInfo& t = it->second;
    if (t.state == ThreadState::BLOCKED || t.state == ThreadState::BUFN) {
      // wake it to throw (REMOVE_THROW path)
      set_state(lk, t, ThreadState::REMOVE_THROW, "remove while blocked");
      t.cv.notify_all();
      return;  // the woken thread removes itself
    }
    threads_.erase(it);
  }

  int pre_alloc(std::unique_lock<std::mutex>& lk, ThreadInfo& t, bool is_gpu) {
    switch (t.state) {
      case ThreadState::RUNNING:
        set_state(lk, t, ThreadState::ALLOC, "pre_alloc");
        return ALLOC_OK;
      case ThreadState::BUFN_WAIT: {
        // must block before allocating again
        set_state(lk, t, ThreadState::BUFN, "pre_alloc while BUFN_WAIT");
        check_and_break_deadlocks(lk);
        int64_t start = now_ns();
        while (t.state == ThreadState::BUFN) t.cv.wait(lk);
        add_block_time(lk, t, now_ns() - start);
        if (t.state == ThreadState::SPLIT_THROW) {
          set_state(lk, t, ThreadState::RUNNING, "split thrown");
          bump_task_metric(lk, t, &TaskMetrics::num_split_retry);
          return THROW_GPU_SPLIT;
        }
        if (t.state == ThreadState::REMOVE_THROW) {
          long tid = t.tid;
          remove_thread(lk, tid);
          return THROW_REMOVED;
        }
        set_state(lk, t, ThreadState::ALLOC, "pre_alloc after BUFN wake");
        return ALLOC_OK;
      }
      case ThreadState::ALLOC:
      case ThreadState::ALLOC_FREE:
        // nested alloc (recursive) — treat as running
        return ALLOC_OK;
      default:
        set_state(lk, t, ThreadState::ALLOC, "pre_alloc from unexpected state");
        return ALLOC_OK;
    }
  }

  void post_alloc_success(std::unique_lock<std::mutex>& lk, ThreadInfo& t,
                          int64_t nbytes) {
    set_state(lk, t, ThreadState::RUNNING, "alloc success");
    if (!t.in_spill_range) {
      for (long task : t.task_ids) {
        auto& m = metrics_[task];
        m.current_memory += nbytes;
        m.max_memory = std::max(m.max_memory, m.current_memory);
      }
    }
    // an allocation succeeded: progress. Wake BUFN threads so they retry.
    for (auto& [tid2, t2] : threads_) {
      if (t2.state == ThreadState::BUFN) {
        set_state(lk, t2, ThreadState::RUNNING, "progress wakes BUFN");
        t2.cv.notify_all();
      }
    }
  }

  int post_alloc_failed(std::unique_lock<std::mutex>& lk, ThreadInfo& t,
                        bool was_oom, bool is_gpu) {
    if (t.retry_oom_injected > 0) {
      t.retry_oom_injected--;
      set_state(lk, t, ThreadState::BUFN_WAIT, "injected retry OOM");
      bump_task_metric(lk, t, &TaskMetrics::num_retry);
      return is_gpu ? THROW_GPU_RETRY : THROW_CPU_RETRY;
    }
    if (t.split_oom_injected > 0) {
      t.split_oom_injected--;
      set_state(lk, t, ThreadState::RUNNING, "injected split OOM");
      bump_task_metric(lk, t, &TaskMetrics::num_split_retry);
      return is_gpu ? THROW_GPU_SPLIT : THROW_CPU_SPLIT;
    }
    if (!was_oom) return ALLOC_FAILED_NO_RETRY;
    if (t.state == ThreadState::ALLOC_FREE) {
      // a free landed while we were allocating: retry immediately
      set_state(lk, t, ThreadState::ALLOC, "retry after concurrent free");
      return ALLOC_OK;
    }
    set_state(lk, t, ThreadState::BLOCKED, "alloc failed; waiting for free");
    check_and_break_deadlocks(lk);
    int64_t start = now_ns();
    while (t.state == ThreadState::BLOCKED) t.cv.wait(lk);
    add_block_time(lk, t, now_ns() - start);
    switch (t.state) {
      case ThreadState::ALLOC:  // woken by a free: retry
        return ALLOC_OK;
      case ThreadState::BUFN_THROW:
        set_state(lk, t, ThreadState::BUFN_WAIT, "retry OOM thrown");
        bump_task_metric(lk, t, &TaskMetrics::num_retry);
        return is_gpu ? THROW_GPU_RETRY : THROW_CPU_RETRY;
      case ThreadState::SPLIT_THROW:
        set_state(lk, t, ThreadState::RUNNING, "split OOM thrown");
        bump_task_metric(lk, t, &TaskMetrics::num_split_retry);
        return is_gpu ? THROW_GPU_SPLIT : THROW_CPU_SPLIT;
      case ThreadState::REMOVE_THROW: {
        long tid = t.tid;
        remove_thread(lk, tid);
        return THROW_REMOVED;
      }
      default:
        return ALLOC_OK;
    }
  }

  void track_dealloc(std::unique_lock<std::mutex>& lk, long tid, int64_t nbytes) {
    auto it = threads_.find(tid);
    if (it != threads_.end() && !it->second.in_spill_range) {
      for (long task : it->second.task_ids) {
        metrics_[task].current_memory -= nbytes;
        if (metrics_[task].current_memory < 0) metrics_[task].current_memory = 0;
      }
    }
  }

  void wake_blocked(std::unique_lock<std::mutex>& lk, const char* why) {
    for (auto& [tid, t] : threads_) {
      if (t.state == ThreadState::BLOCKED) {
        set_state(lk, t, ThreadState::ALLOC, why);
        t.cv.notify_all();
      } else if (t.state == ThreadState::ALLOC) {
        set_state(lk, t, ThreadState::ALLOC_FREE, why);
      }
    }
  }

  // Deadlock rules from docs/memory_management.md: a task is blocked if any
  // dedicated thread is blocked AND all its pool threads are blocked. When
  // every active task is blocked: pick the lowest-priority BLOCKED thread ->
  // BUFN_THROW; if none is BLOCKED (all BUFN) -> lowest-priority -> SPLIT.
  void check_and_break_deadlocks(std::unique_lock<std::mutex>& lk) {
    if (threads_.empty()) return;
    auto is_blocked = [](const ThreadInfo& t) {
      return t.state == ThreadState::BLOCKED || t.state == ThreadState::BUFN;
    };
    std::set<long> blocked_tasks, running_tasks;
    for (auto& [tid, t] : threads_) {
      for (long task : t.task_ids) {
        if (t.is_dedicated) {
          if (is_blocked(t)) blocked_tasks.insert(task);
        } else {
          if (!is_blocked(t)) running_tasks.insert(task);  // pool not blocked
        }
      }
    }
    for (long task : active_tasks_) {
      bool task_blocked =
          blocked_tasks.count(task) > 0 && running_tasks.count(task) == 0;
      if (!task_blocked && has_threads_for(task)) return;  // someone can run
    }
    // deadlock: find victim among BLOCKED first
    ThreadInfo* victim = nullptr;
    for (auto& [tid, t] : threads_) {
      if (t.state == ThreadState::BLOCKED && !t.is_shuffle) {
        if (!victim || t.priority > victim->priority ||
            (t.priority == victim->priority && t.tid > victim->tid))
          victim = &t;
      }
    }
    if (victim) {
      set_state(lk, *victim, ThreadState::BUFN_THROW, "deadlock: rollback");
      victim->cv.notify_all();
      return;
    }
    for (auto& [tid, t] : threads_) {
      if (t.state == ThreadState::BUFN && !t.is_shuffle) {
        if (!victim || t.priority > victim->priority ||
            (t.priority == victim->priority && t.tid > victim->tid))
          victim = &t;
      }
    }
    if (victim) {
      set_state(lk, *victim, ThreadState::SPLIT_THROW, "deadlock: split");
      victim->cv.notify_all();
    }
  }

  bool has_threads_for(long task) {
    for (auto& [tid, t] : threads_)
      if (std::find(t.task_ids.begin(), t.task_ids.end(), task) !=
          t.task_ids.end())
        return true;
    return false;
  }

  void add_block_time(std::unique_lock<std::mutex>& lk, ThreadInfo& t,
                      int64_t ns) {
    for (long task : t.task_ids) metrics_[task].block_time_ns += ns;
  }
  void bump_task_metric(std::unique_lock<std::mutex>& lk, ThreadInfo& t,
                        int64_t TaskMetrics::*field) {
    for (long task : t.task_ids) metrics_[task].*field += 1;
  }

  std::mutex mu_;
  std::map<long, ThreadInfo> threads_;
  std::set<long> active_tasks_;
  std::unordered_map<long, TaskMetrics> metrics_;
  std::unordered_map<long, long> task_prio_;
  long next_prio_ = 0;
  int64_t pool_limit_;
  int64_t host_limit_;
  int64_t pool_used_ = 0;
  bool logging_ = false;
  std::vector<std::string> log_;
};

}  // namespace srj_host

// ---------------------------------------------------------------------------
// Bridge for the torch pluggable allocator (src/gpu/torch_alloc.hip):
// stable C entry points over the CURRENT adaptor so real hipMallocAsync
// allocations run the same pre/post state machine the reference wraps
// around every RMM allocation (SparkResourceAdaptorJni.cpp:2113-2141).
// The GPU-side allocator receives these function addresses from Python
// (no cross-.so link needed).
// ---------------------------------------------------------------------------
static srj_host::SparkResourceAdaptor* g_current_adaptor = nullptr;
static std::mutex g_current_mu;

extern "C" int srj_ra_pre_alloc(long tid) {
  auto* a = g_current_adaptor;
  return a ? a->pre_alloc_external(tid) : 0;
}
extern "C" int srj_ra_post_alloc_failed(long tid) {
  auto* a = g_current_adaptor;
  return a ? a->post_alloc_failed_external(tid, /*was_oom=*/true) : 6;
}
extern "C" void srj_ra_post_alloc_success(long tid, long nbytes) {
  auto* a = g_current_adaptor;
  if (a) a->post_alloc_success_external(tid, nbytes);
}
extern "C" void srj_ra_dealloc(long tid, long nbytes) {
  auto* a = g_current_adaptor;
  if (a) a->dealloc_external(tid, nbytes);
}

void register_resource_adaptor(py::module_& m) {
  using srj_host::SparkResourceAdaptor;
  m.def("install_as_current",
        [](SparkResourceAdaptor& a) {
          std::lock_guard<std::mutex> lk(g_current_mu);
          g_current_adaptor = &a;
        },
        "Route the torch-allocator bridge through this adaptor (the Python "
        "owner must keep it alive until clear_current).");
  m.def("clear_current", []() {
    std::lock_guard<std::mutex> lk(g_current_mu);
    g_current_adaptor = nullptr;
  });
  m.def("hook_addrs", []() {
    return py::make_tuple((uintptr_t)&srj_ra_pre_alloc,
                          (uintptr_t)&srj_ra_post_alloc_failed,
                          (uintptr_t)&srj_ra_post_alloc_success,
                          (uintptr_t)&srj_ra_dealloc);
  });
  py::class_<SparkResourceAdaptor>(m, "SparkResourceAdaptor")
      .def(py::init<int64_t, int64_t>(), py::arg("pool_limit") = -1,
           py::arg("host_limit") = -1)
      .def("start_dedicated_task_thread",
           &SparkResourceAdaptor::start_dedicated_task_thread)
      .def("pool_thread_working_on_tasks",
           &SparkResourceAdaptor::pool_thread_working_on_tasks)
      .def("pool_thread_finished_for_tasks",
           &SparkResourceAdaptor::pool_thread_finished_for_tasks)
      .def("remove_thread_association",
           &SparkResourceAdaptor::remove_thread_association)
      .def("task_done", &SparkResourceAdaptor::task_done)
      .def("alloc_sim", &SparkResourceAdaptor::alloc_sim,
           py::call_guard<py::gil_scoped_release>())
      .def("dealloc_sim", &SparkResourceAdaptor::dealloc_sim)
      .def("pre_alloc_external", &SparkResourceAdaptor::pre_alloc_external,
           py::call_guard<py::gil_scoped_release>())
      .def("post_alloc_success_external",
           &SparkResourceAdaptor::post_alloc_success_external)
      .def("post_alloc_failed_external",
           &SparkResourceAdaptor::post_alloc_failed_external,
           py::call_guard<py::gil_scoped_release>())
      .def("dealloc_external", &SparkResourceAdaptor::dealloc_external)
      .def("block_thread_until_ready",
           &SparkResourceAdaptor::block_thread_until_ready,
           py::call_guard<py::gil_scoped_release>())
      .def("force_retry_oom", &SparkResourceAdaptor::force_retry_oom)
      .def("force_split_and_retry_oom",
           &SparkResourceAdaptor::force_split_and_retry_oom)
      .def("spill_range_start", &SparkResourceAdaptor::spill_range_start)
      .def("spill_range_done", &SparkResourceAdaptor::spill_range_done)
      .def("get_state_of", &SparkResourceAdaptor::get_state_of)
      .def("pool_used", &SparkResourceAdaptor::pool_used)
      .def("get_and_reset_metrics", &SparkResourceAdaptor::get_and_reset_metrics)
      .def("state_log", &SparkResourceAdaptor::state_log)
      .def("enable_logging", &SparkResourceAdaptor::enable_logging);
}
