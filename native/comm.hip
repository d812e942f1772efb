// Native RCCL comm core (SURVEY.md §1b "Comm core (C++)"): the framework's
// own communicator over RCCL/xGMI, bootstrapped out-of-band (the unique id
// travels over the torch.distributed rendezvous store from Python — RCCL
// itself never needs a second rendezvous).
//
// Collectives are enqueued on the CURRENT torch stream (stream-ordered
// semantics, hipGraph-capturable); one communicator per process, one process
// per GPU.
//
// Failure handling (SURVEY.md §5 "clean abort on RCCL error"): an optional
// watchdog thread bounds every collective. Each enqueued collective records
// a hipEvent on its stream; the watchdog polls the event (and RCCL's async
// error state) and calls ncclCommAbort if the collective has not completed
// within the timeout — a peer that died mid-ring then produces a loud abort
// instead of an infinite hang (gloo's rendezvous timeout cannot cover
// in-flight device-side collectives).
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstdio>
#include <deque>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace comm {

inline void RCCL_CHECK_FN(ncclResult_t r) {
  if (r != ncclSuccess)
    throw std::runtime_error(std::string("RCCL error: ") + ncclGetErrorString(r));
}

static ncclComm_t g_comm = nullptr;
static int g_rank = -1, g_world = 0;

// ---- watchdog ------------------------------------------------------------
static std::atomic<bool> g_aborted{false};
static std::atomic<bool> g_watch_stop{false};
static double g_timeout_s = 0.0;  // 0 = watchdog disabled
static std::thread g_watch_thread;
static std::mutex g_mu;
static std::condition_variable g_cv;
struct WatchItem {
  hipEvent_t ev;
  std::chrono::steady_clock::time_point deadline;
};
static std::deque<WatchItem> g_q;
static std::vector<hipEvent_t> g_ev_pool;

static void do_abort(const char* why) {
  std::fprintf(stderr, "[dist_mnist_amd comm watchdog] rank %d: %s — calling "
               "ncclCommAbort\n", g_rank, why);
  std::fflush(stderr);
  if (g_comm != nullptr) ncclCommAbort(g_comm);
  g_aborted.store(true);
}

static void watchdog_loop() {
  for (;;) {
    WatchItem w;
    {
      std::unique_lock<std::mutex> lk(g_mu);
      g_cv.wait(lk, [] { return g_watch_stop.load() || !g_q.empty(); });
      if (g_q.empty()) {
        if (g_watch_stop.load()) return;
        continue;
      }
      w = g_q.front();
      g_q.pop_front();
    }
    for (;;) {
      hipError_t e = hipEventQuery(w.ev);
      if (e == hipSuccess) break;
      if (e != hipErrorNotReady) {
        do_abort(hipGetErrorString(e));
        break;
      }
      ncclResult_t ar = ncclSuccess;
      if (g_comm != nullptr &&
          ncclCommGetAsyncError(g_comm, &ar) == ncclSuccess &&
          ar != ncclSuccess) {
        do_abort(ncclGetErrorString(ar));
        break;
      }
      if (g_watch_stop.load()) break;  // shutting down: stop waiting
      if (std::chrono::steady_clock::now() > w.deadline) {
        char buf[96];
        std::snprintf(buf, sizeof(buf), "collective exceeded %.1f s timeout",
                      g_timeout_s);
        do_abort(buf);
        break;
      }
      std::this_thread::sleep_for(std::chrono::milliseconds(2));
    }
    {
      std::lock_guard<std::mutex> lk(g_mu);
      g_ev_pool.push_back(w.ev);
    }
  }
}

static void watchdog_start_locked() {
  if (!g_watch_thread.joinable()) {
    g_watch_stop.store(false);
    g_watch_thread = std::thread(watchdog_loop);
  }
}

// record an event for the collective just enqueued on `s` and hand it to
// the watchdog (no-op when the watchdog is disabled)
static void watch_collective(hipStream_t s) {
  if (g_timeout_s <= 0.0) return;
  hipEvent_t ev = nullptr;
  {
    std::lock_guard<std::mutex> lk(g_mu);
    if (!g_ev_pool.empty()) {
      ev = g_ev_pool.back();
      g_ev_pool.pop_back();
    }
  }
  if (ev == nullptr) {
    if (hipEventCreateWithFlags(&ev, hipEventDisableTiming) != hipSuccess)
      return;  // watchdog is best-effort; never fail the hot path
  }
  if (hipEventRecord(ev, s) != hipSuccess) {
    std::lock_guard<std::mutex> lk(g_mu);
    g_ev_pool.push_back(ev);
    return;
  }
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::milliseconds((long)(g_timeout_s * 1000.0));
  {
    std::lock_guard<std::mutex> lk(g_mu);
    watchdog_start_locked();
    g_q.push_back({ev, deadline});
  }
  g_cv.notify_one();
}

static void watchdog_shutdown() {
  g_watch_stop.store(true);
  g_cv.notify_all();
  if (g_watch_thread.joinable()) g_watch_thread.join();
  std::lock_guard<std::mutex> lk(g_mu);
  for (auto& w : g_q) g_ev_pool.push_back(w.ev);
  g_q.clear();
  for (auto ev : g_ev_pool) hipEventDestroy(ev);
  g_ev_pool.clear();
}

}  // namespace comm
