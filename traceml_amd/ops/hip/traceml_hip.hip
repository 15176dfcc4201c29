// traceml_amd native timing extension — MI355X (gfx950, CDNA4) only.
//
// Two on-device clocks for non-blocking phase timing:
//
// 1. Ring stamps (primary). `stamp_kernel` is a single-wave kernel that
//    reads `s_memrealtime` — the constant-frequency (~100 MHz) real-time
//    counter, globally consistent across CUs, streams and queues on the
//    device — stages the sample through LDS, and publishes
//    {ticks, seq} into one 16-byte slot of a pinned host ring buffer with
//    a system-scope release. The host side resolves a stamp with a plain
//    memory read (acquire on seq): no hipEventQuery syscall, no
//    synchronize, sub-microsecond device resolution (10 ns ticks).
//    Slot reuse is safe by construction: seq strictly increases, a reused
//    slot shows a stale seq and simply reads as "not ready" until the new
//    kernel lands.
//
// 2. hipEvent pool (secondary). Classic hipEventCreateWithFlags /
//    hipEventRecord / hipEventQuery / hipEventElapsedTime, pooled and
//    id-addressed. Used for cross-validating the ring clock and for any
//    consumer that wants event semantics.
//
// No torch linkage: HIP streams cross the boundary as integer handles
// (torch.cuda.current_stream().cuda_stream), so this object builds with
// plain hipcc + pybind11 and loads anywhere.
//
// Replaces (MI355X-native): torch.cuda.Event pooling in the reference
// (traceml_ai/utils/cuda_event_pool.py:25-52, utils/timing.py:68-93).

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>

#include <atomic>
#include <chrono>
#include <cstdint>
#include <cstring>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _err = (expr);                                                  \
    if (_err != hipSuccess) {                                                  \
      throw std::runtime_error(std::string("HIP error at " #expr ": ") +       \
                               hipGetErrorString(_err));                       \
    }                                                                          \
  } while (0)

struct Slot {
  unsigned long long ticks;
  unsigned long long seq;
};
static_assert(sizeof(Slot) == 16, "slot must be 16 bytes");

// ---------------------------------------------------------------------------
// Kernel
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(64, 1) void stamp_kernel(Slot* __restrict__ ring,
                                                      unsigned long long mask,
                                                      unsigned long long seq) {
  // One wave; lane 0 does the work. LDS staging keeps the VMEM publish as a
  // single contiguous 16-byte transaction built from the sampled value.
  __shared__ Slot staged;
  if (threadIdx.x == 0) {
    staged.ticks = __builtin_amdgcn_s_memrealtime();
    staged.seq = seq;
    Slot* slot = ring + (seq & mask);
    slot->ticks = staged.ticks;
    // Publish: the system-scope RELEASE store orders the ticks write before
    // the seq flip (C++ release semantics cover all prior stores), so no
    // separate __threadfence_system() is needed — saving ~1 us of fence
    // drain per stamp on the hot path.
    __hip_atomic_store(&slot->seq, staged.seq, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_SYSTEM);
  }
}

// ---------------------------------------------------------------------------
// State
// ---------------------------------------------------------------------------

namespace {

struct RingState {
  Slot* host_ptr = nullptr;   // pinned, host view
  Slot* dev_ptr = nullptr;    // device view of the same memory
  unsigned long long capacity = 0;  // power of two
  unsigned long long mask = 0;
  std::atomic<unsigned long long> next_seq{1};  // 0 means "never written"
  double ticks_per_sec = 0.0;
  int device = -1;
  bool initialized = false;
};

RingState g_ring;
std::mutex g_init_mutex;

struct EventPool {
  std::vector<hipEvent_t> events;  // id -> event (never shrinks)
  std::vector<int> free_ids;
  std::mutex mutex;
};

EventPool g_events;

unsigned long long round_up_pow2(unsigned long long v) {
  unsigned long long p = 1;
  while (p < v) p <<= 1;
  return p;
}

double query_wall_clock_rate(int device) {
  int khz = 0;
  hipError_t err =
      hipDeviceGetAttribute(&khz, hipDeviceAttributeWallClockRate, device);
  if (err != hipSuccess || khz <= 0) return 0.0;
  return static_cast<double>(khz) * 1000.0;
}

}  // namespace

// ---------------------------------------------------------------------------
// Ring API
// ---------------------------------------------------------------------------

static unsigned long long ring_mark(uintptr_t stream_handle) {
  RingState& r = g_ring;
  unsigned long long seq = r.next_seq.fetch_add(1, std::memory_order_relaxed);
  hipStream_t stream = reinterpret_cast<hipStream_t>(stream_handle);
  stamp_kernel<<<1, 64, 0, stream>>>(r.dev_ptr, r.mask, seq);
  return seq;
}

static bool ring_ready(unsigned long long seq) {
  const RingState& r = g_ring;
  const Slot* slot = r.host_ptr + (seq & r.mask);
  unsigned long long observed = __atomic_load_n(&slot->seq, __ATOMIC_ACQUIRE);
  return observed == seq;
}

static long long ring_ticks(unsigned long long seq) {
  const RingState& r = g_ring;
  const Slot* slot = r.host_ptr + (seq & r.mask);
  unsigned long long observed = __atomic_load_n(&slot->seq, __ATOMIC_ACQUIRE);
  if (observed != seq) return -1;
  return static_cast<long long>(slot->ticks);
}

static double ring_elapsed_ms(unsigned long long seq_start,
                              unsigned long long seq_end) {
  long long t0 = ring_ticks(seq_start);
  long long t1 = ring_ticks(seq_end);
  if (t0 < 0 || t1 < 0) return -1.0;
  double dticks = static_cast<double>(t1 - t0);
  return dticks * 1000.0 / g_ring.ticks_per_sec;
}

static double calibrate(uintptr_t stream_handle, double window_sec) {
  // Two synchronized stamps around a known CPU interval -> ticks/sec.
  hipStream_t stream = reinterpret_cast<hipStream_t>(stream_handle);
  unsigned long long a = ring_mark(reinterpret_cast<uintptr_t>(stream));
  HIP_CHECK(hipStreamSynchronize(stream));
  auto cpu0 = std::chrono::steady_clock::now();
  long long ta = ring_ticks(a);
  std::this_thread::sleep_for(
      std::chrono::microseconds(static_cast<long long>(window_sec * 1e6)));
  unsigned long long b = ring_mark(reinterpret_cast<uintptr_t>(stream));
  HIP_CHECK(hipStreamSynchronize(stream));
  auto cpu1 = std::chrono::steady_clock::now();
  long long tb = ring_ticks(b);
  double sec =
      std::chrono::duration_cast<std::chrono::duration<double>>(cpu1 - cpu0)
          .count();
  if (ta < 0 || tb < 0 || sec <= 0.0) {
    throw std::runtime_error("traceml_hip: calibration stamps did not land");
  }
  double rate = static_cast<double>(tb - ta) / sec;
  g_ring.ticks_per_sec = rate;
  return rate;
}

static void init(int device, unsigned long long slots) {
  std::lock_guard<std::mutex> guard(g_init_mutex);
  if (g_ring.initialized) return;
  HIP_CHECK(hipSetDevice(device));
  unsigned long long capacity = round_up_pow2(slots < 1024 ? 1024 : slots);
  void* host = nullptr;
  // Coherent pinned allocation: device stores with system-scope release are
  // host-visible without any cache maintenance.
  HIP_CHECK(hipHostMalloc(&host, capacity * sizeof(Slot),
                          hipHostMallocMapped | hipHostMallocCoherent));
  std::memset(host, 0, capacity * sizeof(Slot));
  void* dev = nullptr;
  HIP_CHECK(hipHostGetDevicePointer(&dev, host, 0));
  g_ring.host_ptr = static_cast<Slot*>(host);
  g_ring.dev_ptr = static_cast<Slot*>(dev);
  g_ring.capacity = capacity;
  g_ring.mask = capacity - 1;
  g_ring.device = device;
  g_ring.ticks_per_sec = query_wall_clock_rate(device);
  g_ring.initialized = true;
  // Null stream calibration; overwrites the attribute-derived rate with a
  // measured one (the attribute has been wrong on some ROCm stacks).
  try {
    calibrate(0, 0.050);
  } catch (const std::exception&) {
    if (g_ring.ticks_per_sec <= 0.0) g_ring.ticks_per_sec = 1.0e8;  // spec 100 MHz
  }
}

static bool is_initialized() { return g_ring.initialized; }

static double ticks_per_second() { return g_ring.ticks_per_sec; }

// ---------------------------------------------------------------------------
// hipEvent pool API
// ---------------------------------------------------------------------------

static int event_acquire() {
  std::lock_guard<std::mutex> guard(g_events.mutex);
  if (!g_events.free_ids.empty()) {
    int id = g_events.free_ids.back();
    g_events.free_ids.pop_back();
    return id;
  }
  hipEvent_t event;
  HIP_CHECK(hipEventCreateWithFlags(&event, hipEventDefault));
  g_events.events.push_back(event);
  return static_cast<int>(g_events.events.size()) - 1;
}

static hipEvent_t event_for(int id) {
  if (id < 0 || id >= static_cast<int>(g_events.events.size())) {
    throw std::runtime_error("traceml_hip: bad event id");
  }
  return g_events.events[static_cast<size_t>(id)];
}

static void event_record(int id, uintptr_t stream_handle) {
  HIP_CHECK(hipEventRecord(event_for(id),
                           reinterpret_cast<hipStream_t>(stream_handle)));
}

static bool event_query(int id) {
  hipError_t err = hipEventQuery(event_for(id));
  if (err == hipSuccess) return true;
  if (err == hipErrorNotReady) return false;
  HIP_CHECK(err);
  return false;
}

static double event_elapsed_ms(int start_id, int end_id) {
  float ms = 0.0f;
  HIP_CHECK(hipEventElapsedTime(&ms, event_for(start_id), event_for(end_id)));
  return static_cast<double>(ms);
}

static void event_release(int id) {
  std::lock_guard<std::mutex> guard(g_events.mutex);
  event_for(id);  // bounds check
  g_events.free_ids.push_back(id);
}

static int event_pool_size() {
  std::lock_guard<std::mutex> guard(g_events.mutex);
  return static_cast<int>(g_events.events.size());
}

// ---------------------------------------------------------------------------
// Module
// ---------------------------------------------------------------------------

PYBIND11_MODULE(_traceml_hip, m) {
  m.doc() = "traceml_amd MI355X-native timing: s_memrealtime ring stamps + "
            "hipEvent pool (gfx950)";
  m.def("init", &init, py::arg("device"), py::arg("slots") = 65536);
  m.def("is_initialized", &is_initialized);
  m.def("ticks_per_second", &ticks_per_second);
  m.def("ring_mark", &ring_mark, py::arg("stream"));
  m.def("ring_ready", &ring_ready, py::arg("seq"));
  m.def("ring_ticks", &ring_ticks, py::arg("seq"));
  m.def("ring_elapsed_ms", &ring_elapsed_ms, py::arg("seq_start"),
        py::arg("seq_end"));
  m.def("calibrate", &calibrate, py::arg("stream") = 0,
        py::arg("window_sec") = 0.05);
  m.def("event_acquire", &event_acquire);
  m.def("event_record", &event_record, py::arg("event_id"), py::arg("stream"));
  m.def("event_query", &event_query, py::arg("event_id"));
  m.def("event_elapsed_ms", &event_elapsed_ms, py::arg("start_id"),
        py::arg("end_id"));
  m.def("event_release", &event_release, py::arg("event_id"));
  m.def("event_pool_size", &event_pool_size);
}
