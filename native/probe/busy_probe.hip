// busy_probe.hip — gfx950 busy-loop micro-probe (self-test fixture).
//
// The only device code in this project, by design: the reference has zero
// kernels and this probe exists solely to validate the
// mi355-exporter's counter semantics on real silicon — utilization must read
// > 0 under this load and exactly 0 when idle, in both the instantaneous
// busy-percent and the windowed GR_ENGINE_ACTIVE ratio (SURVEY.md §7
// "Counter semantics parity").
//
// CDNA4 notes (per /opt/skills/guides/cdna_hip_programming.md):
//  * wave64: blocks are a multiple of 64 threads (256 here);
//  * the grid covers all 256 CUs (2 blocks/CU default) so GRBM_GUI_ACTIVE
//    reflects a chip-wide load, not one busy XCD;
//  * every spin is bounded: the kernel self-terminates after `max_ticks`
//    shader cycles (s_memtime) even if the host never sets the stop flag, so
//    a crashed host cannot leave the GPU wedged.
//
// Exposed as a small C ABI (dlopen'd from gpu_pruner_amd/probe.py) — no
// PyTorch dependency.
#include <hip/hip_runtime.h>

#include <atomic>
#include <cstdio>

namespace {

__global__ void busy_kernel(volatile int* stop_flag, unsigned long long max_ticks,
                            float* sink) {
  unsigned long long start = __builtin_amdgcn_s_memtime();
  float a = 1.0f + threadIdx.x;
  float b = 1.000001f;
  // FMA spin: keeps the VALUs of every resident wave busy so GRBM reports
  // graphics-engine activity; checks the stop flag + safety bound between
  // inner bursts.
  while (true) {
    for (int i = 0; i < 4096; i++) a = __builtin_fmaf(a, b, 0.25f);
    if (*stop_flag != 0) break;
    if (__builtin_amdgcn_s_memtime() - start > max_ticks) break;
  }
  if (a == 12345.678f) sink[threadIdx.x] = a;  // defeat DCE; never true
}

struct ProbeState {
  int* stop_flag = nullptr;   // host-pinned, device-visible
  float* sink = nullptr;
  hipStream_t stream = nullptr;
  int device = -1;
  bool running = false;
};

ProbeState g_state;

#define CHECK(expr)                                                      \
  do {                                                                   \
    hipError_t err_ = (expr);                                            \
    if (err_ != hipSuccess) {                                            \
      std::snprintf(g_last_error, sizeof g_last_error, "%s failed: %s",  \
                    #expr, hipGetErrorString(err_));                     \
      return -1;                                                         \
    }                                                                    \
  } while (0)

char g_last_error[512] = {0};

}  // namespace

extern "C" {

const char* busy_probe_last_error() { return g_last_error; }

int busy_probe_device_count() {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

// Launch the persistent busy kernel on `device`. blocks<=0 picks 512
// (2 per CU on MI355X); max_seconds bounds the kernel even without stop.
int busy_probe_start(int device, int blocks, double max_seconds) {
  if (g_state.running) return 0;
  if (blocks <= 0) blocks = 512;
  if (max_seconds <= 0 || max_seconds > 120) max_seconds = 120;
  CHECK(hipSetDevice(device));
  CHECK(hipHostMalloc(reinterpret_cast<void**>(&g_state.stop_flag), sizeof(int),
                      hipHostMallocDefault));
  *g_state.stop_flag = 0;
  CHECK(hipMalloc(reinterpret_cast<void**>(&g_state.sink), 256 * sizeof(float)));
  CHECK(hipStreamCreateWithFlags(&g_state.stream, hipStreamNonBlocking));
  // s_memtime ticks at the shader clock; assume <= 2.5 GHz for the bound
  // (an over-estimate only lengthens the safety window).
  unsigned long long max_ticks =
      static_cast<unsigned long long>(max_seconds * 2.5e9);
  hipLaunchKernelGGL(busy_kernel, dim3(blocks), dim3(256), 0, g_state.stream,
                     g_state.stop_flag, max_ticks, g_state.sink);
  CHECK(hipGetLastError());
  g_state.device = device;
  g_state.running = true;
  return 0;
}

int busy_probe_stop() {
  if (!g_state.running) return 0;
  *g_state.stop_flag = 0x1;
  __atomic_thread_fence(__ATOMIC_SEQ_CST);
  CHECK(hipStreamSynchronize(g_state.stream));
  CHECK(hipStreamDestroy(g_state.stream));
  CHECK(hipHostFree(g_state.stop_flag));
  CHECK(hipFree(g_state.sink));
  g_state = ProbeState{};
  return 0;
}

// Blocking convenience: full load for `ms` milliseconds.
int busy_probe_run_for_ms(int device, int ms) {
  if (busy_probe_start(device, 0, ms / 1000.0 + 30.0) != 0) return -1;
  struct timespec ts {ms / 1000, (ms % 1000) * 1000000L};
  nanosleep(&ts, nullptr);
  return busy_probe_stop();
}

}  // extern "C"
