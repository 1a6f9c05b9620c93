// tsan_compat.hpp — TSan-visible timed condition-variable waits.
//
// GCC 11's libstdc++ implements condition_variable::wait_for/wait_until via
// pthread_cond_clockwait, which the matching libtsan does NOT intercept: the
// sanitizer never observes the mutex being released inside the wait, so every
// later acquisition by another thread is reported as a double lock / "data
// race with both threads holding the mutex" / impossible lock-order
// inversion. Under -fsanitize=thread this shim replaces timed cv waits with
// a short unlock-sleep-relock poll loop built only from intercepted
// primitives, so TSan's model stays accurate and real races remain visible.
// Non-sanitized builds use the real cv wait (no behavior change).
#pragma once

#include <chrono>
#include <condition_variable>
#include <mutex>
#include <thread>

#if defined(__SANITIZE_THREAD__)
#define QX_TSAN_ENABLED 1
#elif defined(__has_feature)
#if __has_feature(thread_sanitizer)
#define QX_TSAN_ENABLED 1
#endif
#endif

namespace qx {

// cv.wait_for(lock, timeout, pred) equivalent. Returns pred() at exit.
template <class Rep, class Period, class Pred>
bool cv_wait_for(std::condition_variable& cv, std::unique_lock<std::mutex>& lock,
                 std::chrono::duration<Rep, Period> timeout, Pred pred) {
#ifdef QX_TSAN_ENABLED
  auto deadline = std::chrono::steady_clock::now() + timeout;
  while (!pred()) {
    if (std::chrono::steady_clock::now() >= deadline) return pred();
    lock.unlock();
    std::this_thread::sleep_for(std::chrono::milliseconds(1));
    lock.lock();
  }
  return true;
#else
  return cv.wait_for(lock, timeout, pred);
#endif
}

// cv.wait_for(lock, timeout) equivalent (no predicate; spurious-wakeup-safe
// callers only). Under TSan this simply sleeps one slice at a time.
template <class Rep, class Period>
void cv_wait_for(std::condition_variable& cv, std::unique_lock<std::mutex>& lock,
                 std::chrono::duration<Rep, Period> timeout) {
#ifdef QX_TSAN_ENABLED
  lock.unlock();
  std::this_thread::sleep_for(
      std::min<std::chrono::duration<Rep, Period>>(
          timeout, std::chrono::duration_cast<std::chrono::duration<Rep, Period>>(
                       std::chrono::milliseconds(50))));
  lock.lock();
#else
  cv.wait_for(lock, timeout);
#endif
}

}  // namespace qx
