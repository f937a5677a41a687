// SPSC byte-ring + futex doorbells over the shared segment (protocol.h).
#pragma once

#include <linux/futex.h>
#include <string.h>
#include <sys/syscall.h>
#include <time.h>
#include <unistd.h>

#include <atomic>
#include <cstdint>

#include "protocol.h"

namespace tfrpc {

inline long futex_wait(volatile uint32_t* addr, uint32_t expect,
                       int timeout_ms) {
  timespec ts{timeout_ms / 1000, (timeout_ms % 1000) * 1000000L};
  return syscall(SYS_futex, (uint32_t*)addr, FUTEX_WAIT, expect,
                 timeout_ms >= 0 ? &ts : nullptr, nullptr, 0);
}

inline void futex_wake(volatile uint32_t* addr) {
  syscall(SYS_futex, (uint32_t*)addr, FUTEX_WAKE, INT32_MAX, nullptr, nullptr,
          0);
}

template <typename T>
inline std::atomic<T>* at(volatile T* p) {
  return reinterpret_cast<std::atomic<T>*>(const_cast<T*>(p));
}

// One side of a ring. Producer and consumer views over the same storage.
class RingView {
 public:
  RingView() = default;
  RingView(Ring* r, uint8_t* buf, size_t cap) : r_(r), buf_(buf), cap_(cap) {}

  // ---------------- producer ----------------

  // Reserve space for a record of `len` bytes (8-aligned internally).
  // Returns pointer to write to, or nullptr if the ring is full.
  uint8_t* try_reserve(size_t len) {
    size_t need = align8(len) + 8;  // length word + payload
    uint64_t head = at(&r_->head)->load(std::memory_order_relaxed);
    uint64_t tail = at(&r_->tail)->load(std::memory_order_acquire);
    size_t used = head - tail;
    size_t off = head % cap_;
    size_t to_end = cap_ - off;
    if (to_end < need + 8) {  // +8: room for a wrap marker
      // wrap: need marker + full record at start
      if (used + to_end + need > cap_) return nullptr;
      if (to_end >= 8) *(uint64_t*)(buf_ + off) = WRAP;
      pending_head_ = head + to_end;
      off = 0;
    } else {
      if (used + need > cap_) return nullptr;
      pending_head_ = head;
    }
    pending_len_ = len;
    *(uint64_t*)(buf_ + off) = 0;  // not yet published
    return buf_ + off + 8;
  }

  void commit() {
    uint64_t off = pending_head_ % cap_;
    // publish: write length last with release so consumer sees full payload
    at((volatile uint64_t*)(buf_ + off))
        ->store(pending_len_, std::memory_order_release);
    uint64_t new_head = pending_head_ + align8(pending_len_) + 8;
    at(&r_->head)->store(new_head, std::memory_order_release);
  }

  void wake_consumer() {
    // load-before-RMW: during bursts the flag is already 1 and the plain
    // load keeps the line shared instead of ping-ponging an exchange
    // between the producer and consumer cores on every record.
    // The fence makes this safe: without it the head.store(release) in
    // commit() can sit in the producer's store buffer while the flag load
    // executes, so the producer could read flag==1 even though the consumer
    // has already cleared the flag and re-checked head — both sides would
    // then skip the wake and the consumer parks (lost wakeup). seq_cst
    // fence + seq_cst flag ops on both sides restore a total order between
    // {head publish, flag load} here and {flag clear, head re-check} in
    // wait_nonempty.
    std::atomic_thread_fence(std::memory_order_seq_cst);
    if (at(&r_->futex_nonempty)->load(std::memory_order_seq_cst) == 1)
      return;
    if (at(&r_->futex_nonempty)->exchange(1, std::memory_order_seq_cst) == 0)
      futex_wake(&r_->futex_nonempty);
  }

  // ---------------- consumer ----------------

  // Peek next record; returns nullptr if empty. len_out = payload length.
  uint8_t* try_next(size_t* len_out) {
    uint64_t tail = at(&r_->tail)->load(std::memory_order_relaxed);
    uint64_t head = at(&r_->head)->load(std::memory_order_acquire);
    while (true) {
      if (tail >= head) return nullptr;
      size_t off = tail % cap_;
      uint64_t len = at((volatile uint64_t*)(buf_ + off))
                         ->load(std::memory_order_acquire);
      if (len == WRAP) {
        tail += cap_ - off;
        at(&r_->tail)->store(tail, std::memory_order_release);
        continue;
      }
      if (len == 0) return nullptr;  // reserved but not yet committed
      *len_out = len;
      cur_tail_ = tail;
      cur_len_ = len;
      return buf_ + off + 8;
    }
  }

  void pop() {
    at(&r_->tail)->store(cur_tail_ + align8(cur_len_) + 8,
                         std::memory_order_release);
  }

  // Park until non-empty (consumer side). Spin first, then futex.
  void wait_nonempty(int spin_us = 50, int park_ms = 100) {
    size_t len;
    for (int i = 0; i < spin_us * 10; ++i) {
      uint64_t tail = at(&r_->tail)->load(std::memory_order_relaxed);
      uint64_t head = at(&r_->head)->load(std::memory_order_acquire);
      if (head > tail) return;
#if defined(__x86_64__)
      __builtin_ia32_pause();
#endif
    }
    // seq_cst exchange, not a plain store: pairs with the seq_cst fence +
    // flag load in wake_consumer — the total order guarantees either the
    // head re-check below observes the producer's commit, or the producer's
    // flag load observes our clear and takes the exchange+wake path
    at(&r_->futex_nonempty)->exchange(0, std::memory_order_seq_cst);
    uint64_t tail = at(&r_->tail)->load(std::memory_order_relaxed);
    uint64_t head = at(&r_->head)->load(std::memory_order_acquire);
    if (head > tail) return;
    futex_wait(&r_->futex_nonempty, 0, park_ms);
    (void)len;
  }

  bool empty() const {
    return at(&r_->tail)->load(std::memory_order_relaxed) >=
           at(&r_->head)->load(std::memory_order_acquire);
  }

 private:
  static constexpr uint64_t WRAP = ~0ull;
  static size_t align8(size_t n) { return (n + 7) & ~size_t(7); }
  Ring* r_ = nullptr;
  uint8_t* buf_ = nullptr;
  size_t cap_ = 0;
  uint64_t pending_head_ = 0;
  uint64_t pending_len_ = 0;
  uint64_t cur_tail_ = 0;
  uint64_t cur_len_ = 0;
};

}  // namespace tfrpc
