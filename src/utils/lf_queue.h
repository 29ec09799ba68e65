/* Bounded lock-free MPMC queue (sequence-numbered ring) for the
 * THREAD_MULTIPLE progress queue.
 * Reference parity: utils/ucc_lock_free_queue.h + the lock-free progress
 * queue option (core/ucc_context.c LOCK_FREE_PROGRESS_Q) — re-derived as
 * the classic sequence-ring design. Falls back to caller's locked path
 * when full (push returns false). */
#ifndef UCC_AMD_LF_QUEUE_H_
#define UCC_AMD_LF_QUEUE_H_

#include <atomic>
#include <cstddef>
#include <cstdint>

namespace ucc {

template <typename T, size_t CapacityPow2 = 1024>
class LfQueue {
    static_assert((CapacityPow2 & (CapacityPow2 - 1)) == 0,
                  "capacity must be a power of two");

    struct Cell {
        std::atomic<uint64_t> seq;
        T                     val;
    };

  public:
    LfQueue()
    {
        for (size_t i = 0; i < CapacityPow2; i++) {
            cells_[i].seq.store(i, std::memory_order_relaxed);
        }
        head_.store(0, std::memory_order_relaxed);
        tail_.store(0, std::memory_order_relaxed);
    }

    bool push(const T &v)
    {
        uint64_t pos = tail_.load(std::memory_order_relaxed);
        for (;;) {
            Cell    &c   = cells_[pos & (CapacityPow2 - 1)];
            uint64_t seq = c.seq.load(std::memory_order_acquire);
            intptr_t dif = (intptr_t)seq - (intptr_t)pos;
            if (dif == 0) {
                if (tail_.compare_exchange_weak(pos, pos + 1,
                                                std::memory_order_relaxed)) {
                    c.val = v;
                    c.seq.store(pos + 1, std::memory_order_release);
                    return true;
                }
            } else if (dif < 0) {
                return false; /* full */
            } else {
                pos = tail_.load(std::memory_order_relaxed);
            }
        }
    }

    bool pop(T *out)
    {
        uint64_t pos = head_.load(std::memory_order_relaxed);
        for (;;) {
            Cell    &c   = cells_[pos & (CapacityPow2 - 1)];
            uint64_t seq = c.seq.load(std::memory_order_acquire);
            intptr_t dif = (intptr_t)seq - (intptr_t)(pos + 1);
            if (dif == 0) {
                if (head_.compare_exchange_weak(pos, pos + 1,
                                                std::memory_order_relaxed)) {
                    *out = c.val;
                    c.seq.store(pos + CapacityPow2,
                                std::memory_order_release);
                    return true;
                }
            } else if (dif < 0) {
                return false; /* empty */
            } else {
                pos = head_.load(std::memory_order_relaxed);
            }
        }
    }

    bool empty() const
    {
        return head_.load(std::memory_order_acquire) ==
               tail_.load(std::memory_order_acquire);
    }

  private:
    alignas(64) Cell cells_[CapacityPow2];
    alignas(64) std::atomic<uint64_t> head_;
    alignas(64) std::atomic<uint64_t> tail_;
};

} // namespace ucc

#endif
