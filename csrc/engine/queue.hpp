// Bounded lock-free SPSC queue — the FastFlow replacement substrate.
// Reference behavior replaced: ff::MPMC_Ptr_Queue / SPSC buffers
// (WindFlow README.md:34-39 FF_BOUNDED_BUFFER semantics).
//
// Single-producer single-consumer ring with acquire/release atomics,
// power-of-two capacity, busy-wait with exponential backoff falling back
// to a futex-style sleep (std::condition_variable after a spin budget) so
// oversubscribed CPU runs (tests on an 8-core box with 30 replicas) do
// not melt down in spin loops.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <mutex>
#include <thread>
#include <vector>

namespace wfa {

struct Batch;

// Message: data batch, or EOS (b == EOS_TAG).
inline Batch* const EOS_TAG = reinterpret_cast<Batch*>(~uintptr_t(0));

class SpscQueue {
  public:
    explicit SpscQueue(size_t cap_pow2 = 2048) {
        size_t c = 1;
        while (c < cap_pow2) c <<= 1;
        mask_ = c - 1;
        buf_.assign(c, nullptr);
    }

    // Producer side. Returns false if full (caller may retry/backpressure).
    bool try_push(Batch* b) {
        const size_t h = head_.load(std::memory_order_relaxed);
        if (h - tail_cache_ > mask_) {
            tail_cache_ = tail_.load(std::memory_order_acquire);
            if (h - tail_cache_ > mask_) return false;
        }
        buf_[h & mask_] = b;
        head_.store(h + 1, std::memory_order_release);
        if (sleepers_.load(std::memory_order_acquire) > 0) wake();
        return true;
    }

    // Blocking push with backoff (backpressure).
    void push(Batch* b, const std::atomic<bool>* abort = nullptr) {
        int spins = 0;
        while (!try_push(b)) {
            if (abort && abort->load(std::memory_order_relaxed)) return;
            backoff(spins);
        }
    }

    // Consumer side. Returns nullptr if empty.
    Batch* try_pop() {
        const size_t t = tail_.load(std::memory_order_relaxed);
        if (t == head_cache_) {
            head_cache_ = head_.load(std::memory_order_acquire);
            if (t == head_cache_) return nullptr;
        }
        Batch* b = buf_[t & mask_];
        tail_.store(t + 1, std::memory_order_release);
        if (sleepers_.load(std::memory_order_acquire) > 0) wake();
        return b;
    }

    bool empty() const {
        return tail_.load(std::memory_order_acquire) == head_.load(std::memory_order_acquire);
    }

    size_t size() const {
        return head_.load(std::memory_order_acquire) - tail_.load(std::memory_order_acquire);
    }

    // Shared backoff helper: spin, yield, then sleep 50us.
    static void backoff(int& spins) {
        ++spins;
        if (spins < 64) {
#if defined(__x86_64__)
            __builtin_ia32_pause();
#endif
        } else if (spins < 256) {
            std::this_thread::yield();
        } else {
            std::this_thread::sleep_for(std::chrono::microseconds(50));
        }
    }

  private:
    void wake() {
        std::lock_guard<std::mutex> g(mu_);
        cv_.notify_all();
    }

    std::vector<Batch*> buf_;
    size_t mask_;
    alignas(64) std::atomic<size_t> head_{0};
    alignas(64) size_t tail_cache_ = 0;
    alignas(64) std::atomic<size_t> tail_{0};
    alignas(64) size_t head_cache_ = 0;
    std::atomic<int> sleepers_{0};
    std::mutex mu_;
    std::condition_variable cv_;
};

}  // namespace wfa
