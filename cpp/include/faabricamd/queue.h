// Concurrency primitives: blocking queue with timeout, token pool, latch,
// barrier, flag-waiter, periodic background thread.
//
// MI355X-native equivalents of the reference's util concurrency layer
// (reference: include/faabric/util/queue.h:25,148,220,245,
//  util/latch.h:11-34, src/util/barrier.cpp,
//  transport/PointToPointBroker.h:165-168 FlagWaiter,
//  util/PeriodicBackgroundThread.h:15-44). Re-designed, not ported:
// std::condition_variable throughout, no third-party lock-free deps.
#pragma once

#include <array>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <deque>
#include <functional>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <thread>
#include <unordered_map>
#include <vector>

#include "faabricamd/util.h"

namespace faabricamd {

inline constexpr int DEFAULT_QUEUE_TIMEOUT_MS = 5000;

class QueueTimeoutException : public FaabricException
{
  public:
    using FaabricException::FaabricException;
};

template<typename T>
class Queue
{
  public:
    void enqueue(T value)
    {
        {
            std::lock_guard<std::mutex> lock(mx);
            q.emplace_back(std::move(value));
        }
        cv.notify_one();
    }

    T dequeue(int timeoutMs = DEFAULT_QUEUE_TIMEOUT_MS)
    {
        std::unique_lock<std::mutex> lock(mx);
        if (timeoutMs <= 0) {
            cv.wait(lock, [this] { return !q.empty(); });
        } else if (!cv.wait_for(lock,
                                std::chrono::milliseconds(timeoutMs),
                                [this] { return !q.empty(); })) {
            throw QueueTimeoutException("queue dequeue timed out");
        }
        T v = std::move(q.front());
        q.pop_front();
        return v;
    }

    bool tryDequeue(T& out)
    {
        std::lock_guard<std::mutex> lock(mx);
        if (q.empty()) {
            return false;
        }
        out = std::move(q.front());
        q.pop_front();
        return true;
    }

    size_t size()
    {
        std::lock_guard<std::mutex> lock(mx);
        return q.size();
    }

    void drain()
    {
        std::lock_guard<std::mutex> lock(mx);
        q.clear();
    }

  private:
    std::mutex mx;
    std::condition_variable cv;
    std::deque<T> q;
};

// Pool of integer tokens (reference: util/queue.h:245 TokenPool)
// Bounded single-producer/single-consumer ring queue. Lock-free on the
// fast path: head/tail are C++20 atomics the producer/consumer each own,
// with acquire/release pairing on the other side's index; blocked sides
// poll with yield so timeouts hold (std::atomic::wait has no timed form). MI355X-native
// replacement for the reference's moodycamel-backed FixedCapacityQueue
// (reference: include/faabric/util/queue.h:148-218) without the
// third-party dependency.
template<typename T>
class FixedCapacityQueue
{
  public:
    explicit FixedCapacityQueue(size_t capacityIn)
      : cap(capacityIn + 1)
      , ring(capacityIn + 1)
    {
        if (capacityIn == 0) {
            throw FaabricException("FixedCapacityQueue capacity must be > 0");
        }
    }

    void enqueue(T value, int timeoutMs = DEFAULT_QUEUE_TIMEOUT_MS)
    {
        size_t t = tail.load(std::memory_order_relaxed);
        size_t next = (t + 1) % cap;
        auto deadline = std::chrono::steady_clock::now() +
                        std::chrono::milliseconds(timeoutMs);
        // Slow path is a bounded poll: std::atomic::wait has no timed
        // variant, and the queue must honour timeoutMs
        while (next == head.load(std::memory_order_acquire)) {
            if (std::chrono::steady_clock::now() > deadline) {
                throw QueueTimeoutException("SPSC enqueue timed out (full)");
            }
            std::this_thread::yield();
        }
        ring[t] = std::move(value);
        tail.store(next, std::memory_order_release);
    }

    T dequeue(int timeoutMs = DEFAULT_QUEUE_TIMEOUT_MS)
    {
        size_t h = head.load(std::memory_order_relaxed);
        auto deadline = std::chrono::steady_clock::now() +
                        std::chrono::milliseconds(timeoutMs);
        while (h == tail.load(std::memory_order_acquire)) {
            if (std::chrono::steady_clock::now() > deadline) {
                throw QueueTimeoutException("SPSC dequeue timed out (empty)");
            }
            std::this_thread::yield();
        }
        T v = std::move(ring[h]);
        head.store((h + 1) % cap, std::memory_order_release);
        return v;
    }

    size_t size() const
    {
        size_t t = tail.load(std::memory_order_acquire);
        size_t h = head.load(std::memory_order_acquire);
        return (t + cap - h) % cap;
    }

  private:
    size_t cap;
    std::vector<T> ring;
    alignas(64) std::atomic<size_t> head{ 0 };
    alignas(64) std::atomic<size_t> tail{ 0 };
};

// Bounded MPMC spin queue: per-slot sequence tickets (Vyukov scheme),
// producers/consumers claim slots with fetch_add and spin until their
// slot's ticket matches. No mutex, no cv — the low-latency local MPI
// delivery option the reference enables with FAABRIC_USE_SPINLOCK
// (reference: include/faabric/util/queue.h:220 SpinLockQueue on
// atomic_queue; selection src/mpi/MpiWorld.h:29-33). Capacity is
// rounded up to a power of two.
template<typename T>
class SpinLockQueue
{
  public:
    explicit SpinLockQueue(size_t capacityIn)
    {
        if (capacityIn == 0) {
            throw FaabricException("SpinLockQueue capacity must be > 0");
        }
        size_t c = 1;
        while (c < capacityIn) {
            c <<= 1;
        }
        mask = c - 1;
        slots = std::make_unique<Slot[]>(c);
        for (size_t i = 0; i < c; i++) {
            slots[i].seq.store(i, std::memory_order_relaxed);
        }
    }

    void enqueue(T value, int timeoutMs = DEFAULT_QUEUE_TIMEOUT_MS)
    {
        size_t pos = tail.fetch_add(1, std::memory_order_relaxed);
        Slot& s = slots[pos & mask];
        auto deadline = std::chrono::steady_clock::now() +
                        std::chrono::milliseconds(timeoutMs);
        // Slot is writable when its ticket equals our position
        while (s.seq.load(std::memory_order_acquire) != pos) {
            if (std::chrono::steady_clock::now() > deadline) {
                throw QueueTimeoutException("spin enqueue timed out (full)");
            }
            std::this_thread::yield();
        }
        s.value = std::move(value);
        s.seq.store(pos + 1, std::memory_order_release);
    }

    T dequeue(int timeoutMs = DEFAULT_QUEUE_TIMEOUT_MS)
    {
        size_t pos = head.fetch_add(1, std::memory_order_relaxed);
        Slot& s = slots[pos & mask];
        auto deadline = std::chrono::steady_clock::now() +
                        std::chrono::milliseconds(timeoutMs);
        while (s.seq.load(std::memory_order_acquire) != pos + 1) {
            if (std::chrono::steady_clock::now() > deadline) {
                throw QueueTimeoutException("spin dequeue timed out (empty)");
            }
            std::this_thread::yield();
        }
        T v = std::move(s.value);
        // Ticket advances a full lap so the slot is writable at pos+cap
        s.seq.store(pos + mask + 1, std::memory_order_release);
        return v;
    }

    size_t size() const
    {
        size_t t = tail.load(std::memory_order_acquire);
        size_t h = head.load(std::memory_order_acquire);
        return t > h ? t - h : 0;
    }

  private:
    struct Slot
    {
        alignas(64) std::atomic<size_t> seq{ 0 };
        T value;
    };
    std::unique_ptr<Slot[]> slots;
    size_t mask = 0;
    alignas(64) std::atomic<size_t> head{ 0 };
    alignas(64) std::atomic<size_t> tail{ 0 };
};

// Sharded concurrent hash map: key hash picks one of NSHARDS
// mutex+unordered_map shards, so hot-path lookups from different keys
// don't serialize on one lock (reference: util/concurrent_map.h, an
// abseil-backed map — re-designed as plain sharding, no third-party
// dependency).
template<typename K, typename V, size_t NSHARDS = 16>
class ConcurrentMap
{
    static_assert((NSHARDS & (NSHARDS - 1)) == 0, "power-of-two shards");

  public:
    // Returns a copy (value semantics keep the lock scope minimal);
    // V is expected to be cheap to copy — use shared_ptr for big values
    bool tryGet(const K& key, V& out) const
    {
        const Shard& s = shardFor(key);
        std::lock_guard<std::mutex> lock(s.mx);
        auto it = s.map.find(key);
        if (it == s.map.end()) {
            return false;
        }
        out = it->second;
        return true;
    }

    // Insert-or-assign
    void set(const K& key, V value)
    {
        Shard& s = shardFor(key);
        std::lock_guard<std::mutex> lock(s.mx);
        s.map[key] = std::move(value);
    }

    // Returns the existing value, or inserts make() and returns that.
    // make() runs under the shard lock — keep it cheap.
    template<typename F>
    V getOrCreate(const K& key, F&& make)
    {
        Shard& s = shardFor(key);
        std::lock_guard<std::mutex> lock(s.mx);
        auto it = s.map.find(key);
        if (it != s.map.end()) {
            return it->second;
        }
        V v = make();
        s.map.emplace(key, v);
        return v;
    }

    bool erase(const K& key)
    {
        Shard& s = shardFor(key);
        std::lock_guard<std::mutex> lock(s.mx);
        return s.map.erase(key) > 0;
    }

    size_t size() const
    {
        size_t n = 0;
        for (const auto& s : shards) {
            std::lock_guard<std::mutex> lock(s.mx);
            n += s.map.size();
        }
        return n;
    }

    void clear()
    {
        for (auto& s : shards) {
            std::lock_guard<std::mutex> lock(s.mx);
            s.map.clear();
        }
    }

    // Removes entries matching pred, one shard locked at a time
    template<typename F>
    size_t eraseIf(F&& pred)
    {
        size_t n = 0;
        for (auto& s : shards) {
            std::lock_guard<std::mutex> lock(s.mx);
            for (auto it = s.map.begin(); it != s.map.end();) {
                if (pred(it->first, it->second)) {
                    it = s.map.erase(it);
                    n++;
                } else {
                    ++it;
                }
            }
        }
        return n;
    }

    // Visits every entry, one shard locked at a time (no global snapshot)
    template<typename F>
    void forEach(F&& fn) const
    {
        for (const auto& s : shards) {
            std::lock_guard<std::mutex> lock(s.mx);
            for (const auto& [k, v] : s.map) {
                fn(k, v);
            }
        }
    }

  private:
    struct Shard
    {
        mutable std::mutex mx;
        std::unordered_map<K, V> map;
    };
    std::array<Shard, NSHARDS> shards;

    Shard& shardFor(const K& key)
    {
        return shards[std::hash<K>{}(key) & (NSHARDS - 1)];
    }
    const Shard& shardFor(const K& key) const
    {
        return shards[std::hash<K>{}(key) & (NSHARDS - 1)];
    }
};

class TokenPool
{
  public:
    explicit TokenPool(int nTokens)
    {
        for (int i = 0; i < nTokens; i++) {
            tokens.enqueue(i);
        }
        nTotal = nTokens;
    }

    int getToken(int timeoutMs = DEFAULT_QUEUE_TIMEOUT_MS)
    {
        return tokens.dequeue(timeoutMs);
    }

    void releaseToken(int token) { tokens.enqueue(token); }
    int size() const { return nTotal; }

  private:
    Queue<int> tokens;
    int nTotal = 0;
};

// Countdown latch with timeout (reference: util/latch.h:11-34 — shared_ptr
// only, wait() with timeout)
class Latch
{
  public:
    static std::shared_ptr<Latch> create(
      int countIn,
      int timeoutMsIn = DEFAULT_QUEUE_TIMEOUT_MS)
    {
        return std::make_shared<Latch>(countIn, timeoutMsIn);
    }

    Latch(int countIn, int timeoutMsIn)
      : count(countIn)
      , timeoutMs(timeoutMsIn)
    {}

    void wait()
    {
        std::unique_lock<std::mutex> lock(mx);
        waiters++;
        if (waiters > count) {
            throw FaabricException("latch already used");
        }
        if (!cv.wait_for(lock, std::chrono::milliseconds(timeoutMs), [this] {
                return waiters >= count;
            })) {
            throw QueueTimeoutException("latch wait timed out");
        }
        cv.notify_all();
    }

  private:
    int count;
    int waiters = 0;
    int timeoutMs;
    std::mutex mx;
    std::condition_variable cv;
};

// Reusable barrier with an optional completion function
// (reference: src/util/barrier.cpp)
class Barrier
{
  public:
    static std::shared_ptr<Barrier> create(
      int count,
      std::function<void()> completionIn = nullptr,
      int timeoutMsIn = DEFAULT_QUEUE_TIMEOUT_MS)
    {
        return std::make_shared<Barrier>(count, completionIn, timeoutMsIn);
    }

    Barrier(int countIn, std::function<void()> completionIn, int timeoutMsIn)
      : count(countIn)
      , completion(std::move(completionIn))
      , timeoutMs(timeoutMsIn)
    {}

    void wait()
    {
        std::unique_lock<std::mutex> lock(mx);
        int phaseAtEntry = phase;
        arrived++;
        if (arrived == count) {
            if (completion) {
                completion();
            }
            arrived = 0;
            phase++;
            cv.notify_all();
            return;
        }
        if (!cv.wait_for(lock, std::chrono::milliseconds(timeoutMs), [&] {
                return phase != phaseAtEntry;
            })) {
            throw QueueTimeoutException("barrier wait timed out");
        }
    }

  private:
    int count;
    int arrived = 0;
    int phase = 0;
    std::function<void()> completion;
    int timeoutMs;
    std::mutex mx;
    std::condition_variable cv;
};

// Set-once flag other threads can block on, used for PTP mapping waits
// (reference: transport/PointToPointBroker.h:165-168)
class FlagWaiter
{
  public:
    explicit FlagWaiter(int timeoutMsIn = DEFAULT_QUEUE_TIMEOUT_MS)
      : timeoutMs(timeoutMsIn)
    {}

    void waitOnFlag()
    {
        std::unique_lock<std::mutex> lock(mx);
        if (!cv.wait_for(lock, std::chrono::milliseconds(timeoutMs), [this] {
                return flag;
            })) {
            throw QueueTimeoutException("flag wait timed out");
        }
    }

    void setFlag(bool value)
    {
        {
            std::lock_guard<std::mutex> lock(mx);
            flag = value;
        }
        cv.notify_all();
    }

    bool isSet()
    {
        std::lock_guard<std::mutex> lock(mx);
        return flag;
    }

    // Bounded wait that reports the flag state instead of throwing
    bool waitMs(int ms)
    {
        std::unique_lock<std::mutex> lock(mx);
        cv.wait_for(lock, std::chrono::milliseconds(ms), [this] {
            return flag;
        });
        return flag;
    }

  private:
    int timeoutMs;
    bool flag = false;
    std::mutex mx;
    std::condition_variable cv;
};

// Base class running doWork() every intervalSeconds until stopped
// (reference: util/PeriodicBackgroundThread.h:15-44)
class PeriodicBackgroundThread
{
  public:
    virtual ~PeriodicBackgroundThread() { stop(); }

    void start(int intervalSecondsIn)
    {
        intervalMs = intervalSecondsIn * 1000;
        startMillis(intervalMs);
    }

    void startMillis(int intervalMsIn)
    {
        intervalMs = intervalMsIn;
        stopped = false;
        worker = std::thread([this] {
            std::unique_lock<std::mutex> lock(mx);
            while (!stopped) {
                if (cv.wait_for(lock,
                                std::chrono::milliseconds(intervalMs),
                                [this] { return stopped; })) {
                    break;
                }
                lock.unlock();
                try {
                    doWork();
                } catch (const std::exception& e) {
                    FAM_ERROR("periodic thread error: %s", e.what());
                }
                lock.lock();
            }
        });
    }

    void stop()
    {
        {
            std::lock_guard<std::mutex> lock(mx);
            if (stopped) {
                if (worker.joinable()) {
                    worker.join();
                }
                return;
            }
            stopped = true;
        }
        cv.notify_all();
        if (worker.joinable()) {
            worker.join();
        }
        tidyUp();
    }

    virtual void doWork() = 0;
    virtual void tidyUp() {}

  protected:
    int intervalMs = 1000;

  private:
    std::thread worker;
    std::mutex mx;
    std::condition_variable cv;
    bool stopped = true;
};

} // namespace faabricamd
