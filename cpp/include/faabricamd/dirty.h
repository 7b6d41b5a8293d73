// Dirty tracking for HOST memory regions (reference:
// include/faabric/util/dirty.h:24-225, src/util/dirty.cpp). Modes,
// selected by DIRTY_TRACKING_MODE:
//  - "compare"  (default): no tracking; diffs compare against the snapshot
//               baseline (the honest mode for HBM, where no mprotect
//               exists — reference NoneDirtyTracker + DIFFING_MODE=xor)
//  - "segfault": mprotect(PROT_READ) + SIGSEGV handler marking faulting
//               pages (reference SegfaultDirtyTracker) — host arenas only
//  - "none":    every page dirty
//  - "uffd":    userfaultfd write-protect mode — kernel-async fault
//               delivery to a poller thread, no SIGSEGV involvement
//               (reference UffdDirtyTracker, wp sub-mode)
//  - "softpte": kernel soft-dirty PTE bit via clear_refs/pagemap
//               (reference SoftPTEDirtyTracker); availability is probed
//               functionally — kernels without CONFIG_MEM_SOFT_DIRTY
//               accept the clear_refs write but never set pagemap bit
//               55, in which case it falls back to segfault tracking.
#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <vector>

namespace faabricamd {

class DirtyTracker
{
  public:
    virtual ~DirtyTracker() = default;
    virtual std::string getType() const = 0;

    // Global channel
    virtual void startTracking(uint8_t* region, size_t size) = 0;
    virtual void stopTracking(uint8_t* region, size_t size) = 0;
    virtual std::vector<char> getDirtyPages(uint8_t* region,
                                            size_t size) = 0;

    // Thread-local channel (per-task tracking inside a THREADS batch)
    virtual void startThreadLocalTracking(uint8_t* region, size_t size) = 0;
    virtual void stopThreadLocalTracking(uint8_t* region, size_t size) = 0;
    virtual std::vector<char> getThreadLocalDirtyPages(uint8_t* region,
                                                       size_t size) = 0;
};

// Every page dirty (reference: util/dirty.h:199-225)
class NoneDirtyTracker : public DirtyTracker
{
  public:
    std::string getType() const override { return "none"; }
    void startTracking(uint8_t*, size_t) override {}
    void stopTracking(uint8_t*, size_t) override {}
    std::vector<char> getDirtyPages(uint8_t* region, size_t size) override;
    void startThreadLocalTracking(uint8_t*, size_t) override {}
    void stopThreadLocalTracking(uint8_t*, size_t) override {}
    std::vector<char> getThreadLocalDirtyPages(uint8_t* region,
                                               size_t size) override;
};

// mprotect + SIGSEGV (reference: src/util/dirty.cpp:136-352). Regions
// must be page-aligned (use PageAlignedBuffer).
class SegfaultDirtyTracker : public DirtyTracker
{
  public:
    SegfaultDirtyTracker();
    std::string getType() const override { return "segfault"; }
    void startTracking(uint8_t* region, size_t size) override;
    void stopTracking(uint8_t* region, size_t size) override;
    std::vector<char> getDirtyPages(uint8_t* region, size_t size) override;
    void startThreadLocalTracking(uint8_t* region, size_t size) override;
    void stopThreadLocalTracking(uint8_t* region, size_t size) override;
    std::vector<char> getThreadLocalDirtyPages(uint8_t* region,
                                               size_t size) override;
};

// userfaultfd write-protect tracker (reference: src/util/dirty.cpp uffd
// modes). Global channel is exact; the thread-local channel attributes by
// time-window (pages dirtied between start/stop on any thread) because
// uffd faults are delivered to a poller thread, not the faulting thread —
// overlapping same-page writes from concurrent THREADS tasks are app
// responsibility, as in the reference.
class UffdDirtyTracker : public DirtyTracker
{
  public:
    UffdDirtyTracker(); // throws if the kernel lacks uffd-wp
    ~UffdDirtyTracker() override;
    std::string getType() const override { return "uffd"; }
    void startTracking(uint8_t* region, size_t size) override;
    void stopTracking(uint8_t* region, size_t size) override;
    std::vector<char> getDirtyPages(uint8_t* region, size_t size) override;
    void startThreadLocalTracking(uint8_t* region, size_t size) override;
    void stopThreadLocalTracking(uint8_t* region, size_t size) override;
    std::vector<char> getThreadLocalDirtyPages(uint8_t* region,
                                               size_t size) override;

    static bool isAvailable();
};

// Kernel soft-dirty PTE bit: "4" to /proc/self/clear_refs resets the
// bits, /proc/self/pagemap bit 55 reads them back (reference:
// util/dirty.h:12-16,58-90 SoftPTEDirtyTracker). No handlers and no
// mprotect, but process-wide: clear_refs resets EVERY mapping, so two
// overlapping tracked regions are fine while an unrelated concurrent
// tracker is not. Thread-local channel is empty (resolution is
// process-wide) — the reference behaves the same way.
class SoftPTEDirtyTracker : public DirtyTracker
{
  public:
    SoftPTEDirtyTracker(); // throws if pagemap/clear_refs unusable
    ~SoftPTEDirtyTracker() override;
    std::string getType() const override { return "softpte"; }
    void startTracking(uint8_t* region, size_t size) override;
    void stopTracking(uint8_t* region, size_t size) override;
    std::vector<char> getDirtyPages(uint8_t* region, size_t size) override;
    void startThreadLocalTracking(uint8_t* region, size_t size) override;
    void stopThreadLocalTracking(uint8_t* region, size_t size) override;
    std::vector<char> getThreadLocalDirtyPages(uint8_t* region,
                                               size_t size) override;

    static bool isAvailable();

  private:
    int clearRefsFd = -1;
    int pagemapFd = -1;
};

std::shared_ptr<DirtyTracker> getDirtyTracker();
void resetDirtyTracker();

// OR-merge of per-thread page flags (reference: src/util/memory.cpp:15-23)
void mergeDirtyPages(std::vector<char>& dest,
                     const std::vector<char>& src);

// Page-aligned growable host buffer for trackable executor arenas
class PageAlignedBuffer
{
  public:
    PageAlignedBuffer() = default;
    ~PageAlignedBuffer();
    PageAlignedBuffer(const PageAlignedBuffer&) = delete;

    void resize(size_t newSize); // rounds up to whole pages, zero-fills
    uint8_t* data() { return base; }
    const uint8_t* data() const { return base; }
    size_t size() const { return usedSize; }

  private:
    uint8_t* base = nullptr;
    size_t allocSize = 0;
    size_t usedSize = 0;
};

} // namespace faabricamd
