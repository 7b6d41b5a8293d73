// Dirty tracking implementation (reference: src/util/dirty.cpp — the
// segfault tracker's structure: install handler, PROT_READ the region,
// mark + unprotect faulting pages, thread-local + global channels).
#include "faabricamd/dirty.h"
#include "faabricamd/util.h"

#include <csignal>
#include <cstdlib>
#include <cstring>
#include <map>
#include <mutex>
#include <sys/mman.h>
#include <unistd.h>

namespace faabricamd {

static constexpr size_t TRACK_PAGE = 4096;

static size_t nPagesOf(size_t size)
{
    return (size + TRACK_PAGE - 1) / TRACK_PAGE;
}

// ------------------------- none ---------------------------------------------

std::vector<char> NoneDirtyTracker::getDirtyPages(uint8_t* region,
                                                  size_t size)
{
    (void)region;
    return std::vector<char>(nPagesOf(size), 1);
}

std::vector<char> NoneDirtyTracker::getThreadLocalDirtyPages(uint8_t* region,
                                                             size_t size)
{
    return getDirtyPages(region, size);
}

// ------------------------- segfault ------------------------------------------

namespace {

struct TrackedRegion
{
    uint8_t* base = nullptr;
    size_t size = 0;
    std::vector<char> globalDirty;
};

std::mutex regionsMx;
std::map<uint8_t*, std::shared_ptr<TrackedRegion>> regions;

// Thread-local dirty channel: region base → flags
thread_local std::map<uint8_t*, std::vector<char>> threadDirty;
thread_local bool threadTrackingOn = false;

struct sigaction oldSegvAction;
bool handlerInstalled = false;

void segvHandler(int sig, siginfo_t* info, void* ucontext)
{
    uint8_t* addr = (uint8_t*)info->si_addr;
    std::shared_ptr<TrackedRegion> hit;
    {
        // NOTE: not strictly async-signal-safe; matches the reference's
        // pragmatic approach (faults only come from tracked regions while
        // tracking is active)
        std::lock_guard<std::mutex> lock(regionsMx);
        for (auto& [base, region] : regions) {
            if (addr >= base && addr < base + region->size) {
                hit = region;
                break;
            }
        }
    }
    if (!hit) {
        // Not ours: restore the previous handler and re-raise
        sigaction(SIGSEGV, &oldSegvAction, nullptr);
        raise(sig);
        return;
    }
    size_t page = (size_t)(addr - hit->base) / TRACK_PAGE;
    hit->globalDirty[page] = 1;
    if (threadTrackingOn) {
        auto& flags = threadDirty[hit->base];
        if (flags.size() <= page) {
            flags.resize(nPagesOf(hit->size), 0);
        }
        flags[page] = 1;
    }
    // Re-enable the page for writing
    mprotect(hit->base + page * TRACK_PAGE, TRACK_PAGE,
             PROT_READ | PROT_WRITE);
    (void)ucontext;
}

void installHandler()
{
    if (handlerInstalled) {
        return;
    }
    struct sigaction sa;
    std::memset(&sa, 0, sizeof(sa));
    sa.sa_sigaction = segvHandler;
    sa.sa_flags = SA_SIGINFO | SA_NODEFER;
    sigaction(SIGSEGV, &sa, &oldSegvAction);
    handlerInstalled = true;
}

} // namespace

SegfaultDirtyTracker::SegfaultDirtyTracker()
{
    installHandler();
}

void SegfaultDirtyTracker::startTracking(uint8_t* region, size_t size)
{
    if (((uintptr_t)region % TRACK_PAGE) != 0) {
        throw FaabricException("segfault tracker needs page-aligned region");
    }
    auto tracked = std::make_shared<TrackedRegion>();
    tracked->base = region;
    tracked->size = size;
    tracked->globalDirty.assign(nPagesOf(size), 0);
    {
        std::lock_guard<std::mutex> lock(regionsMx);
        regions[region] = tracked;
    }
    mprotect(region, nPagesOf(size) * TRACK_PAGE, PROT_READ);
}

void SegfaultDirtyTracker::stopTracking(uint8_t* region, size_t size)
{
    mprotect(region, nPagesOf(size) * TRACK_PAGE, PROT_READ | PROT_WRITE);
}

std::vector<char> SegfaultDirtyTracker::getDirtyPages(uint8_t* region,
                                                      size_t size)
{
    std::lock_guard<std::mutex> lock(regionsMx);
    auto it = regions.find(region);
    if (it == regions.end()) {
        return std::vector<char>(nPagesOf(size), 0);
    }
    return it->second->globalDirty;
}

void SegfaultDirtyTracker::startThreadLocalTracking(uint8_t* region,
                                                    size_t size)
{
    threadDirty[region].assign(nPagesOf(size), 0);
    threadTrackingOn = true;
}

void SegfaultDirtyTracker::stopThreadLocalTracking(uint8_t* region,
                                                   size_t size)
{
    (void)region;
    (void)size;
    threadTrackingOn = false;
}

std::vector<char> SegfaultDirtyTracker::getThreadLocalDirtyPages(
  uint8_t* region,
  size_t size)
{
    auto it = threadDirty.find(region);
    if (it == threadDirty.end()) {
        return std::vector<char>(nPagesOf(size), 0);
    }
    it->second.resize(nPagesOf(size), 0);
    return it->second;
}

// ------------------------- registry ------------------------------------------

static std::shared_ptr<DirtyTracker> trackerInstance;
static std::mutex trackerMx;

std::shared_ptr<DirtyTracker> getDirtyTracker()
{
    std::lock_guard<std::mutex> lock(trackerMx);
    if (trackerInstance) {
        return trackerInstance;
    }
    const std::string& mode = getSystemConfig().dirtyTrackingMode;
    if (mode == "segfault") {
        trackerInstance = std::make_shared<SegfaultDirtyTracker>();
    } else {
        // "compare" and "none" both report every page; "compare" relies
        // on the snapshot diff to refine
        trackerInstance = std::make_shared<NoneDirtyTracker>();
    }
    return trackerInstance;
}

void resetDirtyTracker()
{
    std::lock_guard<std::mutex> lock(trackerMx);
    trackerInstance = nullptr;
}

void mergeDirtyPages(std::vector<char>& dest, const std::vector<char>& src)
{
    if (dest.size() < src.size()) {
        dest.resize(src.size(), 0);
    }
    for (size_t i = 0; i < src.size(); i++) {
        dest[i] |= src[i];
    }
}

// ------------------------- page-aligned buffer -------------------------------

PageAlignedBuffer::~PageAlignedBuffer()
{
    if (base != nullptr) {
        std::free(base);
    }
}

void PageAlignedBuffer::resize(size_t newSize)
{
    size_t rounded = nPagesOf(newSize) * TRACK_PAGE;
    if (rounded > allocSize) {
        void* mem = nullptr;
        if (posix_memalign(&mem, TRACK_PAGE, rounded) != 0) {
            throw FaabricException("page-aligned alloc failed");
        }
        std::memset(mem, 0, rounded);
        if (base != nullptr) {
            std::memcpy(mem, base, usedSize);
            std::free(base);
        }
        base = (uint8_t*)mem;
        allocSize = rounded;
    } else if (newSize > usedSize) {
        std::memset(base + usedSize, 0, newSize - usedSize);
    }
    usedSize = newSize;
}

} // namespace faabricamd
