// Dirty tracking implementation (reference: src/util/dirty.cpp — the
// segfault tracker's structure: install handler, PROT_READ the region,
// mark + unprotect faulting pages, thread-local + global channels).
#include "faabricamd/dirty.h"
#include "faabricamd/util.h"

#include <atomic>
#include <cerrno>
#include <cstdio>
#include <csignal>
#include <cstdlib>
#include <cstring>
#include <fcntl.h>
#include <linux/userfaultfd.h>
#include <map>
#include <mutex>
#include <poll.h>
#include <sys/ioctl.h>
#include <sys/mman.h>
#include <sys/syscall.h>
#include <thread>
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

// Immutable region list for the SIGSEGV handler: a std::mutex is not
// async-signal-safe (a fault while another thread holds regionsMx in
// startTracking/getDirtyPages would deadlock), so the handler reads an
// atomically-swapped snapshot guarded by a hazard counter instead.
struct RegionList
{
    std::vector<std::shared_ptr<TrackedRegion>> items;
};
std::atomic<RegionList*> activeRegions{ nullptr };
std::atomic<int> handlersActive{ 0 };

// Rebuild the handler snapshot from `regions`; caller holds regionsMx.
void publishRegionList()
{
    auto* next = new RegionList;
    next->items.reserve(regions.size());
    for (auto& [base, region] : regions) {
        next->items.push_back(region);
    }
    RegionList* old =
      activeRegions.exchange(next, std::memory_order_acq_rel);
    if (old != nullptr) {
        // Wait for in-flight handlers before reclaiming the old list
        while (handlersActive.load(std::memory_order_acquire) != 0) {
        }
        delete old;
    }
}

// Thread-local dirty channel: region base → flags
thread_local std::map<uint8_t*, std::vector<char>> threadDirty;
thread_local bool threadTrackingOn = false;

struct sigaction oldSegvAction;
bool handlerInstalled = false;

void segvHandler(int sig, siginfo_t* info, void* ucontext)
{
    uint8_t* addr = (uint8_t*)info->si_addr;
    handlersActive.fetch_add(1, std::memory_order_acq_rel);
    RegionList* list = activeRegions.load(std::memory_order_acquire);
    TrackedRegion* hit = nullptr;
    if (list != nullptr) {
        for (auto& region : list->items) {
            if (addr >= region->base &&
                addr < region->base + region->size) {
                hit = region.get();
                break;
            }
        }
    }
    if (hit == nullptr) {
        handlersActive.fetch_sub(1, std::memory_order_acq_rel);
        // Not ours: restore the previous handler and re-raise
        sigaction(SIGSEGV, &oldSegvAction, nullptr);
        raise(sig);
        return;
    }
    size_t page = (size_t)(addr - hit->base) / TRACK_PAGE;
    hit->globalDirty[page] = 1;
    if (threadTrackingOn) {
        // threadDirty is this thread's own map; the entry is preallocated
        // by startThreadLocalTracking so the common path does not allocate
        auto& flags = threadDirty[hit->base];
        if (flags.size() <= page) {
            flags.resize(nPagesOf(hit->size), 0);
        }
        flags[page] = 1;
    }
    // Re-enable the page for writing
    mprotect(hit->base + page * TRACK_PAGE, TRACK_PAGE,
             PROT_READ | PROT_WRITE);
    handlersActive.fetch_sub(1, std::memory_order_acq_rel);
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
        publishRegionList();
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

// ------------------------- soft-dirty PTE ----------------------------------

// Implementation notes: the pagemap is an array of 8-byte entries, one
// per virtual page; bit 55 is the soft-dirty flag set by the kernel on
// the first write after a clear_refs reset (reference semantics:
// util/dirty.h:58-90). pread at (vaddr / PAGE_SIZE) * 8 reads just the
// region's entries.

SoftPTEDirtyTracker::SoftPTEDirtyTracker()
{
    clearRefsFd = ::open("/proc/self/clear_refs", O_WRONLY);
    pagemapFd = ::open("/proc/self/pagemap", O_RDONLY);
    if (clearRefsFd < 0 || pagemapFd < 0) {
        if (clearRefsFd >= 0) {
            ::close(clearRefsFd);
        }
        if (pagemapFd >= 0) {
            ::close(pagemapFd);
        }
        throw FaabricException("soft-dirty PTE tracking unavailable");
    }
}

SoftPTEDirtyTracker::~SoftPTEDirtyTracker()
{
    if (clearRefsFd >= 0) {
        ::close(clearRefsFd);
    }
    if (pagemapFd >= 0) {
        ::close(pagemapFd);
    }
}

bool SoftPTEDirtyTracker::isAvailable()
{
    // Functional probe: kernels without CONFIG_MEM_SOFT_DIRTY accept the
    // clear_refs write but never set pagemap bit 55, so checking the
    // files open is not enough — write a page and look for the bit
    static int cached = -1;
    if (cached >= 0) {
        return cached == 1;
    }
    cached = 0;
    int crFd = ::open("/proc/self/clear_refs", O_WRONLY);
    int pmFd = ::open("/proc/self/pagemap", O_RDONLY);
    if (crFd >= 0 && pmFd >= 0) {
        void* page = ::mmap(nullptr,
                            TRACK_PAGE,
                            PROT_READ | PROT_WRITE,
                            MAP_PRIVATE | MAP_ANONYMOUS,
                            -1,
                            0);
        if (page != MAP_FAILED) {
            *(volatile uint8_t*)page = 1; // fault in
            if (::write(crFd, "4", 1) == 1) {
                *(volatile uint8_t*)page = 2;
                uint64_t entry = 0;
                off_t off = (off_t)((uintptr_t)page / TRACK_PAGE) *
                            (off_t)sizeof(uint64_t);
                if (::pread(pmFd, &entry, sizeof(entry), off) ==
                      (ssize_t)sizeof(entry) &&
                    (entry & (1ULL << 55)) != 0) {
                    cached = 1;
                }
            }
            ::munmap(page, TRACK_PAGE);
        }
    }
    if (crFd >= 0) {
        ::close(crFd);
    }
    if (pmFd >= 0) {
        ::close(pmFd);
    }
    return cached == 1;
}

void SoftPTEDirtyTracker::startTracking(uint8_t*, size_t)
{
    // Process-wide reset of every soft-dirty bit
    if (::write(clearRefsFd, "4", 1) != 1) {
        throw FaabricException("clear_refs write failed");
    }
}

void SoftPTEDirtyTracker::stopTracking(uint8_t*, size_t) {}

std::vector<char> SoftPTEDirtyTracker::getDirtyPages(uint8_t* region,
                                                     size_t size)
{
    size_t nPages = nPagesOf(size);
    std::vector<char> flags(nPages, 0);
    if (nPages == 0) {
        return flags;
    }
    std::vector<uint64_t> entries(nPages);
    off_t off =
      (off_t)((uintptr_t)region / TRACK_PAGE) * (off_t)sizeof(uint64_t);
    ssize_t want = (ssize_t)(nPages * sizeof(uint64_t));
    ssize_t got = ::pread(pagemapFd, entries.data(), want, off);
    if (got != want) {
        throw FaabricException("pagemap read failed");
    }
    constexpr uint64_t SOFT_DIRTY_BIT = 1ULL << 55;
    for (size_t i = 0; i < nPages; i++) {
        flags[i] = (entries[i] & SOFT_DIRTY_BIT) != 0 ? 1 : 0;
    }
    return flags;
}

void SoftPTEDirtyTracker::startThreadLocalTracking(uint8_t*, size_t) {}
void SoftPTEDirtyTracker::stopThreadLocalTracking(uint8_t*, size_t) {}

std::vector<char> SoftPTEDirtyTracker::getThreadLocalDirtyPages(uint8_t*,
                                                                size_t)
{
    // Soft-dirty is process-wide: no per-thread channel (reference
    // SoftPTE tracker returns empty here too)
    return {};
}

// ------------------------- userfaultfd (wp mode) -----------------------------

namespace {

struct UffdRegion
{
    uint8_t* base = nullptr;
    size_t size = 0;
    std::vector<char> dirty; // exact, written only by the poller thread
    std::mutex mx;
};

struct UffdState
{
    int fd = -1;
    int stopPipe[2] = { -1, -1 };
    std::thread poller;
    std::mutex mx;
    std::map<uint8_t*, std::shared_ptr<UffdRegion>> regions;
};

UffdState uffd;

// Thread-local window baselines for the thread-local channel
thread_local std::map<uint8_t*, std::vector<char>> uffdThreadBaseline;

void uffdPollLoop()
{
    while (true) {
        struct pollfd fds[2];
        fds[0] = { uffd.fd, POLLIN, 0 };
        fds[1] = { uffd.stopPipe[0], POLLIN, 0 };
        if (::poll(fds, 2, -1) < 0) {
            if (errno == EINTR) {
                continue;
            }
            break;
        }
        if (fds[1].revents != 0) {
            break; // shutdown
        }
        struct uffd_msg msg;
        ssize_t n = ::read(uffd.fd, &msg, sizeof(msg));
        if (n <= 0) {
            continue;
        }
        if (msg.event != UFFD_EVENT_PAGEFAULT) {
            continue;
        }
        uint8_t* addr = (uint8_t*)(uintptr_t)msg.arg.pagefault.address;
        std::shared_ptr<UffdRegion> hit;
        {
            std::lock_guard<std::mutex> lock(uffd.mx);
            for (auto& [base, region] : uffd.regions) {
                if (addr >= base && addr < base + region->size) {
                    hit = region;
                    break;
                }
            }
        }
        uint8_t* pageAddr = (uint8_t*)((uintptr_t)addr & ~(TRACK_PAGE - 1));
        if (hit) {
            size_t page = (size_t)(pageAddr - hit->base) / TRACK_PAGE;
            std::lock_guard<std::mutex> lock(hit->mx);
            hit->dirty[page] = 1;
        }
        // Drop write-protection on the faulting page and wake the writer
        struct uffdio_writeprotect wp;
        wp.range.start = (uintptr_t)pageAddr;
        wp.range.len = TRACK_PAGE;
        wp.mode = 0; // clear WP
        (void)::ioctl(uffd.fd, UFFDIO_WRITEPROTECT, &wp);
        // Belt-and-braces: wake explicitly (the clear-WP wake alone has
        // been seen to leave a second waiter asleep on this kernel)
        struct uffdio_range wake;
        wake.start = wp.range.start;
        wake.len = wp.range.len;
        (void)::ioctl(uffd.fd, UFFDIO_WAKE, &wake);
    }
}

} // namespace

bool UffdDirtyTracker::isAvailable()
{
    int fd = (int)syscall(SYS_userfaultfd, O_CLOEXEC | O_NONBLOCK);
    if (fd < 0) {
        return false;
    }
    struct uffdio_api api;
    api.api = UFFD_API;
    api.features = UFFD_FEATURE_PAGEFAULT_FLAG_WP;
    bool ok = ::ioctl(fd, UFFDIO_API, &api) == 0 &&
              (api.features & UFFD_FEATURE_PAGEFAULT_FLAG_WP) != 0;
    ::close(fd);
    return ok;
}

UffdDirtyTracker::UffdDirtyTracker()
{
    uffd.fd = (int)syscall(SYS_userfaultfd, O_CLOEXEC | O_NONBLOCK);
    if (uffd.fd < 0) {
        throw FaabricException("userfaultfd syscall unavailable");
    }
    struct uffdio_api api;
    api.api = UFFD_API;
    api.features = UFFD_FEATURE_PAGEFAULT_FLAG_WP;
    if (::ioctl(uffd.fd, UFFDIO_API, &api) != 0 ||
        (api.features & UFFD_FEATURE_PAGEFAULT_FLAG_WP) == 0) {
        ::close(uffd.fd);
        uffd.fd = -1;
        throw FaabricException("kernel lacks uffd write-protect");
    }
    if (::pipe(uffd.stopPipe) != 0) {
        ::close(uffd.fd);
        uffd.fd = -1;
        throw FaabricException("uffd stop pipe failed");
    }
    uffd.poller = std::thread(uffdPollLoop);
}

UffdDirtyTracker::~UffdDirtyTracker()
{
    if (uffd.fd < 0) {
        return;
    }
    char b = 1;
    (void)!::write(uffd.stopPipe[1], &b, 1);
    if (uffd.poller.joinable()) {
        uffd.poller.join();
    }
    ::close(uffd.stopPipe[0]);
    ::close(uffd.stopPipe[1]);
    ::close(uffd.fd);
    uffd.fd = -1;
    uffd.regions.clear();
}

void UffdDirtyTracker::startTracking(uint8_t* region, size_t size)
{
    if (((uintptr_t)region % TRACK_PAGE) != 0) {
        throw FaabricException("uffd tracker needs page-aligned region");
    }
    size_t len = nPagesOf(size) * TRACK_PAGE;
    auto tracked = std::make_shared<UffdRegion>();
    tracked->base = region;
    tracked->size = len;
    tracked->dirty.assign(nPagesOf(size), 0);
    {
        std::lock_guard<std::mutex> lock(uffd.mx);
        uffd.regions[region] = tracked;
    }
    struct uffdio_register reg;
    reg.range.start = (uintptr_t)region;
    reg.range.len = len;
    reg.mode = UFFDIO_REGISTER_MODE_WP;
    if (::ioctl(uffd.fd, UFFDIO_REGISTER, &reg) != 0) {
        throw FaabricException("UFFDIO_REGISTER failed");
    }
    struct uffdio_writeprotect wp;
    wp.range.start = (uintptr_t)region;
    wp.range.len = len;
    wp.mode = UFFDIO_WRITEPROTECT_MODE_WP;
    if (::ioctl(uffd.fd, UFFDIO_WRITEPROTECT, &wp) != 0) {
        throw FaabricException("UFFDIO_WRITEPROTECT arm failed");
    }
}

void UffdDirtyTracker::stopTracking(uint8_t* region, size_t size)
{
    size_t len = nPagesOf(size) * TRACK_PAGE;
    struct uffdio_writeprotect wp;
    wp.range.start = (uintptr_t)region;
    wp.range.len = len;
    wp.mode = 0;
    (void)::ioctl(uffd.fd, UFFDIO_WRITEPROTECT, &wp);
    struct uffdio_range range;
    range.start = (uintptr_t)region;
    range.len = len;
    (void)::ioctl(uffd.fd, UFFDIO_UNREGISTER, &range);
}

std::vector<char> UffdDirtyTracker::getDirtyPages(uint8_t* region,
                                                  size_t size)
{
    std::shared_ptr<UffdRegion> tracked;
    {
        std::lock_guard<std::mutex> lock(uffd.mx);
        auto it = uffd.regions.find(region);
        if (it == uffd.regions.end()) {
            return std::vector<char>(nPagesOf(size), 0);
        }
        tracked = it->second;
    }
    std::lock_guard<std::mutex> lock(tracked->mx);
    return tracked->dirty;
}

void UffdDirtyTracker::startThreadLocalTracking(uint8_t* region, size_t size)
{
    uffdThreadBaseline[region] = getDirtyPages(region, size);
}

void UffdDirtyTracker::stopThreadLocalTracking(uint8_t* region, size_t size)
{
    (void)region;
    (void)size;
}

std::vector<char> UffdDirtyTracker::getThreadLocalDirtyPages(uint8_t* region,
                                                             size_t size)
{
    std::vector<char> now = getDirtyPages(region, size);
    auto it = uffdThreadBaseline.find(region);
    if (it == uffdThreadBaseline.end()) {
        return now;
    }
    for (size_t i = 0; i < now.size() && i < it->second.size(); i++) {
        if (it->second[i]) {
            now[i] = 0; // dirty before this thread's window: not ours
        }
    }
    return now;
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
    } else if (mode == "softpte") {
        if (SoftPTEDirtyTracker::isAvailable()) {
            trackerInstance = std::make_shared<SoftPTEDirtyTracker>();
        } else {
            FAM_WARN("soft-dirty PTE unavailable; falling back to "
                     "segfault dirty tracking");
            trackerInstance = std::make_shared<SegfaultDirtyTracker>();
        }
    } else if (mode == "uffd") {
        if (UffdDirtyTracker::isAvailable()) {
            trackerInstance = std::make_shared<UffdDirtyTracker>();
        } else {
            FAM_WARN("uffd-wp unavailable; falling back to segfault "
                     "dirty tracking");
            trackerInstance = std::make_shared<SegfaultDirtyTracker>();
        }
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
