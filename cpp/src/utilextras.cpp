#include "faabricamd/utilextras.h"
#include "faabricamd/util.h"

#include <atomic>
#include <chrono>
#include <csignal>
#include <cstring>
#include <execinfo.h>
#include <map>
#include <mutex>
#include <pthread.h>
#include <sched.h>
#include <sstream>
#include <thread>
#include <zlib.h>

namespace faabricamd {

// ------------------------- delta codec --------------------------------------

// Command tags (format structurally mirrors src/util/delta.cpp:155-169)
enum DeltaCmd : uint8_t
{
    DELTA_TOTAL_SIZE = 1,
    DELTA_XOR_PAGE = 2,
    DELTA_RAW_RANGE = 3,
    DELTA_COMPRESSED = 4,      // zlib stream
    DELTA_COMPRESSED_ZSTD = 5, // zstd stream (reference's codec)
    DELTA_END = 0xff,
};

// The image ships libzstd.so.1 without headers; the C ABI below is
// stable across zstd 1.x so we declare exactly what the codec needs
// and link -l:libzstd.so.1 (matches the reference's zstd dependency,
// /root/reference/src/util/delta.cpp:15-57)
extern "C" {
size_t ZSTD_compressBound(size_t srcSize);
size_t ZSTD_compress(void* dst,
                     size_t dstCapacity,
                     const void* src,
                     size_t srcSize,
                     int level);
size_t ZSTD_decompress(void* dst,
                       size_t dstCapacity,
                       const void* src,
                       size_t srcSize);
unsigned ZSTD_isError(size_t code);
}
static constexpr uint8_t DELTA_VERSION = 1;

DeltaConfig DeltaConfig::parse(const std::string& s)
{
    DeltaConfig conf;
    conf.xorWithOld = s.find("xor") != std::string::npos;
    conf.compress = s.find("zlib=1") != std::string::npos ||
                    s.find("zstd=1") != std::string::npos;
    conf.useZstd = s.find("zlib=1") == std::string::npos;
    auto pos = s.find("pages=");
    if (pos != std::string::npos) {
        conf.pageSize = (size_t)atoll(s.c_str() + pos + 6);
    }
    return conf;
}

std::string DeltaConfig::str() const
{
    std::string out = "pages=" + std::to_string(pageSize);
    if (xorWithOld) {
        out += ";xor";
    }
    if (compress) {
        out += useZstd ? ";zstd=1" : ";zlib=1";
    }
    return out;
}

static void putU64(std::vector<uint8_t>& out, uint64_t v)
{
    for (int i = 0; i < 8; i++) {
        out.push_back((uint8_t)(v >> (8 * i)));
    }
}

static uint64_t getU64(const uint8_t* p)
{
    uint64_t v = 0;
    for (int i = 0; i < 8; i++) {
        v |= (uint64_t)p[i] << (8 * i);
    }
    return v;
}

static std::vector<uint8_t> zlibCompress(const uint8_t* data, size_t n)
{
    uLongf bound = compressBound((uLong)n);
    std::vector<uint8_t> out(bound);
    if (compress2(out.data(), &bound, data, (uLong)n, Z_BEST_SPEED) !=
        Z_OK) {
        throw FaabricException("zlib compress failed");
    }
    out.resize(bound);
    return out;
}

static std::vector<uint8_t> zlibDecompress(const uint8_t* data,
                                           size_t n,
                                           size_t expected)
{
    std::vector<uint8_t> out(expected);
    uLongf outLen = (uLongf)expected;
    if (uncompress(out.data(), &outLen, data, (uLong)n) != Z_OK) {
        throw FaabricException("zlib decompress failed");
    }
    out.resize(outLen);
    return out;
}

std::vector<uint8_t> deltaEncode(const std::vector<uint8_t>& oldData,
                                 const std::vector<uint8_t>& newData,
                                 const DeltaConfig& conf)
{
    std::vector<uint8_t> cmds;
    cmds.push_back(DELTA_TOTAL_SIZE);
    putU64(cmds, newData.size());

    size_t common = std::min(oldData.size(), newData.size());
    size_t page = conf.pageSize;

    for (size_t off = 0; off < common; off += page) {
        size_t len = std::min(page, common - off);
        if (std::memcmp(oldData.data() + off, newData.data() + off, len) ==
            0) {
            continue;
        }
        if (conf.xorWithOld) {
            cmds.push_back(DELTA_XOR_PAGE);
            putU64(cmds, off);
            putU64(cmds, len);
            size_t base = cmds.size();
            cmds.resize(base + len);
            for (size_t i = 0; i < len; i++) {
                cmds[base + i] = oldData[off + i] ^ newData[off + i];
            }
        } else {
            cmds.push_back(DELTA_RAW_RANGE);
            putU64(cmds, off);
            putU64(cmds, len);
            cmds.insert(cmds.end(),
                        newData.begin() + off,
                        newData.begin() + off + len);
        }
    }
    // Extension tail ships raw
    if (newData.size() > common) {
        cmds.push_back(DELTA_RAW_RANGE);
        putU64(cmds, common);
        putU64(cmds, newData.size() - common);
        cmds.insert(cmds.end(), newData.begin() + common, newData.end());
    }
    cmds.push_back(DELTA_END);

    std::vector<uint8_t> out;
    out.push_back(DELTA_VERSION);
    if (conf.compress && conf.useZstd) {
        size_t bound = ZSTD_compressBound(cmds.size());
        std::vector<uint8_t> compressed(bound);
        size_t n = ZSTD_compress(compressed.data(), bound, cmds.data(),
                                 cmds.size(), /*level=*/1);
        if (ZSTD_isError(n)) {
            throw FaabricException("zstd compress failed");
        }
        compressed.resize(n);
        out.push_back(DELTA_COMPRESSED_ZSTD);
        putU64(out, cmds.size());
        putU64(out, compressed.size());
        out.insert(out.end(), compressed.begin(), compressed.end());
    } else if (conf.compress) {
        auto compressed = zlibCompress(cmds.data(), cmds.size());
        out.push_back(DELTA_COMPRESSED);
        putU64(out, cmds.size());
        putU64(out, compressed.size());
        out.insert(out.end(), compressed.begin(), compressed.end());
    } else {
        out.insert(out.end(), cmds.begin(), cmds.end());
    }
    return out;
}

std::vector<uint8_t> deltaApply(const std::vector<uint8_t>& oldData,
                                const std::vector<uint8_t>& delta)
{
    if (delta.empty() || delta[0] != DELTA_VERSION) {
        throw FaabricException("bad delta version");
    }
    std::vector<uint8_t> cmds;
    size_t pos = 1;
    if (pos < delta.size() && delta[pos] == DELTA_COMPRESSED_ZSTD) {
        uint64_t rawLen = getU64(delta.data() + pos + 1);
        uint64_t compLen = getU64(delta.data() + pos + 9);
        cmds.resize(rawLen);
        size_t n = ZSTD_decompress(cmds.data(), rawLen,
                                   delta.data() + pos + 17, compLen);
        if (ZSTD_isError(n) || n != rawLen) {
            throw FaabricException("zstd decompress failed");
        }
    } else if (pos < delta.size() && delta[pos] == DELTA_COMPRESSED) {
        uint64_t rawLen = getU64(delta.data() + pos + 1);
        uint64_t compLen = getU64(delta.data() + pos + 9);
        cmds = zlibDecompress(delta.data() + pos + 17, compLen, rawLen);
    } else {
        cmds.assign(delta.begin() + 1, delta.end());
    }

    std::vector<uint8_t> out = oldData;
    size_t p = 0;
    while (p < cmds.size()) {
        uint8_t cmd = cmds[p++];
        if (cmd == DELTA_END) {
            break;
        }
        switch (cmd) {
            case DELTA_TOTAL_SIZE: {
                uint64_t total = getU64(cmds.data() + p);
                p += 8;
                out.resize(total, 0);
                break;
            }
            case DELTA_XOR_PAGE: {
                uint64_t off = getU64(cmds.data() + p);
                uint64_t len = getU64(cmds.data() + p + 8);
                p += 16;
                for (uint64_t i = 0; i < len; i++) {
                    out[off + i] ^= cmds[p + i];
                }
                p += len;
                break;
            }
            case DELTA_RAW_RANGE: {
                uint64_t off = getU64(cmds.data() + p);
                uint64_t len = getU64(cmds.data() + p + 8);
                p += 16;
                std::memcpy(out.data() + off, cmds.data() + p, len);
                p += len;
                break;
            }
            default:
                throw FaabricException("bad delta command");
        }
    }
    return out;
}

// ------------------------- PROF timers --------------------------------------

namespace {
struct ProfState
{
    std::mutex mx;
    std::map<std::string, double> totalsMs;
    std::map<std::string, int64_t> counts;
};
ProfState& profState()
{
    static ProfState st;
    return st;
}
thread_local std::map<std::string, std::chrono::steady_clock::time_point>
  profStarts;
} // namespace

// PROF aggregation is opt-in (FAABRIC_PROF=1) — the reference gates its
// PROF macros behind TRACE_ALL for the same reason: map + mutex work on
// every call is measurable in hot paths (3 pairs per message in the
// batch path)
static bool profEnabled()
{
    static const bool v = getEnvVarInt("FAABRIC_PROF", 0) != 0;
    return v;
}

void profStart(const std::string& name)
{
    if (!profEnabled()) {
        return;
    }
    profStarts[name] = std::chrono::steady_clock::now();
}

void profEnd(const std::string& name)
{
    if (!profEnabled()) {
        return;
    }
    auto it = profStarts.find(name);
    if (it == profStarts.end()) {
        return;
    }
    double ms = std::chrono::duration<double, std::milli>(
                  std::chrono::steady_clock::now() - it->second)
                  .count();
    profStarts.erase(it);
    auto& st = profState();
    std::lock_guard<std::mutex> lock(st.mx);
    st.totalsMs[name] += ms;
    st.counts[name] += 1;
}

std::vector<std::pair<std::string, double>> profTotalsMs()
{
    auto& st = profState();
    std::lock_guard<std::mutex> lock(st.mx);
    return { st.totalsMs.begin(), st.totalsMs.end() };
}

void profClear()
{
    auto& st = profState();
    std::lock_guard<std::mutex> lock(st.mx);
    st.totalsMs.clear();
    st.counts.clear();
}

std::string profSummary()
{
    auto& st = profState();
    std::lock_guard<std::mutex> lock(st.mx);
    std::ostringstream out;
    out << "--- PROF totals ---\n";
    for (const auto& [name, ms] : st.totalsMs) {
        out << name << ": " << ms << " ms (" << st.counts[name]
            << " calls)\n";
    }
    return out.str();
}

// ------------------------- crash handler ------------------------------------

static void crashHandler(int sig)
{
    void* frames[64];
    int n = backtrace(frames, 64);
    fprintf(stderr, "--- crash: signal %d (%s) ---\n", sig,
            strsignal(sig));
    backtrace_symbols_fd(frames, n, 2);
    signal(sig, SIG_DFL);
    raise(sig);
}

void setUpCrashHandler()
{
    // SIGSEGV deliberately left alone: the segfault dirty tracker owns it
    // (reference: src/util/crash.cpp:40-67 + tests/dist/main.cpp:3-4)
    signal(SIGABRT, crashHandler);
    signal(SIGILL, crashHandler);
    signal(SIGFPE, crashHandler);
}

// ------------------------- CPU pinning --------------------------------------

namespace {
std::mutex pinMx;
std::vector<bool> pinnedCores;
thread_local int myPinnedCore = -1;
} // namespace

int pinThreadToFreeCpu()
{
    int nCores = (int)std::thread::hardware_concurrency();
    std::lock_guard<std::mutex> lock(pinMx);
    if ((int)pinnedCores.size() != nCores) {
        pinnedCores.assign(nCores, false);
    }
    for (int i = 0; i < nCores; i++) {
        if (!pinnedCores[i]) {
            cpu_set_t set;
            CPU_ZERO(&set);
            CPU_SET(i, &set);
            if (pthread_setaffinity_np(pthread_self(), sizeof(set), &set) ==
                0) {
                pinnedCores[i] = true;
                myPinnedCore = i;
                return i;
            }
        }
    }
    return -1;
}

void unpinThisThread()
{
    std::lock_guard<std::mutex> lock(pinMx);
    if (myPinnedCore >= 0 && myPinnedCore < (int)pinnedCores.size()) {
        pinnedCores[myPinnedCore] = false;
        myPinnedCore = -1;
    }
}

} // namespace faabricamd
