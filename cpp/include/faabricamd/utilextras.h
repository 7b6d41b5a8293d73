// Util parity extras: page-delta codec, PROF timers, crash handler, CPU
// pinning (reference: src/util/delta.cpp, util/timing.h:6-17,
// src/util/crash.cpp:15-67, util/hwloc.h:31).
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace faabricamd {

// ------------------------- delta codec --------------------------------------
// Page-wise XOR-against-old + zstd compression, config string
// "pages=4096;xor;zstd=1" like the reference (src/util/config.cpp:27,
// delta.cpp:15-57); "zlib=1" selects a zlib fallback. Stream format is
// command-tagged like the reference's (delta.cpp:155-169).

struct DeltaConfig
{
    size_t pageSize = 4096;
    bool xorWithOld = true;
    bool compress = true;
    // zstd by default, matching the reference's "pages=4096;xor;zstd=1"
    // (src/util/config.cpp:27); "zlib=1" selects the zlib fallback
    bool useZstd = true;

    static DeltaConfig parse(const std::string& s);
    std::string str() const;
};

// Encode new against old (old may be shorter; the tail ships raw)
std::vector<uint8_t> deltaEncode(const std::vector<uint8_t>& oldData,
                                 const std::vector<uint8_t>& newData,
                                 const DeltaConfig& conf = {});

// Apply a delta to old, producing new
std::vector<uint8_t> deltaApply(const std::vector<uint8_t>& oldData,
                                const std::vector<uint8_t>& delta);

// ------------------------- PROF timers --------------------------------------
// Compile-time-free runtime timers: totals aggregated per name
// (reference: util/timing.h PROF_START/PROF_END under TRACE_ALL)

void profStart(const std::string& name);
void profEnd(const std::string& name);
std::vector<std::pair<std::string, double>> profTotalsMs();
void profClear();
std::string profSummary();

#define PROF_START(name) ::faabricamd::profStart(#name);
#define PROF_END(name) ::faabricamd::profEnd(#name);

// ------------------------- crash handler ------------------------------------
// Backtrace-printing handlers for SIGABRT/SIGILL/SIGFPE; SIGSEGV stays
// free for the segfault dirty tracker (reference: src/util/crash.cpp:40-67)
void setUpCrashHandler();

// ------------------------- CPU pinning --------------------------------------
// Pin the calling thread to a free core from the process-wide pool;
// returns the core id or -1 (reference: util/hwloc.h pinThreadToFreeCpu)
int pinThreadToFreeCpu();
void unpinThisThread();

} // namespace faabricamd
