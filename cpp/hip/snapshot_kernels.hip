// gfx950 (MI355X / CDNA4) kernels for the snapshot engine: dirty-page
// tracking, XOR delta, diff compaction and typed merge application over
// HBM3E-resident snapshots.
//
// These replace the reference's CPU byte-crunching loops
// (reference: src/util/snapshot.cpp:30 diffArrayRegions, :363-402
// applyDiff(s), util/snapshot.h:163-246 calculateDiffValue/applyDiffValue,
// src/util/dirty.cpp trackers — fault-driven tracking has no HBM
// equivalent, so dirty pages come from a compare kernel against the
// baseline, the reference's DIFFING_MODE=xor made native).
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  - memory-bound throughout: 16 B/lane uint4 loads (G13), 256-thread
//    blocks. The page diff/apply kernels launch one wave per page with
//    NO grid cap: a 2048-block grid-stride form measured 4.87 TB/s at
//    25% scattered dirty vs 5.92 uncapped — the chip wants >>256
//    independent workgroups, not resident loops (G11 applies to the
//    small helper kernels only)
//  - one 4 KiB page = one 256-thread block iteration (256 × 16 B);
//    per-wave ballot keeps dirty-flag atomics to ≤4 per dirty page (G12)
//  - compaction via device-scope atomic ticket per dirty page; payload
//    writes stay coalesced within the page

#include <hip/hip_runtime.h>

#include <cstdint>

#define FAM_KERNEL_BLOCK 256
#define FAM_MAX_BLOCKS 2048
#define FAM_PAGE 4096

namespace {

using u8 = uint8_t;
using u32 = uint32_t;
using u64 = uint64_t;

__device__ __forceinline__ bool neq16(const uint4& a, const uint4& b)
{
    return (a.x ^ b.x) | (a.y ^ b.y) | (a.z ^ b.z) | (a.w ^ b.w);
}

// ---------------------------------------------------------------------------
// Dirty-page bitmap: flags[p] = 1 where any byte of 4 KiB page p differs.
// Each block iteration covers one page: 256 lanes × 16 B.
// ---------------------------------------------------------------------------
typedef unsigned int nt_u32v4 __attribute__((ext_vector_type(4)));

template<bool NT>
__device__ inline uint4 ldVec(const uint4* p)
{
    if constexpr (NT) {
        nt_u32v4 v =
          __builtin_nontemporal_load((const nt_u32v4*)p);
        return make_uint4(v.x, v.y, v.z, v.w);
    }
    return *p;
}

template<bool NT>
__device__ inline void stVec(uint4* p, uint4 v)
{
    if constexpr (NT) {
        nt_u32v4 w = { v.x, v.y, v.z, v.w };
        __builtin_nontemporal_store(w, (nt_u32v4*)p);
    } else {
        *p = v;
    }
}

__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void dirtyPagesKernel(
  const uint4* __restrict__ snap,
  const uint4* __restrict__ cur,
  u32 nPages,
  u32* __restrict__ flags)
{
    const u32 vecsPerPage = FAM_PAGE / 16; // 256
    for (u32 page = blockIdx.x; page < nPages; page += gridDim.x) {
        u64 v = (u64)page * vecsPerPage + threadIdx.x;
        bool diff = neq16(ldVec<true>(&snap[v]), ldVec<true>(&cur[v]));
        u64 mask = __ballot(diff);
        if ((threadIdx.x & 63) == 0 && mask != 0) {
            atomicOr(&flags[page], 1u);
        }
    }
}

// ---------------------------------------------------------------------------
// XOR delta over a whole buffer: out = a ^ b (16 B/lane, grid-stride)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void xorBufferKernel(
  const uint4* __restrict__ a,
  const uint4* __restrict__ b,
  uint4* __restrict__ out,
  u64 nVec)
{
    // 2-way unrolled like copyBufferKernel: overlaps the second pair of
    // loads with the first store's drain
    u64 stride = (u64)gridDim.x * blockDim.x;
    u64 i = (u64)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < nVec; i += 2 * stride) {
        uint4 x0 = ldVec<true>(&a[i]);
        uint4 y0 = ldVec<true>(&b[i]);
        uint4 x1 = ldVec<true>(&a[i + stride]);
        uint4 y1 = ldVec<true>(&b[i + stride]);
        stVec<true>(&out[i], make_uint4(x0.x ^ y0.x, x0.y ^ y0.y,
                                        x0.z ^ y0.z, x0.w ^ y0.w));
        stVec<true>(&out[i + stride],
                    make_uint4(x1.x ^ y1.x, x1.y ^ y1.y, x1.z ^ y1.z,
                               x1.w ^ y1.w));
    }
    for (; i < nVec; i += stride) {
        uint4 x = ldVec<true>(&a[i]);
        uint4 y = ldVec<true>(&b[i]);
        stVec<true>(&out[i],
                    make_uint4(x.x ^ y.x, x.y ^ y.y, x.z ^ y.z, x.w ^ y.w));
    }
}

// Plain device-to-device copy, nontemporal both sides: hipMemcpy D2D
// measures ~1.9 TB/s on MI355X while a grid-strided NT copy approaches
// the 2-unit HBM bound (~3.15 TB/s) — used for the single-rank
// collective short-circuits (MpiWorld)
__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void copyBufferKernel(
  const uint4* __restrict__ src,
  uint4* __restrict__ dst,
  u64 nVec)
{
    // 2 vecs per lane per iteration: the second load issues while the
    // first store's cacheline drains
    u64 stride = (u64)gridDim.x * blockDim.x;
    u64 i = (u64)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < nVec; i += 2 * stride) {
        uint4 a = ldVec<true>(&src[i]);
        uint4 b = ldVec<true>(&src[i + stride]);
        stVec<true>(&dst[i], a);
        stVec<true>(&dst[i + stride], b);
    }
    for (; i < nVec; i += stride) {
        stVec<true>(&dst[i], ldVec<true>(&src[i]));
    }
}

// ---------------------------------------------------------------------------
// Page diff, two-kernel pipeline:
//
//  (A) diffXorPagesKernel — one WAVE per page (64 lanes × 4 × 16 B = 4 KiB):
//      compare, and for dirty pages write the XOR payload SPARSELY at the
//      page's own slot plus one bit in a page bitmap. The bitmap atomics
//      are distributed over nPages/32 words, so there is no contention —
//      a single global ticket counter measured 1.07 TB/s at 100% dirty
//      (≈11 ns/atomic on one word, the "dequeue" price), the bitmap form
//      removes that serialization entirely.
//  (B) compactPageIdxKernel — scan the bitmap (2048 pages per wave
//      iteration) and emit the compacted dirty-page index list with ONE
//      ticket atomic per wave-iteration.
//
// Consumers (apply / ship) address the payload via pageIdx[slot] → page.
// ---------------------------------------------------------------------------
// The 4 GiB-scale working sets never fit in L2 (32 MiB/XCD), so the NT
// instantiation streams through with slc-tagged loads/stores
template<bool NT>
__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void diffXorPagesKernel(
  const uint4* __restrict__ snap,
  const uint4* __restrict__ cur,
  u32 nPages,
  u32* __restrict__ bitmap, // nPages/32 words, zeroed before launch
  uint4* __restrict__ payloadOut) // sparse: indexed by page
{
    const u32 lane = threadIdx.x & 63;
    const u32 waveId =
      (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const u32 nWaves = (gridDim.x * blockDim.x) >> 6;
    const u32 vecsPerPage = FAM_PAGE / 16; // 256

    for (u32 page = waveId; page < nPages; page += nWaves) {
        u64 base = (u64)page * vecsPerPage + lane;
        uint4 s0 = ldVec<NT>(&snap[base]);
        uint4 s1 = ldVec<NT>(&snap[base + 64]);
        uint4 s2 = ldVec<NT>(&snap[base + 128]);
        uint4 s3 = ldVec<NT>(&snap[base + 192]);
        uint4 c0 = ldVec<NT>(&cur[base]);
        uint4 c1 = ldVec<NT>(&cur[base + 64]);
        uint4 c2 = ldVec<NT>(&cur[base + 128]);
        uint4 c3 = ldVec<NT>(&cur[base + 192]);
        bool diff = neq16(s0, c0) || neq16(s1, c1) || neq16(s2, c2) ||
                    neq16(s3, c3);
        u64 mask = __ballot(diff);
        if (mask == 0) {
            continue;
        }
        if (lane == 0) {
            atomicOr(&bitmap[page >> 5], 1u << (page & 31));
        }
        stVec<NT>(&payloadOut[base],
          make_uint4(s0.x ^ c0.x, s0.y ^ c0.y, s0.z ^ c0.z, s0.w ^ c0.w));
        stVec<NT>(&payloadOut[base + 64],
          make_uint4(s1.x ^ c1.x, s1.y ^ c1.y, s1.z ^ c1.z, s1.w ^ c1.w));
        stVec<NT>(&payloadOut[base + 128],
          make_uint4(s2.x ^ c2.x, s2.y ^ c2.y, s2.z ^ c2.z, s2.w ^ c2.w));
        stVec<NT>(&payloadOut[base + 192],
          make_uint4(s3.x ^ c3.x, s3.y ^ c3.y, s3.z ^ c3.z, s3.w ^ c3.w));
    }
}

// Compact the dirty-page bitmap into an index list. Each wave sweeps 64
// words (2048 pages) per iteration and takes a single base ticket.
__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void compactPageIdxKernel(
  const u32* __restrict__ bitmap,
  u32 nWords,
  u32* __restrict__ ticket,
  u32* __restrict__ pageIdxOut)
{
    const u32 lane = threadIdx.x & 63;
    const u32 waveId =
      (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const u32 nWaves = (gridDim.x * blockDim.x) >> 6;

    for (u32 w0 = waveId * 64; w0 < nWords; w0 += nWaves * 64) {
        u32 word = (w0 + lane) < nWords ? bitmap[w0 + lane] : 0u;
        u32 mine = __popc(word);
        // Hillis-Steele inclusive scan over the wave's 64 lane counts
        u32 scan = mine;
        for (u32 off = 1; off < 64; off <<= 1) {
            u32 up = (u32)__shfl_up((int)scan, (int)off);
            if (lane >= off) {
                scan += up;
            }
        }
        u32 prefix = scan - mine; // exclusive
        u32 waveTotal = (u32)__shfl((int)scan, 63);
        if (waveTotal == 0) {
            continue;
        }
        u32 base = 0;
        if (lane == 63) {
            base = atomicAdd(ticket, waveTotal);
        }
        base = (u32)__shfl((int)base, 63);
        u32 out = base + prefix;
        u32 pageBase = (w0 + lane) * 32;
        while (word != 0) {
            u32 bit = __ffs(word) - 1;
            pageIdxOut[out++] = pageBase + bit;
            word &= word - 1;
        }
    }
}

// ---------------------------------------------------------------------------
// Apply XOR page diffs: snap[page] ^= payload[...] for page in pageIdx.
// One wave per dirty page. compact=0: payload is sparse (indexed by
// page, the diff kernel's output); compact=1: payload is slot-compacted
// (the shippable wire form).
// ---------------------------------------------------------------------------
template<bool NT>
__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void applyXorPagesKernel(
  uint4* __restrict__ snap,
  const u32* __restrict__ pageIdx,
  const uint4* __restrict__ payload,
  u32 nDirty,
  int compact)
{
    const u32 lane = threadIdx.x & 63;
    const u32 waveId =
      (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const u32 nWaves = (gridDim.x * blockDim.x) >> 6;
    const u32 vecsPerPage = FAM_PAGE / 16;

    for (u32 slot = waveId; slot < nDirty; slot += nWaves) {
        u32 page = pageIdx[slot];
        u64 dst = (u64)page * vecsPerPage + lane;
        u64 src = compact ? (u64)slot * vecsPerPage + lane : dst;
#pragma unroll
        for (int k = 0; k < 4; k++) {
            uint4 p = ldVec<NT>(&payload[src + k * 64]);
            uint4 s = ldVec<NT>(&snap[dst + k * 64]);
            stVec<NT>(&snap[dst + k * 64],
              make_uint4(s.x ^ p.x, s.y ^ p.y, s.z ^ p.z, s.w ^ p.w));
        }
    }
}

// Gather sparse per-page payloads into the compact wire form:
// out[slot] = payload[pageIdx[slot]]
__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void gatherPagesKernel(
  const uint4* __restrict__ payload,
  const u32* __restrict__ pageIdx,
  uint4* __restrict__ out,
  u32 nDirty)
{
    const u32 lane = threadIdx.x & 63;
    const u32 waveId =
      (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const u32 nWaves = (gridDim.x * blockDim.x) >> 6;
    const u32 vecsPerPage = FAM_PAGE / 16;

    for (u32 slot = waveId; slot < nDirty; slot += nWaves) {
        u32 page = pageIdx[slot];
        u64 src = (u64)page * vecsPerPage + lane;
        u64 dst = (u64)slot * vecsPerPage + lane;
#pragma unroll
        for (int k = 0; k < 4; k++) {
            stVec<true>(&out[dst + k * 64],
                        ldVec<true>(&payload[src + k * 64]));
        }
    }
}

// ---------------------------------------------------------------------------
// Typed elementwise merge: inout[i] = op(inout[i], in[i])
// (MPI op_reduce device form + snapshot typed merges; reference CPU loops
// src/mpi/MpiWorld.cpp:1266-1389, util/snapshot.h:216-246)
// ---------------------------------------------------------------------------
enum FamOp : int
{
    FAM_OP_SUM = 0,
    FAM_OP_MAX = 1,
    FAM_OP_MIN = 2,
    FAM_OP_PROD = 3,
    FAM_OP_SUB = 4,
    FAM_OP_XOR = 5,
};

template<typename T>
__device__ __forceinline__ T famApply(int op, T a, T b)
{
    switch (op) {
        case FAM_OP_SUM:
            return a + b;
        case FAM_OP_MAX:
            return a > b ? a : b;
        case FAM_OP_MIN:
            return a < b ? a : b;
        case FAM_OP_PROD:
            return a * b;
        case FAM_OP_SUB:
            return a - b;
    }
    return a;
}

template<typename T>
__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void elementwiseOpKernel(
  T* __restrict__ inout,
  const T* __restrict__ in,
  u64 n,
  int op)
{
    // `in` is read once (NT); `inout` is RMW and keeps cached accesses
    // (same finding as the apply kernel: streaming the RMW side regresses)
    u64 stride = (u64)gridDim.x * blockDim.x;
    u64 i = (u64)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < n; i += 2 * stride) {
        T a0 = __builtin_nontemporal_load(&in[i]);
        T a1 = __builtin_nontemporal_load(&in[i + stride]);
        inout[i] = famApply<T>(op, inout[i], a0);
        inout[i + stride] = famApply<T>(op, inout[i + stride], a1);
    }
    for (; i < n; i += stride) {
        inout[i] = famApply<T>(op, inout[i],
                               __builtin_nontemporal_load(&in[i]));
    }
}

__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void xorBytesKernel(
  u8* __restrict__ inout,
  const u8* __restrict__ in,
  u64 n)
{
    // Unaligned-safe byte XOR for small typed diffs
    for (u64 i = (u64)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (u64)gridDim.x * blockDim.x) {
        inout[i] ^= in[i];
    }
}

__device__ inline u64 xorshift64(u64 x)
{
    x ^= x << 13;
    x ^= x >> 7;
    x ^= x << 17;
    return x;
}

// Fill a buffer with deterministic pseudo-random bytes (bench init for
// the "random-byte region" snapshot shapes — BASELINE.json config 4)
__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void fillRandomKernel(
  u64* __restrict__ out,
  u64 nWords,
  u64 seed)
{
    for (u64 i = (u64)blockIdx.x * blockDim.x + threadIdx.x; i < nWords;
         i += (u64)gridDim.x * blockDim.x) {
        out[i] = xorshift64(seed ^ (i * 0x9e3779b97f4a7c15ULL) ^
                            (i >> 3));
    }
}

// Mutate a scattered set of pages in place (wave per page, every lane
// perturbs its 64-byte slice) so the diff kernels face a randomized
// dirty distribution, not a contiguous memset
__global__ __launch_bounds__(FAM_KERNEL_BLOCK) void touchPagesKernel(
  u64* __restrict__ buf,
  const u32* __restrict__ pages,
  u32 nPages,
  u64 seed)
{
    const u32 wavesPerBlock = FAM_KERNEL_BLOCK / 64;
    const u32 wave = threadIdx.x / 64;
    const u32 lane = threadIdx.x % 64;
    const u32 wordsPerPage = FAM_PAGE / 8; // 512
    for (u32 p = blockIdx.x * wavesPerBlock + wave; p < nPages;
         p += gridDim.x * wavesPerBlock) {
        u64 base = (u64)pages[p] * wordsPerPage;
        for (u32 w = lane; w < wordsPerPage; w += 64) {
            buf[base + w] ^= xorshift64(seed ^ (base + w));
        }
    }
}

inline u32 gridFor(u64 items)
{
    // One element per lane, uncapped: a 2048-block grid-stride copy
    // measured 5.53 TB/s vs 6.42 at one-per-lane on 256 MB (the chip
    // wants >>256 independent workgroups)
    u64 blocks = (items + FAM_KERNEL_BLOCK - 1) / FAM_KERNEL_BLOCK;
    if (blocks == 0) {
        blocks = 1;
    }
    if (blocks > 0xffffffffu) {
        blocks = 0xffffffffu;
    }
    return (u32)blocks;
}

} // namespace

// --------------------------- C API ------------------------------------------

extern "C" {

hipError_t famDirtyPages(const void* snap,
                         const void* cur,
                         uint64_t bytes,
                         uint32_t* flagsDev,
                         hipStream_t stream)
{
    if ((bytes % FAM_PAGE) != 0) {
        // A silently-floored tail would be missed dirty data
        return hipErrorInvalidValue;
    }
    uint32_t nPages = (uint32_t)(bytes / FAM_PAGE);
    uint32_t grid = nPages ? nPages : 1; // block per page, uncapped
    hipLaunchKernelGGL(dirtyPagesKernel,
                       dim3(grid),
                       dim3(FAM_KERNEL_BLOCK),
                       0,
                       stream,
                       (const uint4*)snap,
                       (const uint4*)cur,
                       nPages,
                       flagsDev);
    return hipGetLastError();
}

hipError_t famFillRandom(void* buf,
                         uint64_t bytes,
                         uint64_t seed,
                         hipStream_t stream)
{
    if ((bytes % 8) != 0) {
        return hipErrorInvalidValue;
    }
    u64 nWords = bytes / 8;
    hipLaunchKernelGGL(fillRandomKernel,
                       dim3(gridFor(nWords)),
                       dim3(FAM_KERNEL_BLOCK),
                       0,
                       stream,
                       (u64*)buf,
                       nWords,
                       seed);
    return hipGetLastError();
}

hipError_t famTouchPages(void* buf,
                         const uint32_t* pagesDev,
                         uint32_t nPages,
                         uint64_t seed,
                         hipStream_t stream)
{
    u32 wavesPerBlock = FAM_KERNEL_BLOCK / 64;
    u32 blocks = (nPages + wavesPerBlock - 1) / wavesPerBlock;
    u32 grid = blocks < FAM_MAX_BLOCKS ? (blocks ? blocks : 1)
                                       : FAM_MAX_BLOCKS;
    hipLaunchKernelGGL(touchPagesKernel,
                       dim3(grid),
                       dim3(FAM_KERNEL_BLOCK),
                       0,
                       stream,
                       (u64*)buf,
                       pagesDev,
                       nPages,
                       seed);
    return hipGetLastError();
}

hipError_t famCopyBuffer(const void* src,
                         void* dst,
                         uint64_t bytes,
                         hipStream_t stream)
{
    if ((bytes % 16) != 0) {
        // Tail not vectorisable: fall back to the runtime copy
        return hipMemcpyAsync(dst, src, bytes, hipMemcpyDeviceToDevice,
                              stream);
    }
    uint64_t nVec = bytes / 16;
    static const u32 copyGrid = []() {
        const char* e = getenv("FAM_COPY_GRID");
        return e ? (u32)atoi(e) : 0;
    }();
    u32 grid = copyGrid ? copyGrid : gridFor(nVec);
    if ((u64)grid * FAM_KERNEL_BLOCK > nVec && nVec > 0) {
        grid = gridFor(nVec);
    }
    hipLaunchKernelGGL(copyBufferKernel,
                       dim3(grid),
                       dim3(FAM_KERNEL_BLOCK),
                       0,
                       stream,
                       (const uint4*)src,
                       (uint4*)dst,
                       nVec);
    return hipGetLastError();
}

hipError_t famXorBuffer(const void* a,
                        const void* b,
                        void* out,
                        uint64_t bytes,
                        hipStream_t stream)
{
    if ((bytes % 16) != 0) {
        // No scalar tail path: reject rather than silently drop bytes
        return hipErrorInvalidValue;
    }
    uint64_t nVec = bytes / 16;
    hipLaunchKernelGGL(xorBufferKernel,
                       dim3(gridFor(nVec)),
                       dim3(FAM_KERNEL_BLOCK),
                       0,
                       stream,
                       (const uint4*)a,
                       (const uint4*)b,
                       (uint4*)out,
                       nVec);
    return hipGetLastError();
}

hipError_t famDiffXorPages(const void* snap,
                           const void* cur,
                           uint64_t bytes,
                           uint32_t* ticketDev, // one u32, zeroed by caller
                           uint32_t* pageIdxDev,
                           void* payloadDev,
                           uint32_t* bitmapDev, // nPages/32 words, zeroed
                           hipStream_t stream)
{
    if ((bytes % FAM_PAGE) != 0) {
        return hipErrorInvalidValue;
    }
    uint32_t nPages = (uint32_t)(bytes / FAM_PAGE);
    uint32_t wavesPerBlock = FAM_KERNEL_BLOCK / 64;
    uint32_t blocks = (nPages + wavesPerBlock - 1) / wavesPerBlock;
    // Launch one wave per page with NO grid cap by default: the old
    // 2048-block grid-stride form measured 4.87 TB/s at 25% scattered
    // dirty vs 5.92 uncapped on the same box (the stride loop starves
    // the 256-CU/8-XCD chip of independent blocks)
    static const uint32_t maxBlocks = []() {
        const char* e = getenv("FAM_DIFF_BLOCKS");
        uint32_t v = e ? (uint32_t)atoi(e) : 0;
        return v ? v : 0xffffffffu;
    }();
    uint32_t grid = blocks < maxBlocks ? (blocks ? blocks : 1)
                                       : maxBlocks;
    static const bool useNt = []() {
        const char* e = getenv("FAM_DIFF_NT");
        return e == nullptr || e[0] != '0'; // default on
    }();
    if (useNt) {
        hipLaunchKernelGGL(diffXorPagesKernel<true>,
                           dim3(grid),
                           dim3(FAM_KERNEL_BLOCK),
                           0,
                           stream,
                           (const uint4*)snap,
                           (const uint4*)cur,
                           nPages,
                           bitmapDev,
                           (uint4*)payloadDev);
    } else {
        hipLaunchKernelGGL(diffXorPagesKernel<false>,
                           dim3(grid),
                           dim3(FAM_KERNEL_BLOCK),
                           0,
                           stream,
                           (const uint4*)snap,
                           (const uint4*)cur,
                           nPages,
                           bitmapDev,
                           (uint4*)payloadDev);
    }
    hipError_t err = hipGetLastError();
    if (err != hipSuccess) {
        return err;
    }
    uint32_t nWords = (nPages + 31) / 32;
    uint32_t sweepWaves = (nWords + 63) / 64;
    uint32_t sweepBlocks = (sweepWaves + wavesPerBlock - 1) / wavesPerBlock;
    if (sweepBlocks == 0) {
        sweepBlocks = 1;
    }
    if (sweepBlocks > FAM_MAX_BLOCKS) {
        sweepBlocks = FAM_MAX_BLOCKS;
    }
    hipLaunchKernelGGL(compactPageIdxKernel,
                       dim3(sweepBlocks),
                       dim3(FAM_KERNEL_BLOCK),
                       0,
                       stream,
                       bitmapDev,
                       nWords,
                       ticketDev,
                       pageIdxDev);
    return hipGetLastError();
}

hipError_t famApplyXorPagesEx(void* snap,
                              const uint32_t* pageIdxDev,
                              const void* payloadDev,
                              uint32_t nDirty,
                              int compact,
                              hipStream_t stream)
{
    uint32_t wavesPerBlock = FAM_KERNEL_BLOCK / 64;
    uint32_t blocks = (nDirty + wavesPerBlock - 1) / wavesPerBlock;
    // Uncapped like the diff kernel (see famDiffXorPages)
    static const uint32_t maxApplyBlocks = []() {
        const char* e = getenv("FAM_APPLY_BLOCKS");
        uint32_t v = e ? (uint32_t)atoi(e) : 0;
        return v ? v : 0xffffffffu;
    }();
    uint32_t grid = blocks < maxApplyBlocks ? (blocks ? blocks : 1)
                                            : maxApplyBlocks;
    // Unlike diff, apply is a read-modify-write of the same lines: the
    // store hits L2 brought in by the load, so streaming hurts (A/B on
    // MI355X: 5.15 vs 5.30 TB/s) — default off
    static const bool useNtApply = []() {
        const char* e = getenv("FAM_APPLY_NT");
        return e != nullptr && e[0] == '1';
    }();
    auto* kern =
      useNtApply ? applyXorPagesKernel<true> : applyXorPagesKernel<false>;
    hipLaunchKernelGGL(kern,
                       dim3(grid),
                       dim3(FAM_KERNEL_BLOCK),
                       0,
                       stream,
                       (uint4*)snap,
                       pageIdxDev,
                       (const uint4*)payloadDev,
                       nDirty,
                       compact);
    return hipGetLastError();
}

hipError_t famApplyXorPages(void* snap,
                            const uint32_t* pageIdxDev,
                            const void* payloadDev,
                            uint32_t nDirty,
                            hipStream_t stream)
{
    return famApplyXorPagesEx(snap, pageIdxDev, payloadDev, nDirty, 0,
                              stream);
}

hipError_t famGatherPages(const void* payloadDev,
                          const uint32_t* pageIdxDev,
                          void* outDev,
                          uint32_t nDirty,
                          hipStream_t stream)
{
    uint32_t wavesPerBlock = FAM_KERNEL_BLOCK / 64;
    uint32_t blocks = (nDirty + wavesPerBlock - 1) / wavesPerBlock;
    // Uncapped like the diff kernel (see famDiffXorPages)
    static const uint32_t maxApplyBlocks = []() {
        const char* e = getenv("FAM_APPLY_BLOCKS");
        uint32_t v = e ? (uint32_t)atoi(e) : 0;
        return v ? v : 0xffffffffu;
    }();
    uint32_t grid = blocks < maxApplyBlocks ? (blocks ? blocks : 1)
                                            : maxApplyBlocks;
    hipLaunchKernelGGL(gatherPagesKernel,
                       dim3(grid),
                       dim3(FAM_KERNEL_BLOCK),
                       0,
                       stream,
                       (const uint4*)payloadDev,
                       pageIdxDev,
                       (uint4*)outDev,
                       nDirty);
    return hipGetLastError();
}

// dtype: 0=i32 1=i64 2=u64 3=f32 4=f64 5=byte (matches MpiDataType)
hipError_t famElementwiseOp(void* inout,
                            const void* in,
                            uint64_t count,
                            int dtype,
                            int op,
                            hipStream_t stream)
{
    dim3 grid(gridFor(count));
    dim3 block(FAM_KERNEL_BLOCK);
    switch (dtype) {
        case 0:
            hipLaunchKernelGGL(elementwiseOpKernel<int32_t>,
                               grid, block, 0, stream,
                               (int32_t*)inout, (const int32_t*)in,
                               count, op);
            break;
        case 1:
            hipLaunchKernelGGL(elementwiseOpKernel<int64_t>,
                               grid, block, 0, stream,
                               (int64_t*)inout, (const int64_t*)in,
                               count, op);
            break;
        case 2:
            hipLaunchKernelGGL(elementwiseOpKernel<uint64_t>,
                               grid, block, 0, stream,
                               (uint64_t*)inout, (const uint64_t*)in,
                               count, op);
            break;
        case 3:
            hipLaunchKernelGGL(elementwiseOpKernel<float>,
                               grid, block, 0, stream,
                               (float*)inout, (const float*)in,
                               count, op);
            break;
        case 4:
            hipLaunchKernelGGL(elementwiseOpKernel<double>,
                               grid, block, 0, stream,
                               (double*)inout, (const double*)in,
                               count, op);
            break;
        default:
            if (op == FAM_OP_XOR) {
                hipLaunchKernelGGL(xorBytesKernel,
                                   grid, block, 0, stream,
                                   (u8*)inout, (const u8*)in, count);
            } else {
                hipLaunchKernelGGL(elementwiseOpKernel<u8>,
                                   grid, block, 0, stream,
                                   (u8*)inout, (const u8*)in, count, op);
            }
    }
    return hipGetLastError();
}

} // extern "C"
