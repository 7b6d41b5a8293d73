// FlatBuffers wire codec for the snapshot RPC schema.
//
// The reference serialises its snapshot RPCs with flatbuffers
// (reference: src/flat/faabric.fbs:1-38, snapshot/SnapshotClient.h:40-62,
// src/snapshot/SnapshotServer.cpp:28-62) and Appendix D names that
// schema part of the compatibility surface. This is a self-contained
// implementation of the FlatBuffers binary format (uoffset/vtable/table
// encoding per the official spec) for exactly those five tables — any
// conformant FlatBuffers reader can consume these buffers and vice
// versa. No generated code, no library dependency.
#pragma once

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

namespace faabricamd {

// ---------------------------------------------------------------------------
// Generic minimal builder (back-to-front like flatbuffers::FlatBufferBuilder)
// ---------------------------------------------------------------------------
class FlatWriter
{
  public:
    FlatWriter() { mem.resize(1024); head = mem.size(); }

    // End-relative offset of the next byte to be written
    uint32_t curOffset() const { return (uint32_t)(mem.size() - head); }

    void push(const void* p, size_t n)
    {
        ensure(n);
        head -= n;
        std::memcpy(mem.data() + head, p, n);
    }

    template<typename T>
    void pushScalar(T v)
    {
        push(&v, sizeof(T));
    }

    // Pad so that after pushing `size` MORE bytes the write head is
    // `align`-aligned (end-relative; finish() aligns the total buffer
    // size to the largest alignment seen, which makes end-relative
    // congruence equal final-address congruence)
    void prep(size_t align, size_t size)
    {
        if (align > maxAlign) {
            maxAlign = align;
        }
        size_t want = (size_t)curOffset() + size;
        size_t pad = (align - (want % align)) % align;
        ensure(pad);
        for (size_t i = 0; i < pad; i++) {
            head -= 1;
            mem[head] = 0;
        }
    }

    // [u32 len][bytes]['\0'], len 4-aligned and contiguous with bytes
    // (padding, if any, sits after the terminator)
    uint32_t createString(const std::string& s)
    {
        prep(4, 4 + s.size() + 1);
        uint8_t nul = 0;
        push(&nul, 1);
        push(s.data(), s.size());
        pushScalar<uint32_t>((uint32_t)s.size());
        return curOffset();
    }

    // [u32 count][bytes], count 4-aligned and contiguous
    uint32_t createByteVector(const uint8_t* data, size_t n)
    {
        prep(4, 4 + n);
        push(data, n);
        pushScalar<uint32_t>((uint32_t)n);
        return curOffset();
    }

    // [u32 count][uoffsets...] — offs are end-relative table offsets
    uint32_t createOffsetVector(const std::vector<uint32_t>& offs)
    {
        prep(4, offs.size() * 4 + 4);
        for (size_t i = offs.size(); i > 0; i--) {
            head -= 4;
            uint32_t fieldOff = (uint32_t)(mem.size() - head);
            uint32_t v = fieldOff - offs[i - 1];
            std::memcpy(mem.data() + head, &v, 4);
        }
        pushScalar<uint32_t>((uint32_t)offs.size());
        return curOffset();
    }

    // --- table construction ---
    // Declare fields (id = schema order), then endTable() emits the
    // table + its vtable and returns the table's end-relative offset.
    void startTable() { fields.clear(); }
    void addScalarField(int id, uint64_t value, int size)
    {
        fields.push_back({ id, value, size, false });
    }
    void addOffsetField(int id, uint32_t target)
    {
        if (target == 0) {
            return; // absent
        }
        fields.push_back({ id, target, 4, true });
    }

    uint32_t endTable()
    {
        // Lay the table out forward: [i32 soffset][fields in id order,
        // each aligned to its size]
        int maxId = -1;
        for (auto& f : fields) {
            maxId = f.id > maxId ? f.id : maxId;
        }
        std::vector<uint16_t> fieldPos((size_t)maxId + 1, 0);
        size_t tblSize = 4; // soffset
        size_t maxAlign = 4;
        for (auto& f : fields) {
            size_t a = (size_t)f.size;
            tblSize = (tblSize + a - 1) / a * a;
            fieldPos[f.id] = (uint16_t)tblSize;
            tblSize += a;
            maxAlign = a > maxAlign ? a : maxAlign;
        }

        // Write the table body (soffset patched after the vtable lands;
        // NOTE: absolute indices go stale across ensure() reallocations,
        // so positions are tracked end-relative throughout)
        prep(maxAlign, tblSize);
        ensure(tblSize);
        head -= tblSize;
        std::memset(mem.data() + head, 0, tblSize);
        uint32_t tableOff = curOffset(); // end-relative table start
        for (auto& f : fields) {
            size_t at = (mem.size() - tableOff) + fieldPos[f.id];
            if (f.isOffset) {
                // uoffset: forward distance from the field to the target
                uint32_t fieldEndOff = (uint32_t)(mem.size() - at);
                uint32_t v = fieldEndOff - (uint32_t)f.value;
                std::memcpy(mem.data() + at, &v, 4);
            } else {
                std::memcpy(mem.data() + at, &f.value, f.size);
            }
        }

        // vtable: [u16 vtSize][u16 tblSize][u16 per field]
        size_t vtSize = 4 + 2 * ((size_t)maxId + 1);
        prep(2, vtSize);
        ensure(vtSize);
        head -= vtSize;
        uint8_t* vt = mem.data() + head;
        uint16_t v16 = (uint16_t)vtSize;
        std::memcpy(vt, &v16, 2);
        v16 = (uint16_t)tblSize;
        std::memcpy(vt + 2, &v16, 2);
        for (int i = 0; i <= maxId; i++) {
            v16 = fieldPos[i];
            std::memcpy(vt + 4 + 2 * i, &v16, 2);
        }

        // Patch the table's soffset. With end-relative offsets E and
        // final addresses addr = total - E:
        //   soffset = addr_table - addr_vtable = vtEndOff - tblEndOff
        uint32_t vtOff = curOffset();
        int32_t so = (int32_t)(vtOff - tableOff);
        std::memcpy(mem.data() + (mem.size() - tableOff), &so, 4);
        return tableOff;
    }

    // Root uoffset; returns the finished buffer. The total size is
    // padded to the largest alignment used so that end-relative
    // alignment of every object equals its final address alignment.
    std::string finish(uint32_t rootTable)
    {
        prep(maxAlign, 4);
        uint32_t fieldOffAfter = curOffset() + 4;
        uint32_t v = fieldOffAfter - rootTable;
        pushScalar<uint32_t>(v);
        return std::string((const char*)mem.data() + head,
                           mem.size() - head);
    }

  private:
    struct Field
    {
        int id;
        uint64_t value; // scalar bits or end-relative target offset
        int size;
        bool isOffset;
    };
    std::vector<uint8_t> mem;
    size_t head = 0;
    size_t maxAlign = 4;
    std::vector<Field> fields;

    void ensure(size_t n)
    {
        if (head >= n) {
            return;
        }
        size_t used = mem.size() - head;
        size_t grown = mem.size() * 2 + n;
        std::vector<uint8_t> next(grown);
        std::memcpy(next.data() + grown - used, mem.data() + head, used);
        mem.swap(next);
        head = grown - used;
    }
};

// ---------------------------------------------------------------------------
// Generic reader (follows vtables; tolerant of absent fields)
// ---------------------------------------------------------------------------
class FlatReader
{
  public:
    explicit FlatReader(const std::string& buf)
      : data((const uint8_t*)buf.data())
      , size(buf.size())
    {}

    uint32_t root() const
    {
        uint32_t v = 0;
        if (size >= 4) {
            std::memcpy(&v, data, 4);
        }
        return v; // absolute position of the root table
    }

    // Field position inside the table at `tablePos`, 0 if absent
    uint32_t fieldPos(uint32_t tablePos, int id) const
    {
        if (tablePos + 4 > size) {
            return 0;
        }
        int32_t so = 0;
        std::memcpy(&so, data + tablePos, 4);
        int64_t vt = (int64_t)tablePos - so;
        if (vt < 0 || (uint64_t)vt + 4 > size) {
            return 0;
        }
        uint16_t vtSize = 0;
        std::memcpy(&vtSize, data + vt, 2);
        size_t slot = 4 + 2 * (size_t)id;
        if (slot + 2 > vtSize) {
            return 0;
        }
        uint16_t off = 0;
        std::memcpy(&off, data + vt + slot, 2);
        return off == 0 ? 0 : tablePos + off;
    }

    template<typename T>
    T scalar(uint32_t tablePos, int id, T dflt = 0) const
    {
        uint32_t p = fieldPos(tablePos, id);
        if (p == 0 || p + sizeof(T) > size) {
            return dflt;
        }
        T v;
        std::memcpy(&v, data + p, sizeof(T));
        return v;
    }

    // Absolute position of the object a uoffset field points at
    uint32_t indirect(uint32_t tablePos, int id) const
    {
        uint32_t p = fieldPos(tablePos, id);
        if (p == 0 || p + 4 > size) {
            return 0;
        }
        uint32_t u = 0;
        std::memcpy(&u, data + p, 4);
        return p + u;
    }

    std::string str(uint32_t tablePos, int id) const
    {
        uint32_t p = indirect(tablePos, id);
        if (p == 0 || p + 4 > size) {
            return {};
        }
        uint32_t len = 0;
        std::memcpy(&len, data + p, 4);
        if (p + 4 + len > size) {
            return {};
        }
        return std::string((const char*)data + p + 4, len);
    }

    std::vector<uint8_t> bytes(uint32_t tablePos, int id) const
    {
        uint32_t p = indirect(tablePos, id);
        if (p == 0 || p + 4 > size) {
            return {};
        }
        uint32_t len = 0;
        std::memcpy(&len, data + p, 4);
        if (p + 4 + len > size) {
            return {};
        }
        return { data + p + 4, data + p + 4 + len };
    }

    // Vector of tables: absolute positions of each element
    std::vector<uint32_t> tableVector(uint32_t tablePos, int id) const
    {
        std::vector<uint32_t> out;
        uint32_t p = indirect(tablePos, id);
        if (p == 0 || p + 4 > size) {
            return out;
        }
        uint32_t n = 0;
        std::memcpy(&n, data + p, 4);
        for (uint32_t i = 0; i < n; i++) {
            uint32_t ep = p + 4 + i * 4;
            if (ep + 4 > size) {
                break;
            }
            uint32_t u = 0;
            std::memcpy(&u, data + ep, 4);
            out.push_back(ep + u);
        }
        return out;
    }

  private:
    const uint8_t* data;
    size_t size;
};

// ---------------------------------------------------------------------------
// Schema-specific encode/decode (field ids = faabric.fbs order)
// ---------------------------------------------------------------------------

struct FlatMergeRegion
{
    int32_t offset = 0;    // id 0
    uint64_t length = 0;   // id 1
    int32_t dataType = 0;  // id 2
    int32_t mergeOp = 0;   // id 3
};

struct FlatSnapshotPush
{
    std::string key;              // id 0
    uint64_t maxSize = 0;         // id 1
    std::vector<uint8_t> contents; // id 2
    std::vector<FlatMergeRegion> mergeRegions; // id 3
    std::string encode() const;
    static FlatSnapshotPush decode(const std::string& buf);
};

struct FlatSnapshotDelete
{
    std::string key; // id 0
    std::string encode() const;
    static FlatSnapshotDelete decode(const std::string& buf);
};

struct FlatSnapshotDiff
{
    int32_t offset = 0;   // id 0
    int32_t dataType = 0; // id 1
    int32_t mergeOp = 0;  // id 2
    std::vector<uint8_t> data; // id 3
};

struct FlatSnapshotUpdate
{
    std::string key; // id 0
    std::vector<FlatMergeRegion> mergeRegions; // id 1
    std::vector<FlatSnapshotDiff> diffs;       // id 2
    std::string encode() const;
    static FlatSnapshotUpdate decode(const std::string& buf);
};

struct FlatThreadResult
{
    int32_t appId = 0;       // id 0
    int32_t messageId = 0;   // id 1
    int32_t returnValue = 0; // id 2
    std::string key;         // id 3
    std::vector<FlatSnapshotDiff> diffs; // id 4
    // Extension beyond faabric.fbs (schema-compatible: readers of the
    // original schema ignore ids past their vtable)
    std::string executedHost; // id 5
    std::string encode() const;
    static FlatThreadResult decode(const std::string& buf);
};

} // namespace faabricamd
