// Minimal protobuf-wire-format codec (proto3 semantics) with no protobuf
// library dependency.
//
// The reference serialises its data model with libprotobuf
// (reference: src/proto/faabric.proto, src/planner/planner.proto). This
// container ships no C++ protobuf, so the MI355X build hand-implements the
// wire format — varint / length-delimited encoding with the SAME field
// numbers as the reference schemas, keeping the bytes interoperable with a
// protobuf decoder for the message shapes we use (proto3 default-skipping
// included).
#pragma once

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace faabricamd {

enum class WireType : uint32_t
{
    Varint = 0,
    Fixed64 = 1,
    LengthDelimited = 2,
    Fixed32 = 5,
};

class PbWriter
{
  public:
    std::string& buffer() { return out; }
    std::string take() { return std::move(out); }

    void varint(uint64_t v)
    {
        while (v >= 0x80) {
            out.push_back((char)((v & 0x7f) | 0x80));
            v >>= 7;
        }
        out.push_back((char)v);
    }

    void tag(uint32_t field, WireType t)
    {
        varint(((uint64_t)field << 3) | (uint64_t)t);
    }

    // proto3: scalar fields with default value are omitted
    void putInt32(uint32_t field, int32_t v)
    {
        if (v == 0) {
            return;
        }
        tag(field, WireType::Varint);
        varint((uint64_t)(int64_t)v); // sign-extended like protobuf int32
    }

    void putInt64(uint32_t field, int64_t v)
    {
        if (v == 0) {
            return;
        }
        tag(field, WireType::Varint);
        varint((uint64_t)v);
    }

    void putUInt64(uint32_t field, uint64_t v)
    {
        if (v == 0) {
            return;
        }
        tag(field, WireType::Varint);
        varint(v);
    }

    void putBool(uint32_t field, bool v)
    {
        if (!v) {
            return;
        }
        tag(field, WireType::Varint);
        varint(1);
    }

    void putString(uint32_t field, const std::string& v)
    {
        if (v.empty()) {
            return;
        }
        tag(field, WireType::LengthDelimited);
        varint(v.size());
        out.append(v);
    }

    void putBytes(uint32_t field, const void* data, size_t len)
    {
        if (len == 0) {
            return;
        }
        tag(field, WireType::LengthDelimited);
        varint(len);
        out.append((const char*)data, len);
    }

    void putBytes(uint32_t field, const std::vector<uint8_t>& v)
    {
        putBytes(field, v.data(), v.size());
    }

    // Nested message (always emitted when asked, caller decides emptiness)
    void putMessage(uint32_t field, const std::string& encoded)
    {
        tag(field, WireType::LengthDelimited);
        varint(encoded.size());
        out.append(encoded);
    }

    // Repeated packed varints (proto3 default for repeated int32)
    void putPackedInt32(uint32_t field, const std::vector<int32_t>& vs)
    {
        if (vs.empty()) {
            return;
        }
        PbWriter inner;
        for (int32_t v : vs) {
            inner.varint((uint64_t)(int64_t)v);
        }
        putMessage(field, inner.buffer());
    }

    void putDouble(uint32_t field, double v)
    {
        if (v == 0.0) {
            return;
        }
        tag(field, WireType::Fixed64);
        uint64_t bits;
        std::memcpy(&bits, &v, 8);
        for (int i = 0; i < 8; i++) {
            out.push_back((char)((bits >> (8 * i)) & 0xff));
        }
    }

  private:
    std::string out;
};

class PbReader
{
  public:
    PbReader(const char* data, size_t len)
      : p(data)
      , end(data + len)
    {}
    explicit PbReader(const std::string& s)
      : PbReader(s.data(), s.size())
    {}

    bool next(uint32_t& field, WireType& type)
    {
        if (p >= end) {
            return false;
        }
        uint64_t key = varint();
        field = (uint32_t)(key >> 3);
        type = (WireType)(key & 0x7);
        return true;
    }

    uint64_t varint()
    {
        uint64_t v = 0;
        int shift = 0;
        while (p < end) {
            uint8_t b = (uint8_t)*p++;
            v |= (uint64_t)(b & 0x7f) << shift;
            if ((b & 0x80) == 0) {
                return v;
            }
            shift += 7;
            if (shift >= 64) {
                break;
            }
        }
        throw std::runtime_error("pb: bad varint");
    }

    int32_t asInt32() { return (int32_t)(int64_t)varint(); }
    int64_t asInt64() { return (int64_t)varint(); }
    uint64_t asUInt64() { return varint(); }
    bool asBool() { return varint() != 0; }

    std::string asString()
    {
        uint64_t len = varint();
        checkLen(len);
        std::string s(p, p + len);
        p += len;
        return s;
    }

    std::vector<uint8_t> asBytes()
    {
        uint64_t len = varint();
        checkLen(len);
        std::vector<uint8_t> v((const uint8_t*)p, (const uint8_t*)p + len);
        p += len;
        return v;
    }

    // View of a nested message / length-delimited payload
    PbReader asSub()
    {
        uint64_t len = varint();
        checkLen(len);
        PbReader sub(p, len);
        p += len;
        return sub;
    }

    std::vector<int32_t> asPackedInt32()
    {
        PbReader sub = asSub();
        std::vector<int32_t> out;
        while (sub.p < sub.end) {
            out.push_back((int32_t)(int64_t)sub.varint());
        }
        return out;
    }

    double asDouble()
    {
        checkLen(8);
        uint64_t bits = 0;
        for (int i = 0; i < 8; i++) {
            bits |= (uint64_t)(uint8_t)p[i] << (8 * i);
        }
        p += 8;
        double v;
        std::memcpy(&v, &bits, 8);
        return v;
    }

    void skip(WireType t)
    {
        switch (t) {
            case WireType::Varint:
                varint();
                break;
            case WireType::Fixed64:
                checkLen(8);
                p += 8;
                break;
            case WireType::LengthDelimited: {
                uint64_t len = varint();
                checkLen(len);
                p += len;
                break;
            }
            case WireType::Fixed32:
                checkLen(4);
                p += 4;
                break;
            default:
                throw std::runtime_error("pb: bad wire type");
        }
    }

    bool atEnd() const { return p >= end; }

  private:
    void checkLen(uint64_t len)
    {
        if ((uint64_t)(end - p) < len) {
            throw std::runtime_error("pb: truncated message");
        }
    }

    const char* p;
    const char* end;
};

} // namespace faabricamd
