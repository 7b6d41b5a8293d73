// Minimal JSON value + parser/serializer for the planner's HTTP ops API
// (the reference maps protobuf⇄JSON via libprotobuf's json_util,
// reference: src/util/json.cpp, util/json.h:8-10 — no protobuf here, so
// a small hand-rolled JSON layer with the same json_name field spellings).
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <string>
#include <vector>

namespace faabricamd {

class Json
{
  public:
    enum class Type
    {
        Null,
        Bool,
        Int,
        Double,
        String,
        Array,
        Object,
    };

    Json() = default;
    Json(std::nullptr_t) {}
    Json(bool v)
      : type_(Type::Bool)
      , boolV(v)
    {}
    Json(int v)
      : type_(Type::Int)
      , intV(v)
    {}
    Json(int64_t v)
      : type_(Type::Int)
      , intV(v)
    {}
    Json(double v)
      : type_(Type::Double)
      , dblV(v)
    {}
    Json(const char* v)
      : type_(Type::String)
      , strV(v)
    {}
    Json(std::string v)
      : type_(Type::String)
      , strV(std::move(v))
    {}

    static Json array();
    static Json object();

    Type type() const { return type_; }
    bool isNull() const { return type_ == Type::Null; }

    bool asBool(bool deflt = false) const;
    int64_t asInt(int64_t deflt = 0) const;
    double asDouble(double deflt = 0) const;
    const std::string& asString() const;

    // Object access
    Json& operator[](const std::string& key);
    const Json* find(const std::string& key) const;
    bool getBool(const std::string& key, bool deflt = false) const;
    int64_t getInt(const std::string& key, int64_t deflt = 0) const;
    std::string getString(const std::string& key,
                          const std::string& deflt = "") const;

    // Array access
    void push(Json v);
    const std::vector<Json>& items() const { return arrV; }
    size_t size() const;

    std::string dump() const;
    static Json parse(const std::string& s);

  private:
    void dumpTo(std::string& out) const;

    Type type_ = Type::Null;
    bool boolV = false;
    int64_t intV = 0;
    double dblV = 0;
    std::string strV;
    std::vector<Json> arrV;
    std::vector<std::pair<std::string, Json>> objV; // ordered

    friend class JsonParser;
};

// protobuf-JSON conversions for the data model (json_name spellings from
// src/proto/faabric.proto)
struct Message;
struct BatchExecuteRequest;
struct BatchExecuteRequestStatus;

Json messageToJson(const Message& msg);
Message messageFromJson(const Json& j);
Json berToJson(const BatchExecuteRequest& ber);
BatchExecuteRequest berFromJson(const Json& j);
Json berStatusToJson(const BatchExecuteRequestStatus& status);

} // namespace faabricamd
