#include "faabricamd/json.h"
#include "faabricamd/messages.h"
#include "faabricamd/util.h"

#include <cctype>
#include <cmath>
#include <cstdio>
#include <cstring>

namespace faabricamd {

Json Json::array()
{
    Json j;
    j.type_ = Type::Array;
    return j;
}

Json Json::object()
{
    Json j;
    j.type_ = Type::Object;
    return j;
}

bool Json::asBool(bool deflt) const
{
    if (type_ == Type::Bool) {
        return boolV;
    }
    if (type_ == Type::Int) {
        return intV != 0;
    }
    return deflt;
}

int64_t Json::asInt(int64_t deflt) const
{
    if (type_ == Type::Int) {
        return intV;
    }
    if (type_ == Type::Double) {
        return (int64_t)dblV;
    }
    if (type_ == Type::String) {
        return atoll(strV.c_str());
    }
    return deflt;
}

double Json::asDouble(double deflt) const
{
    if (type_ == Type::Double) {
        return dblV;
    }
    if (type_ == Type::Int) {
        return (double)intV;
    }
    return deflt;
}

const std::string& Json::asString() const
{
    static const std::string empty;
    return type_ == Type::String ? strV : empty;
}

Json& Json::operator[](const std::string& key)
{
    if (type_ == Type::Null) {
        type_ = Type::Object;
    }
    for (auto& [k, v] : objV) {
        if (k == key) {
            return v;
        }
    }
    objV.emplace_back(key, Json());
    return objV.back().second;
}

const Json* Json::find(const std::string& key) const
{
    for (const auto& [k, v] : objV) {
        if (k == key) {
            return &v;
        }
    }
    return nullptr;
}

bool Json::getBool(const std::string& key, bool deflt) const
{
    const Json* v = find(key);
    return v == nullptr ? deflt : v->asBool(deflt);
}

int64_t Json::getInt(const std::string& key, int64_t deflt) const
{
    const Json* v = find(key);
    return v == nullptr ? deflt : v->asInt(deflt);
}

std::string Json::getString(const std::string& key,
                            const std::string& deflt) const
{
    const Json* v = find(key);
    return v == nullptr || v->type() != Type::String ? deflt
                                                     : v->asString();
}

void Json::push(Json v)
{
    if (type_ == Type::Null) {
        type_ = Type::Array;
    }
    arrV.push_back(std::move(v));
}

size_t Json::size() const
{
    if (type_ == Type::Array) {
        return arrV.size();
    }
    if (type_ == Type::Object) {
        return objV.size();
    }
    return 0;
}

static void dumpString(std::string& out, const std::string& s)
{
    out.push_back('"');
    for (char c : s) {
        switch (c) {
            case '"':
                out += "\\\"";
                break;
            case '\\':
                out += "\\\\";
                break;
            case '\n':
                out += "\\n";
                break;
            case '\r':
                out += "\\r";
                break;
            case '\t':
                out += "\\t";
                break;
            default:
                if ((unsigned char)c < 0x20) {
                    char buf[8];
                    snprintf(buf, sizeof(buf), "\\u%04x", c);
                    out += buf;
                } else {
                    out.push_back(c);
                }
        }
    }
    out.push_back('"');
}

void Json::dumpTo(std::string& out) const
{
    switch (type_) {
        case Type::Null:
            out += "null";
            break;
        case Type::Bool:
            out += boolV ? "true" : "false";
            break;
        case Type::Int:
            out += std::to_string(intV);
            break;
        case Type::Double: {
            char buf[32];
            snprintf(buf, sizeof(buf), "%.17g", dblV);
            out += buf;
            break;
        }
        case Type::String:
            dumpString(out, strV);
            break;
        case Type::Array: {
            out.push_back('[');
            for (size_t i = 0; i < arrV.size(); i++) {
                if (i) {
                    out.push_back(',');
                }
                arrV[i].dumpTo(out);
            }
            out.push_back(']');
            break;
        }
        case Type::Object: {
            out.push_back('{');
            for (size_t i = 0; i < objV.size(); i++) {
                if (i) {
                    out.push_back(',');
                }
                dumpString(out, objV[i].first);
                out.push_back(':');
                objV[i].second.dumpTo(out);
            }
            out.push_back('}');
            break;
        }
    }
}

std::string Json::dump() const
{
    std::string out;
    dumpTo(out);
    return out;
}

// ------------------------- parser -------------------------------------------

class JsonParser
{
  public:
    JsonParser(const char* p, const char* end)
      : p(p)
      , end(end)
    {}

    Json parse()
    {
        skipWs();
        Json v = parseValue();
        return v;
    }

  private:
    void skipWs()
    {
        while (p < end && (unsigned char)*p <= ' ') {
            p++;
        }
    }

    [[noreturn]] void fail(const char* why)
    {
        throw FaabricException(std::string("json parse error: ") + why);
    }

    Json parseValue()
    {
        skipWs();
        if (p >= end) {
            fail("unexpected end");
        }
        switch (*p) {
            case '{':
                return parseObject();
            case '[':
                return parseArray();
            case '"':
                return Json(parseString());
            case 't':
                expect("true");
                return Json(true);
            case 'f':
                expect("false");
                return Json(false);
            case 'n':
                expect("null");
                return Json(nullptr);
            default:
                return parseNumber();
        }
    }

    void expect(const char* lit)
    {
        size_t n = strlen(lit);
        if ((size_t)(end - p) < n || strncmp(p, lit, n) != 0) {
            fail("bad literal");
        }
        p += n;
    }

    std::string parseString()
    {
        if (*p != '"') {
            fail("expected string");
        }
        p++;
        std::string out;
        while (p < end && *p != '"') {
            if (*p == '\\') {
                p++;
                if (p >= end) {
                    fail("bad escape");
                }
                switch (*p) {
                    case 'n':
                        out.push_back('\n');
                        break;
                    case 't':
                        out.push_back('\t');
                        break;
                    case 'r':
                        out.push_back('\r');
                        break;
                    case 'b':
                        out.push_back('\b');
                        break;
                    case 'f':
                        out.push_back('\f');
                        break;
                    case 'u': {
                        if (end - p < 5) {
                            fail("bad unicode escape");
                        }
                        unsigned code = 0;
                        sscanf(p + 1, "%4x", &code);
                        p += 4;
                        // UTF-8 encode (BMP only)
                        if (code < 0x80) {
                            out.push_back((char)code);
                        } else if (code < 0x800) {
                            out.push_back((char)(0xc0 | (code >> 6)));
                            out.push_back((char)(0x80 | (code & 0x3f)));
                        } else {
                            out.push_back((char)(0xe0 | (code >> 12)));
                            out.push_back(
                              (char)(0x80 | ((code >> 6) & 0x3f)));
                            out.push_back((char)(0x80 | (code & 0x3f)));
                        }
                        break;
                    }
                    default:
                        out.push_back(*p);
                }
                p++;
            } else {
                out.push_back(*p++);
            }
        }
        if (p >= end) {
            fail("unterminated string");
        }
        p++; // closing quote
        return out;
    }

    Json parseNumber()
    {
        const char* start = p;
        bool isDouble = false;
        if (p < end && (*p == '-' || *p == '+')) {
            p++;
        }
        while (p < end &&
               (isdigit((unsigned char)*p) || *p == '.' || *p == 'e' ||
                *p == 'E' || *p == '-' || *p == '+')) {
            if (*p == '.' || *p == 'e' || *p == 'E') {
                isDouble = true;
            }
            p++;
        }
        std::string num(start, p);
        if (num.empty()) {
            fail("bad number");
        }
        if (isDouble) {
            return Json(atof(num.c_str()));
        }
        return Json((int64_t)atoll(num.c_str()));
    }

    Json parseArray()
    {
        Json out = Json::array();
        p++; // [
        skipWs();
        if (p < end && *p == ']') {
            p++;
            return out;
        }
        while (true) {
            out.push(parseValue());
            skipWs();
            if (p < end && *p == ',') {
                p++;
                continue;
            }
            if (p < end && *p == ']') {
                p++;
                return out;
            }
            fail("bad array");
        }
    }

    Json parseObject()
    {
        Json out = Json::object();
        p++; // {
        skipWs();
        if (p < end && *p == '}') {
            p++;
            return out;
        }
        while (true) {
            skipWs();
            std::string key = parseString();
            skipWs();
            if (p >= end || *p != ':') {
                fail("expected colon");
            }
            p++;
            out[key] = parseValue();
            skipWs();
            if (p < end && *p == ',') {
                p++;
                continue;
            }
            if (p < end && *p == '}') {
                p++;
                return out;
            }
            fail("bad object");
        }
    }

    const char* p;
    const char* end;
};

Json Json::parse(const std::string& s)
{
    JsonParser parser(s.data(), s.data() + s.size());
    return parser.parse();
}

// ------------------------- protobuf-JSON conversions -------------------------

// base64 for bytes fields (protobuf JSON mapping)
static const char b64chars[] =
  "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";

static std::string b64encode(const std::vector<uint8_t>& data)
{
    std::string out;
    out.reserve((data.size() + 2) / 3 * 4);
    size_t i = 0;
    while (i + 3 <= data.size()) {
        uint32_t v = (data[i] << 16) | (data[i + 1] << 8) | data[i + 2];
        out.push_back(b64chars[(v >> 18) & 63]);
        out.push_back(b64chars[(v >> 12) & 63]);
        out.push_back(b64chars[(v >> 6) & 63]);
        out.push_back(b64chars[v & 63]);
        i += 3;
    }
    size_t rem = data.size() - i;
    if (rem == 1) {
        uint32_t v = data[i] << 16;
        out.push_back(b64chars[(v >> 18) & 63]);
        out.push_back(b64chars[(v >> 12) & 63]);
        out += "==";
    } else if (rem == 2) {
        uint32_t v = (data[i] << 16) | (data[i + 1] << 8);
        out.push_back(b64chars[(v >> 18) & 63]);
        out.push_back(b64chars[(v >> 12) & 63]);
        out.push_back(b64chars[(v >> 6) & 63]);
        out.push_back('=');
    }
    return out;
}

static std::vector<uint8_t> b64decode(const std::string& s)
{
    auto val = [](char c) -> int {
        if (c >= 'A' && c <= 'Z') {
            return c - 'A';
        }
        if (c >= 'a' && c <= 'z') {
            return c - 'a' + 26;
        }
        if (c >= '0' && c <= '9') {
            return c - '0' + 52;
        }
        if (c == '+') {
            return 62;
        }
        if (c == '/') {
            return 63;
        }
        return -1;
    };
    std::vector<uint8_t> out;
    uint32_t buf = 0;
    int bits = 0;
    for (char c : s) {
        int v = val(c);
        if (v < 0) {
            continue;
        }
        buf = (buf << 6) | (uint32_t)v;
        bits += 6;
        if (bits >= 8) {
            bits -= 8;
            out.push_back((uint8_t)((buf >> bits) & 0xff));
        }
    }
    return out;
}

Json messageToJson(const Message& msg)
{
    // json_name spellings from src/proto/faabric.proto
    Json j = Json::object();
    j["id"] = Json((int64_t)msg.id);
    j["appId"] = Json((int64_t)msg.appId);
    if (msg.appIdx != 0) {
        j["appIdx"] = Json((int64_t)msg.appIdx);
    }
    if (!msg.mainHost.empty()) {
        j["mainHost"] = Json(msg.mainHost);
    }
    j["user"] = Json(msg.user);
    j["function"] = Json(msg.function);
    if (!msg.inputData.empty()) {
        j["input_data"] = Json(b64encode(msg.inputData));
    }
    if (!msg.outputData.empty()) {
        j["output_data"] = Json(msg.outputData);
    }
    if (msg.returnValue != 0) {
        j["returnValue"] = Json((int64_t)msg.returnValue);
    }
    if (!msg.snapshotKey.empty()) {
        j["snapshotKey"] = Json(msg.snapshotKey);
    }
    if (msg.startTimestamp != 0) {
        j["start_ts"] = Json(msg.startTimestamp);
    }
    if (msg.finishTimestamp != 0) {
        j["finish_ts"] = Json(msg.finishTimestamp);
    }
    if (!msg.executedHost.empty()) {
        j["executedHost"] = Json(msg.executedHost);
    }
    if (msg.groupId != 0) {
        j["groupId"] = Json((int64_t)msg.groupId);
    }
    if (msg.groupIdx != 0) {
        j["groupIdx"] = Json((int64_t)msg.groupIdx);
    }
    if (msg.groupSize != 0) {
        j["groupSize"] = Json((int64_t)msg.groupSize);
    }
    if (msg.isMpi) {
        j["mpi"] = Json(true);
        j["mpiWorldId"] = Json((int64_t)msg.mpiWorldId);
        j["mpiRank"] = Json((int64_t)msg.mpiRank);
        j["mpi_world_size"] = Json((int64_t)msg.mpiWorldSize);
    }
    if (msg.recordExecGraph) {
        j["record_exec_graph"] = Json(true);
    }
    if (!msg.chainedMsgIds.empty()) {
        Json arr = Json::array();
        for (int32_t id : msg.chainedMsgIds) {
            arr.push(Json((int64_t)id));
        }
        j["chainedMsgIds"] = arr;
    }
    return j;
}

Message messageFromJson(const Json& j)
{
    Message m;
    m.id = (int32_t)j.getInt("id");
    m.appId = (int32_t)j.getInt("appId");
    m.appIdx = (int32_t)j.getInt("appIdx");
    m.mainHost = j.getString("mainHost");
    m.user = j.getString("user");
    m.function = j.getString("function");
    m.inputData = b64decode(j.getString("input_data"));
    m.outputData = j.getString("output_data");
    m.returnValue = (int32_t)j.getInt("returnValue");
    m.snapshotKey = j.getString("snapshotKey");
    m.startTimestamp = j.getInt("start_ts");
    m.finishTimestamp = j.getInt("finish_ts");
    m.executedHost = j.getString("executedHost");
    m.groupId = (int32_t)j.getInt("groupId");
    m.groupIdx = (int32_t)j.getInt("groupIdx");
    m.groupSize = (int32_t)j.getInt("groupSize");
    m.isMpi = j.getBool("mpi");
    m.mpiWorldId = (int32_t)j.getInt("mpiWorldId");
    m.mpiRank = (int32_t)j.getInt("mpiRank");
    m.mpiWorldSize = (int32_t)j.getInt("mpi_world_size");
    m.recordExecGraph = j.getBool("record_exec_graph");
    return m;
}

Json berToJson(const BatchExecuteRequest& ber)
{
    Json j = Json::object();
    j["appId"] = Json((int64_t)ber.appId);
    if (ber.groupId != 0) {
        j["groupId"] = Json((int64_t)ber.groupId);
    }
    j["user"] = Json(ber.user);
    j["function"] = Json(ber.function);
    if (ber.type != BatchExecuteType::FUNCTIONS) {
        j["type"] = Json((int64_t)ber.type);
    }
    if (!ber.snapshotKey.empty()) {
        j["snapshotKey"] = Json(ber.snapshotKey);
    }
    Json msgs = Json::array();
    for (const auto& m : ber.messages) {
        msgs.push(messageToJson(m));
    }
    j["messages"] = msgs;
    if (ber.singleHostHint) {
        j["singleHostHint"] = Json(true);
    }
    return j;
}

BatchExecuteRequest berFromJson(const Json& j)
{
    BatchExecuteRequest ber;
    ber.appId = (int32_t)j.getInt("appId");
    ber.groupId = (int32_t)j.getInt("groupId");
    ber.user = j.getString("user");
    ber.function = j.getString("function");
    ber.type = (BatchExecuteType)j.getInt("type");
    ber.snapshotKey = j.getString("snapshotKey");
    ber.singleHostHint = j.getBool("singleHostHint");
    const Json* msgs = j.find("messages");
    if (msgs != nullptr) {
        for (const auto& mj : msgs->items()) {
            ber.messages.push_back(messageFromJson(mj));
        }
    }
    return ber;
}

Json berStatusToJson(const BatchExecuteRequestStatus& status)
{
    Json j = Json::object();
    j["appId"] = Json((int64_t)status.appId);
    j["finished"] = Json(status.finished);
    Json arr = Json::array();
    for (const auto& m : status.messageResults) {
        arr.push(messageToJson(m));
    }
    j["messageResults"] = arr;
    if (status.expectedNumMessages != 0) {
        j["expectedNumMessages"] = Json((int64_t)status.expectedNumMessages);
    }
    return j;
}

} // namespace faabricamd
