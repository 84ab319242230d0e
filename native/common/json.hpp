/*
 * binder-amd: minimal JSON DOM (parse + serialize).
 *
 * ZooKeeper registration node payloads, the server config file, and the
 * structured log lines are all JSON; the reference relies on V8's JSON
 * (see /root/reference/lib/zk.js:139-154, main.js:96-104). This is a small
 * self-contained DOM with int64 preservation, written for this project.
 */
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <optional>
#include <string>
#include <string_view>
#include <vector>

namespace bamd {

class Json;
using JsonArray = std::vector<Json>;
// insertion order is not preserved; binder semantics never depend on it.
using JsonObject = std::map<std::string, Json>;

class Json {
  public:
    enum class Type { Null, Bool, Int, Double, String, Array, Object };

    Json() : type_(Type::Null) {}
    Json(std::nullptr_t) : type_(Type::Null) {}
    Json(bool b) : type_(Type::Bool), bool_(b) {}
    Json(int v) : type_(Type::Int), int_(v) {}
    Json(int64_t v) : type_(Type::Int), int_(v) {}
    Json(uint64_t v) : type_(Type::Int), int_((int64_t)v) {}
    Json(double v) : type_(Type::Double), dbl_(v) {}
    Json(const char* s) : type_(Type::String), str_(s) {}
    Json(std::string s) : type_(Type::String), str_(std::move(s)) {}
    Json(std::string_view s) : type_(Type::String), str_(s) {}
    Json(JsonArray a) : type_(Type::Array),
        arr_(std::make_shared<JsonArray>(std::move(a))) {}
    Json(JsonObject o) : type_(Type::Object),
        obj_(std::make_shared<JsonObject>(std::move(o))) {}

    static Json array() { return Json(JsonArray{}); }
    static Json object() { return Json(JsonObject{}); }

    Type type() const { return type_; }
    bool isNull() const { return type_ == Type::Null; }
    bool isBool() const { return type_ == Type::Bool; }
    bool isNumber() const {
        return type_ == Type::Int || type_ == Type::Double;
    }
    bool isInt() const { return type_ == Type::Int; }
    bool isString() const { return type_ == Type::String; }
    bool isArray() const { return type_ == Type::Array; }
    bool isObject() const { return type_ == Type::Object; }

    bool asBool(bool dflt = false) const {
        return type_ == Type::Bool ? bool_ : dflt;
    }
    int64_t asInt(int64_t dflt = 0) const {
        if (type_ == Type::Int) return int_;
        if (type_ == Type::Double) return (int64_t)dbl_;
        return dflt;
    }
    double asDouble(double dflt = 0.0) const {
        if (type_ == Type::Double) return dbl_;
        if (type_ == Type::Int) return (double)int_;
        return dflt;
    }
    const std::string& asString() const {
        static const std::string kEmpty;
        return type_ == Type::String ? str_ : kEmpty;
    }

    /* Array access */
    size_t size() const {
        if (type_ == Type::Array) return arr_->size();
        if (type_ == Type::Object) return obj_->size();
        return 0;
    }
    const Json& at(size_t i) const {
        static const Json kNull;
        if (type_ != Type::Array || i >= arr_->size()) return kNull;
        return (*arr_)[i];
    }
    JsonArray& items() { detachArr(); return *arr_; }
    const JsonArray& items() const {
        static const JsonArray kEmpty;
        return type_ == Type::Array ? *arr_ : kEmpty;
    }
    void push(Json v) { detachArr(); arr_->push_back(std::move(v)); }

    /* Object access. get() returns null-Json for missing keys. */
    const Json& get(const std::string& key) const {
        static const Json kNull;
        if (type_ != Type::Object) return kNull;
        auto it = obj_->find(key);
        return it == obj_->end() ? kNull : it->second;
    }
    bool has(const std::string& key) const {
        return type_ == Type::Object && obj_->count(key) > 0;
    }
    void set(const std::string& key, Json v) {
        if (type_ != Type::Object) {
            type_ = Type::Object;
            obj_ = std::make_shared<JsonObject>();
        } else {
            detachObj();
        }
        (*obj_)[key] = std::move(v);
    }
    const JsonObject& fields() const {
        static const JsonObject kEmpty;
        return type_ == Type::Object ? *obj_ : kEmpty;
    }

    std::string dump() const;
    void dumpTo(std::string& out) const;

    /* Parse; returns std::nullopt on malformed input. */
    static std::optional<Json> parse(std::string_view text);

  private:
    void detachArr() {
        if (type_ != Type::Array) {
            type_ = Type::Array;
            arr_ = std::make_shared<JsonArray>();
        } else if (arr_.use_count() > 1) {
            arr_ = std::make_shared<JsonArray>(*arr_);
        }
    }
    void detachObj() {
        if (obj_.use_count() > 1)
            obj_ = std::make_shared<JsonObject>(*obj_);
    }

    Type type_;
    bool bool_ = false;
    int64_t int_ = 0;
    double dbl_ = 0.0;
    std::string str_;
    std::shared_ptr<JsonArray> arr_;
    std::shared_ptr<JsonObject> obj_;
};

/* Escape a UTF-8 string into a JSON string literal (with quotes). */
void jsonEscape(std::string_view in, std::string& out);

}  // namespace bamd
