// Minimal JSON DOM — parse + compact serialize (Go json.Marshal style).
// Self-contained (no third-party deps are available in this image).
// Sufficient for the modelx wire format: objects keep insertion order so
// serialized manifests are stable and diffable.
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <utility>
#include <vector>

namespace modelx {
namespace json {

class Value;
using Array = std::vector<Value>;

// Order-preserving object (small N; linear lookup is fine for wire structs).
class Object {
 public:
  Value& operator[](const std::string& k);
  const Value* find(const std::string& k) const;
  bool contains(const std::string& k) const { return find(k) != nullptr; }
  size_t size() const { return items_.size(); }
  auto begin() const { return items_.begin(); }
  auto end() const { return items_.end(); }
  auto begin() { return items_.begin(); }
  auto end() { return items_.end(); }

 private:
  std::vector<std::pair<std::string, Value>> items_;
};

enum class Kind { Null, Bool, Int, Double, String, Array, Object };

class Value {
 public:
  Value() : kind_(Kind::Null) {}
  Value(std::nullptr_t) : kind_(Kind::Null) {}
  Value(bool b) : kind_(Kind::Bool), b_(b) {}
  Value(int i) : kind_(Kind::Int), i_(i) {}
  Value(int64_t i) : kind_(Kind::Int), i_(i) {}
  Value(uint64_t i) : kind_(Kind::Int), i_(static_cast<int64_t>(i)) {}
  Value(double d) : kind_(Kind::Double), d_(d) {}
  Value(const char* s) : kind_(Kind::String), s_(s) {}
  Value(std::string s) : kind_(Kind::String), s_(std::move(s)) {}
  Value(Array a) : kind_(Kind::Array), a_(std::make_shared<Array>(std::move(a))) {}
  Value(Object o) : kind_(Kind::Object), o_(std::make_shared<Object>(std::move(o))) {}

  Kind kind() const { return kind_; }
  bool is_null() const { return kind_ == Kind::Null; }
  bool is_object() const { return kind_ == Kind::Object; }
  bool is_array() const { return kind_ == Kind::Array; }
  bool is_string() const { return kind_ == Kind::String; }

  bool as_bool(bool def = false) const { return kind_ == Kind::Bool ? b_ : def; }
  int64_t as_int(int64_t def = 0) const {
    if (kind_ == Kind::Int) return i_;
    if (kind_ == Kind::Double) return static_cast<int64_t>(d_);
    return def;
  }
  double as_double(double def = 0) const {
    if (kind_ == Kind::Double) return d_;
    if (kind_ == Kind::Int) return static_cast<double>(i_);
    return def;
  }
  const std::string& as_string() const {
    static const std::string empty;
    return kind_ == Kind::String ? s_ : empty;
  }

  // Object access: returns Null value for missing keys.
  const Value& operator[](const std::string& k) const {
    static const Value null_v;
    if (kind_ != Kind::Object || !o_) return null_v;
    const Value* v = o_->find(k);
    return v ? *v : null_v;
  }
  // Mutable object access (converts Null -> Object).
  Value& set(const std::string& k, Value v);

  const Array& items() const {
    static const Array empty;
    return (kind_ == Kind::Array && a_) ? *a_ : empty;
  }
  const Object& object() const {
    static const Object empty;
    return (kind_ == Kind::Object && o_) ? *o_ : empty;
  }
  Array& mutable_array() {
    if (kind_ != Kind::Array) {
      kind_ = Kind::Array;
      a_ = std::make_shared<Array>();
    }
    return *a_;
  }

  void serialize(std::string& out) const;
  std::string dump() const {
    std::string out;
    serialize(out);
    return out;
  }

 private:
  Kind kind_;
  bool b_ = false;
  int64_t i_ = 0;
  double d_ = 0;
  std::string s_;
  std::shared_ptr<Array> a_;
  std::shared_ptr<Object> o_;
};

// Parse; throws std::runtime_error on malformed input.
Value parse(const std::string& text);
Value parse(const char* data, size_t len);

}  // namespace json
}  // namespace modelx
