// Native Avro binary codec + Confluent wire framing (host data plane).
//
// The reference does all topic I/O through confluent_kafka's C serializers
// (scripts/publish_lab1_data.py:158-180); our Python codec (wire/avro.py)
// defines the semantics and this C++ codec is the fast path the broker
// uses for record batches (wire/topics.py): schema compiled once, batch
// encode/decode without per-field Python dispatch.  Covers the lab
// schemas' subset: null/boolean/int/long/float/double/string/bytes,
// record, array, map, union, enum, and the Confluent framing
// magic 0x00 + 4-byte big-endian schema id (publish_lab3_data.py:96-122).
#include <torch/extension.h>
#include <pybind11/numpy.h>

#include <cstring>
#include <memory>
#include <string>
#include <vector>

namespace py = pybind11;

namespace qsa_avro {

enum class T {
  Null, Boolean, Int, Long, Float, Double, String, Bytes,
  Record, Array, Map, Union, Enum
};

struct Node;
using NodeP = std::shared_ptr<Node>;

struct Field {
  std::string name;
  NodeP schema;
  bool has_default = false;
  py::object default_value;  // kept alive with the codec
};

struct Node {
  T type;
  std::vector<Field> fields;        // record
  NodeP items;                      // array
  NodeP values;                     // map
  std::vector<NodeP> branches;      // union
  std::vector<std::string> symbols; // enum
};

static NodeP parse(py::handle defn);

static NodeP parse_typename(const std::string& t, py::handle defn) {
  auto n = std::make_shared<Node>();
  if (t == "null") n->type = T::Null;
  else if (t == "boolean") n->type = T::Boolean;
  else if (t == "int") n->type = T::Int;
  else if (t == "long") n->type = T::Long;
  else if (t == "float") n->type = T::Float;
  else if (t == "double") n->type = T::Double;
  else if (t == "string") n->type = T::String;
  else if (t == "bytes") n->type = T::Bytes;
  else if (t == "record") {
    n->type = T::Record;
    for (auto f : defn.attr("__getitem__")("fields")) {
      Field fld;
      fld.name = py::cast<std::string>(f["name"]);
      fld.schema = parse(f["type"]);
      if (py::cast<py::dict>(f).contains("default")) {
        fld.has_default = true;
        fld.default_value = py::reinterpret_borrow<py::object>(f["default"]);
      }
      n->fields.push_back(std::move(fld));
    }
  } else if (t == "array") {
    n->type = T::Array;
    n->items = parse(defn.attr("__getitem__")("items"));
  } else if (t == "map") {
    n->type = T::Map;
    n->values = parse(defn.attr("__getitem__")("values"));
  } else if (t == "enum") {
    n->type = T::Enum;
    for (auto s : defn.attr("__getitem__")("symbols"))
      n->symbols.push_back(py::cast<std::string>(s));
  } else {
    throw std::runtime_error("unsupported Avro type: " + t);
  }
  return n;
}

static NodeP parse(py::handle defn) {
  if (py::isinstance<py::str>(defn))
    return parse_typename(py::cast<std::string>(defn), defn);
  if (py::isinstance<py::list>(defn)) {
    auto n = std::make_shared<Node>();
    n->type = T::Union;
    for (auto b : defn) n->branches.push_back(parse(b));
    return n;
  }
  // dict: {"type": X, ...}; X may itself be a dict/list (nesting)
  py::handle t = defn.attr("__getitem__")("type");
  if (py::isinstance<py::str>(t))
    return parse_typename(py::cast<std::string>(t), defn);
  return parse(t);
}

// ---- varint -------------------------------------------------------------

static inline void w_long(std::string& out, long long v) {
  unsigned long long n =
      ((unsigned long long)v << 1) ^ (unsigned long long)(v >> 63);
  while (n >= 0x80) {
    out.push_back((char)(n | 0x80));
    n >>= 7;
  }
  out.push_back((char)n);
}

struct Reader {
  const unsigned char* p;
  const unsigned char* end;
  long long vlong() {
    unsigned long long acc = 0;
    int shift = 0;
    while (true) {
      if (p >= end) throw std::runtime_error("truncated varint");
      unsigned char b = *p++;
      acc |= (unsigned long long)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    return (long long)(acc >> 1) ^ -(long long)(acc & 1);
  }
  const unsigned char* take(size_t n) {
    if (p + n > end) throw std::runtime_error("truncated payload");
    const unsigned char* q = p;
    p += n;
    return q;
  }
};

// ---- encode -------------------------------------------------------------

static void encode(const NodeP& s, py::handle v, std::string& out) {
  switch (s->type) {
    case T::Null:
      return;
    case T::Boolean:
      out.push_back(py::cast<bool>(v) ? 1 : 0);
      return;
    case T::Int:
    case T::Long:
      w_long(out, py::cast<long long>(v));
      return;
    case T::Float: {
      float f = py::cast<float>(v);
      out.append(reinterpret_cast<const char*>(&f), 4);
      return;
    }
    case T::Double: {
      double d = py::cast<double>(v);
      out.append(reinterpret_cast<const char*>(&d), 8);
      return;
    }
    case T::String: {
      std::string raw = py::cast<std::string>(py::str(v));
      w_long(out, (long long)raw.size());
      out.append(raw);
      return;
    }
    case T::Bytes: {
      py::bytes b = py::cast<py::bytes>(v);
      std::string raw = b;
      w_long(out, (long long)raw.size());
      out.append(raw);
      return;
    }
    case T::Record: {
      for (const auto& f : s->fields) {
        py::object fv;
        if (py::cast<py::dict>(v).contains(f.name.c_str()))
          fv = py::reinterpret_borrow<py::object>(
              v.attr("__getitem__")(f.name.c_str()));
        else if (f.has_default)
          fv = f.default_value;
        else
          throw std::runtime_error("missing field " + f.name);
        encode(f.schema, fv, out);
      }
      return;
    }
    case T::Array: {
      py::sequence seq = py::cast<py::sequence>(v);
      size_t n = seq.size();
      if (n) {
        w_long(out, (long long)n);
        for (size_t i = 0; i < n; ++i) encode(s->items, seq[i], out);
      }
      w_long(out, 0);
      return;
    }
    case T::Map: {
      py::dict d = py::cast<py::dict>(v);
      size_t n = d.size();
      if (n) {
        w_long(out, (long long)n);
        for (auto item : d) {
          std::string k = py::cast<std::string>(py::str(item.first));
          w_long(out, (long long)k.size());
          out.append(k);
          encode(s->values, item.second, out);
        }
      }
      w_long(out, 0);
      return;
    }
    case T::Union: {
      for (size_t i = 0; i < s->branches.size(); ++i) {
        bool is_null = s->branches[i]->type == T::Null;
        if ((v.is_none() && is_null) || (!v.is_none() && !is_null)) {
          w_long(out, (long long)i);
          encode(s->branches[i], v, out);
          return;
        }
      }
      throw std::runtime_error("no union branch matches value");
    }
    case T::Enum: {
      std::string sym = py::cast<std::string>(py::str(v));
      for (size_t i = 0; i < s->symbols.size(); ++i)
        if (s->symbols[i] == sym) {
          w_long(out, (long long)i);
          return;
        }
      throw std::runtime_error("unknown enum symbol " + sym);
    }
  }
}

// ---- decode -------------------------------------------------------------

static py::object decode(const NodeP& s, Reader& r) {
  switch (s->type) {
    case T::Null:
      return py::none();
    case T::Boolean:
      return py::bool_(*r.take(1) != 0);
    case T::Int:
    case T::Long:
      return py::int_(r.vlong());
    case T::Float: {
      float f;
      std::memcpy(&f, r.take(4), 4);
      return py::float_(f);
    }
    case T::Double: {
      double d;
      std::memcpy(&d, r.take(8), 8);
      return py::float_(d);
    }
    case T::String: {
      long long n = r.vlong();
      const unsigned char* q = r.take((size_t)n);
      return py::str(std::string(reinterpret_cast<const char*>(q),
                                 (size_t)n));
    }
    case T::Bytes: {
      long long n = r.vlong();
      const unsigned char* q = r.take((size_t)n);
      return py::bytes(reinterpret_cast<const char*>(q), (size_t)n);
    }
    case T::Record: {
      py::dict d;
      for (const auto& f : s->fields)
        d[f.name.c_str()] = decode(f.schema, r);
      return d;
    }
    case T::Array: {
      py::list out;
      while (true) {
        long long n = r.vlong();
        if (n == 0) break;
        if (n < 0) {
          n = -n;
          r.vlong();  // skip byte size
        }
        for (long long i = 0; i < n; ++i) out.append(decode(s->items, r));
      }
      return out;
    }
    case T::Map: {
      py::dict out;
      while (true) {
        long long n = r.vlong();
        if (n == 0) break;
        if (n < 0) {
          n = -n;
          r.vlong();
        }
        for (long long i = 0; i < n; ++i) {
          long long klen = r.vlong();
          const unsigned char* q = r.take((size_t)klen);
          // py::str keeps embedded NULs (c_str() would truncate)
          out[py::str(std::string(reinterpret_cast<const char*>(q),
                                  (size_t)klen))] = decode(s->values, r);
        }
      }
      return out;
    }
    case T::Union: {
      long long idx = r.vlong();
      if (idx < 0 || (size_t)idx >= s->branches.size())
        throw std::runtime_error("bad union index");
      return decode(s->branches[(size_t)idx], r);
    }
    case T::Enum: {
      long long idx = r.vlong();
      return py::str(s->symbols.at((size_t)idx));
    }
  }
  return py::none();
}

class Codec {
 public:
  explicit Codec(py::object defn) : root_(parse(defn)), defn_(defn) {}

  py::bytes serialize(long schema_id, py::object value) const {
    std::string out;
    out.reserve(256);
    out.push_back(0);  // magic
    unsigned int id = (unsigned int)schema_id;
    char hdr[4] = {(char)(id >> 24), (char)(id >> 16), (char)(id >> 8),
                   (char)id};
    out.append(hdr, 4);
    encode(root_, value, out);
    return py::bytes(out);
  }

  py::tuple deserialize(py::bytes payload) const {
    std::string raw = payload;
    if (raw.size() < 5 || raw[0] != 0)
      throw std::runtime_error("not Confluent Avro wire format");
    const unsigned char* p =
        reinterpret_cast<const unsigned char*>(raw.data());
    unsigned int id = ((unsigned int)p[1] << 24) | ((unsigned int)p[2] << 16) |
                      ((unsigned int)p[3] << 8) | (unsigned int)p[4];
    Reader r{p + 5, p + raw.size()};
    py::object v = decode(root_, r);
    return py::make_tuple((long)id, v);
  }

  py::list serialize_batch(long schema_id, py::sequence values) const {
    py::list out;
    for (auto v : values)
      out.append(serialize(schema_id,
                           py::reinterpret_borrow<py::object>(v)));
    return out;
  }

  py::list deserialize_batch(py::sequence payloads) const {
    py::list out;
    for (auto pl : payloads)
      out.append(deserialize(py::cast<py::bytes>(pl))[1]);
    return out;
  }

  // Columnar batch decode (K10: record batches feeding columnar GPU
  // buffers): long/int/boolean columns land in int64 numpy arrays,
  // float/double in float64, strings in a py::list — one pass over the
  // wire bytes, no per-record Python dicts.
  py::dict decode_columns(py::sequence payloads,
                          std::vector<std::string> columns) const {
    if (root_->type != T::Record)
      throw std::runtime_error("decode_columns needs a record schema");
    const size_t n = payloads.size();
    // per-column storage
    std::vector<int> kind(columns.size(), -1);  // 0=int64 1=f64 2=obj
    std::vector<std::vector<long long>> icols(columns.size());
    std::vector<std::vector<double>> fcols(columns.size());
    std::vector<py::list> ocols(columns.size());
    auto col_of = [&](const std::string& name) -> int {
      for (size_t c = 0; c < columns.size(); ++c)
        if (columns[c] == name) return (int)c;
      return -1;
    };
    for (size_t i = 0; i < n; ++i) {
      std::string raw = py::cast<py::bytes>(payloads[i]);
      if (raw.size() < 5 || raw[0] != 0)
        throw std::runtime_error("not Confluent wire format");
      const unsigned char* p =
          reinterpret_cast<const unsigned char*>(raw.data());
      Reader r{p + 5, p + raw.size()};
      for (const auto& f : root_->fields) {
        const int c = col_of(f.name);
        const NodeP& fs = f.schema;
        // resolve through a [null, X] union for extraction
        if (c < 0) {
          decode(fs, r);  // skip (still must advance the reader)
          continue;
        }
        NodeP eff = fs;
        bool is_null = false;
        if (fs->type == T::Union) {
          long long idx = r.vlong();
          eff = fs->branches.at((size_t)idx);
          is_null = eff->type == T::Null;
        }
        switch (is_null ? T::Null : eff->type) {
          case T::Int:
          case T::Long: {
            long long v = r.vlong();
            if (kind[c] < 0) kind[c] = 0;
            if (kind[c] == 0) icols[c].push_back(v);
            else if (kind[c] == 1) fcols[c].push_back((double)v);
            else ocols[c].append(py::int_(v));
            break;
          }
          case T::Boolean: {
            bool v = *r.take(1) != 0;
            if (kind[c] < 0) kind[c] = 0;
            if (kind[c] == 0) icols[c].push_back(v ? 1 : 0);
            else ocols[c].append(py::bool_(v));
            break;
          }
          case T::Float: {
            float f32v;
            std::memcpy(&f32v, r.take(4), 4);
            if (kind[c] < 0) kind[c] = 1;
            fcols[c].push_back((double)f32v);
            break;
          }
          case T::Double: {
            double dv;
            std::memcpy(&dv, r.take(8), 8);
            if (kind[c] < 0) kind[c] = 1;
            fcols[c].push_back(dv);
            break;
          }
          default: {
            // strings / complex / null -> object column
            py::object v = is_null ? py::object(py::none())
                                   : decode(eff, r);
            if (kind[c] < 0) kind[c] = 2;
            ocols[c].append(v);
            break;
          }
        }
      }
    }
    py::dict out;
    for (size_t c = 0; c < columns.size(); ++c) {
      if (kind[c] == 0) {
        py::array_t<long long> arr((py::ssize_t)icols[c].size());
        std::memcpy(arr.mutable_data(), icols[c].data(),
                    icols[c].size() * sizeof(long long));
        out[columns[c].c_str()] = arr;
      } else if (kind[c] == 1) {
        py::array_t<double> arr((py::ssize_t)fcols[c].size());
        std::memcpy(arr.mutable_data(), fcols[c].data(),
                    fcols[c].size() * sizeof(double));
        out[columns[c].c_str()] = arr;
      } else {
        out[columns[c].c_str()] = ocols[c];
      }
    }
    return out;
  }

 private:
  NodeP root_;
  py::object defn_;  // keeps default-value objects alive
};

}  // namespace qsa_avro

void register_avro(py::module_& m) {
  py::class_<qsa_avro::Codec>(m, "AvroCodec")
      .def(py::init<py::object>())
      .def("serialize", &qsa_avro::Codec::serialize)
      .def("deserialize", &qsa_avro::Codec::deserialize)
      .def("serialize_batch", &qsa_avro::Codec::serialize_batch)
      .def("decode_columns", &qsa_avro::Codec::decode_columns)
      .def("deserialize_batch", &qsa_avro::Codec::deserialize_batch);
}
