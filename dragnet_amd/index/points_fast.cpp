// Fast path for the `dn index-read` reduce loop (and any tagged-point
// pipe): parse NDJSON points `{"fields":{...},"value":N}` and group-sum
// them per metric in C++, replacing the per-line json.loads +
// Aggregator.write Python loop (reference semantics:
// lib/datasource-manta.js:212-219 reduce phase re-running `dn
// index-read`; our cli.cmd_index_read).
//
// Correctness stance: this is an OPT-IN fast path with a PUNT rule —
// any line that is not a flat-scalar point in the exact common shape
// (escape sequences, nested values, non-ASCII under numeric coercion,
// huge magnitudes, non-shortest float formatting, anything surprising)
// is returned verbatim so the caller can push it through the Python
// oracle (points.Aggregator.write).  Aggregation is commutative, so
// fast-path groups and punted lines merge in any order.  The
// differential test (tests/test_points_fast.py) drives both paths over
// adversarial inputs and asserts identical tables/counters.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cerrno>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <string>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

namespace {

constexpr double kMaxExactD = 9007199254740992.0;   // 2^53
constexpr int64_t kMaxExactI = 9007199254740992LL;  // 2^53
constexpr int64_t kBigOrd = (1LL << 62);

// breakdown-column kinds (mirrors query.py bucketizers / points.canonical)
enum Kind : int { K_CANON = 0, K_DATE = 1, K_P2 = 2, K_LIN = 3 };

struct ColSpec {
  std::string name;
  int kind;
  double step;
};

// One scalar JSON value out of the fields object.
struct Scalar {
  enum Tag { MISSING, NUL, BOOL, INT, DBL, STR } tag = MISSING;
  bool b = false;
  int64_t i = 0;
  double d = 0.0;
  const char* s = nullptr;  // into the input buffer (no escapes)
  size_t slen = 0;
  bool ascii = true;  // STR only: all bytes < 0x80
};

struct Acc {
  int64_t iv = 0;
  double dv = 0.0;
  bool exact = true;  // int64 accumulator still authoritative
  void add_int(int64_t v) {
    if (exact) {
      int64_t r;
      if (__builtin_add_overflow(iv, v, &r)) {
        exact = false;
        dv = static_cast<double>(iv) + static_cast<double>(v);
      } else {
        iv = r;
      }
    } else {
      dv += static_cast<double>(v);
    }
  }
  void add_dbl(double v) {
    if (exact) {
      exact = false;
      dv = static_cast<double>(iv);
    }
    dv += v;
  }
};

struct MetricState {
  std::vector<ColSpec> cols;
  std::unordered_map<std::string, Acc> table;
  int64_t ninputs = 0;
  int64_t nonnumeric = 0;
};

// ---- lexer over one line ----
struct Cursor {
  const char* p;
  const char* end;
  bool fail = false;  // => punt the line
  void ws() {
    while (p < end && (*p == ' ' || *p == '\t')) ++p;
  }
  bool lit(char c) {
    ws();
    if (p < end && *p == c) {
      ++p;
      return true;
    }
    fail = true;
    return false;
  }
  bool peek(char c) {
    ws();
    return p < end && *p == c;
  }
};

// Scan a JSON string token (opening quote consumed by caller).  Punts
// on escape sequences, control chars, invalid UTF-8.
bool scan_string(Cursor& c, const char** out, size_t* len, bool* ascii) {
  const char* s = c.p;
  bool a = true;
  while (c.p < c.end) {
    unsigned char ch = static_cast<unsigned char>(*c.p);
    if (ch == '"') {
      *out = s;
      *len = static_cast<size_t>(c.p - s);
      *ascii = a;
      ++c.p;
      return true;
    }
    if (ch == '\\' || ch < 0x20) {  // escapes / raw control: punt
      c.fail = true;
      return false;
    }
    if (ch < 0x80) {
      ++c.p;
      continue;
    }
    // validate one UTF-8 sequence (so py::str at decode cannot throw)
    a = false;
    int n;
    uint32_t cp;
    if ((ch & 0xE0) == 0xC0) { n = 1; cp = ch & 0x1F; }
    else if ((ch & 0xF0) == 0xE0) { n = 2; cp = ch & 0x0F; }
    else if ((ch & 0xF8) == 0xF0) { n = 3; cp = ch & 0x07; }
    else { c.fail = true; return false; }
    ++c.p;
    for (int k = 0; k < n; ++k, ++c.p) {
      if (c.p >= c.end ||
          (static_cast<unsigned char>(*c.p) & 0xC0) != 0x80) {
        c.fail = true;
        return false;
      }
      cp = (cp << 6) | (static_cast<unsigned char>(*c.p) & 0x3F);
    }
    if (cp > 0x10FFFF || (cp >= 0xD800 && cp <= 0xDFFF) ||
        (n == 1 && cp < 0x80) || (n == 2 && cp < 0x800) ||
        (n == 3 && cp < 0x10000)) {
      c.fail = true;  // overlong/surrogate: json.loads would differ
      return false;
    }
  }
  c.fail = true;
  return false;
}

// Scan a JSON number token into Scalar INT (no frac/exp, fits int64,
// json-strict: no leading zeros) or DBL; punts otherwise.
bool scan_number(Cursor& c, Scalar* out) {
  const char* s = c.p;
  if (c.p < c.end && *c.p == '-') ++c.p;
  const char* dstart = c.p;
  while (c.p < c.end && *c.p >= '0' && *c.p <= '9') ++c.p;
  if (c.p == dstart) { c.fail = true; return false; }
  if (*dstart == '0' && c.p - dstart > 1) { c.fail = true; return false; }
  bool isint = true;
  if (c.p < c.end && *c.p == '.') {
    isint = false;
    ++c.p;
    const char* f = c.p;
    while (c.p < c.end && *c.p >= '0' && *c.p <= '9') ++c.p;
    if (c.p == f) { c.fail = true; return false; }
  }
  if (c.p < c.end && (*c.p == 'e' || *c.p == 'E')) {
    isint = false;
    ++c.p;
    if (c.p < c.end && (*c.p == '+' || *c.p == '-')) ++c.p;
    const char* e = c.p;
    while (c.p < c.end && *c.p >= '0' && *c.p <= '9') ++c.p;
    if (c.p == e) { c.fail = true; return false; }
  }
  std::string tok(s, static_cast<size_t>(c.p - s));
  if (isint) {
    errno = 0;
    char* endp = nullptr;
    long long v = strtoll(tok.c_str(), &endp, 10);
    if (errno == ERANGE || *endp != '\0') { c.fail = true; return false; }
    out->tag = Scalar::INT;
    out->i = v;
  } else {
    char* endp = nullptr;
    double v = strtod(tok.c_str(), &endp);
    if (*endp != '\0' || !std::isfinite(v)) { c.fail = true; return false; }
    out->tag = Scalar::DBL;
    out->d = v;
  }
  return true;
}

bool scan_value(Cursor& c, Scalar* out) {
  c.ws();
  if (c.p >= c.end) { c.fail = true; return false; }
  char ch = *c.p;
  if (ch == '"') {
    ++c.p;
    out->tag = Scalar::STR;
    return scan_string(c, &out->s, &out->slen, &out->ascii);
  }
  if (ch == 't') {
    if (c.end - c.p >= 4 && memcmp(c.p, "true", 4) == 0) {
      c.p += 4;
      out->tag = Scalar::BOOL;
      out->b = true;
      return true;
    }
    c.fail = true;
    return false;
  }
  if (ch == 'f') {
    if (c.end - c.p >= 5 && memcmp(c.p, "false", 5) == 0) {
      c.p += 5;
      out->tag = Scalar::BOOL;
      out->b = false;
      return true;
    }
    c.fail = true;
    return false;
  }
  if (ch == 'n') {
    if (c.end - c.p >= 4 && memcmp(c.p, "null", 4) == 0) {
      c.p += 4;
      out->tag = Scalar::NUL;
      return true;
    }
    c.fail = true;
    return false;
  }
  if (ch == '-' || (ch >= '0' && ch <= '9')) return scan_number(c, out);
  c.fail = true;  // nested {/[ or garbage: punt
  return false;
}

// JS ToNumber of a flat ASCII string under a bucketizer
// (krill.to_number fast subset: trimmed plain decimal, "" -> 0).
// Punts on hex/Infinity/underscore/unicode-whitespace forms.
bool str_to_number(const char* s, size_t n, bool ascii, double* out,
                   bool* punt) {
  if (!ascii) { *punt = true; return false; }
  const char* e = s + n;
  while (s < e && (*s == ' ' || *s == '\t' || *s == '\n' || *s == '\r' ||
                   *s == '\f' || *s == '\v'))
    ++s;
  while (e > s && (e[-1] == ' ' || e[-1] == '\t' || e[-1] == '\n' ||
                   e[-1] == '\r' || e[-1] == '\f' || e[-1] == '\v'))
    --e;
  if (s == e) { *out = 0.0; return true; }
  const char* q = s;
  if (*q == '+' || *q == '-') ++q;
  int digits = 0, dots = 0;
  const char* mant_end = e;
  for (const char* r = q; r < e; ++r) {
    if (*r >= '0' && *r <= '9') { ++digits; continue; }
    if (*r == '.') { if (++dots > 1) { *punt = true; return false; }
      continue; }
    if (*r == 'e' || *r == 'E') { mant_end = r; break; }
    *punt = true;  // hex, Infinity, letters, underscores...
    return false;
  }
  if (digits == 0) { *punt = true; return false; }
  if (mant_end != e) {  // exponent part
    const char* r = mant_end + 1;
    if (r < e && (*r == '+' || *r == '-')) ++r;
    if (r >= e) { *punt = true; return false; }
    for (; r < e; ++r)
      if (*r < '0' || *r > '9') { *punt = true; return false; }
  }
  std::string tok(s, static_cast<size_t>(e - s));
  char* endp = nullptr;
  double v = strtod(tok.c_str(), &endp);
  if (*endp != '\0') { *punt = true; return false; }
  *out = v;  // overflow -> +-inf, matching JS Number("1e999")
  return true;
}

// key-element encodings inside the accumulation map key
void key_int(std::string& k, int64_t v) {
  k.push_back('i');
  k.append(reinterpret_cast<const char*>(&v), 8);
}
void key_str(std::string& k, const char* s, size_t n) {
  uint32_t len = static_cast<uint32_t>(n);
  k.push_back('s');
  k.append(reinterpret_cast<const char*>(&len), 4);
  k.append(s, n);
}

enum LineWhat { L_OK, L_SKIP, L_PUNT };

// Parse + aggregate one line.  Implements cli.cmd_index_read's loop
// body over points.Aggregator.write for the flat-scalar fast shape.
LineWhat do_line(
    const char* lp, const char* le, std::vector<MetricState>& ms) {
  Cursor c{lp, le};
  if (!c.lit('{')) return L_PUNT;

  // collect the flat fields map and the value
  std::vector<std::pair<std::pair<const char*, size_t>, Scalar>> fields;
  bool have_fields = false;
  Scalar value;
  bool have_value = false;

  if (c.peek('}')) {
    ++c.p;  // {}: no fields, no value -> mi missing -> skip
  } else {
    for (;;) {
      if (!c.lit('"')) return L_PUNT;
      const char* kn;
      size_t kl;
      bool ka;
      if (!scan_string(c, &kn, &kl, &ka)) return L_PUNT;
      if (!c.lit(':')) return L_PUNT;
      if (kl == 6 && memcmp(kn, "fields", 6) == 0) {
        if (have_fields) return L_PUNT;  // dup: json.loads
        have_fields = true;                        // keeps last; punt
        if (!c.lit('{')) return L_PUNT;
        if (c.peek('}')) {
          ++c.p;
        } else {
          for (;;) {
            if (!c.lit('"')) return L_PUNT;
            const char* fn;
            size_t fl;
            bool fa;
            if (!scan_string(c, &fn, &fl, &fa)) return L_PUNT;
            if (!c.lit(':')) return L_PUNT;
            Scalar sv;
            if (!scan_value(c, &sv)) return L_PUNT;
            // duplicate field keys: json.loads keeps the LAST
            bool dup = false;
            for (auto& kv : fields)
              if (kv.first.second == fl &&
                  memcmp(kv.first.first, fn, fl) == 0) {
                kv.second = sv;
                dup = true;
                break;
              }
            if (!dup) fields.push_back({{fn, fl}, sv});
            if (c.peek(',')) { ++c.p; continue; }
            if (!c.lit('}')) return L_PUNT;
            break;
          }
        }
      } else if (kl == 5 && memcmp(kn, "value", 5) == 0) {
        if (have_value) return L_PUNT;
        have_value = true;
        if (!scan_value(c, &value)) return L_PUNT;
        if (value.tag != Scalar::INT && value.tag != Scalar::DBL)
          return L_PUNT;  // bool/str/null value: Python path
        if (value.tag == Scalar::INT &&
            (value.i >= kMaxExactI || value.i <= -kMaxExactI))
          return L_PUNT;
      } else {
        return L_PUNT;  // unexpected top-level key
      }
      if (c.peek(',')) { ++c.p; continue; }
      if (!c.lit('}')) return L_PUNT;
      break;
    }
  }
  c.ws();
  if (c.p != c.end) return L_PUNT;  // trailing garbage

  // __dn_metric routing (bool counts as int: Python isinstance)
  int64_t mi = -1;
  bool mi_found = false;
  for (auto& kv : fields)
    if (kv.first.second == 11 &&
        memcmp(kv.first.first, "__dn_metric", 11) == 0) {
      if (kv.second.tag == Scalar::INT) { mi = kv.second.i; mi_found = true; }
      else if (kv.second.tag == Scalar::BOOL) {
        mi = kv.second.b ? 1 : 0;
        mi_found = true;
      }
      break;
    }
  if (!mi_found || mi < 0 || mi >= static_cast<int64_t>(ms.size()))
    return L_SKIP;
  if (!have_value) return L_PUNT;  // KeyError in Python

  // state commits only at terminal OK outcomes below — a punt after
  // this point must leave counters untouched (the Python path will
  // count the line)
  MetricState& m = ms[static_cast<size_t>(mi)];
  std::string key;
  key.reserve(32);
  for (const ColSpec& col : m.cols) {
    // literal-key lookup; with flat scalar fields a dotted pluck
    // fallback always lands on MISSING (points.lookup + krill.pluck)
    const Scalar* v = nullptr;
    for (auto& kv : fields)
      if (kv.first.second == col.name.size() &&
          memcmp(kv.first.first, col.name.data(), col.name.size()) == 0) {
        v = &kv.second;
        break;
      }
    Scalar miss;
    if (v == nullptr) v = &miss;

    if (col.kind == K_P2 || col.kind == K_LIN) {
      double num;
      bool punt = false;
      switch (v->tag) {
        case Scalar::INT: num = static_cast<double>(v->i); break;
        case Scalar::DBL: num = v->d; break;
        case Scalar::STR:
          if (!str_to_number(v->s, v->slen, v->ascii, &num, &punt)) {
            if (punt) return L_PUNT;
            num = NAN;
          }
          break;
        default:
          num = NAN;  // bool/null/missing -> nonnumeric drop
      }
      if (!std::isfinite(num)) {
        m.ninputs += 1;
        m.nonnumeric += 1;
        return L_OK;  // dropped at this column, like Aggregator.write
      }
      int64_t ord;
      if (col.kind == K_P2) {
        if (num < 1.0) {
          ord = 0;
        } else {
          int e;
          std::frexp(num, &e);
          ord = e;
        }
      } else {
        double q = std::floor(num / col.step);
        if (!(std::fabs(q) < static_cast<double>(kBigOrd)))
          return L_PUNT;
        ord = static_cast<int64_t>(q);
      }
      key_int(key, ord);
    } else {  // canonical / canonical-date (points.canonical)
      switch (v->tag) {
        case Scalar::MISSING: key_str(key, "undefined", 9); break;
        case Scalar::NUL: key_str(key, "null", 4); break;
        case Scalar::BOOL:
          if (v->b) key_str(key, "true", 4);
          else key_str(key, "false", 5);
          break;
        case Scalar::INT:
          if (col.kind == K_DATE) {
            key_int(key, v->i);
          } else {
            char buf[24];
            int n = snprintf(buf, sizeof(buf), "%lld",
                             static_cast<long long>(v->i));
            key_str(key, buf, static_cast<size_t>(n));
          }
          break;
        case Scalar::DBL: {
          double d = v->d;
          if (col.kind == K_DATE) {
            double t = std::trunc(d);
            if (!(std::fabs(t) < static_cast<double>(kBigOrd)))
              return L_PUNT;
            key_int(key, static_cast<int64_t>(t));
          } else if (d == std::trunc(d) && std::fabs(d) < kMaxExactD) {
            char buf[24];
            int n = snprintf(buf, sizeof(buf), "%lld",
                             static_cast<long long>(d));
            key_str(key, buf, static_cast<size_t>(n));
          } else {
            return L_PUNT;  // js_num_str uses Python repr
          }
          break;
        }
        case Scalar::STR:
          key_str(key, v->s, v->slen);
          break;
      }
    }
  }

  m.ninputs += 1;
  Acc& a = m.table[key];
  if (value.tag == Scalar::INT) a.add_int(value.i);
  else a.add_dbl(value.d);
  return L_OK;
}

// Decode one accumulated key back into a Python tuple.  Integer key
// elements are bucket ordinals / date seconds (int); the rest str.
py::tuple decode_key(const std::string& k) {
  std::vector<py::object> elems;
  size_t i = 0;
  while (i < k.size()) {
    char tag = k[i++];
    if (tag == 'i') {
      int64_t v;
      memcpy(&v, k.data() + i, 8);
      i += 8;
      elems.push_back(py::int_(v));
    } else {
      uint32_t len;
      memcpy(&len, k.data() + i, 4);
      i += 4;
      elems.push_back(
          py::reinterpret_steal<py::object>(
              PyUnicode_DecodeUTF8(k.data() + i, len, nullptr)));
      i += len;
    }
  }
  py::tuple t(elems.size());
  for (size_t j = 0; j < elems.size(); ++j) t[j] = elems[j];
  return t;
}

// reduce_tagged(data, specs) ->
//   (tables: list[dict[tuple, int|float]], ninputs: list[int],
//    nonnumeric: list[int], punted: list[bytes])
// specs: per metric, list of (name, kind, step) breakdown columns.
py::tuple reduce_tagged(
    py::bytes data,
    const std::vector<std::vector<std::tuple<std::string, int, double>>>&
        specs) {
  char* buf;
  Py_ssize_t blen;
  if (PyBytes_AsStringAndSize(data.ptr(), &buf, &blen) != 0)
    throw py::error_already_set();

  std::vector<MetricState> ms(specs.size());
  for (size_t i = 0; i < specs.size(); ++i)
    for (const auto& t : specs[i])
      ms[i].cols.push_back(
          {std::get<0>(t), std::get<1>(t), std::get<2>(t)});

  std::vector<std::pair<const char*, size_t>> punted;
  {
    py::gil_scoped_release rel;
    const char* p = buf;
    const char* end = buf + blen;
    while (p < end) {
      const char* nl = static_cast<const char*>(
          memchr(p, '\n', static_cast<size_t>(end - p)));
      const char* le = nl ? nl : end;
      // strip() per cmd_index_read: trim ws, skip empties
      const char* s = p;
      const char* e = le;
      while (s < e && (*s == ' ' || *s == '\t' || *s == '\r')) ++s;
      while (e > s && (e[-1] == ' ' || e[-1] == '\t' || e[-1] == '\r'))
        --e;
      if (s < e && do_line(s, e, ms) == L_PUNT)
        punted.push_back({s, static_cast<size_t>(e - s)});
      p = nl ? nl + 1 : end;
    }
  }

  py::list tables, nin, nonn, plines;
  for (auto& m : ms) {
    py::dict t;
    for (auto& kv : m.table) {
      py::object v = kv.second.exact
                         ? static_cast<py::object>(py::int_(kv.second.iv))
                         : static_cast<py::object>(
                               py::float_(kv.second.dv));
      t[decode_key(kv.first)] = v;
    }
    tables.append(t);
    nin.append(py::int_(m.ninputs));
    nonn.append(py::int_(m.nonnumeric));
  }
  for (auto& pl : punted) plines.append(py::bytes(pl.first, pl.second));
  return py::make_tuple(tables, nin, nonn, plines);
}

// ---- emit side: byte-exact json.dumps(separators=(",",":")) ----
// for output_points (dn scan --points / index-scan map emit).  Any
// value outside str/int/float/bool/None falls back to the Python
// point_json callable for THAT point, preserving output order.

void esc_json(std::string& out, PyObject* u) {
  // matches c_encode_basestring_ascii: ensure_ascii, lowercase hex
  static const char* hexd = "0123456789abcdef";
  if (PyUnicode_READY(u) != 0) throw py::error_already_set();
  Py_ssize_t n = PyUnicode_GET_LENGTH(u);
  int kind = PyUnicode_KIND(u);
  const void* dat = PyUnicode_DATA(u);
  out.push_back('"');
  for (Py_ssize_t i = 0; i < n; ++i) {
    Py_UCS4 cp = PyUnicode_READ(kind, dat, i);
    if (cp == '"') { out += "\\\""; continue; }
    if (cp == '\\') { out += "\\\\"; continue; }
    if (cp >= 0x20 && cp < 0x7F) {
      out.push_back(static_cast<char>(cp));
      continue;
    }
    switch (cp) {
      case 0x08: out += "\\b"; continue;
      case 0x09: out += "\\t"; continue;
      case 0x0A: out += "\\n"; continue;
      case 0x0C: out += "\\f"; continue;
      case 0x0D: out += "\\r"; continue;
    }
    auto u4 = [&](uint32_t v) {
      out += "\\u";
      out.push_back(hexd[(v >> 12) & 15]);
      out.push_back(hexd[(v >> 8) & 15]);
      out.push_back(hexd[(v >> 4) & 15]);
      out.push_back(hexd[v & 15]);
    };
    if (cp >= 0x10000) {
      uint32_t v = cp - 0x10000;
      u4(0xD800 + (v >> 10));
      u4(0xDC00 + (v & 0x3FF));
    } else {
      u4(cp);
    }
  }
  out.push_back('"');
}

bool emit_scalar(std::string& out, PyObject* o) {
  if (o == Py_None) { out += "null"; return true; }
  if (PyBool_Check(o)) {
    out += (o == Py_True) ? "true" : "false";
    return true;
  }
  if (PyLong_Check(o)) {
    int overflow = 0;
    long long v = PyLong_AsLongLongAndOverflow(o, &overflow);
    if (!overflow) {
      char buf[24];
      out.append(buf, static_cast<size_t>(
          snprintf(buf, sizeof(buf), "%lld", v)));
      return true;
    }
    PyObject* r = PyObject_Str(o);  // big ints: int.__repr__, ascii
    if (r == nullptr) throw py::error_already_set();
    Py_ssize_t len;
    const char* c = PyUnicode_AsUTF8AndSize(r, &len);
    if (c == nullptr) {
      Py_DECREF(r);
      throw py::error_already_set();
    }
    out.append(c, static_cast<size_t>(len));
    Py_DECREF(r);
    return true;
  }
  if (PyFloat_Check(o)) {
    double d = PyFloat_AS_DOUBLE(o);
    if (std::isnan(d)) { out += "NaN"; return true; }
    if (std::isinf(d)) {
      out += d > 0 ? "Infinity" : "-Infinity";
      return true;
    }
    PyObject* r = PyObject_Repr(o);  // json uses float.__repr__
    if (r == nullptr) throw py::error_already_set();
    Py_ssize_t len;
    const char* c = PyUnicode_AsUTF8AndSize(r, &len);
    if (c == nullptr) {
      Py_DECREF(r);
      throw py::error_already_set();
    }
    out.append(c, static_cast<size_t>(len));
    Py_DECREF(r);
    return true;
  }
  if (PyUnicode_Check(o)) {
    esc_json(out, o);
    return true;
  }
  return false;  // nested/list/other: fall back to point_json
}

// serialize_points(points, fallback) -> bytes (one NDJSON line each;
// fallback(point) -> str handles points with non-scalar values)
py::bytes serialize_points(py::sequence points, py::object fallback) {
  std::string out;
  out.reserve(1 << 16);
  for (py::handle ph : points) {
    size_t mark = out.size();
    PyObject* p = ph.ptr();
    PyObject* fields = PyDict_Check(p)
                           ? PyDict_GetItemString(p, "fields")
                           : nullptr;
    PyObject* value = PyDict_Check(p)
                          ? PyDict_GetItemString(p, "value")
                          : nullptr;
    bool ok = fields != nullptr && PyDict_Check(fields) &&
              value != nullptr;
    if (ok) {
      out += "{\"fields\":{";
      PyObject *k, *v;
      Py_ssize_t pos = 0;
      bool first = true;
      while (ok && PyDict_Next(fields, &pos, &k, &v)) {
        if (!PyUnicode_Check(k)) { ok = false; break; }
        if (!first) out.push_back(',');
        first = false;
        esc_json(out, k);
        out.push_back(':');
        ok = emit_scalar(out, v);
      }
      if (ok) {
        out += "},\"value\":";
        ok = emit_scalar(out, value);
        if (ok) out += "}\n";
      }
    }
    if (!ok) {
      out.resize(mark);  // rewind partial line, use the Python path
      py::str line = fallback(ph);
      Py_ssize_t len;
      const char* c = PyUnicode_AsUTF8AndSize(line.ptr(), &len);
      if (c == nullptr) throw py::error_already_set();
      out.append(c, static_cast<size_t>(len));
      out.push_back('\n');
    }
  }
  return py::bytes(out);
}

}  // namespace

PYBIND11_MODULE(_points, m) {
  m.doc() = "dragnet_amd fast tagged-point reducer (C++)";
  m.def("reduce_tagged", &reduce_tagged, py::arg("data"),
        py::arg("specs"));
  m.def("serialize_points", &serialize_points, py::arg("points"),
        py::arg("fallback"));
}
