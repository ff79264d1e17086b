// Native index sink: aggregated points -> SQLite index file through the
// SQLite C API (the reference's index-sink is backed by the native
// sqlite3 npm binding, lib/index-sink.js:116-230; this is the
// MI355X-framework equivalent of that native layer).
//
// The Python IndexSink drives schema creation (its SQL is pinned by
// golden tests) and hands the hot row stream over columnar: one int64
// array or list[str] per breakdown column + one float64 value array per
// metric, inserted through a prepared statement inside a single
// transaction.
//
// The system image ships libsqlite3.so.0 without headers, so the
// (stable, C) ABI surface used here is declared locally and the
// versioned .so is linked by absolute path (setup.py).

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

extern "C" {
typedef struct sqlite3 sqlite3;
typedef struct sqlite3_stmt sqlite3_stmt;
int sqlite3_open_v2(const char *, sqlite3 **, int, const char *);
int sqlite3_close(sqlite3 *);
int sqlite3_exec(sqlite3 *, const char *,
                 int (*)(void *, int, char **, char **), void *, char **);
void sqlite3_free(void *);
int sqlite3_prepare_v2(sqlite3 *, const char *, int, sqlite3_stmt **,
                       const char **);
int sqlite3_bind_int64(sqlite3_stmt *, int, long long);
int sqlite3_bind_double(sqlite3_stmt *, int, double);
int sqlite3_bind_text(sqlite3_stmt *, int, const char *, int,
                      void (*)(void *));
int sqlite3_bind_null(sqlite3_stmt *, int);
int sqlite3_step(sqlite3_stmt *);
int sqlite3_reset(sqlite3_stmt *);
int sqlite3_finalize(sqlite3_stmt *);
const char *sqlite3_errmsg(sqlite3 *);
double sqlite3_column_double(sqlite3_stmt *, int);
const unsigned char *sqlite3_column_text(sqlite3_stmt *, int);
int sqlite3_column_bytes(sqlite3_stmt *, int);
int sqlite3_column_type(sqlite3_stmt *, int);
}

#define SQLITE_OK 0
#define SQLITE_ROW 100
#define SQLITE_DONE 101
#define SQLITE_NULL 5
#define SQLITE_OPEN_READONLY 0x1
#define SQLITE_OPEN_READWRITE 0x2
#define SQLITE_OPEN_CREATE 0x4
// SQLITE_TRANSIENT: sqlite copies the text before returning
static void (*const kTransient)(void *) =
    reinterpret_cast<void (*)(void *)>(-1);

namespace {

struct CSink {
  sqlite3 *db = nullptr;
  std::vector<sqlite3_stmt *> inserts;
  bool in_txn = false;

  explicit CSink(const std::string &path) {
    if (sqlite3_open_v2(path.c_str(), &db,
                        SQLITE_OPEN_READWRITE | SQLITE_OPEN_CREATE,
                        nullptr) != SQLITE_OK) {
      std::string msg = db ? sqlite3_errmsg(db) : "open failed";
      if (db) sqlite3_close(db);
      db = nullptr;
      throw std::runtime_error("sqlite open " + path + ": " + msg);
    }
    exec("pragma synchronous = off;");
    // rollback is never used (crash safety is tmp+rename at a layer
    // above); skipping the journal removes a second write per page
    exec("pragma journal_mode = off;");
  }

  void check(int rc, const char *what) {
    if (rc != SQLITE_OK)
      throw std::runtime_error(std::string(what) + ": " +
                               sqlite3_errmsg(db));
  }

  void exec(const std::string &sql) {
    char *err = nullptr;
    if (sqlite3_exec(db, sql.c_str(), nullptr, nullptr, &err) !=
        SQLITE_OK) {
      std::string msg = err ? err : "exec failed";
      sqlite3_free(err);
      throw std::runtime_error("sqlite exec: " + msg + " [" + sql + "]");
    }
  }

  // Prepare one INSERT per metric table; opens the write transaction.
  void prepare_inserts(const std::vector<std::string> &sqls) {
    for (const auto &sql : sqls) {
      sqlite3_stmt *st = nullptr;
      check(sqlite3_prepare_v2(db, sql.c_str(), -1, &st, nullptr),
            "prepare");
      inserts.push_back(st);
    }
    exec("BEGIN;");
    in_txn = true;
  }

  void step_row(sqlite3_stmt *st) {
    if (sqlite3_step(st) != SQLITE_DONE)
      throw std::runtime_error(std::string("insert step: ") +
                               sqlite3_errmsg(db));
    sqlite3_reset(st);
  }

  // One row, generic scalars (config/metrics rows and the streaming
  // write_point fallback).
  void insert_row(size_t mi, py::sequence row) {
    sqlite3_stmt *st = inserts.at(mi);
    int col = 1;
    for (auto item : row) {
      py::handle h = item;
      if (h.is_none()) {
        check(sqlite3_bind_null(st, col), "bind null");
      } else if (py::isinstance<py::bool_>(h)) {
        check(sqlite3_bind_int64(st, col, h.cast<bool>() ? 1 : 0),
              "bind bool");
      } else if (py::isinstance<py::int_>(h)) {
        check(sqlite3_bind_int64(st, col, h.cast<long long>()),
              "bind int");
      } else if (py::isinstance<py::float_>(h)) {
        double d = h.cast<double>();
        long long i = static_cast<long long>(d);
        if (static_cast<double>(i) == d)
          check(sqlite3_bind_int64(st, col, i), "bind f->i");
        else
          check(sqlite3_bind_double(st, col, d), "bind double");
      } else {
        std::string s = py::cast<std::string>(h);
        check(sqlite3_bind_text(st, col, s.data(), (int)s.size(),
                                kTransient),
              "bind text");
      }
      col++;
    }
    step_row(st);
  }

  // Hot path: n rows at once, one entry per breakdown column — either
  // an int64 numpy array or a list[str] — plus the float64 values.
  void insert_columnar(size_t mi, py::list cols,
                       py::array_t<double> vals) {
    sqlite3_stmt *st = inserts.at(mi);
    auto v = vals.unchecked<1>();
    const py::ssize_t n = v.shape(0);
    const size_t nc = py::len(cols);

    struct SRef {
      const char *p;
      int len;
    };
    struct Col {
      bool is_int;
      const int64_t *ints = nullptr;
      std::vector<SRef> strs;  // borrowed UTF-8 views into the list's
                               // PyUnicode objects (alive via `cols`)
    };
    std::vector<Col> cv(nc);
    std::vector<py::array_t<int64_t>> keep;  // keep buffers alive
    for (size_t c = 0; c < nc; c++) {
      py::handle h = cols[c];
      if (py::isinstance<py::array>(h)) {
        py::array_t<int64_t> a =
            py::cast<py::array_t<int64_t>>(h);
        if (a.ndim() != 1 || a.shape(0) != n)
          throw std::runtime_error("column shape mismatch");
        keep.push_back(a);
        cv[c].is_int = true;
        cv[c].ints = keep.back().data();
      } else {
        py::list ls = py::cast<py::list>(h);
        if ((py::ssize_t)py::len(ls) != n)
          throw std::runtime_error("column length mismatch");
        cv[c].is_int = false;
        cv[c].strs.reserve(n);
        for (auto s : ls) {
          Py_ssize_t sl = 0;
          const char *sp = PyUnicode_AsUTF8AndSize(s.ptr(), &sl);
          if (sp == nullptr) {
            // rare non-str in a varchar column: coerce the way TEXT
            // affinity would coerce the bound scalar
            PyErr_Clear();
            py::object coerced;
            if (py::isinstance<py::bool_>(s))
              coerced = py::str(s.cast<bool>() ? "1" : "0");
            else
              coerced = py::str(py::reinterpret_borrow<py::object>(s));
            ls[cv[c].strs.size()] = coerced;  // keep alive in the list
            sp = PyUnicode_AsUTF8AndSize(coerced.ptr(), &sl);
            if (sp == nullptr)
              throw std::runtime_error("uncoercible column value");
          }
          cv[c].strs.push_back({sp, (int)sl});
        }
      }
    }
    for (py::ssize_t r = 0; r < n; r++) {
      for (size_t c = 0; c < nc; c++) {
        if (cv[c].is_int) {
          check(sqlite3_bind_int64(st, (int)c + 1, cv[c].ints[r]),
                "bind int col");
        } else {
          const SRef &s = cv[c].strs[r];
          // STATIC: the PyUnicode buffers outlive this call
          check(sqlite3_bind_text(st, (int)c + 1, s.p, s.len,
                                  nullptr),
                "bind str col");
        }
      }
      double d = v(r);
      long long i = static_cast<long long>(d);
      if (static_cast<double>(i) == d)
        check(sqlite3_bind_int64(st, (int)nc + 1, i), "bind value");
      else
        check(sqlite3_bind_double(st, (int)nc + 1, d), "bind value");
      step_row(st);
    }
  }

  void commit() {
    if (in_txn) {
      exec("COMMIT;");
      in_txn = false;
    }
  }

  void close() {
    for (auto *st : inserts) sqlite3_finalize(st);
    inserts.clear();
    if (db) {
      sqlite3_close(db);
      db = nullptr;
    }
  }

  ~CSink() { close(); }
};

// Columnar table extraction for the GPU index-query path (K7): run
// `sql` (the metric table SELECT, value column LAST) and return typed
// columns — f64 arrays for numeric columns, (blob, offs, lens) for
// TEXT columns — at SQLite-C speed instead of per-row Python tuples.
// kinds: one char per non-value column, 'n' = numeric, 's' = string.
py::tuple read_columns(const std::string &path, const std::string &sql,
                       const std::string &kinds) {
  sqlite3 *db = nullptr;
  if (sqlite3_open_v2(path.c_str(), &db, SQLITE_OPEN_READONLY,
                      nullptr) != SQLITE_OK) {
    std::string msg = db ? sqlite3_errmsg(db) : "open failed";
    if (db) sqlite3_close(db);
    throw std::runtime_error("sqlite open " + path + ": " + msg);
  }
  sqlite3_stmt *st = nullptr;
  if (sqlite3_prepare_v2(db, sql.c_str(), -1, &st, nullptr) !=
      SQLITE_OK) {
    std::string msg = sqlite3_errmsg(db);
    sqlite3_close(db);
    throw std::runtime_error("sqlite prepare: " + msg);
  }
  const size_t nc = kinds.size();
  struct SCol {
    std::string blob;
    std::vector<uint32_t> offs, lens;
  };
  std::vector<std::vector<double>> nums(nc);
  std::vector<SCol> strs(nc);
  std::vector<double> values;
  int rc;
  while ((rc = sqlite3_step(st)) == SQLITE_ROW) {
    for (size_t c = 0; c < nc; c++) {
      if (kinds[c] == 'n') {
        nums[c].push_back(sqlite3_column_double(st, (int)c));
      } else {
        // column_text coerces stored integers the way TEXT affinity
        // would; NULL -> empty string (rows are fully populated by
        // the sink)
        const unsigned char *t = sqlite3_column_text(st, (int)c);
        int len = t ? sqlite3_column_bytes(st, (int)c) : 0;
        SCol &sc = strs[c];
        sc.offs.push_back((uint32_t)sc.blob.size());
        sc.lens.push_back((uint32_t)len);
        if (len) sc.blob.append((const char *)t, (size_t)len);
      }
    }
    values.push_back(sqlite3_column_double(st, (int)nc));
  }
  sqlite3_finalize(st);
  sqlite3_close(db);
  if (rc != SQLITE_DONE)
    throw std::runtime_error("sqlite step failed reading " + path);

  py::list cols;
  for (size_t c = 0; c < nc; c++) {
    if (kinds[c] == 'n') {
      py::array_t<double> a((py::ssize_t)nums[c].size());
      std::memcpy(a.mutable_data(), nums[c].data(),
                  nums[c].size() * sizeof(double));
      cols.append(a);
    } else {
      SCol &sc = strs[c];
      py::array_t<uint32_t> offs((py::ssize_t)sc.offs.size());
      py::array_t<uint32_t> lens((py::ssize_t)sc.lens.size());
      std::memcpy(offs.mutable_data(), sc.offs.data(),
                  sc.offs.size() * sizeof(uint32_t));
      std::memcpy(lens.mutable_data(), sc.lens.data(),
                  sc.lens.size() * sizeof(uint32_t));
      cols.append(py::make_tuple(py::bytes(sc.blob), offs, lens));
    }
  }
  py::array_t<double> vals((py::ssize_t)values.size());
  std::memcpy(vals.mutable_data(), values.data(),
              values.size() * sizeof(double));
  return py::make_tuple(cols, vals);
}

}  // namespace

PYBIND11_MODULE(_csink, m) {
  m.doc() = "SQLite C-API index sink (native hot path)";
  py::class_<CSink>(m, "CSink")
      .def(py::init<const std::string &>())
      .def("exec", &CSink::exec)
      .def("prepare_inserts", &CSink::prepare_inserts)
      .def("insert_row", &CSink::insert_row)
      .def("insert_columnar", &CSink::insert_columnar)
      .def("commit", &CSink::commit)
      .def("close", &CSink::close);
  m.def("read_columns", &read_columns,
        "typed columnar extraction of one metric table");
}
