// Shared device-plan layout constants for the dragnet_amd HIP engine.
// Mirrored by dragnet_amd/engine/plan.py — keep in sync (tested by
// tests/test_plan.py::test_constants_match).
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

namespace dn {

// ---- limits ----
constexpr int MAX_FIELDS = 24;   // distinct dotted paths per plan
constexpr int MAX_DEPTH  = 64;   // validated JSON nesting depth
constexpr int SLOT_DEPTH = 12;   // capture-slot tracking depth (deeper
                                 // containers are inside arrays: never captured)
constexpr int MAX_KEY    = 8;    // breakdown columns per metric
constexpr int MAX_SYNTH  = 8;    // synthetic date fields per plan
constexpr int PRED_STACK = 8;    // and/or nesting depth

// ---- predicate bytecode ops (plan.py OP_*) ----
enum PredOp : int32_t {
  OP_AND = 0, OP_OR = 1,
  OP_EQ = 2, OP_NE = 3, OP_LT = 4, OP_LE = 5, OP_GT = 6, OP_GE = 7,
  OP_TRUE = 8,
};

// ---- predicate const kinds ----
enum ConstKind : int32_t { CONST_NUM = 0, CONST_STR = 1, CONST_NULL = 2 };

// ---- bucketizers ----
enum BucketKind : int32_t { BUCKET_NONE = 0, BUCKET_P2 = 1, BUCKET_LIN = 2 };

// ---- extracted field value types ----
enum FieldType : uint8_t {
  T_MISSING = 0, T_NULL = 1, T_FALSE = 2, T_TRUE = 3,
  T_NUM = 4, T_STR = 5, T_OBJ = 6, T_ARR = 7,
};

// ---- group-key column codes (u32; plan.py TAG_*) ----
// tag in bits 30..31, payload in bits 0..29
constexpr uint32_t TAG_ORD     = 0u;
constexpr uint32_t TAG_STR     = 1u;
constexpr uint32_t TAG_NUM     = 2u;
constexpr uint32_t TAG_SPECIAL = 3u;
constexpr uint32_t ORD_BIAS    = 1u << 29;
constexpr uint32_t SPECIAL_NULL = 0, SPECIAL_UNDEF = 1,
                   SPECIAL_TRUE = 2, SPECIAL_FALSE = 3,
                   SPECIAL_OBJECT = 4, SPECIAL_ARRAY = 5,
                   SPECIAL_ARRJSON = 6;  // | (string-dict id << 3)
constexpr uint32_t EMPTY_CODE = 0xFFFFFFFFu;

inline __device__ __host__ uint32_t make_code(uint32_t tag, uint32_t val) {
  return (tag << 30) | (val & 0x3FFFFFFFu);
}

// ---- global counter slots (u64 array; engine/gpu.py COUNTER_*) ----
enum CounterSlot : int {
  C_LINES = 0,          // lines seen (json parser ninputs)
  C_INVALID_JSON = 1,   // parse failures
  C_PARSED = 2,         // json parser noutputs
  C_DS_FILTERED = 3,    // datasource filter nfilteredout
  C_DS_FAILEDEVAL = 4,  // datasource filter nfailedeval
  C_OVERFLOW = 5,       // hash/dict table overflow flag (abort+regrow)
  C_GLOBAL_N = 8,
  // per metric, C_GLOBAL_N + metric * CM_N + slot:
  CM_FILTER_IN = 0, CM_FILTERED = 1, CM_FAILEDEVAL = 2,
  CM_UNDEF = 3, CM_BADDATE = 4, CM_TIME_OUT = 5,
  CM_AGG_IN = 6, CM_NONNUMERIC = 7,
  CM_N = 8,
};

// ---- hash-table slot states ----
constexpr uint32_t SLOT_EMPTY = 0, SLOT_CLAIMED = 1, SLOT_READY = 2;

// Aggregation table: structure-of-arrays in one buffer.
//   state: u32[nslots]
//   keys:  u32[nslots][MAX_KEY]
//   count: double[nslots]
//
// Dense-accumulation mode (partial != nullptr): the table is only a
// slot DIRECTORY (insert assigns a stable slot, no count atomics);
// each workgroup accumulates into its own row of the dense
// partial[prows][nslots] matrix and the MFMA column-sum reduce
// (mfma_reduce_kernel, v_mfma_f64_16x16x4_f64) folds the matrix into
// count[] before extraction — the matrix-shaped accumulation of
// SURVEY §7.7 / the BASELINE north star.
struct AggTable {
  uint32_t* state;
  uint32_t* keys;   // nslots * MAX_KEY
  double*   count;
  uint32_t  nslots; // power of two
  double*   partial;  // [prows][nslots] per-workgroup rows (or null)
  uint32_t  prows;
  uint32_t  pad_;
};

// String-intern table (shared by all metrics):
//   state: u32[nslots]; hash: u64[nslots]; id: u32[nslots];
//   off:   u32[nslots]; len: u32[nslots]
// plus a bump-allocated byte buffer for string payloads.
struct StrDict {
  uint32_t* state;
  uint64_t* hash;
  uint32_t* id;
  uint32_t* off;
  uint32_t* len;
  uint32_t  nslots;
  uint8_t*  data;      // payload bytes (8B-granule stored)
  uint32_t  data_cap;
  uint32_t* data_used; // bump pointer
  uint32_t* next_id;
};

// Number-intern table (doubles; shared):
struct NumDict {
  uint32_t* state;
  uint64_t* bits;   // canonical double bits (-0 -> +0)
  uint32_t* id;
  uint32_t  nslots;
  uint32_t* next_id;
};

// ---- kernel argument structs (shared between kernels and bindings) ----

struct PlanView {
  const uint64_t* field_sigs;   // [nf] (companions carry poison sigs)
  int nf;                       // total physical slots (LDS sizing)
  const int32_t* comp_slot;     // [nf_match] literal-dotted companion
                                // slot per primary field, or -1
  int nf_match;                 // primary field count (agg readout)
  uint64_t sig_bloom;           // OR of 1<<(sig&63): cheap pre-filter
                                // so non-matching keys skip the loop
  const int32_t* prog_nodes;    // [n_nodes][4]
  const int32_t* prog_bounds;   // [n_progs][2]
  const int32_t* const_meta;    // [nc][6] kind, off, len, dvalid, off2, len2
  const double*  const_dvals;   // [nc]
  const uint8_t* const_bytes;
  const int32_t* synth_slots;   // [ns] source field slot
  int ns;
  const int32_t* metric_rows;   // [nm][8]
  const int32_t* synth_req;     // flat
  const int32_t* bd_rows;       // [nb][4]
  const double*  bd_steps;      // [nb]
  int nm;
  int value_slot;               // json-skinner weight slot (-1 if json)
  int fields_slot;              // json-skinner "fields" presence (-1)
  uint64_t fields_parent_sig;   // path_sig("fields") in skinner mode
                                // (fields-root literal-dotted detect)
};

struct ScanArgs {
  const uint8_t* data;
  const uint32_t* nl_pos;      // sorted newline positions
  const uint32_t* nlines_ptr;  // device count (no host sync needed)
  uint32_t pos_cap;            // capacity of nl_pos
  uint32_t first_start;        // byte offset of the first line
  uint32_t tile_cap;           // LDS staging tile bytes (0 = off)
  PlanView P;
  AggTable* tables;       // [nm]
  StrDict sdict;
  NumDict ndict;
  unsigned long long* counters;
  int data_format_skinner;
  // wave-transposed staging (scan_kernel_x; null for linear scans)
  const uint8_t* xdata;              // transposed pool
  const unsigned long long* xwave_base;  // [nwaves] byte base per wave
  const uint32_t* xrec_len;          // [n_slots] record len; ~0u = pad
  uint32_t xn_slots;                 // slots = records rounded to 64
};

}  // namespace dn
