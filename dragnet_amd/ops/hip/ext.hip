// dragnet_amd MI355X scan engine — torch extension entry point.
//
// Single translation unit: includes the gfx950 kernels and exposes the
// host launchers to Python.  Stateless C++: all device state (tables,
// dictionaries, counters, plan buffers) lives in torch tensors owned by
// the Python engine (dragnet_amd/engine/gpu.py).

#include <torch/extension.h>
#include <cstdlib>
#include <c10/hip/HIPStream.h>

#include "scan_kernels.hip"

namespace dn {

namespace {

#define CHECK_GPU(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")

StrDict make_sdict(const torch::Tensor& state, const torch::Tensor& hash,
                   const torch::Tensor& id, const torch::Tensor& off,
                   const torch::Tensor& len, const torch::Tensor& data,
                   const torch::Tensor& data_used,
                   const torch::Tensor& next_id) {
  StrDict d;
  d.state = (uint32_t*)state.data_ptr();
  d.hash = (uint64_t*)hash.data_ptr();
  d.id = (uint32_t*)id.data_ptr();
  d.off = (uint32_t*)off.data_ptr();
  d.len = (uint32_t*)len.data_ptr();
  d.nslots = (uint32_t)state.numel();
  d.data = (uint8_t*)data.data_ptr();
  d.data_cap = (uint32_t)data.numel();
  d.data_used = (uint32_t*)data_used.data_ptr();
  d.next_id = (uint32_t*)next_id.data_ptr();
  return d;
}

NumDict make_ndict(const torch::Tensor& state, const torch::Tensor& bits,
                   const torch::Tensor& id, const torch::Tensor& next_id) {
  NumDict d;
  d.state = (uint32_t*)state.data_ptr();
  d.bits = (uint64_t*)bits.data_ptr();
  d.id = (uint32_t*)id.data_ptr();
  d.nslots = (uint32_t)state.numel();
  d.next_id = (uint32_t*)next_id.data_ptr();
  return d;
}

AggTable make_table(const torch::Tensor& state, const torch::Tensor& keys,
                    const torch::Tensor& count) {
  AggTable t;
  t.state = (uint32_t*)state.data_ptr();
  t.keys = (uint32_t*)keys.data_ptr();
  t.count = (double*)count.data_ptr();
  t.nslots = (uint32_t)state.numel();
  t.partial = nullptr;
  t.prows = 0;
  t.pad_ = 0;
  return t;
}

// Fold a dense partial matrix into count[] with the MFMA column-sum
// reduce (must run before extraction of a dense-mode table).
void dense_reduce_one(const AggTable& T, hipStream_t stream) {
  if (T.partial == nullptr || T.prows == 0) return;
  uint32_t tiles = (T.nslots + 15) / 16;
  uint32_t splits = 32;
  while (splits > 1 && T.prows / splits < 16) splits /= 2;
  uint32_t rows_per_blk = (T.prows + splits - 1) / splits;
  rows_per_blk = (rows_per_blk + 3) & ~3u;  // multiple of the K=4 step
  hipLaunchKernelGGL(mfma_reduce_kernel, dim3(tiles, splits), dim3(64),
                     0, stream, T, rows_per_blk);
  hipError_t err = hipGetLastError();
  TORCH_CHECK(err == hipSuccess, "mfma_reduce launch failed: ",
              hipGetErrorString(err));
}

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

}  // namespace

// Index newlines in a device buffer: fills pos_out (up to its
// capacity) with sorted newline positions and total_out[0] with the
// true count — entirely device-side, no host sync.
void newline_index(torch::Tensor data, int64_t start, int64_t n,
                   torch::Tensor seg_scratch,
                   torch::Tensor pos_out, torch::Tensor total_out) {
  CHECK_GPU(data);
  int64_t span = n - (start & ~(int64_t)15);
  uint32_t nseg = (uint32_t)((span + NL_SEG - 1) / NL_SEG);
  TORCH_CHECK((int64_t)nseg <= seg_scratch.numel(),
              "seg_scratch too small");
  if (span <= 0) {
    total_out.zero_();
    return;
  }
  auto stream = current_stream();
  const uint8_t* d = (const uint8_t*)data.data_ptr();
  uint32_t* segs = (uint32_t*)seg_scratch.data_ptr();
  // wave-per-segment kernels: 256-thread blocks cover 4 segments each
  uint32_t blocks = (nseg + 3) / 4;
  hipLaunchKernelGGL(newline_count_kernel, dim3(blocks), dim3(256), 0,
                     stream, d, (uint32_t)start, (uint32_t)n, segs, nseg);
  hipLaunchKernelGGL(newline_scan_kernel, dim3(1), dim3(1024), 0,
                     stream, segs, nseg, (uint32_t*)total_out.data_ptr());
  hipLaunchKernelGGL(newline_write_kernel, dim3(blocks), dim3(256), 0,
                     stream, d, (uint32_t)start, (uint32_t)n, segs, nseg,
                     (uint32_t*)pos_out.data_ptr(),
                     (uint32_t)pos_out.numel());
  hipError_t err = hipGetLastError();
  TORCH_CHECK(err == hipSuccess, "newline_index launch failed: ",
              hipGetErrorString(err));
}

void scan_chunk(
    torch::Tensor data, torch::Tensor nl_pos, torch::Tensor nlines_dev,
    int64_t first_start,
    torch::Tensor field_sigs, torch::Tensor comp_slot,
    int64_t nf_match, int64_t sig_bloom, int64_t fields_parent_sig,
    torch::Tensor prog_nodes,
    torch::Tensor prog_bounds, torch::Tensor const_meta,
    torch::Tensor const_dvals, torch::Tensor const_bytes,
    torch::Tensor synth_slots, int64_t n_synth,
    torch::Tensor metric_rows, torch::Tensor synth_req,
    torch::Tensor bd_rows, torch::Tensor bd_steps,
    int64_t value_slot, int64_t fields_slot, bool skinner,
    int64_t tile_cap_req,
    torch::Tensor table_descs,
    torch::Tensor sd_state, torch::Tensor sd_hash, torch::Tensor sd_id,
    torch::Tensor sd_off, torch::Tensor sd_len, torch::Tensor sd_data,
    torch::Tensor sd_used, torch::Tensor sd_next,
    torch::Tensor nd_state, torch::Tensor nd_bits, torch::Tensor nd_id,
    torch::Tensor nd_next,
    torch::Tensor counters,
    // wave-transposed staging (xn_slots > 0 selects scan_kernel_x)
    torch::Tensor xdata, torch::Tensor xwave_base,
    torch::Tensor xrec_len, int64_t xn_slots) {
  CHECK_GPU(data);
  CHECK_GPU(nl_pos);
  CHECK_GPU(table_descs);

  ScanArgs A;
  A.data = (const uint8_t*)data.data_ptr();
  A.nl_pos = (const uint32_t*)nl_pos.data_ptr();
  A.nlines_ptr = (const uint32_t*)nlines_dev.data_ptr();
  A.pos_cap = (uint32_t)nl_pos.numel();
  A.first_start = (uint32_t)first_start;

  A.P.field_sigs = (const uint64_t*)field_sigs.data_ptr();
  A.P.nf = (int)field_sigs.numel();
  A.P.comp_slot = (const int32_t*)comp_slot.data_ptr();
  A.P.nf_match = (int)nf_match;
  A.P.sig_bloom = (uint64_t)sig_bloom;
  A.P.fields_parent_sig = (uint64_t)fields_parent_sig;
  A.P.prog_nodes = (const int32_t*)prog_nodes.data_ptr();
  A.P.prog_bounds = (const int32_t*)prog_bounds.data_ptr();
  A.P.const_meta = (const int32_t*)const_meta.data_ptr();
  A.P.const_dvals = (const double*)const_dvals.data_ptr();
  A.P.const_bytes = (const uint8_t*)const_bytes.data_ptr();
  A.P.synth_slots = (const int32_t*)synth_slots.data_ptr();
  A.P.ns = (int)n_synth;
  A.P.metric_rows = (const int32_t*)metric_rows.data_ptr();
  A.P.synth_req = (const int32_t*)synth_req.data_ptr();
  A.P.bd_rows = (const int32_t*)bd_rows.data_ptr();
  A.P.bd_steps = (const double*)bd_steps.data_ptr();
  A.P.nm = (int)(metric_rows.numel() / 8);
  A.P.value_slot = (int)value_slot;
  A.P.fields_slot = (int)fields_slot;

  A.tables = (AggTable*)table_descs.data_ptr();
  A.sdict = make_sdict(sd_state, sd_hash, sd_id, sd_off, sd_len,
                       sd_data, sd_used, sd_next);
  A.ndict = make_ndict(nd_state, nd_bits, nd_id, nd_next);
  A.counters = (unsigned long long*)counters.data_ptr();
  A.data_format_skinner = skinner ? 1 : 0;
  A.xdata = nullptr; A.xwave_base = nullptr; A.xrec_len = nullptr;
  A.xn_slots = 0;
  if (xn_slots > 0) {
    CHECK_GPU(xdata);
    A.xdata = (const uint8_t*)xdata.data_ptr();
    A.xwave_base = (const unsigned long long*)xwave_base.data_ptr();
    A.xrec_len = (const uint32_t*)xrec_len.data_ptr();
    A.xn_slots = (uint32_t)xn_slots;
  }

  int nf = A.P.nf;
  size_t lds = (size_t)nf * BLOCK * (4 + 4 + 1);
  lds = (lds + 15) & ~(size_t)15;
  lds += (size_t)LDS_CACHE * sizeof(LdsCacheEntry);
  lds += (C_GLOBAL_N + (size_t)A.P.nm * CM_N) * 8;
  lds += (size_t)A.P.ns * BLOCK * 8;  // synthetic values (actual count)
  lds += (size_t)6 /*SIG_DEPTH*/ * BLOCK * 8;  // parse sig stack
  lds += 64;  // slack
  TORCH_CHECK(lds <= 160 * 1024, "plan needs too much LDS: ", lds);
  // optional LDS staging tile (parse from LDS); tile_cap_req:
  // 0 = off, -1 = auto (all remaining LDS), >0 = requested bytes.
  // Measured on MI355X: staging costs more occupancy than the
  // global-gather it saves for ~226B records — default off.
  size_t tile_cap = 0;
  if (tile_cap_req != 0) {
    size_t avail = (160 * 1024 - lds - 256) & ~(size_t)15;
    tile_cap = (tile_cap_req < 0) ? avail
                                  : std::min((size_t)tile_cap_req, avail);
    if (tile_cap > 100 * 1024) tile_cap = 100 * 1024;
    if (tile_cap < 24 * 1024) tile_cap = 0;
  }
  A.tile_cap = (uint32_t)tile_cap;
  lds += tile_cap;

  // line count lives on-device; launch a full grid and grid-stride
  uint32_t blocks = (A.pos_cap + BLOCK - 1) / BLOCK;
  if (blocks > 2048) blocks = 2048;

  // default 4 waves/SIMD: the parse kernel is latency-bound and the
  // register-allocator cap measured +17% over the unconstrained build
  const char* mw_env = getenv("DRAGNET_MIN_WAVES");
  int mw = mw_env ? atoi(mw_env) : 4;
  if (A.xn_slots > 0) {
    blocks = (A.xn_slots + BLOCK - 1) / BLOCK;
    if (blocks > 2048) blocks = 2048;
    // r2 sweep on the device-built layout: 64B granules fastest
    // (1457-1462 vs 1436 (32B) vs 1400 (128B) M rec/s)
    const char* xg_env = getenv("DRAGNET_XGRAN");
    int xg = xg_env ? atoi(xg_env) : 64;
    if (xg == 32)
      hipLaunchKernelGGL((scan_kernel_x<4, 5>), dim3(blocks),
                         dim3(BLOCK), lds, current_stream(), A);
    else if (xg == 128)
      hipLaunchKernelGGL((scan_kernel_x<4, 7>), dim3(blocks),
                         dim3(BLOCK), lds, current_stream(), A);
    else if (mw == 2)
      hipLaunchKernelGGL((scan_kernel_x<2, 6>), dim3(blocks),
                         dim3(BLOCK), lds, current_stream(), A);
    else if (mw == 3)
      hipLaunchKernelGGL((scan_kernel_x<3, 6>), dim3(blocks),
                         dim3(BLOCK), lds, current_stream(), A);
    else if (mw == 5)
      hipLaunchKernelGGL((scan_kernel_x<5, 6>), dim3(blocks),
                         dim3(BLOCK), lds, current_stream(), A);
    else
      hipLaunchKernelGGL((scan_kernel_x<4, 6>), dim3(blocks),
                         dim3(BLOCK), lds, current_stream(), A);
    hipError_t xerr = hipGetLastError();
    TORCH_CHECK(xerr == hipSuccess, "scan_kernel_x launch failed: ",
                hipGetErrorString(xerr));
    return;
  }
  if (mw == 2)
    hipLaunchKernelGGL(scan_kernel_mw<2>, dim3(blocks), dim3(BLOCK),
                       lds, current_stream(), A);
  else if (mw == 3)
    hipLaunchKernelGGL(scan_kernel_mw<3>, dim3(blocks), dim3(BLOCK),
                       lds, current_stream(), A);
  else if (mw == 4)
    hipLaunchKernelGGL(scan_kernel_mw<4>, dim3(blocks), dim3(BLOCK),
                       lds, current_stream(), A);
  else if (mw == 5)
    hipLaunchKernelGGL(scan_kernel_mw<5>, dim3(blocks), dim3(BLOCK),
                       lds, current_stream(), A);
  else if (mw == 6)
    hipLaunchKernelGGL(scan_kernel_mw<6>, dim3(blocks), dim3(BLOCK),
                       lds, current_stream(), A);
  else
    hipLaunchKernelGGL(scan_kernel, dim3(blocks), dim3(BLOCK), lds,
                       current_stream(), A);
  hipError_t err = hipGetLastError();
  TORCH_CHECK(err == hipSuccess, "scan_kernel launch failed: ",
              hipGetErrorString(err));
}

// Columnar index-query (K7): build the per-slot column descriptor
// array (CPU bytes; caller copies to GPU) and launch the kernel.
torch::Tensor col_descs_host(std::vector<int64_t> kinds,
                             std::vector<torch::Tensor> nums,
                             std::vector<torch::Tensor> soffs,
                             std::vector<torch::Tensor> slens) {
  int nf = (int)kinds.size();
  auto out = torch::zeros({(long)(nf * sizeof(ColDesc))},
                          torch::dtype(torch::kUInt8));
  ColDesc* d = (ColDesc*)out.data_ptr();
  for (int f = 0; f < nf; f++) {
    d[f].kind = (int32_t)kinds[f];
    d[f].num = kinds[f] == 1 ? (const double*)nums[f].data_ptr()
                             : nullptr;
    d[f].soff = kinds[f] == 2 ? (const uint32_t*)soffs[f].data_ptr()
                              : nullptr;
    d[f].slen = kinds[f] == 2 ? (const uint32_t*)slens[f].data_ptr()
                              : nullptr;
  }
  return out;
}

void columnar_query(
    torch::Tensor col_descs, torch::Tensor blob, torch::Tensor values,
    int64_t nrows,
    torch::Tensor field_sigs, torch::Tensor comp_slot,
    int64_t nf_match, torch::Tensor prog_nodes,
    torch::Tensor prog_bounds, torch::Tensor const_meta,
    torch::Tensor const_dvals, torch::Tensor const_bytes,
    torch::Tensor metric_rows, torch::Tensor synth_req,
    torch::Tensor bd_rows, torch::Tensor bd_steps,
    torch::Tensor table_descs,
    torch::Tensor sd_state, torch::Tensor sd_hash, torch::Tensor sd_id,
    torch::Tensor sd_off, torch::Tensor sd_len, torch::Tensor sd_data,
    torch::Tensor sd_used, torch::Tensor sd_next,
    torch::Tensor nd_state, torch::Tensor nd_bits, torch::Tensor nd_id,
    torch::Tensor nd_next, torch::Tensor counters) {
  CHECK_GPU(col_descs);
  CHECK_GPU(blob);
  ColArgs A;
  A.nrows = (uint32_t)nrows;
  A.values = (const double*)values.data_ptr();
  A.blob = (const uint8_t*)blob.data_ptr();
  A.cols = (const ColDesc*)col_descs.data_ptr();
  A.P.field_sigs = (const uint64_t*)field_sigs.data_ptr();
  A.P.nf = (int)field_sigs.numel();
  A.P.comp_slot = (const int32_t*)comp_slot.data_ptr();
  A.P.nf_match = (int)nf_match;
  A.P.sig_bloom = 0;
  A.P.fields_parent_sig = 0;
  A.P.prog_nodes = (const int32_t*)prog_nodes.data_ptr();
  A.P.prog_bounds = (const int32_t*)prog_bounds.data_ptr();
  A.P.const_meta = (const int32_t*)const_meta.data_ptr();
  A.P.const_dvals = (const double*)const_dvals.data_ptr();
  A.P.const_bytes = (const uint8_t*)const_bytes.data_ptr();
  A.P.synth_slots = (const int32_t*)synth_req.data_ptr();  // unused
  A.P.ns = 0;
  A.P.metric_rows = (const int32_t*)metric_rows.data_ptr();
  A.P.synth_req = (const int32_t*)synth_req.data_ptr();
  A.P.bd_rows = (const int32_t*)bd_rows.data_ptr();
  A.P.bd_steps = (const double*)bd_steps.data_ptr();
  A.P.nm = (int)(metric_rows.numel() / 8);
  A.P.value_slot = -1;
  A.P.fields_slot = -1;
  A.tables = (AggTable*)table_descs.data_ptr();
  A.sdict = make_sdict(sd_state, sd_hash, sd_id, sd_off, sd_len,
                       sd_data, sd_used, sd_next);
  A.ndict = make_ndict(nd_state, nd_bits, nd_id, nd_next);
  A.counters = (unsigned long long*)counters.data_ptr();

  int nf = A.P.nf;
  size_t lds = (size_t)nf * BLOCK * (4 + 4 + 1);
  lds = (lds + 15) & ~(size_t)15;
  lds += (size_t)LDS_CACHE * sizeof(LdsCacheEntry);
  lds += (C_GLOBAL_N + (size_t)A.P.nm * CM_N) * 8;
  lds += 64;
  TORCH_CHECK(lds <= 160 * 1024, "columnar plan needs too much LDS");
  uint32_t blocks = (uint32_t)((nrows + BLOCK - 1) / BLOCK);
  if (blocks > 2048) blocks = 2048;
  if (blocks == 0) return;
  hipLaunchKernelGGL(columnar_query_kernel, dim3(blocks), dim3(BLOCK),
                     lds, current_stream(), A);
  hipError_t err = hipGetLastError();
  TORCH_CHECK(err == hipSuccess, "columnar_query launch failed: ",
              hipGetErrorString(err));
}

// Device-side wave-transpose: scatter length-sorted records into the
// granule-interleaved layout (see xpose_build_kernel).
void xpose_build(torch::Tensor data, torch::Tensor sstart,
                 torch::Tensor slen, torch::Tensor wbase,
                 int64_t n_slots, int64_t gran_log, torch::Tensor xb) {
  CHECK_GPU(data);
  CHECK_GPU(xb);
  uint32_t nw = (uint32_t)(n_slots >> 6);
  TORCH_CHECK(wbase.numel() >= (long)nw + 1, "wbase needs nw+1 rows");
  uint32_t blocks = (nw + 3) / 4;  // 4 waves per 256-thread block
  hipLaunchKernelGGL(xpose_build_kernel, dim3(blocks), dim3(256), 0,
                     current_stream(),
                     (const uint8_t*)data.data_ptr(),
                     (const uint32_t*)sstart.data_ptr(),
                     (const uint32_t*)slen.data_ptr(),
                     (const unsigned long long*)wbase.data_ptr(),
                     (uint32_t)n_slots, (int)gran_log,
                     (uint8_t*)xb.data_ptr());
  hipError_t err = hipGetLastError();
  TORCH_CHECK(err == hipSuccess, "xpose_build launch failed: ",
              hipGetErrorString(err));
}

// One-call zeroing of all scan state: a streaming step otherwise
// issues ~10 separate torch .zero_() dispatches whose python+dispatch
// overhead shows at 5 ms/step.
void scan_reset(std::vector<torch::Tensor> states,
                std::vector<torch::Tensor> counts,
                torch::Tensor sd_state, torch::Tensor sd_used,
                torch::Tensor sd_next, torch::Tensor nd_state,
                torch::Tensor nd_next, torch::Tensor counters,
                std::vector<torch::Tensor> partials) {
  hipStream_t st = current_stream();
  auto z = [&](torch::Tensor& t) {
    hipError_t e = hipMemsetAsync(t.data_ptr(), 0,
                                  (size_t)t.numel() * t.element_size(),
                                  st);
    TORCH_CHECK(e == hipSuccess, "scan_reset memset failed: ",
                hipGetErrorString(e));
  };
  for (auto& t : states) z(t);
  for (auto& t : counts) z(t);
  for (auto& t : partials) z(t);
  z(sd_state); z(sd_used); z(sd_next);
  z(nd_state); z(nd_next); z(counters);
}

// Build the device-side AggTable descriptor array from per-metric
// tensors.  Returns a CPU byte tensor; the caller copies it to the GPU.
// partials (optional, same length): [prows, nslots] f64 dense partial
// matrices — enables the dense-accumulation + MFMA-reduce path.
torch::Tensor agg_descs_host(std::vector<torch::Tensor> states,
                             std::vector<torch::Tensor> keys,
                             std::vector<torch::Tensor> counts,
                             std::vector<torch::Tensor> partials) {
  int nm = (int)states.size();
  auto out = torch::empty({(long)(nm * sizeof(AggTable))},
                          torch::dtype(torch::kUInt8));
  AggTable* descs = (AggTable*)out.data_ptr();
  for (int m = 0; m < nm; m++) {
    descs[m] = make_table(states[m], keys[m], counts[m]);
    if (!partials.empty()) {
      TORCH_CHECK(partials[m].size(1) == (long)descs[m].nslots,
                  "partial width != nslots");
      descs[m].partial = (double*)partials[m].data_ptr();
      descs[m].prows = (uint32_t)partials[m].size(0);
    }
  }
  return out;
}

// Compact one metric's table: returns (keys int32 [n, MAX_KEY],
// counts f64 [n]).
std::vector<torch::Tensor> extract_agg(torch::Tensor state,
                                       torch::Tensor keys,
                                       torch::Tensor count,
                                       int64_t max_out) {
  CHECK_GPU(state);
  AggTable T = make_table(state, keys, count);
  auto i32 = torch::dtype(torch::kInt32).device(state.device());
  auto f64 = torch::dtype(torch::kFloat64).device(state.device());
  auto out_keys = torch::zeros({max_out, (long)MAX_KEY}, i32);
  auto out_counts = torch::zeros({max_out}, f64);
  auto out_n = torch::zeros({1}, i32);
  uint32_t blocks = (T.nslots + 255) / 256;
  hipLaunchKernelGGL(extract_agg_kernel, dim3(blocks), dim3(256), 0,
                     current_stream(), T, (uint32_t*)out_keys.data_ptr(),
                     (double*)out_counts.data_ptr(),
                     (uint32_t*)out_n.data_ptr(), (uint32_t)max_out);
  hipError_t err = hipGetLastError();
  TORCH_CHECK(err == hipSuccess, "extract_agg launch failed: ",
              hipGetErrorString(err));
  int64_t n = out_n.cpu().item<int32_t>();
  if (n > max_out) n = max_out;
  return {out_keys.narrow(0, 0, n), out_counts.narrow(0, 0, n)};
}

// Async variant: no host sync — returns the full-capacity buffers plus
// the device count; the caller slices after its own (later) sync.
std::vector<torch::Tensor> extract_agg_async(torch::Tensor state,
                                             torch::Tensor keys,
                                             torch::Tensor count,
                                             int64_t max_out) {
  CHECK_GPU(state);
  AggTable T = make_table(state, keys, count);
  auto i32 = torch::dtype(torch::kInt32).device(state.device());
  auto f64 = torch::dtype(torch::kFloat64).device(state.device());
  auto out_keys = torch::zeros({max_out, (long)MAX_KEY}, i32);
  auto out_counts = torch::zeros({max_out}, f64);
  auto out_n = torch::zeros({1}, i32);
  uint32_t blocks = (T.nslots + 255) / 256;
  hipLaunchKernelGGL(extract_agg_kernel, dim3(blocks), dim3(256), 0,
                     current_stream(), T, (uint32_t*)out_keys.data_ptr(),
                     (double*)out_counts.data_ptr(),
                     (uint32_t*)out_n.data_ptr(), (uint32_t)max_out);
  hipError_t err = hipGetLastError();
  TORCH_CHECK(err == hipSuccess, "extract_agg_async launch failed: ",
              hipGetErrorString(err));
  return {out_keys, out_counts, out_n};
}

std::vector<torch::Tensor> extract_strdict(
    torch::Tensor sd_state, torch::Tensor sd_hash, torch::Tensor sd_id,
    torch::Tensor sd_off, torch::Tensor sd_len, torch::Tensor sd_data,
    torch::Tensor sd_used, torch::Tensor sd_next, int64_t n_ids) {
  StrDict D = make_sdict(sd_state, sd_hash, sd_id, sd_off, sd_len,
                         sd_data, sd_used, sd_next);
  auto i32 = torch::dtype(torch::kInt32).device(sd_state.device());
  auto out_off = torch::zeros({n_ids > 0 ? n_ids : 1}, i32);
  auto out_len = torch::zeros({n_ids > 0 ? n_ids : 1}, i32);
  if (n_ids > 0) {
    uint32_t blocks = (D.nslots + 255) / 256;
    hipLaunchKernelGGL(extract_strdict_kernel, dim3(blocks), dim3(256), 0,
                       current_stream(), D, (uint32_t*)out_off.data_ptr(),
                       (uint32_t*)out_len.data_ptr(), (uint32_t)n_ids);
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "extract_strdict launch failed: ",
                hipGetErrorString(err));
  }
  return {out_off, out_len};
}

torch::Tensor extract_numdict(torch::Tensor nd_state,
                              torch::Tensor nd_bits, torch::Tensor nd_id,
                              torch::Tensor nd_next, int64_t n_ids) {
  NumDict D = make_ndict(nd_state, nd_bits, nd_id, nd_next);
  auto out = torch::zeros(
      {n_ids > 0 ? n_ids : 1},
      torch::dtype(torch::kFloat64).device(nd_state.device()));
  if (n_ids > 0) {
    uint32_t blocks = (D.nslots + 255) / 256;
    hipLaunchKernelGGL(extract_numdict_kernel, dim3(blocks), dim3(256), 0,
                       current_stream(), D, (double*)out.data_ptr(),
                       (uint32_t)n_ids);
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "extract_numdict launch failed: ",
                hipGetErrorString(err));
  }
  return out;
}

// One-call snapshot + extraction for the pipelined finalize
// (gpu._ScanContext.extract_async): clones of the counters and
// dictionary cursors/blob, the dictionary extractions, and the
// no-sync per-metric aggregate extraction, enqueued on the current
// stream in one binding call instead of ~10 python dispatches.
// Returns [cnt, n_str, n_num, used, blob, str_off, str_len, numbers,
// then (keys, counts, n) per metric].
std::vector<torch::Tensor> extract_all(
    std::vector<torch::Tensor> states, std::vector<torch::Tensor> keys,
    std::vector<torch::Tensor> counts, int64_t max_out,
    torch::Tensor sd_state, torch::Tensor sd_hash, torch::Tensor sd_id,
    torch::Tensor sd_off, torch::Tensor sd_len, torch::Tensor sd_data,
    torch::Tensor sd_used, torch::Tensor sd_next,
    torch::Tensor nd_state, torch::Tensor nd_bits, torch::Tensor nd_id,
    torch::Tensor nd_next, torch::Tensor counters, int64_t dict_slots,
    std::vector<torch::Tensor> partials) {
  // dense mode: fold the per-workgroup partial matrices into count[]
  // (MFMA column-sum) before the table extraction below reads them
  for (size_t m = 0; m < partials.size(); m++) {
    AggTable T = make_table(states[m], keys[m], counts[m]);
    T.partial = (double*)partials[m].data_ptr();
    T.prows = (uint32_t)partials[m].size(0);
    dense_reduce_one(T, current_stream());
  }
  std::vector<torch::Tensor> out;
  out.push_back(counters.clone());
  out.push_back(sd_next.clone());
  out.push_back(nd_next.clone());
  out.push_back(sd_used.clone());
  // the dictionary byte blob is bump-allocated from 0 each scan, so
  // the NEXT step overwrites it — snapshot now
  out.push_back(sd_data.clone());
  auto so = extract_strdict(sd_state, sd_hash, sd_id, sd_off, sd_len,
                            sd_data, sd_used, sd_next, dict_slots);
  out.push_back(so[0]);
  out.push_back(so[1]);
  out.push_back(extract_numdict(nd_state, nd_bits, nd_id, nd_next,
                                dict_slots));
  for (size_t m = 0; m < states.size(); m++) {
    auto a = extract_agg_async(states[m], keys[m], counts[m], max_out);
    out.insert(out.end(), a.begin(), a.end());
  }
  return out;
}

}  // namespace dn

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "dragnet_amd MI355X scan engine (gfx950 HIP kernels)";
  m.def("scan_chunk", &dn::scan_chunk, "fused NDJSON scan over one chunk");
  m.def("newline_index", &dn::newline_index, "device-side newline index");
  m.def("xpose_build", &dn::xpose_build, "device-side wave transpose");
  m.def("col_descs_host", &dn::col_descs_host);
  m.def("columnar_query", &dn::columnar_query,
        "K7: query over uploaded index columns");
  m.def("agg_descs_host", &dn::agg_descs_host);
  m.def("extract_agg", &dn::extract_agg);
  m.def("extract_agg_async", &dn::extract_agg_async);
  m.def("extract_all", &dn::extract_all);
  m.def("scan_reset", &dn::scan_reset);
  m.def("extract_strdict", &dn::extract_strdict);
  m.def("extract_numdict", &dn::extract_numdict);
  m.attr("MAX_KEY") = (int)dn::MAX_KEY;
  m.attr("MAX_FIELDS") = (int)dn::MAX_FIELDS;
}
