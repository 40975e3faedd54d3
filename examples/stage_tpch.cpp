// stage_tpch.cpp — standalone C++ host driving the GPU stage interpreter
// through the C ABI alone: no Python anywhere on the path.  This is the
// shape of the Rust GpuExecutionEngine binding: decode a
// task plan, serialise it to the stage-plan JSON, ONE bg_execute_stage call
// per task.
//
//   ./stage_tpch q6 [rows]   — TPC-H q6 (scan+filter+aggregate) as a stage
//   ./stage_tpch q3 [sf]     — q3 (2 joins + 3-col group-by + top-10)
//   ./stage_tpch shuffle [rows] [k] — a hash-repartition shuffle-write
//                               stage producing the reference's
//                               data.arrow + .index byte format
//
// Data is synthesized directly in HBM (bg_fill_rand; deterministic), the
// same distributions as the Python perf harness.  Builds via
// __graft_entry__.build().
#include <chrono>
#include <cinttypes>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>

#include "ballista_gpu.h"

#define CHK(call)                                                     \
  do {                                                                \
    int rc_ = (call);                                                 \
    if (rc_ != 0) {                                                   \
      fprintf(stderr, "FATAL %s -> %d: %s\n", #call, rc_,             \
              bg_last_error());                                       \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

static void* dmalloc(uint64_t bytes) {
  void* p = nullptr;
  CHK(bg_malloc(bytes, &p));
  return p;
}

static double now_ms() {
  return std::chrono::duration<double, std::milli>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

// Arrow IPC schema message bytes for the shuffle-write schema
// (k int64, v dec128(15,2), d date32), generated once with pyarrow and
// frozen here — the Rust host produces the same bytes with arrow-ipc's
// IpcDataGenerator::schema_to_bytes (the wire format is specified by
// format/Schema.fbs, stable across implementations).
static const char* SHUFFLE_SCHEMA_HEX_FILE = "examples/shuffle_schema.hex";

static std::string read_schema_hex() {
  FILE* f = fopen(SHUFFLE_SCHEMA_HEX_FILE, "rb");
  if (!f) {
    fprintf(stderr,
            "missing %s (generate: python -c \"from "
            "datafusion_ballista_amd import stage; import pyarrow as pa; "
            "print(stage.schema_msg_hex(pa.schema([('k', pa.int64()), "
            "('v', pa.decimal128(15,2)), ('d', pa.date32())])))\" > %s)\n",
            SHUFFLE_SCHEMA_HEX_FILE, SHUFFLE_SCHEMA_HEX_FILE);
    exit(1);
  }
  std::string hex;
  char c;
  while (fread(&c, 1, 1, f) == 1)
    if (c != '\n' && c != '\r' && c != ' ') hex += c;
  fclose(f);
  return hex;
}

static void reg_lineitem(int64_t n) {
  void* okey = dmalloc((uint64_t)n * 8);
  void* ship = dmalloc((uint64_t)n * 4);
  void* qty = dmalloc((uint64_t)n * 16);
  void* price = dmalloc((uint64_t)n * 16);
  void* disc = dmalloc((uint64_t)n * 16);
  CHK(bg_fill_rand(okey, n, 1, 1, n / 4 + 2, 0));
  CHK(bg_fill_rand(ship, n, 2, 8036, 10561, 1));
  CHK(bg_fill_rand(qty, n, 3, 1, 51, 2));  // unscaled 1..50; pred uses 24
  CHK(bg_fill_rand(price, n, 4, 90000, 10495100, 2));
  CHK(bg_fill_rand(disc, n, 5, 0, 11, 2));
  bg_column cols[5] = {};
  const char* names[5] = {"l_orderkey", "l_shipdate", "l_quantity",
                          "l_extendedprice", "l_discount"};
  int dts[5] = {BG_DT_INT64, BG_DT_DATE32, BG_DT_DECIMAL128,
                BG_DT_DECIMAL128, BG_DT_DECIMAL128};
  void* bufs[5] = {okey, ship, qty, price, disc};
  for (int i = 0; i < 5; ++i) {
    cols[i].dtype = dts[i];
    cols[i].precision = 15;
    cols[i].scale = 2;
    cols[i].d_data = bufs[i];
    cols[i].len = n;
  }
  CHK(bg_stage_register_table("lineitem", cols, names, 5, n));
}

static const char* LI_SCHEMA =
    "[{\"name\":\"l_orderkey\",\"dtype\":\"int64\"},"
    "{\"name\":\"l_shipdate\",\"dtype\":\"date32\"},"
    "{\"name\":\"l_quantity\",\"dtype\":\"decimal128\",\"precision\":15,"
    "\"scale\":2},"
    "{\"name\":\"l_extendedprice\",\"dtype\":\"decimal128\",\"precision\":15,"
    "\"scale\":2},"
    "{\"name\":\"l_discount\",\"dtype\":\"decimal128\",\"precision\":15,"
    "\"scale\":2}]";

static int run_q6(int64_t n) {
  reg_lineitem(n);
  // NOTE: l_quantity synthesized UNSCALED (1..50); predicate uses 24
  std::string plan = std::string(
      "{\"job_id\":\"cxx\",\"stage_id\":6,\"task_id\":0,"
      "\"work_dir\":\"/tmp/cxx\",\"plan\":{\"op\":\"collect\",\"input\":{"
      "\"op\":\"hash_aggregate\",\"mode\":\"single\",\"group_by\":[],"
      "\"aggs\":[{\"fn\":\"sum\",\"as\":\"revenue\",\"expr\":{\"mul\":"
      "[{\"col\":\"l_extendedprice\"},{\"col\":\"l_discount\"}]}},"
      "{\"fn\":\"count\",\"as\":\"cnt\"}],"
      "\"input\":{\"op\":\"filter\",\"predicates\":["
      "{\"col\":\"l_shipdate\",\"cmp\":\"ge_lt\",\"lo\":8766,\"hi\":9131},"
      "{\"col\":\"l_discount\",\"cmp\":\"between\",\"lo\":5,\"hi\":7},"
      "{\"col\":\"l_quantity\",\"cmp\":\"lt\",\"hi\":24}],"
      "\"input\":{\"op\":\"scan\",\"schema\":") + LI_SCHEMA +
      ",\"source\":{\"kind\":\"device\",\"table\":\"lineitem\"},"
      "\"projection\":[\"l_shipdate\",\"l_discount\",\"l_quantity\","
      "\"l_extendedprice\"]}}}}}";
  char* out = nullptr;
  CHK(bg_execute_stage(plan.c_str(), &out));  // warmup
  printf("q6 warmup: %s\n", out);
  bg_stage_free(out);
  double best = 1e30;
  for (int it = 0; it < 10; ++it) {
    double t0 = now_ms();
    CHK(bg_execute_stage(plan.c_str(), &out));
    double dt = now_ms() - t0;
    if (dt < best) best = dt;
    bg_stage_free(out);
  }
  double kern = bg_last_kernel_ms();
  printf("q6 stage (C++, no Python): rows=%" PRId64
         " best_wall_ms=%.3f kernel_ms=%.3f overhead=%.3fx\n",
         n, best, kern, best / kern);
  return 0;
}

static int run_shuffle(int64_t n, int k) {
  void* kcol = dmalloc((uint64_t)n * 8);
  void* vcol = dmalloc((uint64_t)n * 16);
  void* dcol = dmalloc((uint64_t)n * 4);
  CHK(bg_fill_rand(kcol, n, 11, -((int64_t)1 << 60), (int64_t)1 << 60, 0));
  CHK(bg_fill_rand(vcol, n, 12, -1000000000, 1000000000, 2));
  CHK(bg_fill_rand(dcol, n, 13, 8000, 11000, 1));
  bg_column cols[3] = {};
  const char* names[3] = {"k", "v", "d"};
  cols[0].dtype = BG_DT_INT64;
  cols[0].d_data = kcol;
  cols[0].len = n;
  cols[1].dtype = BG_DT_DECIMAL128;
  cols[1].precision = 15;
  cols[1].scale = 2;
  cols[1].d_data = vcol;
  cols[1].len = n;
  cols[2].dtype = BG_DT_DATE32;
  cols[2].d_data = dcol;
  cols[2].len = n;
  CHK(bg_stage_register_table("t", cols, names, 3, n));

  std::string schema_hex = read_schema_hex();
  char kbuf[16];
  snprintf(kbuf, sizeof(kbuf), "%d", k);
  std::string plan = std::string(
      "{\"job_id\":\"cxx\",\"stage_id\":2,\"task_id\":0,"
      "\"work_dir\":\"/tmp/cxx-shuffle\",\"schema_msg_hex\":\"") +
      schema_hex +
      "\",\"plan\":{\"op\":\"sort_shuffle_write\",\"k\":" + kbuf +
      ",\"keys\":[{\"col\":\"k\"}],\"input\":{\"op\":\"scan\",\"schema\":"
      "[{\"name\":\"k\",\"dtype\":\"int64\"},"
      "{\"name\":\"v\",\"dtype\":\"decimal128\",\"precision\":15,"
      "\"scale\":2},{\"name\":\"d\",\"dtype\":\"date32\"}],"
      "\"source\":{\"kind\":\"device\",\"table\":\"t\"}}}}";
  char* out = nullptr;
  double t0 = now_ms();
  CHK(bg_execute_stage(plan.c_str(), &out));
  double cold = now_ms() - t0;
  bg_stage_free(out);
  t0 = now_ms();
  CHK(bg_execute_stage(plan.c_str(), &out));
  double warm = now_ms() - t0;
  printf("shuffle-write stage (C++): rows=%" PRId64 " k=%d cold_ms=%.1f "
         "warm_ms=%.1f\nresult=%s\n", n, k, cold, warm, out);
  bg_stage_free(out);
  return 0;
}

static int run_q3(int sf) {
  const int64_t ncust = 150000LL * sf * 10;
  const int64_t nord = 1500000LL * sf * 10;
  const int64_t nli = 6000000LL * sf * 10;  // sf10 units to bound runtime
  reg_lineitem(nli);
  void* ck = dmalloc((uint64_t)ncust * 8);
  void* seg = dmalloc((uint64_t)ncust);
  CHK(bg_fill_rand(ck, ncust, 21, 1, ncust + 1, 3));  // dense 1..n
  {  // seg as u8: fill int32 into a scratch then narrow? use fill+copy:
    void* seg32 = dmalloc((uint64_t)ncust * 4);
    CHK(bg_fill_rand(seg32, ncust, 22, 0, 5, 1));
    // narrow on device via gather trick is overkill; interpret int32
    // lanes' low bytes won't match layout — simplest: host round-trip
    int32_t* h = (int32_t*)malloc((size_t)ncust * 4);
    CHK(bg_memcpy_d2h(h, seg32, (uint64_t)ncust * 4));
    uint8_t* h8 = (uint8_t*)malloc((size_t)ncust);
    for (int64_t i = 0; i < ncust; ++i) h8[i] = (uint8_t)h[i];
    CHK(bg_memcpy_h2d(seg, h8, (uint64_t)ncust));
    free(h);
    free(h8);
    CHK(bg_free(seg32));
  }
  void* ok = dmalloc((uint64_t)nord * 8);
  void* oc = dmalloc((uint64_t)nord * 8);
  void* od = dmalloc((uint64_t)nord * 4);
  void* op = dmalloc((uint64_t)nord * 4);
  CHK(bg_fill_rand(ok, nord, 23, 1, nord + 1, 3));
  CHK(bg_fill_rand(oc, nord, 24, 1, ncust + 1, 0));
  CHK(bg_fill_rand(od, nord, 25, 8036, 10561, 1));
  CHK(bg_fill_rand(op, nord, 26, 0, 3, 1));
  {
    bg_column c[2] = {};
    const char* nm[2] = {"c_custkey", "c_mktsegment"};
    c[0].dtype = BG_DT_INT64;
    c[0].d_data = ck;
    c[0].len = ncust;
    c[1].dtype = BG_DT_DICT8;
    c[1].d_data = seg;
    c[1].len = ncust;
    CHK(bg_stage_register_table("customer", c, nm, 2, ncust));
  }
  {
    bg_column c[4] = {};
    const char* nm[4] = {"o_orderkey", "o_custkey", "o_orderdate",
                         "o_shippriority"};
    c[0].dtype = BG_DT_INT64;
    c[0].d_data = ok;
    c[0].len = nord;
    c[1].dtype = BG_DT_INT64;
    c[1].d_data = oc;
    c[1].len = nord;
    c[2].dtype = BG_DT_DATE32;
    c[2].d_data = od;
    c[2].len = nord;
    c[3].dtype = BG_DT_INT32;
    c[3].d_data = op;
    c[3].len = nord;
    CHK(bg_stage_register_table("orders", c, nm, 4, nord));
  }
  std::string plan = std::string(
      "{\"job_id\":\"cxx\",\"stage_id\":3,\"task_id\":0,"
      "\"work_dir\":\"/tmp/cxx\",\"plan\":{\"op\":\"collect\",\"limit\":10,"
      "\"input\":{\"op\":\"sort\",\"keys\":[{\"col\":\"revenue\","
      "\"desc\":true},{\"col\":\"o_orderdate\",\"desc\":false}],"
      "\"limit\":10,\"input\":{\"op\":\"hash_aggregate\",\"mode\":"
      "\"single\",\"group_by\":[\"l_orderkey\",\"o_orderdate\","
      "\"o_shippriority\"],\"aggs\":[{\"fn\":\"sum\",\"as\":\"revenue\","
      "\"expr\":{\"mul\":[{\"col\":\"l_extendedprice\"},{\"sub\":"
      "[{\"lit\":100},{\"col\":\"l_discount\"}]}]}}],\"input\":{"
      "\"op\":\"hash_join\",\"build\":{\"op\":\"hash_join\",\"build\":{"
      "\"op\":\"filter\",\"predicates\":[{\"col\":\"c_mktsegment\","
      "\"cmp\":\"eq\",\"lo\":1}],\"input\":{\"op\":\"scan\",\"schema\":"
      "[{\"name\":\"c_custkey\",\"dtype\":\"int64\"},"
      "{\"name\":\"c_mktsegment\",\"dtype\":\"dict8\"}],\"source\":{"
      "\"kind\":\"device\",\"table\":\"customer\"}}},\"probe\":{"
      "\"op\":\"filter\",\"predicates\":[{\"col\":\"o_orderdate\","
      "\"cmp\":\"lt\",\"hi\":9204}],\"input\":{\"op\":\"scan\",\"schema\":"
      "[{\"name\":\"o_orderkey\",\"dtype\":\"int64\"},"
      "{\"name\":\"o_custkey\",\"dtype\":\"int64\"},"
      "{\"name\":\"o_orderdate\",\"dtype\":\"date32\"},"
      "{\"name\":\"o_shippriority\",\"dtype\":\"int32\"}],\"source\":{"
      "\"kind\":\"device\",\"table\":\"orders\"}}},\"build_keys\":"
      "[\"c_custkey\"],\"probe_keys\":[\"o_custkey\"],\"join_type\":"
      "\"inner\",\"output\":[{\"side\":\"probe\",\"col\":\"o_orderkey\"},"
      "{\"side\":\"probe\",\"col\":\"o_orderdate\"},"
      "{\"side\":\"probe\",\"col\":\"o_shippriority\"}]},"
      "\"probe\":{\"op\":\"filter\",\"predicates\":[{\"col\":"
      "\"l_shipdate\",\"cmp\":\"gt\",\"lo\":9204}],\"input\":{"
      "\"op\":\"scan\",\"schema\":") + LI_SCHEMA +
      ",\"source\":{\"kind\":\"device\",\"table\":\"lineitem\"},"
      "\"projection\":[\"l_orderkey\",\"l_shipdate\",\"l_extendedprice\","
      "\"l_discount\"]}},\"build_keys\":[\"o_orderkey\"],\"probe_keys\":"
      "[\"l_orderkey\"],\"join_type\":\"inner\",\"output\":["
      "{\"side\":\"probe\",\"col\":\"l_orderkey\"},"
      "{\"side\":\"build\",\"col\":\"o_orderdate\"},"
      "{\"side\":\"build\",\"col\":\"o_shippriority\"},"
      "{\"side\":\"probe\",\"col\":\"l_extendedprice\"},"
      "{\"side\":\"probe\",\"col\":\"l_discount\"}]}}}}}";
  char* out = nullptr;
  double t0 = now_ms();
  CHK(bg_execute_stage(plan.c_str(), &out));
  double cold = now_ms() - t0;
  bg_stage_free(out);
  t0 = now_ms();
  CHK(bg_execute_stage(plan.c_str(), &out));
  double warm = now_ms() - t0;
  printf("q3 stage (C++, sf10-units=%d): cold_ms=%.1f warm_ms=%.1f\n"
         "top10=%s\n", sf, cold, warm, out);
  bg_stage_free(out);
  return 0;
}

int main(int argc, char** argv) {
  const char* mode = argc > 1 ? argv[1] : "q6";
  CHK(bg_init(0));
  if (strcmp(mode, "q6") == 0) {
    int64_t n = argc > 2 ? atoll(argv[2]) : 600037902LL;
    return run_q6(n);
  }
  if (strcmp(mode, "q3") == 0) {
    int sf = argc > 2 ? atoi(argv[2]) : 10;  // units of SF10
    return run_q3(sf);
  }
  if (strcmp(mode, "shuffle") == 0) {
    int64_t n = argc > 2 ? atoll(argv[2]) : 20000000LL;
    int k = argc > 3 ? atoi(argv[3]) : 16;
    return run_shuffle(n, k);
  }
  fprintf(stderr, "usage: %s q6|q3|shuffle [n|sf] [k]\n", argv[0]);
  return 2;
}
