// stage_parquet.h — C++ Parquet scan for the stage interpreter
// (scan{kind:"parquet"}): the DataSourceExec(Parquet) feed of SURVEY.md
// §8f.1 running INSIDE bg_execute_stage.  The host supplies per-column-
// chunk metadata from the footer (pyarrow / parquet-rs both parse footers
// natively); this code walks the page headers (thrift compact protocol,
// restated from the published parquet-format PageHeader struct — the same
// restatement as the Python reader, whose parity is pinned against
// pyarrow's reader in tests) and drives the device decode pipeline:
// snappy batch -> definition levels -> PLAIN/FLBA extract + dictionary
// expansion/gather.
//
// Scope: the TPC-H column shapes — V1 data pages, PLAIN (0) and
// RLE_DICTIONARY (2/8) encodings, INT32/INT64/DOUBLE/FLBA physicals,
// max_def <= 1, SNAPPY/UNCOMPRESSED.  Everything else fails loudly with a
// message pointing at the Python reader (which covers the full encoding
// matrix: DELTA*, BYTE_STREAM_SPLIT, BYTE_ARRAY, DataPageV2, ZSTD bridge).
// Included by stage.cpp only (uses its Table/Col/DBuf machinery).
#ifndef BG_STAGE_PARQUET_H
#define BG_STAGE_PARQUET_H

// ---- thrift compact protocol (PageHeader subset) ----
struct ThriftReader {
  const uint8_t* b;
  size_t n;
  size_t i = 0;

  uint64_t varint() {
    uint64_t x = 0;
    int s = 0;
    while (i < n) {
      uint8_t c = b[i++];
      x |= (uint64_t)(c & 0x7F) << s;
      if (!(c & 0x80)) return x;
      s += 7;
    }
    throw StageError(BG_ERR_INVALID, "parquet: truncated thrift varint");
  }
  int64_t zigzag() {
    uint64_t v = varint();
    return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
  }
  void skip(int t);
  void skip_struct() {
    while (true) {
      if (i >= n) throw StageError(BG_ERR_INVALID, "parquet: thrift eof");
      uint8_t h = b[i++];
      if (h == 0) return;
      int t = h & 0x0F;
      if (!(h >> 4)) zigzag();  // long-form field id
      skip(t);
    }
  }
};

inline void ThriftReader::skip(int t) {
  switch (t) {
    case 1:
    case 2: return;                    // bool true/false
    case 3: i += 1; return;            // byte
    case 4:
    case 5:
    case 6: varint(); return;          // i16/i32/i64
    case 7: i += 8; return;            // double
    case 8: { uint64_t l = varint(); i += l; return; }  // binary
    case 9:
    case 10: {                         // list/set
      uint8_t h = b[i++];
      uint64_t cnt = h >> 4;
      int et = h & 0x0F;
      if (cnt == 15) cnt = varint();
      for (uint64_t k = 0; k < cnt; ++k) skip(et);
      return;
    }
    case 11: {                         // map
      uint64_t cnt = varint();
      if (cnt) {
        uint8_t kv = b[i++];
        for (uint64_t k = 0; k < cnt; ++k) {
          skip(kv >> 4);
          skip(kv & 0x0F);
        }
      }
      return;
    }
    case 12: skip_struct(); return;
    default:
      throw StageError(BG_ERR_INVALID, "parquet: bad thrift type");
  }
}

struct PageHeader {
  int type = 0;          // 0 data, 2 dictionary, 3 data v2
  int64_t usz = 0, csz = 0;
  int64_t nvals = 0;
  int enc = 0;
  int64_t ndict = 0;
  size_t data_pos = 0;   // byte after the header
};

// parse PageHeader{1:type, 2:usz, 3:csz, 5:DataPageHeader{1:nvals,2:enc},
// 7:DictionaryPageHeader{1:nvals}, 8:DataPageHeaderV2}
inline PageHeader parse_page_header(const uint8_t* buf, size_t len,
                                    size_t pos) {
  ThriftReader r{buf, len, pos};
  PageHeader ph;
  bool saw_v2 = false;
  auto parse_sub = [&](int64_t want_nvals_fid, int64_t* out_a,
                       int64_t want_enc_fid, int64_t* out_b) {
    int64_t last = 0;
    while (true) {
      if (r.i >= r.n)
        throw StageError(BG_ERR_INVALID, "parquet: header eof");
      uint8_t sh = r.b[r.i++];
      if (sh == 0) break;
      int sd = sh >> 4;
      int st = sh & 0x0F;
      int64_t sf = sd ? last + sd : r.zigzag();
      last = sf;
      if (sf == want_nvals_fid && out_a) *out_a = r.zigzag();
      else if (sf == want_enc_fid && out_b) *out_b = r.zigzag();
      else r.skip(st);
    }
  };
  int64_t last = 0;
  while (true) {
    if (r.i >= r.n) throw StageError(BG_ERR_INVALID, "parquet: header eof");
    uint8_t h = r.b[r.i++];
    if (h == 0) break;
    int delta = h >> 4;
    int t = h & 0x0F;
    int64_t fid = delta ? last + delta : r.zigzag();
    last = fid;
    switch (fid) {
      case 1: ph.type = (int)r.zigzag(); break;
      case 2: ph.usz = r.zigzag(); break;
      case 3: ph.csz = r.zigzag(); break;
      case 5: {
        int64_t enc64 = 0;
        parse_sub(1, &ph.nvals, 2, &enc64);
        ph.enc = (int)enc64;
        break;
      }
      case 7: parse_sub(1, &ph.ndict, -1, nullptr); break;
      case 8:
        saw_v2 = true;
        r.skip(t);
        break;
      default:
        r.skip(t);
    }
  }
  if (ph.type == 3 || saw_v2)
    throw StageError(BG_ERR_UNSUPPORTED,
                     "parquet scan: DataPageV2 — use the Python reader "
                     "(datafusion_ballista_amd.parquet) for V2 pages");
  ph.data_pos = r.i;
  return ph;
}

// ---- the column decode driver ----

struct PqChunk {
  int64_t start, size, num_values;
};

struct PqColumnSpec {
  std::string phys;    // INT32|INT64|DOUBLE|FLBA
  int flba_len = 0;
  int max_def = 0;
  std::string codec;   // SNAPPY|UNCOMPRESSED
  std::vector<PqChunk> chunks;
};

// Decode one column from the resident file buffer into a Col.
Col parquet_read_column(const uint8_t* h_file, size_t file_len,
                        const DBufPtr& d_file, const PqColumnSpec& spec,
                        const DtSpec& out_dt);

#endif  // BG_STAGE_PARQUET_H
