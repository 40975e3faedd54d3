// stage_ipc.h — Arrow IPC encapsulated-message framing for the C++ stage
// interpreter: the RecordBatch metadata flatbuffer written by hand (C++ port
// of datafusion_ballista_amd/ipc.py, which is validated against pyarrow's
// own reader in tests/test_shuffle_format.py), and the mirror walker that
// locates batch buffer extents on read.
//
// Restates the published Arrow format (format/Message.fbs):
//   Message{version, header union, bodyLength},
//   RecordBatch{length, nodes, buffers, compression},
//   BodyCompression{codec=LZ4_FRAME}.
// The schema message is NOT built here: the host supplies its exact bytes
// (hex in the plan), as arrow-rs/pyarrow produce it — field metadata stays
// byte-identical to the reference writers (ipc.py takes the same shortcut).
#ifndef BG_STAGE_IPC_H
#define BG_STAGE_IPC_H

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace bgipc {

static const uint8_t EOS[8] = {0xff, 0xff, 0xff, 0xff, 0, 0, 0, 0};

// ---------------------------------------------------------------------------
// tiny back-to-front flatbuffers builder (only what Message.fbs needs) —
// port of ipc.py's _FB, byte-for-byte the same layout decisions
// ---------------------------------------------------------------------------

class FB {
 public:
  struct Field {  // (field_id, kind, value)
    int id;
    char kind;  // 'q' i64, 'r' ref, 'h' i16, 'b' u8
    int64_t val;
  };

  int64_t end_rel() const { return size_; }

  int64_t struct_vector(const std::vector<std::pair<int64_t, int64_t>>& elems) {
    std::string body;
    body.reserve(elems.size() * 16);
    for (auto& e : elems) {
      append_le(body, e.first);
      append_le(body, e.second);
    }
    align(8, 0);
    prepend(body);
    align(4, 4);  // count must land 4-aligned
    std::string cnt;
    append_le32(cnt, (uint32_t)elems.size());
    prepend(cnt);
    return size_;
  }

  int64_t table(std::vector<Field> fields) {
    // inline field area ordered i64, ref, i16, u8 (alignment simplicity)
    std::vector<Field> order = fields;
    auto rank = [](char k) {
      switch (k) { case 'q': return 0; case 'r': return 1;
                   case 'h': return 2; default: return 3; }
    };
    for (size_t a = 0; a < order.size(); ++a)  // stable insertion sort
      for (size_t b = a; b > 0 &&
           rank(order[b].kind) < rank(order[b - 1].kind); --b)
        std::swap(order[b], order[b - 1]);

    struct Placed { Field f; int pos; int pad; };
    std::vector<Placed> placed;
    int pos = 4;  // after soffset i32
    for (auto& f : order) {
      int sz = f.kind == 'q' ? 8 : f.kind == 'r' ? 4 : f.kind == 'h' ? 2 : 1;
      int pad = (-pos) % sz;
      if (pad < 0) pad += sz;
      pos += pad;
      placed.push_back({f, pos, pad});
      pos += sz;
    }
    int table_len = pos;
    int pad_tail = (-table_len) % 4;
    if (pad_tail < 0) pad_tail += 4;
    table_len += pad_tail;

    int max_fid = -1;
    for (auto& f : fields) max_fid = f.id > max_fid ? f.id : max_fid;
    int vt_len = 4 + 2 * (max_fid + 1);
    std::vector<uint16_t> vslots((size_t)(max_fid + 1), 0);
    for (auto& p : placed) vslots[(size_t)p.f.id] = (uint16_t)p.pos;
    std::string vtable;
    append_le16(vtable, (uint16_t)vt_len);
    append_le16(vtable, (uint16_t)table_len);
    for (uint16_t s : vslots) append_le16(vtable, s);

    // inline fields (+ tail pad); refs patched once position known
    std::string inl;
    for (auto& p : placed) {
      inl.append((size_t)p.pad, '\0');
      switch (p.f.kind) {
        case 'q': append_le(inl, p.f.val); break;
        case 'h': append_le16(inl, (uint16_t)p.f.val); break;
        case 'b': inl += (char)(uint8_t)p.f.val; break;
        case 'r': append_le32(inl, 0); break;  // patched below
      }
    }
    inl.append((size_t)pad_tail, '\0');

    // absolute alignment: force the table's end_rel to 0 mod 8 (i64 scalars
    // must land 8-aligned absolutely; finish() pads the total to 8)
    int64_t pre_pad = (-(size_ + (int64_t)inl.size() + 4)) % 8;
    if (pre_pad < 0) pre_pad += 8;
    if (pre_pad) prepend(std::string((size_t)pre_pad, '\0'));
    prepend(inl);
    int64_t table_rel = size_ + 4;  // account for soffset written next
    // resolve refs now that the table position is known
    std::string& last = parts_.back();
    for (auto& p : placed) {
      if (p.f.kind == 'r') {
        int64_t field_rel = table_rel - p.pos;
        uint32_t uoff = (uint32_t)(field_rel - p.f.val);
        patch_le32(last, (size_t)(p.pos - 4), uoff);
      }
    }
    std::string soff_part;
    append_le32(soff_part, 0);  // placeholder soffset
    prepend(soff_part);
    table_rel = size_;
    prepend(vtable);
    int64_t vtable_rel = size_;
    // soffset i32 at table start = vtable_rel - table_rel (end-relative)
    int32_t soff = (int32_t)(vtable_rel - table_rel);
    patch_le32(parts_[parts_.size() - 2], 0, (uint32_t)soff);
    return table_rel;
  }

  std::string finish(int64_t root_rel) {
    align(8, 4);
    int64_t root_pos = size_ + 4;
    std::string root;
    append_le32(root, (uint32_t)(root_pos - root_rel));
    prepend(root);
    std::string out;
    out.reserve((size_t)size_);
    for (auto it = parts_.rbegin(); it != parts_.rend(); ++it) out += *it;
    return out;
  }

 private:
  std::vector<std::string> parts_;  // later parts sit closer to the END
  int64_t size_ = 0;

  void prepend(const std::string& b) {
    parts_.push_back(b);
    size_ += (int64_t)b.size();
  }
  void align(int n, int extra) {
    int64_t pad = (-(size_ + extra)) % n;
    if (pad < 0) pad += n;
    if (pad) prepend(std::string((size_t)pad, '\0'));
  }
  static void append_le(std::string& s, int64_t v) {
    for (int i = 0; i < 8; ++i) s += (char)(uint8_t)(((uint64_t)v) >> (8 * i));
  }
  static void append_le32(std::string& s, uint32_t v) {
    for (int i = 0; i < 4; ++i) s += (char)(uint8_t)(v >> (8 * i));
  }
  static void append_le16(std::string& s, uint16_t v) {
    s += (char)(uint8_t)v;
    s += (char)(uint8_t)(v >> 8);
  }
  static void patch_le32(std::string& s, size_t off, uint32_t v) {
    for (int i = 0; i < 4; ++i) s[off + (size_t)i] = (char)(uint8_t)(v >> (8 * i));
  }
};

// MessageHeader union types (format/Message.fbs)
constexpr uint8_t HDR_RECORD_BATCH = 3;
constexpr int16_t V5 = 4;

struct BufSpec { int64_t off, len; };
struct FieldNode { int64_t length, null_count; };

// Metadata flatbuffer (with encapsulated framing) for one RecordBatch.
inline std::string record_batch_message(
    int64_t n_rows, const std::vector<FieldNode>& nodes,
    const std::vector<BufSpec>& buffers, int64_t body_len, bool compressed) {
  FB fb;
  int64_t comp_rel = 0;
  if (compressed)
    comp_rel = fb.table({{0, 'b', 0}});  // codec=LZ4_FRAME(0), explicit
  std::vector<std::pair<int64_t, int64_t>> bufv, nodev;
  for (auto& b : buffers) bufv.push_back({b.off, b.len});
  for (auto& n : nodes) nodev.push_back({n.length, n.null_count});
  int64_t bufs_rel = fb.struct_vector(bufv);
  int64_t nodes_rel = fb.struct_vector(nodev);
  std::vector<FB::Field> rb_fields = {
      {0, 'q', n_rows}, {1, 'r', nodes_rel}, {2, 'r', bufs_rel}};
  if (compressed) rb_fields.push_back({3, 'r', comp_rel});
  int64_t rb_rel = fb.table(rb_fields);
  int64_t msg_rel = fb.table({{0, 'h', V5},
                              {1, 'b', HDR_RECORD_BATCH},
                              {2, 'r', rb_rel},
                              {3, 'q', body_len}});
  std::string meta = fb.finish(msg_rel);
  size_t pad = (size_t)((-(int64_t)(meta.size() + 8)) % 8);
  meta.append(pad, '\0');
  std::string out;
  out += "\xff\xff\xff\xff";
  uint32_t mlen = (uint32_t)meta.size();
  for (int i = 0; i < 4; ++i) out += (char)(uint8_t)(mlen >> (8 * i));
  out += meta;
  return out;
}

// ---------------------------------------------------------------------------
// reader side: walk an IPC stream's messages, parse RecordBatch metadata —
// port of ipc.py walk_stream/parse_record_batch_meta
// ---------------------------------------------------------------------------

struct BatchMeta {
  int64_t n_rows;
  std::vector<BufSpec> bufs;
  bool compressed;
  int64_t body_len;
  int64_t body_off;  // absolute offset of the body within the walked range
};

namespace detail {
inline uint16_t rd16(const uint8_t* p) { return (uint16_t)(p[0] | p[1] << 8); }
inline uint32_t rd32(const uint8_t* p) {
  return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) |
         ((uint32_t)p[3] << 24);
}
inline int64_t rd64(const uint8_t* p) {
  uint64_t v = 0;
  for (int i = 0; i < 8; ++i) v |= ((uint64_t)p[i]) << (8 * i);
  return (int64_t)v;
}
inline std::vector<uint16_t> parse_table(const uint8_t* meta, int64_t tbl) {
  int32_t soff = (int32_t)rd32(meta + tbl);
  int64_t vt = tbl - soff;
  uint16_t vt_len = rd16(meta + vt);
  int n = (vt_len - 4) / 2;
  std::vector<uint16_t> slots((size_t)n);
  for (int i = 0; i < n; ++i) slots[(size_t)i] = rd16(meta + vt + 4 + 2 * i);
  return slots;
}
}  // namespace detail

// Parse one RecordBatch message's metadata; returns false if the message is
// not a RecordBatch (schema/dictionary messages).
inline bool parse_record_batch_meta(const uint8_t* meta, BatchMeta* out) {
  using namespace detail;
  uint32_t root = rd32(meta);
  auto slots = parse_table(meta, (int64_t)root);
  auto fld = [&](size_t i) -> uint16_t {
    return i < slots.size() ? slots[i] : 0;
  };
  if (!fld(1) || meta[root + slots[1]] != HDR_RECORD_BATCH) return false;
  out->body_len = fld(3) ? rd64(meta + root + slots[3]) : 0;
  int64_t hpos = (int64_t)root + slots[2];
  int64_t htbl = hpos + rd32(meta + hpos);
  auto hslots = parse_table(meta, htbl);
  out->n_rows = hslots.size() > 0 && hslots[0] ? rd64(meta + htbl + hslots[0]) : 0;
  int64_t bpos = htbl + hslots[2];
  int64_t bvec = bpos + rd32(meta + bpos);
  uint32_t cnt = rd32(meta + bvec);
  out->bufs.clear();
  for (uint32_t i = 0; i < cnt; ++i) {
    out->bufs.push_back({rd64(meta + bvec + 4 + 16 * i),
                         rd64(meta + bvec + 4 + 16 * i + 8)});
  }
  out->compressed = hslots.size() > 3 && hslots[3] != 0;
  return true;
}

// Walk a partition byte range — a concatenation of complete IPC sub-streams
// (one per input, writer.rs:861-884; multi_stream_reader.rs:17-34) — into
// batch metadata with absolute body offsets.  Skips schema messages and
// crosses EOS markers between sub-streams.
inline std::vector<BatchMeta> walk_partition(const uint8_t* raw, int64_t len) {
  std::vector<BatchMeta> out;
  int64_t pos = 0;
  while (pos + 8 <= len) {
    if (memcmp(raw + pos, "\xff\xff\xff\xff", 4) != 0)
      throw std::runtime_error("IPC stream: missing continuation marker");
    int32_t mlen = (int32_t)detail::rd32(raw + pos + 4);
    if (mlen == 0) { pos += 8; continue; }  // EOS: next sub-stream follows
    const uint8_t* meta = raw + pos + 8;
    BatchMeta bm;
    int64_t body_off = pos + 8 + mlen;
    if (!parse_record_batch_meta(meta, &bm)) {
      pos = body_off;  // schema message: no body
      continue;
    }
    bm.body_off = body_off;
    out.push_back(bm);
    pos = body_off + bm.body_len;
  }
  return out;
}

}  // namespace bgipc

#endif  // BG_STAGE_IPC_H
