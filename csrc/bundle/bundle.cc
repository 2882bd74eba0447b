// TF checkpoint-V2 "tensor bundle" codec — capability C12.
//
// Implements, from the on-disk format spec, a bit-compatible writer and
// reader for the reference stack's checkpoint format
// (TF tensorflow/core/util/tensor_bundle/tensor_bundle.cc +
//  tensorflow/core/lib/io/table — a leveldb-format SSTable):
//
//   prefix.index               leveldb-style table:
//                                key ""      -> BundleHeaderProto
//                                key <name>  -> BundleEntryProto
//                              blocks: prefix-compressed entries, restart
//                              array, 5-byte trailer (type byte +
//                              masked-crc32c); footer: metaindex handle +
//                              index handle + padding + magic
//                              0xdb4775248b80fb57.
//   prefix.data-00000-of-00001 concatenated raw little-endian tensor bytes.
//
// Protos are hand-encoded (protobuf wire format) so there is no
// protobuf dependency. CRC32C is the Castagnoli polynomial with
// leveldb's masking: mask(c) = ((c >> 15) | (c << 17)) + 0xa282ead8.
//
// No TensorFlow code was used; this is written from the public format.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <cstring>
#include <fstream>
#include <map>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

// ---------------------------------------------------------------------------
// CRC32C (Castagnoli), software table implementation + leveldb masking.
// ---------------------------------------------------------------------------
namespace crc32c {

static uint32_t table[8][256];
static bool table_init = [] {
  const uint32_t poly = 0x82f63b78u;  // reflected Castagnoli
  for (uint32_t i = 0; i < 256; i++) {
    uint32_t c = i;
    for (int k = 0; k < 8; k++) c = (c & 1) ? (poly ^ (c >> 1)) : (c >> 1);
    table[0][i] = c;
  }
  for (uint32_t i = 0; i < 256; i++) {
    uint32_t c = table[0][i];
    for (int s = 1; s < 8; s++) {
      c = table[0][c & 0xff] ^ (c >> 8);
      table[s][i] = c;
    }
  }
  return true;
}();

inline uint32_t extend(uint32_t crc, const char* buf, size_t size) {
  uint32_t c = crc ^ 0xffffffffu;
  const uint8_t* p = reinterpret_cast<const uint8_t*>(buf);
  // 8-byte slicing
  while (size >= 8) {
    c ^= static_cast<uint32_t>(p[0]) | (static_cast<uint32_t>(p[1]) << 8) |
         (static_cast<uint32_t>(p[2]) << 16) | (static_cast<uint32_t>(p[3]) << 24);
    uint32_t hi = static_cast<uint32_t>(p[4]) | (static_cast<uint32_t>(p[5]) << 8) |
                  (static_cast<uint32_t>(p[6]) << 16) | (static_cast<uint32_t>(p[7]) << 24);
    c = table[7][c & 0xff] ^ table[6][(c >> 8) & 0xff] ^
        table[5][(c >> 16) & 0xff] ^ table[4][(c >> 24) & 0xff] ^
        table[3][hi & 0xff] ^ table[2][(hi >> 8) & 0xff] ^
        table[1][(hi >> 16) & 0xff] ^ table[0][(hi >> 24) & 0xff];
    p += 8;
    size -= 8;
  }
  while (size--) c = table[0][(c ^ *p++) & 0xff] ^ (c >> 8);
  return c ^ 0xffffffffu;
}

inline uint32_t value(const char* buf, size_t size) { return extend(0, buf, size); }

constexpr uint32_t kMaskDelta = 0xa282ead8u;
inline uint32_t mask(uint32_t crc) {
  return ((crc >> 15) | (crc << 17)) + kMaskDelta;
}
inline uint32_t unmask(uint32_t m) {
  uint32_t rot = m - kMaskDelta;
  return (rot >> 17) | (rot << 15);
}

}  // namespace crc32c

// ---------------------------------------------------------------------------
// Varint / fixed coding (protobuf + leveldb share the varint format)
// ---------------------------------------------------------------------------
static void put_varint32(std::string* dst, uint32_t v) {
  while (v >= 0x80) {
    dst->push_back(static_cast<char>(v | 0x80));
    v >>= 7;
  }
  dst->push_back(static_cast<char>(v));
}
static void put_varint64(std::string* dst, uint64_t v) {
  while (v >= 0x80) {
    dst->push_back(static_cast<char>(v | 0x80));
    v >>= 7;
  }
  dst->push_back(static_cast<char>(v));
}
static void put_fixed32(std::string* dst, uint32_t v) {
  char b[4] = {static_cast<char>(v), static_cast<char>(v >> 8),
               static_cast<char>(v >> 16), static_cast<char>(v >> 24)};
  dst->append(b, 4);
}
static void put_fixed64(std::string* dst, uint64_t v) {
  for (int i = 0; i < 8; i++) dst->push_back(static_cast<char>(v >> (8 * i)));
}
static uint32_t get_fixed32(const char* p) {
  return static_cast<uint32_t>(static_cast<uint8_t>(p[0])) |
         (static_cast<uint32_t>(static_cast<uint8_t>(p[1])) << 8) |
         (static_cast<uint32_t>(static_cast<uint8_t>(p[2])) << 16) |
         (static_cast<uint32_t>(static_cast<uint8_t>(p[3])) << 24);
}
static uint64_t get_fixed64(const char* p) {
  uint64_t v = 0;
  for (int i = 7; i >= 0; i--) v = (v << 8) | static_cast<uint8_t>(p[i]);
  return v;
}
static const char* get_varint64(const char* p, const char* limit, uint64_t* v) {
  uint64_t result = 0;
  for (uint32_t shift = 0; shift <= 63 && p < limit; shift += 7) {
    uint8_t byte = static_cast<uint8_t>(*p++);
    result |= static_cast<uint64_t>(byte & 0x7f) << shift;
    if (!(byte & 0x80)) {
      *v = result;
      return p;
    }
  }
  return nullptr;
}
static const char* get_varint32(const char* p, const char* limit, uint32_t* v) {
  uint64_t u;
  const char* q = get_varint64(p, limit, &u);
  if (q) *v = static_cast<uint32_t>(u);
  return q;
}

// ---------------------------------------------------------------------------
// Hand-rolled proto encode/decode for BundleHeaderProto / BundleEntryProto
// ---------------------------------------------------------------------------
struct BundleEntry {
  int dtype = 0;                    // tensorflow DataType enum value
  std::vector<int64_t> shape;
  int shard_id = 0;
  int64_t offset = 0;
  int64_t size = 0;
  uint32_t crc32c_masked = 0;       // masked crc of tensor bytes
};

static std::string encode_header(int num_shards) {
  std::string s;
  // field 1 (num_shards) varint
  put_varint32(&s, (1 << 3) | 0);
  put_varint64(&s, static_cast<uint64_t>(num_shards));
  // field 2 endianness: LITTLE = 0 → default, omitted
  // field 3 version: VersionDef{ producer = 1 }
  std::string ver;
  put_varint32(&ver, (1 << 3) | 0);
  put_varint64(&ver, 1);
  put_varint32(&s, (3 << 3) | 2);
  put_varint64(&s, ver.size());
  s += ver;
  return s;
}

static std::string encode_entry(const BundleEntry& e) {
  std::string s;
  if (e.dtype != 0) {
    put_varint32(&s, (1 << 3) | 0);
    put_varint64(&s, static_cast<uint64_t>(e.dtype));
  }
  {  // field 2: TensorShapeProto { repeated Dim dim = 2 { int64 size = 1 } }
    std::string shape;
    for (int64_t d : e.shape) {
      std::string dim;
      put_varint32(&dim, (1 << 3) | 0);
      put_varint64(&dim, static_cast<uint64_t>(d));
      put_varint32(&shape, (2 << 3) | 2);
      put_varint64(&shape, dim.size());
      shape += dim;
    }
    put_varint32(&s, (2 << 3) | 2);
    put_varint64(&s, shape.size());
    s += shape;
  }
  if (e.shard_id != 0) {
    put_varint32(&s, (3 << 3) | 0);
    put_varint64(&s, static_cast<uint64_t>(e.shard_id));
  }
  if (e.offset != 0) {
    put_varint32(&s, (4 << 3) | 0);
    put_varint64(&s, static_cast<uint64_t>(e.offset));
  }
  if (e.size != 0) {
    put_varint32(&s, (5 << 3) | 0);
    put_varint64(&s, static_cast<uint64_t>(e.size));
  }
  if (e.crc32c_masked != 0) {
    put_varint32(&s, (6 << 3) | 5);  // fixed32
    put_fixed32(&s, e.crc32c_masked);
  }
  return s;
}

static bool decode_entry(const std::string& buf, BundleEntry* e) {
  const char* p = buf.data();
  const char* limit = p + buf.size();
  while (p < limit) {
    uint32_t tag;
    p = get_varint32(p, limit, &tag);
    if (!p) return false;
    uint32_t field = tag >> 3, wire = tag & 7;
    if (wire == 0) {
      uint64_t v;
      p = get_varint64(p, limit, &v);
      if (!p) return false;
      if (field == 1) e->dtype = static_cast<int>(v);
      else if (field == 3) e->shard_id = static_cast<int>(v);
      else if (field == 4) e->offset = static_cast<int64_t>(v);
      else if (field == 5) e->size = static_cast<int64_t>(v);
    } else if (wire == 2) {
      uint64_t len;
      p = get_varint64(p, limit, &len);
      if (!p || p + len > limit) return false;
      if (field == 2) {  // shape
        const char* sp = p;
        const char* slimit = p + len;
        while (sp < slimit) {
          uint32_t stag;
          sp = get_varint32(sp, slimit, &stag);
          if (!sp) return false;
          if ((stag >> 3) == 2 && (stag & 7) == 2) {  // Dim
            uint64_t dlen;
            sp = get_varint64(sp, slimit, &dlen);
            if (!sp) return false;
            const char* dp = sp;
            const char* dlimit = sp + dlen;
            int64_t dimsize = 0;
            while (dp < dlimit) {
              uint32_t dtag;
              dp = get_varint32(dp, dlimit, &dtag);
              if (!dp) return false;
              if ((dtag >> 3) == 1 && (dtag & 7) == 0) {
                uint64_t v;
                dp = get_varint64(dp, dlimit, &v);
                if (!dp) return false;
                dimsize = static_cast<int64_t>(v);
              } else if ((dtag & 7) == 2) {
                uint64_t l;
                dp = get_varint64(dp, dlimit, &l);
                if (!dp) return false;
                dp += l;
              } else {
                return false;
              }
            }
            e->shape.push_back(dimsize);
            sp += dlen;
          } else if ((stag & 7) == 0) {
            uint64_t v;
            sp = get_varint64(sp, slimit, &v);
            if (!sp) return false;
          } else {
            return false;
          }
        }
      }
      p += len;
    } else if (wire == 5) {
      if (p + 4 > limit) return false;
      if (field == 6) e->crc32c_masked = get_fixed32(p);
      p += 4;
    } else if (wire == 1) {
      if (p + 8 > limit) return false;
      p += 8;
    } else {
      return false;
    }
  }
  return true;
}

// ---------------------------------------------------------------------------
// leveldb-format table building blocks
// ---------------------------------------------------------------------------
static constexpr uint64_t kTableMagic = 0xdb4775248b80fb57ull;
static constexpr int kRestartInterval = 16;
static constexpr size_t kBlockSize = 4096;

class BlockBuilder {
 public:
  void add(const std::string& key, const std::string& value) {
    size_t shared = 0;
    if (counter_ < kRestartInterval) {
      size_t minlen = std::min(last_key_.size(), key.size());
      while (shared < minlen && last_key_[shared] == key[shared]) shared++;
    } else {
      restarts_.push_back(static_cast<uint32_t>(buffer_.size()));
      counter_ = 0;
    }
    size_t non_shared = key.size() - shared;
    put_varint32(&buffer_, static_cast<uint32_t>(shared));
    put_varint32(&buffer_, static_cast<uint32_t>(non_shared));
    put_varint32(&buffer_, static_cast<uint32_t>(value.size()));
    buffer_.append(key.data() + shared, non_shared);
    buffer_ += value;
    last_key_ = key;
    counter_++;
  }

  std::string finish() {
    for (uint32_t r : restarts_) put_fixed32(&buffer_, r);
    put_fixed32(&buffer_, static_cast<uint32_t>(restarts_.size()));
    return buffer_;
  }

  size_t estimated_size() const {
    return buffer_.size() + restarts_.size() * 4 + 4;
  }
  bool empty() const { return buffer_.empty(); }
  void reset() {
    buffer_.clear();
    restarts_.assign(1, 0);
    counter_ = 0;
    last_key_.clear();
  }
  BlockBuilder() { restarts_.assign(1, 0); }

 private:
  std::string buffer_;
  std::vector<uint32_t> restarts_;
  int counter_ = 0;
  std::string last_key_;
};

struct BlockHandle {
  uint64_t offset = 0, size = 0;
  void encode(std::string* dst) const {
    put_varint64(dst, offset);
    put_varint64(dst, size);
  }
  const char* decode(const char* p, const char* limit) {
    p = get_varint64(p, limit, &offset);
    if (p) p = get_varint64(p, limit, &size);
    return p;
  }
};

// Write block + 5-byte trailer to f; returns handle.
static BlockHandle emit_block(std::ofstream& f, uint64_t* file_off,
                              const std::string& contents) {
  BlockHandle h;
  h.offset = *file_off;
  h.size = contents.size();
  f.write(contents.data(), contents.size());
  char type = 0;  // no compression
  uint32_t crc = crc32c::value(contents.data(), contents.size());
  crc = crc32c::extend(crc, &type, 1);
  std::string trailer;
  trailer.push_back(type);
  put_fixed32(&trailer, crc32c::mask(crc));
  f.write(trailer.data(), trailer.size());
  *file_off += contents.size() + 5;
  return h;
}

// ---------------------------------------------------------------------------
// BundleWriter
// ---------------------------------------------------------------------------
class BundleWriter {
 public:
  // num_shards > 1 writes .data-SSSSS-of-NNNNN files with tensors
  // assigned greedily to the least-loaded shard (the reference's
  // sharded SaveV2 produces the same multi-file layout; readers find
  // each tensor through its entry's shard_id).
  explicit BundleWriter(const std::string& prefix, int num_shards = 1)
      : prefix_(prefix), num_shards_(num_shards) {
    if (num_shards < 1 || num_shards > 99999)
      throw std::runtime_error("num_shards out of range");
    data_.resize(num_shards);
    offsets_.assign(num_shards, 0);
    for (int sdx = 0; sdx < num_shards; sdx++) {
      char nm[64];
      snprintf(nm, sizeof(nm), ".data-%05d-of-%05d", sdx, num_shards);
      data_paths_.push_back(prefix_ + nm);
      tmp_data_paths_.push_back(prefix_ + nm + ".tempstate");
      data_[sdx].open(tmp_data_paths_[sdx],
                      std::ios::binary | std::ios::trunc);
      if (!data_[sdx])
        throw std::runtime_error("cannot open " + tmp_data_paths_[sdx]);
    }
  }

  void add(const std::string& name, int dtype,
           const std::vector<int64_t>& shape, py::buffer buf) {
    if (finished_) throw std::runtime_error("writer already finished");
    if (entries_.count(name)) throw std::runtime_error("duplicate key " + name);
    py::buffer_info info = buf.request();
    const char* ptr = static_cast<const char*>(info.ptr);
    size_t nbytes = static_cast<size_t>(info.size) * info.itemsize;
    // greedy least-loaded shard
    int sdx = 0;
    for (int i = 1; i < num_shards_; i++)
      if (offsets_[i] < offsets_[sdx]) sdx = i;
    BundleEntry e;
    e.dtype = dtype;
    e.shape = shape;
    e.shard_id = sdx;
    e.offset = static_cast<int64_t>(offsets_[sdx]);
    e.size = static_cast<int64_t>(nbytes);
    e.crc32c_masked = crc32c::mask(crc32c::value(ptr, nbytes));
    {
      py::gil_scoped_release rel;
      data_[sdx].write(ptr, nbytes);
    }
    offsets_[sdx] += nbytes;
    entries_[name] = e;
  }

  void finish() {
    if (finished_) return;
    for (int sdx = 0; sdx < num_shards_; sdx++) {
      data_[sdx].close();
      if (std::rename(tmp_data_paths_[sdx].c_str(),
                      data_paths_[sdx].c_str()) != 0)
        throw std::runtime_error("rename failed for " + data_paths_[sdx]);
    }

    std::string index_path = prefix_ + ".index";
    std::string tmp_index = index_path + ".tempstate";
    std::ofstream idx(tmp_index, std::ios::binary | std::ios::trunc);
    if (!idx) throw std::runtime_error("cannot open " + tmp_index);
    uint64_t off = 0;

    // Entries sorted; "" (header) first by std::map ordering.
    std::map<std::string, std::string> kv;
    kv[""] = encode_header(num_shards_);
    for (auto& it : entries_) kv[it.first] = encode_entry(it.second);

    BlockBuilder data_block;
    BlockBuilder index_block;
    std::string last_key_in_block;
    std::vector<std::pair<std::string, BlockHandle>> pending_index;
    for (auto& it : kv) {
      data_block.add(it.first, it.second);
      last_key_in_block = it.first;
      if (data_block.estimated_size() >= kBlockSize) {
        BlockHandle h = emit_block(idx, &off, data_block.finish());
        pending_index.emplace_back(last_key_in_block, h);
        data_block.reset();
      }
    }
    if (!data_block.empty()) {
      BlockHandle h = emit_block(idx, &off, data_block.finish());
      pending_index.emplace_back(last_key_in_block, h);
    }

    // metaindex (empty block)
    BlockBuilder meta;
    BlockHandle metaindex_handle = emit_block(idx, &off, meta.finish());

    // index block: key = last key of data block (valid separator)
    for (auto& p : pending_index) {
      std::string hv;
      p.second.encode(&hv);
      index_block.add(p.first, hv);
    }
    BlockHandle index_handle = emit_block(idx, &off, index_block.finish());

    // footer
    std::string footer;
    metaindex_handle.encode(&footer);
    index_handle.encode(&footer);
    footer.resize(40, '\0');
    put_fixed64(&footer, kTableMagic);
    idx.write(footer.data(), footer.size());
    idx.close();
    if (std::rename(tmp_index.c_str(), index_path.c_str()) != 0)
      throw std::runtime_error("rename failed for " + index_path);
    finished_ = true;
  }

 private:
  std::string prefix_;
  int num_shards_ = 1;
  std::vector<std::string> data_paths_, tmp_data_paths_;
  std::vector<std::ofstream> data_;
  std::vector<uint64_t> offsets_;
  std::map<std::string, BundleEntry> entries_;
  bool finished_ = false;
};

// ---------------------------------------------------------------------------
// BundleReader
// ---------------------------------------------------------------------------
class BundleReader {
 public:
  explicit BundleReader(const std::string& prefix) : prefix_(prefix) {
    std::string index_path = prefix + ".index";
    std::ifstream f(index_path, std::ios::binary);
    if (!f) throw std::runtime_error("cannot open " + index_path);
    std::string contents((std::istreambuf_iterator<char>(f)),
                         std::istreambuf_iterator<char>());
    if (contents.size() < 48) throw std::runtime_error("index too small");
    const char* footer = contents.data() + contents.size() - 48;
    if (get_fixed64(footer + 40) != kTableMagic)
      throw std::runtime_error("bad table magic in " + index_path);
    BlockHandle metaindex_h, index_h;
    const char* p = metaindex_h.decode(footer, footer + 40);
    if (!p || !(p = index_h.decode(p, footer + 40)))
      throw std::runtime_error("bad footer");
    // walk the index block → data blocks → entries
    auto index_entries = read_block(contents, index_h);
    for (auto& ie : index_entries) {
      BlockHandle h;
      const char* q = h.decode(ie.second.data(), ie.second.data() + ie.second.size());
      if (!q) throw std::runtime_error("bad block handle");
      for (auto& kv : read_block(contents, h)) {
        if (kv.first.empty()) {
          // BundleHeaderProto field 1 = num_shards (varint)
          const char* hp = kv.second.data();
          const char* he = hp + kv.second.size();
          num_shards_ = 1;
          while (hp < he) {
            uint64_t tag, val;
            if (!(hp = get_varint64(hp, he, &tag))) break;
            if ((tag >> 3) == 1 && (tag & 7) == 0) {
              if ((hp = get_varint64(hp, he, &val))) {
                num_shards_ = static_cast<int>(val);
              }
              break;
            }
            break;  // other fields: stop (only num_shards matters)
          }
          if (num_shards_ < 1) num_shards_ = 1;
          continue;
        }
        BundleEntry e;
        if (!decode_entry(kv.second, &e))
          throw std::runtime_error("bad BundleEntryProto for " + kv.first);
        entries_[kv.first] = e;
      }
    }
  }

  bool contains(const std::string& name) const { return entries_.count(name) > 0; }

  py::list list_tensors() const {
    py::list out;
    for (auto& it : entries_) {
      py::dict d;
      d["name"] = it.first;
      d["dtype"] = it.second.dtype;
      d["shape"] = it.second.shape;
      d["size"] = it.second.size;
      d["shard_id"] = it.second.shard_id;
      d["offset"] = it.second.offset;
      out.append(d);
    }
    return out;
  }

  py::tuple entry_info(const std::string& name) const {
    auto it = entries_.find(name);
    if (it == entries_.end()) throw py::key_error(name);
    return py::make_tuple(it->second.dtype, it->second.shape, it->second.size);
  }

  py::bytes read(const std::string& name) {
    auto it = entries_.find(name);
    if (it == entries_.end()) throw py::key_error(name);
    const BundleEntry& e = it->second;
    char shard_name[64];
    snprintf(shard_name, sizeof(shard_name), ".data-%05d-of-%05d",
             e.shard_id, num_shards_);
    std::ifstream f(prefix_ + shard_name, std::ios::binary);
    if (!f) throw std::runtime_error("cannot open data shard for " + name);
    std::string buf(static_cast<size_t>(e.size), '\0');
    {
      py::gil_scoped_release rel;
      f.seekg(e.offset);
      f.read(&buf[0], e.size);
      if (!f) throw std::runtime_error("short read for " + name);
      uint32_t crc = crc32c::value(buf.data(), buf.size());
      if (e.crc32c_masked != 0 && crc32c::mask(crc) != e.crc32c_masked)
        throw std::runtime_error("crc32c mismatch for " + name);
    }
    return py::bytes(buf);
  }

 private:
  static std::vector<std::pair<std::string, std::string>> read_block(
      const std::string& file, const BlockHandle& h) {
    if (h.offset + h.size + 5 > file.size())
      throw std::runtime_error("block out of range");
    const char* data = file.data() + h.offset;
    // verify trailer crc
    char type = file[h.offset + h.size];
    uint32_t crc = crc32c::value(data, h.size);
    crc = crc32c::extend(crc, &type, 1);
    uint32_t stored = get_fixed32(file.data() + h.offset + h.size + 1);
    if (crc32c::mask(crc) != stored)
      throw std::runtime_error("block crc mismatch");
    if (type != 0) throw std::runtime_error("compressed blocks unsupported");
    // decode entries
    if (h.size < 4) throw std::runtime_error("block too small");
    uint32_t num_restarts = get_fixed32(data + h.size - 4);
    size_t restarts_off = h.size - 4 - 4ull * num_restarts;
    const char* p = data;
    const char* limit = data + restarts_off;
    std::vector<std::pair<std::string, std::string>> out;
    std::string key;
    while (p < limit) {
      uint32_t shared, non_shared, vlen;
      p = get_varint32(p, limit, &shared);
      if (!p) throw std::runtime_error("bad entry varint");
      p = get_varint32(p, limit, &non_shared);
      if (!p) throw std::runtime_error("bad entry varint");
      p = get_varint32(p, limit, &vlen);
      if (!p) throw std::runtime_error("bad entry varint");
      if (p + non_shared + vlen > limit)
        throw std::runtime_error("entry out of range");
      key.resize(shared);
      key.append(p, non_shared);
      p += non_shared;
      out.emplace_back(key, std::string(p, vlen));
      p += vlen;
    }
    return out;
  }

  std::string prefix_;
  int num_shards_ = 1;
  std::map<std::string, BundleEntry> entries_;
};

// ---------------------------------------------------------------------------

PYBIND11_MODULE(_bundle, m) {
  m.doc() = "TF-V2 tensor-bundle checkpoint codec (rocps native)";
  py::class_<BundleWriter>(m, "BundleWriter")
      .def(py::init<const std::string&, int>(), py::arg("prefix"),
           py::arg("num_shards") = 1)
      .def("add", &BundleWriter::add, py::arg("name"), py::arg("dtype"),
           py::arg("shape"), py::arg("data"))
      .def("finish", &BundleWriter::finish);
  py::class_<BundleReader>(m, "BundleReader")
      .def(py::init<const std::string&>())
      .def("contains", &BundleReader::contains)
      .def("list_tensors", &BundleReader::list_tensors)
      .def("entry_info", &BundleReader::entry_info)
      .def("read", &BundleReader::read);
  m.def("masked_crc32c", [](py::buffer b) {
    py::buffer_info info = b.request();
    return crc32c::mask(crc32c::value(static_cast<const char*>(info.ptr),
                                      static_cast<size_t>(info.size) * info.itemsize));
  });
}
