/*
 * fbs.hpp — minimal FlatBuffers reader/writer (header-only), written from
 * the FlatBuffers binary format spec, for the two Gamma schemas:
 *   gamma_api.Table (idl/fbs/table.fbs) — read only
 *   gamma_api.Doc   (idl/fbs/doc.fbs)   — read + write
 * DataType enum (idl/fbs/types.fbs): INT=0, LONG, FLOAT, DOUBLE, STRING,
 * VECTOR, BOOL, DATE, STRINGARRAY.
 */
#pragma once
#include <stdint.h>
#include <string.h>

#include <string>
#include <vector>

namespace gfb {

enum DataType {
  INT = 0, LONG, FLOAT, DOUBLE, STRING, VECTOR, BOOL, DATE, STRINGARRAY
};

/* ---------------------------------------------------------------- reader */
class Reader {
 public:
  Reader(const void *buf, size_t n) : b_((const uint8_t *)buf), n_(n) {}
  bool valid() const { return b_ && n_ >= 8; }
  uint32_t root() const { return rd32(0); }

  /* absolute position of field `id` inside table at `tpos`, 0 if absent */
  uint32_t field(uint32_t tpos, int id) const {
    if ((size_t)tpos + 4 > n_) return 0;
    int32_t soff;
    memcpy(&soff, b_ + tpos, 4);
    int64_t vpos = (int64_t)tpos - soff;
    if (vpos < 0 || (uint64_t)vpos + 4 > n_) return 0;
    uint16_t vbytes = rd16((uint32_t)vpos);
    uint32_t slot = 4 + 2 * (uint32_t)id;
    if (slot + 2 > vbytes) return 0;
    uint16_t voff = rd16((uint32_t)vpos + slot);
    if (voff == 0) return 0;
    return tpos + voff;
  }
  /* follow a uoffset at pos */
  uint32_t indirect(uint32_t pos) const { return pos + rd32(pos); }

  std::string str_at(uint32_t fpos) const { /* fpos = field() result */
    if (!fpos) return "";
    uint32_t s = indirect(fpos);
    uint32_t len = rd32(s);
    if ((size_t)s + 4 + (size_t)len > n_) return "";
    return std::string((const char *)b_ + s + 4, len);
  }
  uint32_t vec_len(uint32_t fpos) const {
    if (!fpos) return 0;
    uint32_t len = rd32(indirect(fpos));
    /* bounds cap: a corrupt buffer cannot claim more elements than
     * bytes (elements are at least 1 byte / 4-byte offsets) */
    return (size_t)len > n_ ? 0 : len;
  }
  /* absolute position of vector element i (elem_size bytes or offsets) */
  uint32_t vec_elem(uint32_t fpos, uint32_t i, uint32_t elem_size) const {
    uint32_t v = indirect(fpos);
    return v + 4 + i * elem_size;
  }
  /* table-vector element: follow the stored uoffset (0 if out of
   * bounds — field() then rejects it) */
  uint32_t vec_table(uint32_t fpos, uint32_t i) const {
    uint32_t e = vec_elem(fpos, i, 4);
    if ((size_t)e + 4 > n_) return 0;
    uint32_t t = indirect(e);
    return (size_t)t + 4 > n_ ? 0 : t;
  }
  template <class T>
  T scalar(uint32_t fpos, T deflt) const {
    if (!fpos || (size_t)fpos + sizeof(T) > n_) return deflt;
    T v;
    memcpy(&v, b_ + fpos, sizeof(T));
    return v;
  }
  const uint8_t *bytes_at(uint32_t fpos, uint32_t *len) const {
    *len = 0;
    if (!fpos) return nullptr;
    uint32_t v = indirect(fpos);
    *len = rd32(v);
    if ((size_t)v + 4 + (size_t)*len > n_) { *len = 0; return nullptr; }
    return b_ + v + 4;
  }

 private:
  uint16_t rd16(uint32_t p) const {
    uint16_t v = 0;
    if ((size_t)p + 2 <= n_) memcpy(&v, b_ + p, 2);
    return v;
  }
  uint32_t rd32(uint32_t p) const {
    uint32_t v = 0;
    if ((size_t)p + 4 <= n_) memcpy(&v, b_ + p, 4);
    return v;
  }
  const uint8_t *b_;
  size_t n_;
};

/* -------------------------------------------------------- parsed schemas */
struct FieldInfo { /* table.fbs FieldInfo: name(0), data_type(1),
                      is_index(2), index_type(3) */
  std::string name;
  int data_type = 0;
  bool is_index = false;
  int index_type = 0;
};
struct VectorInfo { /* table.fbs VectorInfo: name(0), data_type(1),
                       is_index(2), dimension(3), store_type(4),
                       store_param(5) */
  std::string name;
  int data_type = 0;
  bool is_index = false;
  int dimension = 0;
  std::string store_type, store_param;
};
struct IndexInfo { /* table.fbs IndexInfo: name(0), type(1), field_name(2),
                      field_names(3), params(4) */
  std::string name, type, field_name, params;
  std::vector<std::string> field_names;
};
struct TableSchema { /* table.fbs Table: name(0), fields(1),
                        vectors_info(2), index_type(3), index_params(4),
                        refresh_interval(5), enable_id_cache(6),
                        enable_realtime(7), indexes(8) */
  std::string name, index_type, index_params;
  std::vector<FieldInfo> fields;
  std::vector<VectorInfo> vectors;
  std::vector<IndexInfo> indexes;

  bool parse(const void *buf, size_t n) {
    Reader r(buf, n);
    if (!r.valid()) return false;
    uint32_t t = r.root();
    name = r.str_at(r.field(t, 0));
    uint32_t fv = r.field(t, 1);
    for (uint32_t i = 0; i < r.vec_len(fv); i++) {
      uint32_t ft = r.vec_table(fv, i);
      FieldInfo fi;
      fi.name = r.str_at(r.field(ft, 0));
      fi.data_type = r.scalar<int8_t>(r.field(ft, 1), 0);
      fi.is_index = r.scalar<uint8_t>(r.field(ft, 2), 0) != 0;
      fi.index_type = r.scalar<int32_t>(r.field(ft, 3), 0);
      fields.push_back(std::move(fi));
    }
    uint32_t vv = r.field(t, 2);
    for (uint32_t i = 0; i < r.vec_len(vv); i++) {
      uint32_t vt = r.vec_table(vv, i);
      VectorInfo vi;
      vi.name = r.str_at(r.field(vt, 0));
      vi.data_type = r.scalar<int8_t>(r.field(vt, 1), 0);
      vi.is_index = r.scalar<uint8_t>(r.field(vt, 2), 0) != 0;
      vi.dimension = r.scalar<int32_t>(r.field(vt, 3), 0);
      vi.store_type = r.str_at(r.field(vt, 4));
      vi.store_param = r.str_at(r.field(vt, 5));
      vectors.push_back(std::move(vi));
    }
    index_type = r.str_at(r.field(t, 3));
    index_params = r.str_at(r.field(t, 4));
    uint32_t iv = r.field(t, 8);
    for (uint32_t i = 0; i < r.vec_len(iv); i++) {
      uint32_t it = r.vec_table(iv, i);
      IndexInfo ii;
      ii.name = r.str_at(r.field(it, 0));
      ii.type = r.str_at(r.field(it, 1));
      ii.field_name = r.str_at(r.field(it, 2));
      uint32_t fns = r.field(it, 3);
      for (uint32_t j = 0; j < r.vec_len(fns); j++) {
        uint32_t sp = r.vec_elem(fns, j, 4);
        ii.field_names.push_back(str_direct(buf, n, r.indirect(sp)));
      }
      ii.params = r.str_at(r.field(it, 4));
      indexes.push_back(std::move(ii));
    }
    return true;
  }

  static std::string str_direct(const void *buf, size_t n, uint32_t s) {
    const uint8_t *b = (const uint8_t *)buf;
    if ((size_t)s + 4 > n) return "";
    uint32_t len;
    memcpy(&len, b + s, 4);
    if ((size_t)s + 4 + (size_t)len > n) return "";
    return std::string((const char *)b + s + 4, len);
  }
};

struct DocField { /* doc.fbs Field: name(0), value(1) [ubyte],
                     data_type(2) */
  std::string name;
  std::string value;
  int data_type = 0;
};
struct Doc { /* doc.fbs Doc: fields(0) */
  std::vector<DocField> fields;

  bool parse(const void *buf, size_t n) {
    Reader r(buf, n);
    if (!r.valid()) return false;
    uint32_t t = r.root();
    uint32_t fv = r.field(t, 0);
    for (uint32_t i = 0; i < r.vec_len(fv); i++) {
      uint32_t ft = r.vec_table(fv, i);
      DocField df;
      df.name = r.str_at(r.field(ft, 0));
      uint32_t len = 0;
      const uint8_t *p = r.bytes_at(r.field(ft, 1), &len);
      if (p) df.value.assign((const char *)p, len);
      df.data_type = r.scalar<int8_t>(r.field(ft, 2), 0);
      fields.push_back(std::move(df));
    }
    return true;
  }

  /* Serialize to a valid FlatBuffers Doc (forward layout: root table
   * first, then vectors/strings, vtables at the end; all uoffsets point
   * forward, soffsets to vtables are negative — both legal). */
  std::string serialize() const {
    std::string out;
    auto pad4 = [&]() { while (out.size() % 4) out.push_back(0); };
    auto w32 = [&](uint32_t v) { out.append((const char *)&v, 4); };
    auto w16at = [&](size_t pos, uint16_t v) {
      memcpy(&out[pos], &v, 2);
    };
    auto w32at = [&](size_t pos, uint32_t v) {
      memcpy(&out[pos], &v, 4);
    };
    (void)w16at;

    out.reserve(256);
    w32(0);                       /* [0] root uoffset, patched */
    pad4();
    /* Doc table: soffset32 + fields uoffset */
    size_t doc_t = out.size();
    w32(0);                       /* soffset -> doc vtable (patched) */
    size_t doc_fields_slot = out.size();
    w32(0);                       /* uoffset -> fields vector (patched) */
    w32at(0, (uint32_t)doc_t);    /* root points at doc table */

    pad4();
    size_t vec_pos = out.size();
    w32((uint32_t)fields.size());
    size_t vec_elems = out.size();
    for (size_t i = 0; i < fields.size(); i++) w32(0); /* patched */
    w32at(doc_fields_slot, (uint32_t)(vec_pos - doc_fields_slot));

    /* field tables */
    std::vector<size_t> ftab(fields.size());
    std::vector<size_t> fname_slot(fields.size()), fval_slot(fields.size());
    for (size_t i = 0; i < fields.size(); i++) {
      pad4();
      size_t t = out.size();
      ftab[i] = t;
      w32(0);                     /* soffset -> field vtable */
      fname_slot[i] = out.size();
      w32(0);                     /* name uoffset */
      fval_slot[i] = out.size();
      w32(0);                     /* value uoffset */
      out.push_back((char)fields[i].data_type);
      pad4();
      w32at(vec_elems + 4 * i, (uint32_t)(t - (vec_elems + 4 * i)));
    }
    /* strings + byte vectors */
    for (size_t i = 0; i < fields.size(); i++) {
      pad4();
      size_t s = out.size();
      w32((uint32_t)fields[i].name.size());
      out.append(fields[i].name);
      out.push_back(0);
      w32at(fname_slot[i], (uint32_t)(s - fname_slot[i]));
      pad4();
      size_t v = out.size();
      w32((uint32_t)fields[i].value.size());
      out.append(fields[i].value);
      w32at(fval_slot[i], (uint32_t)(v - fval_slot[i]));
    }
    /* vtables (2-byte aligned) */
    while (out.size() % 2) out.push_back(0);
    size_t doc_vt = out.size();
    {
      uint16_t vt[3] = {8, 8, 4}; /* vt_size, table_size, field0 off */
      out.append((const char *)vt, 6);
    }
    {
      int32_t soff = (int32_t)doc_t - (int32_t)doc_vt;
      memcpy(&out[doc_t], &soff, 4);
    }
    while (out.size() % 2) out.push_back(0);
    size_t f_vt = out.size();
    {
      /* Field table: field0 name @4, field1 value @8, field2 dtype @12 */
      uint16_t vt[5] = {10, 13, 4, 8, 12};
      out.append((const char *)vt, 10);
    }
    for (size_t i = 0; i < fields.size(); i++) {
      int32_t soff = (int32_t)ftab[i] - (int32_t)f_vt;
      memcpy(&out[ftab[i]], &soff, 4);
    }
    return out;
  }
};

}  // namespace gfb
