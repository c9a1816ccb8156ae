"""Minimal FlatBuffers builders for the Gamma schemas, used by the test /
bench harness to play the role of the Go caller (which uses the real
flatbuffers library against idl/fbs/table.fbs and doc.fbs).

Forward layout (root table first, strings/vectors after, vtables at the
end) — identical scheme to csrc/fbs.hpp's writer; valid FlatBuffers.
"""
import struct

DATA_INT, DATA_LONG, DATA_FLOAT, DATA_DOUBLE, DATA_STRING, DATA_VECTOR, \
    DATA_BOOL, DATA_DATE, DATA_STRINGARRAY = range(9)


class _Buf:
    def __init__(self):
        self.b = bytearray()

    def pad4(self):
        while len(self.b) % 4:
            self.b.append(0)

    def w32(self, v):
        self.b += struct.pack("<I", v & 0xFFFFFFFF)

    def w32at(self, pos, v):
        self.b[pos:pos + 4] = struct.pack("<I", v & 0xFFFFFFFF)

    def ws32at(self, pos, v):
        self.b[pos:pos + 4] = struct.pack("<i", v)

    def string(self, s):
        """append string object, return its position"""
        self.pad4()
        pos = len(self.b)
        data = s if isinstance(s, bytes) else s.encode()
        self.w32(len(data))
        self.b += data
        self.b.append(0)
        return pos

    def byte_vector(self, data):
        self.pad4()
        pos = len(self.b)
        self.w32(len(data))
        self.b += bytes(data)
        return pos

    def vtable(self, entries):
        """entries: list of u16; returns pos"""
        while len(self.b) % 2:
            self.b.append(0)
        pos = len(self.b)
        for e in entries:
            self.b += struct.pack("<H", e)
        return pos


def build_doc(fields):
    """fields: list of (name, value_bytes, data_type). Returns bytes of a
    gamma_api.Doc (doc.fbs: Doc{fields:[Field{name,value,data_type}]})."""
    B = _Buf()
    B.w32(0)                    # root uoffset
    B.pad4()
    doc_t = len(B.b)
    B.w32(0)                    # soffset -> vtable
    doc_fields_slot = len(B.b)
    B.w32(0)                    # fields vector uoffset
    B.w32at(0, doc_t)

    B.pad4()
    vec_pos = len(B.b)
    B.w32(len(fields))
    vec_elems = len(B.b)
    for _ in fields:
        B.w32(0)
    B.w32at(doc_fields_slot, vec_pos - doc_fields_slot)

    ftab, fname_slot, fval_slot = [], [], []
    for i, (name, value, dt) in enumerate(fields):
        B.pad4()
        t = len(B.b)
        ftab.append(t)
        B.w32(0)                # soffset
        fname_slot.append(len(B.b))
        B.w32(0)
        fval_slot.append(len(B.b))
        B.w32(0)
        B.b.append(dt & 0xFF)
        B.pad4()
        B.w32at(vec_elems + 4 * i, t - (vec_elems + 4 * i))

    for i, (name, value, dt) in enumerate(fields):
        s = B.string(name)
        B.w32at(fname_slot[i], s - fname_slot[i])
        v = B.byte_vector(value)
        B.w32at(fval_slot[i], v - fval_slot[i])

    doc_vt = B.vtable([8, 8, 4])
    B.ws32at(doc_t, doc_t - doc_vt)
    f_vt = B.vtable([10, 13, 4, 8, 12])
    for t in ftab:
        B.ws32at(t, t - f_vt)
    return bytes(B.b)


def build_table(name, scalar_fields, vec_name, dimension, index_type,
                index_params, store_type="MemoryOnly", extra_vecs=()):
    """table.fbs Table: name(0), fields(1), vectors_info(2), index_type(3),
    index_params(4). scalar_fields: list of (name, data_type).
    extra_vecs: additional (name, dimension) vector fields of a
    multi-vector table."""
    B = _Buf()
    B.w32(0)
    B.pad4()
    t = len(B.b)
    B.w32(0)                    # soffset
    slots = {}
    for fid in ("name", "fields", "vectors", "index_type", "index_params"):
        slots[fid] = len(B.b)
        B.w32(0)
    B.w32at(0, t)

    # fields vector of FieldInfo tables
    B.pad4()
    fv = len(B.b)
    B.w32(len(scalar_fields))
    fv_elems = len(B.b)
    for _ in scalar_fields:
        B.w32(0)
    B.w32at(slots["fields"], fv - slots["fields"])

    fi_tabs, fi_name_slots = [], []
    for i, (fname, dt) in enumerate(scalar_fields):
        B.pad4()
        ft = len(B.b)
        fi_tabs.append(ft)
        B.w32(0)                # soffset
        fi_name_slots.append(len(B.b))
        B.w32(0)                # name uoffset
        B.b.append(dt & 0xFF)   # data_type (byte)
        B.b.append(0)           # is_index (bool)
        B.pad4()
        B.w32at(fv_elems + 4 * i, ft - (fv_elems + 4 * i))

    # vectors_info vector of VectorInfo (primary first, then extras)
    all_vecs = [(vec_name, dimension)] + list(extra_vecs)
    B.pad4()
    vv = len(B.b)
    B.w32(len(all_vecs))
    vv_elems = len(B.b)
    for _ in all_vecs:
        B.w32(0)
    B.w32at(slots["vectors"], vv - slots["vectors"])

    v_tabs, v_name_slots, v_store_slots = [], [], []
    for i, (vn, vd) in enumerate(all_vecs):
        B.pad4()
        vt = len(B.b)
        v_tabs.append(vt)
        B.w32(0)                    # soffset
        v_name_slots.append(len(B.b))
        B.w32(0)                    # name
        B.b += struct.pack("<i", vd)  # dimension inline @8
        B.b.append(DATA_VECTOR)     # data_type @12
        B.b.append(1)               # is_index @13
        B.pad4()
        v_store_slots.append(len(B.b))
        B.w32(0)                    # store_type @16
        B.w32at(vv_elems + 4 * i, vt - (vv_elems + 4 * i))

    # strings
    s = B.string(name)
    B.w32at(slots["name"], s - slots["name"])
    for i, (fname, dt) in enumerate(scalar_fields):
        s = B.string(fname)
        B.w32at(fi_name_slots[i], s - fi_name_slots[i])
    for i, (vn, vd) in enumerate(all_vecs):
        s = B.string(vn)
        B.w32at(v_name_slots[i], s - v_name_slots[i])
        s = B.string(store_type)
        B.w32at(v_store_slots[i], s - v_store_slots[i])
    s = B.string(index_type)
    B.w32at(slots["index_type"], s - slots["index_type"])
    s = B.string(index_params)
    B.w32at(slots["index_params"], s - slots["index_params"])

    # vtables
    # Table: fields name=0@4, fields=1@8, vectors_info=2@12, index_type=3@16,
    #        index_params=4@20
    tbl_vt = B.vtable([14, 24, 4, 8, 12, 16, 20])
    B.ws32at(t, t - tbl_vt)
    # FieldInfo: name=0@4, data_type=1@8, is_index=2@9 (index_type absent)
    fi_vt = B.vtable([10, 10, 4, 8, 9])
    for ft in fi_tabs:
        B.ws32at(ft, ft - fi_vt)
    # VectorInfo: name=0@4, data_type=1@12, is_index=2@13, dimension=3@8,
    #             store_type=4@16
    vi_vt = B.vtable([14, 20, 4, 12, 13, 8, 16])
    for vt in v_tabs:
        B.ws32at(vt, vt - vi_vt)
    return bytes(B.b)
