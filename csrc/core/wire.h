// Clean-room implementation of the FlatBuffers binary format (little-endian),
// sufficient for infinistore's four wire tables. Written from the public
// format specification — no code from google/flatbuffers or the reference.
//
// Wire compatibility goal: buffers produced here parse with stock flatbuffers
// readers generated from the reference schemas
// (/root/reference/src/*.fbs — LocalMetaRequest, RemoteMetaRequest,
// RdmaAllocateResponse, GetMatchLastIndexRequest) and vice versa.
//
// Format summary (from the public spec):
//  * Buffers are built back-to-front; all offsets are 32-bit.
//  * Root: uoffset32 at file position 0 pointing at the root table.
//  * Table: starts with soffset32 to its vtable (table_pos - soffset = vtable
//    pos). Vtable: [u16 vtable_bytes][u16 table_bytes][u16 field_off...] where
//    field_off is relative to the table start (0 = field absent).
//  * Scalars are stored inline in the table; strings/vectors/subtables are
//    stored out-of-line and referenced by uoffset32 relative to the field
//    location.
//  * String: [u32 len][bytes][NUL]. Vector: [u32 count][elems]. Struct:
//    inline, fields aligned to their own size, struct padded to max member
//    alignment.
#pragma once

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

namespace ifs {
namespace wire {

// ---------------------------------------------------------------------------
// Builder
// ---------------------------------------------------------------------------
class Builder {
   public:
    explicit Builder(size_t initial = 1024) {
        storage_.resize(initial < 16 ? 16 : initial);
        used_ = 0;
        min_align_ = 1;
    }

    // End-relative offset (bytes from logical buffer end) of the most recently
    // pushed object's start. 0 means "nothing yet".
    using Offset = uint32_t;

    size_t size() const { return used_; }

    void clear() {
        used_ = 0;
        min_align_ = 1;
        fields_.clear();
    }

    // --- low-level pushes (back-to-front) ---
    void align(size_t a) {
        track_align(a);
        size_t pad = (a - (used_ % a)) % a;
        ensure(pad);
        used_ += pad;
        memset(cur(), 0, pad);
    }

    // Pad so that (used_ + len) % a == 0 — aligns the start of the next
    // len-byte push.
    void pre_align(size_t len, size_t a) {
        track_align(a);
        size_t pad = (a - ((used_ + len) % a)) % a;
        ensure(pad);
        used_ += pad;
        memset(cur(), 0, pad);
    }

    void push_bytes(const void* p, size_t n) {
        ensure(n);
        used_ += n;
        memcpy(cur(), p, n);
    }

    template <typename T>
    void push_scalar(T v) {
        pre_align(sizeof(T), sizeof(T));
        push_bytes(&v, sizeof(T));
    }

    // --- strings / vectors (out-of-line objects; return end-relative offset)
    Offset create_string(const char* s, size_t n) {
        // File order [u32 len][bytes][NUL]; built back-to-front, so any
        // alignment padding is pushed first (it lands after the NUL in the
        // file) such that the len field ends up 4-aligned.
        pre_align(n + 1, 4);
        ensure(1);
        used_ += 1;
        *cur() = 0;  // NUL terminator
        push_bytes(s, n);
        uint32_t len = static_cast<uint32_t>(n);
        push_bytes(&len, 4);
        return static_cast<Offset>(used_);
    }
    Offset create_string(const std::string& s) { return create_string(s.data(), s.size()); }

    template <typename T>
    Offset create_vector(const T* data, size_t n) {
        // [u32 count][elems]; element start aligned to max(4, sizeof(T)).
        size_t elem_size = sizeof(T);
        pre_align(4 + n * elem_size, 4);
        pre_align(n * elem_size, elem_size);
        push_bytes(data, n * elem_size);
        uint32_t cnt = static_cast<uint32_t>(n);
        push_bytes(&cnt, 4);
        return static_cast<Offset>(used_);
    }

    // Vector of offsets (e.g. [string]): values are rewritten as uoffsets
    // relative to each element slot.
    Offset create_offset_vector(const std::vector<Offset>& offs) {
        size_t n = offs.size();
        pre_align(4 + n * 4, 4);
        ensure(n * 4);
        used_ += n * 4;
        uint8_t* base = cur();
        for (size_t i = 0; i < n; i++) {
            // element slot end-relative offset (to slot start):
            uint32_t slot = static_cast<uint32_t>(used_ - i * 4);
            uint32_t rel = slot - offs[i];
            memcpy(base + i * 4, &rel, 4);
        }
        uint32_t cnt = static_cast<uint32_t>(n);
        push_bytes(&cnt, 4);
        return static_cast<Offset>(used_);
    }

    // Vector of structs with explicit element size/alignment (structs may have
    // internal padding, e.g. RemoteBlock is 16 bytes, align 8).
    Offset create_struct_vector(const void* data, size_t n, size_t elem_size, size_t elem_align) {
        pre_align(4 + n * elem_size, 4);
        pre_align(n * elem_size, elem_align);
        push_bytes(data, n * elem_size);
        uint32_t cnt = static_cast<uint32_t>(n);
        push_bytes(&cnt, 4);
        return static_cast<Offset>(used_);
    }

    // --- tables ---
    void start_table() { fields_.clear(); }

    template <typename T>
    void add_scalar(int field_id, T v, T default_v) {
        if (v == default_v) return;
        push_scalar(v);
        note_field(field_id, sizeof(T));
    }

    void add_offset(int field_id, Offset off) {
        if (off == 0) return;
        pre_align(4, 4);
        ensure(4);
        used_ += 4;
        uint32_t field_slot = static_cast<uint32_t>(used_);
        uint32_t rel = field_slot - off;
        memcpy(cur(), &rel, 4);
        note_field(field_id, 4);
    }

    Offset end_table() {
        // Push soffset placeholder (table start).
        push_scalar<int32_t>(0);
        uint32_t table_end = static_cast<uint32_t>(used_);

        int max_id = -1;
        for (auto& f : fields_)
            if (f.id > max_id) max_id = f.id;
        size_t nslots = static_cast<size_t>(max_id + 1);
        uint16_t vt_bytes = static_cast<uint16_t>(4 + nslots * 2);

        // Table size: from table start (soffset) through the furthest inline
        // field end. Field offsets from table start = table_end - f.endrel
        // (end-relative offsets shrink toward the file end).
        uint32_t span = 4;  // the soffset itself
        for (auto& f : fields_) {
            uint32_t fo_end = (table_end - f.endrel) + f.size;
            if (fo_end > span) span = fo_end;
        }
        uint16_t tbl_bytes = static_cast<uint16_t>(span);

        // Write vtable back-to-front: fields (id high→low), table bytes,
        // vtable bytes.
        std::vector<uint16_t> slots(nslots, 0);
        for (auto& f : fields_)
            slots[static_cast<size_t>(f.id)] = static_cast<uint16_t>(table_end - f.endrel);
        for (size_t i = nslots; i-- > 0;) push_scalar<uint16_t>(slots[i]);
        push_scalar<uint16_t>(tbl_bytes);
        push_scalar<uint16_t>(vt_bytes);
        uint32_t vtable_end = static_cast<uint32_t>(used_);

        // Patch the table's soffset: value = vtable_pos_from_table (signed,
        // table_file - vtable_file = vtable_endrel - table_endrel > 0).
        int32_t soff = static_cast<int32_t>(vtable_end) - static_cast<int32_t>(table_end);
        memcpy(storage_.data() + (storage_.size() - table_end), &soff, 4);
        fields_.clear();
        return table_end;
    }

    // Finish: push root uoffset; returns pointer/size of the final buffer.
    void finish(Offset root) {
        pre_align(4, min_align_);
        ensure(4);
        used_ += 4;
        uint32_t slot = static_cast<uint32_t>(used_);
        uint32_t rel = slot - root;
        memcpy(cur(), &rel, 4);
    }

    const uint8_t* data() const { return storage_.data() + (storage_.size() - used_); }

    std::vector<uint8_t> release() {
        std::vector<uint8_t> out(data(), data() + used_);
        return out;
    }

   private:
    struct FieldRec {
        int id;
        uint32_t endrel;  // end-relative offset of field value start
        uint32_t size;    // bytes occupied by the value
    };

    uint8_t* cur() { return storage_.data() + (storage_.size() - used_); }

    void ensure(size_t n) {
        if (used_ + n <= storage_.size()) return;
        size_t ns = storage_.size() * 2;
        while (ns < used_ + n) ns *= 2;
        std::vector<uint8_t> bigger(ns);
        memcpy(bigger.data() + (ns - used_), storage_.data() + (storage_.size() - used_), used_);
        storage_ = std::move(bigger);
    }

    void track_align(size_t a) {
        if (a > min_align_) min_align_ = a;
    }

    void note_field(int id, uint32_t size) {
        fields_.push_back({id, static_cast<uint32_t>(used_), size});
    }

    std::vector<uint8_t> storage_;
    size_t used_;
    size_t min_align_;
    std::vector<FieldRec> fields_;
};

// ---------------------------------------------------------------------------
// Reader
// ---------------------------------------------------------------------------
class Table;

class Reader {
   public:
    Reader(const uint8_t* buf, size_t len) : buf_(buf), len_(len) {}
    bool ok() const { return buf_ && len_ >= 8; }
    inline Table root() const;
    const uint8_t* buf() const { return buf_; }
    size_t len() const { return len_; }

    template <typename T>
    T read(size_t pos) const {
        T v{};
        if (pos + sizeof(T) <= len_) memcpy(&v, buf_ + pos, sizeof(T));
        return v;
    }

   private:
    const uint8_t* buf_;
    size_t len_;
};

class Table {
   public:
    Table() : r_(nullptr, 0), pos_(0) {}
    Table(Reader r, size_t pos) : r_(r), pos_(pos) {}
    bool valid() const { return pos_ != 0 && pos_ < r_.len(); }

    // Returns byte offset of field value within buffer, or 0 if absent.
    size_t field_pos(int field_id) const {
        if (!valid()) return 0;
        int32_t soff = r_.read<int32_t>(pos_);
        size_t vt = static_cast<size_t>(static_cast<int64_t>(pos_) - soff);
        if (vt + 4 > r_.len()) return 0;
        uint16_t vt_bytes = r_.read<uint16_t>(vt);
        size_t slot = 4 + static_cast<size_t>(field_id) * 2;
        if (slot + 2 > vt_bytes) return 0;
        uint16_t fo = r_.read<uint16_t>(vt + slot);
        if (fo == 0) return 0;
        return pos_ + fo;
    }

    template <typename T>
    T scalar(int field_id, T default_v) const {
        size_t p = field_pos(field_id);
        if (!p) return default_v;
        return r_.read<T>(p);
    }

    // Resolve a uoffset field to the target object position (0 if absent).
    size_t indirect(int field_id) const {
        size_t p = field_pos(field_id);
        if (!p) return 0;
        uint32_t rel = r_.read<uint32_t>(p);
        return p + rel;
    }

    std::string str_at(size_t pos) const {
        if (!pos || pos + 4 > r_.len()) return {};
        uint32_t n = r_.read<uint32_t>(pos);
        if (pos + 4 + n > r_.len()) return {};
        return std::string(reinterpret_cast<const char*>(r_.buf() + pos + 4), n);
    }

    std::string string_field(int field_id) const { return str_at(indirect(field_id)); }

    // Vector helpers (length clamped: a valid element needs >= 1 byte).
    size_t vec_len(int field_id) const {
        size_t p = indirect(field_id);
        if (!p) return 0;
        uint32_t n = r_.read<uint32_t>(p);
        return static_cast<size_t>(n) > r_.len() ? 0 : n;
    }
    size_t vec_data(int field_id) const {
        size_t p = indirect(field_id);
        if (!p) return 0;
        return p + 4;
    }

    template <typename T>
    std::vector<T> scalar_vector(int field_id) const {
        std::vector<T> out;
        size_t p = indirect(field_id);
        if (!p) return out;
        uint32_t n = r_.read<uint32_t>(p);
        if (p + 4 + static_cast<size_t>(n) * sizeof(T) > r_.len()) return out;
        out.resize(n);
        memcpy(out.data(), r_.buf() + p + 4, n * sizeof(T));
        return out;
    }

    std::vector<std::string> string_vector(int field_id) const {
        std::vector<std::string> out;
        size_t p = indirect(field_id);
        if (!p) return out;
        uint32_t n = r_.read<uint32_t>(p);
        // Malformed-length guard: each element needs >= 4 bytes of buffer.
        if (static_cast<size_t>(n) > r_.len() / 4) return out;
        out.reserve(n);
        for (uint32_t i = 0; i < n; i++) {
            size_t slot = p + 4 + static_cast<size_t>(i) * 4;
            if (slot + 4 > r_.len()) break;
            uint32_t rel = r_.read<uint32_t>(slot);
            out.push_back(str_at(slot + rel));
        }
        return out;
    }

    // Vector of subtables: position of element i's table.
    Table table_at_vec(int field_id, size_t i) const {
        size_t p = indirect(field_id);
        if (!p) return Table();
        uint32_t n = r_.read<uint32_t>(p);
        if (i >= n) return Table();
        size_t slot = p + 4 + i * 4;
        uint32_t rel = r_.read<uint32_t>(slot);
        return Table(r_, slot + rel);
    }

    const Reader& reader() const { return r_; }
    size_t pos() const { return pos_; }

   private:
    Reader r_;
    size_t pos_;
};

inline Table Reader::root() const {
    if (!ok()) return Table();
    uint32_t rel = read<uint32_t>(0);
    return Table(*this, rel);
}

}  // namespace wire
}  // namespace ifs
