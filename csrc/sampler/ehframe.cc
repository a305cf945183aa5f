// .eh_frame stack-delta unwinder (x86-64).
//
// The native replacement for the fork's eBPF DWARF unwinder — SURVEY.md
// §7 calls this "the single biggest piece the fork owns": userspace
// stack-delta table generation from .eh_frame CFI, plus a fast
// table-driven walk over the PERF_SAMPLE_STACK_USER bytes captured by
// the sampler (csrc/sampler/sampler.cc dwarf mode).
//
// Table model (mirrors the classic stack-delta design): one row per CFI
// location with
//   cfa_reg  : 0 = CFA off RSP, 1 = CFA off RBP, 2 = unsupported (expr)
//   cfa_off  : CFA = reg + cfa_off
//   fp_off   : RBP save slot at CFA + fp_off (INT32_MIN = not saved)
// The return address on x86-64 is always at CFA - 8. Unwinding one
// frame: CFA = reg+off; RA = mem[CFA-8]; RBP = mem[CFA+fp_off] (if
// saved); RSP = CFA; repeat against the copied stack bytes.
//
// Parsing covers the CFI subset emitted by gcc/clang for C/C++/Rust:
// def_cfa{,_register,_offset,_sf,_offset_sf}, advance_loc{,1,2,4},
// set_loc, offset{,_extended,_extended_sf} (rbp + ra tracked),
// restore{,_extended}, remember/restore_state, GNU_args_size (ignored),
// def_cfa_expression / expression (marks the range unsupported — the
// walker stops rather than guessing).

#include <algorithm>
#include <cstdint>
#include <cstring>
#include <limits>
#include <map>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

namespace parca_unwind {

constexpr int32_t kNoFp = std::numeric_limits<int32_t>::min();
constexpr uint8_t kCfaSp = 0;
constexpr uint8_t kCfaFp = 1;
constexpr uint8_t kCfaBad = 2;

struct Row {
  uint64_t pc;  // module-relative vaddr
  uint8_t cfa_reg;
  int32_t cfa_off;
  int32_t fp_off;
};

// Compact two-level stack-delta table: the round-1
// SoA cost 17 B/row, which at the 400 MB budget dominated the agent's
// RSS. Rows are now 8-byte packed entries grouped into 64 KiB pc pages
// (page key = pc >> 16): the page index resolves the high pc bits, the
// row keeps only pc_lo u16 plus i16 offsets. CFA/FP offsets beyond
// +-32 KiB (rare: huge frames) escape to a side table addressed by the
// two i16 fields. 8 B/row + ~1% page overhead = 2.1x smaller tables;
// the eBPF fork packs its maps the same way for the same reason.
struct PackedRow {
  uint16_t pc_lo;
  uint8_t flags;  // bits0-1: cfa_reg; bit2: no fp; bit3: offsets escaped
  uint8_t pad;
  int16_t cfa_off;
  int16_t fp_off;
};
static_assert(sizeof(PackedRow) == 8, "packed row layout");

constexpr uint8_t kFlagNoFp = 0x4;
constexpr uint8_t kFlagEscape = 0x8;

struct Module {
  std::vector<uint64_t> page_keys;  // pc >> 16, sorted ascending
  std::vector<uint32_t> page_first;  // [n_pages + 1] row-index prefix
  std::vector<PackedRow> rows;
  std::vector<std::pair<int32_t, int32_t>> escaped;  // (cfa_off, fp_off)

  size_t n_rows() const { return rows.size(); }

  size_t bytes() const {
    return page_keys.size() * 8 + page_first.size() * 4 +
           rows.size() * sizeof(PackedRow) + escaped.size() * 8;
  }

  // Index of the row governing `rel`, or -1. The governing row may live
  // in an earlier page than rel's own (rows cover until the next row;
  // FDE ends emit explicit kCfaBad rows, so crossing back is safe).
  ptrdiff_t lookup(uint64_t rel) const {
    if (rows.empty()) return -1;
    uint64_t page = rel >> 16;
    size_t lo = 0, hi = page_keys.size();
    while (lo < hi) {
      size_t mid = (lo + hi) / 2;
      if (page_keys[mid] <= page)
        lo = mid + 1;
      else
        hi = mid;
    }
    if (lo == 0) return -1;
    size_t p = lo - 1;
    if (page_keys[p] < page)  // rel past the page's rows: its last row
      return static_cast<ptrdiff_t>(page_first[p + 1]) - 1;
    uint16_t target = static_cast<uint16_t>(rel & 0xFFFF);
    size_t rlo = page_first[p], rhi = page_first[p + 1];
    while (rlo < rhi) {
      size_t mid = (rlo + rhi) / 2;
      if (rows[mid].pc_lo <= target)
        rlo = mid + 1;
      else
        rhi = mid;
    }
    if (rlo == page_first[p]) {
      // every row in this page starts above rel: governing row is the
      // previous page's last (if any)
      if (p == 0) return -1;
      return static_cast<ptrdiff_t>(page_first[p]) - 1;
    }
    return static_cast<ptrdiff_t>(rlo) - 1;
  }

  // Decoded view of one row.
  void decode(size_t i, uint8_t* cfa_reg, int32_t* cfa_off,
              int32_t* fp_off) const {
    const PackedRow& r = rows[i];
    *cfa_reg = r.flags & 0x3;
    if (r.flags & kFlagEscape) {
      uint32_t idx = (static_cast<uint32_t>(static_cast<uint16_t>(r.fp_off))
                      << 16) |
                     static_cast<uint16_t>(r.cfa_off);
      *cfa_off = escaped[idx].first;
      *fp_off = escaped[idx].second;
    } else {
      *cfa_off = r.cfa_off;
      *fp_off = r.fp_off;
    }
    if (r.flags & kFlagNoFp) *fp_off = kNoFp;
  }
};

// -- DWARF primitive readers ----------------------------------------------

class Cursor {
 public:
  Cursor(const uint8_t* data, size_t size, size_t pos = 0)
      : data_(data), size_(size), pos_(pos) {}

  bool ok() const { return pos_ <= size_; }
  size_t pos() const { return pos_; }
  void seek(size_t p) { pos_ = p; }
  bool eof() const { return pos_ >= size_; }

  uint8_t u8() { return pos_ < size_ ? data_[pos_++] : (fail(), 0); }
  uint16_t u16() {
    uint16_t v = 0;
    read(&v, 2);
    return v;
  }
  uint32_t u32() {
    uint32_t v = 0;
    read(&v, 4);
    return v;
  }
  uint64_t u64() {
    uint64_t v = 0;
    read(&v, 8);
    return v;
  }
  uint64_t uleb() {
    uint64_t result = 0;
    int shift = 0;
    while (pos_ < size_) {
      uint8_t b = data_[pos_++];
      result |= uint64_t(b & 0x7f) << shift;
      if (!(b & 0x80)) return result;
      shift += 7;
      if (shift > 63) break;
    }
    fail();
    return result;
  }
  int64_t sleb() {
    int64_t result = 0;
    int shift = 0;
    uint8_t b = 0;
    while (pos_ < size_) {
      b = data_[pos_++];
      result |= int64_t(b & 0x7f) << shift;
      shift += 7;
      if (!(b & 0x80)) {
        if (shift < 64 && (b & 0x40)) result |= -(int64_t(1) << shift);
        return result;
      }
      if (shift > 63) break;
    }
    fail();
    return result;
  }
  void skip(size_t n) { pos_ += n; }

  bool failed() const { return failed_; }

 private:
  void read(void* out, size_t n) {
    if (pos_ + n > size_) {
      fail();
      return;
    }
    memcpy(out, data_ + pos_, n);
    pos_ += n;
  }
  void fail() {
    failed_ = true;
    pos_ = size_ + 1;
  }
  const uint8_t* data_;
  size_t size_;
  size_t pos_ = 0;
  bool failed_ = false;
};

// DW_EH_PE pointer decoding. `field_vaddr` = vaddr of the encoded field
// (for pcrel).
uint64_t decode_pointer(Cursor& c, uint8_t enc, uint64_t field_vaddr) {
  if (enc == 0xff /* omit */) return 0;
  uint64_t base = 0;
  switch (enc & 0x70) {
    case 0x10:
      base = field_vaddr;
      break;  // pcrel
    case 0x00:
      break;  // abs
    default:
      base = 0;  // textrel/datarel unsupported; rare in .eh_frame
  }
  uint64_t value = 0;
  switch (enc & 0x0f) {
    case 0x01:
      value = c.uleb();
      break;
    case 0x02:
      value = c.u16();
      break;
    case 0x03:
      value = c.u32();
      break;
    case 0x04:
      value = c.u64();
      break;
    case 0x09:
      value = static_cast<uint64_t>(c.sleb());
      break;
    case 0x0a:
      value = static_cast<uint64_t>(static_cast<int16_t>(c.u16()));
      break;
    case 0x0b:
      value = static_cast<uint64_t>(static_cast<int32_t>(c.u32()));
      break;
    case 0x0c:
      value = c.u64();
      break;
    case 0x00:
      value = c.u64();
      break;  // "absptr"
    default:
      value = c.u32();
  }
  return base + value;
}

size_t pointer_size(uint8_t enc) {
  switch (enc & 0x0f) {
    case 0x02:
    case 0x0a:
      return 2;
    case 0x03:
    case 0x0b:
      return 4;
    case 0x00:
    case 0x04:
    case 0x0c:
      return 8;
    default:
      return 0;  // uleb/sleb variable
  }
}

struct CIE {
  uint64_t code_align = 1;
  int64_t data_align = -8;
  uint8_t fde_encoding = 0x03;  // udata4 default-ish
  bool has_aug_data = false;
  std::vector<uint8_t> initial_instructions;
};

struct RegState {
  uint8_t cfa_reg = kCfaSp;
  int64_t cfa_off = 8;
  int64_t fp_off = kNoFp;
  bool fp_saved = false;
};

// Parse .eh_frame bytes; section_vaddr is the vaddr of the section start
// in the module's link-time address space. Appends rows.
void parse_eh_frame(const uint8_t* data, size_t size, uint64_t section_vaddr,
                    std::vector<Row>& rows, size_t max_rows) {
  std::unordered_map<uint64_t, CIE> cies;  // keyed by section offset
  Cursor top(data, size);

  while (!top.eof() && rows.size() < max_rows) {
    size_t entry_start = top.pos();
    uint64_t length = top.u32();
    if (length == 0) break;  // terminator
    bool dwarf64 = false;
    if (length == 0xffffffff) {
      length = top.u64();
      dwarf64 = true;
    }
    size_t content_start = top.pos();
    size_t entry_end = content_start + length;
    if (entry_end > size || top.failed()) break;

    uint64_t id = dwarf64 ? top.u64() : top.u32();
    if (id == 0) {
      // CIE
      CIE cie;
      uint8_t version = top.u8();
      std::string aug;
      while (true) {
        char ch = static_cast<char>(top.u8());
        if (ch == 0) break;
        aug.push_back(ch);
        if (aug.size() > 8) break;
      }
      cie.code_align = top.uleb();
      cie.data_align = top.sleb();
      if (version == 1)
        top.u8();  // RA register
      else
        top.uleb();
      if (!aug.empty() && aug[0] == 'z') {
        cie.has_aug_data = true;
        uint64_t aug_len = top.uleb();
        size_t aug_end = top.pos() + aug_len;
        for (size_t i = 1; i < aug.size(); ++i) {
          if (aug[i] == 'R') {
            cie.fde_encoding = top.u8();
          } else if (aug[i] == 'P') {
            uint8_t penc = top.u8();
            decode_pointer(top, penc, section_vaddr + top.pos());
          } else if (aug[i] == 'L') {
            top.u8();
          } else if (aug[i] == 'S') {
            // signal frame; no payload
          }
        }
        top.seek(aug_end);
      }
      if (top.pos() < entry_end)
        cie.initial_instructions.assign(data + top.pos(), data + entry_end);
      cies[entry_start] = cie;
    } else {
      // FDE: id is the distance from this field back to the CIE.
      size_t cie_pos = content_start - id;
      auto it = cies.find(cie_pos);
      if (it == cies.end()) {
        top.seek(entry_end);
        continue;
      }
      const CIE& cie = it->second;
      uint64_t pc_begin = decode_pointer(
          top, cie.fde_encoding, section_vaddr + top.pos());
      size_t rng_sz = pointer_size(cie.fde_encoding);
      uint64_t pc_range;
      switch (rng_sz) {
        case 2:
          pc_range = top.u16();
          break;
        case 4:
          pc_range = top.u32();
          break;
        case 8:
          pc_range = top.u64();
          break;
        default:
          pc_range = top.uleb();
      }
      if (cie.has_aug_data) {
        uint64_t aug_len = top.uleb();
        top.skip(aug_len);
      }
      if (top.failed() || top.pos() > entry_end) {
        top.seek(entry_end);
        continue;
      }

      // Run CFI: CIE initial instructions then FDE instructions.
      RegState st;
      std::vector<RegState> stack;
      uint64_t loc = pc_begin;
      size_t rows_before = rows.size();
      auto emit = [&](uint64_t at) {
        uint8_t reg = st.cfa_reg;
        int32_t coff =
            (st.cfa_off > INT32_MAX || st.cfa_off < INT32_MIN)
                ? (reg = kCfaBad, 0)
                : static_cast<int32_t>(st.cfa_off);
        int32_t foff = st.fp_saved && st.fp_off >= INT32_MIN + 1 &&
                               st.fp_off <= INT32_MAX
                           ? static_cast<int32_t>(st.fp_off)
                           : kNoFp;
        if (!rows.empty() && rows.back().pc == at) {
          rows.back() = Row{at, reg, coff, foff};
        } else {
          rows.push_back(Row{at, reg, coff, foff});
        }
      };

      auto run = [&](const uint8_t* ins, size_t n) {
        Cursor c(ins, n);
        emit(loc);
        while (!c.eof() && rows.size() < max_rows) {
          uint8_t op = c.u8();
          uint8_t hi = op >> 6;
          uint8_t low = op & 0x3f;
          if (hi == 1) {  // advance_loc
            loc += low * cie.code_align;
            emit(loc);
          } else if (hi == 2) {  // offset reg, uleb
            uint64_t off = c.uleb();
            if (low == 6) {  // rbp
              st.fp_saved = true;
              st.fp_off = static_cast<int64_t>(off) * cie.data_align;
              emit(loc);
            }
          } else if (hi == 3) {  // restore reg
            if (low == 6) {
              st.fp_saved = false;
              emit(loc);
            }
          } else {
            switch (op) {
              case 0x00:
                break;  // nop
              case 0x01: {  // set_loc
                loc = decode_pointer(c, cie.fde_encoding,
                                     section_vaddr + top.pos() + c.pos());
                emit(loc);
                break;
              }
              case 0x02:
                loc += c.u8() * cie.code_align;
                emit(loc);
                break;
              case 0x03:
                loc += c.u16() * cie.code_align;
                emit(loc);
                break;
              case 0x04:
                loc += c.u32() * cie.code_align;
                emit(loc);
                break;
              case 0x05: {  // offset_extended
                uint64_t reg = c.uleb();
                uint64_t off = c.uleb();
                if (reg == 6) {
                  st.fp_saved = true;
                  st.fp_off = static_cast<int64_t>(off) * cie.data_align;
                  emit(loc);
                }
                break;
              }
              case 0x06: {  // restore_extended
                uint64_t reg = c.uleb();
                if (reg == 6) {
                  st.fp_saved = false;
                  emit(loc);
                }
                break;
              }
              case 0x07:  // undefined
              case 0x08:  // same_value
                c.uleb();
                break;
              case 0x09:  // register
                c.uleb();
                c.uleb();
                break;
              case 0x0a:
                stack.push_back(st);
                break;
              case 0x0b:
                if (!stack.empty()) {
                  st = stack.back();
                  stack.pop_back();
                  emit(loc);
                }
                break;
              case 0x0c: {  // def_cfa reg, off
                uint64_t reg = c.uleb();
                uint64_t off = c.uleb();
                st.cfa_reg = reg == 7 ? kCfaSp : reg == 6 ? kCfaFp : kCfaBad;
                st.cfa_off = static_cast<int64_t>(off);
                emit(loc);
                break;
              }
              case 0x0d: {  // def_cfa_register
                uint64_t reg = c.uleb();
                st.cfa_reg = reg == 7 ? kCfaSp : reg == 6 ? kCfaFp : kCfaBad;
                emit(loc);
                break;
              }
              case 0x0e:  // def_cfa_offset
                st.cfa_off = static_cast<int64_t>(c.uleb());
                emit(loc);
                break;
              case 0x0f: {  // def_cfa_expression
                uint64_t n2 = c.uleb();
                c.skip(n2);
                st.cfa_reg = kCfaBad;
                emit(loc);
                break;
              }
              case 0x10: {  // expression reg
                uint64_t reg = c.uleb();
                uint64_t n2 = c.uleb();
                c.skip(n2);
                if (reg == 6) {
                  st.fp_saved = false;
                  emit(loc);
                }
                break;
              }
              case 0x11: {  // offset_extended_sf
                uint64_t reg = c.uleb();
                int64_t off = c.sleb();
                if (reg == 6) {
                  st.fp_saved = true;
                  st.fp_off = off * cie.data_align;
                  emit(loc);
                }
                break;
              }
              case 0x12: {  // def_cfa_sf
                uint64_t reg = c.uleb();
                int64_t off = c.sleb();
                st.cfa_reg = reg == 7 ? kCfaSp : reg == 6 ? kCfaFp : kCfaBad;
                st.cfa_off = off * cie.data_align;
                emit(loc);
                break;
              }
              case 0x13:  // def_cfa_offset_sf
                st.cfa_off = c.sleb() * cie.data_align;
                emit(loc);
                break;
              case 0x14:  // val_offset
              case 0x15:  // val_offset_sf
                c.uleb();
                c.uleb();
                break;
              case 0x16: {  // val_expression
                c.uleb();
                uint64_t n2 = c.uleb();
                c.skip(n2);
                break;
              }
              case 0x2e:  // GNU_args_size
                c.uleb();
                break;
              default:
                // Unknown opcode: abandon this FDE's remaining rows.
                return;
            }
          }
        }
      };

      run(cie.initial_instructions.data(), cie.initial_instructions.size());
      size_t fde_ins_start = top.pos();
      run(data + fde_ins_start, entry_end - fde_ins_start);

      // Terminate the FDE's range so the walker never applies its last
      // row past pc_end.
      if (rows.size() > rows_before)
        rows.push_back(Row{pc_begin + pc_range, kCfaBad, 0, kNoFp});
    }
    top.seek(entry_end);
    if (top.failed()) break;
  }
}

// -- unwinder -------------------------------------------------------------

struct Mapping {
  uint64_t start;
  uint64_t end;
  uint64_t bias;  // runtime_addr - link_vaddr
  int module_id;
};

class Unwinder {
 public:
  int add_module(std::vector<Row> rows) {
    std::sort(rows.begin(), rows.end(),
              [](const Row& a, const Row& b) { return a.pc < b.pc; });
    // De-duplicate equal pcs IN PLACE (later rows win): libtorch-scale
    // modules parse to tens of millions of rows, and a second vector
    // here doubled the build's transient footprint — which glibc then
    // retained, dominating agent RSS.
    size_t w = 0;
    for (size_t i = 0; i < rows.size(); ++i) {
      if (w != 0 && rows[w - 1].pc == rows[i].pc)
        rows[w - 1] = rows[i];
      else
        rows[w++] = rows[i];
    }
    rows.resize(w);
    Module m;
    m.rows.reserve(rows.size());
    for (auto& r : rows) {
      uint64_t page = r.pc >> 16;
      if (m.page_keys.empty() || m.page_keys.back() != page) {
        m.page_keys.push_back(page);
        m.page_first.push_back(static_cast<uint32_t>(m.rows.size()));
      }
      PackedRow pr{};
      pr.pc_lo = static_cast<uint16_t>(r.pc & 0xFFFF);
      pr.flags = r.cfa_reg & 0x3;
      int32_t fp = r.fp_off;
      if (fp == kNoFp) {
        pr.flags |= kFlagNoFp;
        fp = 0;
      }
      if (r.cfa_off >= INT16_MIN && r.cfa_off <= INT16_MAX &&
          fp >= INT16_MIN && fp <= INT16_MAX) {
        pr.cfa_off = static_cast<int16_t>(r.cfa_off);
        pr.fp_off = static_cast<int16_t>(fp);
      } else {
        pr.flags |= kFlagEscape;
        uint32_t idx = static_cast<uint32_t>(m.escaped.size());
        m.escaped.emplace_back(r.cfa_off, fp);
        pr.cfa_off = static_cast<int16_t>(idx & 0xFFFF);
        pr.fp_off = static_cast<int16_t>(idx >> 16);
      }
      m.rows.push_back(pr);
    }
    m.page_first.push_back(static_cast<uint32_t>(m.rows.size()));
    // reserve() above used the PRE-dedup count; give the slack back
    // (tens of MB per libtorch-scale module).
    m.rows.shrink_to_fit();
    m.page_keys.shrink_to_fit();
    m.page_first.shrink_to_fit();
    m.escaped.shrink_to_fit();
    modules_.push_back(std::move(m));
    return static_cast<int>(modules_.size()) - 1;
  }

  void set_mappings(uint32_t pid, std::vector<Mapping> maps) {
    std::sort(maps.begin(), maps.end(),
              [](const Mapping& a, const Mapping& b) {
                return a.start < b.start;
              });
    mappings_[pid] = std::move(maps);
  }

  void drop_process(uint32_t pid) { mappings_.erase(pid); }

  // Unwind using copied stack bytes. sp_base is the SP register value at
  // capture time == address of stack[0].
  std::vector<uint64_t> unwind(uint32_t pid, uint64_t ip, uint64_t sp,
                               uint64_t bp, const uint8_t* stack,
                               size_t stack_len, int max_frames) const {
    std::vector<uint64_t> out;
    auto mit = mappings_.find(pid);
    if (mit == mappings_.end()) return out;
    const auto& maps = mit->second;
    uint64_t sp_base = sp;

    bool hit_dump_end = false;
    auto read_u64 = [&](uint64_t addr, uint64_t* v) -> bool {
      if (addr < sp_base || addr + 8 > sp_base + stack_len) {
        // CFA walked past the copied stack window: frames beyond this
        // point exist but were not captured (bounded dump) — account
        // for them instead of losing them silently.
        if (addr >= sp_base + stack_len) hit_dump_end = true;
        return false;
      }
      memcpy(v, stack + (addr - sp_base), 8);
      return true;
    };

    out.push_back(ip);
    for (int depth = 0; depth < max_frames; ++depth) {
      // find mapping
      const Mapping* map = nullptr;
      {
        size_t lo = 0, hi = maps.size();
        while (lo < hi) {
          size_t mid = (lo + hi) / 2;
          if (maps[mid].start <= ip)
            lo = mid + 1;
          else
            hi = mid;
        }
        if (lo == 0) break;
        const Mapping& cand = maps[lo - 1];
        if (ip >= cand.end) break;
        map = &cand;
      }
      if (map->module_id < 0 ||
          map->module_id >= static_cast<int>(modules_.size()))
        break;
      const Module& m = modules_[map->module_id];
      uint64_t rel = ip - map->bias;
      ptrdiff_t row = m.lookup(rel);
      if (row < 0) break;
      uint8_t cfa_reg;
      int32_t cfa_off, fp_off;
      m.decode(static_cast<size_t>(row), &cfa_reg, &cfa_off, &fp_off);
      if (cfa_reg == kCfaBad) break;
      uint64_t cfa = (cfa_reg == kCfaSp ? sp : bp) +
                     static_cast<int64_t>(cfa_off);
      uint64_t ra = 0;
      if (!read_u64(cfa - 8, &ra)) break;
      if (ra == 0) break;
      if (fp_off != kNoFp) {
        uint64_t saved_bp;
        if (read_u64(cfa + fp_off, &saved_bp)) bp = saved_bp;
      }
      sp = cfa;
      // Return addresses point AFTER the call; step back one byte for
      // row/symbol attribution of the caller.
      ip = ra - 1;
      out.push_back(ra);
      if (out.size() >= static_cast<size_t>(max_frames)) break;
    }
    if (hit_dump_end) ++stacks_truncated_;
    return out;
  }

  uint64_t stacks_truncated() const { return stacks_truncated_; }

  size_t n_modules() const { return modules_.size(); }
  size_t module_rows(int id) const { return modules_.at(id).n_rows(); }
  size_t module_bytes(int id) const { return modules_.at(id).bytes(); }
  size_t total_bytes() const {
    size_t n = 0;
    for (const auto& m : modules_) n += m.bytes();
    return n;
  }

 private:
  std::vector<Module> modules_;
  std::unordered_map<uint32_t, std::vector<Mapping>> mappings_;
  // Stacks whose CFA walk ran off the end of the copied stack dump
  // (deep C++/torch stacks quietly losing their roots, VERDICT.md
  // next#8). mutable: unwind() is logically const.
  mutable uint64_t stacks_truncated_ = 0;
};

}  // namespace parca_unwind
