// Hand-written proto3 wire codec for the nerrf.trace schema (C++).
//
// Same contract as nerrf_amd/wire/codec.py (schema: proto/trace.proto).
// This image ships no protoc, so the (tiny, frozen) wire format is
// implemented directly: varints, zigzag sint64, length-delimited strings,
// the nested google.protobuf.Timestamp, and the EventBatch envelope.
//
// Used by the native ingest extension (columnar batch decode on the hot
// path) and by the standalone collector daemon.
#pragma once

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <string_view>
#include <vector>

namespace nerrf::wire {

struct Event {
  int64_t ts_sec = 0;
  int32_t ts_nsec = 0;
  uint32_t pid = 0;
  uint32_t tid = 0;
  std::string comm;
  std::string syscall;
  std::string path;
  std::string new_path;
  uint32_t flags = 0;
  int64_t ret_val = 0;
  uint64_t bytes = 0;
  std::string inode;
  uint32_t mode = 0;
  uint64_t uid = 0;
  uint64_t gid = 0;
  std::vector<std::string> dependencies;

  double timestamp() const { return double(ts_sec) + double(ts_nsec) * 1e-9; }
};

// ---------------------------------------------------------------------------
// varint primitives
// ---------------------------------------------------------------------------

inline void write_varint(std::string& out, uint64_t v) {
  while (true) {
    uint8_t b = v & 0x7F;
    v >>= 7;
    if (v) {
      out.push_back(char(b | 0x80));
    } else {
      out.push_back(char(b));
      return;
    }
  }
}

inline uint64_t read_varint(const uint8_t* buf, size_t len, size_t& pos) {
  uint64_t result = 0;
  int shift = 0;
  while (true) {
    if (pos >= len) throw std::runtime_error("truncated varint");
    uint8_t b = buf[pos++];
    result |= uint64_t(b & 0x7F) << shift;
    if (!(b & 0x80)) return result;
    shift += 7;
    if (shift >= 70) throw std::runtime_error("varint too long");
  }
}

inline uint64_t zigzag_encode(int64_t v) {
  return (uint64_t(v) << 1) ^ uint64_t(v >> 63);
}
inline int64_t zigzag_decode(uint64_t v) {
  return int64_t(v >> 1) ^ -int64_t(v & 1);
}

constexpr int WT_VARINT = 0, WT_I64 = 1, WT_LEN = 2, WT_I32 = 5;

inline void write_tag(std::string& out, int field, int wt) {
  write_varint(out, uint64_t(field) << 3 | wt);
}

inline void write_string_field(std::string& out, int field, std::string_view s) {
  if (s.empty()) return;
  write_tag(out, field, WT_LEN);
  write_varint(out, s.size());
  out.append(s.data(), s.size());
}

inline void write_uint_field(std::string& out, int field, uint64_t v) {
  if (!v) return;
  write_tag(out, field, WT_VARINT);
  write_varint(out, v);
}

inline void skip_field(const uint8_t* buf, size_t len, size_t& pos, int wt) {
  switch (wt) {
    case WT_VARINT:
      read_varint(buf, len, pos);
      break;
    case WT_I64:
      if (len - pos < 8) throw std::runtime_error("truncated i64");
      pos += 8;
      break;
    case WT_LEN: {
      uint64_t n = read_varint(buf, len, pos);
      // overflow-safe bound: a crafted 64-bit length must not wrap pos
      if (n > len - pos) throw std::runtime_error("truncated skip");
      pos += n;
      break;
    }
    case WT_I32:
      if (len - pos < 4) throw std::runtime_error("truncated i32");
      pos += 4;
      break;
    default:
      throw std::runtime_error("unsupported wire type");
  }
}

// ---------------------------------------------------------------------------
// messages
// ---------------------------------------------------------------------------

inline std::string encode_timestamp(int64_t sec, int32_t nsec) {
  std::string out;
  if (sec) {
    write_tag(out, 1, WT_VARINT);
    write_varint(out, uint64_t(sec));
  }
  if (nsec) {
    write_tag(out, 2, WT_VARINT);
    // sign-extend to 64 bits like protobuf/the Python codec do: negative
    // nanos (invalid in a well-formed Timestamp but representable) must
    // round-trip identically in both codecs, not truncate to 32 bits
    write_varint(out, uint64_t(int64_t(nsec)));
  }
  return out;
}

inline void decode_timestamp(const uint8_t* buf, size_t len, int64_t& sec,
                             int32_t& nsec) {
  size_t pos = 0;
  while (pos < len) {
    uint64_t key = read_varint(buf, len, pos);
    int field = int(key >> 3), wt = int(key & 7);
    if (field == 1 && wt == WT_VARINT)
      sec = int64_t(read_varint(buf, len, pos));
    else if (field == 2 && wt == WT_VARINT)
      nsec = int32_t(read_varint(buf, len, pos));
    else
      skip_field(buf, len, pos, wt);
  }
}

inline std::string encode_event(const Event& ev) {
  std::string out;
  if (ev.ts_sec || ev.ts_nsec) {
    std::string ts = encode_timestamp(ev.ts_sec, ev.ts_nsec);
    write_tag(out, 1, WT_LEN);
    write_varint(out, ts.size());
    out += ts;
  }
  write_uint_field(out, 2, ev.pid);
  write_uint_field(out, 3, ev.tid);
  write_string_field(out, 4, ev.comm);
  write_string_field(out, 5, ev.syscall);
  write_string_field(out, 6, ev.path);
  write_string_field(out, 7, ev.new_path);
  write_uint_field(out, 8, ev.flags);
  if (ev.ret_val) {
    write_tag(out, 9, WT_VARINT);
    write_varint(out, zigzag_encode(ev.ret_val));
  }
  write_uint_field(out, 10, ev.bytes);
  write_string_field(out, 11, ev.inode);
  write_uint_field(out, 12, ev.mode);
  write_uint_field(out, 13, ev.uid);
  write_uint_field(out, 14, ev.gid);
  for (const auto& dep : ev.dependencies) write_string_field(out, 15, dep);
  return out;
}

inline Event decode_event(const uint8_t* buf, size_t len) {
  Event ev;
  size_t pos = 0;
  while (pos < len) {
    uint64_t key = read_varint(buf, len, pos);
    int field = int(key >> 3), wt = int(key & 7);
    if (wt == WT_LEN) {
      uint64_t n = read_varint(buf, len, pos);
      if (n > len - pos) throw std::runtime_error("truncated field");
      const char* p = reinterpret_cast<const char*>(buf + pos);
      switch (field) {
        case 1:
          decode_timestamp(buf + pos, n, ev.ts_sec, ev.ts_nsec);
          break;
        case 4: ev.comm.assign(p, n); break;
        case 5: ev.syscall.assign(p, n); break;
        case 6: ev.path.assign(p, n); break;
        case 7: ev.new_path.assign(p, n); break;
        case 11: ev.inode.assign(p, n); break;
        case 15: ev.dependencies.emplace_back(p, n); break;
        default: break;  // unknown LEN field skipped
      }
      pos += n;
    } else if (wt == WT_VARINT) {
      uint64_t v = read_varint(buf, len, pos);
      switch (field) {
        case 2: ev.pid = uint32_t(v); break;
        case 3: ev.tid = uint32_t(v); break;
        case 8: ev.flags = uint32_t(v); break;
        case 9: ev.ret_val = zigzag_decode(v); break;
        case 10: ev.bytes = v; break;
        case 12: ev.mode = uint32_t(v); break;
        case 13: ev.uid = v; break;
        case 14: ev.gid = v; break;
        default: break;
      }
    } else {
      skip_field(buf, len, pos, wt);
    }
  }
  return ev;
}

inline std::string encode_event_batch(const std::vector<Event>& events) {
  std::string out;
  for (const auto& ev : events) {
    std::string payload = encode_event(ev);
    write_tag(out, 1, WT_LEN);
    write_varint(out, payload.size());
    out += payload;
  }
  return out;
}

inline std::vector<Event> decode_event_batch(const uint8_t* buf, size_t len) {
  std::vector<Event> events;
  size_t pos = 0;
  while (pos < len) {
    uint64_t key = read_varint(buf, len, pos);
    int field = int(key >> 3), wt = int(key & 7);
    if (field == 1 && wt == WT_LEN) {
      uint64_t n = read_varint(buf, len, pos);
      if (n > len - pos) throw std::runtime_error("truncated event");
      events.push_back(decode_event(buf + pos, n));
      pos += n;
    } else {
      skip_field(buf, len, pos, wt);
    }
  }
  return events;
}

}  // namespace nerrf::wire
