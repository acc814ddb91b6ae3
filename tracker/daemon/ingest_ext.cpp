// nerrf_amd._ingest — native columnar trace ingest.
//
// The hot path of stream consumption: decode nerrf.trace.EventBatch frames
// straight into columnar numpy arrays (one pass, zero Python-object churn)
// with a persistent string interner, mirroring the layout of
// nerrf_amd.data.trace.EventArray.  The reference implementation it must
// agree with byte-for-byte is nerrf_amd/wire/codec.py (tests enforce this).
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <unordered_map>

#include "wire_codec.h"

namespace py = pybind11;

namespace {

int8_t syscall_id(const std::string& name) {
  static const std::unordered_map<std::string, int8_t> ids = {
      {"openat", 1}, {"write", 2}, {"rename", 3}, {"read", 4}, {"unlink", 5},
      {"chmod", 6},  {"close", 7}, {"mkdir", 8},  {"exec", 9},
  };
  auto it = ids.find(name);
  return it == ids.end() ? 0 : it->second;
}

class ColumnarDecoder {
 public:
  // Decode a list of EventBatch frames into one columnar batch.
  // Returns (ts, pid, syscall, path_id, new_path_id, nbytes, ret_val, comm_id).
  py::tuple decode(const std::vector<py::bytes>& frames) {
    std::vector<nerrf::wire::Event> events;
    for (const auto& frame : frames) {
      std::string_view sv = frame;
      auto batch = nerrf::wire::decode_event_batch(
          reinterpret_cast<const uint8_t*>(sv.data()), sv.size());
      events.insert(events.end(), std::make_move_iterator(batch.begin()),
                    std::make_move_iterator(batch.end()));
    }
    const ssize_t n = ssize_t(events.size());
    py::array_t<double> ts(n);
    py::array_t<int64_t> pid(n), path_id(n), new_path_id(n), nbytes(n),
        ret_val(n), comm_id(n);
    py::array_t<int8_t> syscall(n);
    auto* ts_p = ts.mutable_data();
    auto* pid_p = pid.mutable_data();
    auto* sys_p = syscall.mutable_data();
    auto* path_p = path_id.mutable_data();
    auto* newp_p = new_path_id.mutable_data();
    auto* byt_p = nbytes.mutable_data();
    auto* ret_p = ret_val.mutable_data();
    auto* com_p = comm_id.mutable_data();
    for (ssize_t i = 0; i < n; ++i) {
      const auto& ev = events[i];
      ts_p[i] = ev.timestamp();
      pid_p[i] = ev.pid;
      sys_p[i] = syscall_id(ev.syscall);
      path_p[i] = ev.path.empty() ? -1 : intern(paths_, path_strings_, ev.path);
      newp_p[i] =
          ev.new_path.empty() ? -1 : intern(paths_, path_strings_, ev.new_path);
      byt_p[i] = int64_t(ev.bytes);
      ret_p[i] = ev.ret_val;
      com_p[i] = ev.comm.empty() ? -1 : intern(comms_, comm_strings_, ev.comm);
    }
    return py::make_tuple(ts, pid, syscall, path_id, new_path_id, nbytes,
                          ret_val, comm_id);
  }

  // Incremental sync of interned strings into the Python-side StringTable.
  std::vector<std::string> paths_since(size_t start) const {
    return {path_strings_.begin() + std::min(start, path_strings_.size()),
            path_strings_.end()};
  }
  std::vector<std::string> comms_since(size_t start) const {
    return {comm_strings_.begin() + std::min(start, comm_strings_.size()),
            comm_strings_.end()};
  }
  size_t path_count() const { return path_strings_.size(); }
  size_t comm_count() const { return comm_strings_.size(); }

 private:
  static int64_t intern(std::unordered_map<std::string, int64_t>& table,
                        std::vector<std::string>& strings,
                        const std::string& s) {
    auto [it, inserted] = table.try_emplace(s, int64_t(strings.size()));
    if (inserted) strings.push_back(s);
    return it->second;
  }

  std::unordered_map<std::string, int64_t> paths_, comms_;
  std::vector<std::string> path_strings_, comm_strings_;
};

// Round-trip helpers (tests + tracker-sim fast path).
py::bytes encode_batch_py(py::list events) {
  std::vector<nerrf::wire::Event> evs;
  evs.reserve(events.size());
  for (auto item : events) {
    py::dict d = item.cast<py::dict>();
    nerrf::wire::Event ev;
    if (d.contains("ts_sec")) ev.ts_sec = d["ts_sec"].cast<int64_t>();
    if (d.contains("ts_nsec")) ev.ts_nsec = d["ts_nsec"].cast<int32_t>();
    if (d.contains("pid")) ev.pid = d["pid"].cast<uint32_t>();
    if (d.contains("tid")) ev.tid = d["tid"].cast<uint32_t>();
    if (d.contains("comm")) ev.comm = d["comm"].cast<std::string>();
    if (d.contains("syscall")) ev.syscall = d["syscall"].cast<std::string>();
    if (d.contains("path")) ev.path = d["path"].cast<std::string>();
    if (d.contains("new_path")) ev.new_path = d["new_path"].cast<std::string>();
    if (d.contains("flags")) ev.flags = d["flags"].cast<uint32_t>();
    if (d.contains("ret_val")) ev.ret_val = d["ret_val"].cast<int64_t>();
    if (d.contains("bytes")) ev.bytes = d["bytes"].cast<uint64_t>();
    if (d.contains("inode")) ev.inode = d["inode"].cast<std::string>();
    evs.push_back(std::move(ev));
  }
  return py::bytes(nerrf::wire::encode_event_batch(evs));
}

py::list decode_batch_py(py::bytes frame) {
  std::string_view sv = frame;
  auto events = nerrf::wire::decode_event_batch(
      reinterpret_cast<const uint8_t*>(sv.data()), sv.size());
  py::list out;
  for (const auto& ev : events) {
    py::dict d;
    d["ts_sec"] = ev.ts_sec;
    d["ts_nsec"] = ev.ts_nsec;
    d["pid"] = ev.pid;
    d["tid"] = ev.tid;
    d["comm"] = ev.comm;
    d["syscall"] = ev.syscall;
    d["path"] = ev.path;
    d["new_path"] = ev.new_path;
    d["flags"] = ev.flags;
    d["ret_val"] = ev.ret_val;
    d["bytes"] = ev.bytes;
    d["inode"] = ev.inode;
    out.append(d);
  }
  return out;
}

}  // namespace

PYBIND11_MODULE(_ingest, m) {
  m.doc() = "nerrf-amd native trace ingest (wire codec + columnar decode)";
  py::class_<ColumnarDecoder>(m, "ColumnarDecoder")
      .def(py::init<>())
      .def("decode", &ColumnarDecoder::decode)
      .def("paths_since", &ColumnarDecoder::paths_since)
      .def("comms_since", &ColumnarDecoder::comms_since)
      .def("path_count", &ColumnarDecoder::path_count)
      .def("comm_count", &ColumnarDecoder::comm_count);
  m.def("encode_batch", &encode_batch_py);
  m.def("decode_batch", &decode_batch_py);
}
