// Native flow-state tracker + telemetry TSV parser — pure C++ core.
//
// The reference's telemetry ingestion is a per-line Python loop
// (traffic_classifier.py:147-171).  This is the framework's line-rate
// equivalent: bulk-parse a whole poll buffer, update the bidirectional
// flow state with the exact reference arithmetic
// (traffic_classifier.py:63-96 — including division guards and
// ACTIVE/INACTIVE rules).  Column layout matches flow/state.py so the two
// implementations are interchangeable.
//
// Header is Python-free so the sanitizer harness (tools/flowtable_san.cpp,
// -fsanitize=address,undefined) exercises the identical code the pybind11
// extension (flowtable.cpp) ships.
#pragma once

#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <string>
#include <string_view>
#include <unordered_map>
#include <vector>

namespace tcsdn {

// column indices — MUST match flow/state.py
enum Col {
  TIME_START = 0,
  F_PKTS, F_BYTES, F_DELTA_PKTS, F_DELTA_BYTES,
  F_INST_PPS, F_AVG_PPS, F_INST_BPS, F_AVG_BPS, F_LAST_TIME,
  R_PKTS, R_BYTES, R_DELTA_PKTS, R_DELTA_BYTES,
  R_INST_PPS, R_AVG_PPS, R_INST_BPS, R_AVG_BPS, R_LAST_TIME,
  F_ACTIVE, R_ACTIVE,
  F_PREV_PKTS, F_PREV_BYTES, F_PREV_TIME,
  R_PREV_PKTS, R_PREV_BYTES, R_PREV_TIME,
  STATE_COLS
};

constexpr int kFeatureCols[12] = {
    F_DELTA_PKTS, F_DELTA_BYTES, F_INST_PPS, F_AVG_PPS, F_INST_BPS, F_AVG_BPS,
    R_DELTA_PKTS, R_DELTA_BYTES, R_INST_PPS, R_AVG_PPS, R_INST_BPS, R_AVG_BPS};

struct Meta {
  std::string datapath, inport, ethsrc, ethdst, outport;
};

class NativeFlowTable {
 public:
  NativeFlowTable() { state_.reserve(1024 * STATE_COLS); }

  int64_t records = 0;
  int64_t bad_lines = 0;

  size_t size() const { return n_; }
  const double* row(size_t i) const { return state_.data() + i * STATE_COLS; }
  const std::vector<Meta>& metas() const { return metas_; }

  // ---- single-record update (reference traffic_classifier.py:157-165)
  int64_t update(double time, std::string_view datapath, std::string_view inport,
                 std::string_view ethsrc, std::string_view ethdst,
                 std::string_view outport, double packets, double bytes) {
    std::string key = make_key(datapath, ethsrc, ethdst);
    auto it = index_.find(key);
    if (it != index_.end()) {
      update_forward(it->second, packets, bytes, time);
      return (int64_t)it->second;
    }
    std::string rkey = make_key(datapath, ethdst, ethsrc);
    it = index_.find(rkey);
    if (it != index_.end()) {
      update_reverse(it->second, packets, bytes, time);
      return (int64_t)it->second;
    }
    return (int64_t)create(std::move(key), time, datapath, inport, ethsrc, ethdst,
                           outport, packets, bytes);
  }

  // ---- bulk ingestion: parse a whole buffer of telemetry lines
  int64_t feed_buffer(std::string_view buf) {
    int64_t accepted = 0;
    size_t pos = 0;
    while (pos < buf.size()) {
      size_t eol = buf.find('\n', pos);
      std::string_view line =
          buf.substr(pos, eol == std::string_view::npos ? std::string_view::npos
                                                        : eol - pos);
      pos = (eol == std::string_view::npos) ? buf.size() : eol + 1;
      if (!line.empty() && line.back() == '\r') line.remove_suffix(1);
      if (feed_line(line) >= 0) ++accepted;
    }
    return accepted;
  }

  // returns slot or -1
  int64_t feed_line(std::string_view line) {
    if (line.size() < 4 || line.compare(0, 4, "data") != 0) return -1;
    // split on tabs: data, time, dpid, in_port, src, dst, out_port, pkts, bytes
    std::string_view f[9];
    int nf = 0;
    size_t pos = 0;
    while (nf < 9 && pos <= line.size()) {
      size_t tab = line.find('\t', pos);
      f[nf++] = line.substr(pos, tab == std::string_view::npos
                                     ? std::string_view::npos
                                     : tab - pos);
      if (tab == std::string_view::npos) break;
      pos = tab + 1;
    }
    if (nf < 9) { ++bad_lines; return -1; }
    double t, pk, by;
    if (!parse_num(f[1], t) || !parse_num(f[7], pk) || !parse_num(f[8], by)) {
      ++bad_lines;
      return -1;
    }
    int64_t slot = update(t, f[2], f[3], f[4], f[5], f[6], pk, by);
    ++records;
    return slot;
  }

 private:
  std::vector<double> state_;
  std::vector<Meta> metas_;
  std::unordered_map<std::string, size_t> index_;
  size_t n_ = 0;

  double* mrow(size_t i) { return state_.data() + i * STATE_COLS; }

  static std::string make_key(std::string_view dp, std::string_view a,
                              std::string_view b) {
    std::string k;
    k.reserve(dp.size() + a.size() + b.size() + 2);
    k.append(dp).push_back('\x1f');
    k.append(a).push_back('\x1f');
    k.append(b);
    return k;
  }

  static bool parse_num(std::string_view v, double& out) {
    if (v.empty()) return false;
    char buf[32];
    if (v.size() >= sizeof(buf)) return false;
    std::memcpy(buf, v.data(), v.size());
    buf[v.size()] = 0;
    char* end = nullptr;
    out = std::strtod(buf, &end);
    return end == buf + v.size();
  }

  size_t create(std::string key, double time, std::string_view dp,
                std::string_view inport, std::string_view src,
                std::string_view dst, std::string_view outport, double packets,
                double bytes) {
    size_t slot = n_++;
    state_.resize(n_ * STATE_COLS, 0.0);
    index_.emplace(std::move(key), slot);
    metas_.push_back(Meta{std::string(dp), std::string(inport), std::string(src),
                          std::string(dst), std::string(outport)});
    double* s = mrow(slot);
    s[TIME_START] = time;
    s[F_PKTS] = packets;
    s[F_BYTES] = bytes;
    s[F_LAST_TIME] = time;
    s[R_LAST_TIME] = time;
    s[F_ACTIVE] = 1.0;  // traffic_classifier.py:47
    s[R_ACTIVE] = 0.0;  // :59
    s[F_PREV_PKTS] = packets;
    s[F_PREV_BYTES] = bytes;
    s[F_PREV_TIME] = time;
    s[R_PREV_TIME] = time;
    return slot;
  }

  void update_forward(size_t slot, double packets, double bytes, double time) {
    double* s = mrow(slot);
    s[F_PREV_PKTS] = s[F_PKTS];
    s[F_PREV_BYTES] = s[F_BYTES];
    s[F_PREV_TIME] = s[F_LAST_TIME];
    s[F_DELTA_PKTS] = packets - s[F_PKTS];
    s[F_PKTS] = packets;
    if (time != s[TIME_START]) s[F_AVG_PPS] = packets / (time - s[TIME_START]);
    if (time != s[F_LAST_TIME])
      s[F_INST_PPS] = s[F_DELTA_PKTS] / (time - s[F_LAST_TIME]);
    s[F_DELTA_BYTES] = bytes - s[F_BYTES];
    s[F_BYTES] = bytes;
    if (time != s[TIME_START]) s[F_AVG_BPS] = bytes / (time - s[TIME_START]);
    if (time != s[F_LAST_TIME])
      s[F_INST_BPS] = s[F_DELTA_BYTES] / (time - s[F_LAST_TIME]);
    s[F_LAST_TIME] = time;
    s[F_ACTIVE] = (s[F_DELTA_BYTES] == 0.0 || s[F_DELTA_PKTS] == 0.0) ? 0.0 : 1.0;
  }

  void update_reverse(size_t slot, double packets, double bytes, double time) {
    double* s = mrow(slot);
    s[R_PREV_PKTS] = s[R_PKTS];
    s[R_PREV_BYTES] = s[R_BYTES];
    s[R_PREV_TIME] = s[R_LAST_TIME];
    s[R_DELTA_PKTS] = packets - s[R_PKTS];
    s[R_PKTS] = packets;
    if (time != s[TIME_START]) s[R_AVG_PPS] = packets / (time - s[TIME_START]);
    if (time != s[R_LAST_TIME])
      s[R_INST_PPS] = s[R_DELTA_PKTS] / (time - s[R_LAST_TIME]);
    s[R_DELTA_BYTES] = bytes - s[R_BYTES];
    s[R_BYTES] = bytes;
    if (time != s[TIME_START]) s[R_AVG_BPS] = bytes / (time - s[TIME_START]);
    if (time != s[R_LAST_TIME])
      s[R_INST_BPS] = s[R_DELTA_BYTES] / (time - s[R_LAST_TIME]);
    s[R_LAST_TIME] = time;
    s[R_ACTIVE] = (s[R_DELTA_BYTES] == 0.0 || s[R_DELTA_PKTS] == 0.0) ? 0.0 : 1.0;
  }
};

}  // namespace tcsdn
