// Sanitizer harness for the native flow table (SURVEY.md §5 race/sanitizer
// parity): exercises the identical core the pybind11 extension ships
// (csrc/flowtable_core.h) under -fsanitize=address,undefined with
// adversarial and randomized telemetry, plus property checks against the
// reference semantics (traffic_classifier.py:63-96, :144-171).
//
// Built and run by tests/test_native_flowtable.py::test_sanitizer_harness.

#include <cassert>
#include <cinttypes>
#include <cstdio>
#include <random>
#include <sstream>

#include "../traffic_classifier_sdn_amd/csrc/flowtable_core.h"

using tcsdn::NativeFlowTable;

static void adversarial_lines() {
  NativeFlowTable t;
  const char* cases[] = {
      "",
      "\n",
      "data",
      "data\t",
      "garbage\t1\t2\t3\t4\t5\t6\t7\t8",
      "data\tnotanumber\t1\t1\ta\tb\t2\t1\t1",
      "data\t1.0\t1\t1\ta\tb\t2\tNaNx\t1",
      "data\t1.0\t1\t1\ta\tb\t2\t1",                      // 8 fields only
      "data\t1.0\t1\t1\ta\tb\t2\t1\t1\textra\tfields",    // extra fields
      "data\t99999999999999999999999999999999999\t1\t1\ta\tb\t2\t1\t1",
      "data\t1e309\t1\t1\ta\tb\t2\t1\t1",                 // inf time
      "data\t1.0\t\t\t\t\t\t1\t1",                        // empty ids
  };
  for (const char* c : cases) t.feed_line(c);
  // very long MAC-ish fields
  std::string longsrc(100000, 'a');
  std::string line = "data\t1.0\t1\t1\t" + longsrc + "\tb\t2\t1\t1";
  t.feed_line(line);
  // a number field at the 32-byte parse_num boundary
  std::string n31(31, '1');
  t.feed_line("data\t" + n31 + "\t1\t1\tx\ty\t2\t3\t4");
  std::string n32(32, '1');
  t.feed_line("data\t" + n32 + "\t1\t1\tx\ty\t2\t3\t4");
  std::printf("adversarial: size=%zu records=%" PRId64 " bad=%" PRId64 "\n",
              t.size(), t.records, t.bad_lines);
}

static void random_stream() {
  std::mt19937_64 rng(7);
  NativeFlowTable t;
  char buf[256];
  std::string chunk;
  for (int poll = 0; poll < 200; ++poll) {
    chunk.clear();
    for (int f = 0; f < 64; ++f) {
      int a = (int)(rng() % 24), b = (int)(rng() % 24);
      std::snprintf(buf, sizeof(buf),
                    "data\t%d\t1\t%d\th%02d\th%02d\t2\t%llu\t%llu\n", 1000 + poll,
                    1 + (int)(rng() % 3), a, b,
                    (unsigned long long)(rng() % 100000),
                    (unsigned long long)(rng() % 100000000));
      chunk += buf;
    }
    t.feed_buffer(chunk);
  }
  assert(t.size() <= 24 * 24);
  assert(t.records > 0);
  // feature invariants: deltas finite, averages finite, layout readable
  for (size_t i = 0; i < t.size(); ++i) {
    const double* s = t.row(i);
    for (int j = 0; j < 12; ++j) {
      double v = s[tcsdn::kFeatureCols[j]];
      assert(v == v);  // no NaN from the guarded divisions
    }
  }
  std::printf("random: size=%zu records=%" PRId64 " bad=%" PRId64 "\n",
              t.size(), t.records, t.bad_lines);
}

static void forward_reverse_resolution() {
  NativeFlowTable t;
  // first direction observed becomes "forward" (traffic_classifier.py:157-165)
  t.feed_line("data\t1\t1\t1\tA\tB\t2\t10\t1000");
  t.feed_line("data\t1\t1\t2\tB\tA\t1\t5\t500");   // reverse of the same flow
  t.feed_line("data\t2\t1\t1\tA\tB\t2\t20\t2000");
  assert(t.size() == 1);
  const double* s = t.row(0);
  assert(s[tcsdn::F_PKTS] == 20.0);
  assert(s[tcsdn::R_PKTS] == 5.0);
  assert(s[tcsdn::F_DELTA_PKTS] == 10.0);
  // same-timestamp update must not divide by zero (guards at :70-74)
  assert(s[tcsdn::R_INST_PPS] == 0.0);
  std::printf("fwd/rev: ok\n");
}

int main() {
  adversarial_lines();
  random_stream();
  forward_reverse_resolution();
  std::printf("SANITIZER HARNESS OK\n");
  return 0;
}
