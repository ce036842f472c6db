// Per-stage interval/counter profiler (reference: scanner/util/profiler.h).
// One Profiler per pipeline-stage thread; intervals are labeled [start,end)
// nanosecond spans; counters are monotonic. The Python client turns a set of
// these into a Chrome trace (scanner_amd/profiler.py).
#pragma once

#include <atomic>
#include <chrono>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include "common.h"

namespace sca {

enum class ProfilerLevel : i32 { Debug = 0, Info = 1, Important = 2 };

inline i64 now_ns() {
  return std::chrono::duration_cast<std::chrono::nanoseconds>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

class Profiler {
 public:
  struct Interval {
    std::string label;
    i64 start_ns;
    i64 end_ns;
  };

  explicit Profiler(ProfilerLevel level = ProfilerLevel::Info)
      : level_(level) {}

  void add_interval(const std::string& label, i64 start_ns, i64 end_ns,
                    ProfilerLevel level = ProfilerLevel::Info) {
    if (level < level_) return;
    std::lock_guard<std::mutex> l(mu_);
    intervals_.push_back({label, start_ns, end_ns});
  }

  void increment(const std::string& counter, i64 n = 1) {
    std::lock_guard<std::mutex> l(mu_);
    counters_[counter] += n;
  }

  std::vector<Interval> intervals() const {
    std::lock_guard<std::mutex> l(mu_);
    return intervals_;
  }
  std::unordered_map<std::string, i64> counters() const {
    std::lock_guard<std::mutex> l(mu_);
    return counters_;
  }

  // RAII interval helper
  class Scope {
   public:
    Scope(Profiler* p, std::string label,
          ProfilerLevel level = ProfilerLevel::Info)
        : p_(p), label_(std::move(label)), level_(level), start_(now_ns()) {}
    ~Scope() {
      if (p_) p_->add_interval(label_, start_, now_ns(), level_);
    }

   private:
    Profiler* p_;
    std::string label_;
    ProfilerLevel level_;
    i64 start_;
  };

 private:
  ProfilerLevel level_;
  mutable std::mutex mu_;
  std::vector<Interval> intervals_;
  std::unordered_map<std::string, i64> counters_;
};

}  // namespace sca
