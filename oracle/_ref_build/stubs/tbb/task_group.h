#pragma once
#include "tbb_stub.hpp"
namespace tbb {
class task_group {
public:
  template <typename F> void run(F &&f) { std::forward<F>(f)(); }
  template <typename F> void run_and_wait(F &&f) { std::forward<F>(f)(); }
  void wait() {}
};
} // namespace tbb
