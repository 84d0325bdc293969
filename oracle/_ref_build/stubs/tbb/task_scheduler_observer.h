#pragma once
#include "tbb_stub.hpp"
namespace tbb {
class task_scheduler_observer {
public:
  task_scheduler_observer() = default;
  explicit task_scheduler_observer(task_arena &) {}
  virtual ~task_scheduler_observer() = default;
  void observe(bool = true) {}
  virtual void on_scheduler_entry(bool) {}
  virtual void on_scheduler_exit(bool) {}
};
} // namespace tbb
