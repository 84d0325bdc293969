// Serial (single-thread) drop-in for the subset of the oneTBB API that the
// KaMinPar label-propagation hot path uses. This is NOT oneTBB: it executes
// every "parallel" construct sequentially, in deterministic order, so that the
// reference LP path compiled against it reproduces the reference's own
// 1-thread deterministic behaviour (the property pinned by
// /root/reference/tests/endtoend/shm_endtoend_test.cc:189-217).
//
// Written from scratch against the public oneTBB API surface; used only to
// build oracle/_ref (the compiled reference used as a parity oracle).
#pragma once

#include <algorithm>
#include <cstddef>
#include <deque>
#include <memory>
#include <functional>
#include <numeric>
#include <utility>
#include <vector>

namespace tbb {

struct split {};

template <typename Index> class blocked_range {
public:
  using const_iterator = Index;
  blocked_range(Index begin, Index end, std::size_t grainsize = 1)
      : _begin(begin), _end(end), _grainsize(grainsize) {}
  Index begin() const { return _begin; }
  Index end() const { return _end; }
  std::size_t grainsize() const { return _grainsize; }
  bool empty() const { return !(_begin < _end); }
  bool is_divisible() const { return false; }

private:
  Index _begin;
  Index _end;
  std::size_t _grainsize;
};

// ---- parallel_for ----
template <typename Index, typename Body>
void parallel_for(Index first, Index last, const Body &body) {
  for (Index i = first; i < last; ++i) {
    body(i);
  }
}

template <typename Index, typename Body>
void parallel_for(Index first, Index last, Index step, const Body &body) {
  for (Index i = first; i < last; i += step) {
    body(i);
  }
}

template <typename Range, typename Body>
auto parallel_for(const Range &range, const Body &body)
    -> decltype((void)body(range)) {
  body(range);
}

// ---- parallel_reduce ----
template <typename Range, typename Value, typename Func, typename Reduction>
Value parallel_reduce(
    const Range &range, const Value &identity, const Func &func, const Reduction &
) {
  return func(range, identity);
}

// ---- parallel_scan: serial two-pass emulation over blocked_range ----
template <typename Range, typename Body> void parallel_scan(const Range &range, Body &body) {
  body(range, /* is_final_scan */ true);
}

// ---- parallel_invoke ----
template <typename... Fs> void parallel_invoke(Fs &&...fs) {
  (std::forward<Fs>(fs)(), ...);
}

// ---- enumerable_thread_specific ----
struct ets_key_usage_type {};

template <typename T> class enumerable_thread_specific {
public:
  using reference = T &;

  enumerable_thread_specific() : _factory([] { return T(); }) {}

  template <
      typename Factory,
      typename = std::enable_if_t<std::is_invocable_r_v<T, Factory>>>
  enumerable_thread_specific(Factory factory) : _factory(std::move(factory)) {}

  // Construct each thread-local instance from the given arguments.
  template <
      typename Arg,
      typename... Args,
      typename = std::enable_if_t<
          !std::is_invocable_r_v<T, Arg> &&
          std::is_constructible_v<T, Arg, Args...>>,
      typename = void>
  enumerable_thread_specific(Arg arg, Args... args)
      : _factory([=] { return T(arg, args...); }) {}

  // T may be immovable (e.g. a type holding immovable members); store by
  // pointer and rely on C++17 guaranteed elision in `new T(factory())`.
  T &local() {
    if (_instances.empty()) {
      _instances.emplace_back(new T(_factory()));
    }
    return *_instances.front();
  }

  bool empty() const { return _instances.empty(); }

  void clear() { _instances.clear(); }

  // Iterator yielding T& over the pointer storage.
  struct iterator {
    typename std::deque<std::unique_ptr<T>>::iterator it;
    T &operator*() const { return **it; }
    T *operator->() const { return it->get(); }
    iterator &operator++() {
      ++it;
      return *this;
    }
    bool operator!=(const iterator &o) const { return it != o.it; }
    bool operator==(const iterator &o) const { return it == o.it; }
  };

  iterator begin() { return iterator{_instances.begin()}; }
  iterator end() { return iterator{_instances.end()}; }

  // range() is only used with tbb::parallel_for(range, body) which we run
  // serially on the whole range; our parallel_for overload passes the
  // container-like range straight through to the body.
  struct Range {
    iterator _begin, _end;
    iterator begin() const { return _begin; }
    iterator end() const { return _end; }
  };
  Range range() { return Range{begin(), end()}; }

  template <typename BinaryOp> T combine(BinaryOp op) const {
    T result{};
    bool first = true;
    for (auto &p : _instances) {
      if (first) {
        result = *p;
        first = false;
      } else {
        result = op(result, *p);
      }
    }
    return result;
  }

  template <typename UnaryOp> void combine_each(UnaryOp op) const {
    for (auto &p : _instances) {
      op(*p);
    }
  }

  // const iteration (some reference call sites combine on const ETS refs)
  struct const_iterator {
    typename std::deque<std::unique_ptr<T>>::const_iterator it;
    const T &operator*() const { return **it; }
    const T *operator->() const { return it->get(); }
    const_iterator &operator++() {
      ++it;
      return *this;
    }
    bool operator!=(const const_iterator &o) const { return it != o.it; }
    bool operator==(const const_iterator &o) const { return it == o.it; }
  };
  const_iterator begin() const { return const_iterator{_instances.begin()}; }
  const_iterator end() const { return const_iterator{_instances.end()}; }

private:
  std::function<T()> _factory;
  mutable std::deque<std::unique_ptr<T>> _instances;
};

// ---- combinable ----
template <typename T> class combinable {
public:
  combinable() : _factory([] { return T(); }) {}
  template <typename Factory> combinable(Factory f) : _factory(std::move(f)) {}

  T &local() {
    if (_instances.empty()) {
      _instances.push_back(_factory());
    }
    return _instances.front();
  }

  template <typename BinaryOp> T combine(BinaryOp op) {
    T result{};
    bool first = true;
    for (auto &v : _instances) {
      if (first) {
        result = v;
        first = false;
      } else {
        result = op(result, v);
      }
    }
    return result;
  }

private:
  std::function<T()> _factory;
  std::deque<T> _instances;
};

// ---- concurrent_vector ----
template <typename T> class concurrent_vector {
public:
  using iterator = typename std::deque<T>::iterator;
  using const_iterator = typename std::deque<T>::const_iterator;

  void push_back(const T &v) { _data.push_back(v); }
  void push_back(T &&v) { _data.push_back(std::move(v)); }
  template <typename... Args> T &emplace_back(Args &&...args) {
    return _data.emplace_back(std::forward<Args>(args)...);
  }

  std::size_t size() const { return _data.size(); }
  bool empty() const { return _data.empty(); }
  void clear() { _data.clear(); }
  void shrink_to_fit() {}

  T &operator[](std::size_t i) { return _data[i]; }
  const T &operator[](std::size_t i) const { return _data[i]; }

  iterator begin() { return _data.begin(); }
  iterator end() { return _data.end(); }
  const_iterator begin() const { return _data.begin(); }
  const_iterator end() const { return _data.end(); }

private:
  std::deque<T> _data;
};

// ---- spin_mutex ----
class spin_mutex {
public:
  class scoped_lock {
  public:
    scoped_lock() = default;
    explicit scoped_lock(spin_mutex &) {}
    void acquire(spin_mutex &) {}
    void release() {}
  };
  void lock() {}
  void unlock() {}
};

// ---- this_task_arena / task_arena ----
namespace this_task_arena {
inline int current_thread_index() { return 0; }
inline int max_concurrency() { return 1; }
template <typename F> auto isolate(F &&f) { return std::forward<F>(f)(); }
} // namespace this_task_arena

class task_arena {
public:
  task_arena(int = 1, unsigned = 1) {}
  void initialize(int = 1) {}
  template <typename F> auto execute(F &&f) { return std::forward<F>(f)(); }
  int max_concurrency() const { return 1; }
};

// ---- global_control ----
class global_control {
public:
  enum parameter { max_allowed_parallelism, thread_stack_size };
  global_control(parameter, std::size_t) {}
};

// ---- static_partitioner etc. (rarely referenced) ----
struct static_partitioner {};
struct auto_partitioner {};
struct simple_partitioner {};

} // namespace tbb

namespace tbb {

// Imperative-body parallel_reduce: body(range) mutates internal state.
template <typename Range, typename Body>
auto parallel_reduce(const Range &range, Body &body) -> decltype((void)body(range)) {
  body(range);
}

// Functional parallel_scan: scan(range, identity, is_final=true).
template <typename Range, typename Value, typename Scan, typename Combine>
Value parallel_scan(
    const Range &range, const Value &identity, const Scan &scan, const Combine &
) {
  return scan(range, identity, true);
}

template <typename T> using cache_aligned_allocator = std::allocator<T>;

} // namespace tbb
