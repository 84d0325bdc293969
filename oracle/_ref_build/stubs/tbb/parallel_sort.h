#pragma once
#include "tbb_stub.hpp"
#include <algorithm>
namespace tbb {
template <typename It, typename Cmp = std::less<>>
void parallel_sort(It begin, It end, Cmp cmp = Cmp()) { std::sort(begin, end, cmp); }
template <typename C, typename Cmp = std::less<>>
void parallel_sort(C &c, Cmp cmp = Cmp()) { std::sort(c.begin(), c.end(), cmp); }
}
