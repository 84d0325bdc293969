#pragma once
#include "tbb_stub.hpp"
