// hx_internal.h — internals shared between engine.cpp and index.cpp
// (single shared library; not part of the public C-ABI).
#pragma once
#include <string>
#include "../../include/horaedb_hx.h"

namespace hx_int {

// route an error through engine.cpp's thread-local hx_last_error() buffer
hx_status set_error(hx_status code, const std::string& msg);

// the handle's store root directory
const std::string& store_path(hx_handle* h);

}  // namespace hx_int
