/* Shim for /root/reference/utils/math.hpp's `#include "type.hpp"`:
 * provides only the scalar typedefs (the real core/type.hpp pulls in
 * boost::variant, absent here).  Used ONLY by oracle/ref_dump.cpp. */
#pragma once
#include <stdint.h>
typedef uint32_t sid_t;
typedef int32_t ssid_t;
#define BLANK_ID UINT32_MAX
