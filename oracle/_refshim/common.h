/* minimal stand-in for the reference's common.h so that datetime.h's
 * INLINE bit-layout functions compile in isolation (building the parity
 * CHECKER only — no reference code is copied or shipped). */
#pragma once
#include <cstdint>
#include <cstdio>
#include <ctime>
#include <string>

/* referenced by an inline fn the harness never calls; stub to parse */
namespace tso {
inline time_t get_timestamp_internal(long long ts) { return (time_t)ts; }
}
