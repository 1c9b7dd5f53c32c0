// common.hpp — shared macros and basic types for the sboxgates-mi355x engine.
//
// Brand-new MI355X-native implementation of the capabilities of
// dansarie/sboxgates (reference: /root/reference, C11+MPI). Nothing in this
// tree is a translation of the reference sources; reference file:line
// citations in comments mark *behavioral parity points* only.
#pragma once

#include <cstdint>
#include <cstddef>

#if defined(__HIPCC__)
#define SBG_HD __host__ __device__
#define SBG_DEV __device__
#else
#define SBG_HD
#define SBG_DEV
#endif

namespace sbg {

using u8 = uint8_t;
using u16 = uint16_t;
using u32 = uint32_t;
using u64 = uint64_t;
using i8 = int8_t;
using i32 = int32_t;
using i64 = int64_t;

// Graph capacity, matching the reference (state.h:26): gate ids are u16,
// at most 500 gates per circuit.
constexpr int MAX_GATES = 500;
using gatenum = u16;
constexpr gatenum NO_GATE = static_cast<gatenum>(-1);

}  // namespace sbg
