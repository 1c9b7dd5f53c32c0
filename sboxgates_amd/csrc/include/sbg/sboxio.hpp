// sboxio.hpp — S-box table loading (parity: sboxgates.c:988-1040).
#pragma once

#include <string>

#include "sbg/common.hpp"

namespace sbg {

// Loads an S-box from a file of 2^n (1 <= n <= 8) whitespace-separated hex
// values < 256, optionally XOR-permuting the input by `permute`. On
// success fills sbox[256] (tail zeroed) and *num_inputs. Returns false and
// sets *err on failure.
bool load_sbox_file(const std::string& path, int permute, u8 sbox[256],
                    u32* num_inputs, std::string* err);

// Same, from an in-memory table of `len` entries.
bool load_sbox_table(const u8* table, int len, int permute, u8 sbox[256],
                     u32* num_inputs, std::string* err);

}  // namespace sbg
