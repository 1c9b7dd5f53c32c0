// sboxio.cpp — S-box table loading.

#include "sbg/sboxio.hpp"

#include <cstdio>
#include <cstring>

namespace sbg {

bool load_sbox_table(const u8* table, int len, int permute, u8 sbox[256],
                     u32* num_inputs, std::string* err) {
  if (len <= 0 || len > 256 || (len & (len - 1)) != 0) {
    if (err != nullptr) *err = "bad number of items in target S-box";
    return false;
  }
  u32 n = 31 - static_cast<u32>(__builtin_clz(static_cast<u32>(len)));
  *num_inputs = n;
  std::memset(sbox, 0, 256);
  if (permute == 0) {
    std::memcpy(sbox, table, static_cast<size_t>(len));
  } else {
    if (permute < 0 || permute >= (1 << n)) {
      if (err != nullptr) *err = "bad permutation value";
      return false;
    }
    // Input-XOR permutation (parity: sboxgates.c:1021-1031).
    for (int i = 0; i < len; i++) {
      sbox[i] = table[i ^ static_cast<u8>(permute)];
    }
  }
  return true;
}

bool load_sbox_file(const std::string& path, int permute, u8 sbox[256],
                    u32* num_inputs, std::string* err) {
  FILE* fp = std::fopen(path.c_str(), "r");
  if (fp == nullptr) {
    if (err != nullptr) *err = "error opening target S-box file: " + path;
    return false;
  }
  u8 table[256];
  int len = 0;
  unsigned value;
  int ret;
  // Strict parse: an out-of-range entry, a non-hex token, or content beyond
  // 256 entries is a load error. (A silent stop here could truncate a corrupt
  // file to a smaller power-of-two length that then loads "successfully".)
  while ((ret = std::fscanf(fp, " %x", &value)) > 0) {
    if (value >= 256) {
      std::fclose(fp);
      if (err != nullptr) *err = "S-box entry out of range (>= 0x100): " + path;
      return false;
    }
    if (len >= 256) {
      std::fclose(fp);
      if (err != nullptr) *err = "more than 256 entries in S-box file: " + path;
      return false;
    }
    table[len++] = static_cast<u8>(value);
  }
  if (ret == 0) {  // non-hex garbage before EOF
    std::fclose(fp);
    if (err != nullptr) *err = "unparseable token in S-box file: " + path;
    return false;
  }
  std::fclose(fp);
  return load_sbox_table(table, len, permute, sbox, num_inputs, err);
}

}  // namespace sbg
