// Minimal SafeTensors read/write (F32 + BF16) for the parameter-server
// aggregate executor. The checkpoint format the reference reads/writes
// (training.py:61-63, parameter_server.rs:331-446): 8-byte little-endian
// header length, JSON header {tensor: {dtype, shape, data_offsets}}, raw data.
#pragma once

#include <cstdint>
#include <cstdio>
#include <map>
#include <stdexcept>
#include <string>
#include <vector>

#include "json.h"

namespace hypha {

struct TensorMeta {
  std::string dtype;  // "F32" | "BF16"
  std::vector<int64_t> shape;
  size_t begin = 0, end = 0;  // byte offsets into the data section
  int64_t numel() const {
    int64_t n = 1;
    for (auto d : shape) n *= d;
    return n;
  }
};

inline size_t dtype_size(const std::string& dtype) {
  if (dtype == "F32" || dtype == "I32" || dtype == "U32") return 4;
  if (dtype == "BF16" || dtype == "F16" || dtype == "I16" || dtype == "U16") return 2;
  if (dtype == "F64" || dtype == "I64" || dtype == "U64") return 8;
  if (dtype == "F8_E4M3" || dtype == "F8_E5M2" || dtype == "I8" || dtype == "U8" ||
      dtype == "BOOL")
    return 1;
  throw std::runtime_error("safetensors: unsupported dtype " + dtype);
}

class SafeTensors {
 public:
  std::map<std::string, TensorMeta> tensors;  // ordered by name
  std::vector<char> data;

  static SafeTensors load(const std::string& path) {
    FILE* f = fopen(path.c_str(), "rb");
    if (!f) throw std::runtime_error("safetensors: cannot open " + path);
    uint64_t hlen = 0;
    if (fread(&hlen, 8, 1, f) != 1) {
      fclose(f);
      throw std::runtime_error("safetensors: short header");
    }
    if (hlen > (256u << 20)) {  // corrupt/hostile file: bound the allocation
      fclose(f);
      throw std::runtime_error("safetensors: absurd header length");
    }
    std::string hdr(hlen, '\0');
    if (fread(hdr.data(), 1, hlen, f) != hlen) {
      fclose(f);
      throw std::runtime_error("safetensors: short header json");
    }
    SafeTensors st;
    Json j = Json::parse(hdr);
    size_t data_size = 0;
    for (auto& [name, meta] : j.as_object()) {
      if (name == "__metadata__") continue;
      TensorMeta tm;
      tm.dtype = meta.at("dtype").as_string();
      for (auto& d : meta.at("shape").as_array()) {
        int64_t dim = d.as_int();
        if (dim < 0) throw std::runtime_error("safetensors: negative dim in " + name);
        tm.shape.push_back(dim);
      }
      int64_t b = meta.at("data_offsets").as_array()[0].as_int();
      int64_t e = meta.at("data_offsets").as_array()[1].as_int();
      if (b < 0 || e < b)
        throw std::runtime_error("safetensors: bad data_offsets in " + name);
      tm.begin = (size_t)b;
      tm.end = (size_t)e;
      // The byte span must match shape*itemsize exactly: the PS math in
      // ps_math.h walks numel() elements via raw pointers at tm.begin, so a
      // malformed file pushed by a peer must be rejected here, not there.
      size_t isz = dtype_size(tm.dtype);
      uint64_t want = (uint64_t)tm.numel() * isz;
      if ((uint64_t)(tm.end - tm.begin) != want)
        throw std::runtime_error("safetensors: data_offsets span " +
                                 std::to_string(tm.end - tm.begin) +
                                 " != numel*itemsize " + std::to_string(want) +
                                 " for " + name);
      if (tm.end > data_size) data_size = tm.end;
      st.tensors[name] = tm;
    }
    // Bound total data by what the file can actually hold.
    long pos = ftell(f);
    if (pos >= 0 && fseek(f, 0, SEEK_END) == 0) {
      long fend = ftell(f);
      fseek(f, pos, SEEK_SET);
      if (fend >= 0 && (uint64_t)data_size > (uint64_t)(fend - pos)) {
        fclose(f);
        throw std::runtime_error("safetensors: offsets exceed data section size");
      }
    }
    st.data.resize(data_size);
    if (data_size && fread(st.data.data(), 1, data_size, f) != data_size) {
      fclose(f);
      throw std::runtime_error("safetensors: short data");
    }
    fclose(f);
    return st;
  }

  void save(const std::string& path) const {
    JsonObject hdr;
    for (auto& [name, tm] : tensors) {
      JsonObject m;
      m["dtype"] = tm.dtype;
      JsonArray shape;
      for (auto d : tm.shape) shape.push_back(Json(d));
      m["shape"] = shape;
      m["data_offsets"] = JsonArray{Json((int64_t)tm.begin), Json((int64_t)tm.end)};
      hdr[name] = Json(m);
    }
    std::string h = Json(hdr).dump();
    // pad header to 8-byte alignment with spaces (standard practice)
    while (h.size() % 8 != 0) h += ' ';
    FILE* f = fopen(path.c_str(), "wb");
    if (!f) throw std::runtime_error("safetensors: cannot write " + path);
    uint64_t hlen = h.size();
    fwrite(&hlen, 8, 1, f);
    fwrite(h.data(), 1, h.size(), f);
    if (!data.empty()) fwrite(data.data(), 1, data.size(), f);
    fclose(f);
  }

  // elementwise view helpers -----------------------------------------------
  static float bf16_to_f32(uint16_t u) {
    union {
      float f;
      uint32_t i;
    } c;
    c.i = ((uint32_t)u) << 16;
    return c.f;
  }
  static uint16_t f32_to_bf16(float f) {
    union {
      float f;
      uint32_t i;
    } c;
    c.f = f;
    uint32_t x = c.i;
    if ((x & 0x7fffffffu) > 0x7f800000u) return 0x7fc0;
    uint32_t round = 0x7fffu + ((x >> 16) & 1u);
    return (uint16_t)((x + round) >> 16);
  }

  float get_elem(const TensorMeta& tm, int64_t i) const {
    if (tm.dtype == "F32") return *(const float*)(data.data() + tm.begin + i * 4);
    if (tm.dtype == "BF16")
      return bf16_to_f32(*(const uint16_t*)(data.data() + tm.begin + i * 2));
    throw std::runtime_error("safetensors: unsupported dtype " + tm.dtype);
  }
  void set_elem(const TensorMeta& tm, int64_t i, float v) {
    if (tm.dtype == "F32") {
      *(float*)(data.data() + tm.begin + i * 4) = v;
    } else if (tm.dtype == "BF16") {
      *(uint16_t*)(data.data() + tm.begin + i * 2) = f32_to_bf16(v);
    } else {
      throw std::runtime_error("safetensors: unsupported dtype " + tm.dtype);
    }
  }
};

}  // namespace hypha
