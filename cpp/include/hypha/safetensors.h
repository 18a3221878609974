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
      for (auto& d : meta.at("shape").as_array()) tm.shape.push_back(d.as_int());
      tm.begin = (size_t)meta.at("data_offsets").as_array()[0].as_int();
      tm.end = (size_t)meta.at("data_offsets").as_array()[1].as_int();
      if (tm.end > data_size) data_size = tm.end;
      st.tensors[name] = tm;
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
