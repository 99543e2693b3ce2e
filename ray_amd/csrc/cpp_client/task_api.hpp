// ray_amd C++ TASK BODIES (reference: cpp/include/ray/api.h — user
// C++ functions invoked as Ray tasks). The MI355X-native mechanism:
// user functions register by name into a shared library; workers
// dlopen the library and dispatch through one extern-C ABI, so a C++
// function runs INSIDE a normal ray_amd worker process:
//
//   // libmytasks.cc
//   #include "task_api.hpp"
//   static std::string add(const std::string& in) {     // msgpack'd args
//     auto [a, b] = ray::unpack_pair_i64(in);
//     return ray::pack_i64(a + b);
//   }
//   RAY_AMD_CPP_FUNC(add);
//
//   # python driver
//   f = ray_amd.cpp.remote_function("libmytasks.so", "add")
//   ray.get(f.remote(ray_amd.cpp.pack_pair_i64(2, 3)))  # -> 5 (packed)
//
// Payloads are raw bytes end to end; pack_* helpers cover the common
// scalar/vector cases without a serializer dependency.
#pragma once

#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <map>
#include <stdexcept>
#include <string>
#include <vector>

namespace ray {

using CppTaskFn = std::string (*)(const std::string&);

inline std::map<std::string, CppTaskFn>& cpp_task_registry() {
  static std::map<std::string, CppTaskFn> reg;
  return reg;
}

struct CppTaskRegistrar {
  CppTaskRegistrar(const char* name, CppTaskFn fn) {
    cpp_task_registry()[name] = fn;
  }
};

#define RAY_AMD_CPP_FUNC(fn) \
  static ::ray::CppTaskRegistrar _ray_amd_reg_##fn(#fn, fn)

// ---- tiny byte packers (LE) ----

inline std::string pack_i64(int64_t v) {
  std::string s(8, '\0');
  std::memcpy(&s[0], &v, 8);
  return s;
}

inline int64_t unpack_i64(const std::string& s) {
  int64_t v = 0;
  std::memcpy(&v, s.data(), 8);
  return v;
}

inline std::pair<int64_t, int64_t> unpack_pair_i64(const std::string& s) {
  int64_t a = 0, b = 0;
  std::memcpy(&a, s.data(), 8);
  std::memcpy(&b, s.data() + 8, 8);
  return {a, b};
}

inline std::string pack_f64_vec(const std::vector<double>& v) {
  std::string s(8 * v.size(), '\0');
  std::memcpy(&s[0], v.data(), s.size());
  return s;
}

inline std::vector<double> unpack_f64_vec(const std::string& s) {
  std::vector<double> v(s.size() / 8);
  std::memcpy(v.data(), s.data(), s.size());
  return v;
}

}  // namespace ray

// Worker-facing C ABI: list + invoke. Output buffer is heap-allocated
// by the library and freed by ray_amd_cpp_free.
// weak + default visibility: emitted even without in-library uses,
// and multiple TUs including this header merge cleanly
#define RAY_AMD_CPP_ABI __attribute__((weak, visibility("default")))

extern "C" {

RAY_AMD_CPP_ABI const char* ray_amd_cpp_list() {
  static std::string names;
  names.clear();
  for (auto& kv : ray::cpp_task_registry()) {
    if (!names.empty()) names += ",";
    names += kv.first;
  }
  return names.c_str();
}

RAY_AMD_CPP_ABI int ray_amd_cpp_invoke(const char* name, const char* in, long in_len,
                              char** out, long* out_len) {
  auto& reg = ray::cpp_task_registry();
  auto it = reg.find(name);
  if (it == reg.end()) return 1;
  try {
    std::string r = it->second(std::string(in, (size_t)in_len));
    *out = (char*)std::malloc(r.size());
    std::memcpy(*out, r.data(), r.size());
    *out_len = (long)r.size();
    return 0;
  } catch (const std::exception&) {
    return 2;
  }
}

RAY_AMD_CPP_ABI void ray_amd_cpp_free(char* p) { std::free(p); }
}
