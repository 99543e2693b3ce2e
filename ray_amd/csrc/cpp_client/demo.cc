// C++ client demo / integration-test binary.
// Usage: demo <port>   (a ClientServer must listen on 127.0.0.1:<port>
// with a named actor "counter" and a registered task "add")
#include <cstdio>
#include <cstdlib>

#include "ray_client.hpp"

int main(int argc, char** argv) {
  if (argc < 2) {
    std::fprintf(stderr, "usage: demo <port>\n");
    return 2;
  }
  try {
    ray::Client c("127.0.0.1", std::atoi(argv[1]));

    // KV round-trip
    c.kv_put("cpp_key", "cpp_value");
    std::string v = c.kv_get("cpp_key");
    std::printf("kv: %s\n", v.c_str());

    // cluster info
    ray::Value nodes = c.node_table();
    std::printf("nodes: %zu\n", nodes.arr.size());

    // registered task
    ray::Value sum = c.task_call("add", {ray::Value(2), ray::Value(40)});
    std::printf("task add: %lld\n", (long long)sum.as_int());

    // named-actor calls
    ray::Value r1 = c.actor_call("counter", "incr", {ray::Value(5)});
    ray::Value r2 = c.actor_call("counter", "incr", {ray::Value(7)});
    std::printf("actor: %lld %lld\n", (long long)r1.as_int(),
                (long long)r2.as_int());

    // pub/sub publish (subscriber count returned)
    long long n = (long long)c.publish("cpp_events",
                                       ray::Value("hello-from-cpp"));
    std::printf("published: %lld\n", n);

    // error surfacing
    try {
      c.actor_call("no_such_actor", "x", {});
      std::printf("error: MISSED\n");
    } catch (const std::exception& e) {
      std::printf("error: caught\n");
    }
    return 0;
  } catch (const std::exception& e) {
    std::fprintf(stderr, "FAILED: %s\n", e.what());
    return 1;
  }
}
