// Example C++ agent (used by tests/test_cpp_sdk.py as the integration
// binary).  Usage: example_agent <node_id> <agentfield_url> [port]
#include <cstdio>
#include <cstdlib>

#include "agentfield.hpp"

using agentfield::Agent;
using agentfield::AgentConfig;
using agentfield::Json;

int main(int argc, char** argv) {
  AgentConfig cfg;
  cfg.node_id = argc > 1 ? argv[1] : "cppagent";
  if (argc > 2) cfg.agentfield_url = argv[2];
  if (argc > 3) cfg.port = atoi(argv[3]);
  cfg.heartbeat_interval_s = 5;

  Agent app(cfg);

  app.register_reasoner("shout", [](const Json& input) {
    std::string text = input.get_str("text");
    for (auto& c : text) c = toupper(c);
    Json out = Json::object();
    out["shouted"] = text;
    return out;
  });

  app.register_skill("mul", [](const Json& input) {
    Json out = Json::object();
    out["product"] = input.get_num("a") * input.get_num("b");
    return out;
  });

  app.register_reasoner("relay", [&app](const Json& input) {
    // nested cross-agent call through the control plane
    Json args = Json::object();
    args["a"] = input.get_num("x");
    args["b"] = 10.0;
    Json inner = app.call("cppagent.mul", args);
    Json out = Json::object();
    out["relayed"] = inner;
    return out;
  });

  app.on_action("ping", [](const Json& payload) {
    printf("ACTION ping %s\n", payload.dump().c_str());
    fflush(stdout);
  });

  printf("PORT=%d\n", app.port());
  fflush(stdout);
  app.run(true);
  return 0;
}
