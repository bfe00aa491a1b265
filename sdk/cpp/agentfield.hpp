// agentfield-amd C++ SDK (reference parity: the Go SDK niche, SURVEY.md §2.3
// G1-G4 — a compiled-language agent SDK; Go is not in this toolchain, so the
// native SDK is C++).
//
// Capabilities mirrored from the reference Go SDK:
//   * Agent::New-style construction + RegisterReasoner + Run        (G1)
//   * control-plane REST client: register/heartbeat/status callback (G2)
//   * the 202-async + status-callback execution pattern             (G1)
//   * Call() cross-agent invocation through the control plane       (G1)
//   * ai() helper hitting the engine-server fleet (/v1/generate)    (G3)
//
// Zero external dependencies: built-in JSON codec + HTTP/1.1 over POSIX
// sockets (control-plane traffic is host-side localhost plumbing).
#pragma once

#include <atomic>
#include <functional>
#include <map>
#include <memory>
#include <string>
#include <thread>
#include <vector>

namespace agentfield {

// ----------------------------------------------------------- JSON value
class Json {
 public:
  enum Type { Null, Bool, Num, Str, Arr, Obj };
  Type type = Null;
  bool b = false;
  double num = 0;
  std::string str;
  std::vector<Json> arr;
  std::vector<std::pair<std::string, Json>> obj;  // insertion-ordered

  Json() = default;
  Json(bool v) : type(Bool), b(v) {}
  Json(int v) : type(Num), num(v) {}
  Json(double v) : type(Num), num(v) {}
  Json(const char* s) : type(Str), str(s) {}
  Json(const std::string& s) : type(Str), str(s) {}

  static Json object() { Json j; j.type = Obj; return j; }
  static Json array() { Json j; j.type = Arr; return j; }

  Json& operator[](const std::string& key);
  const Json* find(const std::string& key) const;
  std::string get_str(const std::string& key, const std::string& dflt = "") const;
  double get_num(const std::string& key, double dflt = 0) const;

  std::string dump() const;
  static Json parse(const std::string& text, bool* ok = nullptr);
};

// ----------------------------------------------------------- HTTP bits
struct HttpResponse {
  int status = 0;
  std::string body;
  std::map<std::string, std::string> headers;
};

// Minimal HTTP/1.1 client for http://host:port/... URLs.
HttpResponse http_request(const std::string& method, const std::string& url,
                          const std::string& body = "",
                          const std::map<std::string, std::string>& headers = {},
                          int timeout_ms = 90000);

struct HttpRequest {
  std::string method, path, body;
  std::map<std::string, std::string> headers;  // lower-cased keys
};

class HttpServer {
 public:
  using Handler = std::function<HttpResponse(const HttpRequest&)>;
  explicit HttpServer(int port);  // port 0 -> ephemeral
  ~HttpServer();
  void route(const std::string& method, const std::string& prefix, Handler h);
  void start();
  void stop();
  int port() const { return port_; }

 private:
  void serve_loop();
  int fd_ = -1;
  int port_;
  std::atomic<bool> running_{false};
  std::thread thread_;
  std::vector<std::tuple<std::string, std::string, Handler>> routes_;
};

// ----------------------------------------------------------- Agent (G1)
using ReasonerFn = std::function<Json(const Json& input)>;

struct AgentConfig {
  std::string node_id;
  std::string agentfield_url = "http://127.0.0.1:8520";
  std::string engine_url;           // for ai(); empty -> AGENTFIELD_ENGINE_URLS
  int port = 0;                     // 0 -> ephemeral
  int heartbeat_interval_s = 30;
};

class Agent {
 public:
  explicit Agent(AgentConfig cfg);
  ~Agent();

  void register_reasoner(const std::string& name, ReasonerFn fn);
  void register_skill(const std::string& name, ReasonerFn fn);

  // Register with the control plane and serve until stop() (or
  // non-blocking with run(false)).
  bool run(bool block = true);
  void stop();
  int port() const;

  // Cross-agent call through the control plane (sync /execute).
  Json call(const std::string& target, const Json& input);

  // Model call against the engine fleet (/v1/generate).
  std::string ai(const std::string& prompt, int max_tokens = 128,
                 double temperature = 0.0);

  // OpenAI-style chat completion against the engine fleet
  // (/v1/chat/completions); messages are (role, content) pairs.
  // Returns the assistant message content ("" on error).
  std::string chat(
      const std::vector<std::pair<std::string, std::string>>& messages,
      int max_tokens = 128, double temperature = 0.0);

  // Handle a named control-plane action delivered through the claim/ack
  // lease queue (reference G2 AcknowledgeAction; drained each heartbeat).
  void on_action(const std::string& name, std::function<void(const Json&)> fn);

  bool registered() const { return registered_; }

 private:
  bool do_register();
  void heartbeat_loop();
  void drain_actions();
  HttpResponse handle_invoke(const std::string& kind, const std::string& name,
                             const HttpRequest& req);

  AgentConfig cfg_;
  std::map<std::string, ReasonerFn> reasoners_;
  std::map<std::string, ReasonerFn> skills_;
  std::map<std::string, std::function<void(const Json&)>> action_handlers_;
  std::unique_ptr<HttpServer> server_;
  std::thread hb_thread_;
  std::atomic<bool> stopping_{false};
  std::atomic<bool> registered_{false};
};

}  // namespace agentfield
