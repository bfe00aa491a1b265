#include "agentfield.hpp"

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <chrono>
#include <cstring>
#include <sstream>

namespace agentfield {

// ================================================================ JSON
Json& Json::operator[](const std::string& key) {
  type = Obj;
  for (auto& kv : obj)
    if (kv.first == key) return kv.second;
  obj.emplace_back(key, Json());
  return obj.back().second;
}

const Json* Json::find(const std::string& key) const {
  if (type != Obj) return nullptr;
  for (auto& kv : obj)
    if (kv.first == key) return &kv.second;
  return nullptr;
}

std::string Json::get_str(const std::string& key, const std::string& dflt) const {
  const Json* v = find(key);
  return (v && v->type == Str) ? v->str : dflt;
}

double Json::get_num(const std::string& key, double dflt) const {
  const Json* v = find(key);
  return (v && v->type == Num) ? v->num : dflt;
}

static void dump_str(std::string& out, const std::string& s) {
  out += '"';
  for (char c : s) {
    switch (c) {
      case '"': out += "\\\""; break;
      case '\\': out += "\\\\"; break;
      case '\n': out += "\\n"; break;
      case '\r': out += "\\r"; break;
      case '\t': out += "\\t"; break;
      default:
        if ((unsigned char)c < 0x20) {
          char buf[8];
          snprintf(buf, sizeof buf, "\\u%04x", c);
          out += buf;
        } else {
          out += c;
        }
    }
  }
  out += '"';
}

std::string Json::dump() const {
  std::string out;
  switch (type) {
    case Null: out = "null"; break;
    case Bool: out = b ? "true" : "false"; break;
    case Num: {
      char buf[32];
      if (num == (long long)num)
        snprintf(buf, sizeof buf, "%lld", (long long)num);
      else
        snprintf(buf, sizeof buf, "%.17g", num);
      out = buf;
      break;
    }
    case Str: dump_str(out, str); break;
    case Arr: {
      out = "[";
      for (size_t i = 0; i < arr.size(); ++i) {
        if (i) out += ",";
        out += arr[i].dump();
      }
      out += "]";
      break;
    }
    case Obj: {
      out = "{";
      for (size_t i = 0; i < obj.size(); ++i) {
        if (i) out += ",";
        dump_str(out, obj[i].first);
        out += ":";
        out += obj[i].second.dump();
      }
      out += "}";
      break;
    }
  }
  return out;
}

namespace {
struct Parser {
  const char* p;
  const char* end;
  bool ok = true;

  void skip() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r'))
      ++p;
  }

  bool lit(const char* s) {
    size_t n = strlen(s);
    if ((size_t)(end - p) >= n && !strncmp(p, s, n)) { p += n; return true; }
    return false;
  }

  std::string parse_string() {
    std::string out;
    ++p;  // opening quote
    while (p < end && *p != '"') {
      if (*p == '\\' && p + 1 < end) {
        ++p;
        switch (*p) {
          case 'n': out += '\n'; break;
          case 't': out += '\t'; break;
          case 'r': out += '\r'; break;
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'u': {
            if (end - p >= 5) {
              unsigned code = 0;
              sscanf(p + 1, "%4x", &code);
              p += 4;
              if (code < 0x80) {
                out += (char)code;
              } else if (code < 0x800) {
                out += (char)(0xC0 | (code >> 6));
                out += (char)(0x80 | (code & 0x3F));
              } else {
                out += (char)(0xE0 | (code >> 12));
                out += (char)(0x80 | ((code >> 6) & 0x3F));
                out += (char)(0x80 | (code & 0x3F));
              }
            }
            break;
          }
          default: out += *p;
        }
        ++p;
      } else {
        out += *p++;
      }
    }
    if (p < end) ++p;  // closing quote
    else ok = false;
    return out;
  }

  Json value() {
    skip();
    if (p >= end) { ok = false; return Json(); }
    if (*p == '{') {
      Json j = Json::object();
      ++p;
      skip();
      if (p < end && *p == '}') { ++p; return j; }
      while (p < end) {
        skip();
        if (p >= end || *p != '"') { ok = false; break; }
        std::string key = parse_string();
        skip();
        if (p >= end || *p != ':') { ok = false; break; }
        ++p;
        j.obj.emplace_back(key, value());
        skip();
        if (p < end && *p == ',') { ++p; continue; }
        if (p < end && *p == '}') { ++p; break; }
        ok = false;
        break;
      }
      return j;
    }
    if (*p == '[') {
      Json j = Json::array();
      ++p;
      skip();
      if (p < end && *p == ']') { ++p; return j; }
      while (p < end) {
        j.arr.push_back(value());
        skip();
        if (p < end && *p == ',') { ++p; continue; }
        if (p < end && *p == ']') { ++p; break; }
        ok = false;
        break;
      }
      return j;
    }
    if (*p == '"') { Json j; j.type = Json::Str; j.str = parse_string(); return j; }
    if (lit("true")) return Json(true);
    if (lit("false")) return Json(false);
    if (lit("null")) return Json();
    char* endp = nullptr;
    double d = strtod(p, &endp);
    if (endp == p) { ok = false; return Json(); }
    p = endp;
    return Json(d);
  }
};
}  // namespace

Json Json::parse(const std::string& text, bool* ok) {
  Parser ps{text.data(), text.data() + text.size()};
  Json j = ps.value();
  ps.skip();
  bool good = ps.ok && ps.p == ps.end;
  if (ok) *ok = good;
  return j;
}

// ================================================================ HTTP
namespace {
bool parse_url(const std::string& url, std::string& host, int& port,
               std::string& path) {
  if (url.rfind("http://", 0) != 0) return false;
  size_t hstart = 7;
  size_t pstart = url.find('/', hstart);
  std::string hostport =
      url.substr(hstart, (pstart == std::string::npos ? url.size() : pstart) - hstart);
  path = pstart == std::string::npos ? "/" : url.substr(pstart);
  size_t colon = hostport.find(':');
  if (colon == std::string::npos) {
    host = hostport;
    port = 80;
  } else {
    host = hostport.substr(0, colon);
    port = atoi(hostport.c_str() + colon + 1);
  }
  return true;
}

int connect_to(const std::string& host, int port, int timeout_ms) {
  struct addrinfo hints{}, *res = nullptr;
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  char portstr[16];
  snprintf(portstr, sizeof portstr, "%d", port);
  if (getaddrinfo(host.c_str(), portstr, &hints, &res) != 0) return -1;
  int fd = socket(res->ai_family, res->ai_socktype, 0);
  if (fd >= 0) {
    struct timeval tv{timeout_ms / 1000, (timeout_ms % 1000) * 1000};
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
    setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof tv);
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
    if (connect(fd, res->ai_addr, res->ai_addrlen) != 0) {
      close(fd);
      fd = -1;
    }
  }
  freeaddrinfo(res);
  return fd;
}

bool read_http_message(int fd, std::string& head, std::string& body) {
  std::string buf;
  char tmp[4096];
  size_t hdr_end = std::string::npos;
  while (hdr_end == std::string::npos) {
    ssize_t n = recv(fd, tmp, sizeof tmp, 0);
    if (n <= 0) return false;
    buf.append(tmp, n);
    hdr_end = buf.find("\r\n\r\n");
    if (buf.size() > (1u << 22)) return false;
  }
  head = buf.substr(0, hdr_end);
  body = buf.substr(hdr_end + 4);
  // content-length (case-insensitive scan)
  size_t cl = 0;
  {
    std::string lower = head;
    for (auto& c : lower) c = tolower(c);
    size_t pos = lower.find("content-length:");
    if (pos != std::string::npos) cl = strtoul(lower.c_str() + pos + 15, nullptr, 10);
  }
  while (body.size() < cl) {
    ssize_t n = recv(fd, tmp, sizeof tmp, 0);
    if (n <= 0) return false;
    body.append(tmp, n);
  }
  body.resize(std::max(body.size(), cl));
  return true;
}
}  // namespace

HttpResponse http_request(const std::string& method, const std::string& url,
                          const std::string& body,
                          const std::map<std::string, std::string>& headers,
                          int timeout_ms) {
  HttpResponse resp;
  std::string host, path;
  int port = 80;
  if (!parse_url(url, host, port, path)) return resp;
  int fd = connect_to(host, port, timeout_ms);
  if (fd < 0) return resp;
  std::ostringstream req;
  req << method << " " << path << " HTTP/1.1\r\n"
      << "Host: " << host << ":" << port << "\r\n"
      << "Connection: close\r\n"
      << "Content-Type: application/json\r\n"
      << "Content-Length: " << body.size() << "\r\n";
  for (auto& kv : headers) req << kv.first << ": " << kv.second << "\r\n";
  req << "\r\n" << body;
  std::string data = req.str();
  size_t off = 0;
  while (off < data.size()) {
    ssize_t n = send(fd, data.data() + off, data.size() - off, 0);
    if (n <= 0) { close(fd); return resp; }
    off += n;
  }
  std::string head;
  if (read_http_message(fd, head, resp.body)) {
    sscanf(head.c_str(), "HTTP/%*s %d", &resp.status);
    // parse headers
    size_t line_start = head.find("\r\n");
    while (line_start != std::string::npos) {
      size_t line_end = head.find("\r\n", line_start + 2);
      std::string line = head.substr(
          line_start + 2, (line_end == std::string::npos ? head.size() : line_end) -
                              line_start - 2);
      size_t colon = line.find(':');
      if (colon != std::string::npos) {
        std::string k = line.substr(0, colon);
        for (auto& c : k) c = tolower(c);
        size_t vs = line.find_first_not_of(' ', colon + 1);
        resp.headers[k] = vs == std::string::npos ? "" : line.substr(vs);
      }
      line_start = line_end;
    }
  }
  close(fd);
  return resp;
}

HttpServer::HttpServer(int port) : port_(port) {
  fd_ = socket(AF_INET, SOCK_STREAM, 0);
  int one = 1;
  setsockopt(fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof one);
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_addr.s_addr = htonl(INADDR_LOOPBACK);
  addr.sin_port = htons(port);
  if (bind(fd_, (sockaddr*)&addr, sizeof addr) != 0 || listen(fd_, 64) != 0) {
    close(fd_);
    fd_ = -1;
    return;
  }
  socklen_t len = sizeof addr;
  getsockname(fd_, (sockaddr*)&addr, &len);
  port_ = ntohs(addr.sin_port);
}

HttpServer::~HttpServer() { stop(); }

void HttpServer::route(const std::string& method, const std::string& prefix,
                       Handler h) {
  routes_.emplace_back(method, prefix, std::move(h));
}

void HttpServer::start() {
  running_ = true;
  thread_ = std::thread([this] { serve_loop(); });
}

void HttpServer::stop() {
  if (!running_.exchange(false)) return;
  if (fd_ >= 0) {
    shutdown(fd_, SHUT_RDWR);
    close(fd_);
    fd_ = -1;
  }
  if (thread_.joinable()) thread_.join();
}

void HttpServer::serve_loop() {
  while (running_) {
    int cfd = accept(fd_, nullptr, nullptr);
    if (cfd < 0) {
      if (!running_) break;
      continue;
    }
    std::thread([this, cfd] {
      std::string head, body;
      if (read_http_message(cfd, head, body)) {
        HttpRequest req;
        req.body = body;
        {
          std::istringstream first(head.substr(0, head.find("\r\n")));
          std::string version;
          first >> req.method >> req.path >> version;
          size_t line_start = head.find("\r\n");
          while (line_start != std::string::npos) {
            size_t line_end = head.find("\r\n", line_start + 2);
            std::string line = head.substr(
                line_start + 2,
                (line_end == std::string::npos ? head.size() : line_end) -
                    line_start - 2);
            size_t colon = line.find(':');
            if (colon != std::string::npos) {
              std::string k = line.substr(0, colon);
              for (auto& c : k) c = tolower(c);
              size_t vs = line.find_first_not_of(' ', colon + 1);
              req.headers[k] =
                  vs == std::string::npos ? "" : line.substr(vs);
            }
            line_start = line_end;
          }
        }
        HttpResponse resp;
        resp.status = 404;
        resp.body = "{\"error\":\"not found\"}";
        for (auto& [m, prefix, h] : routes_) {
          if (req.method == m && req.path.rfind(prefix, 0) == 0) {
            resp = h(req);
            break;
          }
        }
        std::ostringstream out;
        out << "HTTP/1.1 " << resp.status << " X\r\n"
            << "Content-Type: application/json\r\n"
            << "Content-Length: " << resp.body.size() << "\r\n"
            << "Connection: close\r\n\r\n"
            << resp.body;
        std::string data = out.str();
        send(cfd, data.data(), data.size(), 0);
      }
      close(cfd);
    }).detach();
  }
}

// ================================================================ Agent
Agent::Agent(AgentConfig cfg) : cfg_(std::move(cfg)) {
  server_ = std::make_unique<HttpServer>(cfg_.port);
  server_->route("GET", "/health", [this](const HttpRequest&) {
    HttpResponse r;
    r.status = 200;
    Json j = Json::object();
    j["status"] = "healthy";
    j["node_id"] = cfg_.node_id;
    r.body = j.dump();
    return r;
  });
  server_->route("POST", "/reasoners/", [this](const HttpRequest& req) {
    std::string name = req.path.substr(strlen("/reasoners/"));
    return handle_invoke("reasoner", name, req);
  });
  server_->route("POST", "/skills/", [this](const HttpRequest& req) {
    std::string name = req.path.substr(strlen("/skills/"));
    return handle_invoke("skill", name, req);
  });
}

Agent::~Agent() { stop(); }

void Agent::register_reasoner(const std::string& name, ReasonerFn fn) {
  reasoners_[name] = std::move(fn);
}

void Agent::register_skill(const std::string& name, ReasonerFn fn) {
  skills_[name] = std::move(fn);
}

int Agent::port() const { return server_->port(); }

HttpResponse Agent::handle_invoke(const std::string& kind,
                                  const std::string& name,
                                  const HttpRequest& req) {
  HttpResponse r;
  auto& table = kind == "skill" ? skills_ : reasoners_;
  auto it = table.find(name);
  if (it == table.end()) {
    r.status = 404;
    r.body = "{\"error\":\"no such " + kind + "\"}";
    return r;
  }
  Json input = Json::parse(req.body);
  if (const Json* inner = input.find("input")) input = *inner;
  auto exec_it = req.headers.find("x-execution-id");
  if (exec_it != req.headers.end()) {
    // control-plane invocation: 202 + background + status callback (G1)
    std::string exec_id = exec_it->second;
    ReasonerFn fn = it->second;
    std::string cp = cfg_.agentfield_url;
    std::thread([fn, input, exec_id, cp] {
      auto t0 = std::chrono::steady_clock::now();
      Json cb = Json::object();
      cb["execution_id"] = exec_id;
      try {
        Json result = fn(input);
        cb["status"] = "completed";
        cb["result"] = result;
      } catch (const std::exception& e) {
        cb["status"] = "failed";
        cb["error"] = std::string(e.what());
      }
      cb["duration_ms"] =
          std::chrono::duration<double, std::milli>(
              std::chrono::steady_clock::now() - t0).count();
      http_request("POST", cp + "/api/v1/executions/" + exec_id + "/status",
                   cb.dump());
    }).detach();
    r.status = 202;
    Json j = Json::object();
    j["status"] = "accepted";
    j["execution_id"] = exec_id;
    r.body = j.dump();
    return r;
  }
  try {
    Json result = it->second(input);
    Json j = Json::object();
    j["result"] = result;
    r.status = 200;
    r.body = j.dump();
  } catch (const std::exception& e) {
    r.status = 500;
    Json j = Json::object();
    j["error"] = std::string(e.what());
    r.body = j.dump();
  }
  return r;
}

bool Agent::do_register() {
  Json node = Json::object();
  node["id"] = cfg_.node_id;
  node["base_url"] = "http://127.0.0.1:" + std::to_string(server_->port());
  node["version"] = "0.1.0";
  node["deployment_type"] = "long_running";
  Json rs = Json::array();
  for (auto& kv : reasoners_) {
    Json r = Json::object();
    r["id"] = kv.first;
    rs.arr.push_back(r);
  }
  node["reasoners"] = rs;
  Json sk = Json::array();
  for (auto& kv : skills_) {
    Json s = Json::object();
    s["id"] = kv.first;
    sk.arr.push_back(s);
  }
  node["skills"] = sk;
  Json meta = Json::object();
  meta["sdk"] = "agentfield_amd_cpp";
  node["metadata"] = meta;
  auto resp = http_request("POST", cfg_.agentfield_url + "/api/v1/nodes/register",
                           node.dump());
  registered_ = resp.status == 200;
  return registered_;
}

void Agent::heartbeat_loop() {
  while (!stopping_) {
    for (int i = 0; i < cfg_.heartbeat_interval_s * 10 && !stopping_; ++i)
      std::this_thread::sleep_for(std::chrono::milliseconds(100));
    if (stopping_) break;
    Json hb = Json::object();
    hb["status"] = "active";
    auto resp = http_request(
        "POST", cfg_.agentfield_url + "/api/v1/nodes/" + cfg_.node_id +
                    "/heartbeat", hb.dump());
    if (resp.status == 404) {
      do_register();  // resilient re-register
    } else if (resp.status == 200) {
      drain_actions();  // claim/ack lease queue (G2)
    }
  }
}

bool Agent::run(bool block) {
  server_->start();
  bool ok = do_register();
  hb_thread_ = std::thread([this] { heartbeat_loop(); });
  if (block) {
    while (!stopping_) std::this_thread::sleep_for(std::chrono::milliseconds(200));
  }
  return ok;
}

void Agent::stop() {
  if (stopping_.exchange(true)) return;
  if (hb_thread_.joinable()) hb_thread_.join();
  if (server_) server_->stop();
}

Json Agent::call(const std::string& target, const Json& input) {
  Json body = Json::object();
  body["input"] = input;
  auto resp = http_request("POST",
                           cfg_.agentfield_url + "/api/v1/execute/" + target,
                           body.dump());
  Json out = Json::parse(resp.body);
  if (out.get_str("status") != "completed")
    throw std::runtime_error("call failed: " + resp.body);
  const Json* result = out.find("result");
  if (result) {
    if (const Json* inner = result->find("result")) return *inner;
    return *result;
  }
  return Json();
}

std::string Agent::ai(const std::string& prompt, int max_tokens,
                      double temperature) {
  std::string url = cfg_.engine_url;
  if (url.empty()) {
    const char* env = getenv("AGENTFIELD_ENGINE_URLS");
    if (env) {
      url = env;
      size_t comma = url.find(',');
      if (comma != std::string::npos) url = url.substr(0, comma);
    }
  }
  if (url.empty()) throw std::runtime_error("no engine url configured");
  Json body = Json::object();
  body["prompt"] = prompt;
  body["max_tokens"] = max_tokens;
  body["temperature"] = temperature;
  auto resp = http_request("POST", url + "/v1/generate", body.dump(), {},
                           600000);
  if (resp.status != 200)
    throw std::runtime_error("engine error: " + resp.body);
  return Json::parse(resp.body).get_str("text");
}

std::string Agent::chat(
    const std::vector<std::pair<std::string, std::string>>& messages,
    int max_tokens, double temperature) {
  std::string url = cfg_.engine_url;
  if (url.empty()) {
    const char* env = getenv("AGENTFIELD_ENGINE_URLS");
    if (env) {
      url = env;
      size_t comma = url.find(',');
      if (comma != std::string::npos) url = url.substr(0, comma);
    }
  }
  if (url.empty()) throw std::runtime_error("no engine url configured");
  Json body = Json::object();
  Json msgs = Json::array();
  for (auto& m : messages) {
    Json one = Json::object();
    one["role"] = m.first;
    one["content"] = m.second;
    msgs.arr.push_back(one);
  }
  body["messages"] = msgs;
  body["max_tokens"] = max_tokens;
  body["temperature"] = temperature;
  auto resp = http_request("POST", url + "/v1/chat/completions", body.dump(),
                           {}, 600000);
  if (resp.status != 200)
    throw std::runtime_error("engine error: " + resp.body);
  Json doc = Json::parse(resp.body);
  const Json* choices = doc.find("choices");
  if (!choices || choices->arr.empty()) return "";
  const Json* msg = choices->arr[0].find("message");
  return msg ? msg->get_str("content") : "";
}

void Agent::on_action(const std::string& name,
                      std::function<void(const Json&)> fn) {
  action_handlers_[name] = fn;
}

void Agent::drain_actions() {
  Json req = Json::object();
  req["lease_s"] = 30.0;
  auto resp = http_request(
      "POST", cfg_.agentfield_url + "/api/v1/nodes/" + cfg_.node_id +
                  "/actions/claim", req.dump());
  if (resp.status != 200) return;
  Json doc = Json::parse(resp.body);  // keep alive: find() borrows
  const Json* acts = doc.find("actions");
  if (!acts) return;
  for (const Json& a : acts->arr) {
    std::string action = a.get_str("action");
    std::string status = "done";
    if (action == "stop" || action == "shutdown") {
      stopping_ = true;
    } else {
      auto it = action_handlers_.find(action);
      if (it != action_handlers_.end()) {
        const Json* payload = a.find("payload");
        Json empty = Json::object();
        it->second(payload ? *payload : empty);
      } else {
        status = "ignored";
      }
    }
    Json ack = Json::object();
    ack["action_id"] = a.get_num("id");
    ack["status"] = status;
    http_request("POST", cfg_.agentfield_url + "/api/v1/nodes/" +
                             cfg_.node_id + "/actions/ack", ack.dump());
  }
}

}  // namespace agentfield
