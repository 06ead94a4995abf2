// Copyright 2026 mlrun_amd authors
//
// Licensed under the Apache License, Version 2.0 (the "License");
// you may not use this file except in compliance with the License.
//
// Node-local log-collector daemon (C++).
//
// The reference ships a Go gRPC service (server/log-collector/) that
// tails pod logs into files and serves/monitors them with 6 RPCs
// (proto/log_collector.proto:21-28: StartLog, GetLogs, GetLogSize,
// StopLogs, DeleteLogs, ListRunsInProgress) plus a file-backed state
// store.  This MI355X-native rebuild serves the same 6 operations for
// node-local run processes: it follows each run's source log file
// (the file the runtime points its rank/process stdout at) into a
// persistent per-run log under the log directory, tracks in-progress
// runs in a state file, and answers over a newline-delimited-JSON TCP
// protocol on 127.0.0.1 (no gRPC dependency in the image).
//
// Request:  {"op":"start_log","run_uid":"u","project":"p",
//            "source":"/path/to/live.log"}\n
//           {"op":"get_logs","run_uid":"u","project":"p",
//            "offset":0,"size":0}\n           (size 0 = to EOF)
//           {"op":"get_log_size",...} {"op":"stop_logs",...}
//           {"op":"delete_logs",...} {"op":"list_runs_in_progress"}
// Response: {"success":true,...}\n  (get_logs: header line then raw
//           bytes: {"success":true,"size":N}\n<N bytes>)
//
// Build: g++ -O2 -std=c++17 -pthread log_collector.cpp -o log_collector
// Run:   ./log_collector <port> <log_dir>

#include <arpa/inet.h>
#include <dirent.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <unistd.h>

#include <atomic>
#include <cstring>
#include <fstream>
#include <map>
#include <mutex>
#include <sstream>
#include <string>
#include <thread>
#include <vector>

namespace {

std::string g_log_dir;
std::mutex g_state_mutex;
// run key -> source file being followed ("" = stopped)
std::map<std::string, std::string> g_in_progress;
std::atomic<bool> g_shutdown{false};

std::string run_key(const std::string& project, const std::string& uid) {
  return project + "/" + uid;
}

std::string log_path(const std::string& project, const std::string& uid) {
  return g_log_dir + "/" + project + "_" + uid + ".log";
}

std::string state_path() { return g_log_dir + "/state.json"; }

// --- minimal flat-JSON helpers (string + integer values only) ---

std::string json_escape(const std::string& s) {
  std::string out;
  for (char c : s) {
    switch (c) {
      case '"': out += "\\\""; break;
      case '\\': out += "\\\\"; break;
      case '\n': out += "\\n"; break;
      case '\r': out += "\\r"; break;
      case '\t': out += "\\t"; break;
      default: out += c;
    }
  }
  return out;
}

std::string get_string(const std::string& body, const std::string& key) {
  const std::string pat = "\"" + key + "\"";
  size_t pos = body.find(pat);
  if (pos == std::string::npos) return "";
  pos = body.find(':', pos + pat.size());
  if (pos == std::string::npos) return "";
  ++pos;
  while (pos < body.size() && (body[pos] == ' ')) ++pos;
  if (pos >= body.size() || body[pos] != '"') return "";
  ++pos;
  std::string out;
  while (pos < body.size() && body[pos] != '"') {
    if (body[pos] == '\\' && pos + 1 < body.size()) {
      ++pos;
      switch (body[pos]) {
        case 'n': out += '\n'; break;
        case 't': out += '\t'; break;
        case 'r': out += '\r'; break;
        default: out += body[pos];
      }
    } else {
      out += body[pos];
    }
    ++pos;
  }
  return out;
}

long get_int(const std::string& body, const std::string& key, long dflt) {
  const std::string pat = "\"" + key + "\"";
  size_t pos = body.find(pat);
  if (pos == std::string::npos) return dflt;
  pos = body.find(':', pos + pat.size());
  if (pos == std::string::npos) return dflt;
  ++pos;
  while (pos < body.size() && body[pos] == ' ') ++pos;
  size_t end = pos;
  while (end < body.size() &&
         (isdigit(body[end]) || body[end] == '-')) ++end;
  if (end == pos) return dflt;
  return std::stol(body.substr(pos, end - pos));
}

// --- state store (file persistence, reference statestore/file) ---

void persist_state() {
  std::ostringstream out;
  out << "{";
  bool first = true;
  for (const auto& [key, source] : g_in_progress) {
    if (!first) out << ",";
    out << "\"" << json_escape(key) << "\":\"" << json_escape(source)
        << "\"";
    first = false;
  }
  out << "}";
  std::ofstream fp(state_path() + ".tmp");
  fp << out.str();
  fp.close();
  ::rename((state_path() + ".tmp").c_str(), state_path().c_str());
}

void load_state() {
  std::ifstream fp(state_path());
  if (!fp.good()) return;
  std::stringstream buf;
  buf << fp.rdbuf();
  const std::string body = buf.str();
  // parse {"k":"v",...}
  size_t pos = 0;
  while ((pos = body.find('"', pos)) != std::string::npos) {
    size_t kend = body.find('"', pos + 1);
    if (kend == std::string::npos) break;
    std::string key = body.substr(pos + 1, kend - pos - 1);
    size_t vstart = body.find('"', body.find(':', kend));
    if (vstart == std::string::npos) break;
    size_t vend = body.find('"', vstart + 1);
    if (vend == std::string::npos) break;
    std::string value = body.substr(vstart + 1, vend - vstart - 1);
    g_in_progress[key] = value;
    pos = vend + 1;
  }
}

// --- the follower thread: copy source file growth into the run log ---

void follow_file(const std::string& key, const std::string& source,
                 const std::string& dest) {
  long offset = 0;
  {  // resume from existing collected size
    struct stat st{};
    if (::stat(dest.c_str(), &st) == 0) offset = st.st_size;
  }
  while (!g_shutdown.load()) {
    {
      std::lock_guard<std::mutex> lock(g_state_mutex);
      auto it = g_in_progress.find(key);
      if (it == g_in_progress.end() || it->second != source) return;
    }
    struct stat st{};
    if (::stat(source.c_str(), &st) == 0 && st.st_size > offset) {
      std::ifstream in(source, std::ios::binary);
      in.seekg(offset);
      std::ofstream out(dest, std::ios::binary | std::ios::app);
      std::vector<char> buf(1 << 16);
      while (in.good() && offset < st.st_size) {
        in.read(buf.data(), buf.size());
        std::streamsize got = in.gcount();
        if (got <= 0) break;
        out.write(buf.data(), got);
        offset += got;
      }
    }
    ::usleep(100 * 1000);  // 100 ms poll, like the reference's interval
  }
}

// --- request handlers ---

std::string handle_start_log(const std::string& body) {
  const std::string uid = get_string(body, "run_uid");
  const std::string project = get_string(body, "project");
  const std::string source = get_string(body, "source");
  if (uid.empty() || source.empty())
    return "{\"success\":false,\"error\":\"run_uid and source required\"}";
  const std::string key = run_key(project, uid);
  {
    std::lock_guard<std::mutex> lock(g_state_mutex);
    if (g_in_progress.count(key) && !g_in_progress[key].empty())
      return "{\"success\":true,\"already\":true}";
    g_in_progress[key] = source;
    persist_state();
  }
  std::thread(follow_file, key, source, log_path(project, uid)).detach();
  return "{\"success\":true}";
}

std::string handle_get_log_size(const std::string& body) {
  const std::string uid = get_string(body, "run_uid");
  const std::string project = get_string(body, "project");
  struct stat st{};
  long size = -1;
  if (::stat(log_path(project, uid).c_str(), &st) == 0) size = st.st_size;
  return "{\"success\":true,\"size\":" + std::to_string(size) + "}";
}

std::string handle_stop_logs(const std::string& body) {
  const std::string uid = get_string(body, "run_uid");
  const std::string project = get_string(body, "project");
  std::lock_guard<std::mutex> lock(g_state_mutex);
  if (uid.empty()) {
    // stop every run of the project
    for (auto& [key, source] : g_in_progress)
      if (key.rfind(project + "/", 0) == 0) source.clear();
  } else {
    g_in_progress[run_key(project, uid)].clear();
  }
  persist_state();
  return "{\"success\":true}";
}

std::string handle_delete_logs(const std::string& body) {
  const std::string uid = get_string(body, "run_uid");
  const std::string project = get_string(body, "project");
  {
    std::lock_guard<std::mutex> lock(g_state_mutex);
    g_in_progress.erase(run_key(project, uid));
    persist_state();
  }
  ::unlink(log_path(project, uid).c_str());
  return "{\"success\":true}";
}

std::string handle_list(const std::string&) {
  std::ostringstream out;
  out << "{\"success\":true,\"runs\":[";
  std::lock_guard<std::mutex> lock(g_state_mutex);
  bool first = true;
  for (const auto& [key, source] : g_in_progress) {
    if (source.empty()) continue;
    if (!first) out << ",";
    out << "\"" << json_escape(key) << "\"";
    first = false;
  }
  out << "]}";
  return out.str();
}

void send_all(int fd, const char* data, size_t len) {
  size_t sent = 0;
  while (sent < len) {
    ssize_t n = ::send(fd, data + sent, len - sent, MSG_NOSIGNAL);
    if (n <= 0) return;
    sent += (size_t)n;
  }
}

void handle_get_logs(int fd, const std::string& body) {
  const std::string uid = get_string(body, "run_uid");
  const std::string project = get_string(body, "project");
  long offset = get_int(body, "offset", 0);
  long size = get_int(body, "size", 0);
  const std::string path = log_path(project, uid);
  struct stat st{};
  if (::stat(path.c_str(), &st) != 0) {
    const std::string resp =
        "{\"success\":false,\"error\":\"log not found\"}\n";
    send_all(fd, resp.data(), resp.size());
    return;
  }
  long avail = st.st_size - offset;
  if (avail < 0) avail = 0;
  if (size > 0 && size < avail) avail = size;
  const std::string header =
      "{\"success\":true,\"size\":" + std::to_string(avail) + "}\n";
  send_all(fd, header.data(), header.size());
  // stream the payload in chunks (the GetLogs server-stream analog)
  std::ifstream in(path, std::ios::binary);
  in.seekg(offset);
  std::vector<char> buf(1 << 16);
  long remaining = avail;
  while (remaining > 0 && in.good()) {
    const long want = std::min<long>(remaining, (long)buf.size());
    in.read(buf.data(), want);
    const std::streamsize got = in.gcount();
    if (got <= 0) break;
    send_all(fd, buf.data(), (size_t)got);
    remaining -= got;
  }
}

void serve_client(int fd) {
  std::string pending;
  char buf[4096];
  while (!g_shutdown.load()) {
    ssize_t n = ::recv(fd, buf, sizeof(buf), 0);
    if (n <= 0) break;
    pending.append(buf, (size_t)n);
    size_t nl;
    while ((nl = pending.find('\n')) != std::string::npos) {
      const std::string line = pending.substr(0, nl);
      pending.erase(0, nl + 1);
      if (line.empty()) continue;
      const std::string op = get_string(line, "op");
      if (op == "get_logs") {
        handle_get_logs(fd, line);
        continue;
      }
      std::string resp;
      if (op == "start_log") resp = handle_start_log(line);
      else if (op == "get_log_size") resp = handle_get_log_size(line);
      else if (op == "stop_logs") resp = handle_stop_logs(line);
      else if (op == "delete_logs") resp = handle_delete_logs(line);
      else if (op == "list_runs_in_progress") resp = handle_list(line);
      else if (op == "shutdown") {
        resp = "{\"success\":true}";
        g_shutdown.store(true);
      } else resp = "{\"success\":false,\"error\":\"unknown op\"}";
      resp += "\n";
      send_all(fd, resp.data(), resp.size());
      if (g_shutdown.load()) break;
    }
  }
  ::close(fd);
}

}  // namespace

int main(int argc, char** argv) {
  const int port = argc > 1 ? std::atoi(argv[1]) : 18766;
  g_log_dir = argc > 2 ? argv[2] : "./logs";
  ::mkdir(g_log_dir.c_str(), 0755);
  load_state();
  // resume followers for persisted in-progress runs
  {
    std::lock_guard<std::mutex> lock(g_state_mutex);
    for (const auto& [key, source] : g_in_progress) {
      if (source.empty()) continue;
      const size_t slash = key.find('/');
      std::thread(follow_file, key, source,
                  log_path(key.substr(0, slash), key.substr(slash + 1)))
          .detach();
    }
  }

  int listener = ::socket(AF_INET, SOCK_STREAM, 0);
  int opt = 1;
  ::setsockopt(listener, SOL_SOCKET, SO_REUSEADDR, &opt, sizeof(opt));
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_addr.s_addr = inet_addr("127.0.0.1");
  addr.sin_port = htons((uint16_t)port);
  if (::bind(listener, (sockaddr*)&addr, sizeof(addr)) != 0) {
    fprintf(stderr, "log_collector: bind failed on port %d\n", port);
    return 1;
  }
  ::listen(listener, 16);
  fprintf(stderr, "log_collector: listening on 127.0.0.1:%d dir=%s\n",
          port, g_log_dir.c_str());
  while (!g_shutdown.load()) {
    int fd = ::accept(listener, nullptr, nullptr);
    if (fd < 0) continue;
    std::thread(serve_client, fd).detach();
  }
  ::close(listener);
  // give client threads a moment to flush
  ::usleep(200 * 1000);
  return 0;
}
