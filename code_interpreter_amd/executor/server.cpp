// executor-server: in-sandbox data plane for the MI355X code interpreter.
//
// C++ re-design of the reference's Rust executor (executor/server.rs:1-200):
// same three routes over HTTP/1.1 --
//   PUT  /workspace/{path}   streaming upload (mkdir -p, chunked or sized)
//   GET  /workspace/{path}   download
//   POST /execute            {source_code, timeout?, env?} ->
//                            {stdout, stderr, exit_code, files:[abs paths]}
// plus GET /healthz (readiness: reports whether the pre-warmed runner is up).
//
// Differences by design (MI355X-first, documented in README):
//  - user code runs via a pre-forked Python "zygote" (zygote.py): numpy and
//    the sandbox runtime are imported once per executor, HIP is initialized
//    in a pre-warmed child before any request arrives, so per-request cost
//    is a job handoff instead of a cold interpreter + HIP start
//    (the reference pays upm+pip+xonsh cold start per request,
//    server.rs:126-169 -- including a "TODO ~80ms" it never took).
//  - scripts run under the python sandbox runtime, not xonsh; xonsh's
//    headline `!cmd` shell escapes ARE supported via a compile-gated
//    source transform (sandbox_runtime.transform_shell_escapes);
//    other xonsh syntax ($VAR, $(...)) is not (MIGRATION.md).
//  - dependency auto-install (reference: upm guess + pip, server.rs:126-147)
//    is an AST import scan + pip in the zygote child (depscan in
//    sandbox_runtime.py), pointed at a wheelhouse via APP_PIP_EXTRA_ARGS.
//  - changed-file detection matches the reference (ctime > exec start over
//    top-level /workspace entries, non-recursive, server.rs:98-118);
//    APP_SCAN_RECURSIVE=1 enables the fixed recursive scan.
//
// Config via env: APP_LISTEN_ADDR (host:port) or APP_LISTEN_UNIX (socket
// path), APP_WORKSPACE, APP_PYTHON, APP_RUNTIME_DIR (zygote.py location),
// APP_ZYGOTE=0 to disable pre-fork (cold subprocess per request),
// APP_SCAN_RECURSIVE, plus APP_* forwarded to the sandbox runtime.

#include <arpa/inet.h>
#include <dirent.h>
#include <errno.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <signal.h>
#include <sys/prctl.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <sys/types.h>
#include <sys/un.h>
#include <sys/wait.h>
#include <unistd.h>

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <map>
#include <mutex>
#include <string>
#include <thread>
#include <chrono>
#include <memory>
#include <vector>

// ---------------------------------------------------------------------------
// small JSON (only what the internal pod API needs)
// ---------------------------------------------------------------------------
namespace json {

struct Value;
using Object = std::map<std::string, Value>;
using Array = std::vector<Value>;

struct Value {
  enum Kind { Null, Bool, Num, Str, Arr, Obj } kind = Null;
  bool b = false;
  double num = 0;
  std::string str;
  std::vector<Value> arr;
  std::map<std::string, Value> obj;

  bool is_string() const { return kind == Str; }
  bool is_object() const { return kind == Obj; }
  const Value* get(const std::string& k) const {
    if (kind != Obj) return nullptr;
    auto it = obj.find(k);
    return it == obj.end() ? nullptr : &it->second;
  }
};

struct Parser {
  const char* p;
  const char* end;
  bool ok = true;

  explicit Parser(const std::string& s) : p(s.data()), end(s.data() + s.size()) {}

  void skip_ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) p++;
  }
  bool consume(char c) {
    skip_ws();
    if (p < end && *p == c) { p++; return true; }
    return false;
  }
  Value parse() {
    Value v = parse_value();
    skip_ws();
    if (p != end) ok = false;
    return v;
  }
  Value parse_value() {
    skip_ws();
    if (p >= end) { ok = false; return {}; }
    switch (*p) {
      case '{': return parse_object();
      case '[': return parse_array();
      case '"': return parse_string();
      case 't': case 'f': return parse_bool();
      case 'n': return parse_null();
      default: return parse_number();
    }
  }
  Value parse_object() {
    Value v; v.kind = Value::Obj;
    consume('{');
    skip_ws();
    if (consume('}')) return v;
    while (ok) {
      skip_ws();
      if (p >= end || *p != '"') { ok = false; break; }
      Value key = parse_string();
      if (!consume(':')) { ok = false; break; }
      v.obj[key.str] = parse_value();
      if (consume(',')) continue;
      if (consume('}')) break;
      ok = false;
    }
    return v;
  }
  Value parse_array() {
    Value v; v.kind = Value::Arr;
    consume('[');
    skip_ws();
    if (consume(']')) return v;
    while (ok) {
      v.arr.push_back(parse_value());
      if (consume(',')) continue;
      if (consume(']')) break;
      ok = false;
    }
    return v;
  }
  Value parse_string() {
    Value v; v.kind = Value::Str;
    p++;  // opening quote
    while (p < end && *p != '"') {
      if (*p == '\\') {
        p++;
        if (p >= end) { ok = false; return v; }
        switch (*p) {
          case '"': v.str += '"'; break;
          case '\\': v.str += '\\'; break;
          case '/': v.str += '/'; break;
          case 'b': v.str += '\b'; break;
          case 'f': v.str += '\f'; break;
          case 'n': v.str += '\n'; break;
          case 'r': v.str += '\r'; break;
          case 't': v.str += '\t'; break;
          case 'u': {
            if (end - p < 5) { ok = false; return v; }
            unsigned cp = 0;
            for (int i = 1; i <= 4; i++) {
              char c = p[i];
              cp <<= 4;
              if (c >= '0' && c <= '9') cp |= c - '0';
              else if (c >= 'a' && c <= 'f') cp |= c - 'a' + 10;
              else if (c >= 'A' && c <= 'F') cp |= c - 'A' + 10;
              else { ok = false; return v; }
            }
            p += 4;
            // surrogate pair
            if (cp >= 0xD800 && cp <= 0xDBFF && end - p >= 7 && p[1] == '\\' &&
                p[2] == 'u') {
              unsigned lo = 0;
              bool lo_ok = true;
              for (int i = 3; i <= 6; i++) {
                char c = p[i];
                lo <<= 4;
                if (c >= '0' && c <= '9') lo |= c - '0';
                else if (c >= 'a' && c <= 'f') lo |= c - 'a' + 10;
                else if (c >= 'A' && c <= 'F') lo |= c - 'A' + 10;
                else { lo_ok = false; break; }
              }
              if (lo_ok && lo >= 0xDC00 && lo <= 0xDFFF) {
                cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
                p += 6;
              }
            }
            // encode UTF-8
            if (cp < 0x80) {
              v.str += (char)cp;
            } else if (cp < 0x800) {
              v.str += (char)(0xC0 | (cp >> 6));
              v.str += (char)(0x80 | (cp & 0x3F));
            } else if (cp < 0x10000) {
              v.str += (char)(0xE0 | (cp >> 12));
              v.str += (char)(0x80 | ((cp >> 6) & 0x3F));
              v.str += (char)(0x80 | (cp & 0x3F));
            } else {
              v.str += (char)(0xF0 | (cp >> 18));
              v.str += (char)(0x80 | ((cp >> 12) & 0x3F));
              v.str += (char)(0x80 | ((cp >> 6) & 0x3F));
              v.str += (char)(0x80 | (cp & 0x3F));
            }
            break;
          }
          default: ok = false; return v;
        }
        p++;
      } else {
        v.str += *p++;
      }
    }
    if (p >= end) { ok = false; return v; }
    p++;  // closing quote
    return v;
  }
  Value parse_bool() {
    Value v; v.kind = Value::Bool;
    if (end - p >= 4 && strncmp(p, "true", 4) == 0) { v.b = true; p += 4; }
    else if (end - p >= 5 && strncmp(p, "false", 5) == 0) { v.b = false; p += 5; }
    else ok = false;
    return v;
  }
  Value parse_null() {
    Value v;
    if (end - p >= 4 && strncmp(p, "null", 4) == 0) p += 4;
    else ok = false;
    return v;
  }
  Value parse_number() {
    Value v; v.kind = Value::Num;
    char* num_end = nullptr;
    v.num = strtod(p, &num_end);
    if (num_end == p) ok = false;
    p = num_end;
    return v;
  }
};

inline void escape_to(const std::string& s, std::string& out) {
  for (unsigned char c : s) {
    switch (c) {
      case '"': out += "\\\""; break;
      case '\\': out += "\\\\"; break;
      case '\n': out += "\\n"; break;
      case '\r': out += "\\r"; break;
      case '\t': out += "\\t"; break;
      case '\b': out += "\\b"; break;
      case '\f': out += "\\f"; break;
      default:
        if (c < 0x20) {
          char buf[8];
          snprintf(buf, sizeof buf, "\\u%04x", c);
          out += buf;
        } else {
          out += (char)c;
        }
    }
  }
}

inline std::string quote(const std::string& s) {
  std::string out = "\"";
  escape_to(s, out);
  out += "\"";
  return out;
}

// Replace invalid UTF-8 sequences with U+FFFD so captured stdout/stderr
// always embeds as valid-UTF-8 JSON (reference parity: server.rs uses
// String::from_utf8_lossy; without this, user code printing raw bytes
// turns a 200 into a client-side decode error).
inline std::string to_valid_utf8(const std::string& s) {
  std::string out;
  out.reserve(s.size());
  size_t i = 0, n = s.size();
  auto cont = [&](size_t k) {
    return i + k < n && (static_cast<unsigned char>(s[i + k]) & 0xC0) == 0x80;
  };
  while (i < n) {
    unsigned char c = s[i];
    if (c < 0x80) {
      out += (char)c;
      i += 1;
    } else if ((c & 0xE0) == 0xC0 && c >= 0xC2 && cont(1)) {
      out.append(s, i, 2);
      i += 2;
    } else if ((c & 0xF0) == 0xE0 && cont(1) && cont(2)) {
      unsigned char c1 = s[i + 1];
      // reject overlongs (E0 80-9F) and surrogates (ED A0-BF)
      if ((c == 0xE0 && c1 < 0xA0) || (c == 0xED && c1 > 0x9F)) {
        out += "\xEF\xBF\xBD";
        i += 1;
      } else {
        out.append(s, i, 3);
        i += 3;
      }
    } else if ((c & 0xF8) == 0xF0 && c <= 0xF4 && cont(1) && cont(2) && cont(3)) {
      unsigned char c1 = s[i + 1];
      if ((c == 0xF0 && c1 < 0x90) || (c == 0xF4 && c1 > 0x8F)) {
        out += "\xEF\xBF\xBD";
        i += 1;
      } else {
        out.append(s, i, 4);
        i += 4;
      }
    } else {
      out += "\xEF\xBF\xBD";  // U+FFFD replacement character
      i += 1;
    }
  }
  return out;
}

}  // namespace json

// ---------------------------------------------------------------------------
// config
// ---------------------------------------------------------------------------
static std::string env_or(const char* name, const std::string& dflt) {
  const char* v = getenv(name);
  return (v && *v) ? std::string(v) : dflt;
}

struct ServerConfig {
  std::string listen_addr;   // "host:port" or empty
  std::string listen_unix;   // unix socket path or empty
  int max_connections = 512;  // concurrent handler threads (APP_MAX_CONNECTIONS)
  std::string workspace;
  std::string python;
  std::string runtime_dir;   // where zygote.py lives
  bool zygote = true;
  bool scan_recursive = false;
  double default_timeout = 60.0;
  // request limits: a sandbox host must stay alive under hostile input,
  // so header and body sizes are hard-capped (actix enforces similar
  // defaults in the reference; a hand-written parser has to do it itself)
  size_t max_header_bytes = 64 * 1024;
  size_t max_body_bytes = (size_t)1 << 30;  // 1 GiB, APP_MAX_BODY_BYTES
};

static ServerConfig g_cfg;

// ---------------------------------------------------------------------------
// zygote management
// ---------------------------------------------------------------------------
static double now_ms() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return ts.tv_sec * 1e3 + ts.tv_nsec / 1e6;
}

struct JobState {
  std::mutex mu;
  std::condition_variable cv;
  bool started = false;
  bool done = false;
  long pid = -1;
  int exit_code = -1;
  double t_started_ms = 0;
  double t_done_ms = 0;
};

class Zygote {
 public:
  bool start() {
    int fds[2];
    if (socketpair(AF_UNIX, SOCK_STREAM, 0, fds) != 0) return false;
    // Everything the child needs is built BEFORE fork(): this may be
    // called from a monitor thread, and a forked child of a multithreaded
    // process deadlocks if it allocates while another thread held the
    // heap lock at fork time -- the child below is exec/exit only.
    std::string zygote_py = g_cfg.runtime_dir + "/zygote.py";
    char fd_str[16];
    snprintf(fd_str, sizeof fd_str, "%d", fds[1]);
    const char* argv[] = {g_cfg.python.c_str(), "-u", zygote_py.c_str(),
                          "--fd", fd_str, nullptr};
    pid_t pid = fork();
    if (pid < 0) return false;
    if (pid == 0) {
      close(fds[0]);
      execvp(argv[0], const_cast<char* const*>(argv));
      _exit(127);
    }
    close(fds[1]);
    fd_ = fds[0];
    pid_ = pid;
    reader_ = std::thread([this] { read_loop(); });
    reader_.detach();
    return true;
  }

  bool alive() const { return alive_.load(); }
  bool warm() const { return warm_.load(); }
  pid_t pid() const { return pid_; }

  // Submit a job; returns the JobState used to track it.
  std::shared_ptr<JobState> submit(uint64_t id, const std::string& request_json) {
    auto st = std::make_shared<JobState>();
    {
      std::lock_guard<std::mutex> lk(jobs_mu_);
      jobs_[id] = st;
    }
    std::lock_guard<std::mutex> lk(write_mu_);
    std::string line = request_json + "\n";
    ssize_t off = 0;
    while (off < (ssize_t)line.size()) {
      ssize_t n = ::write(fd_, line.data() + off, line.size() - off);
      if (n <= 0) { alive_.store(false); break; }
      off += n;
    }
    return st;
  }

  void drop(uint64_t id) {
    std::lock_guard<std::mutex> lk(jobs_mu_);
    jobs_.erase(id);
  }

 private:
  void read_loop() {
    std::string buf;
    char chunk[4096];
    while (true) {
      ssize_t n = ::read(fd_, chunk, sizeof chunk);
      if (n <= 0) { alive_.store(false); break; }
      buf.append(chunk, n);
      size_t pos;
      while ((pos = buf.find('\n')) != std::string::npos) {
        std::string line = buf.substr(0, pos);
        buf.erase(0, pos + 1);
        handle_line(line);
      }
    }
  }

  void handle_line(const std::string& line) {
    json::Parser parser(line);
    json::Value msg = parser.parse();
    if (!parser.ok || !msg.is_object()) return;
    const json::Value* ev = msg.get("event");
    if (!ev || !ev->is_string()) return;
    if (ev->str == "warm") { warm_.store(true); return; }
    const json::Value* idv = msg.get("id");
    if (!idv) return;
    uint64_t id = (uint64_t)idv->num;
    std::shared_ptr<JobState> st;
    {
      std::lock_guard<std::mutex> lk(jobs_mu_);
      auto it = jobs_.find(id);
      if (it == jobs_.end()) return;
      st = it->second;
    }
    std::lock_guard<std::mutex> lk(st->mu);
    if (ev->str == "start") {
      const json::Value* pidv = msg.get("pid");
      st->pid = pidv ? (long)pidv->num : -1;
      st->started = true;
      st->t_started_ms = now_ms();
    } else if (ev->str == "exit") {
      const json::Value* code = msg.get("code");
      st->exit_code = code ? (int)code->num : -1;
      st->done = true;
      st->t_done_ms = now_ms();
    }
    st->cv.notify_all();
  }

  int fd_ = -1;
  pid_t pid_ = -1;
  std::thread reader_;
  std::atomic<bool> alive_{true};
  std::atomic<bool> warm_{false};
  std::mutex write_mu_;
  std::mutex jobs_mu_;
  std::map<uint64_t, std::shared_ptr<JobState>> jobs_;
};

// Held atomically: the respawn monitor swaps it from its own thread while
// request threads read it (a plain pointer here is a data race).
static std::atomic<Zygote*> g_zygote{nullptr};
static std::atomic<uint64_t> g_job_id{1};

// ---------------------------------------------------------------------------
// sandbox sessions: one fresh workspace per execution request, served by one
// long-lived engine process (the per-request isolation unit is the forked
// single-use interpreter, not the server)
// ---------------------------------------------------------------------------
struct SandboxRegistry {
  std::mutex mu;
  std::map<std::string, std::string> workspaces;  // id -> dir
};
static SandboxRegistry g_sandboxes;
static std::string g_sessions_root;

static std::string random_hex(int nbytes) {
  static const char* hexd = "0123456789abcdef";
  unsigned char buf[32];
  FILE* f = fopen("/dev/urandom", "rb");
  size_t got = f ? fread(buf, 1, (size_t)nbytes, f) : 0;
  if (f) fclose(f);
  std::string out;
  for (size_t i = 0; i < got; i++) {
    out += hexd[buf[i] >> 4];
    out += hexd[buf[i] & 15];
  }
  return out;
}

static void rmtree(const std::string& path) {
  DIR* d = opendir(path.c_str());
  if (d) {
    struct dirent* ent;
    while ((ent = readdir(d)) != nullptr) {
      std::string name = ent->d_name;
      if (name == "." || name == "..") continue;
      std::string full = path + "/" + name;
      struct stat st;
      if (lstat(full.c_str(), &st) == 0 && S_ISDIR(st.st_mode))
        rmtree(full);
      else
        unlink(full.c_str());
    }
    closedir(d);
  }
  rmdir(path.c_str());
}

// ---------------------------------------------------------------------------
// filesystem helpers
// ---------------------------------------------------------------------------
static bool mkdirs(const std::string& path) {
  std::string cur;
  size_t i = 0;
  if (!path.empty() && path[0] == '/') { cur = "/"; i = 1; }
  while (i <= path.size()) {
    if (i == path.size() || path[i] == '/') {
      if (!cur.empty() && cur != "/") {
        if (mkdir(cur.c_str(), 0777) != 0 && errno != EEXIST) return false;
      }
      if (i < path.size()) cur += '/';
    } else {
      cur += path[i];
    }
    i++;
  }
  return true;
}

// Reject path traversal: no ".." segments, no absolute paths.
static bool safe_rel_path(const std::string& rel) {
  if (rel.empty() || rel[0] == '/') return false;
  size_t start = 0;
  while (start <= rel.size()) {
    size_t slash = rel.find('/', start);
    if (slash == std::string::npos) slash = rel.size();
    std::string seg = rel.substr(start, slash - start);
    if (seg == "..") return false;
    start = slash + 1;
  }
  return true;
}

struct TimeSpec {
  int64_t sec;
  int64_t nsec;
  bool newer_than(const TimeSpec& other) const {
    return sec > other.sec || (sec == other.sec && nsec > other.nsec);
  }
};

// Changed-file scan, reference parity (server.rs:98-118): regular files whose
// ctime is strictly after `since`. Non-recursive unless cfg.scan_recursive.
static void scan_changed(const std::string& dir, const std::string& rel_prefix,
                         const TimeSpec& since, bool recursive,
                         std::vector<std::string>& out) {
  DIR* d = opendir(dir.c_str());
  if (!d) return;
  struct dirent* ent;
  while ((ent = readdir(d)) != nullptr) {
    std::string name = ent->d_name;
    if (name == "." || name == "..") continue;
    std::string full = dir + "/" + name;
    struct stat st;
    if (lstat(full.c_str(), &st) != 0) continue;
    if (S_ISDIR(st.st_mode)) {
      if (recursive)
        scan_changed(full, rel_prefix + name + "/", since, recursive, out);
      continue;
    }
    if (!S_ISREG(st.st_mode)) continue;
    TimeSpec ctime{(int64_t)st.st_ctim.tv_sec, (int64_t)st.st_ctim.tv_nsec};
    if (ctime.newer_than(since)) out.push_back("/workspace/" + rel_prefix + name);
  }
  closedir(d);
}

static std::string read_file(const std::string& path) {
  std::string out;
  int fd = open(path.c_str(), O_RDONLY);
  if (fd < 0) return out;
  char buf[65536];
  ssize_t n;
  while ((n = read(fd, buf, sizeof buf)) > 0) out.append(buf, n);
  close(fd);
  return out;
}

static bool write_file(const std::string& path, const std::string& data) {
  int fd = open(path.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0666);
  if (fd < 0) return false;
  size_t off = 0;
  while (off < data.size()) {
    ssize_t n = write(fd, data.data() + off, data.size() - off);
    if (n <= 0) { close(fd); return false; }
    off += n;
  }
  close(fd);
  return true;
}

// ---------------------------------------------------------------------------
// execution
// ---------------------------------------------------------------------------
struct ExecOutcome {
  std::string stdout_text;
  std::string stderr_text;
  int exit_code = -1;
  // per-stage timings (ms): dispatch = submit -> interpreter running,
  // run = interpreter start -> exit
  double t_dispatch_ms = 0;
  double t_run_ms = 0;
};

// Cold path: plain fork/exec of python (used when the zygote is down or
// APP_ZYGOTE=0; same observable behavior, slower start).
static ExecOutcome run_cold(const std::string& script_path,
                            const std::string& stdout_path,
                            const std::string& stderr_path,
                            const std::map<std::string, std::string>& extra_env,
                            double timeout_s, const std::string& workspace) {
  ExecOutcome out;
  // built before fork(): the child of this multithreaded process must
  // not allocate (heap lock may be held by another thread at fork time)
  std::string runner = g_cfg.runtime_dir + "/runner.py";
  pid_t pid = fork();
  if (pid < 0) {
    out.stderr_text = "fork failed";
    return out;
  }
  if (pid == 0) {
    setsid();
    int so = open(stdout_path.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0666);
    int se = open(stderr_path.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0666);
    int si = open("/dev/null", O_RDONLY);
    if (si >= 0) dup2(si, 0);
    if (so >= 0) dup2(so, 1);
    if (se >= 0) dup2(se, 2);
    for (auto& kv : extra_env) setenv(kv.first.c_str(), kv.second.c_str(), 1);
    if (chdir(workspace.c_str()) != 0) _exit(126);
    // run through the sandbox runtime (runner.py), not bare python: the
    // cold path must keep the zygote path's semantics (import hooks,
    // dep install, rlimits, shell escapes)
    execlp(g_cfg.python.c_str(), g_cfg.python.c_str(), runner.c_str(),
           script_path.c_str(), (char*)nullptr);
    _exit(127);
  }
  // wait with timeout
  int64_t deadline_ms = (int64_t)(timeout_s * 1000);
  int64_t waited = 0;
  int status = 0;
  bool timed_out = false;
  while (true) {
    pid_t r = waitpid(pid, &status, WNOHANG);
    if (r == pid) break;
    if (waited >= deadline_ms) {
      kill(-pid, SIGKILL);
      kill(pid, SIGKILL);
      waitpid(pid, &status, 0);
      timed_out = true;
      break;
    }
    usleep(2000);
    waited += 2;
  }
  if (timed_out) {
    out.stdout_text = "";
    out.stderr_text = "Execution timed out";
    out.exit_code = -1;
    return out;
  }
  out.stdout_text = read_file(stdout_path);
  out.stderr_text = read_file(stderr_path);
  out.exit_code = WIFEXITED(status) ? WEXITSTATUS(status) : -1;
  return out;
}

static ExecOutcome run_via_zygote(const std::string& script_path,
                                  const std::string& stdout_path,
                                  const std::string& stderr_path,
                                  const std::map<std::string, std::string>& extra_env,
                                  double timeout_s, const std::string& workspace) {
  uint64_t id = g_job_id.fetch_add(1);
  std::string req = "{\"event\":\"run\",\"id\":" + std::to_string(id) +
                    ",\"script\":" + json::quote(script_path) +
                    ",\"cwd\":" + json::quote(workspace) +
                    ",\"stdout\":" + json::quote(stdout_path) +
                    ",\"stderr\":" + json::quote(stderr_path) + ",\"env\":{";
  bool first = true;
  for (auto& kv : extra_env) {
    if (!first) req += ",";
    first = false;
    req += json::quote(kv.first) + ":" + json::quote(kv.second);
  }
  req += "}}";

  double t_submit = now_ms();
  Zygote* zygote = g_zygote.load();
  auto st = zygote->submit(id, req);

  ExecOutcome out;
  std::unique_lock<std::mutex> lk(st->mu);
  auto deadline =
      std::chrono::steady_clock::now() +
      std::chrono::milliseconds((int64_t)(timeout_s * 1000));
  bool finished = false;
  while (!finished && std::chrono::steady_clock::now() < deadline) {
    finished = st->cv.wait_for(lk, std::chrono::milliseconds(100),
                               [&] { return st->done; });
    if (!finished && !st->started && !zygote->alive()) {
      // the zygote died before dispatching this job: retry cold
      lk.unlock();
      zygote->drop(id);
      return run_cold(script_path, stdout_path, stderr_path, extra_env,
                      timeout_s, workspace);
    }
  }
  if (!finished) {
    // timeout: kill the child's process group (reference parity:
    // ("", "Execution timed out", -1), server.rs:169)
    if (st->pid > 0) {
      kill((pid_t)-st->pid, SIGKILL);
      kill((pid_t)st->pid, SIGKILL);
    }
    // give the zygote a moment to reap and report
    st->cv.wait_for(lk, std::chrono::seconds(5), [&] { return st->done; });
    zygote->drop(id);
    out.stdout_text = "";
    out.stderr_text = "Execution timed out";
    out.exit_code = -1;
    return out;
  }
  zygote->drop(id);
  out.exit_code = st->exit_code;
  double t_started = st->t_started_ms > 0 ? st->t_started_ms : t_submit;
  double t_done = st->t_done_ms > 0 ? st->t_done_ms : now_ms();
  out.t_dispatch_ms = t_started - t_submit;
  out.t_run_ms = t_done - t_started;
  lk.unlock();
  out.stdout_text = read_file(stdout_path);
  out.stderr_text = read_file(stderr_path);
  return out;
}

// ---------------------------------------------------------------------------
// HTTP
// ---------------------------------------------------------------------------
struct HttpRequest {
  std::string method;
  std::string path;
  std::map<std::string, std::string> headers;  // lowercased keys
  std::string body;
  bool keep_alive = true;
};

// How the body will arrive, decided from the headers; the route handler
// then picks WHERE it goes (memory for JSON routes, straight to the
// workspace file for PUT -- a 1 GiB upload must not cost 1 GiB of RSS).
struct BodyPlan {
  bool chunked = false;
  bool has_length = false;
  size_t length = 0;
  bool has_body() const { return chunked || (has_length && length > 0); }
};

class Conn {
 public:
  explicit Conn(int fd) : fd_(fd) {}
  ~Conn() { close(fd_); }

  // Phase 1: headers only; the body stays on the socket until the route
  // decides its sink. Content-Length above the cap is refused up front.
  bool read_request_headers(HttpRequest& req, BodyPlan& plan) {
    std::string header_block;
    if (!read_until_headers(header_block)) return false;
    if (!parse_headers(header_block, req)) return false;

    auto te = req.headers.find("transfer-encoding");
    if (te != req.headers.end() && te->second.find("chunked") != std::string::npos) {
      plan.chunked = true;
      return true;
    }
    auto cl = req.headers.find("content-length");
    if (cl != req.headers.end()) {
      plan.has_length = true;
      plan.length = (size_t)strtoull(cl->second.c_str(), nullptr, 10);
      if (plan.length > g_cfg.max_body_bytes) {
        respond(413, "Payload Too Large", "{\"error\":\"body too large\"}");
        return false;  // cannot resync mid-body: close the connection
      }
    }
    return true;
  }

  // Phase 2a: body into memory (JSON routes).
  bool read_body(const BodyPlan& plan, std::string& out) {
    if (plan.chunked) return read_chunked_body(out);
    if (plan.has_length) return read_exact_body(plan.length, out);
    return true;
  }

  // Phase 2b: body streamed to an fd in bounded chunks (workspace PUT).
  // On sink failure sets *write_err and keeps draining is NOT attempted
  // (caller closes the connection).
  bool read_body_to_fd(const BodyPlan& plan, int fd, bool* write_err) {
    *write_err = false;
    auto sink = [&](const char* data, size_t n) -> bool {
      size_t off = 0;
      while (off < n) {
        ssize_t w = ::write(fd, data + off, n - off);
        if (w <= 0) { *write_err = true; return false; }
        off += (size_t)w;
      }
      return true;
    };
    if (plan.chunked) return read_chunked_to_sink(sink);
    if (plan.has_length) return read_sized_to_sink(plan.length, sink);
    return true;
  }

  // Phase 2c: body consumed and discarded (error responses on routes
  // whose body we never needed, keeping the connection parseable).
  bool read_body_discard(const BodyPlan& plan) {
    auto sink = [](const char*, size_t) { return true; };
    if (plan.chunked) return read_chunked_to_sink(sink);
    if (plan.has_length) return read_sized_to_sink(plan.length, sink);
    return true;
  }

  // Streamed file response: headers + 256 KiB read/send loop, so a
  // multi-GB workspace download costs O(chunk) memory.
  void respond_file(const std::string& path, int64_t size) {
    int fd = open(path.c_str(), O_RDONLY);
    if (fd < 0) {
      respond(404, "Not Found", "{\"error\":\"not found\"}");
      return;
    }
    std::string head =
        "HTTP/1.1 200 OK\r\nContent-Type: application/octet-stream"
        "\r\nContent-Length: " + std::to_string(size) + "\r\n\r\n";
    send_all(head);
    char chunk[262144];
    int64_t left = size;
    while (left > 0) {
      ssize_t n = read(fd, chunk, std::min<int64_t>(left, sizeof chunk));
      if (n <= 0) break;
      send_all(std::string(chunk, (size_t)n));
      left -= n;
    }
    close(fd);
  }

  void respond(int code, const char* status, const std::string& body,
               const char* content_type = "application/json") {
    std::string resp = "HTTP/1.1 " + std::to_string(code) + " " + status +
                       "\r\nContent-Type: " + content_type +
                       "\r\nContent-Length: " + std::to_string(body.size()) +
                       "\r\n\r\n";
    resp += body;
    send_all(resp);
  }

 private:
  bool read_until_headers(std::string& out) {
    while (true) {
      size_t pos = buf_.find("\r\n\r\n");
      if (pos != std::string::npos) {
        if (pos + 4 > g_cfg.max_header_bytes) {
          respond(431, "Request Header Fields Too Large",
                  "{\"error\":\"headers too large\"}");
          return false;
        }
        out = buf_.substr(0, pos + 4);
        buf_.erase(0, pos + 4);
        return true;
      }
      if (buf_.size() > g_cfg.max_header_bytes) {
        respond(431, "Request Header Fields Too Large",
                "{\"error\":\"headers too large\"}");
        return false;
      }
      if (!fill()) return false;
    }
  }
  bool read_exact_body(size_t len, std::string& out) {
    while (buf_.size() < len) {
      if (!fill()) return false;
    }
    out = buf_.substr(0, len);
    buf_.erase(0, len);
    return true;
  }
  template <typename Sink>
  bool read_sized_to_sink(size_t len, Sink&& sink) {
    // drain what is already buffered, then stream the rest in chunks
    size_t from_buf = std::min(buf_.size(), len);
    if (from_buf) {
      if (!sink(buf_.data(), from_buf)) return false;
      buf_.erase(0, from_buf);
      len -= from_buf;
    }
    char chunk[262144];
    while (len > 0) {
      ssize_t n = recv(fd_, chunk, std::min(len, sizeof chunk), 0);
      if (n <= 0) return false;
      if (!sink(chunk, (size_t)n)) return false;
      len -= (size_t)n;
    }
    return true;
  }
  template <typename Sink>
  bool read_chunked_to_sink(Sink&& sink) {
    size_t total = 0;
    while (true) {
      size_t pos;
      while ((pos = buf_.find("\r\n")) == std::string::npos) {
        if (buf_.size() > 4096) return false;  // chunk-size line is tiny
        if (!fill()) return false;
      }
      size_t chunk_len = strtoull(buf_.substr(0, pos).c_str(), nullptr, 16);
      total += chunk_len;
      if (total > g_cfg.max_body_bytes) {
        respond(413, "Payload Too Large", "{\"error\":\"body too large\"}");
        return false;
      }
      buf_.erase(0, pos + 2);
      if (chunk_len == 0) {
        while (buf_.size() < 2) {
          if (!fill()) return false;
        }
        buf_.erase(0, 2);
        return true;
      }
      // stream this chunk's payload
      size_t remaining = chunk_len;
      while (remaining > 0) {
        if (buf_.empty() && !fill()) return false;
        size_t take = std::min(buf_.size(), remaining);
        if (!sink(buf_.data(), take)) return false;
        buf_.erase(0, take);
        remaining -= take;
      }
      while (buf_.size() < 2) {
        if (!fill()) return false;
      }
      buf_.erase(0, 2);  // chunk-terminating CRLF
    }
  }
  bool read_chunked_body(std::string& out) {
    while (true) {
      size_t pos;
      while ((pos = buf_.find("\r\n")) == std::string::npos) {
        if (buf_.size() > 4096) return false;  // chunk-size line is tiny
        if (!fill()) return false;
      }
      size_t chunk_len = strtoull(buf_.substr(0, pos).c_str(), nullptr, 16);
      if (out.size() + chunk_len > g_cfg.max_body_bytes) {
        respond(413, "Payload Too Large", "{\"error\":\"body too large\"}");
        return false;
      }
      buf_.erase(0, pos + 2);
      if (chunk_len == 0) {
        // trailing CRLF (possibly trailers; we accept bare CRLF)
        while (buf_.size() < 2) {
          if (!fill()) return false;
        }
        buf_.erase(0, 2);
        return true;
      }
      while (buf_.size() < chunk_len + 2) {
        if (!fill()) return false;
      }
      out.append(buf_, 0, chunk_len);
      buf_.erase(0, chunk_len + 2);
    }
  }
  bool parse_headers(const std::string& block, HttpRequest& req) {
    size_t line_end = block.find("\r\n");
    if (line_end == std::string::npos) return false;
    std::string request_line = block.substr(0, line_end);
    size_t sp1 = request_line.find(' ');
    size_t sp2 = request_line.rfind(' ');
    if (sp1 == std::string::npos || sp2 <= sp1) return false;
    req.method = request_line.substr(0, sp1);
    req.path = request_line.substr(sp1 + 1, sp2 - sp1 - 1);
    size_t pos = line_end + 2;
    while (pos < block.size()) {
      size_t eol = block.find("\r\n", pos);
      if (eol == std::string::npos || eol == pos) break;
      std::string line = block.substr(pos, eol - pos);
      size_t colon = line.find(':');
      if (colon != std::string::npos) {
        std::string key = line.substr(0, colon);
        for (auto& c : key) c = (char)tolower((unsigned char)c);
        size_t vstart = colon + 1;
        while (vstart < line.size() && line[vstart] == ' ') vstart++;
        req.headers[key] = line.substr(vstart);
      }
      pos = eol + 2;
    }
    auto conn_hdr = req.headers.find("connection");
    req.keep_alive =
        !(conn_hdr != req.headers.end() && conn_hdr->second == "close");
    return true;
  }
  bool fill() {
    char chunk[65536];
    ssize_t n = recv(fd_, chunk, sizeof chunk, 0);
    if (n <= 0) return false;
    buf_.append(chunk, n);
    return true;
  }
  void send_all(const std::string& data) {
    size_t off = 0;
    while (off < data.size()) {
      ssize_t n = send(fd_, data.data() + off, data.size() - off, MSG_NOSIGNAL);
      if (n <= 0) return;
      off += n;
    }
  }

  int fd_;
  std::string buf_;
};

// URL-decode %XX escapes in a path.
static std::string url_decode(const std::string& s) {
  std::string out;
  for (size_t i = 0; i < s.size(); i++) {
    if (s[i] == '%' && i + 2 < s.size()) {
      auto hex = [](char c) -> int {
        if (c >= '0' && c <= '9') return c - '0';
        if (c >= 'a' && c <= 'f') return c - 'a' + 10;
        if (c >= 'A' && c <= 'F') return c - 'A' + 10;
        return -1;
      };
      int hi = hex(s[i + 1]), lo = hex(s[i + 2]);
      if (hi >= 0 && lo >= 0) {
        out += (char)((hi << 4) | lo);
        i += 2;
        continue;
      }
    }
    out += s[i];
  }
  return out;
}

// returns true when changed files were reported (the caller may keep the
// session alive for downloads); session != nullptr adds it to the response
static bool handle_execute(Conn& conn, const HttpRequest& req,
                           const std::string& workspace,
                           const std::string* session = nullptr) {
  double t_handler0 = now_ms();
  json::Parser parser(req.body);
  json::Value body = parser.parse();
  if (!parser.ok || !body.is_object() || !body.get("source_code") ||
      !body.get("source_code")->is_string()) {
    conn.respond(400, "Bad Request", "{\"error\":\"invalid request body\"}");
    return false;
  }
  const std::string& source = body.get("source_code")->str;
  double timeout_s = g_cfg.default_timeout;
  if (const json::Value* t = body.get("timeout"))
    if (t->kind == json::Value::Num) timeout_s = t->num;
  std::map<std::string, std::string> extra_env;
  if (const json::Value* env = body.get("env"))
    if (env->is_object())
      for (auto& kv : env->obj)
        if (kv.second.is_string()) extra_env[kv.first] = kv.second.str;

  mkdirs(workspace);

  // execution start time for the changed-file scan
  struct timespec now;
  clock_gettime(CLOCK_REALTIME, &now);
  TimeSpec start{(int64_t)now.tv_sec, (int64_t)now.tv_nsec};

  double t_pre0 = now_ms();
  char tmpl[] = "/tmp/exec.XXXXXX";
  char* tmpdir = mkdtemp(tmpl);
  if (!tmpdir) {
    conn.respond(500, "Internal Server Error", "{\"error\":\"mkdtemp failed\"}");
    return false;
  }
  std::string script_path = std::string(tmpdir) + "/script.py";
  std::string stdout_path = std::string(tmpdir) + "/stdout";
  std::string stderr_path = std::string(tmpdir) + "/stderr";
  write_file(script_path, source);

  double t_pre1 = now_ms();
  ExecOutcome outcome;
  Zygote* zyg = g_zygote.load();
  if (g_cfg.zygote && zyg && zyg->alive()) {
    outcome = run_via_zygote(script_path, stdout_path, stderr_path, extra_env,
                             timeout_s, workspace);
  } else {
    outcome = run_cold(script_path, stdout_path, stderr_path, extra_env,
                       timeout_s, workspace);
  }

  double t_post0 = now_ms();
  std::vector<std::string> changed;
  scan_changed(workspace, "", start, g_cfg.scan_recursive, changed);
  double t_post1 = now_ms();

  std::string child_t = read_file(stdout_path + ".t");  // child-side phases
  char timings[256];
  snprintf(timings, sizeof timings,
           ",\"timings\":{\"dispatch_ms\":%.2f,\"run_ms\":%.2f,"
           "\"handler_ms\":%.2f,\"pre_ms\":%.2f,\"scan_ms\":%.2f,"
           "\"wait_ms\":%.2f",
           outcome.t_dispatch_ms, outcome.t_run_ms, now_ms() - t_handler0,
           t_pre1 - t_pre0, t_post1 - t_post0, t_post0 - t_pre1);
  std::string timings_str = timings;
  if (!child_t.empty() && child_t[0] == '{') {
    timings_str += ",\"child\":" + child_t;
  }
  timings_str += "}";
  std::string resp =
      "{\"stdout\":" + json::quote(json::to_valid_utf8(outcome.stdout_text)) +
      ",\"stderr\":" + json::quote(json::to_valid_utf8(outcome.stderr_text)) +
                     ",\"exit_code\":" + std::to_string(outcome.exit_code) +
                     timings_str +
                     ",\"files\":[";
  for (size_t i = 0; i < changed.size(); i++) {
    if (i) resp += ",";
    resp += json::quote(changed[i]);
  }
  resp += "]";
  if (session != nullptr && !changed.empty()) {
    resp += ",\"session\":" + json::quote(*session);
  }
  resp += "}";

  // clean the temp dir
  unlink(script_path.c_str());
  unlink(stdout_path.c_str());
  unlink((stdout_path + ".t").c_str());
  unlink(stderr_path.c_str());
  rmdir(tmpdir);

  conn.respond(200, "OK", resp);
  return !changed.empty();
}

// workspace file routes (PUT/GET), shared by the legacy pod-style routes
// and the per-session routes. Returns false when the connection can no
// longer be reused (body left unread / stream aborted).
// PUT streams the body straight into the destination file and GET
// streams the file out in chunks (reference parity with server.rs:69-96's
// streaming; a 1 GiB round trip stays at O(256 KiB) executor memory).
static bool handle_workspace_io(Conn& conn, const HttpRequest& req,
                                const BodyPlan& plan,
                                const std::string& workspace,
                                const std::string& rel_encoded) {
  std::string rel = url_decode(rel_encoded);
  // a path naming a directory (trailing '/') or nothing is a client
  // error, not a 500 from the failed open
  if (!safe_rel_path(rel) || rel.empty() || rel.back() == '/') {
    bool drained = conn.read_body_discard(plan);
    conn.respond(400, "Bad Request", "{\"error\":\"bad path\"}");
    return drained;
  }
  std::string full = workspace + "/" + rel;
  if (req.method == "PUT") {
    size_t slash = full.rfind('/');
    if (slash != std::string::npos) mkdirs(full.substr(0, slash));
    int fd = open(full.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0666);
    if (fd < 0) {
      bool drained = conn.read_body_discard(plan);
      if (errno == EISDIR || errno == ENOTDIR || errno == ENAMETOOLONG ||
          errno == EINVAL) {
        conn.respond(400, "Bad Request", "{\"error\":\"bad path\"}");
      } else {
        conn.respond(500, "Internal Server Error",
                     "{\"error\":\"write failed\"}");
      }
      return drained;
    }
    bool write_err = false;
    bool ok = conn.read_body_to_fd(plan, fd, &write_err);
    close(fd);
    if (!ok) {
      unlink(full.c_str());  // partial upload must not look complete
      if (write_err)
        conn.respond(500, "Internal Server Error",
                     "{\"error\":\"write failed\"}");
      return false;
    }
    conn.respond(204, "No Content", "");
    return true;
  }
  if (!conn.read_body_discard(plan)) return false;
  if (req.method == "GET") {
    struct stat st;
    if (stat(full.c_str(), &st) != 0 || !S_ISREG(st.st_mode)) {
      conn.respond(404, "Not Found", "{\"error\":\"not found\"}");
    } else {
      conn.respond_file(full, (int64_t)st.st_size);
    }
  } else {
    conn.respond(405, "Method Not Allowed", "{\"error\":\"method\"}");
  }
  return true;
}

static bool lookup_sandbox(const std::string& id, std::string& workspace) {
  std::lock_guard<std::mutex> lk(g_sandboxes.mu);
  auto it = g_sandboxes.workspaces.find(id);
  if (it == g_sandboxes.workspaces.end()) return false;
  workspace = it->second;
  return true;
}

// Bounded concurrency: the acceptor blocks while max_connections handler
// threads are live (the kernel backlog absorbs the burst), so a
// connection storm cannot create unbounded detached threads.
static void handle_conn(int fd);

static std::atomic<int> g_active_conns{0};
static std::mutex g_conn_gate_mu;
static std::condition_variable g_conn_gate_cv;

static void handle_conn_counted(int fd) {
  handle_conn(fd);
  g_active_conns.fetch_sub(1);
  g_conn_gate_cv.notify_one();
}

static void handle_conn(int fd) {
  Conn conn(fd);
  while (true) {
    HttpRequest req;
    BodyPlan plan;
    if (!conn.read_request_headers(req, plan)) return;

    // workspace file routes consume their body streaming; every other
    // route reads it into memory here
    bool is_ws_route = req.path.rfind("/workspace/", 0) == 0;
    bool is_session_ws = false;
    if (!is_ws_route && req.path.rfind("/sandboxes/", 0) == 0) {
      std::string rest = req.path.substr(strlen("/sandboxes/"));
      size_t slash = rest.find('/');
      is_session_ws = slash != std::string::npos &&
                      rest.compare(slash + 1, strlen("workspace/"),
                                   "workspace/", strlen("workspace/")) == 0;
    }
    if (!is_ws_route && !is_session_ws) {
      if (!conn.read_body(plan, req.body)) return;
    }

    if (req.method == "GET" && req.path == "/healthz") {
      Zygote* zyg = g_zygote.load();
      bool warm = zyg && zyg->warm();
      conn.respond(200, "OK",
                   std::string("{\"status\":\"ok\",\"warm\":") +
                       (warm ? "true" : "false") + "}");
    } else if (req.path.rfind("/workspace/", 0) == 0) {
      // legacy pod-style routes: the default workspace
      if (!handle_workspace_io(conn, req, plan, g_cfg.workspace,
                               req.path.substr(strlen("/workspace/"))))
        return;
    } else if (req.method == "POST" && req.path == "/execute") {
      handle_execute(conn, req, g_cfg.workspace);
    } else if (req.method == "POST" && req.path == "/execute-ephemeral") {
      // one-shot: fresh workspace session created, executed and (when no
      // files changed) deleted within a single request -- saves two
      // round trips on the hot path; when files DID change the session
      // id is returned for downloads + explicit DELETE
      std::string id = random_hex(12);
      std::string ws = g_sessions_root + "/" + id;
      mkdirs(ws);
      {
        std::lock_guard<std::mutex> lk(g_sandboxes.mu);
        g_sandboxes.workspaces[id] = ws;
      }
      bool had_files = handle_execute(conn, req, ws, &id);
      if (!had_files) {
        {
          std::lock_guard<std::mutex> lk(g_sandboxes.mu);
          g_sandboxes.workspaces.erase(id);
        }
        rmtree(ws);
      }
    } else if (req.method == "POST" && req.path == "/sandboxes") {
      // fresh single-use workspace session
      std::string id = random_hex(12);
      std::string ws = g_sessions_root + "/" + id;
      mkdirs(ws);
      {
        std::lock_guard<std::mutex> lk(g_sandboxes.mu);
        g_sandboxes.workspaces[id] = ws;
      }
      conn.respond(200, "OK", "{\"id\":" + json::quote(id) + "}");
    } else if (req.path.rfind("/sandboxes/", 0) == 0) {
      std::string rest = req.path.substr(strlen("/sandboxes/"));
      size_t slash = rest.find('/');
      std::string id = rest.substr(0, slash == std::string::npos ? rest.size()
                                                                 : slash);
      std::string ws;
      if (!lookup_sandbox(id, ws)) {
        bool drained = !is_session_ws || conn.read_body_discard(plan);
        conn.respond(404, "Not Found", "{\"error\":\"no such sandbox\"}");
        if (!drained) return;
      } else if (slash == std::string::npos) {
        if (req.method == "DELETE") {
          {
            std::lock_guard<std::mutex> lk(g_sandboxes.mu);
            g_sandboxes.workspaces.erase(id);
          }
          rmtree(ws);
          conn.respond(204, "No Content", "");
        } else {
          conn.respond(405, "Method Not Allowed", "{\"error\":\"method\"}");
        }
      } else {
        std::string sub = rest.substr(slash + 1);
        if (sub.rfind("workspace/", 0) == 0) {
          if (!handle_workspace_io(conn, req, plan, ws,
                                   sub.substr(strlen("workspace/"))))
            return;
        } else if (req.method == "POST" && sub == "execute") {
          handle_execute(conn, req, ws);
        } else {
          conn.respond(404, "Not Found", "{\"error\":\"no route\"}");
        }
      }
    } else {
      conn.respond(404, "Not Found", "{\"error\":\"no route\"}");
    }
    if (!req.keep_alive) return;
  }
}

static int make_listen_socket() {
  if (!g_cfg.listen_unix.empty()) {
    int fd = socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd < 0) { perror("socket"); return -1; }
    struct sockaddr_un addr;
    memset(&addr, 0, sizeof addr);
    addr.sun_family = AF_UNIX;
    strncpy(addr.sun_path, g_cfg.listen_unix.c_str(), sizeof addr.sun_path - 1);
    unlink(g_cfg.listen_unix.c_str());
    if (bind(fd, (struct sockaddr*)&addr, sizeof addr) != 0) {
      perror("bind");
      return -1;
    }
    if (listen(fd, 128) != 0) { perror("listen"); return -1; }
    return fd;
  }
  // TCP host:port
  std::string host = "0.0.0.0";
  int port = 8000;
  size_t colon = g_cfg.listen_addr.rfind(':');
  if (colon != std::string::npos) {
    host = g_cfg.listen_addr.substr(0, colon);
    port = atoi(g_cfg.listen_addr.c_str() + colon + 1);
  }
  int fd = socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) { perror("socket"); return -1; }
  int one = 1;
  setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof one);
  struct sockaddr_in addr;
  memset(&addr, 0, sizeof addr);
  addr.sin_family = AF_INET;
  addr.sin_port = htons((uint16_t)port);
  if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1)
    addr.sin_addr.s_addr = INADDR_ANY;
  if (bind(fd, (struct sockaddr*)&addr, sizeof addr) != 0) {
    perror("bind");
    return -1;
  }
  if (listen(fd, 128) != 0) { perror("listen"); return -1; }
  return fd;
}

// Graceful shutdown: broadcast SIGTERM to the engine's process group
// (zygote, warm children, GPU daemon all share it -- the engine is
// spawned with start_new_session), give the daemon a moment to flush
// (it may be writing profiler output), then exit. Everything here is
// async-signal-safe.
static volatile sig_atomic_t g_shutting_down = 0;
static volatile pid_t g_daemon_pid_for_shutdown = 0;

static void on_terminate(int) {
  g_shutting_down = 1;
  signal(SIGTERM, SIG_IGN);  // the pg broadcast below includes ourselves
  kill(0, SIGTERM);
  pid_t dpid = g_daemon_pid_for_shutdown;
  if (dpid > 0) {
    for (int i = 0; i < 30; i++) {  // <= 3 s for a clean daemon exit
      if (waitpid(dpid, nullptr, WNOHANG) != 0) break;
      struct timespec ts = {0, 100 * 1000 * 1000};
      nanosleep(&ts, nullptr);
    }
  }
  _exit(0);
}

int main(int, char**) {
  signal(SIGPIPE, SIG_IGN);
  signal(SIGTERM, on_terminate);
  signal(SIGINT, on_terminate);
  // safety net against engine leaks: if the control-plane process that
  // spawned us dies without calling aclose (SIGKILL, crash), die too
  prctl(PR_SET_PDEATHSIG, SIGTERM);
  if (getppid() == 1) raise(SIGTERM);  // parent already gone

  g_cfg.listen_addr = env_or("APP_LISTEN_ADDR", "0.0.0.0:8000");
  g_cfg.listen_unix = env_or("APP_LISTEN_UNIX", "");
  g_cfg.workspace = env_or("APP_WORKSPACE", "/workspace");
  g_cfg.python = env_or("APP_PYTHON", "python3");
  g_cfg.zygote = env_or("APP_ZYGOTE", "1") != "0";
  g_cfg.scan_recursive = env_or("APP_SCAN_RECURSIVE", "0") == "1";
  g_cfg.max_body_bytes =
      (size_t)strtoull(env_or("APP_MAX_BODY_BYTES", "1073741824").c_str(),
                       nullptr, 10);
  g_cfg.max_connections =
      atoi(env_or("APP_MAX_CONNECTIONS", "512").c_str());
  if (g_cfg.max_connections < 1) g_cfg.max_connections = 1;

  // default runtime dir: the directory containing this binary
  std::string self_dir;
  {
    char buf[4096];
    ssize_t n = readlink("/proc/self/exe", buf, sizeof buf - 1);
    if (n > 0) {
      buf[n] = 0;
      std::string p(buf);
      size_t slash = p.rfind('/');
      if (slash != std::string::npos) self_dir = p.substr(0, slash);
    }
  }
  g_cfg.runtime_dir = env_or("APP_RUNTIME_DIR", self_dir);

  mkdirs(g_cfg.workspace);
  g_sessions_root = env_or("APP_SESSIONS_DIR", "/tmp/sandboxes");
  mkdirs(g_sessions_root);

  // GPU compute daemon: ONE HIP context per engine (per-process context
  // creation serializes in the driver); sandbox children RPC to it via
  // hipnp's remote backend. If there is no GPU the daemon exits(3), the
  // socket never appears, and children fall back (ops/hipnp.py).
  std::string ops_dir = env_or("APP_OPS_DIR", "");
  if (env_or("APP_HIP_DAEMON", "1") != "0" && !ops_dir.empty()) {
    // spawn + monitor/respawn. The spawner may run from the monitor
    // thread: argv is fully built pre-fork and the child only exec/exits
    // (a forked child of a multithreaded process deadlocks if it
    // allocates while another thread held the heap lock).
    static std::string gpu_sock = g_sessions_root + "/gpu.sock";
    static std::string hipd_script = ops_dir + "/hipd.py";
    auto spawn_daemon = []() -> pid_t {
      const char* argv[] = {g_cfg.python.c_str(), "-u", hipd_script.c_str(),
                            "--socket", gpu_sock.c_str(), nullptr};
      pid_t pid = fork();
      if (pid == 0) {
        execvp(argv[0], const_cast<char* const*>(argv));
        _exit(127);
      }
      return pid;
    };
    pid_t daemon_pid = spawn_daemon();
    if (daemon_pid > 0) {
      setenv("APP_GPU_SERVICE", gpu_sock.c_str(), 1);
      std::thread([daemon_pid, spawn_daemon]() {
        int respawns = 0;
        pid_t pid = daemon_pid;
        while (true) {
          int status = 0;
          if (waitpid(pid, &status, 0) != pid) return;
          if (g_shutting_down) return;
          // exit(3) = no GPU visible: do not respawn; remove the socket
          // and the env hint so sandboxes use their fallback
          if (WIFEXITED(status) && WEXITSTATUS(status) == 3) {
            unlink(gpu_sock.c_str());
            unsetenv("APP_GPU_SERVICE");
            return;
          }
          if (++respawns > 5) {
            fprintf(stderr, "executor-server: gpu daemon kept dying\n");
            unlink(gpu_sock.c_str());
            return;
          }
          fprintf(stderr, "executor-server: gpu daemon died, respawning\n");
          sleep(1);
          pid = spawn_daemon();
          if (pid <= 0) return;
        }
      }).detach();
    }
  }

  if (g_cfg.zygote) {
    Zygote* zyg = new Zygote();
    g_zygote.store(zyg);
    if (!zyg->start()) {
      fprintf(stderr, "executor-server: zygote failed to start; cold mode\n");
      g_cfg.zygote = false;
    } else {
      // respawn a dead zygote (cold fork/exec covers the gap meanwhile)
      std::thread([]() {
        int respawns = 0;
        while (true) {
          Zygote* cur = g_zygote.load();
          while (cur->alive()) sleep(1);
          int status = 0;
          waitpid(cur->pid(), &status, 0);  // reap (no zombie)
          if (++respawns > 5) {
            fprintf(stderr, "executor-server: zygote kept dying; cold mode\n");
            return;
          }
          fprintf(stderr, "executor-server: zygote died, respawning\n");
          Zygote* fresh = new Zygote();
          if (fresh->start()) {
            g_zygote.store(fresh);  // old object intentionally leaked
                                    // (request threads may still hold it)
          } else {
            return;
          }
        }
      }).detach();
    }
  }

  int listen_fd = make_listen_socket();
  if (listen_fd < 0) return 1;
  fprintf(stderr, "executor-server: listening (%s), workspace=%s zygote=%d\n",
          g_cfg.listen_unix.empty() ? g_cfg.listen_addr.c_str()
                                    : g_cfg.listen_unix.c_str(),
          g_cfg.workspace.c_str(), (int)g_cfg.zygote);

  while (true) {
    {
      std::unique_lock<std::mutex> lk(g_conn_gate_mu);
      g_conn_gate_cv.wait(lk, [] {
        return g_active_conns.load() < g_cfg.max_connections;
      });
    }
    int fd = accept(listen_fd, nullptr, nullptr);
    if (fd < 0) {
      if (errno == EINTR) continue;
      perror("accept");
      break;
    }
    g_active_conns.fetch_add(1);
    std::thread(handle_conn_counted, fd).detach();
  }
  return 0;
}
