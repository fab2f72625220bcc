// ckd — PID 1 of every clawker-amd sandbox.
//
// The reference runs `clawkerd` as container PID 1 (clawkerd/ package:
// mTLS gRPC Session stream, spawn with kernel-side privilege drop,
// signal forwarding, two-phase zombie reaping — SURVEY.md §2.5). This
// rebuild keeps those duties but speaks a framed-JSON protocol over a
// Unix control socket in the sandbox's shared runtime dir (single-node
// appliance: filesystem permissions replace the mTLS+OAuth stack; the
// control socket is chmod 0600 inside the 0711 rundir).
//
// Protocol (length-prefixed JSON frames; see clawker_amd/engine/wire.py):
//   -> hello                         <- {t:hello, initialized, cmd_running}
//   -> {t:agent_ready, cmd?:[..]}    spawn the agent CMD (once)
//   -> {t:agent_initialized}         persist one-time init marker
//   -> {t:exec, id, stages:[{argv,uid,gid,cwd}], stdin?, env?}
//         <- {t:started,id} {t:out,id,stream,data(b64)}
//            {t:stage_exit,id,idx,code} {t:done,id,code}
//   -> {t:attach}                    subscribe to console stream
//         <- {t:console, data(b64)}  (agent CMD output)
//   -> {t:stdin, data(b64)} {t:close_stdin} {t:resize,rows,cols}
//   -> {t:signal, sig}               signal the agent process group
//   -> {t:status}                    <- {t:status, ...}
//   <- {t:agent_exit, code}          broadcast when the agent CMD exits
//
// ckd exits with the agent's code (bash convention 128+sig) after a
// bounded orphan drain; the pid-namespace teardown then kills stragglers.

#include <errno.h>
#include <fcntl.h>
#include <grp.h>
#include <poll.h>
#include <pwd.h>
#include <signal.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/ioctl.h>
#include <sys/stat.h>
#include <sys/types.h>
#include <sys/wait.h>
#include <termios.h>
#include <time.h>
#include <unistd.h>

#include <map>
#include <string>
#include <vector>

#include "../common/minijson.hpp"
#include "../common/util.hpp"

using ck::die;
using ck::warn;

namespace {

// Paths come from the spec's "paths" object so ckd works under both
// isolation backends: "ns" (rundir bind-mounted at /run/clawker inside the
// sandbox) and "proc" (no mount namespace available — e.g. a restricted
// container host — so rundir is the per-sandbox host directory itself).
std::string g_rundir = "/run/clawker";
std::string g_marker = "/var/lib/clawker/initialized";

std::string ctl_sock() { return g_rundir + "/ctl.sock"; }
std::string console_log() { return g_rundir + "/console.log"; }
std::string ready_file() { return g_rundir + "/ready"; }

// Load-bearing audit events, one JSON per line in the shared rundir
// (reference: clawkerd session_started/ended + shell_command_started/done
// — the operator triage contract, clawkerd/CLAUDE.md).
void audit(const char* event, std::initializer_list<std::pair<const char*, mj::Value>> kv = {}) {
  mj::Value rec;
  struct timespec ts;
  clock_gettime(CLOCK_REALTIME, &ts);
  rec.set("ts", (double)ts.tv_sec + ts.tv_nsec / 1e9).set("event", event);
  for (auto& p : kv) rec.set(p.first, p.second);
  int fd = open((g_rundir + "/audit.jsonl").c_str(),
                O_WRONLY | O_CREAT | O_APPEND | O_CLOEXEC, 0600);
  if (fd < 0) return;
  std::string line = rec.dump() + "\n";
  ck::write_exact(fd, line.data(), line.size());
  close(fd);
}

mj::Value g_spec;
int g_selfpipe[2];    // SIGCHLD -> poll wakeup

void sigchld(int) {
  char b = 'c';
  ssize_t r = write(g_selfpipe[1], &b, 1);
  (void)r;
}

// ------------------------------------------------------------- users -------

struct Cred { uid_t uid = 0; gid_t gid = 0; std::string home = "/"; std::string name = "root"; };

bool resolve_user(const std::string& spec, Cred& c) {
  // "" or "root" => root; "uid:gid"; or /etc/passwd name. Returns false
  // when a named user cannot be resolved — callers must FAIL the spawn,
  // never degrade to root (reference: clawkerd spawn_unix.go fails on
  // unknown user; a root fallback would hand the workload full host
  // capabilities since there is no user namespace).
  c = Cred{};
  if (spec.empty() || spec == "root") return true;
  size_t colon = spec.find(':');
  if (colon != std::string::npos && spec.find_first_not_of("0123456789:") == std::string::npos) {
    c.uid = atoi(spec.substr(0, colon).c_str());
    c.gid = atoi(spec.substr(colon + 1).c_str());
    c.name = spec;
    return true;
  }
  if (struct passwd* pw = getpwnam(spec.c_str())) {
    c.uid = pw->pw_uid;
    c.gid = pw->pw_gid;
    c.home = pw->pw_dir && *pw->pw_dir ? pw->pw_dir : "/";
    c.name = spec;
    return true;
  }
  return false;
}

void drop_to(const Cred& c) {
  // kernel-side drop ordering: groups -> gid -> uid (reference:
  // clawkerd/spawn_unix.go SysProcAttr.Credential semantics)
  if (c.uid == 0 && c.gid == 0) return;
  if (initgroups(c.name.c_str(), c.gid) != 0) {
    if (setgroups(0, nullptr) != 0) warn("setgroups");
  }
  if (setgid(c.gid) != 0) die("setgid %d", c.gid);
  if (setuid(c.uid) != 0) die("setuid %d", c.uid);
}

std::vector<std::string> build_env(const mj::Value& env_obj, const Cred& c,
                                   bool inherit_spec_env = false) {
  std::vector<std::string> env;
  bool has_path = false, has_home = false, has_user = false, has_term = false;
  std::map<std::string, std::string> merged;
  if (inherit_spec_env) {
    // execs run in the sandbox's environment (proxies, CLAWKER_*, ...)
    for (const auto& kv : g_spec["env"].as_obj())
      merged[kv.first] = kv.second.as_str();
  }
  for (const auto& kv : env_obj.as_obj()) merged[kv.first] = kv.second.as_str();
  for (const auto& kv : merged) {
    if (kv.first == "PATH") has_path = true;
    if (kv.first == "HOME") has_home = true;
    if (kv.first == "USER") has_user = true;
    if (kv.first == "TERM") has_term = true;
    env.push_back(kv.first + "=" + kv.second);
  }
  if (!has_path) env.push_back("PATH=/usr/local/sbin:/usr/local/bin:/usr/sbin:/usr/bin:/sbin:/bin");
  if (!has_home) env.push_back("HOME=" + c.home);
  if (!has_user) { env.push_back("USER=" + c.name); env.push_back("LOGNAME=" + c.name); }
  if (!has_term) env.push_back("TERM=xterm-256color");
  return env;
}

std::vector<char*> to_argv(const std::vector<std::string>& v) {
  std::vector<char*> out;
  for (const auto& s : v) out.push_back(const_cast<char*>(s.c_str()));
  out.push_back(nullptr);
  return out;
}

// ------------------------------------------------------------- clients -----

struct Client {
  int fd = -1;
  std::string buf;          // partial inbound frame bytes
  std::string out;          // pending outbound bytes (nonblocking writes)
  bool attached = false;    // subscribed to console stream
  bool dead = false;
};

std::vector<Client> g_clients;

constexpr size_t kMaxClientBacklog = 8 * 1024 * 1024;

// Non-blocking framed send with per-client backlog: a stalled control
// client must never block PID 1's poll loop (reference: pubsub bounded
// buffers + drop; here the whole frame is buffered or the client is
// marked dead when its backlog cap is exceeded).
void send_to_client(Client& c, const mj::Value& v) {
  if (c.dead) return;
  std::string body = v.dump();
  uint8_t hdr[4] = {
      static_cast<uint8_t>(body.size() >> 24), static_cast<uint8_t>(body.size() >> 16),
      static_cast<uint8_t>(body.size() >> 8), static_cast<uint8_t>(body.size())};
  std::string frame(reinterpret_cast<char*>(hdr), 4);
  frame += body;
  if (c.out.empty()) {
    ssize_t n = send(c.fd, frame.data(), frame.size(), MSG_DONTWAIT | MSG_NOSIGNAL);
    if (n == static_cast<ssize_t>(frame.size())) return;
    if (n < 0) {
      if (errno != EAGAIN && errno != EWOULDBLOCK) {
        c.dead = true;
        return;
      }
      n = 0;
    }
    c.out.append(frame.data() + n, frame.size() - n);
  } else {
    c.out += frame;
  }
  if (c.out.size() > kMaxClientBacklog) c.dead = true;
}

void flush_client(Client& c) {
  if (c.dead || c.out.empty()) return;
  ssize_t n = send(c.fd, c.out.data(), c.out.size(), MSG_DONTWAIT | MSG_NOSIGNAL);
  if (n > 0) c.out.erase(0, n);
  else if (n < 0 && errno != EAGAIN && errno != EWOULDBLOCK) c.dead = true;
}

Client* find_client(int fd) {
  for (auto& c : g_clients)
    if (c.fd == fd && !c.dead) return &c;
  return nullptr;
}

void send_to_fd(int fd, const mj::Value& v) {
  if (Client* c = find_client(fd)) send_to_client(*c, v);
}

void broadcast(const mj::Value& v, bool attached_only = false) {
  for (auto& c : g_clients) {
    if (attached_only && !c.attached) continue;
    send_to_client(c, v);
  }
}

// --------------------------------------------------------------- agent -----

struct Agent {
  pid_t pid = -1;
  int io = -1;              // pty master, or stdout pipe read end
  int err = -1;             // stderr pipe (non-tty only)
  int in = -1;              // stdin write end (non-tty; == io for tty)
  bool tty = false;
  bool running = false;
  int exit_code = -1;
  bool spawned = false;
};

Agent g_agent;
int g_console_log = -1;

// ------------------------------------------------------------- services ----
// Auxiliary in-sandbox daemons from spec "services" (e.g. the ckgw egress
// gateway shims). Root children of PID 1, restarted up to 3 times.

struct Service {
  std::string name;
  std::vector<std::string> argv;
  pid_t pid = -1;
  int restarts = 0;
};

std::vector<Service> g_services;

void spawn_service(Service& svc) {
  pid_t pid = fork();
  if (pid < 0) {
    warn("service fork %s", svc.name.c_str());
    return;
  }
  if (pid == 0) {
    setsid();
    auto eargv = to_argv(svc.argv);
    Cred root_cred;
    auto env = build_env(mj::Value(), root_cred);
    auto eenv = to_argv(env);
    execvpe(eargv[0], eargv.data(), eenv.data());
    fprintf(stderr, "ckd: service %s exec: %s\n", svc.name.c_str(), strerror(errno));
    _exit(127);
  }
  svc.pid = pid;
}

void start_services() {
  for (const auto& sv : g_spec["services"].as_arr()) {
    Service svc;
    svc.name = sv["name"].as_str();
    for (const auto& a : sv["argv"].as_arr()) svc.argv.push_back(a.as_str());
    if (svc.argv.empty()) continue;
    g_services.push_back(std::move(svc));
  }
  for (auto& svc : g_services) spawn_service(svc);
}

void spawn_agent(const mj::Value& cmd_override) {
  if (g_agent.spawned) return;
  std::vector<std::string> argv;
  const mj::Value& cmd = cmd_override.is_arr() && !cmd_override.as_arr().empty()
                             ? cmd_override : g_spec["cmd"];
  for (const auto& a : cmd.as_arr()) argv.push_back(a.as_str());
  if (argv.empty()) argv = {"/bin/sh"};
  Cred cred;
  if (!resolve_user(g_spec["user"].as_str(), cred))
    die("user %s not found in sandbox /etc/passwd; refusing to run as root",
        g_spec["user"].as_str().c_str());
  std::string workdir = g_spec["workdir"].as_str();
  if (workdir.empty()) workdir = "/";
  bool tty = g_spec["tty"].as_bool(false);
  auto env = build_env(g_spec["env"], cred);

  int master = -1, slave = -1;
  int outp[2] = {-1, -1}, errp[2] = {-1, -1}, inp[2] = {-1, -1};
  if (tty) {
    master = posix_openpt(O_RDWR | O_NOCTTY | O_CLOEXEC);
    if (master < 0) die("openpt");
    grantpt(master);
    unlockpt(master);
    slave = open(ptsname(master), O_RDWR | O_NOCTTY);
    if (slave < 0) die("pts open");
  } else {
    if (pipe2(outp, O_CLOEXEC) || pipe2(errp, O_CLOEXEC) || pipe2(inp, O_CLOEXEC))
      die("pipe");
  }

  pid_t pid = fork();
  if (pid < 0) die("fork");
  if (pid == 0) {
    setsid();
    if (tty) {
      ioctl(slave, TIOCSCTTY, 0);
      dup2(slave, 0); dup2(slave, 1); dup2(slave, 2);
      if (slave > 2) close(slave);
      close(master);
    } else {
      dup2(inp[0], 0); dup2(outp[1], 1); dup2(errp[1], 2);
    }
    if (chdir(workdir.c_str()) != 0) {
      // fall back to / rather than failing the spawn
      if (chdir("/") != 0) _exit(111);
    }
    drop_to(cred);
    auto eargv = to_argv(argv);
    auto eenv = to_argv(env);
    execvpe(eargv[0], eargv.data(), eenv.data());
    fprintf(stderr, "ckd: exec %s: %s\n", eargv[0], strerror(errno));
    _exit(127);
  }
  if (tty) {
    close(slave);
    fcntl(master, F_SETFL, O_NONBLOCK);
    g_agent.io = master;
    g_agent.in = master;
  } else {
    close(outp[1]); close(errp[1]); close(inp[0]);
    fcntl(outp[0], F_SETFL, O_NONBLOCK);
    fcntl(errp[0], F_SETFL, O_NONBLOCK);
    g_agent.io = outp[0];
    g_agent.err = errp[0];
    g_agent.in = inp[1];
  }
  g_agent.pid = pid;
  g_agent.tty = tty;
  g_agent.running = true;
  g_agent.spawned = true;
  audit("agent_spawned", {{"pid", mj::Value((int64_t)pid)},
                          {"user", mj::Value(cred.name)},
                          {"argv0", mj::Value(argv[0])}});
}

// ---------------------------------------------------------------- exec -----

struct ExecJob {
  std::string id;
  int client_fd;
  std::vector<pid_t> pids;
  std::vector<int> codes;     // -1 = running
  int out = -1, err = -1;     // last stage stdout, combined stderr
  int in = -1;                // exec stdin (tty: == master)
  bool tty = false;
  int pending_io = 0;
};

std::map<std::string, ExecJob> g_execs;

void start_exec(Client& cl, const mj::Value& req) {
  ExecJob job;
  job.id = req["id"].as_str();
  job.client_fd = cl.fd;
  const auto& stages = req["stages"].as_arr();
  if (stages.empty()) {
    mj::Value e; e.set("t", "error").set("id", job.id).set("msg", "no stages");
    send_to_client(cl, e);
    return;
  }
  std::string init_stdin;
  if (req.has("stdin")) {
    auto raw = ck::b64_decode(req["stdin"].as_str());
    init_stdin.assign(raw.begin(), raw.end());
  }

  if (req["tty"].as_bool(false)) {
    // interactive exec: single stage on its own pty
    const mj::Value& st = stages[0];
    std::vector<std::string> argv;
    for (const auto& a : st["argv"].as_arr()) argv.push_back(a.as_str());
    Cred cred;
    if (st.has("user") && !st["user"].as_str().empty() &&
        !resolve_user(st["user"].as_str(), cred)) {
      mj::Value e;
      e.set("t", "error").set("id", job.id)
       .set("msg", "user not found: " + st["user"].as_str());
      send_to_client(cl, e);
      return;
    }
    std::string cwd = st["cwd"].as_str();
    auto env = build_env(req["env"], cred, /*inherit_spec_env=*/true);
    int master = posix_openpt(O_RDWR | O_NOCTTY | O_CLOEXEC);
    if (master < 0) { warn("exec openpt"); return; }
    grantpt(master);
    unlockpt(master);
    int slave = open(ptsname(master), O_RDWR | O_NOCTTY);
    pid_t pid = fork();
    if (pid == 0) {
      setsid();
      ioctl(slave, TIOCSCTTY, 0);
      dup2(slave, 0); dup2(slave, 1); dup2(slave, 2);
      if (slave > 2) close(slave);
      close(master);
      if (!cwd.empty() && chdir(cwd.c_str()) != 0) _exit(126);
      drop_to(cred);
      auto eargv = to_argv(argv);
      auto eenv = to_argv(env);
      execvpe(eargv[0], eargv.data(), eenv.data());
      _exit(127);
    }
    close(slave);
    fcntl(master, F_SETFL, O_NONBLOCK);
    job.tty = true;
    job.pids.push_back(pid);
    job.codes.push_back(-1);
    job.out = master;
    job.in = master;
    job.pending_io = 1;
    mj::Value started;
    started.set("t", "started").set("id", job.id).set("tty", true);
    send_to_client(cl, started);
    audit("shell_command_started", {{"id", mj::Value(job.id)},
                                    {"tty", mj::Value(true)}});
    g_execs[job.id] = job;
    return;
  }

  int in_fd = -1;   // read end feeding next stage's stdin
  if (!init_stdin.empty()) {
    int p[2];
    if (pipe2(p, O_CLOEXEC)) die("pipe");
    // write initial stdin from a detached writer child to avoid blocking
    pid_t w = fork();
    if (w == 0) {
      close(p[0]);
      ck::write_exact(p[1], init_stdin.data(), init_stdin.size());
      _exit(0);
    }
    close(p[1]);
    in_fd = p[0];
  }

  int errp[2];
  if (pipe2(errp, O_CLOEXEC)) die("pipe");

  for (size_t i = 0; i < stages.size(); i++) {
    const mj::Value& st = stages[i];
    bool last = i + 1 == stages.size();
    int outp[2] = {-1, -1};
    if (pipe2(outp, O_CLOEXEC)) die("pipe");

    std::vector<std::string> argv;
    for (const auto& a : st["argv"].as_arr()) argv.push_back(a.as_str());
    Cred cred;
    cred.uid = (uid_t)st["uid"].as_int(0);
    cred.gid = (gid_t)st["gid"].as_int(0);
    if (st.has("user") && !st["user"].as_str().empty() &&
        !resolve_user(st["user"].as_str(), cred)) {
      mj::Value e;
      e.set("t", "error").set("id", job.id)
       .set("msg", "user not found: " + st["user"].as_str());
      send_to_client(cl, e);
      // reap already-forked earlier stages; the pipeline is aborted
      for (pid_t p : job.pids) kill(p, SIGKILL);
      return;
    }
    std::string cwd = st["cwd"].as_str();
    auto env = build_env(req["env"], cred, /*inherit_spec_env=*/true);

    pid_t pid = fork();
    if (pid < 0) die("fork");
    if (pid == 0) {
      if (in_fd >= 0) dup2(in_fd, 0);
      else { int nul = open("/dev/null", O_RDONLY); dup2(nul, 0); }
      dup2(outp[1], 1);
      dup2(errp[1], 2);
      if (!cwd.empty() && chdir(cwd.c_str()) != 0) _exit(126);
      if (cred.uid || cred.gid) {
        if (setgroups(0, nullptr) != 0) { /* best effort */ }
        if (setgid(cred.gid) != 0 || setuid(cred.uid) != 0) _exit(126);
      }
      auto eargv = to_argv(argv);
      auto eenv = to_argv(env);
      execvpe(eargv[0], eargv.data(), eenv.data());
      fprintf(stderr, "exec %s: %s\n", eargv[0], strerror(errno));
      _exit(127);
    }
    job.pids.push_back(pid);
    job.codes.push_back(-1);
    if (in_fd >= 0) close(in_fd);
    close(outp[1]);
    if (last) {
      fcntl(outp[0], F_SETFL, O_NONBLOCK);
      job.out = outp[0];
    } else {
      in_fd = outp[0];   // next stage reads previous stdout
    }
  }
  close(errp[1]);
  fcntl(errp[0], F_SETFL, O_NONBLOCK);
  job.err = errp[0];
  job.pending_io = 2;

  mj::Value started;
  started.set("t", "started").set("id", job.id);
  send_to_client(cl, started);
  audit("shell_command_started", {{"id", mj::Value(job.id)},
                                  {"stages", mj::Value((int64_t)job.pids.size())}});
  g_execs[job.id] = job;
}

// drain an exec output fd; returns false when EOF
bool pump_exec_fd(ExecJob& job, int which) {
  int fd = which == 1 ? job.out : job.err;
  if (fd < 0) return false;
  char buf[65536];
  while (true) {
    ssize_t n = read(fd, buf, sizeof buf);
    if (n > 0) {
      mj::Value out;
      out.set("t", "out").set("id", job.id).set("stream", (int64_t)which)
         .set("data", ck::b64_encode(reinterpret_cast<uint8_t*>(buf), n));
      send_to_fd(job.client_fd, out);
      continue;
    }
    if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) return true;
    // EOF or error
    close(fd);
    if (job.in == fd) job.in = -1;   // tty: master serves both directions
    if (which == 1) job.out = -1; else job.err = -1;
    job.pending_io--;
    return false;
  }
}

void maybe_finish_exec(ExecJob& job) {
  if (job.pending_io > 0) return;
  for (int c : job.codes)
    if (c < 0) return;
  mj::Value done;
  done.set("t", "done").set("id", job.id).set("code", (int64_t)job.codes.back());
  send_to_fd(job.client_fd, done);
  audit("shell_command_done", {{"id", mj::Value(job.id)},
                               {"code", mj::Value((int64_t)job.codes.back())}});
  g_execs.erase(job.id);
}

// --------------------------------------------------------------- reaper ----

void reap() {
  // two-phase: identify the agent CMD specially, then drain any orphans
  int wstatus;
  pid_t pid;
  while ((pid = waitpid(-1, &wstatus, WNOHANG)) > 0) {
    int code = WIFEXITED(wstatus) ? WEXITSTATUS(wstatus)
             : WIFSIGNALED(wstatus) ? 128 + WTERMSIG(wstatus) : 1;
    if (pid == g_agent.pid) {
      g_agent.running = false;
      g_agent.exit_code = code;
      continue;
    }
    bool was_service = false;
    for (auto& svc : g_services) {
      if (svc.pid == pid) {
        was_service = true;
        svc.pid = -1;
        if (g_agent.exit_code < 0 && svc.restarts < 3) {
          svc.restarts++;
          warn("service %s died (code %d); restart %d/3",
               svc.name.c_str(), code, svc.restarts);
          spawn_service(svc);
        }
      }
    }
    if (was_service) continue;
    for (auto& kv : g_execs) {
      ExecJob& job = kv.second;
      for (size_t i = 0; i < job.pids.size(); i++) {
        if (job.pids[i] == pid && job.codes[i] < 0) {
          job.codes[i] = code;
          mj::Value se;
          se.set("t", "stage_exit").set("id", job.id)
            .set("idx", (int64_t)i).set("code", (int64_t)code);
          send_to_fd(job.client_fd, se);
        }
      }
    }
  }
}

// -------------------------------------------------------------- console ----

// console.log rotation: an agent printing unbounded output must not
// fill the host disk (reference: docker/lumberjack log rotation).
// At 64 MiB the current file becomes console.log.1 (replacing any
// previous rotation) and a fresh file is opened.
constexpr int64_t kConsoleMax = 64 * 1024 * 1024;
int64_t g_console_written = -1;

void maybe_rotate_console(ssize_t about_to_write) {
  if (g_console_log < 0) return;
  if (g_console_written < 0) {
    off_t cur = lseek(g_console_log, 0, SEEK_END);
    g_console_written = cur < 0 ? 0 : cur;
  }
  if (g_console_written + about_to_write < kConsoleMax) {
    g_console_written += about_to_write;
    return;
  }
  close(g_console_log);
  std::string path = console_log();
  rename(path.c_str(), (path + ".1").c_str());
  g_console_log = open(path.c_str(),
                       O_WRONLY | O_CREAT | O_APPEND | O_CLOEXEC, 0600);
  g_console_written = about_to_write;
}

void pump_console(int fd, int stream) {
  char buf[65536];
  while (true) {
    ssize_t n = read(fd, buf, sizeof buf);
    if (n > 0) {
      maybe_rotate_console(n);
      if (g_console_log >= 0) ck::write_exact(g_console_log, buf, n);
      mj::Value out;
      out.set("t", "console").set("stream", (int64_t)stream)
         .set("data", ck::b64_encode(reinterpret_cast<uint8_t*>(buf), n));
      broadcast(out, /*attached_only=*/true);
      continue;
    }
    if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) return;
    // EOF / EIO (pty closed)
    if (stream == 1 && g_agent.io == fd) { close(fd); g_agent.io = -1; }
    if (stream == 2 && g_agent.err == fd) { close(fd); g_agent.err = -1; }
    return;
  }
}

// ------------------------------------------------------------- handlers ----


void handle_frame(Client& cl, const mj::Value& req) {
  const std::string& t = req["t"].as_str();
  if (t == "hello") {
    mj::Value r;
    r.set("t", "hello")
     .set("initialized", ck::exists(g_marker))
     .set("cmd_running", g_agent.running)
     .set("pid", (int64_t)(g_agent.running ? g_agent.pid : -1))
     // connected control clients INCLUDING the asker: the CP watcher
     // uses clients==1 + !cmd_running to detect a gated sandbox whose
     // starting client died before driving the boot plans
     .set("clients", (int64_t)g_clients.size())
     // the sandbox's configured user: `exec` defaults to it (docker
     // semantics); internal plans send explicit stages instead
     .set("user", g_spec["user"].as_str())
     .set("version", "0.2.0");
    send_to_client(cl, r);
  } else if (t == "agent_ready") {
    // CP releases the user CMD (reference: boot_steps.go AgentReady)
    spawn_agent(req["cmd"]);
    mj::Value r;
    r.set("t", "ready_ack").set("pid", (int64_t)g_agent.pid);
    send_to_client(cl, r);
  } else if (t == "agent_initialized") {
    size_t slash = g_marker.rfind('/');
    if (slash != std::string::npos) ck::mkdirs(g_marker.substr(0, slash));
    ck::write_file(g_marker, std::to_string(time(nullptr)));
    mj::Value r; r.set("t", "ok");
    send_to_client(cl, r);
  } else if (t == "exec") {
    start_exec(cl, req);
  } else if (t == "attach") {
    cl.attached = true;
    mj::Value r; r.set("t", "attached").set("tty", g_agent.tty);
    send_to_client(cl, r);
  } else if (t == "exec_stdin") {
    auto it = g_execs.find(req["id"].as_str());
    if (it != g_execs.end() && it->second.in >= 0) {
      auto data = ck::b64_decode(req["data"].as_str());
      if (!data.empty())
        ck::write_exact(it->second.in, data.data(), data.size());
    }
  } else if (t == "exec_resize") {
    auto it = g_execs.find(req["id"].as_str());
    if (it != g_execs.end() && it->second.tty && it->second.out >= 0) {
      struct winsize ws{};
      ws.ws_row = (unsigned short)req["rows"].as_int(24);
      ws.ws_col = (unsigned short)req["cols"].as_int(80);
      ioctl(it->second.out, TIOCSWINSZ, &ws);
    }
  } else if (t == "exec_close_stdin") {
    auto it = g_execs.find(req["id"].as_str());
    if (it != g_execs.end() && it->second.in >= 0 && !it->second.tty) {
      close(it->second.in);
      it->second.in = -1;
    }
  } else if (t == "stdin") {
    auto data = ck::b64_decode(req["data"].as_str());
    if (g_agent.in >= 0 && !data.empty())
      ck::write_exact(g_agent.in, data.data(), data.size());
  } else if (t == "close_stdin") {
    if (!g_agent.tty && g_agent.in >= 0) { close(g_agent.in); g_agent.in = -1; }
  } else if (t == "resize") {
    if (g_agent.tty && g_agent.io >= 0) {
      struct winsize ws{};
      ws.ws_row = (unsigned short)req["rows"].as_int(24);
      ws.ws_col = (unsigned short)req["cols"].as_int(80);
      ioctl(g_agent.io, TIOCSWINSZ, &ws);
    }
  } else if (t == "signal") {
    int sig = (int)req["sig"].as_int(SIGTERM);
    if (g_agent.pid > 0) kill(-g_agent.pid, sig);   // whole process group
  } else if (t == "status") {
    mj::Value r;
    r.set("t", "status").set("cmd_running", g_agent.running)
     .set("exit_code", (int64_t)g_agent.exit_code)
     .set("initialized", ck::exists(g_marker));
    send_to_client(cl, r);
  } else {
    mj::Value r; r.set("t", "error").set("msg", "unknown command: " + t);
    send_to_client(cl, r);
  }
}

void drain_client(Client& cl, bool& drop) {
  char buf[65536];
  while (true) {
    ssize_t n = read(cl.fd, buf, sizeof buf);
    if (n > 0) {
      cl.buf.append(buf, n);
      // extract complete frames
      while (cl.buf.size() >= 4) {
        size_t len = (size_t(uint8_t(cl.buf[0])) << 24) | (size_t(uint8_t(cl.buf[1])) << 16) |
                     (size_t(uint8_t(cl.buf[2])) << 8) | uint8_t(cl.buf[3]);
        if (len > ck::kMaxFrame) { drop = true; return; }
        if (cl.buf.size() < 4 + len) break;
        std::string body = cl.buf.substr(4, len);
        cl.buf.erase(0, 4 + len);
        try {
          handle_frame(cl, mj::parse(body));
        } catch (const std::exception& e) {
          mj::Value r; r.set("t", "error").set("msg", e.what());
          send_to_client(cl, r);
        }
      }
      continue;
    }
    if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) return;
    drop = true;   // EOF or error
    return;
  }
}

}  // namespace

int main() {
  const char* spec_path = getenv("CKD_SPEC");
  if (!spec_path) die("CKD_SPEC not set");
  g_spec = mj::parse(ck::read_file(spec_path));
  const mj::Value& paths = g_spec["paths"];
  if (paths.has("rundir")) g_rundir = paths["rundir"].as_str();
  if (paths.has("marker")) g_marker = paths["marker"].as_str();

  if (pipe2(g_selfpipe, O_CLOEXEC | O_NONBLOCK) != 0) die("selfpipe");
  struct sigaction sa{};
  sa.sa_handler = sigchld;
  sa.sa_flags = SA_RESTART | SA_NOCLDSTOP;
  sigaction(SIGCHLD, &sa, nullptr);
  signal(SIGPIPE, SIG_IGN);

  // forward termination signals to the agent's process group (reference:
  // clawkerd signal forwarding excl. SIGCHLD/SIGURG)
  auto fwd = [](int sig) {
    if (g_agent.pid > 0) kill(-g_agent.pid, sig);
    else if (sig == SIGTERM || sig == SIGINT) _exit(128 + sig);
  };
  struct sigaction fsa{};
  fsa.sa_handler = fwd;
  fsa.sa_flags = SA_RESTART;
  for (int sig : {SIGTERM, SIGINT, SIGHUP, SIGQUIT, SIGUSR1, SIGUSR2})
    sigaction(sig, &fsa, nullptr);

  int listen_fd = ck::unix_listen(ctl_sock());
  if (listen_fd < 0) die("listen %s", ctl_sock().c_str());
  // root-only admin surface: the rundir is 0711 so the in-sandbox agent
  // user can traverse it for its own material — the control socket must
  // not be connectable by anyone but root (reference: clawkerd's STRICT
  // listener, listener.go:145 — here trust is the socket mode itself)
  chmod(ctl_sock().c_str(), 0600);
  fcntl(listen_fd, F_SETFL, O_NONBLOCK);

  g_console_log = open(console_log().c_str(),
                       O_WRONLY | O_CREAT | O_APPEND | O_CLOEXEC, 0600);

  // ready file: the HEALTHCHECK analog (reference: Dockerfile.base.tmpl:245)
  ck::write_file(ready_file(), "1");

  start_services();
  if (g_spec["autostart"].as_bool(false)) spawn_agent(mj::Value());

  bool exiting = false;
  int64_t exit_deadline_ms = 0;
  auto now_ms = [] {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return (int64_t)ts.tv_sec * 1000 + ts.tv_nsec / 1000000;
  };

  while (true) {
    std::vector<pollfd> pfds;
    pfds.push_back({g_selfpipe[0], POLLIN, 0});
    pfds.push_back({listen_fd, POLLIN, 0});
    size_t client_base = pfds.size();
    for (auto& c : g_clients) {
      short ev = POLLIN;
      if (!c.out.empty()) ev |= POLLOUT;
      pfds.push_back({c.fd, ev, 0});
    }
    int agent_io_idx = -1, agent_err_idx = -1;
    if (g_agent.io >= 0) { agent_io_idx = pfds.size(); pfds.push_back({g_agent.io, POLLIN, 0}); }
    if (g_agent.err >= 0) { agent_err_idx = pfds.size(); pfds.push_back({g_agent.err, POLLIN, 0}); }
    std::vector<std::pair<std::string, int>> exec_fds;   // (id, which)
    for (auto& kv : g_execs) {
      if (kv.second.out >= 0) { exec_fds.push_back({kv.first, 1}); pfds.push_back({kv.second.out, POLLIN, 0}); }
      if (kv.second.err >= 0) { exec_fds.push_back({kv.first, 2}); pfds.push_back({kv.second.err, POLLIN, 0}); }
    }

    int timeout = exiting ? 10 : 1000;
    int rc = poll(pfds.data(), pfds.size(), timeout);
    if (rc < 0 && errno != EINTR) die("poll");

    // SIGCHLD wakeup
    if (pfds[0].revents & POLLIN) {
      char b[64];
      while (read(g_selfpipe[0], b, sizeof b) > 0) {}
    }
    reap();

    // accept new control clients
    if (pfds[1].revents & POLLIN) {
      int cfd;
      while ((cfd = accept4(listen_fd, nullptr, nullptr, SOCK_NONBLOCK | SOCK_CLOEXEC)) >= 0) {
        Client nc;
        nc.fd = cfd;
        g_clients.push_back(std::move(nc));
        audit("session_started", {{"clients", mj::Value((int64_t)g_clients.size())}});
      }
    }

    // client traffic
    std::vector<int> drop_fds;
    for (size_t i = 0; i < g_clients.size(); i++) {
      if (client_base + i >= pfds.size()) break;
      auto& pe = pfds[client_base + i];
      if (pe.revents & POLLOUT) flush_client(g_clients[i]);
      if (pe.revents & (POLLIN | POLLHUP | POLLERR)) {
        bool drop = false;
        drain_client(g_clients[i], drop);
        if (drop) drop_fds.push_back(g_clients[i].fd);
      }
      if (g_clients[i].dead) drop_fds.push_back(g_clients[i].fd);
    }
    for (int fd : drop_fds) {
      for (size_t i = 0; i < g_clients.size(); i++) {
        if (g_clients[i].fd == fd) {
          close(fd);
          g_clients.erase(g_clients.begin() + i);
          break;
        }
      }
      // orphan exec jobs owned by a gone client keep running; their
      // output is discarded when send fails (SIGPIPE ignored)
    }

    // console output
    if (agent_io_idx >= 0 && (pfds[agent_io_idx].revents & (POLLIN | POLLHUP)))
      pump_console(g_agent.io, 1);
    if (agent_err_idx >= 0 && (pfds[agent_err_idx].revents & (POLLIN | POLLHUP)))
      pump_console(g_agent.err, 2);

    // exec output
    for (auto& [id, which] : exec_fds) {
      auto it = g_execs.find(id);
      if (it != g_execs.end()) pump_exec_fd(it->second, which);
    }
    // exec completion
    for (auto it = g_execs.begin(); it != g_execs.end();) {
      auto cur = it++;
      maybe_finish_exec(cur->second);
    }

    // agent exit: announce once, then drain orphans briefly and leave —
    // pidns teardown reclaims anything still running
    if (g_agent.spawned && !g_agent.running && !exiting) {
      // flush any trailing console output first
      if (g_agent.io >= 0) pump_console(g_agent.io, 1);
      if (g_agent.err >= 0) pump_console(g_agent.err, 2);
      mj::Value ev;
      ev.set("t", "agent_exit").set("code", (int64_t)g_agent.exit_code);
      broadcast(ev);
      audit("agent_exit", {{"code", mj::Value((int64_t)g_agent.exit_code)}});
      exiting = true;
      exit_deadline_ms = now_ms() + 500;
      // reclaim our service daemons right away so the orphan drain can go
      // dry; in the ns backend we are PID 1 of a private pidns, so a
      // namespace-wide sweep is safe and catches agent orphans too. In the
      // proc backend kill(-1) would hit the HOST — only signal known pids.
      bool ns_backend = g_spec["backend"].as_str() != "proc";
      for (auto& svc : g_services) {
        if (svc.pid > 0) kill(-svc.pid, SIGKILL);
      }
      if (ns_backend) {
        kill(-1, SIGKILL);
      } else if (g_agent.pid > 0) {
        kill(-g_agent.pid, SIGKILL);
      }
    }
    if (exiting) {
      // leave as soon as the orphan drain is dry (no remaining children)
      // or after the bounded grace; pidns teardown reclaims stragglers
      int st;
      pid_t r = waitpid(-1, &st, WNOHANG);
      bool no_children = (r < 0 && errno == ECHILD);
      if (no_children || now_ms() >= exit_deadline_ms)
        return g_agent.exit_code < 0 ? 0 : g_agent.exit_code;
    }
  }
}
