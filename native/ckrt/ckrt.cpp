// ckrt — the clawker-amd sandbox runtime.
//
// The MI355X-native replacement for the reference's Docker engine layer
// (pkg/whail + dockerd): there is no Docker on the target node, so ckrt
// creates sandboxes directly with Linux primitives, designed for a
// dedicated rootful 8xMI355X box:
//
//   * namespaces: mount + pid + uts + ipc (+ net when the egress firewall
//     is on — deny-by-default BY CONSTRUCTION: a fresh netns has no uplink)
//   * rootfs: overlayfs over the host filesystem ("hostfs base") plus
//     content-addressed image layers and a per-sandbox writable upper —
//     zero image pull, copy-on-write everywhere
//   * /dev: private tmpfs with only standard nodes plus the *allocated*
//     GPU devices (/dev/kfd + /dev/dri/renderD<N>) — this is the primary
//     GPU pinning mechanism. Kernel-side device rules are applied on top
//     as defense in depth against an in-sandbox root mknod'ing usable
//     nodes: the v1 devices controller where present, and on pure-v2
//     hosts a hand-emitted BPF_PROG_TYPE_CGROUP_DEVICE allow-list
//     (devbpf.hpp) attached to the sandbox cgroup
//   * cgroups: memory/pids limits (v1 or v2 trees)
//
// ckrt stays resident as a per-sandbox shim (containerd-shim analog): it
// reaps the sandbox's PID 1 (ckd), records the exit status, and tears down
// cgroups. Usage: ckrt run <spec.json>
//
// Reference behavior being reproduced (not ported): container create/start
// semantics of internal/cmd/container/shared/container_create.go and
// pkg/whail engine.go — see SURVEY.md §2.4/§2.2.

#include <fcntl.h>
#include <sched.h>
#include <signal.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/ioctl.h>
#include <sys/mount.h>
#include <sys/prctl.h>
#include <sys/stat.h>
#include <sys/syscall.h>
#include <sys/sysmacros.h>
#include <sys/types.h>
#include <sys/wait.h>
#include <net/if.h>
#include <sys/socket.h>
#include <time.h>
#include <unistd.h>

#include <string>
#include <vector>

#include "../common/minijson.hpp"
#include "../common/util.hpp"
#include "devbpf.hpp"

using ck::die;
using ck::warn;

// mount_setattr(2) ABI — defined locally: <linux/mount.h> conflicts with
// <sys/mount.h> on glibc 2.35, and glibc has no wrapper before 2.36.
#ifndef MOUNT_ATTR_RDONLY
#define MOUNT_ATTR_RDONLY 0x00000001
#define MOUNT_ATTR_NOSUID 0x00000002
#endif
#ifndef AT_RECURSIVE
#define AT_RECURSIVE 0x8000
#endif
#ifndef SYS_mount_setattr
#define SYS_mount_setattr 442
#endif
struct ck_mount_attr {
  uint64_t attr_set;
  uint64_t attr_clr;
  uint64_t propagation;
  uint64_t userns_fd;
};

namespace {

struct Spec {
  std::string name;
  std::string backend = "ns";  // ns: full namespaces+overlay; proc: plain
                               // child process (hosts without CAP_SYS_ADMIN
                               // or with user.max_user_namespaces=0)
  std::string spec_path;       // host path of the spec file (proc backend)
  std::string rundir;          // host dir bind-mounted at /run/clawker
  std::vector<std::string> lowerdirs;  // top-most first (overlay order)
  std::string upper, work, merged;
  std::string hostname = "clawker";
  bool netns = true;
  bool tty = false;
  mj::Value mounts;            // array of {src,dst,ro,type,opts}
  mj::Value devices;           // array of {path}
  int64_t mem_bytes = 0;
  int64_t pids_max = 0;
  bool device_allow_only = true;
  std::string restart = "no";  // no | on-failure
  int64_t restart_max = 3;
  mj::Value raw;               // full spec for ckd
};

Spec g_spec;
pid_t g_child = -1;

// CK_TRACE=1: microsecond timing of each boot phase to stderr (shim.log)
bool g_trace = false;
int64_t g_t0 = 0;
int64_t now_us() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return (int64_t)ts.tv_sec * 1000000 + ts.tv_nsec / 1000;
}
void trace(const char* what) {
  if (g_trace) fprintf(stderr, "ckrt-trace %s %+lld us\n", what,
                       (long long)(now_us() - g_t0));
}

Spec parse_spec(const std::string& path) {
  Spec s;
  s.raw = mj::parse(ck::read_file(path));
  const mj::Value& v = s.raw;
  s.name = v["name"].as_str();
  if (v.has("backend")) s.backend = v["backend"].as_str();
  s.rundir = v["rundir"].as_str();
  const mj::Value& rootfs = v["rootfs"];
  for (const auto& l : rootfs["lowerdirs"].as_arr()) s.lowerdirs.push_back(l.as_str());
  s.upper = rootfs["upper"].as_str();
  s.work = rootfs["work"].as_str();
  s.merged = rootfs["merged"].as_str();
  if (v.has("hostname")) s.hostname = v["hostname"].as_str();
  s.netns = v["netns"].as_bool(true);
  s.tty = v["tty"].as_bool(false);
  s.mounts = v["mounts"];
  s.devices = v["devices"];
  const mj::Value& cg = v["cgroup"];
  s.mem_bytes = cg["mem_bytes"].as_int(0);
  s.pids_max = cg["pids"].as_int(0);
  s.device_allow_only = cg["device_allow_only"].as_bool(true);
  if (v.has("restart")) {
    s.restart = v["restart"]["policy"].as_str();
    if (s.restart.empty()) s.restart = "no";
    s.restart_max = v["restart"]["max"].as_int(3);
  }
  if (s.name.empty() || s.rundir.empty())
    die("spec: name/rundir required");
  if (s.backend == "ns" && (s.merged.empty() || s.lowerdirs.empty()))
    die("spec: rootfs required for ns backend");
  return s;
}

// ------------------------------------------------------------- cgroups -----

struct Cgroups {
  bool v2 = false;
  std::vector<std::string> dirs;   // created dirs, for cleanup
};

Cgroups g_cg;

bool cg_write(const std::string& path, const std::string& val) {
  return ck::write_file(path, val);
}

std::string dev_rule(const std::string& path) {
  struct stat st;
  if (stat(path.c_str(), &st) != 0) return "";
  char type = S_ISBLK(st.st_mode) ? 'b' : 'c';
  char buf[64];
  snprintf(buf, sizeof buf, "%c %u:%u rwm", type, major(st.st_rdev), minor(st.st_rdev));
  return buf;
}

// pure-v2: the clawker subtree root. Prefer the top of the unified tree
// (rootful dedicated node); when that mkdir is refused (running inside a
// delegated subtree, e.g. a CI container) fall back to a child of OUR
// current cgroup — creation there is always permitted for the owner.
std::string cg2_base() {
  std::string base = "/sys/fs/cgroup/clawker";
  if (mkdir(base.c_str(), 0755) == 0 || errno == EEXIST) return base;
  std::string self = ck::read_file("/proc/self/cgroup");
  // "0::<path>\n"
  size_t pos = self.find("0::");
  if (pos == std::string::npos) return base;
  std::string path = self.substr(pos + 3);
  size_t nl = path.find('\n');
  if (nl != std::string::npos) path = path.substr(0, nl);
  std::string own = "/sys/fs/cgroup" + path + "/clawker";
  if (mkdir(own.c_str(), 0755) == 0 || errno == EEXIST) {
    // child cgroups only get controllers the parent delegates; enable
    // best-effort (fails under the no-internal-process rule when our own
    // process sits in the parent — device BPF still attaches fine)
    cg_write("/sys/fs/cgroup" + path + "/cgroup.subtree_control", "+memory +pids");
    return own;
  }
  return base;
}

void cgroups_setup(const Spec& s) {
  errno = 0;
  if (ck::exists("/sys/fs/cgroup/cgroup.controllers")) {
    // pure v2: memory/pids via the unified tree; device enforcement is a
    // hand-emitted BPF_PROG_TYPE_CGROUP_DEVICE allow-list attached to the
    // sandbox cgroup at cgroups_attach time (devbpf.hpp) — kernel-side
    // defense in depth on top of the private /dev construction.
    g_cg.v2 = true;
    std::string base = cg2_base();
    std::string dir = base + "/" + s.name;
    if (mkdir(dir.c_str(), 0755) != 0 && errno != EEXIST) {
      warn("cgroup2 mkdir %s (continuing without cgroup limits)", dir.c_str());
      return;
    }
    g_cg.dirs.push_back(dir);
    if (s.mem_bytes > 0)
      cg_write(dir + "/memory.max", std::to_string(s.mem_bytes));
    if (s.pids_max > 0)
      cg_write(dir + "/pids.max", std::to_string(s.pids_max));
    if (s.device_allow_only && !s.devices.as_arr().empty()) {
      std::vector<devbpf::Rule> rules = {
          // std nodes mirroring the v1 allow-list below
          {2, 1, 3}, {2, 1, 5}, {2, 1, 7}, {2, 1, 8}, {2, 1, 9},
          {2, 5, 0}, {2, 5, 2}, {2, 136, ~0u},
      };
      for (const auto& d : s.devices.as_arr()) {
        struct stat st;
        const std::string& p = d["path"].as_str();
        if (stat(p.c_str(), &st) != 0) continue;
        if (!(S_ISCHR(st.st_mode) || S_ISBLK(st.st_mode))) continue;
        rules.push_back({S_ISBLK(st.st_mode) ? 1u : 2u,
                         major(st.st_rdev), minor(st.st_rdev)});
      }
      std::string err;
      if (devbpf::attach(dir, rules, &err) != 0)
        warn("device_bpf_unavailable (%s); /dev construction remains the "
             "device enforcement", err.c_str());
    }
    return;
  }
  // v1 hybrid: per-controller hierarchies (only those with something
  // to enforce — cgroup mkdir+attach costs milliseconds under load)
  for (const char* ctl : {"memory", "pids", "devices"}) {
    if (strcmp(ctl, "memory") == 0 && s.mem_bytes <= 0) continue;
    if (strcmp(ctl, "pids") == 0 && s.pids_max <= 0) continue;
    if (strcmp(ctl, "devices") == 0 &&
        (!s.device_allow_only || s.devices.as_arr().empty())) continue;
    std::string root = std::string("/sys/fs/cgroup/") + ctl;
    if (!ck::exists(root)) continue;
    std::string base = root + "/clawker";
    mkdir(base.c_str(), 0755);
    std::string dir = base + "/" + s.name;
    if (mkdir(dir.c_str(), 0755) != 0 && errno != EEXIST) {
      warn("cgroup mkdir %s", dir.c_str());
      continue;
    }
    g_cg.dirs.push_back(dir);
    if (strcmp(ctl, "memory") == 0 && s.mem_bytes > 0) {
      cg_write(dir + "/memory.limit_in_bytes", std::to_string(s.mem_bytes));
    } else if (strcmp(ctl, "pids") == 0 && s.pids_max > 0) {
      cg_write(dir + "/pids.max", std::to_string(s.pids_max));
    } else if (strcmp(ctl, "devices") == 0 && s.device_allow_only &&
               !s.devices.as_arr().empty()) {
      // only sandboxes with passed-through GPUs need the device
      // allow-list (defense in depth for the /dev construction)
      // deny-all then allow-list: std nodes + the sandbox's allocated GPUs
      // (the amdgpu device-cgroup pinning of BASELINE.json)
      cg_write(dir + "/devices.deny", "a *:* rwm");
      for (const char* rule :
           {"c 1:3 rwm",  "c 1:5 rwm",  "c 1:7 rwm", "c 1:8 rwm", "c 1:9 rwm",
            "c 5:0 rwm",  "c 5:2 rwm",  "c 136:* rwm"}) {
        cg_write(dir + "/devices.allow", rule);
      }
      for (const auto& d : s.devices.as_arr()) {
        std::string rule = dev_rule(d["path"].as_str());
        if (!rule.empty()) cg_write(dir + "/devices.allow", rule);
      }
    }
  }
}

void cgroups_attach(pid_t pid) {
  for (const auto& dir : g_cg.dirs) {
    std::string procs = dir + "/cgroup.procs";
    if (!ck::exists(procs)) procs = dir + "/tasks";
    cg_write(procs, std::to_string(pid));
  }
}

void cgroups_cleanup() {
  for (const auto& dir : g_cg.dirs) rmdir(dir.c_str());
}

// --------------------------------------------------------- child (ns) ------

void mnt(const char* src, const char* dst, const char* type, unsigned long flags,
         const char* opts) {
  if (mount(src, dst, type, flags, opts) != 0)
    die("mount %s -> %s (%s)", src ? src : "-", dst, type ? type : "bind");
}

void bind_file(const std::string& src, const std::string& dst, bool ro) {
  struct stat st;
  if (stat(src.c_str(), &st) != 0) die("bind src missing: %s", src.c_str());
  if (S_ISDIR(st.st_mode)) {
    ck::mkdirs(dst);
  } else {
    // ensure parent + target file exist
    size_t slash = dst.rfind('/');
    if (slash != std::string::npos) ck::mkdirs(dst.substr(0, slash));
    int fd = open(dst.c_str(), O_WRONLY | O_CREAT | O_CLOEXEC, 0644);
    if (fd >= 0) close(fd);
  }
  mnt(src.c_str(), dst.c_str(), nullptr, MS_BIND | MS_REC, nullptr);
  if (ro) {
    // MS_BIND|MS_REC binds the whole subtree, but a classic remount is
    // NON-recursive — host submounts under the bind would stay writable.
    // mount_setattr(AT_RECURSIVE) locks the entire subtree read-only;
    // fall back to the single-mount remount on pre-5.12 kernels.
    struct ck_mount_attr ma {};
    ma.attr_set = MOUNT_ATTR_RDONLY | MOUNT_ATTR_NOSUID;
    if (syscall(SYS_mount_setattr, AT_FDCWD, dst.c_str(),
                AT_RECURSIVE, &ma, sizeof ma) != 0) {
      if (mount(nullptr, dst.c_str(), nullptr,
                MS_BIND | MS_REMOUNT | MS_RDONLY | MS_NOSUID | MS_NODEV,
                nullptr) != 0)
        warn("ro remount %s", dst.c_str());
    }
  }
}

void lo_up() {
  int fd = socket(AF_INET, SOCK_DGRAM, 0);
  if (fd < 0) { warn("lo: socket"); return; }
  struct ifreq ifr{};
  strncpy(ifr.ifr_name, "lo", IFNAMSIZ - 1);
  if (ioctl(fd, SIOCGIFFLAGS, &ifr) == 0) {
    ifr.ifr_flags |= IFF_UP | IFF_RUNNING;
    if (ioctl(fd, SIOCSIFFLAGS, &ifr) != 0) warn("lo: up");
  }
  close(fd);
}

int g_sync_pipe[2];   // parent writes 1 byte after cgroup attach

int child_main(void*) {
  const Spec& s = g_spec;
  // wait for the shim to finish cgroup attachment
  char b;
  close(g_sync_pipe[1]);
  if (read(g_sync_pipe[0], &b, 1) != 1) die("sync pipe");
  close(g_sync_pipe[0]);

  trace("child_released");
  if (sethostname(s.hostname.c_str(), s.hostname.size()) != 0) warn("sethostname");
  if (s.netns) lo_up();
  trace("lo_up");

  // our mount ops must not propagate back to the host
  mnt(nullptr, "/", nullptr, MS_REC | MS_PRIVATE, nullptr);

  // rootfs overlay: lowerdirs (top first) + upper/work
  std::string lower;
  for (size_t i = 0; i < s.lowerdirs.size(); i++) {
    if (i) lower += ':';
    lower += s.lowerdirs[i];
  }
  ck::mkdirs(s.merged);
  std::string opts = "lowerdir=" + lower + ",upperdir=" + s.upper + ",workdir=" + s.work;
  // "userxattr" not needed (rootful); index off for hostfs lower reuse
  if (mount("overlay", s.merged.c_str(), "overlay", 0, opts.c_str()) != 0)
    die("overlay mount (%s)", opts.c_str());
  trace("overlay_mounted");

  const std::string& m = s.merged;

  // /proc of the new pidns
  ck::mkdirs(m + "/proc");
  mnt("proc", (m + "/proc").c_str(), "proc", MS_NOSUID | MS_NODEV | MS_NOEXEC, nullptr);

  // /sys: fresh sysfs so /sys/class/net reflects the sandbox's netns
  // (ROCm's /sys/class/kfd + /sys/class/drm + hwmon are netns-global and
  // fully visible); fall back to a ro bind of the host /sys if the fresh
  // mount is refused.
  ck::mkdirs(m + "/sys");
  if (mount("sysfs", (m + "/sys").c_str(), "sysfs",
            MS_RDONLY | MS_NOSUID | MS_NODEV | MS_NOEXEC, nullptr) != 0) {
    if (mount("/sys", (m + "/sys").c_str(), nullptr, MS_BIND | MS_REC, nullptr) != 0)
      warn("sys bind");
    mount(nullptr, (m + "/sys").c_str(), nullptr,
          MS_BIND | MS_REMOUNT | MS_RDONLY | MS_NOSUID | MS_NODEV | MS_NOEXEC, nullptr);
  }

  // /dev: private tmpfs with only the standard nodes + this sandbox's
  // allocated GPU devices. Nodes are mknod'd (a bind mount costs ~1.5 ms
  // each; mknod is microseconds) — access control comes from the device
  // cgroup allow-list plus the fact that agents run unprivileged (only
  // our own ckd/ckgw are root inside).
  ck::mkdirs(m + "/dev");
  mnt("tmpfs", (m + "/dev").c_str(), "tmpfs", MS_NOSUID | MS_STRICTATIME,
      "mode=755,size=65536k");
  auto make_node = [&](const char* host_path) {
    struct stat st;
    if (stat(host_path, &st) != 0 || !(S_ISCHR(st.st_mode) || S_ISBLK(st.st_mode))) {
      return;
    }
    std::string dst = m + host_path;
    size_t slash = dst.rfind('/');
    if (slash != std::string::npos) ck::mkdirs(dst.substr(0, slash));
    mode_t mode = (S_ISCHR(st.st_mode) ? S_IFCHR : S_IFBLK) | 0666;
    if (mknod(dst.c_str(), mode, st.st_rdev) != 0) {
      // e.g. no CAP_MKNOD: fall back to a bind mount
      bind_file(host_path, dst, false);
    } else {
      // mknod's mode is masked by umask — the unprivileged agent user
      // must still be able to open the standard nodes and its GPUs
      chmod(dst.c_str(), 0666);
    }
  };
  for (const char* d : {"/dev/null", "/dev/zero", "/dev/full", "/dev/random",
                        "/dev/urandom", "/dev/tty"}) {
    make_node(d);
  }
  for (const auto& d : s.devices.as_arr()) {
    const std::string& p = d["path"].as_str();
    if (ck::exists(p)) make_node(p.c_str());
    else warn("gpu device missing: %s", p.c_str());
  }
  ck::mkdirs(m + "/dev/pts");
  mnt("devpts", (m + "/dev/pts").c_str(), "devpts", MS_NOSUID | MS_NOEXEC,
      "newinstance,ptmxmode=0666,mode=0620,gid=5");
  bind_file(m + "/dev/pts/ptmx", m + "/dev/ptmx", false);
  ck::mkdirs(m + "/dev/shm");
  mnt("tmpfs", (m + "/dev/shm").c_str(), "tmpfs", MS_NOSUID | MS_NODEV,
      "mode=1777,size=65536k");
  if (symlink("/proc/self/fd", (m + "/dev/fd").c_str()) != 0) warn("symlink");
  if (symlink("/proc/self/fd/0", (m + "/dev/stdin").c_str()) != 0) warn("symlink");
  if (symlink("/proc/self/fd/1", (m + "/dev/stdout").c_str()) != 0) warn("symlink");
  if (symlink("/proc/self/fd/2", (m + "/dev/stderr").c_str()) != 0) warn("symlink");

  // writable scratch
  ck::mkdirs(m + "/tmp");
  mnt("tmpfs", (m + "/tmp").c_str(), "tmpfs", MS_NOSUID | MS_NODEV, "mode=1777");
  ck::mkdirs(m + "/run");
  mnt("tmpfs", (m + "/run").c_str(), "tmpfs", MS_NOSUID | MS_NODEV, "mode=755");

  // the per-sandbox runtime dir (control socket, spec, ckd binary, logs)
  bind_file(s.rundir, m + "/run/clawker", false);

  // per-sandbox identity files: the engine writes them straight into the
  // overlay upper (statedir/upper/etc/) — zero mounts. Legacy rundir
  // copies are still honored via bind for hand-written specs.
  for (const char* f : {"resolv.conf", "hosts", "hostname"}) {
    std::string src = s.rundir + "/" + f;
    if (ck::exists(src) && !ck::exists(s.upper + "/etc/" + f))
      bind_file(src, m + "/etc/" + f, true);
  }

  // user-requested mounts (workspace, volumes, host-state)
  for (const auto& mt : s.mounts.as_arr()) {
    const std::string& type = mt["type"].as_str();
    const std::string& dst = m + mt["dst"].as_str();
    bool ro = mt["ro"].as_bool(false);
    if (type == "tmpfs") {
      ck::mkdirs(dst);
      mnt("tmpfs", dst.c_str(), "tmpfs", MS_NOSUID | MS_NODEV,
          mt["opts"].as_str().empty() ? "mode=755" : mt["opts"].as_str().c_str());
    } else {
      bind_file(mt["src"].as_str(), dst, ro);
    }
  }

  trace("mounts_done");
  // pivot into the sandbox rootfs
  std::string oldroot = m + "/.oldroot";
  ck::mkdirs(oldroot, 0700);
  if (syscall(SYS_pivot_root, m.c_str(), oldroot.c_str()) != 0) die("pivot_root");
  if (chdir("/") != 0) die("chdir /");
  if (umount2("/.oldroot", MNT_DETACH) != 0) die("umount oldroot");
  rmdir("/.oldroot");

  // exec the PID-1 supervisor (staged into the rundir by the engine)
  trace("pivoted");
  const char* ckd = "/run/clawker/bin/ckd";
  std::string specs = "CKD_SPEC=/run/clawker/spec.json";
  std::vector<char*> envp;
  envp.push_back(const_cast<char*>(specs.c_str()));
  envp.push_back(const_cast<char*>("PATH=/usr/local/sbin:/usr/local/bin:/usr/sbin:/usr/bin:/sbin:/bin"));
  envp.push_back(nullptr);
  char* argv[] = {const_cast<char*>("ckd"), nullptr};
  execve(ckd, argv, envp.data());
  die("exec ckd");
}

// ------------------------------------------------------------ shim ---------

void on_signal(int sig) {
  if (g_child > 0) kill(g_child, sig);
}

// proc backend: no namespaces available on this host — plain child that
// execs ckd against the host filesystem (rundir paths stay host paths).
int proc_child(const Spec& s) {
  char b;
  close(g_sync_pipe[1]);
  if (read(g_sync_pipe[0], &b, 1) != 1) die("sync pipe");
  close(g_sync_pipe[0]);
  std::string ckd = s.rundir + "/bin/ckd";
  std::string specs = "CKD_SPEC=" + s.spec_path;
  std::vector<char*> envp;
  envp.push_back(const_cast<char*>(specs.c_str()));
  envp.push_back(const_cast<char*>(
      "PATH=/usr/local/sbin:/usr/local/bin:/usr/sbin:/usr/bin:/sbin:/bin"));
  envp.push_back(nullptr);
  char* argv[] = {const_cast<char*>("ckd"), nullptr};
  execve(ckd.c_str(), argv, envp.data());
  die("exec ckd (proc backend)");
}

int run(const std::string& spec_path) {
  g_trace = getenv("CK_TRACE") != nullptr;
  g_t0 = now_us();
  g_spec = parse_spec(spec_path);
  g_spec.spec_path = spec_path;
  trace("spec_parsed");
  const Spec& s = g_spec;

  if (s.backend == "ns") {
    ck::mkdirs(s.upper);
    ck::mkdirs(s.work);
  }
  ck::mkdirs(s.rundir, 0711);

  cgroups_setup(s);
  trace("cgroups_setup");

  int attempts = 0;
restart_attempt:
  if (pipe2(g_sync_pipe, O_CLOEXEC) != 0) die("pipe");

  if (s.backend == "proc") {
    g_child = fork();
    if (g_child < 0) die("fork");
    if (g_child == 0) {
      setsid();
      return proc_child(s);
    }
  } else {
    int flags = CLONE_NEWNS | CLONE_NEWPID | CLONE_NEWUTS | CLONE_NEWIPC | SIGCHLD;
    if (s.netns) flags |= CLONE_NEWNET;
    constexpr size_t kStack = 1 << 20;
    static char stack[kStack];
    g_child = clone(child_main, stack + kStack, flags, nullptr);
    if (g_child < 0) die("clone");
  }

  trace("cloned");
  cgroups_attach(g_child);
  trace("cgroups_attached");

  // pidfile + status for the engine
  ck::write_file(s.rundir + "/pid", std::to_string(g_child));
  {
    mj::Value st;
    st.set("state", "running").set("pid", (int64_t)g_child).set("shim_pid", (int64_t)getpid());
    ck::write_file(s.rundir + "/status.json", st.dump());
  }

  // release the child past the sync barrier
  close(g_sync_pipe[0]);
  if (write(g_sync_pipe[1], "g", 1) != 1) die("sync write");
  close(g_sync_pipe[1]);

  // forward termination signals to the sandbox's PID 1
  struct sigaction sa{};
  sa.sa_handler = on_signal;
  sigaction(SIGTERM, &sa, nullptr);
  sigaction(SIGINT, &sa, nullptr);
  sigaction(SIGHUP, &sa, nullptr);
  signal(SIGPIPE, SIG_IGN);

  int wstatus = 0;
  pid_t r;
  do {
    r = waitpid(g_child, &wstatus, 0);
  } while (r < 0 && errno == EINTR);

  int code = 0, sig = 0;
  if (WIFEXITED(wstatus)) code = WEXITSTATUS(wstatus);
  else if (WIFSIGNALED(wstatus)) { sig = WTERMSIG(wstatus); code = 128 + sig; }

  // restart policy (reference: docker-style on-failure:N — the CP
  // container itself runs with on-failure:3, bootstrap.go:501)
  if (s.restart == "on-failure" && code != 0 && attempts < s.restart_max) {
    attempts++;
    warn("restarting sandbox (exit %d, attempt %d/%lld)", code, attempts,
         (long long)s.restart_max);
    mj::Value st;
    st.set("state", "restarting").set("attempt", (int64_t)attempts);
    ck::write_file(s.rundir + "/status.json", st.dump());
    unlink((s.rundir + "/" + "ctl.sock").c_str());
    usleep(200000 * attempts);   // linear backoff
    goto restart_attempt;
  }

  {
    mj::Value ex;
    ex.set("code", (int64_t)code).set("signal", (int64_t)sig)
      .set("at", (int64_t)time(nullptr));
    ck::write_file(s.rundir + "/exit.json", ex.dump());
    mj::Value st;
    st.set("state", "exited").set("code", (int64_t)code);
    ck::write_file(s.rundir + "/status.json", st.dump());
  }
  cgroups_cleanup();
  return code;
}

}  // namespace

int main(int argc, char** argv) {
  if (argc >= 3 && strcmp(argv[1], "run") == 0) return run(argv[2]);
  fprintf(stderr, "usage: ckrt run <spec.json>\n");
  return 2;
}
