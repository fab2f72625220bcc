// Functional probe for the cgroup-v2 device BPF allow-list
// (devbpf.hpp). Usage: devbpf_probe <cgroup-v2-dir>
//
// Creates a child cgroup, attaches an allow-list WITHOUT the null
// device (1:3) but WITH zero (1:5), moves a forked child into it, and
// checks from inside: open(/dev/zero) must succeed, open(/dev/null)
// must fail EPERM. Prints one JSON line; exit 0 = verified, 3 = bpf
// unavailable on this host (EPERM/ENOSYS — caller should skip).
#include <fcntl.h>
#include <stdio.h>
#include <string.h>
#include <sys/stat.h>
#include <sys/types.h>
#include <sys/wait.h>
#include <unistd.h>

#include <string>

#include "../ckrt/devbpf.hpp"

int main(int argc, char** argv) {
  if (argc < 2) {
    fprintf(stderr, "usage: devbpf_probe <cgroup2-dir>\n");
    return 2;
  }
  std::string dir = std::string(argv[1]) + "/devbpf-probe";
  rmdir(dir.c_str());
  if (mkdir(dir.c_str(), 0755) != 0 && errno != EEXIST) {
    printf("{\"status\":\"error\",\"msg\":\"mkdir: %s\"}\n", strerror(errno));
    return 2;
  }
  std::vector<devbpf::Rule> rules = {{2, 1, 5}};   // zero only, NOT null
  std::string err;
  int rc = devbpf::attach(dir, rules, &err);
  if (rc == -EPERM || rc == -ENOSYS || rc == -EACCES) {
    printf("{\"status\":\"unavailable\",\"msg\":\"%s\"}\n", err.c_str());
    rmdir(dir.c_str());
    return 3;
  }
  if (rc != 0) {
    printf("{\"status\":\"error\",\"msg\":\"%s\"}\n", err.c_str());
    rmdir(dir.c_str());
    return 2;
  }
  pid_t pid = fork();
  if (pid == 0) {
    // join the enforced cgroup, then probe both nodes
    FILE* f = fopen((dir + "/cgroup.procs").c_str(), "w");
    if (!f) _exit(10);
    fprintf(f, "%d", getpid());
    fclose(f);
    int zero_fd = open("/dev/zero", O_RDONLY);
    int null_fd = open("/dev/null", O_WRONLY);
    int null_errno = null_fd < 0 ? errno : 0;
    if (zero_fd < 0) _exit(11);          // allowed node blocked: broken
    if (null_fd >= 0) _exit(12);         // denied node open: not enforced
    if (null_errno != EPERM) _exit(13);  // denied with the wrong errno
    _exit(0);
  }
  int st = 0;
  waitpid(pid, &st, 0);
  int code = WIFEXITED(st) ? WEXITSTATUS(st) : 99;
  rmdir(dir.c_str());
  if (code == 0) {
    printf("{\"status\":\"enforced\",\"allowed\":\"/dev/zero\","
           "\"denied\":\"/dev/null EPERM\"}\n");
    return 0;
  }
  printf("{\"status\":\"failed\",\"child_code\":%d}\n", code);
  return 1;
}
