// Shared helpers for ckrt/ckd: error handling, file IO, base64, unix
// sockets with length-prefixed JSON frames and SCM_RIGHTS fd passing.
#pragma once

#include <errno.h>
#include <fcntl.h>
#include <stdarg.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <sys/types.h>
#include <sys/un.h>
#include <unistd.h>

#include <string>
#include <vector>

#include "minijson.hpp"

namespace ck {

[[noreturn]] inline void die(const char* fmt, ...) {
  va_list ap;
  va_start(ap, fmt);
  fprintf(stderr, "ck: fatal: ");
  vfprintf(stderr, fmt, ap);
  if (errno) fprintf(stderr, ": %s", strerror(errno));
  fprintf(stderr, "\n");
  va_end(ap);
  _exit(111);
}

inline void warn(const char* fmt, ...) {
  va_list ap;
  va_start(ap, fmt);
  fprintf(stderr, "ck: warn: ");
  vfprintf(stderr, fmt, ap);
  if (errno) fprintf(stderr, ": %s", strerror(errno));
  fprintf(stderr, "\n");
  va_end(ap);
}

inline std::string read_file(const std::string& path) {
  int fd = open(path.c_str(), O_RDONLY | O_CLOEXEC);
  if (fd < 0) die("open %s", path.c_str());
  std::string out;
  char buf[65536];
  ssize_t n;
  while ((n = read(fd, buf, sizeof buf)) > 0) out.append(buf, n);
  close(fd);
  return out;
}

inline bool write_file(const std::string& path, const std::string& data) {
  int fd = open(path.c_str(), O_WRONLY | O_CREAT | O_TRUNC | O_CLOEXEC, 0644);
  if (fd < 0) return false;
  size_t off = 0;
  while (off < data.size()) {
    ssize_t n = write(fd, data.data() + off, data.size() - off);
    if (n < 0) { close(fd); return false; }
    off += n;
  }
  close(fd);
  return true;
}

inline bool exists(const std::string& path) {
  struct stat st;
  return lstat(path.c_str(), &st) == 0;
}

inline void mkdirs(const std::string& path, mode_t mode = 0755) {
  std::string cur;
  for (size_t i = 0; i < path.size(); i++) {
    cur += path[i];
    if (path[i] == '/' && cur.size() > 1) {
      mkdir(cur.c_str(), mode);
    }
  }
  if (!cur.empty()) mkdir(cur.c_str(), mode);
}

// ----------------------------------------------------------------- base64 --
inline std::string b64_encode(const uint8_t* data, size_t len) {
  static const char tbl[] =
      "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";
  std::string out;
  out.reserve((len + 2) / 3 * 4);
  for (size_t i = 0; i < len; i += 3) {
    uint32_t v = data[i] << 16;
    if (i + 1 < len) v |= data[i + 1] << 8;
    if (i + 2 < len) v |= data[i + 2];
    out += tbl[(v >> 18) & 63];
    out += tbl[(v >> 12) & 63];
    out += i + 1 < len ? tbl[(v >> 6) & 63] : '=';
    out += i + 2 < len ? tbl[v & 63] : '=';
  }
  return out;
}

inline std::string b64_encode(const std::string& s) {
  return b64_encode(reinterpret_cast<const uint8_t*>(s.data()), s.size());
}

inline std::vector<uint8_t> b64_decode(const std::string& in) {
  auto val = [](char c) -> int {
    if (c >= 'A' && c <= 'Z') return c - 'A';
    if (c >= 'a' && c <= 'z') return c - 'a' + 26;
    if (c >= '0' && c <= '9') return c - '0' + 52;
    if (c == '+') return 62;
    if (c == '/') return 63;
    return -1;
  };
  std::vector<uint8_t> out;
  uint32_t acc = 0;
  int bits = 0;
  for (char c : in) {
    int v = val(c);
    if (v < 0) continue;   // skip '=' and whitespace
    acc = (acc << 6) | v;
    bits += 6;
    if (bits >= 8) {
      bits -= 8;
      out.push_back((acc >> bits) & 0xFF);
    }
  }
  return out;
}

// -------------------------------------------------------------- unix sock --
inline int unix_listen(const std::string& path, int backlog = 128) {
  int fd = socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (fd < 0) return -1;
  sockaddr_un addr{};
  addr.sun_family = AF_UNIX;
  if (path.size() >= sizeof(addr.sun_path)) { close(fd); errno = ENAMETOOLONG; return -1; }
  strncpy(addr.sun_path, path.c_str(), sizeof(addr.sun_path) - 1);
  unlink(path.c_str());
  if (bind(fd, reinterpret_cast<sockaddr*>(&addr), sizeof addr) < 0) { close(fd); return -1; }
  if (listen(fd, backlog) < 0) { close(fd); return -1; }
  return fd;
}

inline int unix_connect(const std::string& path) {
  int fd = socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (fd < 0) return -1;
  sockaddr_un addr{};
  addr.sun_family = AF_UNIX;
  if (path.size() >= sizeof(addr.sun_path)) { close(fd); errno = ENAMETOOLONG; return -1; }
  strncpy(addr.sun_path, path.c_str(), sizeof(addr.sun_path) - 1);
  if (connect(fd, reinterpret_cast<sockaddr*>(&addr), sizeof addr) < 0) { close(fd); return -1; }
  return fd;
}

// ------------------------------------------------------------------ frames -
// Wire format: 4-byte big-endian length N, then N bytes of UTF-8 JSON.
// Mirrored by clawker_amd/engine/wire.py — keep in lockstep.

inline bool read_exact(int fd, void* buf, size_t n) {
  auto* p = static_cast<uint8_t*>(buf);
  while (n > 0) {
    ssize_t r = read(fd, p, n);
    if (r == 0) return false;
    if (r < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    p += r;
    n -= r;
  }
  return true;
}

inline bool write_exact(int fd, const void* buf, size_t n) {
  auto* p = static_cast<const uint8_t*>(buf);
  while (n > 0) {
    ssize_t r = write(fd, p, n);
    if (r < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    p += r;
    n -= r;
  }
  return true;
}

constexpr size_t kMaxFrame = 16 * 1024 * 1024;

inline bool send_frame(int fd, const mj::Value& v) {
  std::string body = v.dump();
  if (body.size() > kMaxFrame) return false;
  uint8_t hdr[4] = {
      static_cast<uint8_t>(body.size() >> 24), static_cast<uint8_t>(body.size() >> 16),
      static_cast<uint8_t>(body.size() >> 8), static_cast<uint8_t>(body.size())};
  return write_exact(fd, hdr, 4) && write_exact(fd, body.data(), body.size());
}

inline bool recv_frame(int fd, mj::Value* out) {
  uint8_t hdr[4];
  if (!read_exact(fd, hdr, 4)) return false;
  size_t len = (size_t(hdr[0]) << 24) | (size_t(hdr[1]) << 16) | (size_t(hdr[2]) << 8) | hdr[3];
  if (len > kMaxFrame) return false;
  std::string body(len, '\0');
  if (!read_exact(fd, body.data(), len)) return false;
  try {
    *out = mj::parse(body);
  } catch (const std::exception&) {
    return false;
  }
  return true;
}

// Pass an fd alongside a 1-byte payload (SCM_RIGHTS).
inline bool send_fd(int sock, int fd) {
  char dummy = 'F';
  iovec iov{&dummy, 1};
  char cbuf[CMSG_SPACE(sizeof(int))] = {};
  msghdr msg{};
  msg.msg_iov = &iov;
  msg.msg_iovlen = 1;
  msg.msg_control = cbuf;
  msg.msg_controllen = sizeof cbuf;
  cmsghdr* cm = CMSG_FIRSTHDR(&msg);
  cm->cmsg_level = SOL_SOCKET;
  cm->cmsg_type = SCM_RIGHTS;
  cm->cmsg_len = CMSG_LEN(sizeof(int));
  memcpy(CMSG_DATA(cm), &fd, sizeof(int));
  return sendmsg(sock, &msg, 0) == 1;
}

inline int recv_fd(int sock) {
  char dummy;
  iovec iov{&dummy, 1};
  char cbuf[CMSG_SPACE(sizeof(int))] = {};
  msghdr msg{};
  msg.msg_iov = &iov;
  msg.msg_iovlen = 1;
  msg.msg_control = cbuf;
  msg.msg_controllen = sizeof cbuf;
  if (recvmsg(sock, &msg, 0) != 1) return -1;
  for (cmsghdr* cm = CMSG_FIRSTHDR(&msg); cm; cm = CMSG_NXTHDR(&msg, cm)) {
    if (cm->cmsg_level == SOL_SOCKET && cm->cmsg_type == SCM_RIGHTS) {
      int fd;
      memcpy(&fd, CMSG_DATA(cm), sizeof(int));
      return fd;
    }
  }
  return -1;
}

}  // namespace ck
