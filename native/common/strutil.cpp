#include "strutil.hpp"

#include <fcntl.h>
#include <unistd.h>

#include <cstring>
#include <random>

namespace strutil {

std::string uuid4_simple() {
  unsigned char bytes[16];
  bool ok = false;
  int fd = ::open("/dev/urandom", O_RDONLY);
  if (fd >= 0) {
    ok = ::read(fd, bytes, sizeof bytes) == static_cast<ssize_t>(sizeof bytes);
    ::close(fd);
  }
  if (!ok) {
    static thread_local std::mt19937_64 rng{std::random_device{}()};
    for (int i = 0; i < 16; i += 8) {
      uint64_t v = rng();
      std::memcpy(bytes + i, &v, 8);
    }
  }
  // set uuid4 version/variant bits
  bytes[6] = static_cast<unsigned char>((bytes[6] & 0x0F) | 0x40);
  bytes[8] = static_cast<unsigned char>((bytes[8] & 0x3F) | 0x80);
  static const char* hex = "0123456789abcdef";
  std::string out;
  out.reserve(32);
  for (unsigned char b : bytes) {
    out += hex[b >> 4];
    out += hex[b & 0xF];
  }
  return out;
}

}  // namespace strutil
