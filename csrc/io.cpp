// Native raw-image block I/O.
//
// C++ equivalent of the reference's per-rank fread/fseek scatter and
// gather-free per-rank writes: each rank reads
// or writes only its own block at computed offsets; parallel writers own
// disjoint byte ranges of one file (pwrite, no locking needed).

#include <fcntl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cerrno>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <utility>
#include <vector>

namespace pconv {

static void throw_errno(const std::string& what) {
  throw std::runtime_error(what + ": " + std::strerror(errno));
}

// Read h rows of rowBytes each, the r-th row at file offset
// (y0+r)*imageRowBytes + xByteOff, into out (h*rowBytes, contiguous).
void read_block_file(const std::string& path, int64_t imageRowBytes,
                     int64_t xByteOff, int64_t y0, int64_t h, int64_t rowBytes,
                     char* out) {
  int fd = ::open(path.c_str(), O_RDONLY);
  if (fd < 0) throw_errno("open " + path);
  for (int64_t r = 0; r < h; ++r) {
    int64_t off = (y0 + r) * imageRowBytes + xByteOff;
    int64_t done = 0;
    while (done < rowBytes) {
      ssize_t n = ::pread(fd, out + r * rowBytes + done, rowBytes - done, off + done);
      if (n < 0) {
        ::close(fd);
        throw_errno("pread " + path);
      }
      if (n == 0) {
        ::close(fd);
        throw std::runtime_error("short read: " + path);
      }
      done += n;
    }
  }
  ::close(fd);
}

void write_block_file(const std::string& path, int64_t imageRowBytes,
                      int64_t totalBytes, int64_t xByteOff, int64_t y0,
                      int64_t h, int64_t rowBytes, const char* in) {
  int fd = ::open(path.c_str(), O_WRONLY | O_CREAT, 0644);
  if (fd < 0) throw_errno("open " + path);
  struct stat st;
  if (::fstat(fd, &st) != 0) {
    ::close(fd);
    throw_errno("fstat " + path);
  }
  if (st.st_size < totalBytes && ::ftruncate(fd, totalBytes) != 0) {
    ::close(fd);
    throw_errno("ftruncate " + path);
  }
  for (int64_t r = 0; r < h; ++r) {
    int64_t off = (y0 + r) * imageRowBytes + xByteOff;
    int64_t done = 0;
    while (done < rowBytes) {
      ssize_t n = ::pwrite(fd, in + r * rowBytes + done, rowBytes - done, off + done);
      if (n < 0) {
        ::close(fd);
        throw_errno("pwrite " + path);
      }
      done += n;
    }
  }
  ::close(fd);
}

// Filter-file parser (format defined in
// models/filters.py): first token k (odd), then k*k floats; '#' comments.
std::pair<int, std::vector<float>> parse_filter_file(const std::string& text);

}  // namespace pconv

#include <sstream>

namespace pconv {

std::pair<int, std::vector<float>> parse_filter_file(const std::string& text) {
  std::string clean;
  clean.reserve(text.size());
  bool comment = false;
  for (char c : text) {
    if (c == '#') comment = true;
    if (c == '\n') comment = false;
    clean.push_back(comment ? ' ' : c);
  }
  std::istringstream is(clean);
  int k = 0;
  if (!(is >> k) || k <= 0 || k % 2 == 0)
    throw std::runtime_error("filter file: bad kernel size");
  std::vector<float> vals(k * k);
  for (auto& v : vals)
    if (!(is >> v)) throw std::runtime_error("filter file: too few coefficients");
  float extra;
  if (is >> extra) throw std::runtime_error("filter file: trailing values");
  return {k, vals};
}

}  // namespace pconv
