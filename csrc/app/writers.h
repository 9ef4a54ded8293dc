// Native detection-product writers: ${prefix}${counter}.bin / .N.npy /
// .L.tim (formats identical to the reference write_signal_pipe.hpp:150-280
// and the unit-tested Python twin srtb_amd/io/writers.py).
#pragma once

#include <fcntl.h>
#include <unistd.h>

#include <cerrno>
#include <complex>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <fstream>
#include <sstream>
#include <string>
#include <sys/stat.h>
#include <vector>

namespace srtb_app {

inline bool file_exists(const std::string& p) {
  struct stat st;
  return ::stat(p.c_str(), &st) == 0;
}

inline void write_baseband_bin(const std::string& prefix, uint64_t counter,
                               const uint8_t* data, size_t bytes) {
  const std::string path = prefix + std::to_string(counter) + ".bin";
  const int fd = ::open(path.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0644);
  if (fd < 0) throw std::runtime_error("cannot open " + path);
  size_t off = 0;
  while (off < bytes) {
    const ssize_t w = ::write(fd, data + off, bytes - off);
    if (w <= 0) break;
    off += (size_t)w;
  }
  ::fdatasync(fd);  // reference fdatasyncs baseband dumps
  ::close(fd);
}

// minimal .npy (format 1.0) writer for complex64 [rows][cols]; the
// per-polarization index is claimed with O_EXCL so concurrent pool writers
// (both pols of one block) get distinct .i.npy names
inline void write_spectrum_npy(const std::string& prefix, uint64_t counter,
                               const std::complex<float>* data, size_t rows,
                               size_t cols) {
  int fd = -1;
  for (int i = 0;; ++i) {  // multiple polarizations: first free index
    const std::string path = prefix + std::to_string(counter) + "." +
                             std::to_string(i) + ".npy";
    fd = ::open(path.c_str(), O_WRONLY | O_CREAT | O_EXCL, 0644);
    if (fd >= 0) break;
    if (errno != EEXIST)
      throw std::runtime_error("cannot create " + path);
  }
  std::ostringstream hd;
  hd << "{'descr': '<c8', 'fortran_order': False, 'shape': (" << rows << ", "
     << cols << "), }";
  std::string header = hd.str();
  const size_t total = 10 + header.size() + 1;
  const size_t pad = (64 - total % 64) % 64;
  header += std::string(pad, ' ');
  header += '\n';
  std::string head;
  head += std::string("\x93NUMPY\x01", 8 - 1);
  head += '\0';
  const uint16_t hlen = (uint16_t)header.size();
  head.append(reinterpret_cast<const char*>(&hlen), 2);
  head += header;
  size_t off = 0;
  while (off < head.size()) {
    const ssize_t w = ::write(fd, head.data() + off, head.size() - off);
    if (w <= 0) break;
    off += (size_t)w;
  }
  const char* body = reinterpret_cast<const char*>(data);
  const size_t nbytes = rows * cols * sizeof(std::complex<float>);
  off = 0;
  while (off < nbytes) {
    const ssize_t w = ::write(fd, body + off, nbytes - off);
    if (w <= 0) break;
    off += (size_t)w;
  }
  ::close(fd);
}

inline void write_time_series_tim(const std::string& prefix, uint64_t counter,
                                  size_t boxcar, const float* data,
                                  size_t n) {
  const std::string path = prefix + std::to_string(counter) + "." +
                           std::to_string(boxcar) + ".tim";
  std::ofstream f(path, std::ios::binary);
  f.write(reinterpret_cast<const char*>(data), n * sizeof(float));
}

}  // namespace srtb_app
