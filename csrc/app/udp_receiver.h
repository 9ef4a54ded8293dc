// Native UDP baseband ingest: recvmmsg batch provider + counter-gap block
// assembler (reference io/udp/recvmmsg_packet_provider.hpp:41-139 and
// io/udp/udp_receiver.hpp:180-272).  Designed for >=1 GB/s sustained:
// 128-packet recvmmsg batches into a preallocated ring, SO_RCVBUF maxed,
// receiver thread pinned to a configured core, payloads land directly in
// the (pinned) block buffer at (counter - begin) * payload.
#pragma once

#include <arpa/inet.h>
#include <netinet/in.h>
#include <sys/mman.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstdint>
#include <cstring>
#include <functional>
#include <stdexcept>
#include <string>
#include <vector>

#include "runtime.h"

namespace srtb_app {

struct PacketFormat {
  // parse (counter, timestamp) from a packet; returns payload offset
  size_t header_size = 8;
  size_t payload_size = 4096;  // bytes of baseband per packet
  // counter = little-endian u64 at offset 0 (fastmb_roach2 / naocpsr_snap1);
  // gznupsr_a1 reads VDIF words 6,7 at offset 24
  size_t counter_offset = 0;

  uint64_t parse_counter(const uint8_t* pkt) const {
    uint64_t c = 0;
    std::memcpy(&c, pkt + counter_offset, 8);  // little-endian hosts
    return c;
  }

  // VDIF-framed formats (gznupsr_a1): the invalid-data flag is bit 31 of
  // word 0 (reference io/vdif_header.hpp:28-61); others have no validity
  // marker.  Invalid frames are zero-filled by the assembler like losses.
  bool vdif = false;
  bool packet_valid(const uint8_t* pkt) const {
    if (!vdif) return true;
    uint32_t w0 = 0;
    std::memcpy(&w0, pkt, 4);
    return (w0 >> 31) == 0;  // invalid_data bit clear
  }

  size_t packet_size() const { return header_size + payload_size; }

  // 'simple' (reference backend_registry.hpp:36-39) is a headerless linear
  // sample stream with no counter: packets append sequentially, any size
  bool headerless = false;

  static PacketFormat for_backend(const std::string& name) {
    PacketFormat f;
    if (name == "fastmb_roach2" || name == "naocpsr_roach2" ||
        name == "naocpsr_snap1") {
      f.header_size = 8;
      f.payload_size = 4096;
      f.counter_offset = 0;
    } else if (name == "gznupsr_a1") {
      f.header_size = 64;
      f.payload_size = 8192;
      f.counter_offset = 24;  // VDIF words 6..7
      f.vdif = true;
    } else if (name == "simple") {
      f.header_size = 0;
      f.payload_size = 4096;  // nominal; any size accepted (headerless)
      f.counter_offset = 0;
      f.headerless = true;
    } else {
      throw std::runtime_error("unknown backend format: " + name);
    }
    return f;
  }
};

class RecvmmsgProvider {
 public:
  static constexpr int kBatch = 128;

  RecvmmsgProvider(const std::string& address, int port, size_t packet_size,
                   int rcvbuf_bytes = 1 << 28)
      : packet_size_(packet_size) {
    fd_ = ::socket(AF_INET, SOCK_DGRAM, 0);
    if (fd_ < 0) throw std::runtime_error("socket() failed");
    ::setsockopt(fd_, SOL_SOCKET, SO_RCVBUF, &rcvbuf_bytes,
                 sizeof(rcvbuf_bytes));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port);
    if (::inet_pton(AF_INET, address.c_str(), &addr.sin_addr) != 1)
      throw std::runtime_error("bad address " + address);
    if (::bind(fd_, (sockaddr*)&addr, sizeof(addr)) != 0)
      throw std::runtime_error("bind failed on " + address + ":" +
                               std::to_string(port));
    // receive timeout so the stop flag is polled
    timeval tv{0, 100000};
    ::setsockopt(fd_, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));

    // hugepage-backed receive ring (reference recvmmsg provider uses
    // 2 MB-aligned hugepage buffers); plain pages when unavailable
    const size_t ring_bytes = (size_t)kBatch * packet_size_;
    const size_t huge_bytes = (ring_bytes + (2u << 20) - 1) & ~((size_t)(2u << 20) - 1);
    void* hp = ::mmap(nullptr, huge_bytes, PROT_READ | PROT_WRITE,
                      MAP_PRIVATE | MAP_ANONYMOUS | MAP_HUGETLB, -1, 0);
    if (hp != MAP_FAILED) {
      huge_ = static_cast<uint8_t*>(hp);
      huge_bytes_ = huge_bytes;
    } else {
      buf_.resize(ring_bytes);
    }
    uint8_t* ring = huge_ ? huge_ : buf_.data();
    iov_.resize(kBatch);
    msgs_.resize(kBatch);
    for (int i = 0; i < kBatch; ++i) {
      iov_[i].iov_base = ring + (size_t)i * packet_size_;
      iov_[i].iov_len = packet_size_;
      std::memset(&msgs_[i], 0, sizeof(mmsghdr));
      msgs_[i].msg_hdr.msg_iov = &iov_[i];
      msgs_[i].msg_hdr.msg_iovlen = 1;
    }
  }

  ~RecvmmsgProvider() {
    if (fd_ >= 0) ::close(fd_);
    if (huge_) ::munmap(huge_, huge_bytes_);
  }

  // receive up to kBatch packets; returns count (0 on timeout)
  int receive() { return std::max(0, (int)::recvmmsg(fd_, msgs_.data(),
                                                     kBatch, 0, nullptr)); }

  const uint8_t* packet(int i) const {
    return (huge_ ? huge_ : buf_.data()) + (size_t)i * packet_size_;
  }
  bool hugepages() const { return huge_ != nullptr; }
  size_t packet_len(int i) const { return msgs_[i].msg_len; }

 private:
  int fd_ = -1;
  size_t packet_size_;
  std::vector<uint8_t> buf_;
  uint8_t* huge_ = nullptr;
  size_t huge_bytes_ = 0;
  std::vector<iovec> iov_;
  std::vector<mmsghdr> msgs_;
};

struct LossStats {
  uint64_t received = 0, lost = 0, out_of_order = 0, wrong_size = 0;
  uint64_t duplicate = 0, invalid = 0;
  double loss_rate() const {
    const uint64_t t = received + lost;
    return t ? (double)lost / (double)t : 0.0;
  }
};

// Counter-gap block assembler: identical protocol logic to the (unit-tested)
// Python twin srtb_amd/io/udp.py BlockAssembler.
class BlockAssembler {
 public:
  BlockAssembler(PacketFormat fmt, size_t block_bytes, uint8_t* block_buf)
      : fmt_(fmt), block_bytes_(block_bytes), buf_(block_buf) {
    if (!fmt.headerless && block_bytes % fmt.payload_size != 0)
      throw std::runtime_error("block not a multiple of payload");
    packets_per_block_ =
        fmt.headerless ? 0 : block_bytes / fmt.payload_size;
    occupied_.assign(packets_per_block_, 0);
    std::memset(buf_, 0, block_bytes_);
  }

  // feed one packet; returns true when the current block completed (caller
  // consumes buf_ and MUST call begin_next() before pushing more)
  bool push(const uint8_t* pkt, size_t len) {
    if (fmt_.headerless) return push_headerless(pkt, len);
    if (len != fmt_.packet_size()) {
      ++stats_.wrong_size;
      return false;
    }
    if (!fmt_.packet_valid(pkt)) {
      ++stats_.invalid;
      return false;  // frame slot stays zero, counted when the block closes
    }
    const uint64_t counter = fmt_.parse_counter(pkt);
    if (!started_) {
      begin_ = counter;
      started_ = true;
      first_timestamp_ = counter;
    }
    const int64_t idx = (int64_t)(counter - begin_);
    if (idx < 0) {
      ++stats_.out_of_order;
      return false;
    }
    if ((uint64_t)idx >= packets_per_block_) {
      // complete current block; remember the packet for the next one
      pending_.assign(pkt, pkt + len);
      stats_.lost += packets_per_block_ - filled_;
      const uint64_t skip = (uint64_t)idx / packets_per_block_;
      stats_.lost += (skip - 1) * packets_per_block_;
      // the counter that identifies the block being handed to the caller
      // (reference udp_receiver.hpp block_first_counter — NOT the stream's
      // first counter ever: product files are named with this value)
      completed_begin_ = begin_;
      next_begin_ = begin_ + skip * packets_per_block_;
      return true;
    }
    if (occupied_[(size_t)idx]) {
      ++stats_.duplicate;
      return false;
    }
    occupied_[(size_t)idx] = 1;
    std::memcpy(buf_ + (size_t)idx * fmt_.payload_size,
                pkt + fmt_.header_size, fmt_.payload_size);
    ++filled_;
    ++stats_.received;
    return false;
  }

  void begin_next() {
    begin_ = next_begin_;
    filled_ = 0;
    std::fill(occupied_.begin(), occupied_.end(), 0);
    std::memset(buf_, 0, block_bytes_);
    if (!pending_.empty()) {
      std::vector<uint8_t> p;
      p.swap(pending_);
      push(p.data(), p.size());
    }
  }

  void set_block_buffer(uint8_t* b) {
    buf_ = b;
    std::memset(buf_, 0, block_bytes_);
  }

  uint64_t first_timestamp() const { return first_timestamp_; }
  // begin counter of the most recently COMPLETED block (valid after push()
  // returned true, until the next completion) — use this to stamp/name the
  // block's products, matching the reference's per-block first counter
  uint64_t block_begin_counter() const { return completed_begin_; }
  const LossStats& stats() const { return stats_; }

 private:
  // 'simple' headerless mode: sequential append of raw sample bytes; block
  // completes when full, remainder carries into the next block.  The block
  // counter is the byte offset of the block start in the stream.
  bool push_headerless(const uint8_t* pkt, size_t len) {
    const size_t take = std::min(len, block_bytes_ - fill_bytes_);
    std::memcpy(buf_ + fill_bytes_, pkt, take);
    fill_bytes_ += take;
    stats_.received += 1;
    if (fill_bytes_ >= block_bytes_) {
      completed_begin_ = stream_offset_;
      stream_offset_ += block_bytes_;
      if (take < len) pending_.assign(pkt + take, pkt + len);
      fill_bytes_ = 0;
      return true;
    }
    return false;
  }

  PacketFormat fmt_;
  size_t block_bytes_, packets_per_block_;
  uint8_t* buf_;
  uint64_t begin_ = 0, next_begin_ = 0, filled_ = 0, first_timestamp_ = 0;
  uint64_t completed_begin_ = 0;
  uint64_t fill_bytes_ = 0, stream_offset_ = 0;  // headerless mode
  bool started_ = false;
  std::vector<uint8_t> pending_;
  std::vector<uint8_t> occupied_;
  LossStats stats_;
};

}  // namespace srtb_app
