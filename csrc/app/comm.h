// Native multi-GPU communicator for the srtb apps: RCCL over xGMI, one
// process per GPU (SURVEY §2c MI355X equivalent; BASELINE.json north star:
// stream/polarization/beam sharding across the 8 GPUs of one node with
// RCCL reductions of detection statistics).
//
// Bootstrap without MPI: rank 0 writes the ncclUniqueId to a file
// (SRTB_RCCL_ID_FILE, default under /tmp keyed by MASTER_PORT), other
// ranks poll it — the standard single-node pattern.  Rank/world come from
// the torchrun-style env (RANK / WORLD_SIZE / LOCAL_RANK), so
// `python -m torch.distributed.run --nproc-per-node 8 ... srtb-backend ...`
// or a plain shell loop both work as launchers.
#pragma once

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <ctime>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

#include "runtime.h"

namespace srtb_app {

inline void check_nccl(ncclResult_t r, const char* what) {
  if (r != ncclSuccess)
    throw std::runtime_error(std::string("rccl error in ") + what + ": " +
                             ncclGetErrorString(r));
}

class Comm {
 public:
  // Reads RANK/WORLD_SIZE/LOCAL_RANK from the environment; world_size <= 1
  // leaves the communicator inactive (all ops become local no-ops).
  Comm() {
    const char* ws = std::getenv("WORLD_SIZE");
    const char* rk = std::getenv("RANK");
    world_ = ws ? std::atoi(ws) : 1;
    rank_ = rk ? std::atoi(rk) : 0;
    const char* lr = std::getenv("LOCAL_RANK");
    local_rank_ = lr ? std::atoi(lr) : rank_;
    // SRTB_FORCE_COMM=1 initializes a single-rank communicator anyway —
    // lets 1-GPU boxes exercise the real RCCL init + allreduce path
    if (world_ <= 1 && !std::getenv("SRTB_FORCE_COMM")) return;

    int ndev = 0;
    srtb_hip::check_hip(hipGetDeviceCount(&ndev), "device count");
    srtb_hip::check_hip(hipSetDevice(local_rank_ % ndev), "set device");

    ncclUniqueId id{};
    const std::string path = id_file_path();
    if (rank_ == 0) {
      check_nccl(ncclGetUniqueId(&id), "get unique id");
      const std::string tmp = path + ".tmp";
      FILE* f = std::fopen(tmp.c_str(), "wb");
      if (!f) throw std::runtime_error("cannot write " + tmp);
      std::fwrite(&id, sizeof(id), 1, f);
      std::fclose(f);
      if (std::rename(tmp.c_str(), path.c_str()) != 0)
        throw std::runtime_error("cannot publish " + path);
    } else {
      // poll for rank 0's id (bounded); a file older than 120 s is a
      // leftover from a crashed earlier run and must not be joined
      for (int tries = 0;; ++tries) {
        struct stat st {};
        if (::stat(path.c_str(), &st) == 0 &&
            std::time(nullptr) - st.st_mtime < 120) {
          FILE* f = std::fopen(path.c_str(), "rb");
          if (f) {
            const size_t got = std::fread(&id, 1, sizeof(id), f);
            std::fclose(f);
            if (got == sizeof(id)) break;
          }
        }
        if (tries > 3000)
          throw std::runtime_error("timed out waiting for " + path);
        std::this_thread::sleep_for(std::chrono::milliseconds(100));
      }
    }
    check_nccl(ncclCommInitRank(&comm_, world_, id, rank_), "comm init");
    srtb_hip::check_hip(hipStreamCreateWithFlags(&stream_,
                                                 hipStreamNonBlocking),
                        "comm stream");
    srtb_hip::check_hip(hipMalloc(&dev_buf_, kMax * sizeof(uint64_t)),
                        "comm buf");
    if (rank_ == 0) (void)std::remove(path.c_str());
  }

  ~Comm() {
    if (dev_buf_) (void)hipFree(dev_buf_);
    if (stream_) (void)hipStreamDestroy(stream_);
    if (comm_) (void)ncclCommDestroy(comm_);
  }
  Comm(const Comm&) = delete;
  Comm& operator=(const Comm&) = delete;

  int rank() const { return rank_; }
  int world() const { return world_; }
  int local_rank() const { return local_rank_; }
  bool active() const { return comm_ != nullptr; }

  // Sum-all-reduce a small vector of counters across ranks (blocking; used
  // for end-of-run stats, off the per-block hot path).
  void allreduce_sum(std::vector<uint64_t>& v) {
    if (!active() || v.empty()) return;
    if (v.size() > kMax) throw std::runtime_error("allreduce too large");
    srtb_hip::check_hip(hipMemcpyAsync(dev_buf_, v.data(),
                                       v.size() * sizeof(uint64_t),
                                       hipMemcpyHostToDevice, stream_),
                        "comm h2d");
    check_nccl(ncclAllReduce(dev_buf_, dev_buf_, v.size(), ncclUint64,
                             ncclSum, comm_, stream_),
               "allreduce");
    srtb_hip::check_hip(hipMemcpyAsync(v.data(), dev_buf_,
                                       v.size() * sizeof(uint64_t),
                                       hipMemcpyDeviceToHost, stream_),
                        "comm d2h");
    srtb_hip::check_hip(hipStreamSynchronize(stream_), "comm sync");
  }

 private:
  static std::string id_file_path() {
    if (const char* p = std::getenv("SRTB_RCCL_ID_FILE")) return p;
    const char* port = std::getenv("MASTER_PORT");
    return std::string("/tmp/srtb_rccl_id_") + (port ? port : "0");
  }

  static constexpr size_t kMax = 64;
  int rank_ = 0, world_ = 1, local_rank_ = 0;
  ncclComm_t comm_ = nullptr;
  hipStream_t stream_ = nullptr;
  uint64_t* dev_buf_ = nullptr;
};

// round-robin shard of `count` items for this rank (Python twin:
// srtb_amd/parallel/sharding.py shard_streams)
inline std::vector<int> shard_indices(int count, int world, int rank) {
  std::vector<int> out;
  for (int i = rank; i < count; i += world) out.push_back(i);
  return out;
}

}  // namespace srtb_app
