// srtb-baseband-receiver — record-only UDP baseband capture
// (reference src/baseband_receiver.cpp:37-87): recvmmsg ingest → counter-gap
// block assembly → composite (stamp ∘ write) pipe appending blocks to
// ${prefix}recording.bin.  No GPU.  Built on the generic pipe framework
// (pipe.h) exactly as the reference builds on composite_pipe.

#include <atomic>
#include <cinttypes>
#include <cstdio>
#include <cstring>
#include <fstream>
#include <memory>
#include <optional>
#include <vector>

#include "config.h"
#include "pipe.h"
#include "runtime.h"
#include "udp_receiver.h"

using namespace srtb_app;

namespace {

struct BlockWork {
  std::shared_ptr<std::vector<uint8_t>> data;
  uint64_t counter = 0;
};

}  // namespace

int main(int argc, char** argv) {
  install_termination_handler();
  long long max_blocks = -1;
  std::vector<char*> cfg_argv{argv[0]};
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    if (a == "--max-blocks") max_blocks = std::stoll(argv[++i]);
    else if (a.rfind("--max-blocks=", 0) == 0) max_blocks = std::stoll(a.substr(13));
    else cfg_argv.push_back(argv[i]);
  }
  Config cfg;
  cfg.parse_args((int)cfg_argv.size(), cfg_argv.data());
  log_level().store(cfg.log_level);

  const auto fmt = PacketFormat::for_backend(cfg.baseband_format_type);
  const size_t block_bytes =
      cfg.baseband_input_count * (size_t)std::abs(cfg.baseband_input_bits) / 8;
  std::vector<uint8_t> block(block_bytes);
  if (!cfg.udp_receiver_cpu_preferred.empty())
    set_thread_affinity(cfg.udp_receiver_cpu_preferred[0]);
  RecvmmsgProvider prov(cfg.udp_receiver_address.at(0),
                        cfg.udp_receiver_port.at(0), fmt.packet_size());
  BlockAssembler assembler(fmt, block_bytes, block.data());
  const std::string out = cfg.baseband_output_file_prefix + "recording.bin";

  // pipeline: [this thread] receive+assemble → queue → [writer pipe thread]
  // composite (stamp ∘ write_file), the reference's cast∘write composite
  SpscQueue<BlockWork, 4> q;
  StopFlag stop;
  std::atomic<long long> written{0};

  auto stamp = [](const StopFlag&, BlockWork w) -> std::optional<BlockWork> {
    SRTB_APP_LOGD("block " << w.counter << " (" << w.data->size()
                           << " bytes) -> writer");
    return w;
  };
  std::ofstream f(out, std::ios::binary | std::ios::app);
  auto write_file = [&f, &written](const StopFlag&,
                                   BlockWork w) -> std::optional<int> {
    f.write(reinterpret_cast<const char*>(w.data->data()),
            (std::streamsize)w.data->size());
    f.flush();
    written.fetch_add(1);
    return 0;
  };
  auto writer = start_pipe("srtb-writer", compose(stamp, write_file),
                           QueueIn<SpscQueue<BlockWork, 4>>{&q},
                           [](const StopFlag&, int) {}, stop);

  long long count = 0;
  while (max_blocks < 0 || count < max_blocks) {
    const int got = prov.receive();
    for (int i = 0; i < got; ++i) {
      if (assembler.push(prov.packet(i), prov.packet_len(i))) {
        auto data = std::make_shared<std::vector<uint8_t>>(
            block.begin(), block.end());
        q.push(BlockWork{std::move(data), assembler.block_begin_counter()},
               stop);
        ++count;
        SRTB_APP_LOGI("block " << count << " assembled (loss_rate="
                               << assembler.stats().loss_rate() << ")");
        assembler.begin_next();
        if (max_blocks >= 0 && count >= max_blocks) break;
      }
    }
  }
  while (written.load() < count) std::this_thread::yield();
  stop.request_stop();
  writer.join();
  std::printf("[srtb-baseband-receiver] wrote %lld blocks to %s\n", count,
              out.c_str());
  return 0;
}
