#!/bin/bash
# Diagnose the native UDP ingest path on a GPU box: run srtb-backend with
# full logging while a python sender streams counter-stamped packets.
set -x
cd /root/repo
mkdir -p gpurun_out
OUT=gpurun_out/udp_debug.log
PORT=29957

python3 - <<'EOF' &
import socket, time, numpy as np, subprocess, os
sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
pay = np.random.default_rng(9).integers(0, 256, 4096, dtype=np.uint8).tobytes()
c = 0
t0 = time.time()
while time.time() - t0 < 100:
    if c != 7:
        sock.sendto(c.to_bytes(8, "little") + pay, ("127.0.0.1", 29957))
    c += 1
    time.sleep(0.0005)
EOF
SENDER=$!

timeout 110 ./bin/srtb-backend \
  --log_level 4 \
  --baseband_format_type fastmb_roach2 \
  --baseband_input_count 65536 --baseband_input_bits 8 \
  --spectrum_channel_count 32 \
  --baseband_freq_low 1400 --baseband_bandwidth 64 \
  --baseband_sample_rate 128e6 --dm 0.5 --baseband_reserve_sample 1 \
  --mitigate_rfi_average_method_threshold 1e30 \
  --mitigate_rfi_spectral_kurtosis_threshold 1e30 \
  --signal_detect_signal_noise_threshold 1e30 \
  --udp_receiver_address 127.0.0.1 --udp_receiver_port $PORT \
  --baseband_output_file_prefix gpurun_out/ud_ \
  --max-blocks 2 > $OUT 2>&1
echo "backend rc=$?" >> $OUT
kill $SENDER 2>/dev/null
wait $SENDER 2>/dev/null
tail -50 $OUT
