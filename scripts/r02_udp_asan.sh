#!/bin/bash
# Build srtb-backend with host AddressSanitizer + line info and reproduce
# the UDP segfault under it.
set -x
cd /root/repo
mkdir -p gpurun_out
OBJ=build/hip_obj
/opt/rocm/bin/hipcc --offload-arch=gfx950 -O1 -g -std=c++17 -fPIC \
  -fsanitize=address -fno-omit-frame-pointer -x hip csrc/app/srtb_backend.cpp \
  -x none $OBJ/unpack.o $OBJ/spectrum.o $OBJ/display.o $OBJ/fft.o $OBJ/engine.o \
  -Icsrc/include -L/opt/rocm/lib -lhipfft -lroctx64 -lrccl \
  -o gpurun_out/srtb-backend-asan 2> gpurun_out/asan_build.log || { tail -20 gpurun_out/asan_build.log; exit 1; }

python3 - <<'PY' &
import socket, time, numpy as np
sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
pay = np.random.default_rng(9).integers(0, 256, 4096, dtype=np.uint8).tobytes()
c = 0
t0 = time.time()
while time.time() - t0 < 160:
    if c != 7:
        sock.sendto(c.to_bytes(8, "little") + pay, ("127.0.0.1", 29958))
    c += 1
    time.sleep(0.0005)
PY
SENDER=$!

ASAN_OPTIONS=detect_leaks=0 timeout 170 gpurun_out/srtb-backend-asan \
  --log_level 4 \
  --baseband_format_type fastmb_roach2 \
  --baseband_input_count 65536 --baseband_input_bits 8 \
  --spectrum_channel_count 32 \
  --baseband_freq_low 1400 --baseband_bandwidth 64 \
  --baseband_sample_rate 128e6 --dm 0.5 --baseband_reserve_sample 1 \
  --mitigate_rfi_average_method_threshold 1e30 \
  --mitigate_rfi_spectral_kurtosis_threshold 1e30 \
  --signal_detect_signal_noise_threshold 1e30 \
  --udp_receiver_address 127.0.0.1 --udp_receiver_port 29958 \
  --baseband_output_file_prefix gpurun_out/ua_ \
  --max-blocks 2 > gpurun_out/udp_asan.log 2>&1
echo "rc=$?" >> gpurun_out/udp_asan.log
kill $SENDER 2>/dev/null; wait $SENDER 2>/dev/null
tail -60 gpurun_out/udp_asan.log
