#!/bin/bash
# Build and run the sanitizer harnesses (TSan + ASan/UBSan) over the
# host-side native layers: reliable transport, codec, planner/executor.
# These catch what the pytest tier cannot (races, OOB, UB) because the
# torch extension cannot easily be sanitizer-instrumented.
set -e
cd "$(dirname "$0")/.."
OUT=${TMPDIR:-/tmp}
for san in thread address,undefined; do
  tag=${san%%,*}
  # TSan builds define UCCL_SAN_NO_TIMED_WAIT everywhere: this libtsan
  # lacks pthread_cond_clockwait interception, so any cv wait_for reports
  # phantom double-locks/races (minimal repro verified)
  tw=""
  [ "$tag" = thread ] && tw=-DUCCL_SAN_NO_TIMED_WAIT
  g++ -O1 -g -std=c++17 -fsanitize=$san $tw tools/san_transport.cpp \
      uccl_amd/csrc/transport/reliable.cpp \
      uccl_amd/csrc/transport/udp_fabric.cpp \
      uccl_amd/csrc/transport/verbs_fabric.cpp \
      uccl_amd/csrc/core/trace.cpp \
      -o "$OUT/san_tp_$tag" -pthread -ldl
  g++ -O1 -g -std=c++17 -fsanitize=$san tools/san_codec_ukernel.cpp \
      uccl_amd/csrc/p2p/compress.cpp uccl_amd/csrc/ukernel/ukernel.cpp \
      uccl_amd/csrc/core/trace.cpp -o "$OUT/san_cu_$tag" -pthread -lz
  g++ -O1 -g -std=c++17 -fsanitize=$san $tw -D__HIP_PLATFORM_AMD__=1 \
      -I/opt/rocm/include tools/san_p2p.cpp \
      uccl_amd/csrc/p2p/endpoint.cpp uccl_amd/csrc/p2p/rccl_plane.cpp \
      uccl_amd/csrc/transport/reliable.cpp \
      uccl_amd/csrc/transport/udp_fabric.cpp \
      uccl_amd/csrc/transport/verbs_fabric.cpp \
      uccl_amd/csrc/core/trace.cpp -o "$OUT/san_p2p_$tag" -pthread -ldl \
      -L/opt/rocm/lib -lamdhip64 -Wl,-rpath,/opt/rocm/lib
  # net plugin: dlopen harness against a sanitizer-built .so. The TSan
  # build swaps timed cv waits for untimed ones (UCCL_SAN_NO_TIMED_WAIT):
  # this libtsan lacks pthread_cond_clockwait interception, so wait_for
  # reports false double-lock/races (verified with a minimal repro).
  extra="$tw"
  g++ -O1 -g -std=c++17 -fsanitize=$san $extra -fPIC -shared \
      uccl_amd/csrc/plugin/tcp_plugin.cpp \
      uccl_amd/csrc/transport/reliable.cpp \
      uccl_amd/csrc/transport/udp_fabric.cpp \
      uccl_amd/csrc/transport/verbs_fabric.cpp \
      uccl_amd/csrc/core/trace.cpp \
      -o "$OUT/librccl-net-uccl-$tag.so" -pthread -ldl
  g++ -O1 -g -std=c++17 -fsanitize=$san \
      uccl_amd/csrc/plugin/plugin_test_main.cpp \
      -o "$OUT/plugin_test_$tag" -ldl -pthread
  echo "== $tag: transport =="; "$OUT/san_tp_$tag"
  echo "== $tag: codec+ukernel =="; "$OUT/san_cu_$tag"
  echo "== $tag: p2p endpoint =="; "$OUT/san_p2p_$tag"
  echo "== $tag: plugin (multipath) =="
  "$OUT/plugin_test_$tag" "$OUT/librccl-net-uccl-$tag.so" | tail -1
  echo "== $tag: plugin (tcp) =="
  UCCL_NET_TRANSPORT=tcp "$OUT/plugin_test_$tag" \
      "$OUT/librccl-net-uccl-$tag.so" | tail -1
done
echo "ALL SANITIZERS CLEAN"
