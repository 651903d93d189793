#!/bin/bash
# ASan build + exercise of the native core (C++ sanitizer jobs).
# Builds csrc/core.cpp with -fsanitize=address into a scratch module and runs
# the shm-ring + codec exercises under it. Any leak/overflow aborts.
set -e
cd "$(dirname "$0")/.."
SCRATCH=$(mktemp -d)
python - <<PY
import os, subprocess, sys, sysconfig
import pybind11
out = "$SCRATCH/_core" + sysconfig.get_config_var("EXT_SUFFIX")
cmd = [
    "g++", "-O1", "-g", "-std=c++20", "-shared", "-fPIC",
    "-fsanitize=address", "-fno-omit-frame-pointer",
    f"-I{pybind11.get_include()}",
    f"-I{sysconfig.get_paths()['include']}",
    "csrc/core.cpp", "-o", out,
]
subprocess.run(cmd, check=True)
print("built", out)
PY
ASAN_LIB=$(gcc -print-file-name=libasan.so)
LD_PRELOAD=$ASAN_LIB ASAN_OPTIONS=detect_leaks=0 python - <<PY
import sys, os
sys.path.insert(0, "$SCRATCH")
import importlib, importlib.util, sysconfig
spec = importlib.util.spec_from_file_location(
    "_core", "$SCRATCH/_core" + sysconfig.get_config_var("EXT_SUFFIX"))
core = importlib.util.module_from_spec(spec)
spec.loader.exec_module(core)
# shm ring: push/pop across wrap boundaries
ring_path = "$SCRATCH/ring"
a = core.ShmRing(ring_path, 1 << 16, True)
b = core.ShmRing(ring_path, 1 << 16, False)
import os as _os
for i in range(2000):
    payload = bytes([i % 251]) * (17 + (i * 37) % 4000)
    assert a.push(payload)
    got = b.pop_all()
    assert got == [payload], i
# codec: random + repetitive + cross-sized
import random
rnd = random.Random(7)
for n in (0, 1, 4095, 4096, 4097, 100_000, 1_000_003):
    data = bytes(rnd.randrange(5) for _ in range(min(n, 50_000))) * (n // min(n, 50_000) if n else 1)
    data = data[:n]
    blob = core.malz_compress(data, 0.99)
    if blob is not None:
        assert core.malz_decompress(blob) == data
# pack payloads round trip
bufs = [_os.urandom(rnd.randrange(1, 5000)) for _ in range(64)]
packed = core.pack_payloads(bufs)
assert core.unpack_payloads(packed) == bufs
print("ASAN CHECK OK")
PY
rm -rf "$SCRATCH"
