#!/usr/bin/env python3
"""Generates crc64 golden vectors from the REFERENCE's own crc.cpp compiled
standalone (oracle/_ref/libcrc_ref.so — built by oracle/Makefile from
/root/reference/src/utils/crc.cpp, reference algorithm crc.cpp:45-88,289-481).

Run in the container that has the reference tree; the committed JSON travels
to GPU boxes where the reference is absent.
    python tests/golden/gen_crc64_golden.py > tests/golden/crc64_golden.json
"""
import ctypes
import json
import os
import random

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
ref = ctypes.CDLL(os.path.join(REPO, "oracle", "_ref", "libcrc_ref.so"))
f = getattr(ref, "_ZN3dsn5utils10crc64_calcEPKvmm")  # dsn::utils::crc64_calc
f.restype = ctypes.c_uint64
f.argtypes = [ctypes.c_char_p, ctypes.c_size_t, ctypes.c_uint64]

rnd = random.Random(20260915)
cases = []
fixed = [b"", b"a", b"123456789", b"hashkey", b"sortkey", b"u:00000000000042",
         bytes(range(256)), b"\x00" * 32, b"\xff" * 32]
for b in fixed:
    cases.append({"data": b.hex(), "init": 0, "crc": f(b, len(b), 0)})
for _ in range(120):
    n = rnd.randrange(0, 128)
    b = bytes(rnd.randrange(256) for _ in range(n))
    init = rnd.choice([0, 0xFFFFFFFFFFFFFFFF, rnd.getrandbits(64)])
    cases.append({"data": b.hex(), "init": init, "crc": f(b, n, init)})
print(json.dumps({"source": "reference src/utils/crc.cpp compiled standalone (oracle/_ref)",
                  "cases": cases}, indent=0))
