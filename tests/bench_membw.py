#!/usr/bin/env python3
"""HBM read-bandwidth probe (measurement infrastructure).

Pins the "achievable" read rate that the roofline `frac` is quoted
against: the same b128 grid-stride load pattern as the dense agg kernel's
input stream (rw_amd.hip `membw_probe_kernel`), on a buffer far larger
than L2. Prints one JSON line; not a pytest test and not a product path.
"""
import ctypes
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import risingwave_amd

risingwave_amd.load_library()
L = ctypes.CDLL(risingwave_amd.lib_path())
L.rw_membw_probe.restype = ctypes.c_int
L.rw_membw_probe.argtypes = [ctypes.c_uint64, ctypes.c_int,
                             ctypes.POINTER(ctypes.c_double)]

out = []
for gib in (1, 4, 16):
    g = ctypes.c_double(0.0)
    rc = L.rw_membw_probe(gib << 30, 8, ctypes.byref(g))
    if rc != 0:
        print(json.dumps({"error": rc, "GiB": gib}))
        sys.exit(1)
    out.append({"GiB": gib, "read_GBps": round(g.value, 1)})

# RANDOM 64-B line touches (the hash probe/insert pattern): the honest
# ceiling for q8's random accesses (rw_membw_rand_probe)
L.rw_membw_rand_probe.restype = ctypes.c_int
L.rw_membw_rand_probe.argtypes = [ctypes.c_uint64, ctypes.c_int,
                                  ctypes.POINTER(ctypes.c_double),
                                  ctypes.POINTER(ctypes.c_double)]
rand = []
for mib, name in ((128, "LIC-resident (128 MiB)"), (1024, "1 GiB"),
                  (8192, "8 GiB")):
    gl = ctypes.c_double(0.0)
    gb = ctypes.c_double(0.0)
    rc = L.rw_membw_rand_probe(mib << 20, 4096, ctypes.byref(gl),
                               ctypes.byref(gb))
    if rc != 0:
        print(json.dumps({"error": rc, "MiB": mib}))
        sys.exit(1)
    rand.append({"region": name, "Glines_per_s": round(gl.value, 2),
                 "line_GBps": round(gb.value, 1)})
L.rw_membw_rand_atomic.restype = ctypes.c_int
L.rw_membw_rand_atomic.argtypes = [ctypes.c_uint64, ctypes.c_int,
                                   ctypes.POINTER(ctypes.c_double)]
L.rw_membw_rand_chase.restype = ctypes.c_int
L.rw_membw_rand_chase.argtypes = [ctypes.c_uint64, ctypes.c_int,
                                  ctypes.c_int,
                                  ctypes.POINTER(ctypes.c_double)]
at = []
for mib in (128, 1024):
    g = ctypes.c_double(0.0)
    rc = L.rw_membw_rand_atomic(mib << 20, 2048, ctypes.byref(g))
    assert rc == 0
    at.append({"MiB": mib, "Gatomics_per_s": round(g.value, 2)})
ch = []
for mlp in (1, 2, 4, 8):
    g = ctypes.c_double(0.0)
    rc = L.rw_membw_rand_chase(1 << 30, 512, mlp, ctypes.byref(g))
    assert rc == 0
    ch.append({"mlp": mlp, "Glines_per_s": round(g.value, 2)})
print(json.dumps({"membw_probe": out, "membw_random_64B": rand,
                  "membw_random_atomic": at,
                  "membw_dependent_chase_1GiB": ch}))

# q8-shape ladder (see rw_q8shape_probe)
L.rw_q8shape_probe.restype = ctypes.c_int
L.rw_q8shape_probe.argtypes = [ctypes.c_uint32, ctypes.c_int,
                               ctypes.POINTER(ctypes.c_double)]
shape = []
for mode in (0, 8, 9, 25, 24 + 64, 24 + 64 + 2, 24 + 128, 24 + 128 + 2, 31, 24 + 128 + 2 + 4):
    us = ctypes.c_double(0.0)
    rc = L.rw_q8shape_probe(1 << 20, mode, ctypes.byref(us))
    assert rc == 0
    shape.append({"mode": mode, "us_per_1M_rows": round(us.value, 1)})
print(json.dumps({"q8_shape_ladder": shape}))
