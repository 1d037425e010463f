#!/usr/bin/env python3
"""Derive the ds_read_b64_tr_b16 lane->element mapping empirically."""
import torch

import realhf_amd._C as C

# Each lane passes the element address of an 8-byte (4 x bf16) chunk.
# Candidate: lane l -> chunk l (linear): addr_elem = 4*l
for name, addr in [
    ("linear 4*l", [4 * l for l in range(64)]),
    ("group-local 4*(l&15)", [4 * (l & 15) for l in range(64)]),
    ("uniform 0", [0] * 64),
]:
    out = C.tr16_probe(torch.tensor(addr, dtype=torch.int32))
    print(f"--- {name}")
    for l in [0, 1, 2, 15, 16, 17, 31, 32, 48, 63]:
        print(f"lane {l:2d}: {[int(v) for v in out[l].tolist()]}")
