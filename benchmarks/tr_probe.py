"""Resolve ds_read_b64_tr_b16 lane->element semantics on hardware.

LDS is filled with bf16 value == element index.  For several per-lane
address patterns, print what each lane receives.  (guide T10's wording
is ambiguous between a per-lane gather and a cross-lane exchange.)
"""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from scaffold.ops.backend import require_ext


def run(name, off):
    m = require_ext()
    out = m.probe_tr(torch.tensor(off, dtype=torch.int32, device="cuda"))
    torch.cuda.synchronize()
    o = out.cpu().to(torch.int64)
    print(f"--- {name}")
    for l in range(0, 64, 1):
        if l in (0, 1, 2, 3, 4, 5, 15, 16, 17, 31, 32, 33, 48):
            print(f"lane {l:2d} addr {off[l]:4d}: {o[l].tolist()}")


def main():
    # pattern A: my gather model — addr = 2*(l&15) + 128*(l>>4)
    run("A: 2*(l&15)+128*(l>>4)", [2 * (l & 15) + 128 * (l >> 4) for l in range(64)])
    # pattern B: row-piece model — addr = 8*(l&15) + 128*(l>>4)
    run("B: 8*(l&15)+128*(l>>4)", [8 * (l & 15) + 128 * (l >> 4) for l in range(64)])
    # pattern C: uniform base 0
    run("C: uniform 0", [0] * 64)
    # pattern D: uniform 8 (one 8B piece in)
    run("D: uniform 8", [8] * 64)
    # pattern E: lane*8 linear across whole wave
    run("E: 8*l", [8 * l for l in range(64)])


if __name__ == "__main__":
    main()
