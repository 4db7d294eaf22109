"""One-shot hardware probe for ds_read_b64_tr_b16 lane->element mapping.

Fills LDS with element indices (lds[i] = i) and dumps what each lane's two
tr_reads (offset:0 and offset:128) deliver for three addressing modes:
  mode 0: addr = base + 8*lane      (v4 kernel's pattern — 4 tiles/wave)
  mode 1: addr = base               (uniform)
  mode 2: addr = base + 8*(lane&15) (one tile, all four groups)

Expected under the guide's m162 mapping (lane l, elem j reads
lds[(l&15) + j*16 + (l>>4)*64] for mode 0): out[l][j] = (l&15) + j*16 +
(l>>4)*64 and the offset:128 read = same + 64 elements.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ctypes
import torch
from kubeflow_amd.ops import _backend


def main():
    lib = _backend.require()
    lib.kf_tr16_probe.restype = ctypes.c_int
    dev = torch.device("cuda", 0)
    inp = torch.arange(1024, dtype=torch.int16, device=dev)
    for mode in (0, 1, 2):
        out = torch.full((64, 8), -1, dtype=torch.int16, device=dev)
        rc = lib.kf_tr16_probe(
            ctypes.c_void_p(out.data_ptr()), ctypes.c_void_p(inp.data_ptr()),
            ctypes.c_int(mode),
            ctypes.c_void_p(torch.cuda.current_stream().cuda_stream))
        torch.cuda.synchronize()
        a = out.cpu().numpy()
        print(f"--- mode {mode} (rc={rc})")
        for lane in (0, 1, 5, 15, 16, 17, 31, 32, 47, 48, 63):
            print(f"lane {lane:2d}: r0={list(a[lane][:4])} r1={list(a[lane][4:])}")
        if mode == 0:
            ok = all(
                a[l][j] == (l & 15) + j * 16 + (l >> 4) * 64
                and a[l][4 + j] == (l & 15) + j * 16 + (l >> 4) * 64 + 64
                for l in range(64) for j in range(4))
            print(f"m162 mapping (addr=base+8*lane) holds: {ok}")


if __name__ == "__main__":
    main()
