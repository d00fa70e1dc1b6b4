"""Empirical semantics probe for ds_read_b64_tr_b16 (gfx950 hardware
transpose read).  Fills LDS with identity values and dumps which 4 bf16
elements land in each lane for several addressing patterns — used to
derive the LDS image the fa backward's transposed operands need."""
import os
import subprocess
import sys

import torch

SRC = r'''
#include <hip/hip_runtime.h>
typedef int i32x2p __attribute__((ext_vector_type(2)));
extern "C" __global__ void tr_probe(short* out, const int* addrs) {
  __shared__ short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += 64) lds[i] = (short)i;
  __syncthreads();
  int lane = threadIdx.x;
  int addr = addrs[lane];
  i32x2p v;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr));
  short* pv = reinterpret_cast<short*>(&v);
  for (int j = 0; j < 4; j++) out[lane * 4 + j] = pv[j];
}
'''


def main():
    work = "/tmp/tr16"
    os.makedirs(work, exist_ok=True)
    with open(f"{work}/p.hip", "w") as f:
        f.write(SRC)
    subprocess.run(["hipcc", "--offload-arch=gfx950", "-O3", "--genco",
                    f"{work}/p.hip", "-o", f"{work}/p.hsaco"], check=True)
    import ctypes
    torch.zeros(1, device="cuda")  # init context
    hip = ctypes.CDLL("libamdhip64.so")
    mod = ctypes.c_void_p()
    assert hip.hipModuleLoad(ctypes.byref(mod), f"{work}/p.hsaco".encode()) == 0
    fn = ctypes.c_void_p()
    assert hip.hipModuleGetFunction(ctypes.byref(fn), mod, b"tr_probe") == 0

    def run(addrs):
        out = torch.zeros(256, dtype=torch.int16, device="cuda")
        a = torch.tensor(addrs, dtype=torch.int32, device="cuda")
        out_p = ctypes.c_void_p(out.data_ptr())
        a_p = ctypes.c_void_p(a.data_ptr())
        args = (ctypes.c_void_p * 2)(ctypes.cast(ctypes.byref(out_p), ctypes.c_void_p),
                                     ctypes.cast(ctypes.byref(a_p), ctypes.c_void_p))
        class KP(ctypes.Structure):
            pass
        argv = (ctypes.c_void_p * 2)(ctypes.addressof(out_p), ctypes.addressof(a_p))
        assert hip.hipModuleLaunchKernel(fn, 1, 1, 1, 64, 1, 1, 0, None,
                                         argv, None) == 0
        torch.cuda.synchronize()
        return out.view(64, 4).tolist()

    for name, addrs in [
        ("linear lane*8", [l * 8 for l in range(64)]),
        ("all-same 0", [0] * 64),
        ("row-major rows of 16B: lane*16", [l * 16 for l in range(64)]),
    ]:
        res = run(addrs)
        print(f"== {name}")
        for l in range(0, 64, 1):
            print(f"lane {l:2d} addr {addrs[l]:5d}: {res[l]}")


if __name__ == "__main__":
    main()
