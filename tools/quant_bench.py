#!/usr/bin/env python3
"""Quant pack/unpack kernel throughput microbench."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), '..'))
import torch

def main():
    from adaqp_amd.ops.kernels import native
    from adaqp_amd.comm.buffers import bytes_per_node
    C = native()
    n, F = 200_000, 256
    for dtype in (torch.float32, torch.bfloat16):
        x = torch.randn(n, F, device='cuda', dtype=dtype)
        rows = torch.arange(n, dtype=torch.int64, device='cuda')
        for bits in (2, 4, 8):
            bpn = bytes_per_node(F, bits)
            off = rows * bpn
            payload = torch.zeros(n * bpn, dtype=torch.uint8, device='cuda')
            params = torch.zeros(2 * n, dtype=torch.bfloat16, device='cuda')
            out = torch.zeros(n, F, device='cuda', dtype=dtype)
            for _ in range(3):
                C.quant_pack(x, rows, rows, off, bits, 3, payload, params)
            torch.cuda.synchronize(); t0 = time.perf_counter()
            for i in range(20):
                C.quant_pack(x, rows, rows, off, bits, i, payload, params)
            torch.cuda.synchronize(); tp = (time.perf_counter() - t0) / 20
            for _ in range(3):
                C.quant_unpack(payload, params, rows, rows, off, bits, F, out)
            torch.cuda.synchronize(); t0 = time.perf_counter()
            for _ in range(20):
                C.quant_unpack(payload, params, rows, rows, off, bits, F, out)
            torch.cuda.synchronize(); tu = (time.perf_counter() - t0) / 20
            inb = n * F * x.element_size()
            print(f'{str(dtype)[6:]} bits={bits}: pack {tp*1e6:.0f}us '
                  f'({2*inb/tp/1e12:.2f} TB/s rd, 2-pass) | unpack {tu*1e6:.0f}us '
                  f'({inb/tu/1e12:.2f} TB/s wr)')

if __name__ == '__main__':
    main()
