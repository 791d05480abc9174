#!/usr/bin/env python3
"""Quant pack/unpack kernel throughput microbench.

Default: per-bit uniform shapes. ``--mixed``: realistic mixed-bit
boundary sets at reddit-4-part scale — ~100k boundary
nodes x F in {608, 256}, bits drawn {2,4,8} uniformly across 3 peers,
through the SAME SidePlan/mixed_quantize path the training step uses.
Reports kernel time, effective bandwidth, wire bytes vs the fp32
exchange volume replaced, and % of the measured reddit epoch."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), '..'))
import torch


def bench_mixed():
    from adaqp_amd.ops.kernels import mixed_quantize, mixed_dequantize
    from adaqp_amd.comm.buffers import _layout, BITS_SET
    g = torch.Generator().manual_seed(7)
    n_send = 100_000
    peers = 3                       # reddit 4-part: 3 channels per rank
    print('# mixed-bit boundary quant at reddit-4-part shape '
          f'({n_send} send nodes, {peers} peers, bits uniform over {BITS_SET})')
    for F in (608, 256):
        per = n_send // peers
        bits_pp, rows_pp = [], []
        base = 0
        for p in range(peers):
            bits = torch.tensor(BITS_SET)[torch.randint(0, 3, (per,), generator=g)]
            bits_pp.append(bits)
            rows_pp.append(torch.randperm(per * 2, generator=g)[:per] + base)
            base += per * 2
        plan = _layout(bits_pp, rows_pp, F).to('cuda')
        # recv side: same bit mix, but rows index the REMOTE block
        # (arange blocks, like build_key_plan's recv side) — using the
        # send rows here would scatter out of bounds
        rbase, rrows = 0, []
        for p in range(peers):
            rrows.append(torch.arange(rbase, rbase + per, dtype=torch.int64))
            rbase += per
        rplan = _layout(bits_pp, rrows, F).to('cuda')
        x = torch.randn(base, F, device='cuda')
        out = torch.zeros(rplan.total_nodes, F, device='cuda')
        for _ in range(3):
            payload, params = mixed_quantize(x, plan, 3)
            mixed_dequantize(payload, params, rplan, out)
        torch.cuda.synchronize(); t0 = time.perf_counter()
        iters = 20
        for i in range(iters):
            payload, params = mixed_quantize(x, plan, i)
        torch.cuda.synchronize(); tq = (time.perf_counter() - t0) / iters
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for i in range(iters):
            mixed_dequantize(payload, params, rplan, out)
        torch.cuda.synchronize(); td = (time.perf_counter() - t0) / iters
        fp32_bytes = plan.total_nodes * F * 4
        wire = plan.total_bytes + 4 * plan.total_nodes   # payload + bf16 params
        print(f'F={F}: pack {tq*1e6:.0f}us + unpack {td*1e6:.0f}us '
              f'(read {fp32_bytes/tq/1e12:.2f} TB/s); wire {wire/2**20:.1f} MiB '
              f'vs fp32 {fp32_bytes/2**20:.1f} MiB ({fp32_bytes/wire:.2f}x '
              f'compression); total {1e3*(tq+td):.3f} ms')


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
    if '--mixed' in sys.argv:
        bench_mixed()
    else:
        main()
