#!/usr/bin/env python3
"""Training entry point.

Same CLI surface as the reference (``/root/reference/main.py:6-15``):
dataset, num_parts (implied by WORLD_SIZE under torchrun), backend,
init_method, model_name, mode, assign_scheme, logger_level.

Launch (one process per GPU over RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
        --master-addr 127.0.0.1 main.py --dataset reddit --model_name gcn \
        --mode AdaQP
"""
import argparse

from adaqp_amd.runtime.trainer import Trainer


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--dataset', type=str, default='reddit',
                   choices=['reddit', 'yelp', 'ogbn-products', 'amazonProducts'])
    p.add_argument('--model_name', type=str, default='gcn',
                   choices=['gcn', 'sage'])
    p.add_argument('--mode', type=str, default='AdaQP',
                   choices=['Vanilla', 'AdaQP', 'AdaQP-q', 'AdaQP-p'])
    p.add_argument('--assign_scheme', type=str, default=None,
                   choices=[None, 'uniform', 'random', 'adaptive'])
    p.add_argument('--backend', type=str, default=None,
                   help='torch.distributed backend (default: RCCL on GPU, gloo on CPU)')
    p.add_argument('--init_method', type=str, default='env://')
    p.add_argument('--logger_level', type=str, default='INFO')
    p.add_argument('--num_parts', type=int, default=None,
                   help='expected partition count; validated against WORLD_SIZE')
    p.add_argument('--partition_dir', type=str, default='part_data')
    p.add_argument('--exp_dir', type=str, default='exp',
                   help='results root (metrics/time artifacts + logs)')
    p.add_argument('--scale', type=float, default=None,
                   help='synthetic graph scale; validated against the '
                        'partition cache (omit to adopt the cached scale)')
    p.add_argument('--num_epochs', type=int, default=None)
    p.add_argument('--lr', type=float, default=None)
    p.add_argument('--log_steps', type=int, default=None)
    p.add_argument('--eval_every', type=int, default=None,
                   help='evaluate every N epochs (default: config, 1)')
    p.add_argument('--seed', type=int, default=None)
    p.add_argument('--dtype', type=str, default='fp32', choices=['fp32', 'bf16'])
    p.add_argument('--ckpt_path', type=str, default=None)
    p.add_argument('--ckpt_every', type=int, default=None)
    p.add_argument('--resume', action='store_true')
    p.add_argument('--time_breakdown', action='store_true',
                   help='enable per-epoch comm/quant/agg span timing '
                        '(adds sync fences; use rocprofv3 for kernel evidence)')
    args = p.parse_args()

    # allow plain `python main.py` without torchrun (world_size 1)
    import os
    if os.environ.get('ADAQP_HANG_DEBUG'):
        # dump all thread stacks and exit if a run wedges for N seconds
        import faulthandler
        faulthandler.dump_traceback_later(
            int(os.environ['ADAQP_HANG_DEBUG']), exit=True)
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    os.environ.setdefault('MASTER_PORT', '29501')
    os.environ.setdefault('RANK', '0')
    os.environ.setdefault('WORLD_SIZE', '1')
    os.environ.setdefault('LOCAL_RANK', os.environ.get('RANK', '0'))

    ws = int(os.environ.get('WORLD_SIZE', '1'))
    if args.num_parts is not None and args.num_parts != ws:
        raise SystemExit(f'--num_parts {args.num_parts} != WORLD_SIZE {ws}: '
                         'launch with --nproc-per-node equal to num_parts')

    trainer = Trainer(args)
    start = 0
    if args.resume and args.ckpt_path and os.path.exists(args.ckpt_path):
        start = trainer.load_checkpoint(args.ckpt_path)
        print(f'resumed from {args.ckpt_path} at epoch {start}')
    best = trainer.train(start_epoch=start)
    trainer.save(root=args.exp_dir)
    if trainer.comm.rank == 0:
        print(f'best: epoch {best["epoch"]} val {best["val"]:.4f} '
              f'test {best["test"]:.4f}')


if __name__ == '__main__':
    main()
