#!/usr/bin/env python3
"""torch.profiler over one bench step: top ops by device time with shapes
(identifies eager torch kernels that rocprof can't attribute)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    model_name = sys.argv[1] if len(sys.argv) > 1 else 'efficientnet_b0'
    import bench as bench_mod
    sys.argv = ['bench.py', '--model', model_name, '--steps', '3',
                '--warmup', '1', '--no-graph']
    bench = bench_mod.parse_args()
    device = torch.device('cuda')
    torch.manual_seed(0)
    model, args, image_size, num_classes = bench_mod.build_secondary(
        bench, device, torch.bfloat16) if model_name != 'noisynet' else (None,) * 4
    from noisynet_amd import ops
    from noisynet_amd import optim as native_optim
    from noisynet_amd.quant import finish_calibration, start_calibration
    opt = native_optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    x = torch.rand(bench.batch, 3, image_size, image_size, device=device,
                   dtype=torch.bfloat16).contiguous(
                       memory_format=torch.channels_last)
    y = torch.randint(0, num_classes, (bench.batch,), device=device)

    def step(i):
        out = model(x, 0, i) if model_name != 'efficientnet_b0' else model(x)
        loss = ops.cross_entropy(out, y)
        opt.zero_grad(set_to_none=False)
        loss.backward()
        opt.step()

    start_calibration(model)
    with torch.no_grad():
        for i in range(5):
            model(x, 0, i) if model_name != 'efficientnet_b0' else model(x)
    finish_calibration(model, device)
    model.train()
    for i in range(3):
        step(100 + i)
    torch.cuda.synchronize()
    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CUDA, ProfilerActivity.CPU],
                 record_shapes=True) as prof:
        step(1000)
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_input_shape=True).table(
        sort_by='cuda_time_total', row_limit=25, max_src_column_width=60))


if __name__ == '__main__':
    main()
