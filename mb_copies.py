"""Attribute aten::copy_ calls in one flagship training step to Python stacks."""
import torch
import bench as bench_mod
from noisynet_amd.models.noisynet import Net
from noisynet_amd import optim as native_optim, utils
from noisynet_amd.quant import start_calibration, finish_calibration

b = bench_mod.parse_args()
b.batch = 2048
args = bench_mod.flagship_args(b)
model = Net(args)
utils.init_model(model, args)
model = model.to('cuda').bfloat16()
for m in model.modules():
    if isinstance(m, (torch.nn.BatchNorm1d, torch.nn.BatchNorm2d)):
        m.float()
model = model.to(memory_format=torch.channels_last)
data = (torch.randint(0, 16, (b.batch, 3, 32, 32)).cuda().to(torch.bfloat16) / 15.0)
data = data.contiguous(memory_format=torch.channels_last)
labels = torch.randint(0, 10, (b.batch,)).cuda()
opt = native_optim.SGD(model.parameters(), lr=0.01, momentum=0.9, nesterov=True)
crit = torch.nn.CrossEntropyLoss()

start_calibration(model)
with torch.no_grad():
    for i in range(5):
        model(data, 0, i)
finish_calibration(model, 'cuda')


def step(i):
    model.train()
    out = model(data, 0, i)
    loss = crit(out.float(), labels)
    opt.zero_grad(set_to_none=False)
    loss.backward()
    opt.step()


for i in range(5):
    step(i + 100)
torch.cuda.synchronize()

from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             with_stack=True) as prof:
    for i in range(2):
        step(i + 1000)
    torch.cuda.synchronize()

evs = prof.key_averages(group_by_stack_n=7)
rows = [e for e in evs if 'copy_' in e.key or 'contiguous' in e.key or 'clone' in e.key]
rows.sort(key=lambda e: -e.self_device_time_total)
for e in rows[:12]:
    print('=' * 80)
    print(f"{e.key}  calls={e.count}  device={e.self_device_time_total/1000:.3f}ms")
    if e.stack:
        for line in e.stack[:7]:
            print('   ', line)
