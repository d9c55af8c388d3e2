"""Find which ops copy big activations in one flagship step (TorchDispatchMode)."""
import traceback
from collections import Counter

import torch
from torch.utils._python_dispatch import TorchDispatchMode

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


for i in range(3):
    step(i + 100)
torch.cuda.synchronize()

log = Counter()


class CopySpy(TorchDispatchMode):
    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        kwargs = kwargs or {}
        name = str(func)
        if any(s in name for s in ('copy_', 'clone', '_to_copy', 'cat', 'mul', 'add', 'fill', 'zero')):
            tens = [a for a in args if isinstance(a, torch.Tensor)]
            big = max((a.numel() for a in tens), default=0)
            if big >= (1 << 19):
                shp = tuple(tuple(a.shape) for a in tens[:2])
                st = traceback.extract_stack()
                src = '<-'.join(
                    f"{f.filename.split('/')[-1]}:{f.lineno}" for f in reversed(st)
                    if ('noisynet_amd' in f.filename or 'bench' in f.filename
                        or 'mb_copies' in f.filename))[:140] or 'backward-thread'
                log[(name, shp, src)] += 1
        return func(*args, **kwargs)


with CopySpy():
    step(1000)
torch.cuda.synchronize()

for (name, shp, src), cnt in log.most_common(30):
    print(f"x{cnt:<3} {name:<28} {str(shp):<50} {src}")
