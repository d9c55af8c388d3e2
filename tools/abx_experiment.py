import sys, torch
sys.path.insert(0, '/root/repo')
from noisynet_amd import utils, ops
from noisynet_amd import optim as native_optim
from noisynet_amd.config import broadcast_per_layer, build_noisynet_parser
from noisynet_amd.models.noisynet import Net
from noisynet_amd.quant import start_calibration, finish_calibration

dev = torch.device('cuda')
torch.manual_seed(0)
X = torch.rand(256, 3, 32, 32, device=dev)
y = torch.randint(0, 10, (256,), device=dev)

def build(cl):
    argv = ['--q_a','4','--act_max','5','--LR','0.005','--batch_size','64',
            '--stochastic','0','--calculate_running','--no-augment']
    args = build_noisynet_parser().parse_args(argv); broadcast_per_layer(args)
    torch.manual_seed(11)
    m = Net(args); utils.init_model(m, args); m = m.to(dev)
    if cl: m = m.to(memory_format=torch.channels_last)
    opt = native_optim.AdamW(m.parameters(), lr=0.005)
    start_calibration(m)
    with torch.no_grad():
        for i in range(6): m(X[:64], 0, i)
    finish_calibration(m, dev)
    return m, opt

mn, on = build(False)
mc, oc = build(True)
for step in range(4):
    s = step * 64
    for m, opt in ((mn, on), (mc, oc)):
        out = m(X[s:s+64], 0, 1000 + step)
        loss = ops.cross_entropy(out, y[s:s+64])
        opt.zero_grad(set_to_none=False); loss.backward(); opt.step()
    print(f"after step {step}:")
    for (n, pn), (_, pc) in zip(mn.named_parameters(), mc.named_parameters()):
        d = (pn.detach().float().contiguous() - pc.detach().float().contiguous().view_as(pn)).abs().max().item()
        if d > 1e-7:
            print(f"  {n} diff {d:.3e}")
            # compare grads and moments for the first divergent param
            gd = (pn.grad.float().contiguous() - pc.grad.float().contiguous().view_as(pn)).abs().max().item()
            stn = on.state[pn]; stc = oc.state[pc]
            md = (stn['exp_avg'].float().contiguous().view(-1) - stc['exp_avg'].float().contiguous().view(-1)).abs().max().item()
            # careful: flat raw order differs between layouts; compare via logical
            md_l = (stn['exp_avg'].float().contiguous() - stc['exp_avg'].float().contiguous(memory_format=torch.contiguous_format).view_as(stn['exp_avg'])).abs().max().item()
            vd_l = (stn['exp_avg_sq'].float().contiguous() - stc['exp_avg_sq'].float().contiguous(memory_format=torch.contiguous_format).view_as(stn['exp_avg_sq'])).abs().max().item()
            print(f"    grad diff {gd:.3e} exp_avg(logical) {md_l:.3e} exp_avg_sq(logical) {vd_l:.3e}")
            break
