"""Micro-benchmark: custom HIP wgrad/dgrad/fwd kernels vs rocBLAS matmul
on the exact flagship shapes (batch 2048, fm1=65, fm2=120, fs=5, fc=390)."""
import torch, time
from noisynet_amd.ops import _ext
E = _ext.ext()

dev = 'cuda'
dt = torch.bfloat16
B = 2048


def t(fn, n=50, warm=10):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


# conv wgrad im2col shapes: conv1 M=2048*28*28, cols=5*5*8=200; K=65
# conv2 M=2048*10*10, cols=5*5*120->per-tap pad 120 ->cols 25*120=3000; K=120
shapes = [
    ('conv1-wgrad', B * 28 * 28, 65, 200),
    ('conv2-wgrad', B * 10 * 10, 120, 3000),
    ('fc1-wgrad', B, 390, 3000),
    ('fc2-wgrad', B, 10, 390),
]
for name, M, K, C in shapes:
    g = torch.randn(M, K, device=dev, dtype=dt)
    x = torch.randn(M, C, device=dev, dtype=dt)
    ours = t(lambda: E.linear_wgrad(g, x))
    blas = t(lambda: g.t().matmul(x))
    r = torch.allclose(E.linear_wgrad(g, x).float(), g.t().matmul(x).float(),
                       rtol=1e-2, atol=2.0)
    print(f'{name:<14} M={M:>9} K={K:>4} C={C:>4}  ours {ours:7.3f} ms  rocBLAS {blas:7.3f} ms  match={r}')

# linear fwd/dgrad shapes
for name, M, O, I in [('fc1-fwd', B, 390, 3000), ('fc2-fwd', B, 10, 390)]:
    x = torch.randn(M, I, device=dev, dtype=dt)
    w = torch.randn(O, I, device=dev, dtype=dt)
    g = torch.randn(M, O, device=dev, dtype=dt)
    ours_f = t(lambda: E.linear_fwd(x, w))
    blas_f = t(lambda: x.matmul(w.t()))
    ours_d = t(lambda: E.linear_dgrad(g, w))
    blas_d = t(lambda: g.matmul(w))
    print(f'{name:<14} fwd ours {ours_f:7.3f} vs blas {blas_f:7.3f} | dgrad ours {ours_d:7.3f} vs blas {blas_d:7.3f}')

# im2col materialize cost for conv1/conv2
x1 = torch.randn(B, 3, 32, 32, device=dev, dtype=dt).contiguous(memory_format=torch.channels_last)
x2 = torch.randn(B, 65, 14, 14, device=dev, dtype=dt).contiguous(memory_format=torch.channels_last)
print('im2col conv1', t(lambda: E.im2col_materialize(x1, 65, 1, 0, 5, 5)), 'ms')
print('im2col conv2', t(lambda: E.im2col_materialize(x2, 120, 1, 0, 5, 5)), 'ms')
