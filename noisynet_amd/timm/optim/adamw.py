"""AdamW (decoupled weight decay) -- alias to the fused-kernel optimizer
(reference timm/optim/adamw.py is a plain PyTorch AdamW; ours runs one fused
HIP update kernel per parameter)."""

from ... import optim as native_optim

AdamW = native_optim.AdamW
