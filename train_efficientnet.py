#!/usr/bin/env python3
"""timm-style EfficientNet training entrypoint (flag-compatible with the
reference train_efficientnet.py; implementation in
noisynet_amd/drivers/efficientnet_train.py)."""

from noisynet_amd.drivers.efficientnet_train import main

if __name__ == '__main__':
    main()
