#!/usr/bin/env python3
"""CIFAR-10 NoisyNet training entrypoint (flag-compatible with the reference
noisynet.py; see noisynet_amd/drivers/cifar.py for the implementation).

Examples (reference README.md:6-13):
  python noisynet.py --current 1 --act_max 5 --w_max1 0.3 --LR 0.005 \
      --L2_1 0.0005 --L2_2 0.0002
  python noisynet.py --L2 0.0005 --dropout 0.1 --nepochs 450
"""

from noisynet_amd.drivers.cifar import main

if __name__ == '__main__':
    main()
