#!/usr/bin/env python3
"""ImageNet training entrypoint (flag-compatible with the reference main.py;
implementation in noisynet_amd/drivers/imagenet.py).

Example: python main.py -a resnet18 --q_a 4 --calculate_running
"""

from noisynet_amd.drivers.imagenet import main

if __name__ == '__main__':
    main()
