#!/usr/bin/env python3
"""MNIST chip MLP entrypoint (flag-compatible with the reference
chip_mnist.py; implementation in noisynet_amd/drivers/mnist.py)."""

from noisynet_amd.drivers.mnist import main

if __name__ == '__main__':
    main()
