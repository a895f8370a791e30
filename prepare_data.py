#!/usr/bin/env python3
"""Dataset preparation (parity with reference prepare_data.py).

The reference downloads FashionMNIST/CIFAR-10/CIFAR-100 into ./data.
This environment is offline, and every benchmarked config runs on
synthetic data of the real datasets' shapes (BASELINE.json); the
synthetic generators need no preparation.  When a network IS available
this script fetches the real datasets so users can swap them in.
"""

def main():
    try:
        from torchvision import datasets

        for ctor in (datasets.FashionMNIST, datasets.CIFAR10,
                     datasets.CIFAR100):
            for train in (True, False):
                ctor("./data", train=train, download=True)
        print("real datasets downloaded to ./data")
    except Exception as e:  # offline: synthetic generators need nothing
        print(f"download unavailable ({e}); synthetic datasets "
              "(dynamic_load_balance_distributeddnn_amd.data) need no "
              "preparation")


if __name__ == "__main__":
    main()
