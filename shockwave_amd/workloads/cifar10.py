"""ResNet-18 / CIFAR-10 workload entry (flagship job family).

Thin module wrapper over the shared implementation in
``families.cifar10_main`` (generic lease loop + Accordion/GNS adaptation).
"""

from .families import cifar10_main as main

if __name__ == "__main__":
    main()
