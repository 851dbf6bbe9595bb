#!/usr/bin/env python3
"""Compatibility entry point (reference: ``src/data_generation_offloading.py``)."""
from multihop_offload_amd.datagen import main

if __name__ == "__main__":
    main()
