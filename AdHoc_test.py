#!/usr/bin/env python3
"""Compatibility entry point (reference: ``src/AdHoc_test.py``)."""
from multihop_offload_amd.harness.adhoc_test import main

if __name__ == "__main__":
    main()
