#!/usr/bin/env python3
"""Process entry — `python main.py --mode train_test ...` or
`torchrun --nnodes 1 --nproc_per_node N main.py ...`
(CLI parity with the reference's main.py)."""

from seist_amd.cli import main

if __name__ == "__main__":
    main()
