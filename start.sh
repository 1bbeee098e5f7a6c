#!/bin/bash
# Single-process launch (parity with the reference's start.sh).
nohup python -u main.py --mode train_test > nohup.out 2>&1 &
echo "pid: $!"
