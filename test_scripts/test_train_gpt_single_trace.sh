#!/bin/bash
# MegaScan smoke (reference test_scripts/test_train_gpt_single_trace.sh)
bash examples/gpt3/train_gpt3_345m_distributed.sh --train-iters 12
python scripts/aggregate.py --trace-dir trace_output --detect
