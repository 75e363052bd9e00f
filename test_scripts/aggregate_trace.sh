#!/bin/bash
python scripts/aggregate.py --trace-dir ${1:-trace_output} --detect
