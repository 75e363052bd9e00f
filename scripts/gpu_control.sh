#!/bin/bash
# Fault-injection helper for MegaScan demos: down/up-clock one MI355X so the
# slow-rank detector has something to find.  MI355X equivalent of the
# reference's nvidia-smi -lgc script (scripts/gpu_control.sh).
#
# usage: gpu_control.sh slow <gpu_id> [sclk_level]   # cap core clock
#        gpu_control.sh reset <gpu_id>               # restore defaults
set -e
CMD=${1:?"usage: gpu_control.sh slow|reset <gpu_id> [level]"}
GPU=${2:?"gpu id required"}
case "$CMD" in
  slow)
    LEVEL=${3:-0}   # perf level 0 = lowest sclk
    rocm-smi -d "$GPU" --setperflevel manual
    rocm-smi -d "$GPU" --setsclk "$LEVEL"
    echo "GPU $GPU capped to sclk level $LEVEL"
    ;;
  reset)
    rocm-smi -d "$GPU" --setperflevel auto
    echo "GPU $GPU restored to auto perf level"
    ;;
  *)
    echo "unknown command $CMD" >&2; exit 1
    ;;
esac
