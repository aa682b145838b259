#!/bin/bash
# podworker startup-latency experiment (runs on the GPU box via gpurun).
#
# Question: why does the pod-Ready p50 sit at ~300 ms in bench waves when an
# isolated podworker (smoke) reaches Ready in ~90 ms? Hypothesis: a fresh HIP
# context init serializes against the *previous* pod process's KFD/VRAM
# teardown when launched back-to-back on the same GPU.
#
# Writes gpurun_out/pw_timing.txt with per-run wall ms for:
#   a) back-to-back runs (exit -> immediately start next)
#   b) 1 s gap between runs (teardown fully drained)
#   c) overlapped: next run starts while a holder process is being SIGTERMed
set -u
PW=k8s_runpod_kubelet_amd/ops/podworker/podworker
OUT=gpurun_out/pw_timing.txt
mkdir -p gpurun_out
: > "$OUT"

now_ms() { echo $(( $(date +%s%N) / 1000000 )); }

run_once() {
  local t0 t1
  t0=$(now_ms)
  ROCR_VISIBLE_DEVICES=0 timeout 30 "$PW" --expect-gpus 1 --run-for 0 >/dev/null 2>&1
  t1=$(now_ms)
  echo $(( t1 - t0 ))
}

echo "# warmup (cold page cache)" >> "$OUT"
run_once >> "$OUT"
run_once >> "$OUT"

echo "# a) back-to-back x10" >> "$OUT"
for i in $(seq 1 10); do run_once >> "$OUT"; done

echo "# b) 1s gap x6" >> "$OUT"
for i in $(seq 1 6); do sleep 1; run_once >> "$OUT"; done

echo "# c) start while previous is terminating x6" >> "$OUT"
for i in $(seq 1 6); do
  ROCR_VISIBLE_DEVICES=0 timeout 30 "$PW" --expect-gpus 1 --hold >/dev/null 2>&1 &
  HOLDER=$!
  sleep 0.4   # holder reaches Ready
  kill -TERM "$HOLDER" 2>/dev/null
  run_once >> "$OUT"
  wait "$HOLDER" 2>/dev/null
done

echo "# d) two concurrent on same GPU x4 (second while first inits)" >> "$OUT"
for i in $(seq 1 4); do
  ROCR_VISIBLE_DEVICES=0 timeout 30 "$PW" --expect-gpus 1 --run-for 0 >/dev/null 2>&1 &
  run_once >> "$OUT"
  wait
done
