#!/bin/bash
# Hyperparameter sweep job generator.
# Parity: /root/reference/demo/gpu-training/generate_job.sh:17-38 — stamps a
# parameterized Job manifest per (lr, batch) combination.
set -euo pipefail

LRS=${LRS:-"1e-4 3e-4 1e-3"}
BATCHES=${BATCHES:-"32 64 128"}

for lr in ${LRS}; do
  for batch in ${BATCHES}; do
    name="train-lr${lr//[.e-]/}-b${batch}"
    cat <<EOF
apiVersion: batch/v1
kind: Job
metadata:
  name: ${name}
spec:
  template:
    spec:
      restartPolicy: Never
      containers:
        - name: train
          image: cea-amd/gpu-device-plugin:latest
          command: ["python3", "/opt/cea-amd/demo/train.py",
                    "--lr", "${lr}", "--batch", "${batch}"]
          resources:
            limits:
              amd.com/gpu: 1
---
EOF
  done
done
