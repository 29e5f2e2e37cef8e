#!/bin/bash
# Serving launcher (reference server.sh runs tensorflow/serving:1.14.0 in
# docker on port 8500; here the native gRPC server serves the exported
# model with hipGraph-captured inference on the same port).
MODEL=${1:-bert_bilstm_crf}
PORT=${2:-8500}
exec python -m chinesener_amd.serve.server --model "$MODEL" --port "$PORT"
