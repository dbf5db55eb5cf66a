#!/usr/bin/env python3
"""CLI entry point for the example pipelines (ref: bin/run-pipeline.sh +
the per-app mains of pipelines/{images,speech,text}).

Usage (single GPU/CPU):
    python bin/run_pipeline.py newsgroups --n-docs 2000
    python bin/run_pipeline.py timit --n 50000 --num-features 8192
    python bin/run_pipeline.py imagenet-fv --images 256 --gmm-k 64

Multi-GPU (one rank per GPU over RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 bin/run_pipeline.py timit ...
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from keystone_amd._cli import main

if __name__ == "__main__":
    main()
