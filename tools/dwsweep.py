#!/usr/bin/env python3
"""Sweep the conv dW tr-GEMM block-count target (DMNIST_DW_BLOCKS) on GPU."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from distributedmnist_amd import _C
from tools.kernbench import timeit

bf16 = torch.bfloat16
ext = _C.ext()
for B in [1024, 8192]:
    y1 = (torch.randn(B, 14, 14, 32) * 0.1).to("cuda", bf16)
    dact2 = (torch.randn(B, 14, 14, 64) * 0.1).to("cuda", bf16)
    dw = torch.zeros(5, 5, 32, 64, device="cuda")
    h2 = (torch.randn(B, 3136) * 0.1).to("cuda", bf16)
    dyeff1 = (torch.randn(B, 512) * 0.1).to("cuda", bf16)
    dwf = torch.zeros(3136, 512, device="cuda")
    print(f"## B={B}")
    print("| blocks | conv2 dW us | fc1 dW us |")
    print("|---|---|---|")
    for blocks in [1024, 2048, 4096, 8192, 16384]:
        os.environ["DMNIST_DW_BLOCKS"] = str(blocks)
        t1 = timeit(lambda: ext.conv_dw_into(y1, dact2, dw))
        t2 = timeit(lambda: ext.linear_dw_into(h2, dyeff1, dwf))
        print(f"| {blocks} | {t1:.1f} | {t2:.1f} |", flush=True)
    del os.environ["DMNIST_DW_BLOCKS"]
