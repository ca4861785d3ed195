#!/bin/bash
echo "== 16x16 body (CC_GEMM_WIDE=0)"; CC_GEMM_WIDE=0 python tools/gemm_body_ab.py 2>/dev/null | tail -6
echo "== 32x32 body (CC_GEMM_WIDE=1)"; CC_GEMM_WIDE=1 python tools/gemm_body_ab.py 2>/dev/null | tail -6
