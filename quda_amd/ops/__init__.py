"""Kernel launchers + CPU oracle implementations.

Every op exists twice:
  reference.py  — plain-PyTorch complex-tensor oracle (fp64/fp32), used on CPU
                  and as the ground truth for GPU numerics tests
  dispatch.py   — dispatches to the HIP extension (quda_amd_hip) on GPU
                  tensors, to the oracle otherwise; fails loudly if a GPU
                  tensor arrives and the extension is not built
"""
