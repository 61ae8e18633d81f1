"""In-process multi-GPU context (vega_gpu_init(G>1)): one process drives all
devices over RCCL. Runs fully only on a multi-GPU box; on 1-GPU boxes it
verifies argument validation."""
import os
import sys

import numpy as np
import pytest

import oracle_ctypes as oc
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from vega_amd import datagen

pytestmark = pytest.mark.gpu


def test_init_validation():
    from vega_amd import gpu
    with pytest.raises(gpu.VegaGpuError):
        gpu.VegaContext(ngpus=99)


def test_multigpu_reduce_all_devices():
    import torch
    from vega_amd import gpu
    ndev = torch.cuda.device_count()
    if ndev < 2:
        pytest.skip("needs >= 2 GPUs")
    n = 4_000_000
    with gpu.VegaContext(ngpus=ndev) as ctx:
        rdd = ctx.gen_rdd_uniform(n, seed=91, key_bits=18)
        red = rdd.reduce_by_key(gpu.OP_SUM_I64)
        gk, gv = red.collect()
    hk, hv = datagen.uniform_pairs(91, n, key_bits=18)
    ok, ov = oc.reduce_by_key_i64(hk, hv, 256, 256)
    assert sorted(zip(gk.tolist(), gv.tolist())) == sorted(zip(ok.tolist(), ov.tolist()))
