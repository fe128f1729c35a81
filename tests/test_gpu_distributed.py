"""2-rank data-parallel training on ONE GPU (device oversubscription).

RCCL refuses two ranks on one device ("Duplicate GPU detected"), so this
uses the gloo backend over CUDA tensors -- every framework-side distributed
code path (torchrun rendezvous, DWBP GradReducer bucketing + side-stream
events, SFB factor all-gather + MFMA outer-product reconstruction,
device-staged metric all-reduce, barriers) runs exactly as in the 8-GPU
RCCL configuration; only the transport differs.
"""

import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.skipif(not torch.cuda.is_available(), reason="GPU only")
def test_two_rank_dwbp_sfb_on_one_gpu():
    env = dict(os.environ)
    env.update(MASTER_ADDR="127.0.0.1", PS_BACKEND="gloo")
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", "2",
           "--master-addr", "127.0.0.1", "--master-port", "29517",
           os.path.join(REPO, "bench.py"),
           "--gpus", "2", "--steps", "2", "--warmup", "1", "--batch", "32"]
    out = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                         text=True, timeout=300)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["sfb"] is True
    assert rec["value"] > 0
