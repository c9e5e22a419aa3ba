"""CPU-staged host-tier gather path: correctness vs plain torch indexing.

Runs in a subprocess with QUIVER_STAGED_GATHER=1 because the mode flag is
read once per process (static).
"""
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

SCRIPT = r"""
import torch
from quiver.shard_tensor import ShardTensor, ShardTensorConfig

t = torch.randn(60000, 100)
st = ShardTensor(0, ShardTensorConfig({}))
st.append(t[:20000], 0)    # HBM
st.append(t[20000:], -1)   # pinned host
g = torch.Generator().manual_seed(3)
for trial in range(3):
    idx = torch.randint(0, 60000, (30000,), generator=g)
    got = st[idx.cuda()].cpu()
    assert torch.equal(got, t[idx]), (got - t[idx]).abs().max()
print("STAGED-GATHER-OK")
"""


def test_staged_gather_correct():
    env = dict(os.environ)
    env["QUIVER_STAGED_GATHER"] = "1"
    res = subprocess.run([sys.executable, "-c", SCRIPT], env=env,
                         capture_output=True, text=True, timeout=600,
                         cwd=os.path.dirname(os.path.dirname(
                             os.path.abspath(__file__))))
    assert res.returncode == 0, res.stdout + res.stderr
    assert "STAGED-GATHER-OK" in res.stdout
