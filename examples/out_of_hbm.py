"""Beyond-HBM training: the shard lives in pinned host memory and streams
through the GPU block by block (SpilledDistArray)."""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.join(_os.path.dirname(
    _os.path.abspath(__file__)), ".."))
import torch

from mi355ml.core.array import shard_rows
from mi355ml.core.spill import spill_rows
from mi355ml.linear_model import LogisticRegression
from mi355ml.parallel import comm

c = comm.init()
n, d = 20_000_000, 100  # scale rows up to exceed HBM per GPU
g = torch.Generator().manual_seed(0)
x = torch.randn(n, d, generator=g)
w = torch.randn(d, generator=g)
y = (torch.sigmoid(x @ w) > torch.rand(n, generator=g)).float()
X = spill_rows(x, block_rows=2_000_000)  # host-pinned, streamed
clf = LogisticRegression(C=1.0, max_iter=50).fit(X, shard_rows(y))
if c.rank == 0:
    print("fit on spilled shard, iters:", clf.n_iter_)
comm.shutdown()
