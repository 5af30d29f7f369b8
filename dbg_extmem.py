import numpy as np
import torch
from xgboost_amd import Booster
from xgboost_amd.extmem import ExtMemQuantileDMatrix, ExtMemOps, _StreamedPage
from xgboost_amd.backend.cpu import GradQuantizer


class It:
    def __init__(self):
        self.i = 0

    def reset(self):
        self.i = 0

    def next(self, input_data):
        if self.i >= 3:
            return False
        rng = np.random.RandomState(self.i)
        Xb = rng.randn(5000, 8).astype(np.float32)
        yb = (Xb[:, 0] > 0).astype(np.float32)
        input_data(data=Xb, label=yb)
        self.i += 1
        return True


d = ExtMemQuantileDMatrix(It(), max_bin=64)
dev = torch.device("cuda")
ops_s = ExtMemOps(d, dev, device_cache_bytes=0)
ops_c = ExtMemOps(d, dev)
print("streamed:", [type(p).__name__ for p in ops_s.page_ops])
print("cached:", [type(p).__name__ for p in ops_c.page_ops])

n = d.num_row()
rng = np.random.RandomState(42)
g = torch.tensor(rng.randn(n).astype(np.float32), device=dev)
h = torch.rand(n, device=dev) + 0.5
gpair = torch.stack([g, h], dim=1).contiguous()
q = GradQuantizer(gpair)
qg = q.quantize(gpair)

for ops in (ops_s, ops_c):
    ops.reset(n)
hs = ops_s.build_hist_nodes(qg, [0])
hc = ops_c.build_hist_nodes(qg, [0])
torch.cuda.synchronize()
print("root hist equal:", torch.equal(hs, hc),
      "sum s/c:", hs.sum().item(), hc.sum().item())
print("per-page:")
for i in range(3):
    ps = ops_s.page_ops[i].build_hist_nodes(
        ops_s._page_gpair(qg, i), [0])
    pc = ops_c.page_ops[i].build_hist_nodes(
        ops_c._page_gpair(qg, i), [0])
    torch.cuda.synchronize()
    print(f"  page {i}: equal={torch.equal(ps, pc)} "
          f"s={ps.sum().item()} c={pc.sum().item()}")

# node sizes after reset
print("node0 size:", ops_s.node_size(0), ops_c.node_size(0))

# now run one full tree through each and compare
from xgboost_amd.grower import TreeGrower
from xgboost_amd.tree_model import RegTree
from xgboost_amd.params import TrainParam

param = TrainParam(max_depth=4, max_bin=64)
for name, ops in (("streamed", ops_s), ("cached", ops_c)):
    t = RegTree(8)
    gr = TreeGrower(ops, param, q, n, seed=0)
    t, pos = gr.grow(qg.clone(), t)
    print(name, "n_nodes:", t.n_nodes,
          "splits:", t.split_index[:min(t.n_nodes, 7)].tolist())
