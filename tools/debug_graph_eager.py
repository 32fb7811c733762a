"""Lockstep graph-vs-eager divergence probe (run on a GPU box)."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from distributedmnist_amd.engine.train import Trainer, make_dataset
from distributedmnist_amd.utils.flags import build_train_parser

SEGS = [("conv1_w", 0, 800), ("conv1_b", 800, 832), ("conv2_w", 832, 52032),
        ("conv2_b", 52032, 52096), ("fc1_w", 52096, 1657728),
        ("fc1_b", 1657728, 1658240), ("fc2_w", 1658240, 1663360),
        ("fc2_b", 1663360, 1663370)]

def mk(graph):
    argv = ["--synthetic_data", "--train_dir", f"/tmp/dbg_{graph}",
            "--batch_size", "128", "--max_steps", "9", "--model", "lenet",
            "--initial_learning_rate", "0.05", "--save_interval_secs", "100000"]
    if not graph:
        argv += ["--hip_graph", "off"]
    flags = build_train_parser().parse_args(argv)
    t = Trainer(flags, device=torch.device("cuda:0"))
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    return t, ds

te, dse = mk(False)
tg, dsg = mk(True)
for s in range(6):
    xe, ye = dse.next_batch(128)
    xg, yg = dsg.next_batch(128)
    assert torch.equal(xe, xg) and torch.equal(ye, yg), f"data mismatch at {s}"
    _, le, ae, _ = te.graph_or_eager_step(xe, ye)
    _, lg, ag, _ = tg.graph_or_eager_step(xg, yg)
    torch.cuda.synchronize()
    we = te.fp.flat_master
    wg = tg.fp.flat_master
    d = (we - wg).abs()
    print(f"step {s}: loss eager={float(le):.6f} graph={float(lg):.6f} "
          f"max|dw|={float(d.max()):.3e} n_diff={(d > 1e-6).sum().item()}")
    if float(d.max()) > 1e-5:
        for name, a, b in SEGS:
            dd = d[a:b]
            print(f"   {name}: max={float(dd.max()):.3e} "
                  f"n={(dd > 1e-6).sum().item()}/{b-a}")
        break
print("graph built:", tg._graph is not None)
