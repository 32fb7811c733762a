import torch, sys
sys.path.insert(0, "/root/repo")
from distributedmnist_amd.models import LeNet5
from distributedmnist_amd.parallel import FlatParams
bf16 = torch.bfloat16
torch.manual_seed(7)
x = torch.rand(64, 28, 28, 1) - 0.5
labels = torch.randint(0, 10, (64,))
mc = LeNet5(seed=123); fpc = FlatParams(mc); fpc.zero_grad()
loss_c, acc = mc.loss_and_accuracy(mc(x, train=False), labels); loss_c.backward(); fpc.fix_grad_views()
mg = LeNet5(seed=123, compute_dtype=bf16).cuda(); fpg = FlatParams(mg, compute_dtype=bf16); fpg.zero_grad()
loss_g, acc_g = mg.loss_and_accuracy(mg(x.cuda().to(bf16), train=False), labels.cuda()); loss_g.backward(); fpg.fix_grad_views()
gc, gg = fpc.flat_grad, fpg.flat_grad.cpu()
for name, off, sz in zip(fpc.names, fpc.offsets, fpc.numels):
    r, g = gc[off:off+sz], gg[off:off+sz]
    scale = float(r.abs().max())
    err = (g - r).abs()
    rel = err / (r.abs() + 0.02*scale)
    print(f"{name:10s} scale={scale:10.3e} maxerr={float(err.max()):10.3e} maxerr/scale={float(err.max())/scale:8.4f} p99rel={float(rel.quantile(0.99)):.4f}")
