import sys, torch
import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tfmesos_amd import ops
from tfmesos_amd.models.inception import InceptionV3, synthetic_images
from tfmesos_amd.ps.module_trainer import ModuleReplicaTrainer

dev = torch.device("cuda", 0)
model = InceptionV3(num_classes=1000)
tr = ModuleReplicaTrainer(model, optimizer="sgd", hparams={"lr": 0.01},
                          device=dev, n_ps=1, colocate_ps=True)
x, y = synthetic_images(32, size=299, classes=1000, device=dev,
                        dtype=torch.bfloat16, seed=0)

def fwd():
    return ops.softmax_xent_loss(model(x).contiguous(), y)

def fwd_bwd():
    tr.zero_grad()
    loss = fwd()
    loss.backward()

def full():
    fwd_bwd()
    tr.step()

for name, fn in [("fwd", lambda: fwd()), ("fwd_bwd", fwd_bwd), ("full", full)]:
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(2):
            fn()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    try:
        g = torch.cuda.CUDAGraph()
        with torch.autograd.set_multithreading_enabled(False):
            with torch.cuda.graph(g):
                fn()
        g.replay(); torch.cuda.synchronize()
        print(name, "CAPTURE OK")
    except Exception as e:
        print(name, "FAIL:", repr(e)[:300])
        torch.cuda.synchronize()
