import torch, time
from skypilot_amd.serve.engine import Engine
eng = Engine("llama3-8b", device="cuda", max_batch=8)
g, st = eng._get_graph(1)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50):
    g.replay()
torch.cuda.synchronize()
print("bucket-1 replay: %.3f ms" % ((time.perf_counter()-t0)/50*1e3))
