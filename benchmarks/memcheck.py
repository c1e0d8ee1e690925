"""Per-round wall clock + allocator stats (used to isolate a ~35 ms
CPython gen-2 GC pause; see bench.py's gc pinning)."""
import sys, torch
sys.path.insert(0, "/root/repo")
from bflc_amd.config import FLConfig
from bflc_amd.comm import Transport
from bflc_amd.data import make_federated
from bflc_amd.fl import FLEngine
cfg = FLConfig.for_world(1, model="resnet20", n_class=10,
                         samples_per_client=2048, batch_size=512,
                         partition="dirichlet", eval_samples=1024,
                         learning_rate=0.01)
shards, test = make_federated(cfg)
eng = FLEngine(cfg, Transport(device=torch.device("cuda", 0)), shards, test)
import gc
gc.collect()
gc.disable()
for i in range(16):
    st = eng.run_round()
    print(f"r{i}: {st.wall_s*1e3:6.1f}ms alloc={torch.cuda.memory_allocated()/2**20:8.1f}MB "
          f"reserved={torch.cuda.memory_reserved()/2**20:8.1f}MB", flush=True)
