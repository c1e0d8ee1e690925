"""Transport-layer unit tests (gloo, world_size=2): the var-length blob
all-gather (the PBFT-total-order replacement) and
the tensor primitives, exercised directly rather than through the
engine."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import json, os, sys
import torch
sys.path.insert(0, {repo!r})
from bflc_amd.comm import Transport

t = Transport(backend="gloo", device=torch.device("cpu"))
r = t.rank

# var-length blobs, including an empty one on rank 1
blob = b"x" * (10 + 100 * r) if r == 0 else b""
blobs = t.all_gather_blobs(blob)
assert blobs[0] == b"x" * 10, blobs
assert blobs[1] == b"", blobs

# pickled objects keep rank order
objs = t.all_gather_objects({{"rank": r, "v": [r] * (r + 1)}})
assert [o["rank"] for o in objs] == [0, 1]
assert objs[1]["v"] == [1, 1]

# equal-shape tensor gather
x = torch.full((4,), float(r))
outs = t.all_gather_tensor(x)
assert torch.equal(outs[0], torch.zeros(4))
assert torch.equal(outs[1], torch.ones(4))

# broadcast from rank 0
b = torch.full((3,), 7.0) if r == 0 else torch.zeros(3)
b = t.broadcast_tensor(b, src=0)
assert torch.equal(b, torch.full((3,), 7.0))

with open(os.path.join({outdir!r}, f"ok{{r}}"), "w") as f:
    f.write("ok")
t.barrier()
t.close()
"""


def test_world2_transport_primitives(tmp_path):
    script = tmp_path / "w.py"
    script.write_text(WORKER.format(repo=REPO, outdir=str(tmp_path)))
    procs = []
    for rank in range(2):
        env = dict(os.environ, RANK=str(rank), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT="29561",
                   OMP_NUM_THREADS="2")
        procs.append(subprocess.Popen([sys.executable, str(script)],
                                      env=env, stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE))
    for p in procs:
        _, err = p.communicate(timeout=300)
        assert p.returncode == 0, err.decode()[-3000:]
    assert (tmp_path / "ok0").exists() and (tmp_path / "ok1").exists()
