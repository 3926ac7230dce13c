"""RcclMesh exchange semantics over gloo (world_size 2, CPU): each rank sees
every rank's batch with correct metadata, in rank order."""

import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

WORKER = r"""
import os, sys, torch
sys.path.insert(0, os.environ["PUSHCDN_REPO"])
from pushcdn_amd.parallel.mesh import RcclMesh

rank = int(os.environ["RANK"])
mesh = RcclMesh(torch.device("cpu"), batch_capacity=64)
payload = f"batch-from-{rank}".encode()
batch = torch.zeros(64, dtype=torch.uint8)
batch[: len(payload)] = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
out = mesh.exchange(batch, n_messages=rank + 1, batch_bytes=len(payload))
assert len(out) == 2
for r, view, n_msgs, nbytes in out:
    got = bytes(view[:nbytes].numpy().tobytes())
    assert got == f"batch-from-{r}".encode(), (r, got)
    assert n_msgs == r + 1
assert mesh.max_over_ranks(float(rank)) == 1.0
mesh.barrier()
print(f"rank {rank} OK")
"""


def test_mesh_exchange_gloo(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env["PUSHCDN_REPO"] = str(REPO)
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr", "127.0.0.1", "--master-port", "29517",
            str(script),
        ],
        capture_output=True, text=True, timeout=300, env=env,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    assert "rank 0 OK" in out.stdout and "rank 1 OK" in out.stdout


P2P_WORKER = r"""
import os, sys, torch
sys.path.insert(0, os.environ["PUSHCDN_REPO"])
from pushcdn_amd.parallel.mesh import RcclMesh

rank = int(os.environ["RANK"])
mesh = RcclMesh(torch.device("cpu"), batch_capacity=64)
payload = f"p2p-from-{rank}".encode()
batch = torch.zeros(64, dtype=torch.uint8)
batch[: len(payload)] = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
# rank 0 targets rank 1; rank 1 targets nobody
targets = [1] if rank == 0 else []
out = mesh.exchange_p2p(batch, n_messages=5 + rank, batch_bytes=len(payload),
                        targets=targets)
senders = sorted(r for r, _, _, _ in out)
if rank == 0:
    assert senders == [0], senders          # nobody targeted rank 0
else:
    assert senders == [0, 1], senders       # self + rank 0's send
    for r, view, n_msgs, nbytes in out:
        got = bytes(view[:nbytes].numpy().tobytes())
        assert got == f"p2p-from-{r}".encode()
        assert n_msgs == 5 + r
mesh.barrier()
print(f"rank {rank} P2P OK")
"""


def test_mesh_p2p_targeted_gloo(tmp_path):
    script = tmp_path / "worker_p2p.py"
    script.write_text(P2P_WORKER)
    env = dict(os.environ)
    env["PUSHCDN_REPO"] = str(REPO)
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr", "127.0.0.1", "--master-port", "29519",
            str(script),
        ],
        capture_output=True, text=True, timeout=300, env=env,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    assert "rank 0 P2P OK" in out.stdout and "rank 1 P2P OK" in out.stdout
