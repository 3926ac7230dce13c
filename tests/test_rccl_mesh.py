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


INTEREST_WORKER = r"""
import os, sys, torch
sys.path.insert(0, os.environ["PUSHCDN_REPO"])
from pushcdn_amd.parallel.mesh import RcclMesh

rank = int(os.environ["RANK"])
mesh = RcclMesh(torch.device("cpu"), batch_capacity=64)
payload = f"int-from-{rank}".encode()
batch = torch.zeros(64, dtype=torch.uint8)
batch[: len(payload)] = torch.frombuffer(bytearray(payload), dtype=torch.uint8)

# tick 1: rank 0's batch has topic 7; only rank 1 is interested in topic 7
out = mesh.exchange_interest(batch, n_messages=1, batch_bytes=len(payload),
                             batch_topics=(1 << 7) if rank == 0 else 0,
                             interests=(1 << 7) if rank == 1 else (1 << 3))
senders = sorted(r for r, _, _, _ in out)
if rank == 0:
    assert senders == [0], senders          # rank 1's batch (topic-less) not shipped
else:
    assert senders == [0, 1], senders
    for r, view, n_msgs, nbytes in out:
        assert bytes(view[:nbytes].numpy().tobytes()) == f"int-from-{r}".encode()

# tick 2: direct digests — rank 0 carries a direct whose recipient digest
# hits rank 1's owned digest; rank 1's digest hits nobody
out = mesh.exchange_interest(batch, n_messages=1, batch_bytes=len(payload),
                             batch_topics=0, interests=0,
                             direct_bits=(1 << 9) if rank == 0 else (1 << 11),
                             owned_bits=(1 << 9) if rank == 1 else 0)
senders = sorted(r for r, _, _, _ in out)
assert senders == ([0] if rank == 0 else [0, 1]), senders

# tick 3: no overlap anywhere -> nobody ships (prune works)
out = mesh.exchange_interest(batch, n_messages=1, batch_bytes=len(payload),
                             batch_topics=1, interests=2,
                             direct_bits=4, owned_bits=8)
assert [r for r, _, _, _ in out] == [rank]
mesh.barrier()
print(f"rank {rank} INTEREST OK")
"""


def test_mesh_interest_digest_routing_gloo(tmp_path):
    """exchange_interest ships on topic intersection OR direct-digest
    intersection, and prunes when neither matches."""
    script = tmp_path / "worker_interest.py"
    script.write_text(INTEREST_WORKER)
    env = dict(os.environ)
    env["PUSHCDN_REPO"] = str(REPO)
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr", "127.0.0.1", "--master-port", "29537",
            str(script),
        ],
        capture_output=True, text=True, timeout=300, env=env,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    assert "rank 0 INTEREST OK" in out.stdout and "rank 1 INTEREST OK" in out.stdout
