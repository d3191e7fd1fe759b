"""Distributed pager on the HIP engine (single GPU box, gloo transport,
2 ranks sharing device 0) + DLPack zero-copy view checks.

This validates the HIP side of the exchange protocol without an 8-GPU node:
N pages on 1 GPU must implement exactly the gate semantics of N pages on N
GPUs (the reference's CI strategy: test_main.cpp:277-283 runs QPager over
repeated device IDs).
"""

import multiprocessing as mp
import os

import numpy as np
import pytest

import qrack_amd as qa

pytestmark = pytest.mark.gpu

import os as _os
PORT = 28000 + (_os.getpid() % 8000)


def _worker(rank, world, qubits, seed, port, fn_name):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    from qrack_amd.dist_pager import DistQPager

    pager = DistQPager(qubits, engine="hip", seed=seed, device_id=0)
    globals()[fn_name](pager, rank)
    dist.barrier()
    dist.destroy_process_group()


def run_distributed(fn_name, world=2, qubits=5, seed=7, port_off=0):
    ctx = mp.get_context("spawn")
    procs = []
    for r in range(world):
        p = ctx.Process(target=_worker, args=(r, world, qubits, seed, PORT + port_off, fn_name))
        p.start()
        procs.append(p)
    ok = True
    for p in procs:
        p.join(timeout=300)
        if p.exitcode != 0:
            ok = False
    assert ok, f"workers failed for {fn_name}"


def _body_ghz_hip(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(0)
    pager.h(0)
    for i in range(n - 1):
        pager.cnot(i, i + 1)
    sv = pager.get_state_vector()
    assert abs(abs(sv[0]) - 1 / np.sqrt(2)) < 1e-4
    assert abs(abs(sv[-1]) - 1 / np.sqrt(2)) < 1e-4


def _body_qft_hip(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(5)
    pager.qft(0, n)
    sv = pager.get_state_vector().astype(np.complex128)
    N = 1 << n
    k = np.arange(N)
    expected = np.exp(2j * np.pi * 5 * k / N) / np.sqrt(N)
    rev = np.array([int(format(i, f"0{n}b")[::-1], 2) for i in range(N)])
    inner = np.vdot(expected, sv[rev])
    assert abs(abs(inner) - 1.0) < 1e-3


def _body_measure_hip(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(0)
    pager.h(0)
    pager.cnot(0, n - 1)
    res = pager.multi_shot_measure_mask([1, 1 << (n - 1)], 200)
    assert sum(res.values()) == 200
    assert set(res.keys()) <= {0, 3}


def test_ghz_hip_world2():
    run_distributed("_body_ghz_hip", world=2, qubits=5, port_off=1)


def test_qft_hip_world2():
    run_distributed("_body_qft_hip", world=2, qubits=6, port_off=2)


def test_measure_hip_world2():
    run_distributed("_body_measure_hip", world=2, qubits=5, port_off=3)


_DLPACK_BODY = """
import torch
torch.cuda.init()   # torch first, engine second — bench.py ordering
import sys
sys.path.insert(0, {repo!r})
import numpy as np
import qrack_amd as qa
q = qa.create_simulator(4, engine="hip", seed=3)
q.h(0)
q.cnot(0, 1)
cap = q.dlpack_view(0, 16)
t = torch.from_dlpack(cap)
assert t.shape == (16,)
assert t.is_cuda
sv = np.asarray(q.get_state_vector())
assert np.allclose(t.cpu().numpy(), sv, atol=1e-6)
t[3] = 0.5 + 0.25j
torch.cuda.synchronize()
amp = q.get_amplitude(3)
assert abs(amp - (0.5 + 0.25j)) < 1e-6
print("DLPACK_OK")
"""


def test_dlpack_view_roundtrip():
    # fresh process: torch's HIP runtime must win the init race (same order
    # bench.py uses: torch/process-group first, engines second)
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-c", _DLPACK_BODY.format(repo=repo)],
        capture_output=True,
        text=True,
        timeout=240,
    )
    assert "DLPACK_OK" in out.stdout, f"stdout={out.stdout}\nstderr={out.stderr}"
