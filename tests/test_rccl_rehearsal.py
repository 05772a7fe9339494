"""RCCL (nccl backend) rehearsal on a single GPU box.

The round-end driver runs bench.py at N=1..8, one rank per GPU over RCCL —
a path that must not fail on its first-ever execution there. These tests
execute the real nccl(=RCCL) backend on a 1-GPU box:

* a 1-rank process group running every collective op our code issues
  (int64/float64/float32 allreduce, max-reduce, broadcast, object
  allgather, barrier) — real RCCL kernels, real stream semantics;
* the full trainer path (distributed cuts, per-level histogram allreduce,
  fused metric aggregation) under a 1-rank nccl communicator;
* a 2-rank nccl attempt with both ranks on device 0 — RCCL, like NCCL,
  may refuse duplicate devices; if it does, the test records that and is
  skipped (the gloo 2-rank test covers collective ordering instead).

Reference semantics being rehearsed: the implicit per-round rabit
allreduce inside xgb.train (reference distributed.py:219).
"""
import json
import multiprocessing as mp
import os
import socket

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def _find_open_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _single_rank_worker(port, q):
    try:
        import datetime

        import torch
        import torch.distributed as dist

        from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
        from sagemaker_xgboost_container_amd.models import trainer
        from sagemaker_xgboost_container_amd.parallel.comm import Communicator

        torch.cuda.set_device(0)
        dist.init_process_group(
            backend="nccl", init_method=f"tcp://127.0.0.1:{port}", rank=0, world_size=1,
            timeout=datetime.timedelta(seconds=120),
        )
        comm = Communicator()
        dev = torch.device("cuda", 0)

        # every collective op + dtype the training path issues
        t64 = torch.arange(4096, dtype=torch.int64, device=dev).reshape(4, 1024)
        comm.allreduce_(t64)
        assert int(t64.sum()) == 4096 * 4095 // 2
        f64 = torch.tensor([1.5, -2.5], dtype=torch.float64, device=dev)
        comm.allreduce_(f64)
        f32 = torch.tensor([3.0], dtype=torch.float32, device=dev)
        comm.allreduce_max_(f32)
        assert float(f32[0]) == 3.0
        b = torch.full((8,), 7.0, device=dev)
        comm.broadcast_(b, src=0)
        objs = comm.allgather_object({"host": "a", "ok": True})
        assert objs == [{"host": "a", "ok": True}]
        comm.barrier()

        # full training path with a live nccl communicator (grower's
        # per-level int64 allreduce + float64 root sum run real RCCL)
        rng = np.random.default_rng(3)
        X = rng.normal(size=(100_000, 10)).astype(np.float32)
        y = (X[:, 0] - 0.3 * X[:, 2] > 0).astype(np.float32)
        params = {"objective": "binary:logistic", "max_depth": 6, "eta": 0.4,
                  "device": "cuda", "eval_metric": ["logloss"]}
        res = {}
        bst = trainer.train(
            params,
            DMatrix(X, label=y),
            num_boost_round=5,
            evals=[(DMatrix(X, label=y), "train")],
            evals_result=res,
            verbose_eval=False,
            comm=comm,
        )
        # a 1-rank allreduce is the identity: the comm path (per-level
        # compacted histogram allreduce) must grow EXACTLY the trees the
        # communicator-free device path grows
        bst_ref = trainer.train(
            params, DMatrix(X, label=y), num_boost_round=5, verbose_eval=False,
        )

        def _trees(b):
            return json.dumps(
                b.save_json()["learner"]["gradient_booster"]["model"]["trees"], sort_keys=True
            )

        same_trees = _trees(bst) == _trees(bst_ref)

        # chunked-overlap variant through REAL nccl async allreduce handles
        os.environ["SMXGB_COMM_OVERLAP"] = "1"
        try:
            bst_ov = trainer.train(
                params, DMatrix(X, label=y), num_boost_round=5, verbose_eval=False, comm=comm,
            )
        finally:
            os.environ.pop("SMXGB_COMM_OVERLAP", None)
        same_trees = same_trees and (_trees(bst_ov) == _trees(bst_ref))
        q.put(("ok", res["train"]["logloss"][-1], same_trees))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001 - relayed to the test process
        q.put(("error", f"{type(e).__name__}: {e}"))


def test_nccl_single_rank_full_path():
    port = _find_open_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_single_rank_worker, args=(port, q))
    p.start()
    result = q.get(timeout=420)
    p.join(timeout=120)
    assert result[0] == "ok", f"nccl 1-rank path failed: {result[1]}"
    assert result[1] < 0.5  # loss decreased through the nccl-comm grower
    assert result[2], "comm-path trees differ from the communicator-free path"
    assert p.exitcode == 0


def _two_rank_worker(rank, world, port, q):
    try:
        import datetime

        import torch
        import torch.distributed as dist

        from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
        from sagemaker_xgboost_container_amd.models import trainer
        from sagemaker_xgboost_container_amd.parallel.comm import Communicator

        torch.cuda.set_device(0)  # both ranks share the only device
        dist.init_process_group(
            backend="nccl", init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=90),
        )
        comm = Communicator()
        # a first collective is where duplicate-device setups fail
        probe = torch.ones(4, device="cuda")
        comm.allreduce_(probe)

        rng = np.random.default_rng(11)
        X = rng.normal(size=(100_000, 10)).astype(np.float32)
        y = (X[:, 0] + 0.5 * X[:, 3] > 0).astype(np.float32)
        sl = slice(rank, None, world)
        res = {}
        bst = trainer.train(
            {"objective": "binary:logistic", "max_depth": 6, "eta": 0.4, "device": "cuda",
             "eval_metric": ["logloss", "auc"]},
            DMatrix(X[sl], label=y[sl]),
            num_boost_round=5,
            evals=[(DMatrix(X[sl], label=y[sl]), "train")],
            evals_result=res,
            verbose_eval=False,
            comm=comm,
        )
        import hashlib

        sig = hashlib.sha256(
            json.dumps(bst.save_json()["learner"]["gradient_booster"]["model"]["trees"],
                       sort_keys=True).encode()
        ).hexdigest()
        q.put(("ok", rank, sig, res["train"]["logloss"][-1]))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put(("error", rank, f"{type(e).__name__}: {e}", None))


def test_nccl_two_rank_one_device():
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    port = _find_open_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_two_rank_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=420) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
            p.join(timeout=30)
    errors = [r for r in results if r[0] == "error"]
    if errors:
        msg = "; ".join(str(e[2]) for e in errors)
        if any(s in msg.lower() for s in ("duplicate", "invalid usage", "invalid argument")):
            pytest.skip(f"RCCL refuses two ranks on one device (expected): {msg[:300]}")
        pytest.fail(f"nccl 2-rank-1-device failed for a non-duplicate reason: {msg[:800]}")
    sigs = {r[2] for r in results}
    assert len(sigs) == 1, "ranks grew different trees under nccl"
    assert results[0][3] == pytest.approx(results[1][3], abs=1e-12)
