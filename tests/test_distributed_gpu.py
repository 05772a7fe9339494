"""Distributed training on a real GPU: two ranks sharing cuda:0 over gloo.

The round-end scaling bench runs one rank per GPU over RCCL; this test
exercises the same code path (device grower + per-level histogram allreduce
+ distributed cuts + eval-metric aggregation) with two processes sharing
one device, which a 1-GPU box can run. gloo accepts CUDA tensors for
allreduce, so the device-resident collectives run end to end.
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


def _worker(rank, world, port, q):
    os.environ["GLOO_SOCKET_IFNAME"] = os.environ.get("GLOO_SOCKET_IFNAME", "lo")
    try:
        import datetime

        import torch
        import torch.distributed as dist

        from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
        from sagemaker_xgboost_container_amd.models import trainer
        from sagemaker_xgboost_container_amd.parallel.comm import Communicator

        dist.init_process_group(
            backend="gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=180),
        )
        comm = Communicator()
        rng = np.random.default_rng(11)
        X = rng.normal(size=(200_000, 12)).astype(np.float32)
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
        q.put(("ok", rank, sig, res["train"]["logloss"][-1], res["train"]["auc"][-1]))
        dist.barrier()
        dist.destroy_process_group()
    except Exception:  # noqa: BLE001 - full traceback relayed to the test
        import traceback

        q.put(("error", rank, traceback.format_exc(), None, None))


def test_two_rank_gpu_training_identical_trees():
    port = _find_open_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in procs:
        p.join(timeout=120)
        if p.is_alive():
            p.terminate()
            p.join(timeout=30)
    errors = [r for r in results if r[0] == "error"]
    assert not errors, "worker rank(s) failed:\n" + "\n".join(str(e[2]) for e in errors)
    sigs = {s for _ok, _r, s, _l, _a in results}
    assert len(sigs) == 1, "ranks grew different trees on GPU"
    # aggregated eval metrics must be identical across ranks
    assert results[0][3] == pytest.approx(results[1][3], abs=1e-12)
    assert results[0][4] == pytest.approx(results[1][4], abs=1e-12)
    assert results[0][3] < 0.5  # loss actually decreased over 5 rounds


def _overlap_worker(rank, world, port, overlap, q):
    os.environ["GLOO_SOCKET_IFNAME"] = os.environ.get("GLOO_SOCKET_IFNAME", "lo")
    os.environ["SMXGB_COMM_OVERLAP"] = "1" if overlap else "0"
    try:
        import datetime
        import hashlib

        import torch.distributed as dist

        from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
        from sagemaker_xgboost_container_amd.models import trainer
        from sagemaker_xgboost_container_amd.parallel.comm import Communicator

        dist.init_process_group(
            backend="gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=180),
        )
        comm = Communicator()
        rng = np.random.default_rng(17)
        X = rng.normal(size=(150_000, 10)).astype(np.float32)
        y = (X[:, 0] - 0.4 * X[:, 2] > 0).astype(np.float32)
        sl = slice(rank, None, world)
        bst = trainer.train(
            {"objective": "binary:logistic", "max_depth": 6, "eta": 0.4, "device": "cuda"},
            DMatrix(X[sl], label=y[sl]),
            num_boost_round=4,
            verbose_eval=False,
            comm=comm,
        )
        sig = hashlib.sha256(
            json.dumps(bst.save_json()["learner"]["gradient_booster"]["model"]["trees"],
                       sort_keys=True).encode()
        ).hexdigest()
        q.put(("ok", rank, sig))
        dist.barrier()
        dist.destroy_process_group()
    except Exception:  # noqa: BLE001
        import traceback

        q.put(("error", rank, traceback.format_exc()))


def test_comm_overlap_bit_identical_trees():
    """SMXGB_COMM_OVERLAP=1 pipelines each level's histogram build with the
    allreduce of the previous half. Chunked int64 sums are bit-exact, so
    the overlapped path must grow EXACTLY the trees the plain path grows."""
    ctx = mp.get_context("spawn")
    sigs = {}
    for overlap in (False, True):
        port = _find_open_port()
        q = ctx.Queue()
        procs = [ctx.Process(target=_overlap_worker, args=(r, 2, port, overlap, q))
                 for r in range(2)]
        for p in procs:
            p.start()
        results = [q.get(timeout=300) for _ in range(2)]
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
        errors = [r for r in results if r[0] == "error"]
        assert not errors, f"overlap={overlap}:\n" + "\n".join(str(e[2]) for e in errors)
        rank_sigs = {r[2] for r in results}
        assert len(rank_sigs) == 1, f"overlap={overlap}: ranks grew different trees"
        sigs[overlap] = rank_sigs.pop()
    assert sigs[False] == sigs[True], "overlapped allreduce changed the trees"
