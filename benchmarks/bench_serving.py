#!/usr/bin/env python3
"""Serving latency benchmark (BASELINE config 5).

Loads a Booster the serving path way (serve_utils.get_loaded_booster),
then measures end-to-end /invocations latency through the ASGI app
(payload parse -> DMatrix -> predict [HIP forest kernel on GPU] -> encode)
at a fixed request size, plus the raw predict-kernel latency.

    python benchmarks/bench_serving.py [--rows 1000] [--requests 200] [--trees 500]
"""
import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=1000)
    ap.add_argument("--features", type=int, default=28)
    ap.add_argument("--trees", type=int, default=500)
    ap.add_argument("--max-depth", type=int, default=6)
    ap.add_argument("--requests", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=20)
    args = ap.parse_args()

    import tempfile

    import torch

    from sagemaker_xgboost_container_amd.algorithm_mode import serve
    from sagemaker_xgboost_container_amd.data.dmatrix import DeviceDMatrix, DMatrix
    from sagemaker_xgboost_container_amd.models import trainer

    use_gpu = torch.cuda.is_available()
    device = "cuda" if use_gpu else "cpu"

    # train a flagship-shaped model (500 trees depth 6)
    rng = np.random.default_rng(0)
    n_train = 200_000 if use_gpu else 20_000
    Xt = torch.randn((n_train, args.features), device=device)
    yt = (Xt[:, 0] * 2 - Xt[:, 1] > 0).float()
    dtrain = DeviceDMatrix(Xt, label=yt) if use_gpu else DMatrix(Xt.numpy(), label=yt.numpy())
    bst = trainer.train(
        {"objective": "binary:logistic", "max_depth": args.max_depth, "device": device},
        dtrain,
        num_boost_round=args.trees,
        verbose_eval=False,
    )
    model_dir = tempfile.mkdtemp(prefix="serve_bench_")
    bst.save_model(os.path.join(model_dir, "xgboost-model"))

    os.environ["SM_MODEL_DIR"] = model_dir
    serve.ScoringService.MODEL_PATH = model_dir
    serve.ScoringService.reset()

    # real server: uvicorn in a background thread + raw stdlib HTTP client
    import http.client as http_client
    import logging as _logging
    import threading

    import uvicorn

    _logging.getLogger("uvicorn").setLevel(_logging.WARNING)
    _logging.getLogger("uvicorn.access").setLevel(_logging.WARNING)
    port = 18080
    config = uvicorn.Config(serve.ScoringService.app, host="127.0.0.1", port=port, log_level="warning")
    server = uvicorn.Server(config)
    thread = threading.Thread(target=server.run, daemon=True)
    thread.start()
    for _ in range(100):
        if server.started:
            break
        time.sleep(0.05)

    X = rng.normal(size=(args.rows, args.features)).astype(np.float32)
    payload = "\n".join(",".join(f"{v:.6f}" for v in row) for row in X).encode()

    conn = http_client.HTTPConnection("127.0.0.1", port)
    lat = []
    for i in range(args.warmup + args.requests):
        t0 = time.perf_counter()
        conn.request("POST", "/invocations", body=payload, headers={"Content-Type": "text/csv"})
        resp = conn.getresponse()
        body = resp.read()
        assert resp.status == 200, body[:200]
        if i >= args.warmup:
            lat.append((time.perf_counter() - t0) * 1000)
    lat = np.array(lat)
    server.should_exit = True

    # raw predict path (model loaded, DMatrix pre-parsed)
    from sagemaker_xgboost_container_amd.algorithm_mode import serve_utils

    dtest, ct = serve_utils.parse_content_data(payload, "text/csv")
    raw = []
    for i in range(args.warmup + args.requests):
        t0 = time.perf_counter()
        serve.ScoringService.predict(dtest, content_type=ct, model_format=serve.ScoringService.format)
        if use_gpu:
            torch.cuda.synchronize()
        if i >= args.warmup:
            raw.append((time.perf_counter() - t0) * 1000)
    raw = np.array(raw)

    print(
        json.dumps(
            {
                "metric": "p50 predict ms @1k-row batch",
                "p50_ms": float(np.percentile(lat, 50)),
                "p90_ms": float(np.percentile(lat, 90)),
                "p99_ms": float(np.percentile(lat, 99)),
                "raw_predict_p50_ms": float(np.percentile(raw, 50)),
                "rows": args.rows,
                "trees": args.trees,
                "max_depth": args.max_depth,
                "device": device,
                "requests": args.requests,
            }
        )
    )


if __name__ == "__main__":
    main()
