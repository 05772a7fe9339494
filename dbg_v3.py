import os, numpy as np, torch, json
from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
from sagemaker_xgboost_container_amd.models import trainer

rng = np.random.default_rng(11)
X = rng.normal(size=(300_000, 28)).astype(np.float32)
y = (X[:, 0]*2 - X[:, 1] + 0.5*X[:, 2]*X[:, 3] > 0).astype(np.float32)
params = {"objective":"binary:logistic","max_depth":6,"eta":0.3,"device":"cuda"}

os.environ["SMXGB_NO_DEVICE_GROW"]="1"
ref = trainer.train(dict(params), DMatrix(X,label=y), 2, verbose_eval=False)
del os.environ["SMXGB_NO_DEVICE_GROW"]
dev = trainer.train(dict(params), DMatrix(X,label=y), 2, verbose_eval=False)

for t in range(len(ref.trees)):
    a, b = ref.trees[t], dev.trees[t]
    if a.feature.tolist() != b.feature.tolist() or a.split_bin.tolist() != b.split_bin.tolist():
        print(f"tree {t} differs")
        # find first differing node in BFS order
        for nid in range(min(a.num_nodes, b.num_nodes)):
            fa = (int(a.feature[nid]), int(a.split_bin[nid]), int(a.left[nid]))
            fb = (int(b.feature[nid]), int(b.split_bin[nid]), int(b.left[nid]))
            if fa != fb:
                print(f"  node {nid}: ref={fa} gain={a.gain[nid]:.5f} vs dev={fb} gain={b.gain[nid]:.5f}")
                print(f"  ref value={a.value[nid]:.5f} hess={a.sum_hess[nid]:.2f}; dev value={b.value[nid]:.5f} hess={b.sum_hess[nid]:.2f}")
                if nid < 20: pass
        print("  ref nodes", a.num_nodes, "dev nodes", b.num_nodes)
        break
else:
    print("trees identical")
    pa, pb = ref.predict(X[:1000]), dev.predict(X[:1000])
    print("pred maxdiff", np.abs(pa-pb).max())
