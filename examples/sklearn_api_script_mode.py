"""Example script-mode entry using the xgboost sklearn API + xgb.cv —
the reference's public "boston" customer-script pattern
(test/resources/boston/single_machine_customer_script.py), runnable on
this container verbatim thanks to the xgboost compatibility package.

SageMaker runs it via the `sagemaker_program` hyperparameter; synthetic
data replaces the sklearn dataset download (containers have no egress).
"""
import argparse
import os

import numpy as np
import pandas as pd
import xgboost as xgb

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--objective", type=str, default="reg:squarederror")
    parser.add_argument("--learning-rate", type=float, default=0.1)
    parser.add_argument("--max-depth", type=int, default=5)
    parser.add_argument("--n-estimators", type=int, default=10)
    parser.add_argument("--output-data-dir", type=str, default=os.environ.get("SM_OUTPUT_DATA_DIR"))
    parser.add_argument("--model-dir", type=str, default=os.environ.get("SM_MODEL_DIR"))
    args, _ = parser.parse_known_args()

    rng = np.random.default_rng(0)
    X = pd.DataFrame(rng.normal(size=(5000, 8)), columns=[f"x{i}" for i in range(8)])
    y = X["x0"] * 2 - X["x1"] + rng.normal(scale=0.2, size=5000)

    reg = xgb.XGBRegressor(
        objective=args.objective,
        learning_rate=args.learning_rate,
        max_depth=args.max_depth,
        n_estimators=args.n_estimators,
    )
    reg.fit(X.iloc[:4000], y[:4000])
    preds = reg.predict(X.iloc[4000:])
    rmse = float(np.sqrt(np.mean((preds - y[4000:].to_numpy()) ** 2)))
    print(f"holdout RMSE: {rmse:.4f}")
    reg.get_booster().save_model(os.path.join(args.model_dir, "xgboost-model"))

    cv_results = xgb.cv(
        params={"objective": args.objective, "max_depth": args.max_depth,
                "learning_rate": args.learning_rate},
        dtrain=xgb.DMatrix(X, label=y),
        nfold=5,
        num_boost_round=args.n_estimators,
        early_stopping_rounds=5,
        metrics="rmse",
        as_pandas=True,
        seed=100,
    )
    os.makedirs(args.output_data_dir, exist_ok=True)
    cv_results.to_csv(os.path.join(args.output_data_dir, "cv_results.csv"))
    print(cv_results.tail(1).to_string())
