"""creditcore CLI — the local train -> package -> serve lifecycle.

Replaces the reference's Databricks bundle + GitHub Actions + Bicep pipeline
 with one-command entrypoints on a single 8x MI355X node:

    python -m creditcore train  [--model-dir ./model] [--max-evals 10] ...
    python -m creditcore pack   --model-dir ./model --out ./model/packed.npz
    python -m creditcore serve  [--model-directory ./model] [--port 5000] ...
    python -m creditcore smoke  [--url http://127.0.0.1:5000]
"""

from __future__ import annotations

import argparse
import json
import sys


def _cmd_train(argv):
    p = argparse.ArgumentParser(prog="creditcore train")
    p.add_argument("--model-dir", default="./model")
    p.add_argument("--model-name", default="credit-default-uci-custom")
    p.add_argument("--registry-root", default=None)
    p.add_argument("--max-evals", type=int, default=10)
    p.add_argument("--n-rows", type=int, default=20_000)
    p.add_argument("--seed", type=int, default=2024)
    p.add_argument("--no-register", action="store_true")
    p.add_argument("--min-roc-auc", type=float, default=None,
                   help="refuse to register below this validation ROC-AUC "
                        "(the quality gate the reference lacks)")
    p.add_argument("--algorithm", default="rf", choices=["rf", "gbt", "et"],
                   help="rf = the reference's RandomForest; et = ExtraTrees "
                        "(same packed format/kernel); gbt = "
                        "gradient-boosted trees (same HIP traversal kernel)")
    p.add_argument("--data", default=None,
                   help="CSV with the UCI schema (the reference's curated "
                        "table, 00-create-external-table.ipynb); synthetic "
                        "data is generated when omitted")
    a = p.parse_args(argv)
    from .train import train_and_register

    df = None
    if a.data:
        import pandas as pd

        df = pd.read_csv(a.data)
    uri = train_and_register(
        model_dir=a.model_dir,
        model_name=a.model_name,
        registry_root=a.registry_root,
        max_evals=a.max_evals,
        n_rows=a.n_rows,
        seed=a.seed,
        register=not a.no_register,
        df=df,
        algorithm=a.algorithm,
        min_roc_auc=a.min_roc_auc,
    )
    print(uri)


def _cmd_pack(argv):
    p = argparse.ArgumentParser(prog="creditcore pack")
    p.add_argument("--model-dir", default="./model")
    p.add_argument("--out", default=None)
    a = p.parse_args(argv)
    from .pack import pack_pyfunc_dir

    packed = pack_pyfunc_dir(a.model_dir)
    out = a.out or f"{a.model_dir.rstrip('/')}/packed.npz"
    packed.save(out)
    print(
        json.dumps(
            {
                "out": out,
                "cls_trees": packed.cls_n_trees,
                "cls_nodes": int(len(packed.cls_nodes)),
                "if_trees": packed.if_n_trees,
                "n_onehot": packed.n_onehot,
                "n_ref": packed.n_ref,
            }
        )
    )


def _cmd_serve(argv):
    from .serve import main as serve_main

    serve_main(argv)


def _cmd_smoke(argv):
    """The CI smoke test (reference deploy-kubernetes.yml:206-271): POST the
    sample request, require HTTP 200 + schema-valid body."""
    p = argparse.ArgumentParser(prog="creditcore smoke")
    p.add_argument("--url", default="http://127.0.0.1:5000")
    a = p.parse_args(argv)
    import httpx

    from .schema import SAMPLE_REQUEST, ModelOutput

    r = httpx.post(f"{a.url}/predict", json=SAMPLE_REQUEST, timeout=30.0)
    assert r.status_code == 200, f"smoke test failed: HTTP {r.status_code}: {r.text}"
    ModelOutput.model_validate(r.json())
    print(json.dumps(r.json()))
    print("SMOKE OK")


def _cmd_pipeline(argv):
    """The full local CD pipeline (reference deploy workflow analog)."""
    p = argparse.ArgumentParser(prog="creditcore pipeline")
    p.add_argument("--model-dir", default="./model")
    p.add_argument("--registry-root", default=None)
    p.add_argument("--model-name", default="credit-default-uci-custom")
    p.add_argument("--max-evals", type=int, default=10)
    p.add_argument("--n-rows", type=int, default=20_000)
    p.add_argument("--staging-port", type=int, default=5001)
    p.add_argument("--production-port", type=int, default=5000)
    p.add_argument("--device", default="auto")
    p.add_argument("--auto-approve", action="store_true")
    p.add_argument("--keep-production", action="store_true")
    p.add_argument("--docker-image", default=None)
    a = p.parse_args(argv)
    from .pipeline import run_pipeline

    report = run_pipeline(
        model_dir=a.model_dir,
        registry_root=a.registry_root,
        model_name=a.model_name,
        max_evals=a.max_evals,
        n_rows=a.n_rows,
        staging_port=a.staging_port,
        production_port=a.production_port,
        device=a.device,
        auto_approve=a.auto_approve,
        keep_production=a.keep_production,
        docker_image=a.docker_image,
    )
    print(json.dumps(report, indent=2))


def _cmd_train_dense(argv):
    """Train the dense wide-tabular family (BASELINE config 5)."""
    p = argparse.ArgumentParser(prog="creditcore train-dense")
    p.add_argument("--model-dir", default="./dense_model")
    p.add_argument("--train-rows", type=int, default=10_000_000)
    p.add_argument("--feats", type=int, default=1000)
    p.add_argument("--ref-rows", type=int, default=None)
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--device", default="auto")
    a = p.parse_args(argv)
    from .dense import train_dense

    model = train_dense(
        n_rows=a.train_rows,
        n_feats=a.feats,
        ref_rows=a.ref_rows,
        epochs=a.epochs,
        device=a.device,
    )
    model.save(a.model_dir)
    print(json.dumps({"model_dir": a.model_dir, "n_features": model.n_features,
                      "ref_rows": model.n_ref}))


def _cmd_score_batch(argv):
    """Offline batch scoring job: CSV in, scored CSV out, sharded
    round-robin across all visible GPUs (SURVEY.md §5.7 row-parallel
    scaling; the reference's only offline path was pasting inference.csv
    rows into Swagger)."""
    p = argparse.ArgumentParser(prog="creditcore score-batch")
    p.add_argument("--model-dir", default="./model")
    p.add_argument("--input", required=True, help="CSV with the request schema")
    p.add_argument("--output", required=True, help="CSV: input + prediction/outlier")
    p.add_argument("--chunk-rows", type=int, default=16384)
    p.add_argument("--device", default="auto")
    a = p.parse_args(argv)
    import numpy as np
    import pandas as pd

    from .config import ServeConfig
    from .engine import ScoringEngine, load_engine
    from .pack import encode_batch

    device = a.device if a.device != "auto" else ServeConfig().resolve_device()
    engines = [load_engine(a.model_dir, device=device, device_index=0)]
    if device == "cuda":
        import torch

        for i in range(1, torch.cuda.device_count()):
            engines.append(ScoringEngine(engines[0].packed, device="cuda", device_index=i))

    df = pd.read_csv(a.input)
    preds = np.empty(len(df))
    outl = np.empty(len(df))
    chunks = range(0, len(df), a.chunk_rows)
    for k, lo in enumerate(chunks):
        hi = min(lo + a.chunk_rows, len(df))
        eng = engines[k % len(engines)]  # round-robin shard across GPUs
        codes, nums = encode_batch(df.iloc[lo:hi], eng.packed.vocabs)
        out = eng.score_arrays(codes, nums, with_drift=False)
        preds[lo:hi] = out["predictions"]
        outl[lo:hi] = out["outliers"]
    df_out = df.copy()
    df_out["prediction"] = preds
    df_out["is_outlier"] = outl
    df_out.to_csv(a.output, index=False)
    print(json.dumps({"rows": len(df), "output": a.output,
                      "engines": len(engines), "device": device}))


def _cmd_generate_data(argv):
    """Write a UCI-shaped synthetic CSV (the reference's curated.csv analog,
    reference databricks/data/; the real UCI CSV is not available offline)."""
    p = argparse.ArgumentParser(prog="creditcore generate-data")
    p.add_argument("--out", default="./data/curated.csv")
    p.add_argument("--n-rows", type=int, default=30_000)
    p.add_argument("--seed", type=int, default=2024)
    p.add_argument("--inference-sample", default=None,
                   help="also write an N-row inference.csv-style sample")
    a = p.parse_args(argv)
    import os

    from .data import make_uci_shaped_frame

    os.makedirs(os.path.dirname(a.out) or ".", exist_ok=True)
    df = make_uci_shaped_frame(n_rows=a.n_rows, seed=a.seed)
    df.to_csv(a.out, index=False)
    print(f"wrote {a.out} ({len(df)} rows)")
    if a.inference_sample:
        df2 = make_uci_shaped_frame(n_rows=80, seed=a.seed + 1, include_target=False)
        df2.to_csv(a.inference_sample, index=False)
        print(f"wrote {a.inference_sample} (80 rows)")


def main():
    cmds = {
        "train": _cmd_train,
        "pack": _cmd_pack,
        "serve": _cmd_serve,
        "smoke": _cmd_smoke,
        "pipeline": _cmd_pipeline,
        "train-dense": _cmd_train_dense,
        "score-batch": _cmd_score_batch,
        "generate-data": _cmd_generate_data,
    }
    if len(sys.argv) < 2 or sys.argv[1] not in cmds:
        print(f"usage: python -m creditcore {{{'|'.join(cmds)}}} [args]", file=sys.stderr)
        sys.exit(2)
    cmds[sys.argv[1]](sys.argv[2:])


if __name__ == "__main__":
    main()
